// Metadata C ABI — analog of the reference's rust/lakesoul-metadata-c
// (lib.rs:116-560): open the catalog store, look up table info, resolve
// the latest snapshot to a file list, and commit new data with the MVCC
// version-CAS retry loop — everything a JVM/foreign-engine connector
// needs to participate in the commit protocol (DBManager.java routes
// the same operations through metadata-c).
//
// Backed by the sqlite catalog (same schema as meta/store.py, which
// mirrors script/meta_init.sql). sqlite3 is declared extern and linked
// against the system libsqlite3.so.0 (no dev headers in the image,
// same technique as compress.h for libzstd). Results cross the ABI as
// JSON strings (callers free them with lakesoul_meta_free_string).

#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <stdexcept>
#include <random>
#include <map>
#include <string>
#include <vector>

// ---- minimal sqlite3 API (stable C ABI of libsqlite3.so.0) ----
extern "C" {
typedef struct sqlite3 sqlite3;
typedef struct sqlite3_stmt sqlite3_stmt;
int sqlite3_open(const char*, sqlite3**);
int sqlite3_close(sqlite3*);
int sqlite3_prepare_v2(sqlite3*, const char*, int, sqlite3_stmt**,
                       const char**);
int sqlite3_bind_text(sqlite3_stmt*, int, const char*, int, void (*)(void*));
int sqlite3_bind_int64(sqlite3_stmt*, int, int64_t);
int sqlite3_step(sqlite3_stmt*);
int sqlite3_finalize(sqlite3_stmt*);
const unsigned char* sqlite3_column_text(sqlite3_stmt*, int);
int64_t sqlite3_column_int64(sqlite3_stmt*, int);
int sqlite3_exec(sqlite3*, const char*, int (*)(void*, int, char**, char**),
                 void*, char**);
int sqlite3_busy_timeout(sqlite3*, int);
#define SQLITE_ROW 100
#define SQLITE_DONE 101
#define SQLITE_OK 0
#define SQLITE_CONSTRAINT 19
}

static thread_local std::string g_meta_err;

extern "C" const char* lakesoul_meta_last_error(void) {
  return g_meta_err.c_str();
}

namespace {

struct MetaHandle {
  sqlite3* db = nullptr;
};

std::string jesc(const std::string& s) {
  std::string o;
  for (char c : s) {
    if (c == '"' || c == '\\') {
      o += '\\';
      o += c;
    } else if ((unsigned char)c < 0x20) {
      char buf[8];
      snprintf(buf, sizeof buf, "\\u%04x", c);
      o += buf;
    } else {
      o += c;
    }
  }
  return o;
}

// tiny JSON scanners for the store's own serializations (lists of
// strings / file-op dicts written by meta/store.py via json.dumps)
std::vector<std::string> parse_str_list(const std::string& j) {
  std::vector<std::string> out;
  size_t i = 0;
  while ((i = j.find('"', i)) != std::string::npos) {
    size_t e = i + 1;
    std::string cur;
    while (e < j.size() && j[e] != '"') {
      if (j[e] == '\\' && e + 1 < j.size()) e++;
      cur += j[e++];
    }
    out.push_back(cur);
    i = e + 1;
  }
  return out;
}

std::string q1(sqlite3* db, const std::string& sql,
               const std::vector<std::string>& binds, bool* found) {
  sqlite3_stmt* st = nullptr;
  if (sqlite3_prepare_v2(db, sql.c_str(), -1, &st, nullptr) != SQLITE_OK)
    throw std::runtime_error("sqlite prepare failed: " + sql);
  for (size_t i = 0; i < binds.size(); i++)
    sqlite3_bind_text(st, (int)i + 1, binds[i].c_str(), -1, nullptr);
  std::string out;
  *found = false;
  if (sqlite3_step(st) == SQLITE_ROW) {
    const unsigned char* t = sqlite3_column_text(st, 0);
    out = t ? (const char*)t : "";
    *found = true;
  }
  sqlite3_finalize(st);
  return out;
}

std::string rand_hex(int n) {
  static thread_local std::mt19937_64 rng{std::random_device{}()};
  static const char* hexd = "0123456789abcdef";
  std::string s;
  for (int i = 0; i < n; i++) s += hexd[rng() & 15];
  return s;
}

}  // namespace

#define META_TRY(body)             \
  try {                            \
    body;                          \
    return 0;                      \
  } catch (std::exception & e) {   \
    g_meta_err = e.what();         \
    return -1;                     \
  }

extern "C" void* lakesoul_meta_open(const char* db_path) {
  auto* h = new MetaHandle();
  if (sqlite3_open(db_path, &h->db) != SQLITE_OK) {
    g_meta_err = "cannot open metadata db";
    delete h;
    return nullptr;
  }
  sqlite3_busy_timeout(h->db, 60000);
  return h;
}

extern "C" void lakesoul_meta_close(void* hp) {
  auto* h = (MetaHandle*)hp;
  if (h) {
    sqlite3_close(h->db);
    delete h;
  }
}

extern "C" void lakesoul_meta_free_string(char* s) { free(s); }

// table info as JSON {"table_id","table_path","table_schema",...} or NULL
extern "C" char* lakesoul_meta_table_info(void* hp, const char* name,
                                          const char* ns) {
  auto* h = (MetaHandle*)hp;
  try {
    sqlite3_stmt* st = nullptr;
    const char* sql =
        "SELECT table_id, table_path, table_schema, properties, partitions"
        " FROM table_info WHERE table_name=? AND table_namespace=?";
    if (sqlite3_prepare_v2(h->db, sql, -1, &st, nullptr) != SQLITE_OK)
      throw std::runtime_error("prepare failed");
    sqlite3_bind_text(st, 1, name, -1, nullptr);
    sqlite3_bind_text(st, 2, ns && *ns ? ns : "default", -1, nullptr);
    char* out = nullptr;
    if (sqlite3_step(st) == SQLITE_ROW) {
      auto col = [&](int i) {
        const unsigned char* t = sqlite3_column_text(st, i);
        return std::string(t ? (const char*)t : "");
      };
      std::string j = "{\"table_id\":\"" + jesc(col(0)) + "\",\"table_path\":\"" +
                      jesc(col(1)) + "\",\"table_schema\":" +
                      (col(2).empty() ? "null" : col(2)) +
                      ",\"properties\":" + (col(3).empty() ? "{}" : col(3)) +
                      ",\"partitions\":\"" + jesc(col(4)) + "\"}";
      out = strdup(j.c_str());
    }
    sqlite3_finalize(st);
    if (!out) g_meta_err = "table not found";
    return out;
  } catch (std::exception& e) {
    g_meta_err = e.what();
    return nullptr;
  }
}

// latest snapshot resolved to files: JSON [{"path":...,"size":N},...]
extern "C" char* lakesoul_meta_files_for_latest(void* hp, const char* table_id,
                                                const char* partition_desc) {
  auto* h = (MetaHandle*)hp;
  try {
    bool found = false;
    std::string snap = q1(
        h->db,
        "SELECT snapshot FROM partition_info WHERE table_id=? AND"
        " partition_desc=? ORDER BY version DESC LIMIT 1",
        {table_id, partition_desc}, &found);
    std::string out = "[";
    bool first = true;
    if (found) {
      // resolve add/del ops across the snapshot's commits in order
      std::vector<std::pair<std::string, int64_t>> files;
      for (auto& cid : parse_str_list(snap)) {
        bool f2 = false;
        std::string ops = q1(
            h->db,
            "SELECT file_ops FROM data_commit_info WHERE table_id=? AND"
            " partition_desc=? AND commit_id=?",
            {table_id, partition_desc, cid}, &f2);
        if (!f2) continue;
        // ops JSON: [{"path": "...", "file_op": "add|del", "size": N}, ...]
        size_t i = 0;
        while ((i = ops.find("{", i)) != std::string::npos) {
          size_t e = ops.find("}", i);
          std::string obj = ops.substr(i, e - i);
          auto grab = [&](const char* key) {
            size_t k = obj.find(std::string("\"") + key + "\"");
            if (k == std::string::npos) return std::string();
            k = obj.find(':', k) + 1;
            while (k < obj.size() && (obj[k] == ' ')) k++;
            if (obj[k] == '"') {
              size_t q = obj.find('"', k + 1);
              return obj.substr(k + 1, q - k - 1);
            }
            size_t q = obj.find_first_of(",}", k);
            return obj.substr(k, q - k);
          };
          std::string path = grab("path");
          std::string op = grab("file_op");
          std::string sz = grab("size");
          if (op == "del") {
            for (auto it = files.begin(); it != files.end(); ++it)
              if (it->first == path) {
                files.erase(it);
                break;
              }
          } else {
            files.push_back({path, sz.empty() ? 0 : atoll(sz.c_str())});
          }
          i = e + 1;
        }
      }
      for (auto& f : files) {
        if (!first) out += ",";
        first = false;
        out += "{\"path\":\"" + jesc(f.first) +
               "\",\"size\":" + std::to_string(f.second) + "}";
      }
    }
    out += "]";
    return strdup(out.c_str());
  } catch (std::exception& e) {
    g_meta_err = e.what();
    return nullptr;
  }
}

// commit new files (Append/Merge semantics: snapshot extend) with the
// MVCC version-CAS retry loop. paths/sizes: parallel arrays.
extern "C" int lakesoul_meta_commit_add_files(void* hp, const char* table_id,
                                              const char* partition_desc,
                                              const char** paths,
                                              const int64_t* sizes, int nfiles,
                                              const char* commit_op) {
  auto* h = (MetaHandle*)hp;
  META_TRY({
    std::string op = commit_op && *commit_op ? commit_op : "MergeCommit";
    std::string cid = rand_hex(32);
    std::string ops = "[";
    for (int i = 0; i < nfiles; i++) {
      if (i) ops += ", ";
      ops += "{\"path\": \"" + jesc(paths[i]) + "\", \"file_op\": \"add\", "
             "\"size\": " + std::to_string(sizes ? sizes[i] : 0) + "}";
    }
    ops += "]";
    {
      sqlite3_stmt* st = nullptr;
      const char* sql =
          "INSERT INTO data_commit_info VALUES (?,?,?,?,?,0,"
          "CAST(strftime('%s','now') AS INTEGER)*1000,'public')";
      if (sqlite3_prepare_v2(h->db, sql, -1, &st, nullptr) != SQLITE_OK)
        throw std::runtime_error("prepare dci failed");
      sqlite3_bind_text(st, 1, table_id, -1, nullptr);
      sqlite3_bind_text(st, 2, partition_desc, -1, nullptr);
      sqlite3_bind_text(st, 3, cid.c_str(), -1, nullptr);
      sqlite3_bind_text(st, 4, ops.c_str(), -1, nullptr);
      sqlite3_bind_text(st, 5, op.c_str(), -1, nullptr);
      if (sqlite3_step(st) != SQLITE_DONE) {
        sqlite3_finalize(st);
        throw std::runtime_error("insert data_commit_info failed");
      }
      sqlite3_finalize(st);
    }
    // CAS loop: read latest version+snapshot, insert version+1
    for (int attempt = 0; attempt < 25; attempt++) {
      bool found = false;
      std::string snap = q1(
          h->db,
          "SELECT snapshot FROM partition_info WHERE table_id=? AND"
          " partition_desc=? ORDER BY version DESC LIMIT 1",
          {table_id, partition_desc}, &found);
      int64_t ver = -1;
      if (found) {
        bool f2 = false;
        std::string v = q1(
            h->db,
            "SELECT version FROM partition_info WHERE table_id=? AND"
            " partition_desc=? ORDER BY version DESC LIMIT 1",
            {table_id, partition_desc}, &f2);
        ver = atoll(v.c_str());
      }
      std::string new_snap;
      if (found && !snap.empty() && snap != "[]") {
        new_snap = snap.substr(0, snap.size() - 1) + ", \"" + cid + "\"]";
      } else {
        new_snap = "[\"" + cid + "\"]";
      }
      sqlite3_stmt* st = nullptr;
      const char* sql =
          "INSERT INTO partition_info VALUES (?,?,?,?,"
          "CAST(strftime('%s','now') AS INTEGER)*1000,?,'','public')";
      if (sqlite3_prepare_v2(h->db, sql, -1, &st, nullptr) != SQLITE_OK)
        throw std::runtime_error("prepare pi failed");
      sqlite3_bind_text(st, 1, table_id, -1, nullptr);
      sqlite3_bind_text(st, 2, partition_desc, -1, nullptr);
      sqlite3_bind_int64(st, 3, ver + 1);
      sqlite3_bind_text(st, 4, op.c_str(), -1, nullptr);
      sqlite3_bind_text(st, 5, new_snap.c_str(), -1, nullptr);
      int rc = sqlite3_step(st);
      sqlite3_finalize(st);
      if (rc == SQLITE_DONE) {
        char* err = nullptr;
        std::string upd =
            "UPDATE data_commit_info SET committed=1 WHERE commit_id='" + cid +
            "'";
        sqlite3_exec(h->db, upd.c_str(), nullptr, nullptr, &err);
        return 0;  // committed
      }
      // CAS conflict (PK violation): retry with fresh read
    }
    throw std::runtime_error("commit_add_files: CAS retries exhausted");
  });
}


// ===================================================================== //
// extended DAO surface (reference lakesoul-metadata-c lib.rs DaoType
// families: namespace/table listing, partition listing, versioned +
// incremental snapshot queries, lookup by path)
// ===================================================================== //

namespace {

std::string rows_to_json_array(sqlite3* db, const std::string& sql,
                               const std::vector<std::string>& binds) {
  sqlite3_stmt* st = nullptr;
  if (sqlite3_prepare_v2(db, sql.c_str(), -1, &st, nullptr) != SQLITE_OK)
    throw std::runtime_error("sqlite prepare failed: " + sql);
  for (size_t i = 0; i < binds.size(); i++)
    sqlite3_bind_text(st, (int)i + 1, binds[i].c_str(), -1, nullptr);
  std::string out = "[";
  bool first = true;
  while (sqlite3_step(st) == SQLITE_ROW) {
    const unsigned char* t = sqlite3_column_text(st, 0);
    if (!first) out += ",";
    first = false;
    out += "\"" + jesc(t ? (const char*)t : "") + "\"";
  }
  sqlite3_finalize(st);
  out += "]";
  return out;
}

// resolve a snapshot (JSON list of commit ids) to surviving files
std::string resolve_snapshot_files(sqlite3* db, const std::string& table_id,
                                   const std::string& desc,
                                   const std::string& snap) {
  std::vector<std::pair<std::string, int64_t>> files;
  for (auto& cid : parse_str_list(snap)) {
    bool f2 = false;
    std::string ops = q1(db,
        "SELECT file_ops FROM data_commit_info WHERE table_id=? AND"
        " partition_desc=? AND commit_id=?",
        {table_id, desc, cid}, &f2);
    if (!f2) continue;
    size_t i = 0;
    while ((i = ops.find("{", i)) != std::string::npos) {
      size_t e = ops.find("}", i);
      std::string obj = ops.substr(i, e - i);
      auto grab = [&](const char* key) {
        size_t k = obj.find(std::string("\"") + key + "\"");
        if (k == std::string::npos) return std::string();
        k = obj.find(':', k) + 1;
        while (k < obj.size() && (obj[k] == ' ')) k++;
        if (obj[k] == '"') {
          size_t q = obj.find('"', k + 1);
          return obj.substr(k + 1, q - k - 1);
        }
        size_t q = obj.find_first_of(",}", k);
        return obj.substr(k, q - k);
      };
      std::string path = grab("path");
      std::string op = grab("file_op");
      std::string sz = grab("size");
      if (op == "del") {
        for (auto it = files.begin(); it != files.end(); ++it)
          if (it->first == path) { files.erase(it); break; }
      } else {
        files.push_back({path, sz.empty() ? 0 : atoll(sz.c_str())});
      }
      i = e + 1;
    }
  }
  std::string out = "[";
  bool first = true;
  for (auto& f : files) {
    if (!first) out += ",";
    first = false;
    out += "{\"path\":\"" + jesc(f.first) +
           "\",\"size\":" + std::to_string(f.second) + "}";
  }
  out += "]";
  return out;
}

}  // namespace

extern "C" char* lakesoul_meta_list_namespaces(void* hp) {
  auto* h = (MetaHandle*)hp;
  try {
    return strdup(rows_to_json_array(
        h->db, "SELECT namespace FROM namespace ORDER BY namespace", {})
        .c_str());
  } catch (std::exception& e) { g_meta_err = e.what(); return nullptr; }
}

extern "C" char* lakesoul_meta_list_tables(void* hp, const char* ns) {
  auto* h = (MetaHandle*)hp;
  try {
    return strdup(rows_to_json_array(
        h->db,
        "SELECT table_name FROM table_info WHERE table_namespace=?"
        " ORDER BY table_name",
        {ns && *ns ? ns : "default"}).c_str());
  } catch (std::exception& e) { g_meta_err = e.what(); return nullptr; }
}

extern "C" char* lakesoul_meta_partition_descs(void* hp, const char* table_id) {
  auto* h = (MetaHandle*)hp;
  try {
    return strdup(rows_to_json_array(
        h->db,
        "SELECT DISTINCT partition_desc FROM partition_info WHERE table_id=?"
        " ORDER BY partition_desc",
        {table_id}).c_str());
  } catch (std::exception& e) { g_meta_err = e.what(); return nullptr; }
}

extern "C" int64_t lakesoul_meta_latest_version(void* hp, const char* table_id,
                                                const char* desc) {
  auto* h = (MetaHandle*)hp;
  try {
    bool found = false;
    std::string v = q1(h->db,
        "SELECT version FROM partition_info WHERE table_id=? AND"
        " partition_desc=? ORDER BY version DESC LIMIT 1",
        {table_id, desc}, &found);
    return found ? atoll(v.c_str()) : -1;
  } catch (std::exception& e) { g_meta_err = e.what(); return -1; }
}

// files of a SPECIFIC partition version (time travel)
extern "C" char* lakesoul_meta_files_for_version(void* hp,
                                                 const char* table_id,
                                                 const char* desc,
                                                 int64_t version) {
  auto* h = (MetaHandle*)hp;
  try {
    bool found = false;
    std::string snap = q1(h->db,
        "SELECT snapshot FROM partition_info WHERE table_id=? AND"
        " partition_desc=? AND version=?",
        {table_id, desc, std::to_string(version)}, &found);
    if (!found) { g_meta_err = "version not found"; return nullptr; }
    return strdup(resolve_snapshot_files(h->db, table_id, desc, snap).c_str());
  } catch (std::exception& e) { g_meta_err = e.what(); return nullptr; }
}

// files ADDED in (from_version, to_version] minus later deletes — the
// incremental-read query (reference metadata_client.rs:1052-1126)
extern "C" char* lakesoul_meta_incremental_files(void* hp,
                                                 const char* table_id,
                                                 const char* desc,
                                                 int64_t from_version,
                                                 int64_t to_version) {
  auto* h = (MetaHandle*)hp;
  try {
    bool f0 = false, f1 = false;
    std::string snap_from =
        from_version < 0 ? std::string("[]")
        : q1(h->db,
             "SELECT snapshot FROM partition_info WHERE table_id=? AND"
             " partition_desc=? AND version=?",
             {table_id, desc, std::to_string(from_version)}, &f0);
    std::string snap_to = q1(h->db,
        "SELECT snapshot FROM partition_info WHERE table_id=? AND"
        " partition_desc=? AND version=?",
        {table_id, desc, std::to_string(to_version)}, &f1);
    if (!f1) { g_meta_err = "to_version not found"; return nullptr; }
    auto from_ids = parse_str_list(snap_from);
    std::string inc = "[";
    bool first = true;
    for (auto& cid : parse_str_list(snap_to)) {
      bool seen = false;
      for (auto& p : from_ids) if (p == cid) { seen = true; break; }
      if (!seen) {
        if (!first) inc += ",";
        first = false;
        inc += "\"" + jesc(cid) + "\"";
      }
    }
    inc += "]";
    return strdup(resolve_snapshot_files(h->db, table_id, desc, inc).c_str());
  } catch (std::exception& e) { g_meta_err = e.what(); return nullptr; }
}

extern "C" char* lakesoul_meta_table_info_by_path(void* hp, const char* path) {
  auto* h = (MetaHandle*)hp;
  try {
    bool f1 = false, f2 = false;
    std::string name = q1(h->db,
        "SELECT table_name FROM table_info WHERE table_path=?", {path}, &f1);
    std::string ns = q1(h->db,
        "SELECT table_namespace FROM table_info WHERE table_path=?", {path},
        &f2);
    if (!f1 || !f2) { g_meta_err = "path not found"; return nullptr; }
    return lakesoul_meta_table_info(hp, name.c_str(), ns.c_str());
  } catch (std::exception& e) { g_meta_err = e.what(); return nullptr; }
}

// SplitDesc array — the scan-planning unit engine connectors consume
// (reference lakesoul-metadata-c lib.rs:560 create_split_desc_array +
// transfusion.rs:316,403 SplitDesc{file_paths, primary_keys,
// range_partition_desc, table_schema}): one entry per (partition desc,
// hash bucket) over the LATEST snapshot, files in commit order.
extern "C" char* lakesoul_meta_split_descs(void* hp, const char* table_name,
                                           const char* ns) {
  auto* h = (MetaHandle*)hp;
  try {
    bool f1 = false, f2 = false, f3 = false;
    std::string tid = q1(h->db,
        "SELECT table_id FROM table_info WHERE table_name=? AND"
        " table_namespace=?", {table_name, ns}, &f1);
    if (!f1) { g_meta_err = "table not found"; return nullptr; }
    std::string schema = q1(h->db,
        "SELECT table_schema FROM table_info WHERE table_id=?", {tid}, &f2);
    std::string parts = q1(h->db,
        "SELECT partitions FROM table_info WHERE table_id=?", {tid}, &f3);
    // partitions = "rangeKeys;hashKeys" (comma-split)
    std::string pks;
    auto semi = parts.find(';');
    if (semi != std::string::npos) pks = parts.substr(semi + 1);
    std::string pks_json = "[";
    {
      bool first = true;
      size_t i = 0;
      while (i < pks.size()) {
        size_t e = pks.find(',', i);
        std::string k = pks.substr(i, e == std::string::npos ? e : e - i);
        if (!k.empty()) {
          if (!first) pks_json += ",";
          first = false;
          pks_json += "\"" + jesc(k) + "\"";
        }
        if (e == std::string::npos) break;
        i = e + 1;
      }
      pks_json += "]";
    }

    std::string out = "[";
    bool first_sd = true;
    // latest version per partition desc
    sqlite3_stmt* st = nullptr;
    if (sqlite3_prepare_v2(h->db,
            "SELECT partition_desc, MAX(version) FROM partition_info WHERE"
            " table_id=? GROUP BY partition_desc",
            -1, &st, nullptr) != SQLITE_OK)
      throw std::runtime_error("prepare failed");
    sqlite3_bind_text(st, 1, tid.c_str(), -1, nullptr);
    std::vector<std::pair<std::string, int64_t>> descs;
    while (sqlite3_step(st) == SQLITE_ROW) {
      descs.push_back({(const char*)sqlite3_column_text(st, 0),
                       sqlite3_column_int64(st, 1)});
    }
    sqlite3_finalize(st);
    for (auto& dv : descs) {
      bool fs = false;
      std::string snap = q1(h->db,
          "SELECT snapshot FROM partition_info WHERE table_id=? AND"
          " partition_desc=? AND version=?",
          {tid, dv.first, std::to_string(dv.second)}, &fs);
      if (!fs) continue;
      std::string files_json =
          resolve_snapshot_files(h->db, tid, dv.first, snap);
      // group file paths by bucket id parsed from part-..._NNNN.parquet
      std::map<int, std::vector<std::string>> by_bucket;
      size_t i = 0;
      while ((i = files_json.find("\"path\":\"", i)) != std::string::npos) {
        i += 8;
        size_t e = files_json.find('"', i);
        std::string p = files_json.substr(i, e - i);
        int bucket = -1;
        auto us = p.rfind('_');
        auto dot = p.rfind('.');
        if (us != std::string::npos && dot != std::string::npos && dot > us)
          bucket = atoi(p.substr(us + 1, dot - us - 1).c_str());
        by_bucket[bucket].push_back(p);
        i = e;
      }
      for (auto& bb : by_bucket) {
        if (!first_sd) out += ",";
        first_sd = false;
        out += "{\"file_paths\":[";
        for (size_t k = 0; k < bb.second.size(); k++) {
          if (k) out += ",";
          out += "\"" + jesc(bb.second[k]) + "\"";
        }
        out += "],\"primary_keys\":" + pks_json;
        out += ",\"partition_desc\":\"" + jesc(dv.first) + "\"";
        out += ",\"hash_bucket\":" + std::to_string(bb.first);
        out += ",\"table_schema\":\"" + jesc(schema) + "\"}";
      }
    }
    out += "]";
    return strdup(out.c_str());
  } catch (std::exception& e) { g_meta_err = e.what(); return nullptr; }
}

// JWT-shaped tokens (reference lakesoul-metadata-c lib.rs:465-521 /
// jwt.rs): HMAC-SHA256 over a base64url JSON payload, wire-compatible
// with the python gateway's TokenService (service/server.py) so a C
// consumer can mint/verify the same bearer tokens.
#include <openssl/hmac.h>
#include <ctime>

namespace {
std::string b64url(const unsigned char* d, size_t n) {
  static const char* tbl =
      "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789-_";
  std::string out;
  for (size_t i = 0; i < n; i += 3) {
    uint32_t v = d[i] << 16;
    if (i + 1 < n) v |= d[i + 1] << 8;
    if (i + 2 < n) v |= d[i + 2];
    out += tbl[(v >> 18) & 63];
    out += tbl[(v >> 12) & 63];
    if (i + 1 < n) out += tbl[(v >> 6) & 63];
    if (i + 2 < n) out += tbl[v & 63];
  }
  return out;  // no padding (rstrip'd '=' on the python side)
}

std::string hmac_b64(const std::string& key, const std::string& msg) {
  unsigned char mac[32];
  unsigned int maclen = 0;
  HMAC(EVP_sha256(), key.data(), (int)key.size(),
       (const unsigned char*)msg.data(), msg.size(), mac, &maclen);
  return b64url(mac, maclen);
}
}  // namespace

extern "C" char* lakesoul_meta_jwt_encode(const char* sub, const char* domain,
                                          int64_t ttl_s, const char* secret) {
  try {
    std::string payload = std::string("{\"sub\": \"") + jesc(sub) +
        "\", \"domain\": \"" + jesc(domain) + "\", \"exp\": " +
        std::to_string((int64_t)time(nullptr) + ttl_s) + "}";
    std::string b = b64url((const unsigned char*)payload.data(),
                           payload.size());
    return strdup((b + "." + hmac_b64(secret, b)).c_str());
  } catch (std::exception& e) { g_meta_err = e.what(); return nullptr; }
}

extern "C" char* lakesoul_meta_jwt_decode(const char* token,
                                          const char* secret) {
  try {
    std::string t = token;
    auto dot = t.find('.');
    if (dot == std::string::npos) { g_meta_err = "malformed token"; return nullptr; }
    std::string b = t.substr(0, dot), sig = t.substr(dot + 1);
    if (hmac_b64(secret, b) != sig) { g_meta_err = "bad signature"; return nullptr; }
    // base64url decode the payload
    auto inv = [](char c) -> int {
      if (c >= 'A' && c <= 'Z') return c - 'A';
      if (c >= 'a' && c <= 'z') return c - 'a' + 26;
      if (c >= '0' && c <= '9') return c - '0' + 52;
      if (c == '-') return 62;
      if (c == '_') return 63;
      return -1;
    };
    std::string out;
    uint32_t buf = 0;
    int bits = 0;
    for (char c : b) {
      int v = inv(c);
      if (v < 0) { g_meta_err = "bad base64"; return nullptr; }
      buf = (buf << 6) | (uint32_t)v;
      bits += 6;
      if (bits >= 8) {
        bits -= 8;
        out += (char)((buf >> bits) & 0xFF);
      }
    }
    // expiry check
    auto k = out.find("\"exp\"");
    if (k != std::string::npos) {
      int64_t exp = atoll(out.c_str() + out.find(':', k) + 1);
      if (exp < (int64_t)time(nullptr)) { g_meta_err = "expired"; return nullptr; }
    }
    return strdup(out.c_str());
  } catch (std::exception& e) { g_meta_err = e.what(); return nullptr; }
}
