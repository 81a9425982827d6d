// lakesoul_amd C ABI implementation — torch-free; links only libzstd.
// Build: g++ -O3 -std=c++17 -shared -fPIC lakesoul_c.cc -o
//        liblakesoul_amd_c.so -l:libzstd.so.1 -pthread
#include "lakesoul_c.h"

#include <algorithm>
#include <cstring>
#include <memory>
#include <string>
#include <vector>

#include "../cpp/murmur3.h"
#include "../cpp/parquet_file.h"

using namespace lakesoul;

static thread_local std::string g_err;

extern "C" const char* lakesoul_c_last_error(void) { return g_err.c_str(); }

#define C_TRY(body)                 \
  try {                             \
    body                            \
  } catch (std::exception & e) {    \
    g_err = e.what();               \
    return -1;                      \
  }

// ===================================================================== //
// Arrow C Data export helpers
// ===================================================================== //

namespace {

struct ExportedColumn {
  std::string name;
  std::string format;           // arrow format string
  std::vector<uint8_t> validity_bits;  // empty => no nulls
  std::vector<uint8_t> data;    // fixed-width values
  std::vector<int32_t> offsets; // strings
  std::vector<uint8_t> bytes;   // strings
  int64_t null_count = 0;
  int64_t length = 0;
};

struct ExportPrivate {
  std::vector<std::unique_ptr<ExportedColumn>> cols;
  std::vector<ArrowArray*> child_arrays;
  std::vector<ArrowSchema*> child_schemas;
  std::vector<std::vector<const void*>> buffers;
  std::vector<std::string> names;
};

void release_schema(ArrowSchema* s) {
  if (!s || !s->release) return;
  for (int64_t i = 0; i < s->n_children; i++) {
    if (s->children[i] && s->children[i]->release)
      s->children[i]->release(s->children[i]);
  }
  if (s->private_data) delete (ExportPrivate*)s->private_data;
  s->release = nullptr;
}

void release_child_schema(ArrowSchema* s) { s->release = nullptr; }

void release_array(ArrowArray* a) {
  if (!a || !a->release) return;
  for (int64_t i = 0; i < a->n_children; i++) {
    if (a->children[i] && a->children[i]->release)
      a->children[i]->release(a->children[i]);
  }
  if (a->private_data) delete (ExportPrivate*)a->private_data;
  a->release = nullptr;
}

void release_child_array(ArrowArray* a) { a->release = nullptr; }

std::string dtype_to_arrow_fmt(const ColumnDesc& c) {
  switch (c.physical) {
    case PT_BOOLEAN: return "b";
    case PT_INT32:
      if (c.converted == CV_INT_8) return "c";
      if (c.converted == CV_INT_16) return "s";
      if (c.converted == CV_DATE) return "tdD";
      return "i";
    case PT_INT64:
      if (c.converted == CV_TIMESTAMP_MILLIS) return "tsm:UTC";
      if (c.converted == CV_TIMESTAMP_MICROS) return "tsu:UTC";
      return "l";
    case PT_FLOAT: return "f";
    case PT_DOUBLE: return "g";
    case PT_BYTE_ARRAY:
      return (c.converted == CV_UTF8) ? "u" : "z";
    default: throw std::runtime_error("unsupported physical type for export");
  }
}

}  // namespace

// ===================================================================== //
// reader
// ===================================================================== //

struct LakesoulCReader {
  std::vector<std::string> files;
  std::vector<std::string> columns;
  std::vector<std::string> pks;
  int64_t batch_size = 8192;

  // merged result
  std::vector<ExportedColumn> result;  // full columns; batches slice them
  std::vector<std::string> out_names;
  std::vector<std::string> out_formats;
  int64_t total_rows = 0;
  int64_t cursor = 0;
  bool started = false;
};

extern "C" LakesoulCReader* lakesoul_c_reader_create(void) {
  return new LakesoulCReader();
}

extern "C" int lakesoul_c_reader_add_file(LakesoulCReader* r, const char* p) {
  r->files.push_back(p);
  return 0;
}
extern "C" int lakesoul_c_reader_add_column(LakesoulCReader* r, const char* n) {
  r->columns.push_back(n);
  return 0;
}
extern "C" int lakesoul_c_reader_add_primary_key(LakesoulCReader* r, const char* n) {
  r->pks.push_back(n);
  return 0;
}
extern "C" int lakesoul_c_reader_set_batch_size(LakesoulCReader* r, int64_t b) {
  r->batch_size = b;
  return 0;
}

namespace {

struct FileCols {
  std::vector<DecodedColumn> cols;   // per requested column
  std::vector<ColumnDesc> descs;
  int64_t rows = 0;
};

// key for merge ordering: (pk..., seq, row). integer pks only.
struct RowRef {
  int64_t seq;
  int64_t row;
};

int64_t int_value(const DecodedColumn& dc, const ColumnDesc& cd, int64_t row) {
  switch (cd.physical) {
    case PT_INT64: {
      int64_t v;
      std::memcpy(&v, dc.data.data() + row * 8, 8);
      return v;
    }
    case PT_INT32: {
      int32_t v;
      std::memcpy(&v, dc.data.data() + row * 4, 4);
      return v;
    }
    default:
      throw std::runtime_error(
          "C reader merge supports integer and string primary keys");
  }
}

// three-way PK value compare: integers numerically, BYTE_ARRAY (string)
// byte-wise lexicographic — same order the writer sorted by
int cmp_value(const DecodedColumn& da, const ColumnDesc& ca, int64_t ra,
              const DecodedColumn& db, const ColumnDesc& cb, int64_t rb) {
  if (ca.physical == PT_BYTE_ARRAY && cb.physical == PT_BYTE_ARRAY) {
    int32_t a0 = da.offsets[(size_t)ra], a1 = da.offsets[(size_t)ra + 1];
    int32_t b0 = db.offsets[(size_t)rb], b1 = db.offsets[(size_t)rb + 1];
    size_t la = (size_t)(a1 - a0), lb = (size_t)(b1 - b0);
    int c = std::memcmp(da.bytes.data() + a0, db.bytes.data() + b0,
                        la < lb ? la : lb);
    if (c != 0) return c < 0 ? -1 : 1;
    if (la != lb) return la < lb ? -1 : 1;
    return 0;
  }
  int64_t va = int_value(da, ca, ra);
  int64_t vb = int_value(db, cb, rb);
  return va < vb ? -1 : (va > vb ? 1 : 0);
}

}  // namespace

extern "C" int lakesoul_c_reader_start(LakesoulCReader* r) {
  C_TRY({
    if (r->files.empty()) throw std::runtime_error("no files configured");
    std::vector<FileCols> fcs(r->files.size());
    // resolve requested columns from the first file when unspecified
    {
      ParquetFile f0(r->files[0]);
      if (r->columns.empty())
        for (auto& c : f0.columns()) r->columns.push_back(c.name);
    }
    // pk columns must be read
    std::vector<std::string> read_cols = r->pks;
    for (auto& c : r->columns)
      if (std::find(read_cols.begin(), read_cols.end(), c) == read_cols.end())
        read_cols.push_back(c);

    for (size_t fi = 0; fi < r->files.size(); fi++) {
      ParquetFile f(r->files[fi]);
      FileCols& fc = fcs[fi];
      fc.rows = f.num_rows();
      for (auto& name : read_cols) {
        int ci = f.column_index(name);
        if (ci < 0) throw std::runtime_error("column missing in file: " + name);
        fc.descs.push_back(f.columns()[ci]);
        DecodedColumn dc;
        for (size_t rg = 0; rg < f.num_row_groups(); rg++) {
          auto ch = f.read_chunk(rg, ci);
          DecodedColumn part = decode_chunk_cpu(ch);
          // append
          if (dc.num_values == 0) {
            dc = std::move(part);
          } else {
            int64_t old_n = dc.num_values;
            if (!part.validity.empty() || !dc.validity.empty()) {
              if (dc.validity.empty()) dc.validity.assign((size_t)old_n, 1);
              if (part.validity.empty()) part.validity.assign((size_t)part.num_values, 1);
              dc.validity.insert(dc.validity.end(), part.validity.begin(), part.validity.end());
            }
            if (!part.offsets.empty()) {
              int32_t base = dc.offsets.empty() ? 0 : dc.offsets.back();
              if (dc.offsets.empty()) dc.offsets.push_back(0);
              for (size_t i = 1; i < part.offsets.size(); i++)
                dc.offsets.push_back(base + part.offsets[i]);
              dc.bytes.insert(dc.bytes.end(), part.bytes.begin(), part.bytes.end());
            } else {
              dc.data.insert(dc.data.end(), part.data.begin(), part.data.end());
            }
            dc.num_values += part.num_values;
          }
        }
        fc.cols.push_back(std::move(dc));
      }
    }

    // order refs
    std::vector<RowRef> refs;
    for (size_t fi = 0; fi < fcs.size(); fi++)
      for (int64_t i = 0; i < fcs[fi].rows; i++) refs.push_back({(int64_t)fi, i});

    std::vector<size_t> pk_idx;
    for (auto& p : r->pks) {
      auto it = std::find(read_cols.begin(), read_cols.end(), p);
      pk_idx.push_back((size_t)(it - read_cols.begin()));
    }
    if (!r->pks.empty()) {
      std::stable_sort(refs.begin(), refs.end(), [&](const RowRef& a, const RowRef& b) {
        for (size_t k : pk_idx) {
          int c = cmp_value(fcs[a.seq].cols[k], fcs[a.seq].descs[k], a.row,
                            fcs[b.seq].cols[k], fcs[b.seq].descs[k], b.row);
          if (c != 0) return c < 0;
        }
        if (a.seq != b.seq) return a.seq < b.seq;
        return a.row < b.row;
      });
      // dedup keep-last (UseLast)
      std::vector<RowRef> dedup;
      for (size_t i = 0; i < refs.size(); i++) {
        bool last = (i + 1 == refs.size());
        if (!last) {
          bool same = true;
          for (size_t k : pk_idx) {
            if (cmp_value(fcs[refs[i].seq].cols[k], fcs[refs[i].seq].descs[k],
                          refs[i].row,
                          fcs[refs[i + 1].seq].cols[k],
                          fcs[refs[i + 1].seq].descs[k], refs[i + 1].row) != 0) {
              same = false;
              break;
            }
          }
          if (same) continue;  // a newer row with the same PK follows
        }
        dedup.push_back(refs[i]);
      }
      refs = std::move(dedup);
    }

    // gather requested columns in ref order
    for (auto& name : r->columns) {
      auto it = std::find(read_cols.begin(), read_cols.end(), name);
      size_t k = (size_t)(it - read_cols.begin());
      const ColumnDesc& cd = fcs[0].descs[k];
      ExportedColumn ec;
      ec.name = name;
      ec.format = dtype_to_arrow_fmt(cd);
      ec.length = (int64_t)refs.size();
      bool is_str = cd.physical == PT_BYTE_ARRAY;
      bool is_bool = cd.physical == PT_BOOLEAN;
      int es = physical_elem_size(cd.physical);
      bool any_null = false;
      for (auto& fc : fcs)
        if (!fc.cols[k].validity.empty()) any_null = true;
      if (any_null) ec.validity_bits.assign((refs.size() + 7) / 8, 0);
      if (is_str) ec.offsets.push_back(0);
      if (is_bool) ec.data.assign((refs.size() + 7) / 8, 0);

      for (size_t i = 0; i < refs.size(); i++) {
        const DecodedColumn& dc = fcs[refs[i].seq].cols[k];
        int64_t row = refs[i].row;
        bool valid = dc.validity.empty() || dc.validity[row];
        if (any_null) {
          if (valid) ec.validity_bits[i >> 3] |= (uint8_t)(1 << (i & 7));
          else ec.null_count++;
        }
        if (is_str) {
          if (valid) {
            ec.bytes.insert(ec.bytes.end(), dc.bytes.data() + dc.offsets[row],
                            dc.bytes.data() + dc.offsets[row + 1]);
          }
          ec.offsets.push_back((int32_t)ec.bytes.size());
        } else if (is_bool) {
          if (valid && dc.data[row]) ec.data[i >> 3] |= (uint8_t)(1 << (i & 7));
        } else {
          const uint8_t* src = dc.data.data() + row * es;
          ec.data.insert(ec.data.end(), src, src + es);
        }
      }
      r->result.push_back(std::move(ec));
      r->out_names.push_back(name);
      r->out_formats.push_back(dtype_to_arrow_fmt(cd));
    }
    r->total_rows = r->result.empty() ? 0 : r->result[0].length;
    r->cursor = 0;
    r->started = true;
    return 0;
  })
}

extern "C" int lakesoul_c_reader_schema(LakesoulCReader* r, struct ArrowSchema* out) {
  C_TRY({
    if (!r->started) throw std::runtime_error("reader not started");
    auto* priv = new ExportPrivate();
    size_t n = r->out_names.size();
    priv->child_schemas.resize(n);
    priv->names = r->out_names;
    static std::vector<std::string>* fmt_keep = nullptr;  // formats live in priv
    for (size_t i = 0; i < n; i++) {
      auto* cs = new ArrowSchema();
      std::memset(cs, 0, sizeof(ArrowSchema));
      auto* ec = new ExportedColumn();
      ec->name = r->out_names[i];
      ec->format = r->out_formats[i];
      priv->cols.push_back(std::unique_ptr<ExportedColumn>(ec));
      cs->format = priv->cols.back()->format.c_str();
      cs->name = priv->cols.back()->name.c_str();
      cs->flags = 2;  // ARROW_FLAG_NULLABLE
      cs->release = release_child_schema;
      priv->child_schemas[i] = cs;
    }
    (void)fmt_keep;
    std::memset(out, 0, sizeof(ArrowSchema));
    out->format = "+s";
    out->name = "";
    out->n_children = (int64_t)n;
    out->children = priv->child_schemas.data();
    out->release = release_schema;
    out->private_data = priv;
    return 0;
  })
}

extern "C" int lakesoul_c_reader_next(LakesoulCReader* r, struct ArrowArray* out) {
  C_TRY({
    if (!r->started) throw std::runtime_error("reader not started");
    if (r->cursor >= r->total_rows) return 0;
    int64_t n = std::min(r->batch_size, r->total_rows - r->cursor);
    int64_t a = r->cursor;
    r->cursor += n;

    auto* priv = new ExportPrivate();
    size_t nc = r->result.size();
    priv->child_arrays.resize(nc);
    priv->buffers.resize(nc);
    for (size_t i = 0; i < nc; i++) {
      ExportedColumn& full = r->result[i];
      // slice [a, a+n): copy into a fresh column (C ABI owns its buffers)
      auto ec = std::make_unique<ExportedColumn>();
      ec->format = full.format;
      ec->length = n;
      bool is_str = full.format == "u" || full.format == "z";
      bool is_bool = full.format == "b";
      if (!full.validity_bits.empty()) {
        ec->validity_bits.assign((n + 7) / 8, 0);
        for (int64_t j = 0; j < n; j++) {
          if (full.validity_bits[(a + j) >> 3] & (1 << ((a + j) & 7)))
            ec->validity_bits[j >> 3] |= (uint8_t)(1 << (j & 7));
          else
            ec->null_count++;
        }
      }
      if (is_str) {
        int32_t base = full.offsets[a];
        ec->offsets.reserve(n + 1);
        for (int64_t j = 0; j <= n; j++) ec->offsets.push_back(full.offsets[a + j] - base);
        ec->bytes.assign(full.bytes.begin() + base, full.bytes.begin() + full.offsets[a + n]);
      } else if (is_bool) {
        ec->data.assign((n + 7) / 8, 0);
        for (int64_t j = 0; j < n; j++)
          if (full.data[(a + j) >> 3] & (1 << ((a + j) & 7)))
            ec->data[j >> 3] |= (uint8_t)(1 << (j & 7));
      } else {
        int es = (full.format == "l" || full.format == "g" ||
                  full.format.rfind("ts", 0) == 0) ? 8
                 : (full.format == "s") ? 2
                 : (full.format == "c") ? 1
                 : 4;
        ec->data.assign(full.data.begin() + a * es, full.data.begin() + (a + n) * es);
      }

      auto* ca = new ArrowArray();
      std::memset(ca, 0, sizeof(ArrowArray));
      ca->length = n;
      ca->null_count = ec->null_count;
      auto& bufs = priv->buffers[i];
      bufs.push_back(ec->validity_bits.empty() ? nullptr : ec->validity_bits.data());
      if (is_str) {
        bufs.push_back(ec->offsets.data());
        bufs.push_back(ec->bytes.data());
        ca->n_buffers = 3;
      } else {
        bufs.push_back(ec->data.data());
        ca->n_buffers = 2;
      }
      ca->buffers = bufs.data();
      ca->release = release_child_array;
      priv->child_arrays[i] = ca;
      priv->cols.push_back(std::move(ec));
    }
    std::memset(out, 0, sizeof(ArrowArray));
    out->length = n;
    out->n_children = (int64_t)nc;
    out->children = priv->child_arrays.data();
    out->n_buffers = 1;
    static const void* struct_bufs[1] = {nullptr};
    out->buffers = struct_bufs;
    out->release = release_array;
    out->private_data = priv;
    return 1;
  })
}

extern "C" void lakesoul_c_reader_close(LakesoulCReader* r) { delete r; }

// ===================================================================== //
// writer
// ===================================================================== //

struct LakesoulCWriter {
  std::string path;
  int codec = CODEC_ZSTD;
  int level = 1;
  int64_t row_group = 250000;
  std::vector<ColumnDesc> descs;
  // accumulated rows (row-major append of column buffers)
  std::vector<ExportedColumn> acc;  // reuse as column accumulators (byte-validity in validity_bits)
  std::unique_ptr<ParquetWriter> w;
  bool open = false;
};

extern "C" LakesoulCWriter* lakesoul_c_writer_create(const char* path) {
  auto* w = new LakesoulCWriter();
  w->path = path;
  return w;
}

extern "C" int lakesoul_c_writer_set_compression(LakesoulCWriter* w, const char* codec, int level) {
  std::string c = codec;
  if (c == "zstd") w->codec = CODEC_ZSTD;
  else if (c == "none" || c == "uncompressed") w->codec = CODEC_UNCOMPRESSED;
  else {
    g_err = "unsupported codec " + c;
    return -1;
  }
  w->level = level;
  return 0;
}

extern "C" int lakesoul_c_writer_set_row_group_size(LakesoulCWriter* w, int64_t r) {
  w->row_group = r;
  return 0;
}

extern "C" int lakesoul_c_writer_set_schema(LakesoulCWriter* w, struct ArrowSchema* s) {
  C_TRY({
    if (std::string(s->format) != "+s") throw std::runtime_error("expected struct schema");
    for (int64_t i = 0; i < s->n_children; i++) {
      ArrowSchema* c = s->children[i];
      ColumnDesc d;
      d.name = c->name ? c->name : "";
      d.nullable = (c->flags & 2) != 0;
      std::string f = c->format;
      if (f == "l") d.physical = PT_INT64;
      else if (f == "i") d.physical = PT_INT32;
      else if (f == "f") d.physical = PT_FLOAT;
      else if (f == "g") d.physical = PT_DOUBLE;
      else if (f == "b") d.physical = PT_BOOLEAN;
      else if (f == "u") { d.physical = PT_BYTE_ARRAY; d.converted = CV_UTF8; d.logical = LogicalTag::STRING; }
      else if (f == "z") d.physical = PT_BYTE_ARRAY;
      else throw std::runtime_error("unsupported arrow format: " + f);
      w->descs.push_back(d);
    }
    w->w = std::make_unique<ParquetWriter>(w->path, w->descs, w->codec, w->level, w->row_group);
    w->open = true;
    return 0;
  })
}

extern "C" int lakesoul_c_writer_write(LakesoulCWriter* w, struct ArrowArray* batch) {
  C_TRY({
    if (!w->open) throw std::runtime_error("schema not set");
    if ((size_t)batch->n_children != w->descs.size())
      throw std::runtime_error("column count mismatch");
    int64_t n = batch->length;
    std::vector<ColumnData> data(w->descs.size());
    std::vector<std::vector<uint8_t>> vmasks(w->descs.size());
    std::vector<std::vector<uint8_t>> bool_bytes(w->descs.size());
    std::vector<std::vector<int32_t>> off_keep(w->descs.size());
    for (size_t i = 0; i < w->descs.size(); i++) {
      ArrowArray* c = batch->children[i];
      const uint8_t* vbits = (const uint8_t*)c->buffers[0];
      if (vbits) {
        vmasks[i].resize(n);
        for (int64_t j = 0; j < n; j++)
          vmasks[i][j] = (vbits[(c->offset + j) >> 3] >> ((c->offset + j) & 7)) & 1;
        data[i].validity = vmasks[i].data();
      }
      if (w->descs[i].physical == PT_BYTE_ARRAY) {
        const int32_t* offs = (const int32_t*)c->buffers[1];
        data[i].offsets = offs + c->offset;
        data[i].bytes = (const uint8_t*)c->buffers[2];
        if (c->offset) {
          // rebase offsets
          off_keep[i].assign(offs + c->offset, offs + c->offset + n + 1);
          data[i].offsets = off_keep[i].data();
        }
      } else if (w->descs[i].physical == PT_BOOLEAN) {
        const uint8_t* bits = (const uint8_t*)c->buffers[1];
        bool_bytes[i].resize(n);
        for (int64_t j = 0; j < n; j++)
          bool_bytes[i][j] = (bits[(c->offset + j) >> 3] >> ((c->offset + j) & 7)) & 1;
        data[i].data = bool_bytes[i].data();
      } else {
        int es = physical_elem_size(w->descs[i].physical);
        data[i].data = (const uint8_t*)c->buffers[1] + (int64_t)c->offset * es;
      }
    }
    w->w->write_row_group(data, n);
    return 0;
  })
}

extern "C" int64_t lakesoul_c_writer_close(LakesoulCWriter* w) {
  int64_t size = -1;
  try {
    if (w->open) size = w->w->close();
  } catch (std::exception& e) {
    g_err = e.what();
  }
  delete w;
  return size;
}

extern "C" void lakesoul_c_writer_abort(LakesoulCWriter* w) {
  try {
    if (w->open) {
      w->w.reset();
      std::remove(w->path.c_str());
    }
  } catch (...) {
  }
  delete w;
}

// ===================================================================== //
// murmur3
// ===================================================================== //

extern "C" uint32_t lakesoul_c_murmur3_bytes(const uint8_t* d, int64_t len, uint32_t seed) {
  return spark_hash_bytes(d, len, seed);
}
extern "C" uint32_t lakesoul_c_murmur3_i32(int32_t v, uint32_t seed) {
  return spark_hash_u32((uint32_t)v, seed);
}
extern "C" uint32_t lakesoul_c_murmur3_i64(int64_t v, uint32_t seed) {
  return spark_hash_u64((uint64_t)v, seed);
}
