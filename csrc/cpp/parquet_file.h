// Parquet file writer + reader (host side).
//
// MI355X-native replacement for the reference's arrow-rs `parquet`
// dependency (wired at rust/lakesoul-io/src/writer/mod.rs:53,
// file_format.rs:19-23). Defaults follow the reference's measured config:
// zstd(1), dictionary off, row-group <= 250k rows (writer/mod.rs:224-245).
//
// The reader separates host work (footer/page walk, decompression,
// def-level decode) from value decode: PLAIN fixed-width payloads are
// returned as contiguous buffers ready for H2D + GPU cast/scatter, and
// dictionary-index pages are returned raw with run descriptors for the
// HIP expansion kernel (csrc/hip/decode.hip). decode_chunk_cpu() is the
// CPU fallback used in non-GPU tests.
#pragma once

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

#include <mutex>
#include <unordered_map>

#include "compress.h"
#include "parquet_types.h"
#include "rle.h"

namespace lakesoul {

// foreign physical-type translation (read side): FLBA -> length-prefixed
// byte_array stream; INT96 (legacy impala/spark timestamps: 8B
// nanos-of-day + 4B julian day, both LE) -> int64 nanoseconds since epoch
inline void flba_to_byte_array(const uint8_t* src, int64_t n, int32_t len,
                               std::vector<uint8_t>& out) {
  out.reserve(out.size() + (size_t)n * (len + 4));
  for (int64_t i = 0; i < n; i++) {
    uint32_t l = (uint32_t)len;
    const uint8_t* lp = (const uint8_t*)&l;
    out.insert(out.end(), lp, lp + 4);
    out.insert(out.end(), src + i * len, src + (i + 1) * len);
  }
}

inline void int96_to_ns(const uint8_t* src, int64_t n,
                        std::vector<uint8_t>& out) {
  size_t base = out.size();
  out.resize(base + (size_t)n * 8);
  for (int64_t i = 0; i < n; i++) {
    uint64_t nanos;
    uint32_t jd;
    std::memcpy(&nanos, src + i * 12, 8);
    std::memcpy(&jd, src + i * 12 + 8, 4);
    int64_t ns = ((int64_t)jd - 2440588) * 86400000000000LL + (int64_t)nanos;
    std::memcpy(out.data() + base + i * 8, &ns, 8);
  }
}

// ---------------------------------------------------------------------- //
// column description (writer input / reader output)
// ---------------------------------------------------------------------- //

struct ColumnDesc {
  std::string name;
  int32_t physical = PT_INT64;
  bool nullable = true;
  int32_t converted = CV_NONE;
  LogicalTag logical = LogicalTag::NONE;
  int32_t int_bit_width = 0;
  bool int_signed = true;
  int32_t dec_precision = 0;
  int32_t dec_scale = 0;
  int32_t type_length = 0;  // FLBA
  // LIST columns (3-level standard layout): the desc describes the leaf
  // ELEMENT; list structure decodes from rep/def levels
  bool is_list = false;
  bool elem_nullable = false;
  int32_t max_def = 0;      // 0 for flat columns (unused)
  // struct/map member leaves (writer): the enclosing group's name and
  // kind. Struct members write like flat nullable columns (optional
  // group + required leaf = max_def 1); map key/value leaves write like
  // list elements under group{MAP}/key_value.
  std::string parent;       // empty = top-level column
  int parent_kind = 0;      // 0 none, 1 struct, 2 map
};

inline int physical_elem_size(int32_t pt) {
  switch (pt) {
    case PT_INT32:
    case PT_FLOAT:
      return 4;
    case PT_INT64:
    case PT_DOUBLE:
      return 8;
    case PT_BOOLEAN:
      return 1;  // in-memory byte-per-value
    default:
      return -1;  // variable
  }
}

// ---------------------------------------------------------------------- //
// writer
// ---------------------------------------------------------------------- //

struct ColumnData {
  // fixed-width: `data` points at n*elem values (bool: byte per value)
  // byte_array: `offsets` (n+1 int32) + `bytes`
  const uint8_t* data = nullptr;
  const int32_t* offsets = nullptr;
  const uint8_t* bytes = nullptr;
  const uint8_t* validity = nullptr;  // byte per row, 1=valid; null=all valid
  // LIST columns: `data` holds the ELEMENT values; list_offsets (n+1,
  // row -> element range) defines rows; validity is per ROW (null list)
  const int64_t* list_offsets = nullptr;
};

class ParquetWriter {
 public:
  ParquetWriter(const std::string& path, std::vector<ColumnDesc> cols,
                int codec = CODEC_ZSTD, int level = 1,
                int64_t max_row_group_rows = 250000)
      : cols_(std::move(cols)),
        codec_(codec),
        level_(level),
        max_rg_rows_(max_row_group_rows) {
    f_ = std::fopen(path.c_str(), "wb");
    if (!f_) throw std::runtime_error("cannot open for write: " + path);
    const char magic[4] = {'P', 'A', 'R', '1'};
    fwrite_all(magic, 4);
  }

  ~ParquetWriter() {
    if (f_) std::fclose(f_);
  }

  // write one row group (caller may pass more rows; we split internally)
  void write_row_group(const std::vector<ColumnData>& data, int64_t num_rows) {
    int64_t off = 0;
    while (off < num_rows) {
      int64_t n = std::min(max_rg_rows_, num_rows - off);
      write_one_row_group(data, off, n);
      off += n;
    }
  }

  // bytes emitted so far — a stable part boundary after every
  // write_row_group (each call fully flushes its pages), so a streaming
  // uploader can ship [uploaded, bytes_written) while the next row group
  // encodes (reference multipart_writer.rs:43 overlap)
  int64_t bytes_written() const { return pos_; }

  void flush_os() {
    if (f_) std::fflush(f_);
  }

  int64_t close() {
    FileMetaData fm;
    fm.version = 2;
    fm.created_by = "lakesoul_amd 0.1.0";
    SchemaElement root;
    root.name = "schema";
    root.type = -1;
    root.repetition = REP_REQUIRED;
    // top-level children: each group (struct/map) counts once
    int32_t top = 0;
    for (size_t i = 0; i < cols_.size(); i++)
      if (cols_[i].parent.empty() || i == 0 ||
          cols_[i - 1].parent != cols_[i].parent)
        top++;
    root.num_children = top;
    fm.schema.push_back(root);
    for (size_t ci = 0; ci < cols_.size(); ci++) {
      auto& c = cols_[ci];
      if (c.parent_kind == 1) {
        // struct: optional group { required members... } — consecutive
        // descs sharing the parent are the members
        if (ci == 0 || cols_[ci - 1].parent != c.parent) {
          int32_t nmem = 0;
          for (size_t j = ci; j < cols_.size() &&
                              cols_[j].parent == c.parent; j++)
            nmem++;
          SchemaElement g;
          g.name = c.parent;
          g.type = -1;
          g.repetition = c.nullable ? REP_OPTIONAL : REP_REQUIRED;
          g.num_children = nmem;
          fm.schema.push_back(g);
        }
        SchemaElement leaf;
        leaf.name = c.name;
        leaf.type = c.physical;
        leaf.repetition = REP_REQUIRED;
        leaf.converted = c.converted;
        leaf.logical = c.logical;
        leaf.int_bit_width = c.int_bit_width;
        leaf.int_signed = c.int_signed;
        leaf.dec_precision = c.dec_precision;
        leaf.dec_scale = c.dec_scale;
        fm.schema.push_back(leaf);
        continue;
      }
      if (c.parent_kind == 2) {
        // map: optional group (MAP) { repeated key_value { key; value } }
        if (ci == 0 || cols_[ci - 1].parent != c.parent) {
          SchemaElement g;
          g.name = c.parent;
          g.type = -1;
          g.repetition = c.nullable ? REP_OPTIONAL : REP_REQUIRED;
          g.converted = 1;  // MAP
          g.num_children = 1;
          fm.schema.push_back(g);
          SchemaElement kv;
          kv.name = "key_value";
          kv.type = -1;
          kv.repetition = REP_REPEATED;
          kv.num_children = 2;
          fm.schema.push_back(kv);
        }
        SchemaElement leaf;
        leaf.name = c.name;
        leaf.type = c.physical;
        leaf.repetition = REP_REQUIRED;
        leaf.converted = c.converted;
        leaf.logical = c.logical;
        leaf.int_bit_width = c.int_bit_width;
        leaf.int_signed = c.int_signed;
        leaf.dec_precision = c.dec_precision;
        leaf.dec_scale = c.dec_scale;
        fm.schema.push_back(leaf);
        continue;
      }
      if (c.is_list) {
        // standard 3-level LIST: optional group (LIST) { repeated group
        // list { required <element>; } } — mirrors the reader's
        // flatten_element LIST shape
        SchemaElement g;
        g.name = c.name;
        g.type = -1;
        g.repetition = REP_OPTIONAL;
        g.converted = 3;  // LIST
        g.num_children = 1;
        fm.schema.push_back(g);
        SchemaElement mid;
        mid.name = "list";
        mid.type = -1;
        mid.repetition = REP_REPEATED;
        mid.num_children = 1;
        fm.schema.push_back(mid);
        SchemaElement leaf;
        leaf.name = "element";
        leaf.type = c.physical;
        leaf.repetition = REP_REQUIRED;
        leaf.converted = c.converted;
        leaf.logical = c.logical;
        leaf.int_bit_width = c.int_bit_width;
        leaf.int_signed = c.int_signed;
        fm.schema.push_back(leaf);
        continue;
      }
      SchemaElement e;
      e.name = c.name;
      e.type = c.physical;
      e.repetition = c.nullable ? REP_OPTIONAL : REP_REQUIRED;
      e.converted = c.converted;
      e.logical = c.logical;
      e.int_bit_width = c.int_bit_width;
      e.int_signed = c.int_signed;
      e.dec_precision = c.dec_precision;
      e.dec_scale = c.dec_scale;
      fm.schema.push_back(e);
    }
    fm.num_rows = total_rows_;
    fm.row_groups = row_groups_;
    auto meta = serialize_file_meta(fm);
    fwrite_all(meta.data(), meta.size());
    uint32_t len = (uint32_t)meta.size();
    fwrite_all(&len, 4);
    const char magic[4] = {'P', 'A', 'R', '1'};
    fwrite_all(magic, 4);
    int64_t size = pos_;
    std::fclose(f_);
    f_ = nullptr;
    return size;
  }

 private:
  void fwrite_all(const void* p, size_t n) {
    if (std::fwrite(p, 1, n, f_) != n) throw std::runtime_error("write failed");
    pos_ += (int64_t)n;
  }

  void write_one_row_group(const std::vector<ColumnData>& data, int64_t row_off,
                           int64_t n) {
    RowGroup rg;
    rg.num_rows = n;
    for (size_t ci = 0; ci < cols_.size(); ci++) {
      const ColumnDesc& cd = cols_[ci];
      const ColumnData& col = data[ci];
      if (cd.is_list) {
        write_list_chunk(cd, col, row_off, n, rg);
        continue;
      }

      // page split: target ~LAKESOUL_PAGE_BYTES decompressed bytes per
      // page. Default 128 KB: measured best for the host zstd decode
      // path (32 KB costs ~15% on the headline scan from extra headers
      // and worse zstd ratio). Set 32768 when running the experimental
      // GPU zstd kernel (profiles/r01_gpu_zstd.md) — it wants thousands
      // of pages in flight per scan unit.
      // Strings keep one page per chunk (host-assembled path).
      static const int64_t kPageBytes = []() {
        const char* e = std::getenv("LAKESOUL_PAGE_BYTES");
        int64_t v = e ? atoll(e) : 0;
        return v > 0 ? v : (int64_t)(128 << 10);
      }();
      int64_t page_rows = n;
      if (cd.physical != PT_BYTE_ARRAY) {
        int es = physical_elem_size(cd.physical);
        if (es < 1) es = 1;
        page_rows = kPageBytes / es;
        if (page_rows < 1) page_rows = 1;
      }

      Statistics stats;
      stats.null_count = 0;
      ColumnMeta cm;
      cm.type = cd.physical;
      cm.encodings = {ENC_PLAIN, ENC_RLE};
      cm.path_in_schema = cd.parent.empty()
          ? std::vector<std::string>{cd.name}
          : std::vector<std::string>{cd.parent, cd.name};
      cm.codec = codec_;
      cm.num_values = n;
      cm.data_page_offset = pos_;

      // LAKESOUL_PAGE_V2=1 emits DataPageV2 (levels outside the
      // compressed region; reader + pyarrow both handle it)
      static const bool kPageV2 = []() {
        const char* e = std::getenv("LAKESOUL_PAGE_V2");
        return e && *e == '1';
      }();
      for (int64_t poff = 0; poff < n || n == 0; poff += page_rows) {
        int64_t pn = n - poff < page_rows ? n - poff : page_rows;
        // ---- assemble page payload (def levels + PLAIN values) ----
        std::vector<uint8_t> payload;
        std::vector<uint8_t> levels;
        int64_t page_nulls = 0;
        const uint8_t* validity =
            cd.nullable && col.validity ? col.validity + row_off + poff : nullptr;
        if (cd.nullable) {
          levels = encode_def_levels(validity, pn);
          if (!kPageV2) {
            uint32_t lv_len = (uint32_t)levels.size();
            payload.insert(payload.end(), (uint8_t*)&lv_len,
                           (uint8_t*)&lv_len + 4);
            payload.insert(payload.end(), levels.begin(), levels.end());
          }
          if (validity)
            for (int64_t i = 0; i < pn; i++) page_nulls += validity[i] ? 0 : 1;
          stats.null_count += page_nulls;
        }
        append_plain_values(cd, col, row_off + poff, pn, validity, payload, stats);

        // ---- compress ----
        std::vector<uint8_t> compressed;
        const uint8_t* body = payload.data();
        size_t body_n = payload.size();
        if (codec_ == CODEC_ZSTD) {
          compressed = zstd_compress(payload.data(), payload.size(), level_);
          body = compressed.data();
          body_n = compressed.size();
        } else if (codec_ != CODEC_UNCOMPRESSED) {
          throw std::runtime_error("writer supports zstd/uncompressed only");
        }

        // ---- page header + emit ----
        PageHeader ph;
        ph.type = kPageV2 ? PAGE_DATA_V2 : PAGE_DATA;
        ph.num_values = (int32_t)pn;
        ph.encoding = ENC_PLAIN;
        ph.def_encoding = ENC_RLE;
        ph.rep_encoding = ENC_RLE;
        if (kPageV2) {
          // v2: levels sit uncompressed in front of the compressed values
          ph.num_nulls = (int32_t)page_nulls;
          ph.num_rows = (int32_t)pn;
          ph.def_levels_byte_length = (int32_t)levels.size();
          ph.rep_levels_byte_length = 0;
          ph.v2_is_compressed = codec_ != CODEC_UNCOMPRESSED;
          ph.uncompressed_size = (int32_t)(levels.size() + payload.size());
          ph.compressed_size = (int32_t)(levels.size() + body_n);
        } else {
          ph.uncompressed_size = (int32_t)payload.size();
          ph.compressed_size = (int32_t)body_n;
        }
        auto ph_bytes = serialize_page_header(ph);
        cm.total_uncompressed_size +=
            (int64_t)(ph_bytes.size() + ph.uncompressed_size);
        cm.total_compressed_size +=
            (int64_t)(ph_bytes.size() + ph.compressed_size);
        fwrite_all(ph_bytes.data(), ph_bytes.size());
        if (kPageV2 && !levels.empty())
          fwrite_all(levels.data(), levels.size());
        fwrite_all(body, body_n);
        if (n == 0) break;
      }
      cm.stats = stats;
      rg.columns.push_back(cm);
      rg.total_byte_size += cm.total_uncompressed_size;
    }
    row_groups_.push_back(std::move(rg));
    total_rows_ += n;
  }

  template <typename T>
  void minmax_update(const uint8_t* p, Statistics& s) {
    T v;
    std::memcpy(&v, p, sizeof(T));
    T mn, mx;
    if (!s.has_min_max) {
      mn = mx = v;
      s.has_min_max = true;
    } else {
      std::memcpy(&mn, s.min_value.data(), sizeof(T));
      std::memcpy(&mx, s.max_value.data(), sizeof(T));
      if (v < mn) mn = v;
      if (v > mx) mx = v;
    }
    s.min_value.assign((char*)&mn, sizeof(T));
    s.max_value.assign((char*)&mx, sizeof(T));
  }

  // One-page LIST chunk (3-level standard layout, v1 page:
  // [u32][rep RLE][u32][def RLE][PLAIN element values]) — the exact
  // shape the reader's list path decodes (read side ~line 655).
  void write_list_chunk(const ColumnDesc& cd, const ColumnData& col,
                        int64_t row_off, int64_t n, RowGroup& rg) {
    if (!col.list_offsets) throw std::runtime_error("list column without offsets");
    const int64_t* offs = col.list_offsets;
    std::vector<uint8_t> reps, defs;
    int64_t lo = offs[row_off], hi = offs[row_off + n];
    reps.reserve((size_t)(hi - lo + n));
    defs.reserve((size_t)(hi - lo + n));
    // optional list group + repeated: max_def = 2, elements required
    for (int64_t i = 0; i < n; i++) {
      bool valid = !col.validity || col.validity[row_off + i];
      int64_t len = offs[row_off + i + 1] - offs[row_off + i];
      if (!valid) {
        reps.push_back(0);
        defs.push_back(0);
      } else if (len == 0) {
        reps.push_back(0);
        defs.push_back(1);
      } else {
        for (int64_t j = 0; j < len; j++) {
          reps.push_back(j == 0 ? 0 : 1);
          defs.push_back(2);
        }
      }
    }
    int64_t entries = (int64_t)reps.size();
    auto rep_rle = encode_levels(reps.data(), entries);
    auto def_rle = encode_levels(defs.data(), entries);

    std::vector<uint8_t> payload;
    uint32_t rl = (uint32_t)rep_rle.size(), dl = (uint32_t)def_rle.size();
    payload.insert(payload.end(), (uint8_t*)&rl, (uint8_t*)&rl + 4);
    payload.insert(payload.end(), rep_rle.begin(), rep_rle.end());
    payload.insert(payload.end(), (uint8_t*)&dl, (uint8_t*)&dl + 4);
    payload.insert(payload.end(), def_rle.begin(), def_rle.end());
    if (cd.physical == PT_BYTE_ARRAY) {
      // list<string>: elements as PLAIN len-prefixed byte arrays
      // (col.offsets = element byte offsets, col.bytes = payload)
      if (!col.offsets) throw std::runtime_error("list<string> without element offsets");
      for (int64_t e = lo; e < hi; e++) {
        uint32_t len = (uint32_t)(col.offsets[e + 1] - col.offsets[e]);
        payload.insert(payload.end(), (uint8_t*)&len, (uint8_t*)&len + 4);
        payload.insert(payload.end(), col.bytes + col.offsets[e],
                       col.bytes + col.offsets[e + 1]);
      }
    } else {
      int es = physical_elem_size(cd.physical);
      payload.insert(payload.end(), col.data + lo * es, col.data + hi * es);
    }

    std::vector<uint8_t> compressed;
    const uint8_t* body = payload.data();
    size_t body_n = payload.size();
    if (codec_ == CODEC_ZSTD) {
      compressed = zstd_compress(payload.data(), payload.size(), level_);
      body = compressed.data();
      body_n = compressed.size();
    } else if (codec_ != CODEC_UNCOMPRESSED) {
      throw std::runtime_error("writer supports zstd/uncompressed only");
    }

    ColumnMeta cm;
    cm.type = cd.physical;
    cm.encodings = {ENC_PLAIN, ENC_RLE};
    cm.path_in_schema = cd.parent_kind == 2
        ? std::vector<std::string>{cd.parent, "key_value", cd.name}
        : std::vector<std::string>{cd.name, "list", "element"};
    cm.codec = codec_;
    cm.num_values = entries;
    cm.data_page_offset = pos_;

    PageHeader ph;
    ph.type = PAGE_DATA;
    ph.num_values = (int32_t)entries;
    ph.encoding = ENC_PLAIN;
    ph.def_encoding = ENC_RLE;
    ph.rep_encoding = ENC_RLE;
    ph.uncompressed_size = (int32_t)payload.size();
    ph.compressed_size = (int32_t)body_n;
    auto ph_bytes = serialize_page_header(ph);
    cm.total_uncompressed_size = (int64_t)(ph_bytes.size() + payload.size());
    cm.total_compressed_size = (int64_t)(ph_bytes.size() + body_n);
    fwrite_all(ph_bytes.data(), ph_bytes.size());
    fwrite_all(body, body_n);
    Statistics stats;
    int64_t nulls = 0;
    if (col.validity)
      for (int64_t i = 0; i < n; i++) nulls += col.validity[row_off + i] ? 0 : 1;
    stats.null_count = nulls;
    cm.stats = stats;
    rg.columns.push_back(cm);
    rg.total_byte_size += cm.total_uncompressed_size;
  }

  void append_plain_values(const ColumnDesc& cd, const ColumnData& col,
                           int64_t row_off, int64_t n, const uint8_t* validity,
                           std::vector<uint8_t>& payload, Statistics& stats) {
    if (cd.physical == PT_BYTE_ARRAY) {
      const int32_t* offs = col.offsets + row_off;
      for (int64_t i = 0; i < n; i++) {
        if (validity && !validity[i]) continue;
        uint32_t len = (uint32_t)(offs[i + 1] - offs[i]);
        payload.insert(payload.end(), (uint8_t*)&len, (uint8_t*)&len + 4);
        payload.insert(payload.end(), col.bytes + offs[i],
                       col.bytes + offs[i + 1]);
        // lexicographic min/max
        std::string v((const char*)(col.bytes + offs[i]), len);
        if (!stats.has_min_max) {
          stats.min_value = stats.max_value = v;
          stats.has_min_max = true;
        } else {
          if (v < stats.min_value) stats.min_value = v;
          if (v > stats.max_value) stats.max_value = v;
        }
      }
      return;
    }
    if (cd.physical == PT_BOOLEAN) {
      // bit-pack non-null byte values LSB-first
      uint8_t cur = 0;
      int nb = 0;
      const uint8_t* d = col.data + row_off;
      for (int64_t i = 0; i < n; i++) {
        if (validity && !validity[i]) continue;
        if (d[i]) cur |= (uint8_t)(1 << nb);
        if (++nb == 8) {
          payload.push_back(cur);
          cur = 0;
          nb = 0;
        }
      }
      if (nb) payload.push_back(cur);
      return;
    }
    int es = physical_elem_size(cd.physical);
    const uint8_t* d = col.data + row_off * es;
    if (!validity) {
      payload.insert(payload.end(), d, d + n * es);
      switch (cd.physical) {
        case PT_INT32: minmax_range<int32_t>(d, n, nullptr, stats); break;
        case PT_INT64: minmax_range<int64_t>(d, n, nullptr, stats); break;
        case PT_FLOAT: minmax_range<float>(d, n, nullptr, stats); break;
        case PT_DOUBLE: minmax_range<double>(d, n, nullptr, stats); break;
      }
    } else {
      // dense-pack valid runs with bulk copies (run-length segments)
      size_t base = payload.size();
      int64_t i = 0;
      while (i < n) {
        while (i < n && !validity[i]) i++;
        int64_t j = i;
        while (j < n && validity[j]) j++;
        if (j > i)
          payload.insert(payload.end(), d + i * es, d + j * es);
        i = j;
      }
      (void)base;
      switch (cd.physical) {
        case PT_INT32: minmax_range<int32_t>(d, n, validity, stats); break;
        case PT_INT64: minmax_range<int64_t>(d, n, validity, stats); break;
        case PT_FLOAT: minmax_range<float>(d, n, validity, stats); break;
        case PT_DOUBLE: minmax_range<double>(d, n, validity, stats); break;
      }
    }
  }

  // range min/max with register accumulation (the per-element
  // string-assign version cost ~80% of encode time; measured 226 MB/s ->
  // memcpy-bound after this)
  template <typename T>
  void minmax_range(const uint8_t* d, int64_t n, const uint8_t* validity,
                    Statistics& s) {
    T mn = T(), mx = T();
    bool any = false;
    for (int64_t i = 0; i < n; i++) {
      if (validity && !validity[i]) continue;
      T v;
      std::memcpy(&v, d + i * (int64_t)sizeof(T), sizeof(T));
      if (!any) {
        mn = mx = v;
        any = true;
      } else {
        if (v < mn) mn = v;
        if (v > mx) mx = v;
      }
    }
    if (!any) return;
    if (s.has_min_max) {
      T omn, omx;
      std::memcpy(&omn, s.min_value.data(), sizeof(T));
      std::memcpy(&omx, s.max_value.data(), sizeof(T));
      if (omn < mn) mn = omn;
      if (omx > mx) mx = omx;
    }
    s.min_value.assign((char*)&mn, sizeof(T));
    s.max_value.assign((char*)&mx, sizeof(T));
    s.has_min_max = true;
  }

  std::FILE* f_ = nullptr;
  std::vector<ColumnDesc> cols_;
  int codec_;
  int level_;
  int64_t max_rg_rows_;
  int64_t pos_ = 0;
  int64_t total_rows_ = 0;
  std::vector<RowGroup> row_groups_;
};

// ---------------------------------------------------------------------- //
// reader
// ---------------------------------------------------------------------- //

class ParquetFile {
 public:
  explicit ParquetFile(const std::string& path) : path_(path) {
    fd_ = ::open(path.c_str(), O_RDONLY);
    if (fd_ < 0) throw std::runtime_error("cannot open " + path);
    struct stat st;
    if (fstat(fd_, &st) != 0) throw std::runtime_error("fstat failed " + path);
    size_ = (size_t)st.st_size;
    if (size_ < 12) throw std::runtime_error("file too small: " + path);
    map_ = (uint8_t*)mmap(nullptr, size_, PROT_READ, MAP_PRIVATE, fd_, 0);
    if (map_ == MAP_FAILED) throw std::runtime_error("mmap failed " + path);
    if (std::memcmp(map_, "PAR1", 4) != 0 ||
        std::memcmp(map_ + size_ - 4, "PAR1", 4) != 0)
      throw std::runtime_error("not a parquet file: " + path);
    uint32_t meta_len;
    std::memcpy(&meta_len, map_ + size_ - 8, 4);
    if (meta_len + 12 > size_) throw std::runtime_error("bad footer length");
    meta_ = parse_file_meta(map_ + size_ - 8 - meta_len, meta_len);
    // build leaf column list (flat schemas: children of root)
    build_columns();
  }

  ~ParquetFile() {
    if (map_ && map_ != MAP_FAILED) munmap(map_, size_);
    if (fd_ >= 0) ::close(fd_);
  }

  const FileMetaData& meta() const { return meta_; }
  const std::vector<ColumnDesc>& columns() const { return cols_; }
  const std::string& path() const { return path_; }
  int64_t num_rows() const { return meta_.num_rows; }
  size_t num_row_groups() const { return meta_.row_groups.size(); }

  int column_index(const std::string& name) const {
    for (size_t i = 0; i < cols_.size(); i++)
      if (cols_[i].name == name) return (int)i;
    return -1;
  }

  struct IdxPage {
    int64_t out_off;
    int64_t n;            // non-null values in page
    int64_t payload_off;  // offset into ChunkData.values
    int64_t payload_len;
    int32_t bit_width;
  };

  struct CompPage {  // GPU-decompress job (snappy/zstd, levels-free pages)
    int64_t comp_off;   // into comp buffer
    int64_t comp_len;
    int64_t out_off;    // into the chunk's values stream
    int64_t out_len;
    int32_t codec = CODEC_SNAPPY;
  };

  struct DeferPage {  // host direct-decompress job (zstd/uncompressed,
    int64_t file_off;  // levels-free PLAIN pages): source bytes stay in
    int64_t comp_len;  // the mmap until fill() decompresses straight into
    int64_t out_off;   // the caller's pinned buffer
    int64_t out_len;
    int32_t codec;
  };

  struct ChunkData {
    int32_t physical = 0;
    int64_t num_values = 0;  // rows in chunk
    int64_t null_count = 0;
    bool is_dict = false;
    std::vector<uint8_t> values;    // PLAIN payload (dense non-null) or idx payloads
    std::vector<uint8_t> validity;  // byte/value, empty if no nulls present
    std::vector<uint8_t> dict;      // PLAIN dictionary payload
    int64_t dict_num_values = 0;
    std::vector<IdxPage> idx_pages;
    // gpu_compressed mode: values stays EMPTY; `comp` holds the raw
    // snappy page bodies and `comp_pages` the decompress jobs; values
    // stream length is values_len (sum of uncompressed page payloads)
    bool gpu_compressed = false;
    int64_t values_len = 0;
    std::vector<uint8_t> comp;
    std::vector<CompPage> comp_pages;
    // host_deferred mode: values stays EMPTY; defer_pages decompress
    // later directly into the final buffer (no staging copy)
    bool host_deferred = false;
    std::vector<DeferPage> defer_pages;
    // LIST columns: rows -> element ranges (+1 sentinel) and per-row
    // validity; values/validity above then describe the ELEMENTS
    bool is_list = false;
    std::vector<int64_t> list_offsets;
    std::vector<uint8_t> list_validity;
  };

  const uint8_t* data_at(int64_t off) const { return map_ + off; }

  // gpu_snappy / gpu_zstd: defer page decompression to the GPU kernels
  // when the column is REQUIRED (no def-level section inside the
  // compressed blob), PLAIN-encoded, non-boolean v1 pages.
  ChunkData read_chunk(size_t rg, size_t col, bool gpu_snappy = false,
                       bool defer_host = false, bool gpu_zstd = false) const {
    // descriptor cache: lakehouse files are immutable, and deferred /
    // gpu-staged chunks are pure page descriptors (no payload) — cache
    // them so repeated scans skip the per-page thrift header walk
    // (reference analog: the global metadata cache, session.rs:86-105)
    uint64_t ckey = ((uint64_t)rg << 24) | ((uint64_t)col << 4) |
                    (defer_host ? 1u : 0u) | (gpu_snappy ? 2u : 0u) |
                    (gpu_zstd ? 4u : 0u);
    if (defer_host || gpu_snappy || gpu_zstd) {
      std::lock_guard<std::mutex> lk(chunk_mu_);
      auto it = chunk_meta_cache_.find(ckey);
      if (it != chunk_meta_cache_.end()) return *it->second;
    }
    ChunkData out_cached = read_chunk_impl(rg, col, gpu_snappy, defer_host,
                                           gpu_zstd);
    if ((out_cached.host_deferred || out_cached.gpu_compressed) &&
        out_cached.values.empty() && out_cached.validity.empty() &&
        out_cached.dict.empty() && out_cached.comp.empty()) {
      std::lock_guard<std::mutex> lk(chunk_mu_);
      chunk_meta_cache_.emplace(
          ckey, std::make_shared<const ChunkData>(out_cached));
    }
    return out_cached;
  }

  ChunkData read_chunk_impl(size_t rg, size_t col, bool gpu_snappy = false,
                            bool defer_host = false,
                            bool gpu_zstd = false) const {
    const RowGroup& g = meta_.row_groups.at(rg);
    const ColumnMeta& cm = g.columns.at(col);
    const ColumnDesc& cd = cols_.at(col);
    ChunkData out;
    // foreign-file physical types are translated at page-decode time:
    // FLBA becomes a length-prefixed byte_array stream, INT96 becomes
    // timestamp[ns] int64 — everything downstream sees standard types
    out.physical = cd.physical == PT_FLBA ? PT_BYTE_ARRAY
                   : cd.physical == PT_INT96 ? PT_INT64
                                             : cd.physical;
    out.num_values = cm.num_values;

    int64_t off = cm.dictionary_page_offset >= 0
                      ? cm.dictionary_page_offset
                      : cm.data_page_offset;
    // some writers put data_page_offset < dictionary_page_offset wrongly;
    // start at the min positive
    if (cm.dictionary_page_offset >= 0 &&
        cm.data_page_offset < cm.dictionary_page_offset)
      off = cm.data_page_offset;
    int64_t end = off + cm.total_compressed_size;
    int64_t values_seen = 0;
    out.values.reserve((size_t)cm.total_uncompressed_size);
    bool any_null_page = false;
    // validity allocated lazily on the first null (REQUIRED columns and
    // fully-valid chunks never pay the memset)
    auto ensure_validity = [&]() {
      if (out.validity.empty()) out.validity.assign((size_t)cm.num_values, 1);
    };

    while (off < end && values_seen < cm.num_values) {
      ThriftReader r(map_ + off, (size_t)(end - off));
      const uint8_t* hdr_start = map_ + off;
      PageHeader ph = parse_page_header(r);
      int64_t hdr_len = (int64_t)r.consumed(hdr_start);
      const uint8_t* body = map_ + off + hdr_len;
      off += hdr_len + ph.compressed_size;

      if (ph.type == PAGE_DICTIONARY) {
        std::vector<uint8_t> draw((size_t)ph.uncompressed_size);
        decompress_into(cm.codec, draw.data(), draw.size(), body,
                        ph.compressed_size);
        if (cd.physical == PT_FLBA)
          flba_to_byte_array(draw.data(), ph.dict_num_values, cd.type_length,
                             out.dict);
        else if (cd.physical == PT_INT96)
          int96_to_ns(draw.data(), ph.dict_num_values, out.dict);
        else
          out.dict = std::move(draw);
        out.dict_num_values = ph.dict_num_values;
        continue;
      }
      if (ph.type != PAGE_DATA && ph.type != PAGE_DATA_V2) continue;

      if (out.gpu_compressed &&
          !((cm.codec == CODEC_SNAPPY ||
             (cm.codec == CODEC_ZSTD &&
              ph.uncompressed_size <= (512 << 10))) &&
            ph.type == PAGE_DATA && ph.encoding == ENC_PLAIN)) {
        // mixed chunk (dict fallback / oversized page) — redo on host
        return read_chunk_impl(rg, col, false);
      }
      if (((gpu_snappy && cm.codec == CODEC_SNAPPY) ||
           (gpu_zstd && cm.codec == CODEC_ZSTD &&
            ph.uncompressed_size <= (512 << 10))) &&
          ph.type == PAGE_DATA && ph.encoding == ENC_PLAIN && !cd.nullable &&
          cd.physical != PT_BOOLEAN && cd.physical != PT_BYTE_ARRAY &&
          cd.physical != PT_FLBA && cd.physical != PT_INT96 && !cd.is_list &&
          out.dict.empty() && out.values.empty() && !out.host_deferred) {
        out.gpu_compressed = true;
        CompPage cp;
        cp.comp_off = (int64_t)out.comp.size();
        cp.comp_len = ph.compressed_size;
        cp.out_off = out.values_len;
        cp.out_len = ph.uncompressed_size;
        cp.codec = cm.codec;
        out.comp.insert(out.comp.end(), body, body + ph.compressed_size);
        out.comp_pages.push_back(cp);
        out.values_len += ph.uncompressed_size;
        values_seen += ph.num_values;
        continue;
      }
      if (defer_host &&
          (cm.codec == CODEC_ZSTD || cm.codec == CODEC_UNCOMPRESSED) &&
          ph.type == PAGE_DATA && ph.encoding == ENC_PLAIN && !cd.nullable &&
          cd.physical != PT_BOOLEAN && cd.physical != PT_BYTE_ARRAY &&
          cd.physical != PT_FLBA && cd.physical != PT_INT96 && !cd.is_list &&
          out.dict.empty() && out.values.empty() && !out.gpu_compressed) {
        // levels-free PLAIN page: defer decompression — fill() writes it
        // straight into the destination buffer
        out.host_deferred = true;
        DeferPage dp;
        dp.file_off = (int64_t)(body - map_);
        dp.comp_len = ph.compressed_size;
        dp.out_off = out.values_len;
        dp.out_len = ph.uncompressed_size;
        dp.codec = cm.codec;
        out.defer_pages.push_back(dp);
        out.values_len += ph.uncompressed_size;
        values_seen += ph.num_values;
        continue;
      }
      if (out.host_deferred &&
          !((cm.codec == CODEC_ZSTD || cm.codec == CODEC_UNCOMPRESSED) &&
            ph.type == PAGE_DATA && ph.encoding == ENC_PLAIN)) {
        return read_chunk_impl(rg, col, false, false);  // mixed: redo staged
      }

      std::vector<uint8_t> page;
      const uint8_t* vals;
      size_t vals_len;
      int64_t nv = ph.num_values;
      int64_t page_nulls = 0;          // entries without a stored value
      int64_t page_markers = 0;        // list row-marker entries (not null
                                       // ELEMENTS — excluded from null_count)
      int64_t elem_slots_page_start = (int64_t)out.validity.size();

      if (ph.type == PAGE_DATA) {
        page.resize(ph.uncompressed_size);
        decompress_into(cm.codec, page.data(), page.size(), body,
                        ph.compressed_size);
        const uint8_t* p = page.data();
        size_t rem = page.size();
        if (cd.is_list) {
          // standard 3-level LIST: [u32][rep RLE][u32][def RLE][values].
          // rep==0 starts a row; def encodes null-list / empty-list /
          // null-element / value (parquet.thrift nested encoding)
          out.is_list = true;
          uint32_t rl_len, dl_len;
          std::memcpy(&rl_len, p, 4);
          std::vector<uint8_t> reps((size_t)nv);
          rle_decode<uint8_t>(p + 4, rl_len, 1, nv, reps.data());
          p += 4 + rl_len;
          rem -= 4 + rl_len;
          std::memcpy(&dl_len, p, 4);
          int def_bw = 1;
          while ((1 << def_bw) <= cd.max_def) def_bw++;
          std::vector<uint8_t> defs((size_t)nv);
          rle_decode<uint8_t>(p + 4, dl_len, def_bw, nv, defs.data());
          p += 4 + dl_len;
          rem -= 4 + dl_len;
          int elem_threshold = cd.max_def - (cd.elem_nullable ? 1 : 0);
          for (int64_t i = 0; i < nv; i++) {
            if (reps[i] == 0) {
              out.list_offsets.push_back((int64_t)out.validity.size());
              out.list_validity.push_back(
                  (cd.nullable && defs[i] == 0) ? 0 : 1);
            }
            if (defs[i] >= elem_threshold) {
              bool present = defs[i] == cd.max_def;
              out.validity.push_back(present ? 1 : 0);
              if (!present) page_nulls++;  // null ELEMENT (no stored value)
            } else {
              page_nulls++;      // row marker entry carries no value
              page_markers++;
            }
          }
          vals = p;
          vals_len = rem;
        } else if (cd.nullable) {
          if (ph.def_encoding != ENC_RLE)
            throw std::runtime_error("unsupported def-level encoding");
          uint32_t lv_len;
          std::memcpy(&lv_len, p, 4);
          int def_bw = 1;
          while ((1 << def_bw) <= cd.max_def) def_bw++;
          std::vector<uint8_t> levels((size_t)nv);
          rle_decode<uint8_t>(p + 4, lv_len, def_bw, nv, levels.data());
          uint8_t full = (uint8_t)cd.max_def;
          for (int64_t i = 0; i < nv; i++) {
            if (levels[i] != full) {   // null at leaf OR a null ancestor
              ensure_validity();
              out.validity[values_seen + i] = 0;
              page_nulls++;
            }
          }
          p += 4 + lv_len;
          rem -= 4 + lv_len;
        }
        vals = p;
        vals_len = rem;
      } else {  // DATA_PAGE_V2: levels uncompressed, values possibly compressed
        if (cd.is_list)
          throw std::runtime_error("LIST columns: DataPageV2 not supported yet");
        int64_t lv = ph.rep_levels_byte_length + ph.def_levels_byte_length;
        if (cd.nullable && ph.def_levels_byte_length > 0) {
          std::vector<uint8_t> levels((size_t)nv);
          rle_decode<uint8_t>(body + ph.rep_levels_byte_length,
                              ph.def_levels_byte_length, 1, nv, levels.data());
          for (int64_t i = 0; i < nv; i++) {
            if (!levels[i]) {
              ensure_validity();
              out.validity[values_seen + i] = 0;
              page_nulls++;
            }
          }
        }
        size_t comp_vals = (size_t)(ph.compressed_size - lv);
        size_t uncomp_vals = (size_t)(ph.uncompressed_size - lv);
        page.resize(uncomp_vals);
        if (ph.v2_is_compressed && cm.codec != CODEC_UNCOMPRESSED) {
          decompress_into(cm.codec, page.data(), uncomp_vals, body + lv,
                          comp_vals);
        } else {
          std::memcpy(page.data(), body + lv, uncomp_vals);
        }
        vals = page.data();
        vals_len = uncomp_vals;
      }
      if (page_nulls) any_null_page = true;
      out.null_count += page_nulls - page_markers;  // null ELEMENTS only
      int64_t nonnull = nv - page_nulls;            // stored values

      if (ph.encoding == ENC_PLAIN) {
        if (cd.physical == PT_BOOLEAN) {
          // unpack bits to bytes here (cheap, host)
          size_t base = out.values.size();
          out.values.resize(base + (size_t)nonnull);
          for (int64_t i = 0; i < nonnull; i++)
            out.values[base + i] = (vals[i >> 3] >> (i & 7)) & 1;
        } else if (cd.physical == PT_FLBA) {
          flba_to_byte_array(vals, nonnull, cd.type_length, out.values);
        } else if (cd.physical == PT_INT96) {
          int96_to_ns(vals, nonnull, out.values);
        } else {
          out.values.insert(out.values.end(), vals, vals + vals_len);
        }
      } else if (ph.encoding == ENC_RLE_DICTIONARY ||
                 ph.encoding == ENC_PLAIN_DICTIONARY) {
        out.is_dict = true;
        int bw = vals[0];
        IdxPage ip;
        // row offset incl. nulls (element-slot offset for LIST columns);
        // indices themselves are dense
        ip.out_off = cd.is_list ? elem_slots_page_start : values_seen;
        ip.n = nonnull;
        ip.payload_off = (int64_t)out.values.size();
        ip.payload_len = (int64_t)(vals_len - 1);
        ip.bit_width = bw;
        out.idx_pages.push_back(ip);
        out.values.insert(out.values.end(), vals + 1, vals + vals_len);
      } else {
        throw std::runtime_error("unsupported data encoding " +
                                 std::to_string(ph.encoding));
      }
      values_seen += nv;
    }
    (void)any_null_page;
    if (cd.is_list) {
      out.list_offsets.push_back((int64_t)out.validity.size());
      out.num_values = (int64_t)out.validity.size();  // element slots
      if (out.null_count == 0) out.validity.clear();  // all elements valid
    }
    return out;
  }

 private:
  void build_columns() {
    if (meta_.schema.empty()) throw std::runtime_error("empty schema");
    // flat schema: every element after root with num_children==0
    size_t i = 1;
    while (i < meta_.schema.size())
      i = flatten_element(i, "", 0);
  }

  // recursive schema flatten: returns the index after the subtree.
  // prefix names nested leaves "outer.inner"; def_base accumulates
  // optional ancestors so leaf def-levels decode with the right width
  // and null threshold. LIST groups use the dedicated 3-level path.
  size_t flatten_element(size_t i, const std::string& prefix, int def_base) {
    const SchemaElement& e = meta_.schema[i];
    std::string name = prefix.empty() ? e.name : prefix + "." + e.name;
    if (e.num_children > 0) {
      if (e.converted == 3) {  // LIST
        if (i + 2 < meta_.schema.size() &&
            meta_.schema[i + 1].repetition == REP_REPEATED &&
            meta_.schema[i + 1].num_children == 1 &&
            meta_.schema[i + 2].num_children == 0 && def_base == 0) {
          const SchemaElement& leaf = meta_.schema[i + 2];
          ColumnDesc c;
          c.name = name;
          c.physical = leaf.type;
          c.nullable = e.repetition == REP_OPTIONAL;
          c.converted = leaf.converted;
          c.logical = leaf.logical;
          c.int_bit_width = leaf.int_bit_width;
          c.int_signed = leaf.int_signed;
          c.dec_precision = leaf.dec_precision;
          c.dec_scale = leaf.dec_scale;
          c.type_length = leaf.type_length;
          c.is_list = true;
          c.elem_nullable = leaf.repetition == REP_OPTIONAL;
          c.max_def = (c.nullable ? 1 : 0) + 1 + (c.elem_nullable ? 1 : 0);
          cols_.push_back(c);
          return i + 3;
        }
        throw std::runtime_error("unsupported LIST shape: " + name);
      }
      if ((e.converted == 1 || e.converted == 2) && def_base == 0) {  // MAP / legacy MAP_KEY_VALUE
        // standard shape: group (MAP) { repeated group key_value {
        //   required key; [optional] value; } } -> surface as TWO
        // parallel list columns name.key / name.value (same row ranges)
        if (i + 3 < meta_.schema.size() &&
            meta_.schema[i + 1].repetition == REP_REPEATED &&
            meta_.schema[i + 1].num_children == 2 &&
            meta_.schema[i + 2].num_children == 0 &&
            meta_.schema[i + 3].num_children == 0) {
          for (int k = 0; k < 2; k++) {
            const SchemaElement& leaf = meta_.schema[i + 2 + k];
            ColumnDesc c;
            c.name = name + "." + leaf.name;
            c.physical = leaf.type;
            c.nullable = e.repetition == REP_OPTIONAL;
            c.converted = leaf.converted;
            c.logical = leaf.logical;
            c.int_bit_width = leaf.int_bit_width;
            c.int_signed = leaf.int_signed;
            c.dec_precision = leaf.dec_precision;
            c.dec_scale = leaf.dec_scale;
            c.type_length = leaf.type_length;
            c.is_list = true;
            c.elem_nullable = leaf.repetition == REP_OPTIONAL;
            c.max_def = (c.nullable ? 1 : 0) + 1 + (c.elem_nullable ? 1 : 0);
            cols_.push_back(c);
          }
          return i + 4;
        }
        throw std::runtime_error("unsupported MAP shape: " + name);
      }
      if (e.repetition == REP_REPEATED)
        throw std::runtime_error("repeated groups (MAP/legacy lists) not "
                                 "supported yet: " + name);
      // plain struct: flatten children as dotted leaf columns
      int base = def_base + (e.repetition == REP_OPTIONAL ? 1 : 0);
      size_t j = i + 1;
      for (int k = 0; k < e.num_children; k++)
        j = flatten_element(j, name, base);
      return j;
    }
    ColumnDesc c;
    c.name = name;
    c.physical = e.type;
    c.max_def = def_base + (e.repetition == REP_OPTIONAL ? 1 : 0);
    c.nullable = c.max_def > 0;  // null if ANY optional ancestor is null
    c.converted = e.converted;
    c.logical = e.logical;
    c.int_bit_width = e.int_bit_width;
    c.int_signed = e.int_signed;
    c.dec_precision = e.dec_precision;
    c.dec_scale = e.dec_scale;
    c.type_length = e.type_length;
    cols_.push_back(c);
    return i + 1;
  }

  mutable std::mutex chunk_mu_;
  mutable std::unordered_map<uint64_t, std::shared_ptr<const ChunkData>>
      chunk_meta_cache_;

  std::string path_;
  int fd_ = -1;
  uint8_t* map_ = nullptr;
  size_t size_ = 0;
  FileMetaData meta_;
  std::vector<ColumnDesc> cols_;
};

// ---------------------------------------------------------------------- //
// CPU decode of a ChunkData into a full-length typed column
// ---------------------------------------------------------------------- //

struct DecodedColumn {
  std::vector<uint8_t> data;      // elem_size * num_values (fixed width)
  std::vector<int32_t> offsets;   // byte_array
  std::vector<uint8_t> bytes;     // byte_array
  std::vector<uint8_t> validity;  // empty if no nulls
  int64_t num_values = 0;
  // LIST columns: rows -> element ranges (+1 sentinel) + per-row validity
  bool is_list = false;
  std::vector<int64_t> list_offsets;
  std::vector<uint8_t> list_validity;
};

inline DecodedColumn decode_chunk_cpu(const ParquetFile::ChunkData& ch) {
  DecodedColumn out;
  out.num_values = ch.num_values;
  out.validity = ch.validity;
  if (ch.is_list) {
    out.is_list = true;
    out.list_offsets = ch.list_offsets;
    out.list_validity = ch.list_validity;
  }
  const uint8_t* validity = ch.validity.empty() ? nullptr : ch.validity.data();

  // 1) materialize dense (non-null) values
  std::vector<uint8_t> dense;         // fixed width
  std::vector<int32_t> dense_offs;    // byte_array
  std::vector<uint8_t> dense_bytes;
  int es = physical_elem_size(ch.physical);
  int64_t nonnull = ch.num_values - ch.null_count;

  if (!ch.is_dict) {
    if (ch.physical == PT_BYTE_ARRAY) {
      dense_offs.reserve(nonnull + 1);
      dense_offs.push_back(0);
      const uint8_t* p = ch.values.data();
      const uint8_t* endp = p + ch.values.size();
      while (p + 4 <= endp && (int64_t)dense_offs.size() <= nonnull) {
        uint32_t len;
        std::memcpy(&len, p, 4);
        p += 4;
        dense_bytes.insert(dense_bytes.end(), p, p + len);
        p += len;
        dense_offs.push_back((int32_t)dense_bytes.size());
      }
    } else {
      dense = ch.values;
    }
  } else {
    // dict: decode indices then gather
    std::vector<int32_t> idx((size_t)nonnull);
    for (auto& ip : ch.idx_pages) {
      // out_off counts rows incl. nulls; compute dense offset by counting
      // valid rows before out_off
      int64_t dense_off = 0;
      if (validity) {
        for (int64_t i = 0; i < ip.out_off; i++) dense_off += validity[i];
      } else {
        dense_off = ip.out_off;
      }
      rle_decode<int32_t>(ch.values.data() + ip.payload_off,
                          (size_t)ip.payload_len, ip.bit_width, ip.n,
                          idx.data() + dense_off);
    }
    // parse dict values
    if (ch.physical == PT_BYTE_ARRAY) {
      std::vector<int32_t> doffs;
      std::vector<uint8_t> dbytes;
      doffs.push_back(0);
      const uint8_t* p = ch.dict.data();
      for (int64_t i = 0; i < ch.dict_num_values; i++) {
        uint32_t len;
        std::memcpy(&len, p, 4);
        p += 4;
        dbytes.insert(dbytes.end(), p, p + len);
        p += len;
        doffs.push_back((int32_t)dbytes.size());
      }
      dense_offs.push_back(0);
      for (int64_t i = 0; i < nonnull; i++) {
        int32_t k = idx[i];
        dense_bytes.insert(dense_bytes.end(), dbytes.data() + doffs[k],
                           dbytes.data() + doffs[k + 1]);
        dense_offs.push_back((int32_t)dense_bytes.size());
      }
    } else {
      int des = es;
      dense.resize((size_t)nonnull * des);
      for (int64_t i = 0; i < nonnull; i++)
        std::memcpy(dense.data() + i * des, ch.dict.data() + idx[i] * des, des);
    }
  }

  // 2) scatter through validity to full length
  if (ch.physical == PT_BYTE_ARRAY) {
    out.offsets.resize(ch.num_values + 1);
    out.bytes = std::move(dense_bytes);
    int64_t di = 0;
    out.offsets[0] = 0;
    for (int64_t i = 0; i < ch.num_values; i++) {
      if (!validity || validity[i]) {
        out.offsets[i + 1] = dense_offs[di + 1];
        di++;
      } else {
        out.offsets[i + 1] = out.offsets[i];
      }
    }
  } else {
    out.data.resize((size_t)ch.num_values * es, 0);
    if (!validity) {
      std::memcpy(out.data.data(), dense.data(),
                  std::min(dense.size(), out.data.size()));
    } else {
      int64_t di = 0;
      for (int64_t i = 0; i < ch.num_values; i++) {
        if (validity[i]) {
          std::memcpy(out.data.data() + i * es, dense.data() + di * es, es);
          di++;
        }
      }
    }
  }
  return out;
}

}  // namespace lakesoul
