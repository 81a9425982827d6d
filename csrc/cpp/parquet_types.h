// Parquet metadata model + thrift (de)serialization — written against the
// apache/parquet-format spec (parquet.thrift), minimal subset for flat
// schemas. This replaces what the reference gets from the arrow-rs
// `parquet` crate.
#pragma once

#include <cstdint>
#include <optional>
#include <string>
#include <vector>

#include "thrift_compact.h"

namespace lakesoul {

// parquet physical types
enum PhysicalType : int32_t {
  PT_BOOLEAN = 0,
  PT_INT32 = 1,
  PT_INT64 = 2,
  PT_INT96 = 3,
  PT_FLOAT = 4,
  PT_DOUBLE = 5,
  PT_BYTE_ARRAY = 6,
  PT_FLBA = 7,
};

enum Encoding : int32_t {
  ENC_PLAIN = 0,
  ENC_PLAIN_DICTIONARY = 2,
  ENC_RLE = 3,
  ENC_BIT_PACKED = 4,
  ENC_DELTA_BINARY_PACKED = 5,
  ENC_DELTA_LENGTH_BYTE_ARRAY = 6,
  ENC_DELTA_BYTE_ARRAY = 7,
  ENC_RLE_DICTIONARY = 8,
  ENC_BYTE_STREAM_SPLIT = 9,
};

enum Codec : int32_t {
  CODEC_UNCOMPRESSED = 0,
  CODEC_SNAPPY = 1,
  CODEC_GZIP = 2,
  CODEC_BROTLI = 4,
  CODEC_LZ4 = 5,
  CODEC_ZSTD = 6,
  CODEC_LZ4_RAW = 7,
};

enum PageType : int32_t {
  PAGE_DATA = 0,
  PAGE_INDEX = 1,
  PAGE_DICTIONARY = 2,
  PAGE_DATA_V2 = 3,
};

enum Repetition : int32_t { REP_REQUIRED = 0, REP_OPTIONAL = 1, REP_REPEATED = 2 };

// ConvertedType values (subset)
enum Converted : int32_t {
  CV_NONE = -1,
  CV_UTF8 = 0,
  CV_DECIMAL = 5,
  CV_DATE = 6,
  CV_TIMESTAMP_MILLIS = 9,
  CV_TIMESTAMP_MICROS = 10,
  CV_INT_8 = 15,
  CV_INT_16 = 16,
  CV_INT_32 = 17,
  CV_INT_64 = 18,
};

// logical type tags we emit/understand
enum class LogicalTag {
  NONE,
  STRING,
  DATE,
  TIMESTAMP_MILLIS,
  TIMESTAMP_MICROS,
  TIMESTAMP_NANOS,
  INT,       // with bit_width / signed
  FLOAT16,
  DECIMAL,   // with dec_precision / dec_scale
};

struct SchemaElement {
  std::string name;
  int32_t type = -1;  // physical; -1 for group
  int32_t type_length = 0;
  int32_t repetition = REP_OPTIONAL;
  int32_t num_children = 0;
  int32_t converted = CV_NONE;
  LogicalTag logical = LogicalTag::NONE;
  int32_t int_bit_width = 0;
  bool int_signed = true;
  bool ts_utc = true;
  int32_t dec_precision = 0;
  int32_t dec_scale = 0;
};

struct Statistics {
  std::string min_value, max_value;
  int64_t null_count = -1;
  bool has_min_max = false;
};

struct ColumnMeta {
  int32_t type = 0;
  std::vector<int32_t> encodings;
  std::vector<std::string> path_in_schema;
  int32_t codec = CODEC_UNCOMPRESSED;
  int64_t num_values = 0;
  int64_t total_uncompressed_size = 0;
  int64_t total_compressed_size = 0;
  int64_t data_page_offset = 0;
  int64_t dictionary_page_offset = -1;
  Statistics stats;
};

struct RowGroup {
  std::vector<ColumnMeta> columns;
  int64_t total_byte_size = 0;
  int64_t num_rows = 0;
};

struct FileMetaData {
  int32_t version = 2;
  std::vector<SchemaElement> schema;  // flattened, root first
  int64_t num_rows = 0;
  std::vector<RowGroup> row_groups;
  std::string created_by;
};

struct PageHeader {
  int32_t type = PAGE_DATA;
  int32_t uncompressed_size = 0;
  int32_t compressed_size = 0;
  // v1 data page
  int32_t num_values = 0;
  int32_t encoding = ENC_PLAIN;
  int32_t def_encoding = ENC_RLE;
  int32_t rep_encoding = ENC_RLE;
  // v2 data page
  int32_t num_nulls = 0;
  int32_t num_rows = 0;
  int32_t def_levels_byte_length = 0;
  int32_t rep_levels_byte_length = 0;
  bool v2_is_compressed = true;
  // dictionary page
  int32_t dict_num_values = 0;
};

// ---------------------------------------------------------------------- //
// parsing
// ---------------------------------------------------------------------- //

inline Statistics parse_statistics(ThriftReader& r) {
  Statistics s;
  int16_t fid = 0;
  CType t;
  std::string min_dep, max_dep;
  bool has_min_dep = false, has_max_dep = false, has_min = false, has_max = false;
  while (r.read_field_header(fid, t)) {
    switch (fid) {
      case 1: max_dep = r.read_binary(); has_max_dep = true; break;
      case 2: min_dep = r.read_binary(); has_min_dep = true; break;
      case 3: s.null_count = r.read_zigzag(); break;
      case 4: r.skip(t); break;  // distinct_count
      case 5: s.max_value = r.read_binary(); has_max = true; break;
      case 6: s.min_value = r.read_binary(); has_min = true; break;
      default: r.skip(t);
    }
  }
  if (has_min && has_max) {
    s.has_min_max = true;
  } else if (has_min_dep && has_max_dep) {
    s.min_value = min_dep;
    s.max_value = max_dep;
    s.has_min_max = true;
  }
  return s;
}

inline SchemaElement parse_schema_element(ThriftReader& r) {
  SchemaElement e;
  int16_t fid = 0;
  CType t;
  bool has_type = false;
  while (r.read_field_header(fid, t)) {
    switch (fid) {
      case 1: e.type = (int32_t)r.read_zigzag(); has_type = true; break;
      case 2: e.type_length = (int32_t)r.read_zigzag(); break;
      case 3: e.repetition = (int32_t)r.read_zigzag(); break;
      case 4: e.name = r.read_binary(); break;
      case 5: e.num_children = (int32_t)r.read_zigzag(); break;
      case 6: e.converted = (int32_t)r.read_zigzag(); break;
      case 7: e.dec_scale = (int32_t)r.read_zigzag(); break;
      case 8: e.dec_precision = (int32_t)r.read_zigzag(); break;
      case 10: {  // LogicalType union
        int16_t f2 = 0;
        CType t2;
        while (r.read_field_header(f2, t2)) {
          switch (f2) {
            case 1: e.logical = LogicalTag::STRING; r.skip(t2); break;
            case 5: {  // DecimalType { 1: i32 scale, 2: i32 precision }
              e.logical = LogicalTag::DECIMAL;
              int16_t f3 = 0;
              CType t3;
              while (r.read_field_header(f3, t3)) {
                if (f3 == 1) e.dec_scale = (int32_t)r.read_zigzag();
                else if (f3 == 2) e.dec_precision = (int32_t)r.read_zigzag();
                else r.skip(t3);
              }
              break;
            }
            case 6: e.logical = LogicalTag::DATE; r.skip(t2); break;
            case 8: {  // TIMESTAMP
              int16_t f3 = 0;
              CType t3;
              while (r.read_field_header(f3, t3)) {
                if (f3 == 1) {
                  e.ts_utc = (t3 == CType::BOOL_TRUE);
                } else if (f3 == 2) {  // TimeUnit union
                  int16_t f4 = 0;
                  CType t4;
                  while (r.read_field_header(f4, t4)) {
                    if (f4 == 1) e.logical = LogicalTag::TIMESTAMP_MILLIS;
                    else if (f4 == 2) e.logical = LogicalTag::TIMESTAMP_MICROS;
                    else if (f4 == 3) e.logical = LogicalTag::TIMESTAMP_NANOS;
                    r.skip(t4);
                  }
                } else {
                  r.skip(t3);
                }
              }
              break;
            }
            case 10: {  // IntType { 1: i8 bitWidth, 2: bool isSigned }
              e.logical = LogicalTag::INT;
              int16_t f3 = 0;
              CType t3;
              while (r.read_field_header(f3, t3)) {
                if (f3 == 1 && t3 == CType::BYTE) {
                  e.int_bit_width = (int32_t)(int8_t)r.read_byte();
                } else if (f3 == 2) {
                  e.int_signed = (t3 == CType::BOOL_TRUE);
                } else {
                  r.skip(t3);
                }
              }
              break;
            }
            case 15: e.logical = LogicalTag::FLOAT16; r.skip(t2); break;
            default: r.skip(t2);
          }
        }
        break;
      }
      default: r.skip(t);
    }
  }
  (void)has_type;
  return e;
}

inline ColumnMeta parse_column_meta(ThriftReader& r) {
  ColumnMeta m;
  int16_t fid = 0;
  CType t;
  while (r.read_field_header(fid, t)) {
    switch (fid) {
      case 1: m.type = (int32_t)r.read_zigzag(); break;
      case 2: {
        CType elem; uint32_t n;
        r.read_list_header(elem, n);
        for (uint32_t i = 0; i < n; i++) m.encodings.push_back((int32_t)r.read_zigzag());
        break;
      }
      case 3: {
        CType elem; uint32_t n;
        r.read_list_header(elem, n);
        for (uint32_t i = 0; i < n; i++) m.path_in_schema.push_back(r.read_binary());
        break;
      }
      case 4: m.codec = (int32_t)r.read_zigzag(); break;
      case 5: m.num_values = r.read_zigzag(); break;
      case 6: m.total_uncompressed_size = r.read_zigzag(); break;
      case 7: m.total_compressed_size = r.read_zigzag(); break;
      case 9: m.data_page_offset = r.read_zigzag(); break;
      case 11: m.dictionary_page_offset = r.read_zigzag(); break;
      case 12: m.stats = parse_statistics(r); break;
      default: r.skip(t);
    }
  }
  return m;
}

inline RowGroup parse_row_group(ThriftReader& r) {
  RowGroup g;
  int16_t fid = 0;
  CType t;
  while (r.read_field_header(fid, t)) {
    switch (fid) {
      case 1: {
        CType elem; uint32_t n;
        r.read_list_header(elem, n);
        for (uint32_t i = 0; i < n; i++) {
          // ColumnChunk struct
          ColumnMeta m;
          int16_t f2 = 0;
          CType t2;
          while (r.read_field_header(f2, t2)) {
            if (f2 == 3) {
              m = parse_column_meta(r);
            } else {
              r.skip(t2);
            }
          }
          g.columns.push_back(std::move(m));
        }
        break;
      }
      case 2: g.total_byte_size = r.read_zigzag(); break;
      case 3: g.num_rows = r.read_zigzag(); break;
      default: r.skip(t);
    }
  }
  return g;
}

inline FileMetaData parse_file_meta(const uint8_t* data, size_t len) {
  ThriftReader r(data, len);
  FileMetaData fm;
  int16_t fid = 0;
  CType t;
  while (r.read_field_header(fid, t)) {
    switch (fid) {
      case 1: fm.version = (int32_t)r.read_zigzag(); break;
      case 2: {
        CType elem; uint32_t n;
        r.read_list_header(elem, n);
        for (uint32_t i = 0; i < n; i++) fm.schema.push_back(parse_schema_element(r));
        break;
      }
      case 3: fm.num_rows = r.read_zigzag(); break;
      case 4: {
        CType elem; uint32_t n;
        r.read_list_header(elem, n);
        for (uint32_t i = 0; i < n; i++) fm.row_groups.push_back(parse_row_group(r));
        break;
      }
      case 6: fm.created_by = r.read_binary(); break;
      default: r.skip(t);
    }
  }
  return fm;
}

inline PageHeader parse_page_header(ThriftReader& r) {
  PageHeader h;
  int16_t fid = 0;
  CType t;
  while (r.read_field_header(fid, t)) {
    switch (fid) {
      case 1: h.type = (int32_t)r.read_zigzag(); break;
      case 2: h.uncompressed_size = (int32_t)r.read_zigzag(); break;
      case 3: h.compressed_size = (int32_t)r.read_zigzag(); break;
      case 5: {  // DataPageHeader
        int16_t f2 = 0;
        CType t2;
        while (r.read_field_header(f2, t2)) {
          switch (f2) {
            case 1: h.num_values = (int32_t)r.read_zigzag(); break;
            case 2: h.encoding = (int32_t)r.read_zigzag(); break;
            case 3: h.def_encoding = (int32_t)r.read_zigzag(); break;
            case 4: h.rep_encoding = (int32_t)r.read_zigzag(); break;
            default: r.skip(t2);
          }
        }
        break;
      }
      case 7: {  // DictionaryPageHeader
        int16_t f2 = 0;
        CType t2;
        while (r.read_field_header(f2, t2)) {
          switch (f2) {
            case 1: h.dict_num_values = (int32_t)r.read_zigzag(); break;
            case 2: h.encoding = (int32_t)r.read_zigzag(); break;
            default: r.skip(t2);
          }
        }
        break;
      }
      case 8: {  // DataPageHeaderV2
        h.type = PAGE_DATA_V2;
        int16_t f2 = 0;
        CType t2;
        while (r.read_field_header(f2, t2)) {
          switch (f2) {
            case 1: h.num_values = (int32_t)r.read_zigzag(); break;
            case 2: h.num_nulls = (int32_t)r.read_zigzag(); break;
            case 3: h.num_rows = (int32_t)r.read_zigzag(); break;
            case 4: h.encoding = (int32_t)r.read_zigzag(); break;
            case 5: h.def_levels_byte_length = (int32_t)r.read_zigzag(); break;
            case 6: h.rep_levels_byte_length = (int32_t)r.read_zigzag(); break;
            case 7: h.v2_is_compressed = (t2 == CType::BOOL_TRUE); break;
            default: r.skip(t2);
          }
        }
        break;
      }
      default: r.skip(t);
    }
  }
  return h;
}

// ---------------------------------------------------------------------- //
// serialization (writer side)
// ---------------------------------------------------------------------- //

inline void write_statistics(ThriftWriter& w, const Statistics& s) {
  int16_t last = 0;
  if (s.null_count >= 0) w.field_i64(last, 3, s.null_count);
  if (s.has_min_max) {
    w.field_binary(last, 5, s.max_value);
    w.field_binary(last, 6, s.min_value);
  }
  w.stop();
}

inline void write_schema_element(ThriftWriter& w, const SchemaElement& e) {
  int16_t last = 0;
  if (e.type >= 0) w.field_i32(last, 1, e.type);
  if (e.type == PT_FLBA) w.field_i32(last, 2, e.type_length);
  w.field_i32(last, 3, e.repetition);
  w.field_binary(last, 4, e.name);
  if (e.num_children > 0) w.field_i32(last, 5, e.num_children);
  if (e.converted != CV_NONE) w.field_i32(last, 6, e.converted);
  if (e.converted == CV_DECIMAL || e.logical == LogicalTag::DECIMAL) {
    w.field_i32(last, 7, e.dec_scale);
    w.field_i32(last, 8, e.dec_precision);
  }
  if (e.logical != LogicalTag::NONE) {
    w.field(last, 10, CType::STRUCT);
    int16_t l2 = 0;
    switch (e.logical) {
      case LogicalTag::STRING:
        w.field(l2, 1, CType::STRUCT);
        w.stop();
        break;
      case LogicalTag::DATE:
        w.field(l2, 6, CType::STRUCT);
        w.stop();
        break;
      case LogicalTag::TIMESTAMP_MILLIS:
      case LogicalTag::TIMESTAMP_MICROS:
      case LogicalTag::TIMESTAMP_NANOS: {
        w.field(l2, 8, CType::STRUCT);
        int16_t l3 = 0;
        w.field_bool(l3, 1, e.ts_utc);
        w.field(l3, 2, CType::STRUCT);
        int16_t l4 = 0;
        int16_t unit_fid =
            e.logical == LogicalTag::TIMESTAMP_MILLIS ? 1
            : e.logical == LogicalTag::TIMESTAMP_MICROS ? 2 : 3;
        w.field(l4, unit_fid, CType::STRUCT);
        w.stop();  // empty unit struct
        w.stop();  // TimeUnit union
        w.stop();  // TimestampType
        break;
      }
      case LogicalTag::INT: {
        w.field(l2, 10, CType::STRUCT);
        int16_t l3 = 0;
        w.field(l3, 1, CType::BYTE);
        w.buf.push_back((uint8_t)e.int_bit_width);
        w.field_bool(l3, 2, e.int_signed);
        w.stop();
        break;
      }
      case LogicalTag::FLOAT16:
        w.field(l2, 15, CType::STRUCT);
        w.stop();
        break;
      case LogicalTag::DECIMAL: {
        w.field(l2, 5, CType::STRUCT);
        int16_t l3 = 0;
        w.field_i32(l3, 1, e.dec_scale);
        w.field_i32(l3, 2, e.dec_precision);
        w.stop();
        break;
      }
      default:
        break;
    }
    w.stop();  // LogicalType union
  }
  w.stop();
}

inline void write_column_meta(ThriftWriter& w, const ColumnMeta& m) {
  int16_t last = 0;
  w.field_i32(last, 1, m.type);
  w.field(last, 2, CType::LIST);
  w.list_header(CType::I32, (uint32_t)m.encodings.size());
  for (auto e : m.encodings) w.write_zigzag(e);
  w.field(last, 3, CType::LIST);
  w.list_header(CType::BINARY, (uint32_t)m.path_in_schema.size());
  for (auto& p : m.path_in_schema) {
    w.write_varint(p.size());
    w.buf.insert(w.buf.end(), p.begin(), p.end());
  }
  w.field_i32(last, 4, m.codec);
  w.field_i64(last, 5, m.num_values);
  w.field_i64(last, 6, m.total_uncompressed_size);
  w.field_i64(last, 7, m.total_compressed_size);
  w.field_i64(last, 9, m.data_page_offset);
  if (m.dictionary_page_offset >= 0) w.field_i64(last, 11, m.dictionary_page_offset);
  if (m.stats.has_min_max || m.stats.null_count >= 0) {
    w.field(last, 12, CType::STRUCT);
    write_statistics(w, m.stats);
  }
  w.stop();
}

inline void write_row_group(ThriftWriter& w, const RowGroup& g, int64_t file_offset_base) {
  int16_t last = 0;
  w.field(last, 1, CType::LIST);
  w.list_header(CType::STRUCT, (uint32_t)g.columns.size());
  for (auto& c : g.columns) {
    // ColumnChunk
    int16_t l2 = 0;
    // field 2: file_offset (i64) — deprecated but required by some readers
    int64_t off = c.dictionary_page_offset >= 0 ? c.dictionary_page_offset : c.data_page_offset;
    {
      ThriftWriter tmp;  // write fields in order 2 then 3
      (void)tmp;
    }
    w.field_i64(l2, 2, off);
    w.field(l2, 3, CType::STRUCT);
    write_column_meta(w, c);
    w.stop();
  }
  w.field_i64(last, 2, g.total_byte_size);
  w.field_i64(last, 3, g.num_rows);
  w.stop();
  (void)file_offset_base;
}

inline std::vector<uint8_t> serialize_file_meta(const FileMetaData& fm) {
  ThriftWriter w;
  int16_t last = 0;
  w.field_i32(last, 1, fm.version);
  w.field(last, 2, CType::LIST);
  w.list_header(CType::STRUCT, (uint32_t)fm.schema.size());
  for (auto& e : fm.schema) write_schema_element(w, e);
  w.field_i64(last, 3, fm.num_rows);
  w.field(last, 4, CType::LIST);
  w.list_header(CType::STRUCT, (uint32_t)fm.row_groups.size());
  for (auto& g : fm.row_groups) write_row_group(w, g, 0);
  w.field_binary(last, 6, fm.created_by);
  // column_orders: TYPE_ORDER for every leaf — required for readers to
  // trust min_value/max_value statistics (parquet-format spec).
  size_t nleaf = 0;
  for (size_t i = 1; i < fm.schema.size(); i++)
    if (fm.schema[i].num_children == 0) nleaf++;
  w.field(last, 7, CType::LIST);
  w.list_header(CType::STRUCT, (uint32_t)nleaf);
  for (size_t i = 0; i < nleaf; i++) {
    int16_t l2 = 0;
    w.field(l2, 1, CType::STRUCT);  // TYPE_ORDER (empty struct)
    w.stop();
    w.stop();
  }
  w.stop();
  return std::move(w.buf);
}

inline std::vector<uint8_t> serialize_page_header(const PageHeader& h) {
  ThriftWriter w;
  int16_t last = 0;
  w.field_i32(last, 1, h.type);
  w.field_i32(last, 2, h.uncompressed_size);
  w.field_i32(last, 3, h.compressed_size);
  if (h.type == PAGE_DATA) {
    w.field(last, 5, CType::STRUCT);
    int16_t l2 = 0;
    w.field_i32(l2, 1, h.num_values);
    w.field_i32(l2, 2, h.encoding);
    w.field_i32(l2, 3, h.def_encoding);
    w.field_i32(l2, 4, h.rep_encoding);
    w.stop();
  } else if (h.type == PAGE_DICTIONARY) {
    w.field(last, 7, CType::STRUCT);
    int16_t l2 = 0;
    w.field_i32(l2, 1, h.dict_num_values);
    w.field_i32(l2, 2, h.encoding);
    w.stop();
  } else if (h.type == PAGE_DATA_V2) {
    // DataPageHeaderV2 { 1:num_values 2:num_nulls 3:num_rows 4:encoding
    //   5:def_levels_byte_length 6:rep_levels_byte_length 7:is_compressed }
    w.field(last, 8, CType::STRUCT);
    int16_t l2 = 0;
    w.field_i32(l2, 1, h.num_values);
    w.field_i32(l2, 2, h.num_nulls);
    w.field_i32(l2, 3, h.num_rows);
    w.field_i32(l2, 4, h.encoding);
    w.field_i32(l2, 5, h.def_levels_byte_length);
    w.field_i32(l2, 6, h.rep_levels_byte_length);
    w.field_bool(l2, 7, h.v2_is_compressed);
    w.stop();
  }
  w.stop();
  return std::move(w.buf);
}

}  // namespace lakesoul
