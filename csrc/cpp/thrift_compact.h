// Thrift compact-protocol reader/writer — minimal subset for Parquet
// metadata. Written from the Thrift compact protocol spec; no generated
// code. (Reference counterpart: the arrow-rs parquet crate's thrift
// module, used by rust/lakesoul-io via its parquet dependency.)
#pragma once

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace lakesoul {

// compact protocol wire types
enum class CType : uint8_t {
  STOP = 0,
  BOOL_TRUE = 1,
  BOOL_FALSE = 2,
  BYTE = 3,
  I16 = 4,
  I32 = 5,
  I64 = 6,
  DOUBLE = 7,
  BINARY = 8,
  LIST = 9,
  SET = 10,
  MAP = 11,
  STRUCT = 12,
};

class ThriftReader {
 public:
  ThriftReader(const uint8_t* data, size_t len) : p_(data), end_(data + len) {}

  size_t remaining() const { return end_ - p_; }
  size_t consumed(const uint8_t* start) const { return p_ - start; }
  const uint8_t* pos() const { return p_; }

  uint64_t read_varint() {
    uint64_t v = 0;
    int shift = 0;
    while (true) {
      if (p_ >= end_) throw std::runtime_error("thrift: varint overrun");
      uint8_t b = *p_++;
      v |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
      if (shift > 63) throw std::runtime_error("thrift: varint too long");
    }
    return v;
  }

  int64_t read_zigzag() {
    uint64_t v = read_varint();
    return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
  }

  uint8_t read_byte() {
    if (p_ >= end_) throw std::runtime_error("thrift: byte overrun");
    return *p_++;
  }

  double read_double() {
    if (p_ + 8 > end_) throw std::runtime_error("thrift: double overrun");
    double d;
    std::memcpy(&d, p_, 8);  // compact protocol: little-endian
    p_ += 8;
    return d;
  }

  std::string read_binary() {
    uint64_t n = read_varint();
    if (p_ + n > end_) throw std::runtime_error("thrift: binary overrun");
    std::string s((const char*)p_, n);
    p_ += n;
    return s;
  }

  // Struct field header. Returns false at STOP. fid updated in place.
  bool read_field_header(int16_t& fid, CType& type) {
    if (p_ >= end_) throw std::runtime_error("thrift: field header overrun");
    uint8_t b = *p_++;
    if (b == 0) return false;
    uint8_t delta = b >> 4;
    type = (CType)(b & 0x0F);
    if (delta == 0) {
      fid = (int16_t)read_zigzag();
    } else {
      fid = (int16_t)(fid + delta);
    }
    return true;
  }

  void read_list_header(CType& elem, uint32_t& size) {
    if (p_ >= end_) throw std::runtime_error("thrift: list header overrun");
    uint8_t b = *p_++;
    elem = (CType)(b & 0x0F);
    uint32_t s = b >> 4;
    if (s == 15) s = (uint32_t)read_varint();
    size = s;
  }

  void skip(CType type) {
    switch (type) {
      case CType::BOOL_TRUE:
      case CType::BOOL_FALSE:
        return;  // value embedded in field header
      case CType::BYTE:
        p_ += 1;
        return;
      case CType::I16:
      case CType::I32:
      case CType::I64:
        read_zigzag();
        return;
      case CType::DOUBLE:
        p_ += 8;
        return;
      case CType::BINARY: {
        uint64_t n = read_varint();
        p_ += n;
        return;
      }
      case CType::LIST:
      case CType::SET: {
        CType elem;
        uint32_t size;
        read_list_header(elem, size);
        for (uint32_t i = 0; i < size; i++) skip(elem);
        return;
      }
      case CType::MAP: {
        uint64_t size = read_varint();
        if (size > 0) {
          if (p_ >= end_) throw std::runtime_error("thrift: map overrun");
          uint8_t kv = *p_++;
          CType kt = (CType)(kv >> 4), vt = (CType)(kv & 0x0F);
          for (uint64_t i = 0; i < size; i++) {
            skip(kt);
            skip(vt);
          }
        }
        return;
      }
      case CType::STRUCT: {
        int16_t fid = 0;
        CType t;
        while (read_field_header(fid, t)) skip(t);
        return;
      }
      default:
        throw std::runtime_error("thrift: cannot skip type");
    }
  }

 private:
  const uint8_t* p_;
  const uint8_t* end_;
};

class ThriftWriter {
 public:
  std::vector<uint8_t> buf;

  void write_varint(uint64_t v) {
    while (v >= 0x80) {
      buf.push_back((uint8_t)(v | 0x80));
      v >>= 7;
    }
    buf.push_back((uint8_t)v);
  }

  void write_zigzag(int64_t v) {
    write_varint(((uint64_t)v << 1) ^ (uint64_t)(v >> 63));
  }

  void field(int16_t& last_fid, int16_t fid, CType type) {
    int delta = fid - last_fid;
    if (delta >= 1 && delta <= 15) {
      buf.push_back((uint8_t)((delta << 4) | (int)type));
    } else {
      buf.push_back((uint8_t)type);
      write_zigzag(fid);
    }
    last_fid = fid;
  }

  void field_i32(int16_t& last, int16_t fid, int32_t v) {
    field(last, fid, CType::I32);
    write_zigzag(v);
  }
  void field_i64(int16_t& last, int16_t fid, int64_t v) {
    field(last, fid, CType::I64);
    write_zigzag(v);
  }
  void field_bool(int16_t& last, int16_t fid, bool v) {
    field(last, fid, v ? CType::BOOL_TRUE : CType::BOOL_FALSE);
  }
  void field_binary(int16_t& last, int16_t fid, const std::string& s) {
    field(last, fid, CType::BINARY);
    write_varint(s.size());
    buf.insert(buf.end(), s.begin(), s.end());
  }
  void list_header(CType elem, uint32_t size) {
    if (size < 15) {
      buf.push_back((uint8_t)((size << 4) | (int)elem));
    } else {
      buf.push_back((uint8_t)(0xF0 | (int)elem));
      write_varint(size);
    }
  }
  void stop() { buf.push_back(0); }
};

}  // namespace lakesoul
