// lakesoul_amd._cpp — host-side native core: parquet read/write, spark
// murmur3, CPU decode fallbacks. GPU kernels live in lakesoul_amd._hip.
#include <torch/extension.h>

#include <atomic>
#include <cstring>
#include <memory>
#include <map>
#include <mutex>
#include <thread>
#include <unordered_map>
#include <chrono>

#include "murmur3.h"
#include "parquet_file.h"
#include "read_unit.h"
#include "zstd_dec.h"

namespace py = pybind11;
using namespace lakesoul;

// ---------------------------------------------------------------------- //
// dtype mapping: canonical dtype string <-> parquet physical/logical
// ---------------------------------------------------------------------- //

struct DtypeInfo {
  int32_t physical;
  int32_t converted;
  LogicalTag logical;
  int32_t bit_width;
  int32_t dec_precision = 0;
  int32_t dec_scale = 0;
};

static DtypeInfo dtype_info(const std::string& dt) {
  if (dt == "bool") return {PT_BOOLEAN, CV_NONE, LogicalTag::NONE, 0};
  if (dt == "int8") return {PT_INT32, CV_INT_8, LogicalTag::INT, 8};
  if (dt == "int16") return {PT_INT32, CV_INT_16, LogicalTag::INT, 16};
  if (dt == "int32") return {PT_INT32, CV_NONE, LogicalTag::NONE, 0};
  if (dt == "int64") return {PT_INT64, CV_NONE, LogicalTag::NONE, 0};
  if (dt == "float32") return {PT_FLOAT, CV_NONE, LogicalTag::NONE, 0};
  if (dt == "float64") return {PT_DOUBLE, CV_NONE, LogicalTag::NONE, 0};
  if (dt == "string") return {PT_BYTE_ARRAY, CV_UTF8, LogicalTag::STRING, 0};
  if (dt == "binary") return {PT_BYTE_ARRAY, CV_NONE, LogicalTag::NONE, 0};
  if (dt == "date32") return {PT_INT32, CV_DATE, LogicalTag::DATE, 0};
  if (dt == "timestamp[ms]")
    return {PT_INT64, CV_TIMESTAMP_MILLIS, LogicalTag::TIMESTAMP_MILLIS, 0};
  if (dt == "timestamp[us]")
    return {PT_INT64, CV_TIMESTAMP_MICROS, LogicalTag::TIMESTAMP_MICROS, 0};
  if (dt == "timestamp[ns]")
    return {PT_INT64, CV_NONE, LogicalTag::TIMESTAMP_NANOS, 0};
  if (dt.rfind("decimal(", 0) == 0 && dt.back() == ')') {
    // decimal(p,s), p<=18: stored as INT64 unscaled (parquet spec allows
    // INT32/INT64 physical for small precisions; arrow reads it with
    // store_decimal_as_integer parity)
    int p = 0, sc = 0;
    if (sscanf(dt.c_str(), "decimal(%d,%d)", &p, &sc) != 2 || p <= 0 || p > 18)
      throw std::runtime_error("unsupported decimal dtype: " + dt);
    DtypeInfo di{PT_INT64, CV_DECIMAL, LogicalTag::DECIMAL, 0};
    di.dec_precision = p;
    di.dec_scale = sc;
    return di;
  }
  throw std::runtime_error("unsupported dtype: " + dt);
}

static std::string dtype_name_scalar(const ColumnDesc& c) {
  switch (c.physical) {
    case PT_BOOLEAN:
      return "bool";
    case PT_INT32:
      if (c.converted == CV_INT_8 || (c.logical == LogicalTag::INT && c.int_bit_width == 8))
        return "int8";
      if (c.converted == CV_INT_16 || (c.logical == LogicalTag::INT && c.int_bit_width == 16))
        return "int16";
      if (c.converted == CV_DATE || c.logical == LogicalTag::DATE) return "date32";
      return "int32";
    case PT_INT64:
      if (c.converted == CV_DECIMAL || c.logical == LogicalTag::DECIMAL)
        return "decimal(" + std::to_string(c.dec_precision) + "," +
               std::to_string(c.dec_scale) + ")";
      if (c.converted == CV_TIMESTAMP_MILLIS || c.logical == LogicalTag::TIMESTAMP_MILLIS)
        return "timestamp[ms]";
      if (c.converted == CV_TIMESTAMP_MICROS || c.logical == LogicalTag::TIMESTAMP_MICROS)
        return "timestamp[us]";
      if (c.logical == LogicalTag::TIMESTAMP_NANOS) return "timestamp[ns]";
      return "int64";
    case PT_FLOAT:
      return "float32";
    case PT_DOUBLE:
      return "float64";
    case PT_BYTE_ARRAY:
      if (c.converted == CV_UTF8 || c.logical == LogicalTag::STRING) return "string";
      return "binary";
    case PT_FLBA:
      return "binary";   // fixed-len values surface as binary cells
    case PT_INT96:
      return "timestamp[ns]";
    default:
      return "unsupported";
  }
}

// ---------------------------------------------------------------------- //
// writer binding
// ---------------------------------------------------------------------- //

static std::string dtype_name(const ColumnDesc& c) {
  if (c.is_list) return "list<" + dtype_name_scalar(c) + ">";
  return dtype_name_scalar(c);
}

static int64_t write_parquet(
    const std::string& path, const std::vector<std::string>& names,
    const std::vector<std::string>& dtypes, const std::vector<torch::Tensor>& columns,
    const std::vector<c10::optional<torch::Tensor>>& offsets,
    const std::vector<c10::optional<torch::Tensor>>& validity,
    const std::vector<bool>& nullable, int64_t row_group_size, int64_t codec,
    int64_t level,
    const std::vector<c10::optional<torch::Tensor>>& elem_offsets = {},
    const std::vector<std::string>& parents = {}) {
  size_t ncol = names.size();
  TORCH_CHECK(dtypes.size() == ncol && columns.size() == ncol);
  std::vector<ColumnDesc> descs(ncol);
  std::vector<ColumnData> data(ncol);
  int64_t num_rows = -1;
  std::vector<torch::Tensor> keep;  // hold contiguous refs

  for (size_t i = 0; i < ncol; i++) {
    bool is_list = dtypes[i].rfind("list<", 0) == 0;
    std::string scalar_dtype = is_list
        ? dtypes[i].substr(5, dtypes[i].size() - 6)
        : dtypes[i];
    DtypeInfo di = dtype_info(scalar_dtype);
    descs[i].name = names[i];
    descs[i].physical = di.physical;
    descs[i].converted = di.converted;
    descs[i].logical = di.logical;
    descs[i].int_bit_width = di.bit_width;
    descs[i].dec_precision = di.dec_precision;
    descs[i].dec_scale = di.dec_scale;
    descs[i].nullable = nullable[i];
    descs[i].is_list = is_list;
    if (is_list) {
      descs[i].max_def = 2;  // optional list group + repeated, required elem
    }
    if (i < parents.size() && !parents[i].empty()) {
      // "struct:NAME" / "map:NAME" — the enclosing group
      const std::string& p = parents[i];
      auto colon = p.find(':');
      TORCH_CHECK(colon != std::string::npos, "bad parent spec ", p);
      descs[i].parent = p.substr(colon + 1);
      descs[i].parent_kind = p.rfind("map", 0) == 0 ? 2 : 1;
    }

    torch::Tensor col = columns[i].contiguous().cpu();
    keep.push_back(col);
    if (is_list) {
      TORCH_CHECK(offsets[i].has_value(), "list column needs offsets");
      torch::Tensor off = offsets[i]->contiguous().cpu().to(torch::kInt64);
      keep.push_back(off);
      data[i].list_offsets = off.data_ptr<int64_t>();
      if (di.physical == PT_BYTE_ARRAY) {
        // list<string>: col = payload bytes, elem_offsets = element
        // byte offsets (int32)
        TORCH_CHECK(i < elem_offsets.size() && elem_offsets[i].has_value(),
                    "list<string> column needs elem_offsets");
        torch::Tensor eo = elem_offsets[i]->contiguous().cpu().to(torch::kInt32);
        keep.push_back(eo);
        data[i].offsets = eo.data_ptr<int32_t>();
        data[i].bytes = (const uint8_t*)col.data_ptr();
      } else {
        data[i].data = (const uint8_t*)col.data_ptr();
        int es = physical_elem_size(di.physical);
        TORCH_CHECK(col.element_size() == es, "list element size mismatch");
      }
      int64_t n = off.numel() - 1;
      TORCH_CHECK(num_rows < 0 || n == num_rows, "row count mismatch");
      num_rows = n;
    } else if (di.physical == PT_BYTE_ARRAY) {
      TORCH_CHECK(offsets[i].has_value(), "byte_array column needs offsets");
      torch::Tensor off = offsets[i]->contiguous().cpu().to(torch::kInt32);
      keep.push_back(off);
      data[i].offsets = off.data_ptr<int32_t>();
      data[i].bytes = (const uint8_t*)col.data_ptr();
      int64_t n = off.numel() - 1;
      TORCH_CHECK(num_rows < 0 || n == num_rows, "row count mismatch");
      num_rows = n;
    } else {
      // dtype check: int8/int16 must be passed widened to int32
      data[i].data = (const uint8_t*)col.data_ptr();
      int64_t n = col.numel();
      TORCH_CHECK(num_rows < 0 || n == num_rows, "row count mismatch");
      num_rows = n;
      int es = physical_elem_size(di.physical);
      TORCH_CHECK(col.element_size() == es, "column ", names[i],
                  " element size ", col.element_size(), " != physical ", es);
    }
    if (validity[i].has_value()) {
      torch::Tensor v = validity[i]->contiguous().cpu().to(torch::kUInt8);
      keep.push_back(v);
      data[i].validity = v.data_ptr<uint8_t>();
    }
  }
  if (num_rows < 0) num_rows = 0;

  int64_t size;
  {
    py::gil_scoped_release rel;
    ParquetWriter w(path, descs, (int)codec, (int)level, row_group_size);
    w.write_row_group(data, num_rows);
    size = w.close();
  }
  return size;
}

// ---------------------------------------------------------------------- //
// incremental writer handles (streaming multipart upload support):
// writer_open -> write batches one by one (each call appends row groups
// and flushes) -> writer_bytes gives the stable part boundary ->
// writer_finish emits the footer. The marshaling mirrors write_parquet.
// ---------------------------------------------------------------------- //

namespace {
struct OpenWriter {
  std::unique_ptr<ParquetWriter> w;
  std::vector<ColumnDesc> descs;
  std::vector<std::string> dtypes;
  std::string path;
};
std::mutex g_wmu;
std::map<int64_t, std::shared_ptr<OpenWriter>> g_writers;
int64_t g_wnext = 1;

std::shared_ptr<OpenWriter> get_writer(int64_t h) {
  std::lock_guard<std::mutex> lk(g_wmu);
  auto it = g_writers.find(h);
  if (it == g_writers.end()) throw std::runtime_error("bad writer handle");
  return it->second;
}
}  // namespace

static int64_t writer_open(const std::string& path,
                           const std::vector<std::string>& names,
                           const std::vector<std::string>& dtypes,
                           const std::vector<bool>& nullable,
                           int64_t row_group_size, int64_t codec,
                           int64_t level,
                           const std::vector<std::string>& parents = {}) {
  auto ow = std::make_shared<OpenWriter>();
  ow->path = path;
  ow->dtypes = dtypes;
  ow->descs.resize(names.size());
  for (size_t i = 0; i < names.size(); i++) {
    bool is_list = dtypes[i].rfind("list<", 0) == 0;
    std::string sd = is_list ? dtypes[i].substr(5, dtypes[i].size() - 6)
                             : dtypes[i];
    DtypeInfo di = dtype_info(sd);
    ColumnDesc& d = ow->descs[i];
    d.name = names[i];
    d.physical = di.physical;
    d.converted = di.converted;
    d.logical = di.logical;
    d.int_bit_width = di.bit_width;
    d.dec_precision = di.dec_precision;
    d.dec_scale = di.dec_scale;
    d.nullable = nullable[i];
    d.is_list = is_list;
    if (is_list) d.max_def = 2;
    if (i < parents.size() && !parents[i].empty()) {
      const std::string& p = parents[i];
      auto colon = p.find(':');
      TORCH_CHECK(colon != std::string::npos, "bad parent spec ", p);
      d.parent = p.substr(colon + 1);
      d.parent_kind = p.rfind("map", 0) == 0 ? 2 : 1;
    }
  }
  ow->w = std::make_unique<ParquetWriter>(path, ow->descs, (int)codec,
                                          (int)level, row_group_size);
  std::lock_guard<std::mutex> lk(g_wmu);
  int64_t h = g_wnext++;
  g_writers[h] = ow;
  return h;
}

static void writer_write(int64_t h, const std::vector<torch::Tensor>& columns,
                         const std::vector<c10::optional<torch::Tensor>>& offsets,
                         const std::vector<c10::optional<torch::Tensor>>& validity,
                         const std::vector<c10::optional<torch::Tensor>>& elem_offsets = {}) {
  auto ow = get_writer(h);
  size_t ncol = ow->descs.size();
  TORCH_CHECK(columns.size() == ncol);
  std::vector<ColumnData> data(ncol);
  std::vector<torch::Tensor> keep;
  int64_t num_rows = -1;
  for (size_t i = 0; i < ncol; i++) {
    const ColumnDesc& d = ow->descs[i];
    torch::Tensor col = columns[i].contiguous().cpu();
    keep.push_back(col);
    if (d.is_list) {
      TORCH_CHECK(offsets[i].has_value(), "list column needs offsets");
      torch::Tensor off = offsets[i]->contiguous().cpu().to(torch::kInt64);
      keep.push_back(off);
      data[i].list_offsets = off.data_ptr<int64_t>();
      if (d.physical == PT_BYTE_ARRAY) {
        TORCH_CHECK(i < elem_offsets.size() && elem_offsets[i].has_value(),
                    "list<string> column needs elem_offsets");
        torch::Tensor eo = elem_offsets[i]->contiguous().cpu().to(torch::kInt32);
        keep.push_back(eo);
        data[i].offsets = eo.data_ptr<int32_t>();
        data[i].bytes = (const uint8_t*)col.data_ptr();
      } else {
        data[i].data = (const uint8_t*)col.data_ptr();
      }
      int64_t n = off.numel() - 1;
      TORCH_CHECK(num_rows < 0 || n == num_rows);
      num_rows = n;
    } else if (d.physical == PT_BYTE_ARRAY) {
      TORCH_CHECK(offsets[i].has_value(), "byte_array column needs offsets");
      torch::Tensor off = offsets[i]->contiguous().cpu().to(torch::kInt32);
      keep.push_back(off);
      data[i].offsets = off.data_ptr<int32_t>();
      data[i].bytes = (const uint8_t*)col.data_ptr();
      int64_t n = off.numel() - 1;
      TORCH_CHECK(num_rows < 0 || n == num_rows);
      num_rows = n;
    } else {
      data[i].data = (const uint8_t*)col.data_ptr();
      int64_t n = col.numel();
      TORCH_CHECK(num_rows < 0 || n == num_rows);
      num_rows = n;
    }
    if (validity[i].has_value()) {
      torch::Tensor v = validity[i]->contiguous().cpu().to(torch::kUInt8);
      keep.push_back(v);
      data[i].validity = v.data_ptr<uint8_t>();
    }
  }
  if (num_rows < 0) num_rows = 0;
  py::gil_scoped_release rel;
  ow->w->write_row_group(data, num_rows);
  ow->w->flush_os();
}

static int64_t writer_bytes(int64_t h) { return get_writer(h)->w->bytes_written(); }

static int64_t writer_finish(int64_t h) {
  auto ow = get_writer(h);
  int64_t size;
  {
    py::gil_scoped_release rel;
    size = ow->w->close();
  }
  std::lock_guard<std::mutex> lk(g_wmu);
  g_writers.erase(h);
  return size;
}

static void writer_abort(int64_t h) {
  std::shared_ptr<OpenWriter> ow;
  {
    std::lock_guard<std::mutex> lk(g_wmu);
    auto it = g_writers.find(h);
    if (it == g_writers.end()) return;
    ow = it->second;
    g_writers.erase(it);
  }
  ow->w.reset();
  std::remove(ow->path.c_str());
}

// ---------------------------------------------------------------------- //
// reader bindings
// ---------------------------------------------------------------------- //

static std::unordered_map<int64_t, std::shared_ptr<ParquetFile>> g_files;
static std::mutex g_files_mu;
static std::atomic<int64_t> g_next_handle{1};

static int64_t open_parquet(const std::string& path) {
  auto f = std::make_shared<ParquetFile>(path);
  std::lock_guard<std::mutex> lk(g_files_mu);
  int64_t h = g_next_handle++;
  g_files[h] = f;
  return h;
}

static std::shared_ptr<ParquetFile> get_file(int64_t h) {
  std::lock_guard<std::mutex> lk(g_files_mu);
  auto it = g_files.find(h);
  TORCH_CHECK(it != g_files.end(), "bad parquet handle");
  return it->second;
}

static void close_parquet(int64_t h) {
  std::lock_guard<std::mutex> lk(g_files_mu);
  g_files.erase(h);
}

static py::dict parquet_meta(int64_t h) {
  auto f = get_file(h);
  py::dict d;
  d["num_rows"] = f->num_rows();
  d["num_row_groups"] = (int64_t)f->num_row_groups();
  py::list cols;
  for (auto& c : f->columns()) {
    py::dict cd;
    cd["name"] = c.name;
    cd["dtype"] = dtype_name(c);
    cd["nullable"] = c.nullable;
    cols.append(cd);
  }
  d["columns"] = cols;
  py::list rgs;
  for (auto& g : f->meta().row_groups) {
    py::dict gd;
    gd["num_rows"] = g.num_rows;
    py::list cstats;
    for (auto& cm : g.columns) {
      py::dict sd;
      sd["num_values"] = cm.num_values;
      sd["null_count"] = cm.stats.null_count;
      if (cm.stats.has_min_max) {
        sd["min"] = py::bytes(cm.stats.min_value);
        sd["max"] = py::bytes(cm.stats.max_value);
      }
      cstats.append(sd);
    }
    gd["columns"] = cstats;
    rgs.append(gd);
  }
  d["row_groups"] = rgs;
  return d;
}

static torch::Tensor vec_to_tensor_u8(std::vector<uint8_t>&& v) {
  auto t = torch::empty({(int64_t)v.size()}, torch::kUInt8);
  std::memcpy(t.data_ptr(), v.data(), v.size());
  return t;
}

static torch::Tensor vec_to_tensor_i32(std::vector<int32_t>&& v) {
  auto t = torch::empty({(int64_t)v.size()}, torch::kInt32);
  std::memcpy(t.data_ptr(), v.data(), v.size() * 4);
  return t;
}

// raw chunk for GPU decode
static py::dict chunk_to_dict_raw(ParquetFile::ChunkData&& ch) {
  py::dict d;
  d["physical"] = ch.physical;
  d["num_values"] = ch.num_values;
  d["null_count"] = ch.null_count;
  d["is_dict"] = ch.is_dict;
  d["dict_num_values"] = ch.dict_num_values;
  d["values"] = vec_to_tensor_u8(std::move(ch.values));
  d["validity"] = vec_to_tensor_u8(std::move(ch.validity));
  d["dict"] = vec_to_tensor_u8(std::move(ch.dict));
  // idx pages as int64 [n,5]
  auto pt = torch::empty({(int64_t)ch.idx_pages.size(), 5}, torch::kInt64);
  auto pa = pt.accessor<int64_t, 2>();
  for (size_t i = 0; i < ch.idx_pages.size(); i++) {
    pa[i][0] = ch.idx_pages[i].out_off;
    pa[i][1] = ch.idx_pages[i].n;
    pa[i][2] = ch.idx_pages[i].payload_off;
    pa[i][3] = ch.idx_pages[i].payload_len;
    pa[i][4] = ch.idx_pages[i].bit_width;
  }
  d["idx_pages"] = pt;
  return d;
}

static py::dict read_chunk_raw(int64_t h, int64_t rg, int64_t col) {
  auto f = get_file(h);
  ParquetFile::ChunkData ch;
  {
    py::gil_scoped_release rel;
    ch = f->read_chunk((size_t)rg, (size_t)col);
  }
  return chunk_to_dict_raw(std::move(ch));
}

static py::dict decoded_to_dict(DecodedColumn&& dc) {
  py::dict d;
  d["num_values"] = dc.num_values;
  d["data"] = vec_to_tensor_u8(std::move(dc.data));
  d["offsets"] = vec_to_tensor_i32(std::move(dc.offsets));
  d["bytes"] = vec_to_tensor_u8(std::move(dc.bytes));
  d["validity"] = vec_to_tensor_u8(std::move(dc.validity));
  if (dc.is_list) {
    auto lo = torch::empty({(int64_t)dc.list_offsets.size()}, torch::kInt64);
    std::memcpy(lo.data_ptr(), dc.list_offsets.data(),
                dc.list_offsets.size() * 8);
    d["list_offsets"] = lo;
    auto lv = torch::empty({(int64_t)dc.list_validity.size()}, torch::kUInt8);
    if (!dc.list_validity.empty())
      std::memcpy(lv.data_ptr(), dc.list_validity.data(),
                  dc.list_validity.size());
    d["list_validity"] = lv;
  }
  return d;
}

static py::dict read_chunk_cpu(int64_t h, int64_t rg, int64_t col) {
  auto f = get_file(h);
  DecodedColumn dc;
  torch::Tensor list_offs, list_valid;
  bool is_list = false;
  {
    py::gil_scoped_release rel;
    auto ch = f->read_chunk((size_t)rg, (size_t)col);
    if (ch.is_list) {
      is_list = true;
      list_offs = torch::empty({(int64_t)ch.list_offsets.size()}, torch::kInt64);
      std::memcpy(list_offs.data_ptr(), ch.list_offsets.data(),
                  ch.list_offsets.size() * 8);
      list_valid = torch::empty({(int64_t)ch.list_validity.size()}, torch::kUInt8);
      if (!ch.list_validity.empty())
        std::memcpy(list_valid.data_ptr(), ch.list_validity.data(),
                    ch.list_validity.size());
    }
    dc = decode_chunk_cpu(ch);
  }
  py::dict d = decoded_to_dict(std::move(dc));
  if (is_list) {
    d["list_offsets"] = list_offs;
    d["list_validity"] = list_valid;
  }
  return d;
}

// batch: parallel host decode across (rg,col) pairs
static py::list read_chunks_cpu_batch(int64_t h,
                                      const std::vector<std::pair<int64_t, int64_t>>& rc,
                                      int64_t nthreads) {
  auto f = get_file(h);
  std::vector<DecodedColumn> out(rc.size());
  {
    py::gil_scoped_release rel;
    std::string err;
    std::mutex err_mu;
    ThreadPool::instance().parallel_for((int64_t)rc.size(), [&](int64_t i) {
      try {
        auto ch = f->read_chunk((size_t)rc[i].first, (size_t)rc[i].second);
        out[i] = decode_chunk_cpu(ch);
      } catch (std::exception& e) {
        std::lock_guard<std::mutex> lk(err_mu);
        err = e.what();
      }
    });
    if (!err.empty()) throw std::runtime_error(err);
  }
  py::list result;
  for (auto& dc : out) result.append(decoded_to_dict(std::move(dc)));
  return result;
}

static py::list read_chunks_raw_batch(int64_t h,
                                      const std::vector<std::pair<int64_t, int64_t>>& rc,
                                      int64_t nthreads) {
  auto f = get_file(h);
  std::vector<ParquetFile::ChunkData> out(rc.size());
  {
    py::gil_scoped_release rel;
    std::string err;
    std::mutex err_mu;
    ThreadPool::instance().parallel_for((int64_t)rc.size(), [&](int64_t i) {
      try {
        out[i] = f->read_chunk((size_t)rc[i].first, (size_t)rc[i].second);
      } catch (std::exception& e) {
        std::lock_guard<std::mutex> lk(err_mu);
        err = e.what();
      }
    });
    if (!err.empty()) throw std::runtime_error(err);
  }
  py::list result;
  for (auto& ch : out) result.append(chunk_to_dict_raw(std::move(ch)));
  return result;
}

// prep dictionary-index RLE runs for the GPU expansion kernel:
// idx_pages int64 [p,5] = {row_off, n_nonnull, payload_off, payload_len, bw}
// returns (payload padded +8 bytes, runs int64 [m,5] =
//          {dense_out_off, n, is_literal, value_or_abs_bitoff, bit_width},
//          total_dense_n)
static py::tuple prep_rle_runs(torch::Tensor values, torch::Tensor idx_pages) {
  auto v = values.contiguous();
  auto pages = idx_pages.contiguous();
  const uint8_t* base = v.data_ptr<uint8_t>();
  auto pa = pages.accessor<int64_t, 2>();
  std::vector<RleRun> all;
  int64_t dense_off = 0;
  std::vector<int64_t> bws;
  for (int64_t p = 0; p < pages.size(0); p++) {
    int64_t n = pa[p][1], off = pa[p][2], len = pa[p][3];
    int bw = (int)pa[p][4];
    std::vector<RleRun> runs;
    parse_rle_runs(base + off, (size_t)len, bw, n, off * 8, runs);
    for (auto& r : runs) {
      r.out_off += dense_off;
      all.push_back(r);
      bws.push_back(bw);
    }
    dense_off += n;
  }
  auto runs_t = torch::empty({(int64_t)all.size(), 6}, torch::kInt64);
  auto ra = runs_t.accessor<int64_t, 2>();
  for (size_t i = 0; i < all.size(); i++) {
    ra[i][0] = all[i].out_off;
    ra[i][1] = all[i].n;
    ra[i][2] = all[i].is_literal;
    ra[i][3] = all[i].is_literal ? all[i].bit_off : (int64_t)all[i].value;
    ra[i][4] = bws[i];
    ra[i][5] = 0;
  }
  auto padded = torch::zeros({v.numel() + 8}, torch::kUInt8);
  std::memcpy(padded.data_ptr(), base, v.numel());
  return py::make_tuple(padded, runs_t, dense_off);
}

// read all files of a scan unit in one call; big contiguous buffers for
// single-H2D GPU decode (see read_unit.h). ``pin`` requests pinned host
// memory for the big buffers (GPU boxes).
static py::dict read_unit_raw_py(const std::vector<std::string>& paths,
                                 const std::vector<std::string>& names,
                                 int64_t nthreads, bool pin, bool gpu_snappy,
                                 bool gpu_zstd, double gpu_zstd_frac) {
  (void)nthreads;
  std::unique_ptr<UnitStage> st;
  auto t0 = std::chrono::steady_clock::now();
  {
    py::gil_scoped_release rel;
    st = read_unit_stage1(paths, names, gpu_snappy, gpu_zstd, gpu_zstd_frac);
  }
  auto t1 = std::chrono::steady_clock::now();
  UnitStage& ud = *st;
  auto alloc_u8 = [&](int64_t n) {
    auto opts = torch::TensorOptions().dtype(torch::kUInt8);
    if (pin) {
      try {
        return torch::empty({n}, opts.pinned_memory(true));
      } catch (...) {
      }
    }
    return torch::empty({n}, opts);
  };
  torch::Tensor values = alloc_u8(ud.values_size);
  torch::Tensor validity = alloc_u8(ud.validity_size);
  torch::Tensor dicts = alloc_u8(ud.dicts_size);
  torch::Tensor soffs = torch::empty({ud.soffs_size}, torch::kInt64);
  torch::Tensor comp = alloc_u8(ud.comp_size);
  torch::Tensor runs = torch::empty({(int64_t)ud.runs.size()}, torch::kInt64);
  if (!ud.runs.empty())
    std::memcpy(runs.data_ptr(), ud.runs.data(), ud.runs.size() * 8);
  torch::Tensor sjobs = torch::empty({(int64_t)ud.snappy_jobs.size()}, torch::kInt64);
  if (!ud.snappy_jobs.empty())
    std::memcpy(sjobs.data_ptr(), ud.snappy_jobs.data(), ud.snappy_jobs.size() * 8);
  torch::Tensor zjobs = torch::empty({(int64_t)ud.zstd_jobs.size()}, torch::kInt64);
  if (!ud.zstd_jobs.empty())
    std::memcpy(zjobs.data_ptr(), ud.zstd_jobs.data(), ud.zstd_jobs.size() * 8);
  auto t2 = std::chrono::steady_clock::now();
  {
    py::gil_scoped_release rel;
    read_unit_fill(ud, (uint8_t*)values.data_ptr(), (uint8_t*)validity.data_ptr(),
                   (uint8_t*)dicts.data_ptr(), soffs.data_ptr<int64_t>(),
                   ud.comp_size ? (uint8_t*)comp.data_ptr() : nullptr);
  }
  auto t3 = std::chrono::steady_clock::now();
  py::dict d;
  auto us = [](auto a, auto b) {
    return std::chrono::duration_cast<std::chrono::microseconds>(b - a).count();
  };
  d["t_stage1_us"] = us(t0, t1);
  d["t_open_us"] = ud.t_open_us;
  d["t_chunks_us"] = ud.t_chunks_us;
  d["t_layout_us"] = ud.t_layout_us;
  d["t_alloc_us"] = us(t1, t2);
  d["t_fill_us"] = us(t2, t3);
  d["values"] = values;
  d["validity"] = validity;
  d["dicts"] = dicts;
  d["runs"] = runs;
  d["soffs"] = soffs;
  d["comp"] = comp;
  d["snappy_jobs"] = sjobs;
  d["zstd_jobs"] = zjobs;
  py::list frows;
  for (auto r : ud.file_rows) frows.append(r);
  d["file_rows"] = frows;
  // descriptor tensor for the C++ scan driver: [nuc, 17]
  {
    auto dt = torch::empty({(int64_t)ud.cols.size(), 17}, torch::kInt64);
    auto da = dt.accessor<int64_t, 2>();
    for (size_t i = 0; i < ud.cols.size(); i++) {
      const UnitColumn& c = ud.cols[i];
      int es = c.present && !c.is_string ? physical_elem_size(c.physical) : 0;
      da[i][0] = c.present;
      da[i][1] = c.is_string;
      da[i][2] = c.is_dict;
      da[i][3] = es;
      da[i][4] = c.num_values;
      da[i][5] = c.null_count;
      da[i][6] = c.val_off;
      da[i][7] = c.val_len;
      da[i][8] = c.validity_off;
      da[i][9] = c.dict_off;
      da[i][10] = c.dict_len;
      da[i][11] = c.run_off;
      da[i][12] = c.run_cnt;
      da[i][13] = c.dense_n;
      da[i][14] = c.soff_off;
      da[i][15] = c.sbytes_off;
      da[i][16] = c.sbytes_len;
    }
    d["desc"] = dt;
  }
  py::list cols;
  for (auto& c : ud.cols) {
    py::dict cd;
    cd["file_idx"] = c.file_idx;
    cd["name"] = c.name;
    cd["present"] = c.present;
    cd["is_string"] = c.is_string;
    cd["is_list"] = c.is_list;
    cd["is_dict"] = c.is_dict;
    cd["num_values"] = c.num_values;
    cd["null_count"] = c.null_count;
    cd["val_off"] = c.val_off;
    cd["val_len"] = c.val_len;
    cd["validity_off"] = c.validity_off;
    cd["dict_off"] = c.dict_off;
    cd["dict_len"] = c.dict_len;
    cd["run_off"] = c.run_off;
    cd["run_cnt"] = c.run_cnt;
    cd["dense_n"] = c.dense_n;
    cd["soff_off"] = c.soff_off;
    cd["sbytes_off"] = c.sbytes_off;
    cd["sbytes_len"] = c.sbytes_len;
    cols.append(cd);
  }
  d["cols"] = cols;
  // free the staging structures (dozens of MB of chunk vectors) off the
  // critical path — destructor cost was visible in the fetch time
  std::thread([p = st.release()]() { delete p; }).detach();
  return d;
}

// ---------------------------------------------------------------------- //
// murmur3 (CPU)
// ---------------------------------------------------------------------- //

// Hash fixed-width columns with seed chaining; returns int64 tensor of u32
// hashes. columns: CPU tensors (bool/int8/16/32/64/float32/64). Strings:
// pass (offsets,bytes) via hash_string_column step.
static torch::Tensor hash_columns_cpu(
    const std::vector<torch::Tensor>& columns,
    const std::vector<c10::optional<torch::Tensor>>& validity) {
  TORCH_CHECK(!columns.empty());
  int64_t n = columns[0].numel();
  auto out = torch::empty({n}, torch::kInt64);
  int64_t* hp = out.data_ptr<int64_t>();
  std::vector<torch::Tensor> cols;
  for (auto& c : columns) cols.push_back(c.contiguous().cpu());

  for (size_t ci = 0; ci < cols.size(); ci++) {
    torch::Tensor c = cols[ci];
    const uint8_t* vmask = nullptr;
    torch::Tensor vt;
    if (validity[ci].has_value()) {
      vt = validity[ci]->contiguous().cpu().to(torch::kUInt8);
      vmask = vt.data_ptr<uint8_t>();
    }
    auto st = c.scalar_type();
    at::parallel_for(0, n, 4096, [&](int64_t b, int64_t e) {
      for (int64_t i = b; i < e; i++) {
        uint32_t seed = ci == 0 ? kHashSeed : (uint32_t)hp[i];
        if (vmask && !vmask[i]) {
          if (ci == 0) hp[i] = 0;
          continue;
        }
        uint32_t hv;
        switch (st) {
          case torch::kBool:
            hv = spark_hash_u32((uint32_t)c.data_ptr<bool>()[i], seed);
            break;
          case torch::kInt8:
            hv = spark_hash_u32((uint32_t)(int32_t)c.data_ptr<int8_t>()[i], seed);
            break;
          case torch::kInt16:
            hv = spark_hash_u32((uint32_t)(int32_t)c.data_ptr<int16_t>()[i], seed);
            break;
          case torch::kInt32:
            hv = spark_hash_u32((uint32_t)c.data_ptr<int32_t>()[i], seed);
            break;
          case torch::kInt64:
            hv = spark_hash_u64((uint64_t)c.data_ptr<int64_t>()[i], seed);
            break;
          case torch::kFloat:
            hv = spark_hash_f32(c.data_ptr<float>()[i], seed);
            break;
          case torch::kDouble:
            hv = spark_hash_f64(c.data_ptr<double>()[i], seed);
            break;
          default:
            hv = 0;  // checked below
        }
        hp[i] = (int64_t)hv;
      }
    });
    TORCH_CHECK(st == torch::kBool || st == torch::kInt8 || st == torch::kInt16 ||
                    st == torch::kInt32 || st == torch::kInt64 ||
                    st == torch::kFloat || st == torch::kDouble,
                "unsupported dtype for murmur3");
  }
  return out;
}

// string column step: chain into existing hashes (or start, col_index==0)
static torch::Tensor hash_string_column_cpu(torch::Tensor offsets,
                                            torch::Tensor bytes,
                                            c10::optional<torch::Tensor> validity,
                                            c10::optional<torch::Tensor> prev,
                                            bool first_column) {
  auto off = offsets.contiguous().cpu().to(torch::kInt32);
  auto by = bytes.contiguous().cpu();
  int64_t n = off.numel() - 1;
  auto out = prev.has_value() ? prev->clone() : torch::zeros({n}, torch::kInt64);
  int64_t* hp = out.data_ptr<int64_t>();
  const int32_t* op = off.data_ptr<int32_t>();
  const uint8_t* bp = (const uint8_t*)by.data_ptr();
  const uint8_t* vmask = nullptr;
  torch::Tensor vt;
  if (validity.has_value()) {
    vt = validity->contiguous().cpu().to(torch::kUInt8);
    vmask = vt.data_ptr<uint8_t>();
  }
  at::parallel_for(0, n, 4096, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; i++) {
      if (vmask && !vmask[i]) {
        if (first_column) hp[i] = 0;
        continue;
      }
      uint32_t seed = first_column ? kHashSeed : (uint32_t)hp[i];
      hp[i] = (int64_t)spark_hash_bytes(bp + op[i], op[i + 1] - op[i], seed);
    }
  });
  return out;
}

static torch::Tensor bucket_ids_from_hashes(torch::Tensor hashes, int64_t nbuckets) {
  auto h = hashes.contiguous().cpu();
  int64_t n = h.numel();
  auto out = torch::empty({n}, torch::kInt32);
  const int64_t* hp = h.data_ptr<int64_t>();
  int32_t* op = out.data_ptr<int32_t>();
  for (int64_t i = 0; i < n; i++)
    op[i] = (int32_t)((uint32_t)hp[i] % (uint32_t)nbuckets);
  return out;
}

// list<string> columns ride the MOR merge as opaque per-row byte blobs
// whose payload is the PLAIN parquet representation ([u32 len][bytes]
// per element). After merge this parses the stream back into row ->
// element and element -> byte offsets (the chain walk is sequential, so
// it lives here at memcpy speed instead of a python loop).
static py::dict split_len_prefixed(const torch::Tensor& bytes,
                                   const torch::Tensor& row_byte_offsets) {
  torch::Tensor b = bytes.contiguous().cpu();
  torch::Tensor ro = row_byte_offsets.contiguous().cpu().to(torch::kInt64);
  const uint8_t* bp = b.data_ptr<uint8_t>();
  const int64_t* rp = ro.data_ptr<int64_t>();
  int64_t nrows = ro.numel() - 1;
  int64_t total = b.numel();
  // count elements first (u32 walk)
  int64_t m = 0;
  {
    int64_t p = 0;
    while (p + 4 <= total) {
      uint32_t len;
      std::memcpy(&len, bp + p, 4);
      p += 4 + len;
      m++;
    }
    TORCH_CHECK(p == total, "corrupt len-prefixed stream");
  }
  auto row_eoffs = torch::empty({nrows + 1}, torch::kInt64);
  auto eoffs = torch::empty({m + 1}, torch::kInt64);
  int64_t out_bytes = total - 4 * m;
  auto payload = torch::empty({out_bytes}, torch::kUInt8);
  int64_t* rep = row_eoffs.data_ptr<int64_t>();
  int64_t* ep = eoffs.data_ptr<int64_t>();
  uint8_t* pp = payload.data_ptr<uint8_t>();
  int64_t p = 0, e = 0, w = 0, r = 0;
  ep[0] = 0;
  rep[0] = 0;
  while (p + 4 <= total) {
    while (r < nrows && rp[r] == p) rep[r++] = e;  // rows ending here
    uint32_t len;
    std::memcpy(&len, bp + p, 4);
    p += 4;
    std::memcpy(pp + w, bp + p, len);
    p += len;
    w += len;
    ep[++e] = w;
  }
  while (r <= nrows) rep[r++] = e;  // trailing (incl. empty) rows
  py::dict d;
  d["row_offsets"] = row_eoffs;
  d["elem_offsets"] = eoffs;
  d["bytes"] = payload;
  return d;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "lakesoul_amd host-native core";
  m.def("write_parquet", &write_parquet, "write a parquet file",
        py::arg("path"), py::arg("names"), py::arg("dtypes"),
        py::arg("columns"), py::arg("offsets"), py::arg("validity"),
        py::arg("nullable"), py::arg("row_group_size"), py::arg("codec"),
        py::arg("level"),
        py::arg("elem_offsets") = std::vector<c10::optional<torch::Tensor>>{},
        py::arg("parents") = std::vector<std::string>{});
  m.def("writer_open", &writer_open, py::arg("path"), py::arg("names"),
        py::arg("dtypes"), py::arg("nullable"), py::arg("row_group_size"),
        py::arg("codec"), py::arg("level"),
        py::arg("parents") = std::vector<std::string>{});
  m.def("writer_write", &writer_write, py::arg("h"), py::arg("columns"),
        py::arg("offsets"), py::arg("validity"),
        py::arg("elem_offsets") = std::vector<c10::optional<torch::Tensor>>{});
  m.def("split_len_prefixed", &split_len_prefixed,
        "parse a PLAIN len-prefixed byte-array stream back into row/element offsets");
  m.def("writer_bytes", &writer_bytes);
  m.def("writer_finish", &writer_finish);
  m.def("writer_abort", &writer_abort);
  m.def("open_parquet", &open_parquet);
  m.def("close_parquet", &close_parquet);
  m.def("parquet_meta", &parquet_meta);
  m.def("read_chunk_raw", &read_chunk_raw);
  m.def("read_chunk_cpu", &read_chunk_cpu);
  m.def("read_chunks_cpu_batch", &read_chunks_cpu_batch);
  m.def("prep_rle_runs", &prep_rle_runs);
  m.def("read_unit_raw", &read_unit_raw_py, py::arg("paths"), py::arg("names"),
        py::arg("nthreads") = 0, py::arg("pin") = true,
        py::arg("gpu_snappy") = false, py::arg("gpu_zstd") = false,
        py::arg("gpu_zstd_frac") = 1.0);
  m.def("read_chunks_raw_batch", &read_chunks_raw_batch);
  m.def("zstd_compress_ref", [](py::bytes src, int64_t level) {
    std::string b = src;
    auto out = zstd_compress((const uint8_t*)b.data(), b.size(), (int)level);
    return py::bytes((const char*)out.data(), out.size());
  });
  m.def("zstd_decompress_ref", [](py::bytes src, int64_t cap) {
    std::string b = src;
    std::vector<uint8_t> out((size_t)cap);
    zstd_decompress_into(out.data(), (size_t)cap, (const uint8_t*)b.data(),
                         b.size());
    return py::bytes((const char*)out.data(), (size_t)cap);
  });
  m.def("zstd_decode_ref", [](py::bytes src, int64_t cap) {
    // differential-test hook for the from-scratch zstd decoder
    std::string b = src;
    std::vector<uint8_t> out((size_t)cap);
    static thread_local std::unique_ptr<lszstd::Ctx> ctx;
    if (!ctx) ctx.reset(new lszstd::Ctx());
    int64_t n = lszstd::decode((const uint8_t*)b.data(), (int64_t)b.size(),
                               out.data(), cap, ctx.get());
    if (n < 0) throw std::runtime_error("lszstd decode failed");
    return py::bytes((const char*)out.data(), (size_t)n);
  });
  m.def("hash_columns_cpu", &hash_columns_cpu);
  m.def("hash_string_column_cpu", &hash_string_column_cpu);
  m.def("bucket_ids_from_hashes", &bucket_ids_from_hashes);
  m.def("spark_hash_bytes", [](py::bytes b, int64_t seed) {
    std::string s = b;
    return (int64_t)spark_hash_bytes((const uint8_t*)s.data(), (int64_t)s.size(),
                                     (uint32_t)seed);
  });
}
