// Unit reader: read ALL files of one scan unit (bucket) in one call.
//
// Two-stage design so the python binding can allocate pinned torch
// tensors and have chunk payloads land in them directly (no second
// copy):
//   stage1: open files (via a footer/mmap cache — the analog of the
//           reference's file-meta cache, session.rs:86-105), decompress
//           every (file, column, row-group) chunk across the persistent
//           thread pool, compute the buffer layout and RLE run table.
//   fill:   parallel memcpy of chunk payloads straight into the caller's
//           destination buffers.
//
// Buffer layout (8-byte aligned so torch dtype views work):
//   values  : per-(file,col) dense PLAIN payloads, row-group chunks
//             contiguous (a no-null column is ONE device view);
//             dict-index payloads (+8 pad); string bytes
//   validity: per-(file,col) validity bytes, contiguous across RGs
//   dicts   : concatenated per-RG dictionaries (element bias applied in
//             the GPU expansion kernel)
//   runs    : int64 [m][6] {dense_out_off, n, is_literal,
//             value_or_abs_bitoff, bit_width, dict_elem_bias}
//   soffs   : int64 string offsets per (file,col), rebased across chunks
#pragma once

#include <atomic>
#include <cstring>
#include <list>
#include <memory>
#include <mutex>
#include <thread>
#include <unordered_map>
#include <vector>

#include <chrono>

#include "parquet_file.h"
#include "rle.h"
#include "thread_pool.h"

namespace lakesoul {

// ---------------------------------------------------------------------- //
// file handle cache (footer + mmap reuse across scans/steps)
// ---------------------------------------------------------------------- //

class FileCache {
 public:
  static FileCache& instance() {
    static FileCache* c = new FileCache();
    return *c;
  }

  std::shared_ptr<ParquetFile> open(const std::string& path) {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = map_.find(path);
    if (it != map_.end()) {
      lru_.splice(lru_.begin(), lru_, it->second.second);
      return it->second.first;
    }
    auto f = std::make_shared<ParquetFile>(path);
    lru_.push_front(path);
    map_[path] = {f, lru_.begin()};
    while (map_.size() > capacity_) {
      map_.erase(lru_.back());
      lru_.pop_back();
    }
    return f;
  }

  void invalidate(const std::string& path) {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = map_.find(path);
    if (it != map_.end()) {
      lru_.erase(it->second.second);
      map_.erase(it);
    }
  }

 private:
  std::mutex mu_;
  size_t capacity_ = 512;
  std::list<std::string> lru_;
  std::unordered_map<
      std::string,
      std::pair<std::shared_ptr<ParquetFile>, std::list<std::string>::iterator>>
      map_;
};

// ---------------------------------------------------------------------- //

struct UnitColumn {
  int file_idx;
  std::string name;
  bool present = false;
  bool is_string = false;
  bool is_dict = false;
  int physical = 0;
  int64_t num_values = 0;
  int64_t null_count = 0;
  int64_t val_off = 0, val_len = 0;
  int64_t validity_off = -1;
  int64_t dict_off = 0, dict_len = 0;
  int64_t run_off = 0, run_cnt = 0;
  int64_t dense_n = 0;
  int64_t soff_off = -1;
  int64_t sbytes_off = 0, sbytes_len = 0;
  bool is_list = false;  // element values in the sbytes region, element
                         // offsets in soffs, ROW validity in validity
};

inline int64_t ru_align8(int64_t x) { return (x + 7) & ~7LL; }

struct UnitStage {
  struct FileData {
    std::shared_ptr<ParquetFile> f;
    std::vector<std::vector<ParquetFile::ChunkData>> chunks;
    std::vector<int> col_idx;
  };
  struct StrDecoded {
    std::vector<int64_t> offs;
    std::vector<uint8_t> bytes;
    std::vector<uint8_t> row_valid;  // lists: per-row validity (may be empty)
  };

  int64_t t_open_us = 0, t_chunks_us = 0, t_layout_us = 0;
  size_t ncols = 0;
  std::vector<FileData> files;
  std::vector<UnitColumn> cols;
  std::vector<int64_t> file_rows;
  std::vector<int64_t> runs;  // [m][6]
  std::vector<int64_t> snappy_jobs;  // [m][4]: comp_off, comp_len, dst_off(values), dst_len
  std::vector<int64_t> zstd_jobs;    // [m][4]: same layout, zstd frames
  struct HostJob {
    int file_idx;
    int64_t file_off, comp_len, dst_off, dst_len;
    int32_t codec;
  };
  std::vector<HostJob> host_jobs;
  std::vector<std::unique_ptr<StrDecoded>> str_cols;
  int64_t values_size = 0, validity_size = 0, dicts_size = 0, soffs_size = 0,
          comp_size = 0;
};

inline std::unique_ptr<UnitStage> read_unit_stage1(
    const std::vector<std::string>& paths, const std::vector<std::string>& names,
    bool gpu_snappy = false, bool gpu_zstd = false,
    double gpu_zstd_frac = 1.0) {
  auto st = std::make_unique<UnitStage>();
  UnitStage& S = *st;
  auto now = [] { return std::chrono::steady_clock::now(); };
  auto us = [](auto a, auto b) {
    return std::chrono::duration_cast<std::chrono::microseconds>(b - a).count();
  };
  auto tp0 = now();
  size_t nfiles = paths.size();
  S.ncols = names.size();
  S.files.resize(nfiles);
  S.file_rows.resize(nfiles);

  // open via cache + build the (file, col, rg) task list
  struct Task {
    size_t fi, c, rg;
  };
  std::vector<Task> tasks;
  for (size_t i = 0; i < nfiles; i++) {
    auto& fd = S.files[i];
    fd.f = FileCache::instance().open(paths[i]);
    S.file_rows[i] = fd.f->num_rows();
    fd.chunks.resize(names.size());
    fd.col_idx.resize(names.size());
    size_t nrg = fd.f->num_row_groups();
    for (size_t c = 0; c < names.size(); c++) {
      int ci = fd.f->column_index(names[c]);
      fd.col_idx[c] = ci;
      if (ci < 0) continue;
      fd.chunks[c].resize(nrg);
      for (size_t rg = 0; rg < nrg; rg++) tasks.push_back({i, c, rg});
    }
  }
  auto tp1 = now();
  {
    std::string err;
    std::mutex err_mu;
    ThreadPool::instance().parallel_for((int64_t)tasks.size(), [&](int64_t i) {
      try {
        const Task& t = tasks[i];
        auto& fd = S.files[t.fi];
        // hybrid host/GPU decompress split (profiles/r01_gpu_zstd.md
        // round-2 plan item 3): route gpu_zstd_frac of the chunks to the
        // GPU zstd kernel, the rest to the host pool — the prefetch
        // pipeline overlaps them across units, so steady-state unit cost
        // is max(host share, gpu share) instead of the loser's total
        bool this_gpu_zstd =
            gpu_zstd && ((double)((i * 2654435761u) % 1000) < gpu_zstd_frac * 1000.0);
        fd.chunks[t.c][t.rg] =
            fd.f->read_chunk(t.rg, fd.col_idx[t.c], gpu_snappy, true, this_gpu_zstd);
      } catch (std::exception& e) {
        std::lock_guard<std::mutex> lk(err_mu);
        err = e.what();
      }
    });
    if (!err.empty()) throw std::runtime_error(err);
  }
  auto tp2 = now();
  S.t_open_us = us(tp0, tp1);
  S.t_chunks_us = us(tp1, tp2);

  // layout
  int64_t vpos = 0, vapos = 0, dpos = 0, spos = 0;
  for (size_t fi = 0; fi < nfiles; fi++) {
    auto& fd = S.files[fi];
    for (size_t c = 0; c < names.size(); c++) {
      UnitColumn uc;
      uc.file_idx = (int)fi;
      uc.name = names[c];
      if (fd.col_idx[c] < 0) {
        S.cols.push_back(uc);
        continue;
      }
      uc.present = true;
      const ColumnDesc& cd = fd.f->columns()[fd.col_idx[c]];
      uc.physical = cd.physical;
      uc.is_list = cd.is_list;
      uc.is_string = !cd.is_list && cd.physical == PT_BYTE_ARRAY;
      auto& chs = fd.chunks[c];
      if (uc.is_list) {
        // rows (not element slots) are the unit currency for lists;
        // payload layout happens in the decoded pre-pass below (same
        // host-decode lane as strings)
        int64_t rows = 0;
        for (auto& ch : chs)
          rows += ch.list_offsets.empty() ? 0
                  : (int64_t)ch.list_offsets.size() - 1;
        uc.num_values = rows;
        bool any_null_row = false;
        for (auto& ch : chs)
          for (auto v : ch.list_validity)
            if (!v) any_null_row = true;
        uc.null_count = 0;
        for (auto& ch : chs) {
          size_t rows_c = ch.list_offsets.empty() ? 0 : ch.list_offsets.size() - 1;
          for (size_t i = 0; i < rows_c && i < ch.list_validity.size(); i++)
            if (!ch.list_validity[i]) uc.null_count++;
        }
        if (any_null_row) {
          uc.validity_off = vapos;
          vapos += rows;
        }
        uc.soff_off = spos;
        spos += rows + 1;
        S.cols.push_back(uc);
        continue;
      }
      int64_t nv = 0, nulls = 0;
      bool any_dict = false, any_valid = false;
      for (auto& ch : chs) {
        nv += ch.num_values;
        nulls += ch.null_count;
        if (ch.is_dict) any_dict = true;
        if (!ch.validity.empty()) any_valid = true;
      }
      uc.num_values = nv;
      uc.null_count = nulls;
      uc.is_dict = any_dict;
      if (any_valid) {
        uc.validity_off = vapos;
        vapos += nv;
      }
      if (uc.is_string) {
        uc.soff_off = spos;
        spos += nv + 1;
      } else if (uc.is_dict) {
        uc.val_off = vpos;
        int64_t plen = 0, dlen = 0, dense = 0;
        for (auto& ch : chs) {
          plen += ru_align8((int64_t)ch.values.size());
          dlen += ru_align8((int64_t)ch.dict.size());
          dense += ch.num_values - ch.null_count;
        }
        uc.val_len = plen + 8;
        vpos += uc.val_len;
        uc.dict_off = dpos;
        uc.dict_len = dlen;
        dpos += dlen;
        uc.dense_n = dense;
        // runs (bit offsets are relative to the values buffer)
        uc.run_off = (int64_t)S.runs.size() / 6;
        int64_t poff = uc.val_off;
        int64_t dense_off = 0;
        int64_t bias = 0;
        int es = physical_elem_size(uc.physical);
        for (auto& ch : chs) {
          for (auto& ip : ch.idx_pages) {
            std::vector<RleRun> rr;
            parse_rle_runs(ch.values.data() + ip.payload_off,
                           (size_t)ip.payload_len, ip.bit_width, ip.n,
                           (poff + ip.payload_off) * 8, rr);
            for (auto& r : rr) {
              S.runs.push_back(r.out_off + dense_off);
              S.runs.push_back(r.n);
              S.runs.push_back(r.is_literal);
              S.runs.push_back(r.is_literal ? r.bit_off : (int64_t)r.value + bias);
              S.runs.push_back(ip.bit_width);
              S.runs.push_back(bias);
            }
            dense_off += ip.n;
          }
          poff += ru_align8((int64_t)ch.values.size());
          bias += (int64_t)(ru_align8((int64_t)ch.dict.size()) / es);
        }
        uc.run_cnt = (int64_t)S.runs.size() / 6 - uc.run_off;
      } else {
        uc.val_off = vpos;
        int64_t plen = 0;
        for (auto& ch : chs) {
          if (ch.gpu_compressed) {
            for (auto& cp : ch.comp_pages) {
              auto& jobs = (cp.codec == CODEC_ZSTD) ? S.zstd_jobs : S.snappy_jobs;
              jobs.push_back(S.comp_size + cp.comp_off);
              jobs.push_back(cp.comp_len);
              jobs.push_back(vpos + plen + cp.out_off);
              jobs.push_back(cp.out_len);
            }
            S.comp_size += (int64_t)ch.comp.size();
            plen += ch.values_len;
          } else if (ch.host_deferred) {
            for (auto& dp : ch.defer_pages) {
              S.host_jobs.push_back(
                  {(int)fi, dp.file_off, dp.comp_len, vpos + plen + dp.out_off,
                   dp.out_len, dp.codec});
            }
            plen += ch.values_len;
          } else {
            plen += (int64_t)ch.values.size();
          }
        }
        uc.val_len = plen;
        vpos += ru_align8(plen);
      }
      S.cols.push_back(uc);
    }
  }

  // string pre-pass (decode to learn byte totals) — parallel per column
  S.str_cols.resize(S.cols.size());
  {
    std::vector<size_t> str_idx;
    for (size_t u = 0; u < S.cols.size(); u++)
      if (S.cols[u].present && (S.cols[u].is_string || S.cols[u].is_list))
        str_idx.push_back(u);
    std::string err;
    std::mutex err_mu;
    ThreadPool::instance().parallel_for((int64_t)str_idx.size(), [&](int64_t k) {
      try {
        size_t u = str_idx[k];
        UnitColumn& uc = S.cols[u];
        auto& fd = S.files[uc.file_idx];
        auto sd = std::make_unique<UnitStage::StrDecoded>();
        sd->offs.push_back(0);
        if (uc.is_list) {
          int es = physical_elem_size(uc.physical);
          for (auto& ch : fd.chunks[u % S.ncols]) {
            DecodedColumn dc = decode_chunk_cpu(ch);
            int64_t elem_base = (int64_t)sd->bytes.size() / es;
            sd->bytes.insert(sd->bytes.end(), dc.data.begin(), dc.data.end());
            for (size_t i = 1; i < dc.list_offsets.size(); i++)
              sd->offs.push_back(elem_base + dc.list_offsets[i]);
            size_t rows = dc.list_offsets.empty()
                              ? 0 : dc.list_offsets.size() - 1;
            if (!dc.list_validity.empty())
              sd->row_valid.insert(sd->row_valid.end(),
                                   dc.list_validity.begin(),
                                   dc.list_validity.begin() + rows);
            else
              sd->row_valid.insert(sd->row_valid.end(), rows, 1);
          }
        } else {
          for (auto& ch : fd.chunks[u % S.ncols]) {
            DecodedColumn dc = decode_chunk_cpu(ch);
            int64_t base = (int64_t)sd->bytes.size();
            sd->bytes.insert(sd->bytes.end(), dc.bytes.begin(), dc.bytes.end());
            for (size_t i = 1; i < dc.offsets.size(); i++)
              sd->offs.push_back(base + dc.offsets[i]);
          }
        }
        S.str_cols[u] = std::move(sd);
      } catch (std::exception& e) {
        std::lock_guard<std::mutex> lk(err_mu);
        err = e.what();
      }
    });
    if (!err.empty()) throw std::runtime_error(err);
    for (size_t u : str_idx) {
      UnitColumn& uc = S.cols[u];
      uc.sbytes_off = vpos;
      uc.sbytes_len = (int64_t)S.str_cols[u]->bytes.size();
      vpos += ru_align8(uc.sbytes_len);
    }
  }

  S.values_size = vpos;
  S.validity_size = vapos;
  S.dicts_size = dpos;
  S.soffs_size = spos;
  S.t_layout_us = us(tp2, now());
  return st;
}

// Fill caller-allocated buffers (sized from stage1) — pure parallel memcpy.
inline void read_unit_fill(UnitStage& S, uint8_t* values, uint8_t* validity,
                           uint8_t* dicts, int64_t* soffs, uint8_t* comp) {
  // compressed snappy page bodies, in the same chunk order the layout
  // assigned (serial: tiny relative to values)
  if (comp) {
    int64_t cpos = 0;
    for (size_t u = 0; u < S.cols.size(); u++) {
      UnitColumn& uc = S.cols[u];
      if (!uc.present || uc.is_string || uc.is_dict) continue;
      auto& chs = S.files[uc.file_idx].chunks[u % S.ncols];
      for (auto& ch : chs) {
        if (ch.gpu_compressed && !ch.comp.empty()) {
          std::memcpy(comp + cpos, ch.comp.data(), ch.comp.size());
          cpos += (int64_t)ch.comp.size();
        }
      }
    }
  }
  if (S.validity_size) std::memset(validity, 1, (size_t)S.validity_size);
  ThreadPool::instance().parallel_for((int64_t)S.cols.size(), [&](int64_t u) {
    UnitColumn& uc = S.cols[u];
    if (!uc.present) return;
    auto& fd = S.files[uc.file_idx];
    auto& chs = fd.chunks[u % S.ncols];
    if (uc.validity_off >= 0 && !uc.is_list) {
      int64_t off = uc.validity_off;
      for (auto& ch : chs) {
        if (!ch.validity.empty())
          std::memcpy(validity + off, ch.validity.data(), ch.validity.size());
        off += ch.num_values;
      }
    }
    if (uc.is_string || uc.is_list) {
      auto& sd = *S.str_cols[u];
      std::memcpy(soffs + uc.soff_off, sd.offs.data(), sd.offs.size() * 8);
      if (!sd.bytes.empty())
        std::memcpy(values + uc.sbytes_off, sd.bytes.data(), sd.bytes.size());
      if (uc.is_list && uc.validity_off >= 0 && !sd.row_valid.empty())
        std::memcpy(validity + uc.validity_off, sd.row_valid.data(),
                    sd.row_valid.size());
      return;
    }
    if (uc.is_dict) {
      int64_t poff = uc.val_off;
      int64_t doff = uc.dict_off;
      for (auto& ch : chs) {
        std::memcpy(values + poff, ch.values.data(), ch.values.size());
        std::memcpy(dicts + doff, ch.dict.data(), ch.dict.size());
        poff += ru_align8((int64_t)ch.values.size());
        doff += ru_align8((int64_t)ch.dict.size());
      }
      return;
    }
    int64_t off = uc.val_off;
    for (auto& ch : chs) {
      if (ch.gpu_compressed || ch.host_deferred) {
        off += ch.values_len;  // GPU snappy kernel / host job fills it
        continue;
      }
      std::memcpy(values + off, ch.values.data(), ch.values.size());
      off += (int64_t)ch.values.size();
    }
  });
  // deferred pages: decompress straight from the file mmap into the
  // destination buffer (one pass, zero staging)
  ThreadPool::instance().parallel_for((int64_t)S.host_jobs.size(), [&](int64_t i) {
    const UnitStage::HostJob& hj = S.host_jobs[i];
    const uint8_t* src = S.files[hj.file_idx].f->data_at(hj.file_off);
    decompress_into(hj.codec, values + hj.dst_off, (size_t)hj.dst_len, src,
                    (size_t)hj.comp_len);
  });
}

}  // namespace lakesoul
