// lakesoul_amd._hip — torch bindings for the gfx950 kernels (kernels.hip).
#include <torch/extension.h>

#include <hip/hip_runtime.h>

#include <c10/hip/HIPStream.h>

namespace py = pybind11;

namespace lakesoul {

struct GatherTable {
  const void* src[16];
  void* dst[16];
  int esize[16];
  int ncols;
};

void launch_hash_fixed(int dt, const void*, const uint8_t*, const int64_t*, int64_t*, int64_t, int, hipStream_t);
void launch_hash_string(const int32_t*, const uint8_t*, const uint8_t*, const int64_t*, int64_t*, int64_t, int, hipStream_t);
void launch_bucket_ids(const int64_t*, int32_t*, int64_t, uint32_t, hipStream_t);
void launch_rle_expand(const uint8_t*, const int64_t*, int64_t, int32_t*, int64_t, hipStream_t);
template <typename T>
void launch_dict_gather_scatter(const T*, const int32_t*, const uint8_t*, const int64_t*, T*, int64_t, hipStream_t);
template <typename T>
void launch_scatter_valid(const T*, const uint8_t*, const int64_t*, T*, int64_t, hipStream_t);
void launch_merge_pairs(const uint64_t*, const uint64_t*, int64_t, const uint64_t*, const uint64_t*, int64_t, uint64_t*, uint64_t*, hipStream_t);
void launch_keep_last(const uint64_t*, uint8_t*, int64_t, hipStream_t);
void launch_group_start(const uint64_t*, uint8_t*, int64_t, hipStream_t);
void launch_pack_key_i64(const int64_t*, uint64_t*, int64_t, hipStream_t);
void launch_pack_key_2xi32(const int32_t*, const int32_t*, uint64_t*, int64_t, hipStream_t);
void launch_gather_fixed_multi(const GatherTable&, const int64_t*, int64_t, hipStream_t);
void launch_gather_strings(const uint8_t*, const int64_t*, const int64_t*, const int64_t*, uint8_t*, int64_t, hipStream_t);
void launch_bytes_ne_mask(const int64_t*, const uint8_t*, const uint8_t*, int, uint8_t*, int64_t, hipStream_t);
template <typename T, typename ACC>
void launch_segmented_sum(const T*, const int64_t*, const uint8_t*, const uint8_t*, ACC*, int32_t*, int64_t, hipStream_t);
void launch_segmented_last(const int64_t*, const uint8_t*, const uint8_t*, int64_t*, int64_t, hipStream_t);
void launch_hamming_scores(const uint64_t*, const uint64_t*, int32_t*, int64_t,
                           int, int, hipStream_t);
void launch_str_chunk_keys(const int64_t*, const uint8_t*, int64_t, int64_t*,
                           int64_t, hipStream_t);
void launch_ann_scores(const short*, const short*, float*, int64_t, int32_t, int32_t, hipStream_t);
void launch_ann_scores_t(const short*, const short*, float*, int64_t, int32_t, int32_t, hipStream_t);
void launch_fastscan_lut(const uint8_t*, const float*, float*, int64_t,
                         int32_t, int32_t, int32_t, hipStream_t);
void launch_fastscan_ex_dot(const uint8_t*, const float*, float*, int64_t,
                            int32_t, int32_t, int32_t, hipStream_t);
void launch_fastscan_ex_dot_pairs(const uint8_t*, const int64_t*,
                                  const float*, float*, int32_t, int32_t,
                                  int32_t, int32_t, hipStream_t);
void launch_fastscan_est(const uint8_t*, const float*, const float*,
                         const float*, const int32_t*, const float*,
                         const float*, float*, int64_t, int32_t, int32_t,
                         int32_t, int32_t, hipStream_t);
void launch_snappy_decompress(const uint8_t*, const int64_t*, int64_t, uint8_t*, int32_t*, hipStream_t);
void launch_zstd_decompress(const uint8_t*, const int64_t*, int64_t, uint8_t*,
                            uint8_t*, int64_t, int32_t*, hipStream_t);
int64_t lsz_gpu_litbuf_bytes();

}  // namespace lakesoul

using namespace lakesoul;

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_GPU(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU"); \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

// optional/empty tensors may be placeholder CPU empties; any tensor the
// kernel will actually dereference must be device-resident (a CPU
// pointer reaching a kernel = memory fault, see Column.take fix)
#define CHECK_GPU_NONEMPTY(t)                                   \
  if ((t).numel()) {                                            \
    TORCH_CHECK((t).is_cuda(), #t " must be on GPU");           \
    TORCH_CHECK((t).is_contiguous(), #t " must be contiguous"); \
  }

static const uint8_t* opt_u8(const torch::Tensor& t) {
  return t.numel() ? t.data_ptr<uint8_t>() : nullptr;
}

// ---- hashing ---------------------------------------------------------- //

static torch::Tensor hash_fixed_column(torch::Tensor data, torch::Tensor validity,
                                       torch::Tensor prev, bool first) {
  CHECK_GPU_NONEMPTY(validity);
  CHECK_GPU_NONEMPTY(prev);
  CHECK_GPU(data);
  int64_t n = data.numel();
  auto out = torch::empty({n}, data.options().dtype(torch::kInt64));
  int dt;
  switch (data.scalar_type()) {
    case torch::kUInt8:
    case torch::kBool: dt = 0; break;
    case torch::kInt8: dt = 1; break;
    case torch::kInt16: dt = 2; break;
    case torch::kInt32: dt = 3; break;
    case torch::kInt64: dt = 4; break;
    case torch::kFloat: dt = 5; break;
    case torch::kDouble: dt = 6; break;
    default: TORCH_CHECK(false, "unsupported dtype for hash");
  }
  launch_hash_fixed(dt, data.data_ptr(), opt_u8(validity),
                    first ? nullptr : prev.data_ptr<int64_t>(),
                    out.data_ptr<int64_t>(), n, first ? 1 : 0, cur_stream());
  return out;
}

static torch::Tensor hash_string_column(torch::Tensor offsets, torch::Tensor bytes,
                                        torch::Tensor validity, torch::Tensor prev,
                                        bool first) {
  CHECK_GPU_NONEMPTY(bytes);
  CHECK_GPU_NONEMPTY(validity);
  CHECK_GPU_NONEMPTY(prev);
  CHECK_GPU(offsets);
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, offsets.options().dtype(torch::kInt64));
  launch_hash_string(offsets.data_ptr<int32_t>(), bytes.data_ptr<uint8_t>(),
                     opt_u8(validity), first ? nullptr : prev.data_ptr<int64_t>(),
                     out.data_ptr<int64_t>(), n, first ? 1 : 0, cur_stream());
  return out;
}

static torch::Tensor bucket_ids(torch::Tensor hashes, int64_t nbuckets) {
  CHECK_GPU(hashes);
  int64_t n = hashes.numel();
  auto out = torch::empty({n}, hashes.options().dtype(torch::kInt32));
  launch_bucket_ids(hashes.data_ptr<int64_t>(), out.data_ptr<int32_t>(), n,
                    (uint32_t)nbuckets, cur_stream());
  return out;
}

// ---- decode ----------------------------------------------------------- //

static torch::Tensor rle_expand(torch::Tensor payload, torch::Tensor runs, int64_t n) {
  CHECK_GPU(payload);
  CHECK_GPU(runs);
  auto out = torch::empty({n}, payload.options().dtype(torch::kInt32));
  launch_rle_expand(payload.data_ptr<uint8_t>(), runs.data_ptr<int64_t>(),
                    runs.size(0), out.data_ptr<int32_t>(), n, cur_stream());
  return out;
}

static torch::Tensor dict_gather_scatter(torch::Tensor dict_vals, torch::Tensor idx,
                                         torch::Tensor validity, torch::Tensor positions,
                                         int64_t elem_size, int64_t n) {
  CHECK_GPU_NONEMPTY(validity);
  CHECK_GPU_NONEMPTY(positions);
  CHECK_GPU(dict_vals);
  CHECK_GPU(idx);
  auto out = torch::empty({n * elem_size}, dict_vals.options().dtype(torch::kUInt8));
  const uint8_t* vmask = opt_u8(validity);
  const int64_t* pos = validity.numel() ? positions.data_ptr<int64_t>() : nullptr;
  if (elem_size == 4) {
    launch_dict_gather_scatter<uint32_t>(
        (const uint32_t*)dict_vals.data_ptr(), idx.data_ptr<int32_t>(), vmask,
        pos, (uint32_t*)out.data_ptr(), n, cur_stream());
  } else if (elem_size == 8) {
    launch_dict_gather_scatter<uint64_t>(
        (const uint64_t*)dict_vals.data_ptr(), idx.data_ptr<int32_t>(), vmask,
        pos, (uint64_t*)out.data_ptr(), n, cur_stream());
  } else if (elem_size == 1) {
    launch_dict_gather_scatter<uint8_t>(
        dict_vals.data_ptr<uint8_t>(), idx.data_ptr<int32_t>(), vmask, pos,
        out.data_ptr<uint8_t>(), n, cur_stream());
  } else {
    TORCH_CHECK(false, "unsupported elem size");
  }
  return out;
}

static torch::Tensor scatter_valid(torch::Tensor dense, torch::Tensor validity,
                                   torch::Tensor positions, int64_t elem_size,
                                   int64_t n) {
  CHECK_GPU_NONEMPTY(validity);
  CHECK_GPU_NONEMPTY(positions);
  CHECK_GPU(dense);
  auto out = torch::zeros({n * elem_size}, dense.options().dtype(torch::kUInt8));
  if (elem_size == 1)
    launch_scatter_valid<uint8_t>(dense.data_ptr<uint8_t>(), validity.data_ptr<uint8_t>(), positions.data_ptr<int64_t>(), out.data_ptr<uint8_t>(), n, cur_stream());
  else if (elem_size == 2)
    launch_scatter_valid<uint16_t>((const uint16_t*)dense.data_ptr(), validity.data_ptr<uint8_t>(), positions.data_ptr<int64_t>(), (uint16_t*)out.data_ptr(), n, cur_stream());
  else if (elem_size == 4)
    launch_scatter_valid<uint32_t>((const uint32_t*)dense.data_ptr(), validity.data_ptr<uint8_t>(), positions.data_ptr<int64_t>(), (uint32_t*)out.data_ptr(), n, cur_stream());
  else
    launch_scatter_valid<uint64_t>((const uint64_t*)dense.data_ptr(), validity.data_ptr<uint8_t>(), positions.data_ptr<int64_t>(), (uint64_t*)out.data_ptr(), n, cur_stream());
  return out;
}

// ---- merge ------------------------------------------------------------ //

static std::vector<torch::Tensor> merge_pairs(torch::Tensor kA, torch::Tensor vA,
                                              torch::Tensor kB, torch::Tensor vB) {
  CHECK_GPU_NONEMPTY(vA);
  CHECK_GPU_NONEMPTY(vB);
  CHECK_GPU(kA);
  CHECK_GPU(kB);
  int64_t nA = kA.numel(), nB = kB.numel();
  auto kOut = torch::empty({nA + nB}, kA.options());
  auto vOut = torch::empty({nA + nB}, vA.options());
  launch_merge_pairs((const uint64_t*)kA.data_ptr(), (const uint64_t*)vA.data_ptr(), nA,
                     (const uint64_t*)kB.data_ptr(), (const uint64_t*)vB.data_ptr(), nB,
                     (uint64_t*)kOut.data_ptr(), (uint64_t*)vOut.data_ptr(),
                     cur_stream());
  return {kOut, vOut};
}

static torch::Tensor keep_last_mask(torch::Tensor keys) {
  CHECK_GPU(keys);
  int64_t n = keys.numel();
  auto out = torch::empty({n}, keys.options().dtype(torch::kUInt8));
  launch_keep_last((const uint64_t*)keys.data_ptr(), out.data_ptr<uint8_t>(), n,
                   cur_stream());
  return out;
}

static torch::Tensor group_start_mask(torch::Tensor keys) {
  CHECK_GPU(keys);
  int64_t n = keys.numel();
  auto out = torch::empty({n}, keys.options().dtype(torch::kUInt8));
  launch_group_start((const uint64_t*)keys.data_ptr(), out.data_ptr<uint8_t>(), n,
                     cur_stream());
  return out;
}

static torch::Tensor pack_key_i64(torch::Tensor x) {
  CHECK_GPU(x);
  auto out = torch::empty_like(x);
  launch_pack_key_i64(x.data_ptr<int64_t>(), (uint64_t*)out.data_ptr(), x.numel(),
                      cur_stream());
  return out;
}

static torch::Tensor pack_key_2xi32(torch::Tensor hi, torch::Tensor lo) {
  CHECK_GPU_NONEMPTY(lo);
  CHECK_GPU(hi);
  auto out = torch::empty({hi.numel()}, hi.options().dtype(torch::kInt64));
  launch_pack_key_2xi32(hi.data_ptr<int32_t>(), lo.data_ptr<int32_t>(),
                        (uint64_t*)out.data_ptr(), hi.numel(), cur_stream());
  return out;
}

// ---- gathers ----------------------------------------------------------- //

static std::vector<torch::Tensor> gather_fixed_multi(std::vector<torch::Tensor> cols,
                                                     torch::Tensor idx) {
  for (auto& c : cols) CHECK_GPU_NONEMPTY(c);
  CHECK_GPU(idx);
  int64_t n = idx.numel();
  std::vector<torch::Tensor> outs;
  size_t c = 0;
  while (c < cols.size()) {
    GatherTable tbl;
    tbl.ncols = 0;
    for (; c < cols.size() && tbl.ncols < 16; c++) {
      CHECK_GPU(cols[c]);
      auto out = torch::empty({n}, cols[c].options());
      tbl.src[tbl.ncols] = cols[c].data_ptr();
      tbl.dst[tbl.ncols] = out.data_ptr();
      tbl.esize[tbl.ncols] = (int)cols[c].element_size();
      tbl.ncols++;
      outs.push_back(out);
    }
    launch_gather_fixed_multi(tbl, idx.data_ptr<int64_t>(), n, cur_stream());
  }
  return outs;
}

static torch::Tensor gather_strings(torch::Tensor src_bytes, torch::Tensor src_offsets,
                                    torch::Tensor idx, torch::Tensor dst_offsets) {
  CHECK_GPU_NONEMPTY(src_offsets);
  CHECK_GPU_NONEMPTY(idx);
  CHECK_GPU_NONEMPTY(dst_offsets);
  CHECK_GPU(src_bytes);
  int64_t n = idx.numel();
  int64_t total = dst_offsets.numel() ? dst_offsets[n].item<int64_t>() : 0;
  auto out = torch::empty({total}, src_bytes.options());
  launch_gather_strings(src_bytes.data_ptr<uint8_t>(), src_offsets.data_ptr<int64_t>(),
                        idx.data_ptr<int64_t>(), dst_offsets.data_ptr<int64_t>(),
                        out.data_ptr<uint8_t>(), n, cur_stream());
  return out;
}

static torch::Tensor bytes_ne_mask(torch::Tensor offsets, torch::Tensor bytes,
                                   torch::Tensor pattern) {
  CHECK_GPU_NONEMPTY(bytes);
  CHECK_GPU_NONEMPTY(pattern);
  CHECK_GPU(offsets);
  int64_t n = offsets.numel() - 1;
  auto out = torch::empty({n}, offsets.options().dtype(torch::kUInt8));
  launch_bytes_ne_mask(offsets.data_ptr<int64_t>(), bytes.data_ptr<uint8_t>(),
                       pattern.data_ptr<uint8_t>(), (int)pattern.numel(),
                       out.data_ptr<uint8_t>(), n, cur_stream());
  return out;
}

// ---- segmented merge ops ---------------------------------------------- //

static std::vector<torch::Tensor> segmented_sum(torch::Tensor vals, torch::Tensor grp,
                                                torch::Tensor contrib, torch::Tensor validity,
                                                int64_t ngroups) {
  CHECK_GPU_NONEMPTY(grp);
  CHECK_GPU_NONEMPTY(contrib);
  CHECK_GPU_NONEMPTY(validity);
  CHECK_GPU(vals);
  int64_t n = vals.numel();
  auto sums = torch::zeros({ngroups}, vals.options());
  auto has_null = torch::zeros({ngroups}, vals.options().dtype(torch::kInt32));
  const uint8_t* cb = opt_u8(contrib);
  const uint8_t* vb = opt_u8(validity);
  switch (vals.scalar_type()) {
    case torch::kFloat:
      launch_segmented_sum<float, float>(vals.data_ptr<float>(), grp.data_ptr<int64_t>(), cb, vb, sums.data_ptr<float>(), has_null.data_ptr<int32_t>(), n, cur_stream());
      break;
    case torch::kDouble:
      launch_segmented_sum<double, double>(vals.data_ptr<double>(), grp.data_ptr<int64_t>(), cb, vb, sums.data_ptr<double>(), has_null.data_ptr<int32_t>(), n, cur_stream());
      break;
    case torch::kInt32:
      launch_segmented_sum<int32_t, int32_t>(vals.data_ptr<int32_t>(), grp.data_ptr<int64_t>(), cb, vb, sums.data_ptr<int32_t>(), has_null.data_ptr<int32_t>(), n, cur_stream());
      break;
    case torch::kInt64:
      launch_segmented_sum<int64_t, int64_t>(vals.data_ptr<int64_t>(), grp.data_ptr<int64_t>(), cb, vb, sums.data_ptr<int64_t>(), has_null.data_ptr<int32_t>(), n, cur_stream());
      break;
    default:
      TORCH_CHECK(false, "unsupported dtype for segmented_sum");
  }
  return {sums, has_null};
}

static torch::Tensor segmented_last(torch::Tensor grp, torch::Tensor contrib,
                                    torch::Tensor validity, int64_t ngroups, int64_t n) {
  CHECK_GPU_NONEMPTY(contrib);
  CHECK_GPU_NONEMPTY(validity);
  CHECK_GPU(grp);
  auto out = torch::zeros({ngroups}, grp.options());
  launch_segmented_last(grp.data_ptr<int64_t>(), opt_u8(contrib), opt_u8(validity),
                        out.data_ptr<int64_t>(), n, cur_stream());
  return out - 1;  // -1 = no contributing row
}

// ---- ANN scoring (MFMA) ------------------------------------------------ //

static torch::Tensor ann_scores(torch::Tensor X, torch::Tensor Q) {
  CHECK_GPU(X);
  CHECK_GPU(Q);
  TORCH_CHECK(X.scalar_type() == torch::kBFloat16 && Q.scalar_type() == torch::kBFloat16,
              "ann_scores expects bf16");
  int64_t n = X.size(0);
  int64_t nq = Q.size(0);
  int64_t K = X.size(1);
  TORCH_CHECK(Q.size(1) == K, "dim mismatch");
  TORCH_CHECK(K % 32 == 0, "K must be a multiple of 32");
  TORCH_CHECK(nq % 16 == 0, "nq must be a multiple of 16 (pad queries)");
  auto out = torch::empty({n, nq}, X.options().dtype(torch::kFloat32));
  launch_ann_scores((const short*)X.data_ptr(), (const short*)Q.data_ptr(),
                    out.data_ptr<float>(), n, (int32_t)nq, (int32_t)K,
                    cur_stream());
  return out;
}

// query-major output (nq, n): the per-query top-k then reads contiguous
// rows (28 ms -> 2.5 ms for 5M x 64 k=10; benchmarks/topk_micro.py)
static torch::Tensor ann_scores_t(torch::Tensor X, torch::Tensor Q) {
  CHECK_GPU(X);
  CHECK_GPU(Q);
  TORCH_CHECK(X.scalar_type() == torch::kBFloat16 &&
                  Q.scalar_type() == torch::kBFloat16,
              "ann_scores_t expects bf16");
  int64_t n = X.size(0);
  int64_t nq = Q.size(0);
  int64_t K = X.size(1);
  TORCH_CHECK(Q.size(1) == K, "dim mismatch");
  TORCH_CHECK(K % 32 == 0, "K must be a multiple of 32");
  TORCH_CHECK(nq % 16 == 0, "nq must be a multiple of 16 (pad queries)");
  auto out = torch::empty({nq, n}, X.options().dtype(torch::kFloat32));
  launch_ann_scores_t((const short*)X.data_ptr(), (const short*)Q.data_ptr(),
                      out.data_ptr<float>(), n, (int32_t)nq, (int32_t)K,
                      cur_stream());
  return out;
}

// jobs int64 [n,4] = {src_off, src_len, dst_off, dst_len}; returns
// (dst_buffer, status int32[n]) — status 0 = ok.
static std::vector<torch::Tensor> snappy_decompress(torch::Tensor src,
                                                    torch::Tensor jobs,
                                                    int64_t total_dst) {
  CHECK_GPU(src);
  CHECK_GPU(jobs);
  auto dst = torch::empty({total_dst}, src.options());
  auto status = torch::empty({jobs.size(0)}, src.options().dtype(torch::kInt32));
  launch_snappy_decompress(src.data_ptr<uint8_t>(), jobs.data_ptr<int64_t>(),
                           jobs.size(0), dst.data_ptr<uint8_t>(),
                           status.data_ptr<int32_t>(), cur_stream());
  return {dst, status};
}

// ---------------------------------------------------------------------- //
// C++ scan-unit driver (simple path): decode + UseLast merge + gather for
// one unit entirely from C++ — the python per-unit op storm (GIL-bound)
// was the scan's critical path. Covers: integer single-PK (int64/int32),
// every-column UseLast, no CDC. Python falls back otherwise.
//
// desc: int64 [nuc,17] from _cpp.read_unit_raw (see module.cc):
// {present,is_string,is_dict,esize,num_values,null_count,val_off,val_len,
//  validity_off,dict_off,dict_len,run_off,run_cnt,dense_n,soff_off,
//  sbytes_off,sbytes_len}
// Returns: [n_out scalar?] -> list: for each read col:
//   fixed: [data_u8_gathered, validity_or_empty]
//   string: [offsets_i64, bytes_u8, validity_or_empty]
// plus the merged row count implicit in tensor sizes.
static py::list scan_unit_uselast(torch::Tensor vals, torch::Tensor validity_buf,
                                  torch::Tensor dicts, torch::Tensor runs,
                                  torch::Tensor soffs, torch::Tensor desc,
                                  int64_t nfiles, int64_t ncols, int64_t pk_ci,
                                  int64_t pk_es) {
  CHECK_GPU_NONEMPTY(validity_buf);
  CHECK_GPU_NONEMPTY(dicts);
  CHECK_GPU_NONEMPTY(runs);
  CHECK_GPU_NONEMPTY(soffs);
  CHECK_GPU(vals);
  py::gil_scoped_release rel;  // long device-side section (incl. one sync)
  auto device = vals.device();
  auto u8 = torch::TensorOptions().dtype(torch::kUInt8).device(device);
  auto i64 = torch::TensorOptions().dtype(torch::kInt64).device(device);
  auto da = desc.accessor<int64_t, 2>();
  auto stream = cur_stream();

  torch::Tensor empty_u8 = torch::empty({0}, u8);
  torch::Tensor runs2 = runs.numel() ? runs.view({-1, 6}) : runs;

  // ---- decode every (file, col) into dense device columns ----
  // decoded[f][c] = {data_u8 (es per elem), validity_u8 (or undef),
  //                  offsets_i64/bytes_u8 for strings}
  struct Col {
    torch::Tensor data, validity, offsets, bytes;
    bool is_string = false;
  };
  std::vector<std::vector<Col>> cols((size_t)nfiles,
                                     std::vector<Col>((size_t)ncols));
  for (int64_t f = 0; f < nfiles; f++) {
    for (int64_t c = 0; c < ncols; c++) {
      int64_t u = f * ncols + c;
      Col& out = cols[f][c];
      TORCH_CHECK(da[u][0], "scan_unit_uselast: absent column (fallback)");
      int64_t nv = da[u][4];
      bool has_null = da[u][8] >= 0 && da[u][5] > 0;
      torch::Tensor vmask;
      if (has_null) vmask = validity_buf.narrow(0, da[u][8], nv);
      if (da[u][1]) {  // string
        out.is_string = true;
        out.offsets = soffs.narrow(0, da[u][14], nv + 1);
        out.bytes = vals.narrow(0, da[u][15], da[u][16]);
        out.validity = vmask;
        continue;
      }
      int64_t es = da[u][3];
      if (da[u][2]) {  // dict
        auto rr = runs2.narrow(0, da[u][11], da[u][12]);
        auto payload = vals.narrow(0, da[u][6], da[u][7]);
        auto idx = torch::empty({da[u][13]}, torch::TensorOptions().dtype(torch::kInt32).device(device));
        launch_rle_expand(payload.data_ptr<uint8_t>(), rr.data_ptr<int64_t>(),
                          rr.size(0), idx.data_ptr<int32_t>(), da[u][13], stream);
        auto dv = dicts.narrow(0, da[u][9], da[u][10]);
        auto out_d = torch::empty({nv * es}, u8);
        const uint8_t* vm = has_null ? vmask.data_ptr<uint8_t>() : nullptr;
        torch::Tensor pos;
        const int64_t* posp = nullptr;
        if (has_null) {
          pos = at::cumsum(vmask, 0, torch::kInt64) - 1;
          posp = pos.data_ptr<int64_t>();
        }
        if (es == 4)
          launch_dict_gather_scatter<uint32_t>((const uint32_t*)dv.data_ptr(), idx.data_ptr<int32_t>(), vm, posp, (uint32_t*)out_d.data_ptr(), nv, stream);
        else if (es == 8)
          launch_dict_gather_scatter<uint64_t>((const uint64_t*)dv.data_ptr(), idx.data_ptr<int32_t>(), vm, posp, (uint64_t*)out_d.data_ptr(), nv, stream);
        else
          launch_dict_gather_scatter<uint8_t>(dv.data_ptr<uint8_t>(), idx.data_ptr<int32_t>(), vm, posp, out_d.data_ptr<uint8_t>(), nv, stream);
        out.data = out_d;
      } else {
        auto dense = vals.narrow(0, da[u][6], da[u][7]);
        if (has_null) {
          auto pos = at::cumsum(vmask, 0, torch::kInt64) - 1;
          auto out_d = torch::zeros({nv * es}, u8);
          if (es == 1)
            launch_scatter_valid<uint8_t>(dense.data_ptr<uint8_t>(), vmask.data_ptr<uint8_t>(), pos.data_ptr<int64_t>(), out_d.data_ptr<uint8_t>(), nv, stream);
          else if (es == 2)
            launch_scatter_valid<uint16_t>((const uint16_t*)dense.data_ptr(), vmask.data_ptr<uint8_t>(), pos.data_ptr<int64_t>(), (uint16_t*)out_d.data_ptr(), nv, stream);
          else if (es == 4)
            launch_scatter_valid<uint32_t>((const uint32_t*)dense.data_ptr(), vmask.data_ptr<uint8_t>(), pos.data_ptr<int64_t>(), (uint32_t*)out_d.data_ptr(), nv, stream);
          else
            launch_scatter_valid<uint64_t>((const uint64_t*)dense.data_ptr(), vmask.data_ptr<uint8_t>(), pos.data_ptr<int64_t>(), (uint64_t*)out_d.data_ptr(), nv, stream);
          out.data = out_d;
        } else {
          out.data = dense;
        }
      }
      out.validity = vmask;
    }
  }

  // ---- pack keys per file + pairwise merge-path merges ----
  std::vector<std::pair<torch::Tensor, torch::Tensor>> streams;
  int64_t base = 0;
  for (int64_t f = 0; f < nfiles; f++) {
    int64_t nv = da[f * ncols + pk_ci][4];
    auto& pkc = cols[f][pk_ci];
    torch::Tensor key64;
    if (pk_es == 8) {
      key64 = pkc.data.view(torch::kInt64);
    } else {
      key64 = pkc.data.view(torch::kInt32).to(torch::kInt64);
    }
    auto keys = torch::empty({nv}, i64);
    launch_pack_key_i64(key64.data_ptr<int64_t>(), (uint64_t*)keys.data_ptr(), nv, stream);
    auto vals_idx = torch::arange(base, base + nv, i64);
    streams.emplace_back(keys, vals_idx);
    base += nv;
  }
  while (streams.size() > 1) {
    std::vector<std::pair<torch::Tensor, torch::Tensor>> nxt;
    for (size_t j = 0; j + 1 < streams.size(); j += 2) {
      auto& A = streams[j];
      auto& B = streams[j + 1];
      int64_t nA = A.first.numel(), nB = B.first.numel();
      auto kO = torch::empty({nA + nB}, i64);
      auto vO = torch::empty({nA + nB}, i64);
      launch_merge_pairs((const uint64_t*)A.first.data_ptr(), (const uint64_t*)A.second.data_ptr(), nA,
                         (const uint64_t*)B.first.data_ptr(), (const uint64_t*)B.second.data_ptr(), nB,
                         (uint64_t*)kO.data_ptr(), (uint64_t*)vO.data_ptr(), stream);
      nxt.emplace_back(kO, vO);
    }
    if (streams.size() % 2) nxt.push_back(streams.back());
    streams = std::move(nxt);
  }
  torch::Tensor keys = streams[0].first;
  torch::Tensor order = streams[0].second;
  int64_t n = keys.numel();

  // ---- dedup keep-last + survivor source indices ----
  auto keep = torch::empty({n}, u8);
  launch_keep_last((const uint64_t*)keys.data_ptr(), keep.data_ptr<uint8_t>(), n, stream);
  auto surv = at::nonzero(keep).view(-1);  // the unit's single device sync
  auto src_idx = order.index({surv});

  // ---- gather every column (concat across files first) ----
  std::vector<torch::Tensor> g_in;
  std::vector<int64_t> g_col;  // read col index per g_in entry (fixed/validity)
  std::vector<int> g_kind;     // 0=data, 1=validity
  for (int64_t c = 0; c < ncols; c++) {
    bool is_str = cols[0][c].is_string;
    bool any_valid = false;
    for (int64_t f = 0; f < nfiles; f++)
      if (cols[f][c].validity.defined()) any_valid = true;
    if (!is_str) {
      std::vector<torch::Tensor> parts;
      int64_t es = da[c][3];
      for (int64_t f = 0; f < nfiles; f++)
        parts.push_back(cols[f][c].data.view({-1, es}));
      auto cat = nfiles == 1 ? parts[0] : at::cat(parts, 0);
      // gather as typed elements (view to es-wide rows then index)
      g_in.push_back(cat);
      g_col.push_back(c);
      g_kind.push_back(0);
    }
    if (any_valid || is_str) {
      std::vector<torch::Tensor> parts;
      for (int64_t f = 0; f < nfiles; f++) {
        auto v = cols[f][c].validity;
        if (v.defined()) parts.push_back(v);
        else parts.push_back(torch::ones({da[f * ncols + c][4]}, u8));
      }
      if (any_valid) {
        g_in.push_back(nfiles == 1 ? parts[0] : at::cat(parts, 0));
        g_col.push_back(c);
        g_kind.push_back(1);
      }
    }
  }
  // fused fixed-width gathers (16 cols per launch)
  std::vector<torch::Tensor> g_out(g_in.size());
  {
    size_t i = 0;
    int64_t nsurv = src_idx.numel();
    while (i < g_in.size()) {
      GatherTable tbl;
      tbl.ncols = 0;
      size_t start = i;
      for (; i < g_in.size() && tbl.ncols < 16; i++) {
        auto& t = g_in[i];
        int64_t es = t.dim() == 2 ? t.size(1) * t.element_size() : t.element_size();
        auto o = torch::empty({nsurv, t.dim() == 2 ? t.size(1) : 1},
                              t.options());
        tbl.src[tbl.ncols] = t.data_ptr();
        tbl.dst[tbl.ncols] = o.data_ptr();
        tbl.esize[tbl.ncols] = (int)es;
        tbl.ncols++;
        g_out[i] = o;
      }
      launch_gather_fixed_multi(tbl, src_idx.data_ptr<int64_t>(), nsurv, stream);
      (void)start;
    }
  }
  // build outputs per column
  std::vector<torch::Tensor> col_data((size_t)ncols), col_valid((size_t)ncols);
  std::vector<std::vector<torch::Tensor>> out_cols;
  for (size_t i = 0; i < g_in.size(); i++) {
    if (g_kind[i] == 0) col_data[(size_t)g_col[i]] = g_out[i].view(-1);
    else col_valid[(size_t)g_col[i]] = g_out[i].view(-1);
  }
  for (int64_t c = 0; c < ncols; c++) {
    std::vector<torch::Tensor> entry;
    if (!cols[0][c].is_string) {
      entry.push_back(col_data[(size_t)c]);
      entry.push_back(col_valid[(size_t)c]);  // may be undefined
    } else {
      // string gather: offsets via cumsum of gathered lens, bytes kernel
      std::vector<torch::Tensor> offs_parts, bytes_parts;
      int64_t byte_base = 0;
      std::vector<torch::Tensor> adj_offs;
      for (int64_t f = 0; f < nfiles; f++) {
        auto o = cols[f][c].offsets;
        adj_offs.push_back((f == 0 ? o : o + byte_base));
        bytes_parts.push_back(cols[f][c].bytes);
        byte_base += cols[f][c].bytes.numel();
      }
      // concatenated offsets: drop leading 0 of subsequent files
      std::vector<torch::Tensor> oparts;
      for (int64_t f = 0; f < nfiles; f++)
        oparts.push_back(f == 0 ? adj_offs[f] : adj_offs[f].narrow(0, 1, adj_offs[f].numel() - 1));
      auto cat_offs = nfiles == 1 ? oparts[0] : at::cat(oparts, 0);
      auto cat_bytes = nfiles == 1 ? bytes_parts[0] : at::cat(bytes_parts, 0);
      auto lens = cat_offs.narrow(0, 1, cat_offs.numel() - 1) -
                  cat_offs.narrow(0, 0, cat_offs.numel() - 1);
      auto sel_lens = lens.index({src_idx});
      int64_t nsurv = src_idx.numel();
      auto new_offs = torch::zeros({nsurv + 1}, i64);
      new_offs.narrow(0, 1, nsurv).copy_(at::cumsum(sel_lens, 0));
      int64_t total = nsurv ? new_offs[nsurv].item<int64_t>() : 0;
      auto new_bytes = torch::empty({total}, u8);
      launch_gather_strings(cat_bytes.data_ptr<uint8_t>(), cat_offs.data_ptr<int64_t>(),
                            src_idx.data_ptr<int64_t>(), new_offs.data_ptr<int64_t>(),
                            new_bytes.data_ptr<uint8_t>(), nsurv, stream);
      entry.push_back(new_offs);
      entry.push_back(new_bytes);
      entry.push_back(col_valid[(size_t)c]);  // may be undefined
    }
    out_cols.push_back(std::move(entry));
  }
  py::gil_scoped_acquire acq;
  py::list out_list;
  for (auto& entry : out_cols) {
    py::list e;
    for (auto& t : entry) {
      if (t.defined()) e.append(t);
      else e.append(py::none());
    }
    out_list.append(e);
  }
  return out_list;
}

// decompress into an existing device buffer (jobs dst offsets index it)
static torch::Tensor snappy_decompress_into(torch::Tensor src, torch::Tensor jobs,
                                            torch::Tensor dst) {
  CHECK_GPU(src);
  CHECK_GPU(jobs);
  CHECK_GPU(dst);
  auto status = torch::empty({jobs.size(0)}, src.options().dtype(torch::kInt32));
  launch_snappy_decompress(src.data_ptr<uint8_t>(), jobs.data_ptr<int64_t>(),
                           jobs.size(0), dst.data_ptr<uint8_t>(),
                           status.data_ptr<int32_t>(), cur_stream());
  return status;
}

// GPU zstd: one wave per page, scratch literals buffer per block slot
static torch::Tensor zstd_decompress_into(torch::Tensor src, torch::Tensor jobs,
                                          torch::Tensor dst) {
  CHECK_GPU(src);
  CHECK_GPU(jobs);
  CHECK_GPU(dst);
  int64_t njobs = jobs.size(0);
  int64_t cap = 2048;
  if (const char* e = std::getenv("LAKESOUL_ZSTD_BLOCKS")) {
    int64_t v = atoll(e);
    if (v > 0) cap = v;
  }
  int64_t nblocks = njobs < cap ? njobs : cap;
  auto scratch = torch::empty({nblocks * lsz_gpu_litbuf_bytes()},
                              src.options().dtype(torch::kUInt8));
  auto status = torch::empty({njobs}, src.options().dtype(torch::kInt32));
  launch_zstd_decompress(src.data_ptr<uint8_t>(), jobs.data_ptr<int64_t>(),
                         njobs, dst.data_ptr<uint8_t>(),
                         scratch.data_ptr<uint8_t>(), nblocks,
                         status.data_ptr<int32_t>(), cur_stream());
  return status;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("ann_scores", &ann_scores);
  m.def("ann_scores_t", &ann_scores_t);
  m.def("zstd_decompress_into", &zstd_decompress_into);
  m.def("str_chunk_keys", [](torch::Tensor offsets, torch::Tensor bytes,
                             int64_t chunk) {
    CHECK_GPU(offsets);
    CHECK_GPU(bytes);
    int64_t n = offsets.numel() - 1;
    auto out = torch::empty({n}, offsets.options().dtype(torch::kInt64));
    launch_str_chunk_keys(offsets.data_ptr<int64_t>(),
                          bytes.data_ptr<uint8_t>(), chunk,
                          out.data_ptr<int64_t>(), n, cur_stream());
    return out;
  });
  m.def("hamming_scores", [](torch::Tensor codes, torch::Tensor qcodes) {
    CHECK_GPU(codes);
    CHECK_GPU(qcodes);
    int64_t n = codes.size(0);
    int nq = (int)qcodes.size(0);
    int words = (int)codes.size(1);
    TORCH_CHECK(words <= 16 && qcodes.size(1) == words);
    auto out = torch::empty({n, nq}, codes.options().dtype(torch::kInt32));
    launch_hamming_scores((const uint64_t*)codes.data_ptr<int64_t>(),
                          (const uint64_t*)qcodes.data_ptr<int64_t>(),
                          out.data_ptr<int32_t>(), n, nq, words, cur_stream());
    return out;
  });
  m.def("fastscan_bit_dot", [](torch::Tensor bits, torch::Tensor q,
                               int64_t dim) {
    // bits (m, w) uint8 cuda; q (nq, dim) f32 cuda -> (m, nq) f32 of
    // <sign_bits, q> (RaBitQ stage-1 FastScan, LDS-LUT kernel)
    CHECK_GPU(bits);
    CHECK_GPU(q);
    TORCH_CHECK(bits.dtype() == torch::kUInt8 && bits.is_contiguous());
    TORCH_CHECK(q.dtype() == torch::kFloat32);
    int64_t m = bits.size(0);
    int w = (int)bits.size(1);
    int nq = (int)q.size(0);
    int g = (int)((dim + 3) / 4);
    TORCH_CHECK(w == (dim + 7) / 8, "bits width mismatch");
    // per-query LUT: (nq, g, 16); entry v = sum_j q[4g+j]*((v>>j)&1)
    auto qp = torch::zeros({nq, (int64_t)g * 4}, q.options());
    qp.slice(1, 0, dim).copy_(q);
    auto qg = qp.view({nq, g, 4});
    static float pat_host[16 * 4];
    static bool pat_init = false;
    if (!pat_init) {
      for (int v = 0; v < 16; v++)
        for (int j = 0; j < 4; j++) pat_host[v * 4 + j] = (float)((v >> j) & 1);
      pat_init = true;
    }
    auto pat = torch::from_blob(pat_host, {16, 4},
                                torch::TensorOptions().dtype(torch::kFloat32))
                   .to(q.device());
    auto lut = torch::matmul(qg, pat.t()).contiguous();  // (nq, g, 16)
    auto out = torch::empty({m, nq}, q.options());
    launch_fastscan_lut(bits.data_ptr<uint8_t>(), lut.data_ptr<float>(),
                        out.data_ptr<float>(), m, nq, w, g, cur_stream());
    return out;
  });
  m.def("fastscan_est", [](torch::Tensor bits, torch::Tensor q, int64_t dim,
                           torch::Tensor f_add, torch::Tensor f_rescale,
                           torch::Tensor cl_of_row, torch::Tensor g_add,
                           torch::Tensor c1_sum_q) {
    // fused RaBitQ stage-1: (nq, n) estimated distances in one pass
    CHECK_GPU(bits);
    CHECK_GPU(q);
    CHECK_GPU(f_add);
    CHECK_GPU(f_rescale);
    CHECK_GPU(cl_of_row);
    CHECK_GPU(g_add);
    CHECK_GPU(c1_sum_q);
    TORCH_CHECK(bits.dtype() == torch::kUInt8 && bits.is_contiguous());
    TORCH_CHECK(cl_of_row.dtype() == torch::kInt32);
    int64_t m = bits.size(0);
    int w = (int)bits.size(1);
    int nq = (int)q.size(0);
    int g = (int)((dim + 3) / 4);
    int kc = (int)g_add.size(1);
    TORCH_CHECK(w == (dim + 7) / 8, "bits width mismatch");
    TORCH_CHECK(g_add.size(0) == nq && c1_sum_q.size(0) == nq);
    TORCH_CHECK(f_add.size(0) == m && f_rescale.size(0) == m &&
                cl_of_row.size(0) == m);
    auto qp = torch::zeros({nq, (int64_t)g * 4}, q.options());
    qp.slice(1, 0, dim).copy_(q);
    auto qg = qp.view({nq, g, 4});
    static float pat_host2[16 * 4];
    static bool pat_init2 = false;
    if (!pat_init2) {
      for (int v = 0; v < 16; v++)
        for (int j = 0; j < 4; j++) pat_host2[v * 4 + j] = (float)((v >> j) & 1);
      pat_init2 = true;
    }
    auto pat = torch::from_blob(pat_host2, {16, 4},
                                torch::TensorOptions().dtype(torch::kFloat32))
                   .to(q.device());
    auto lut = torch::matmul(qg, pat.t()).contiguous();
    auto out = torch::empty({nq, m}, q.options());
    launch_fastscan_est(
        bits.data_ptr<uint8_t>(), lut.data_ptr<float>(),
        f_add.contiguous().data_ptr<float>(),
        f_rescale.contiguous().data_ptr<float>(),
        cl_of_row.contiguous().data_ptr<int32_t>(),
        g_add.contiguous().data_ptr<float>(),
        c1_sum_q.contiguous().data_ptr<float>(), out.data_ptr<float>(), m, nq,
        w, g, kc, cur_stream());
    return out;
  });
  m.def("fastscan_ex_dot", [](torch::Tensor ex, torch::Tensor q, int64_t dim) {
    // ex (m, wn) uint8 nibbles cuda; q (nq, dim) f32 -> (m, nq) f32
    CHECK_GPU(ex);
    CHECK_GPU(q);
    TORCH_CHECK(ex.dtype() == torch::kUInt8 && ex.is_contiguous());
    TORCH_CHECK(q.dtype() == torch::kFloat32);
    int64_t m = ex.size(0);
    int wn = (int)ex.size(1);
    int nq = (int)q.size(0);
    TORCH_CHECK(wn == (dim + 1) / 2, "ex width mismatch");
    auto qc = q.contiguous();
    auto out = torch::empty({m, nq}, q.options());
    launch_fastscan_ex_dot(ex.data_ptr<uint8_t>(), qc.data_ptr<float>(),
                           out.data_ptr<float>(), m, nq, wn, (int)dim,
                           cur_stream());
    return out;
  });
  m.def("fastscan_ex_dot_pairs",
        [](torch::Tensor ex, torch::Tensor cand, torch::Tensor q,
           int64_t dim) {
    // ex (n, wn) uint8 (FULL table); cand (nq, C) int64 row indices
    // (-1 = missing); q (nq, dim) f32 -> (nq, C) f32 per-pair dots
    CHECK_GPU(ex);
    CHECK_GPU(cand);
    CHECK_GPU(q);
    TORCH_CHECK(ex.dtype() == torch::kUInt8 && ex.is_contiguous());
    TORCH_CHECK(cand.dtype() == torch::kInt64);
    TORCH_CHECK(q.dtype() == torch::kFloat32);
    int wn = (int)ex.size(1);
    int nq = (int)cand.size(0);
    int C = (int)cand.size(1);
    TORCH_CHECK(wn == (dim + 1) / 2, "ex width mismatch");
    TORCH_CHECK(q.size(0) == nq, "query count mismatch");
    auto qc = q.contiguous();
    auto cc = cand.contiguous();
    auto out = torch::empty({nq, C}, q.options());
    launch_fastscan_ex_dot_pairs(ex.data_ptr<uint8_t>(),
                                 cc.data_ptr<int64_t>(),
                                 qc.data_ptr<float>(), out.data_ptr<float>(),
                                 C, nq, wn, (int)dim, cur_stream());
    return out;
  });
  m.def("snappy_decompress", &snappy_decompress);
  m.def("snappy_decompress_into", &snappy_decompress_into);
  m.def("scan_unit_uselast", &scan_unit_uselast);
  m.doc() = "lakesoul_amd gfx950 HIP kernels";
  m.def("hash_fixed_column", &hash_fixed_column);
  m.def("hash_string_column", &hash_string_column);
  m.def("bucket_ids", &bucket_ids);
  m.def("rle_expand", &rle_expand);
  m.def("dict_gather_scatter", &dict_gather_scatter);
  m.def("scatter_valid", &scatter_valid);
  m.def("merge_pairs", &merge_pairs);
  m.def("keep_last_mask", &keep_last_mask);
  m.def("group_start_mask", &group_start_mask);
  m.def("pack_key_i64", &pack_key_i64);
  m.def("pack_key_2xi32", &pack_key_2xi32);
  m.def("gather_fixed_multi", &gather_fixed_multi);
  m.def("gather_strings", &gather_strings);
  m.def("bytes_ne_mask", &bytes_ne_mask);
  m.def("segmented_sum", &segmented_sum);
  m.def("segmented_last", &segmented_last);
}
