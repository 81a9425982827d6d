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
void launch_ann_scores(const short*, const short*, float*, int64_t, int32_t, int32_t, hipStream_t);
void launch_snappy_decompress(const uint8_t*, const int64_t*, int64_t, uint8_t*, int32_t*, hipStream_t);

}  // namespace lakesoul

using namespace lakesoul;

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_GPU(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU"); \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

static const uint8_t* opt_u8(const torch::Tensor& t) {
  return t.numel() ? t.data_ptr<uint8_t>() : nullptr;
}

// ---- hashing ---------------------------------------------------------- //

static torch::Tensor hash_fixed_column(torch::Tensor data, torch::Tensor validity,
                                       torch::Tensor prev, bool first) {
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
  CHECK_GPU(hi);
  auto out = torch::empty({hi.numel()}, hi.options().dtype(torch::kInt64));
  launch_pack_key_2xi32(hi.data_ptr<int32_t>(), lo.data_ptr<int32_t>(),
                        (uint64_t*)out.data_ptr(), hi.numel(), cur_stream());
  return out;
}

// ---- gathers ----------------------------------------------------------- //

static std::vector<torch::Tensor> gather_fixed_multi(std::vector<torch::Tensor> cols,
                                                     torch::Tensor idx) {
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

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("ann_scores", &ann_scores);
  m.def("snappy_decompress", &snappy_decompress);
  m.def("snappy_decompress_into", &snappy_decompress_into);
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
