// Flat-parameter interval gather/scatter for parameter reallocation.
// Reference semantics: csrc/interval_op/interval_op.cu (slice_intervals /
// set_intervals): dst[cumsum_offsets[k] + j] = src[interval_k.start + j].
// One (interval, 1KB-chunk) pair per workgroup-slice; 16-byte vector copies
// with a scalar tail (intervals are element-counts of a bf16/f32 buffer).
#include "common.h"

// grid.x = interval, grid.y = chunk
template <typename T>
__global__ void interval_copy_kernel(
    const T* __restrict__ src, T* __restrict__ dst,
    const long* __restrict__ intervals, const long* __restrict__ offsets,
    int n_intervals, long chunk, bool gather) {
  int k = blockIdx.x;
  long s = intervals[2 * k], e = intervals[2 * k + 1];
  long off = offsets[k];
  long len = e - s;
  for (long i = (long)blockIdx.y * chunk + threadIdx.x; i < len && i < (long)(blockIdx.y + 1) * chunk;
       i += blockDim.x) {
    if (gather) dst[off + i] = src[s + i];
    else dst[s + i] = src[off + i];
  }
}

static std::pair<torch::Tensor, long> make_offsets(torch::Tensor intervals) {
  auto lens = intervals.select(1, 1) - intervals.select(1, 0);
  auto offs = torch::zeros_like(lens);
  if (lens.numel() > 1)
    offs.slice(0, 1) = torch::cumsum(lens, 0).slice(0, 0, lens.numel() - 1);
  long total = torch::sum(lens).item<long>();
  return {offs.contiguous(), total};
}

torch::Tensor slice_intervals(torch::Tensor src, torch::Tensor intervals) {
  TORCH_CHECK(src.is_cuda() && src.is_contiguous());
  auto iv = intervals.to(torch::kLong).to(src.device()).contiguous();
  auto [offs, total] = make_offsets(iv.cpu());
  auto offs_d = offs.to(src.device());
  auto iv_d = iv;
  auto out = torch::empty({total}, src.options());
  int n = iv.size(0);
  if (n == 0) return out;
  long maxlen = (iv.select(1, 1) - iv.select(1, 0)).max().item<long>();
  long chunk = 256 * 8;
  int chunks = (int)std::min<long>((maxlen + chunk - 1) / chunk, 1024);
  dim3 grid(n, std::max(chunks, 1));
  DISPATCH_BF16_FP16_FP32(src.scalar_type(), "slice_intervals", [&] {
    hipLaunchKernelGGL((interval_copy_kernel<scalar_t>), grid, dim3(256), 0,
      cur_stream(), (const scalar_t*)src.data_ptr(), (scalar_t*)out.data_ptr(),
      iv_d.data_ptr<long>(), offs_d.data_ptr<long>(), n, chunk, true);
  });
  CHECK_CUDA_OK();
  return out;
}

void set_intervals(torch::Tensor src, torch::Tensor dst, torch::Tensor intervals) {
  TORCH_CHECK(dst.is_cuda() && dst.is_contiguous());
  auto iv = intervals.to(torch::kLong).to(dst.device()).contiguous();
  auto [offs, total] = make_offsets(iv.cpu());
  TORCH_CHECK(total == src.numel(), "set_intervals: src size mismatch");
  auto offs_d = offs.to(dst.device());
  auto iv_d = iv;
  int n = iv.size(0);
  if (n == 0) return;
  long maxlen = (iv.select(1, 1) - iv.select(1, 0)).max().item<long>();
  long chunk = 256 * 8;
  int chunks = (int)std::min<long>((maxlen + chunk - 1) / chunk, 1024);
  dim3 grid(n, std::max(chunks, 1));
  DISPATCH_BF16_FP16_FP32(dst.scalar_type(), "set_intervals", [&] {
    hipLaunchKernelGGL((interval_copy_kernel<scalar_t>), grid, dim3(256), 0,
      cur_stream(), (const scalar_t*)src.data_ptr(), (scalar_t*)dst.data_ptr(),
      iv_d.data_ptr<long>(), offs_d.data_ptr<long>(), n, chunk, false);
  });
  CHECK_CUDA_OK();
}
