// Fused RMSNorm forward/backward for gfx950.
// HBM-bound: vectorized 16-byte loads (guide G13: scalar bf16 is 2x slower).
// One 256-thread workgroup per row (hidden <= 16384), grid-stride over rows.
// Replaces the reference's TransformerEngine RMSNorm (SURVEY.md §2.2 ext deps).
#include "common.h"

template <typename T, int VEC, bool RESID = false>
__global__ void rmsnorm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, T* __restrict__ out,
    float* __restrict__ rstd, int rows, int H, float eps,
    const T* __restrict__ resid = nullptr, T* __restrict__ sum_out = nullptr) {
  __shared__ float red[8];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    T* yr = out + (long)row * H;
    float ss = 0.f;
    for (int i = threadIdx.x * VEC; i < H; i += blockDim.x * VEC) {
      T v[VEC];
      *(float4v*)v = *(const float4v*)(xr + i);  // 16B when VEC matches
      if constexpr (RESID) {
        T r[VEC];
        *(float4v*)r = *(const float4v*)(resid + (long)row * H + i);
        #pragma unroll
        for (int j = 0; j < VEC; j++)
          v[j] = from_f32<T>(to_f32<T>(v[j]) + to_f32<T>(r[j]));
        *(float4v*)(sum_out + (long)row * H + i) = *(float4v*)v;
      }
      #pragma unroll
      for (int j = 0; j < VEC; j++) { float f = to_f32<T>(v[j]); ss += f * f; }
    }
    ss = wave_sum(ss);
    int wid = threadIdx.x / WAVE;
    int nw = blockDim.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) red[wid] = ss;
    __syncthreads();
    if (threadIdx.x < 8) {
      float v2 = (threadIdx.x < nw) ? red[threadIdx.x] : 0.f;
      for (int off = 4; off > 0; off >>= 1) v2 += __shfl_down(v2, off, 64);
      if (threadIdx.x == 0) red[0] = v2;
    }
    __syncthreads();
    float rs = rsqrtf(red[0] / H + eps);
    if (threadIdx.x == 0 && rstd) rstd[row] = rs;
    const T* src2 = RESID ? (sum_out + (long)row * H) : xr;
    for (int i = threadIdx.x * VEC; i < H; i += blockDim.x * VEC) {
      T v[VEC], wv[VEC], o[VEC];
      *(float4v*)v = *(const float4v*)(src2 + i);
      *(float4v*)wv = *(const float4v*)(w + i);
      #pragma unroll
      for (int j = 0; j < VEC; j++)
        o[j] = from_f32<T>(to_f32<T>(v[j]) * rs * to_f32<T>(wv[j]));
      *(float4v*)(yr + i) = *(float4v*)o;
    }
    __syncthreads();
  }
}

// slab variant of the fused add+rmsnorm: x comes in as fp32 split-K
// partial slabs [nks, rows, H] from skinny_gemm_nc (launch-boundary
// reduce); sum_out = sum_s(parts) + resid, out = rmsnorm(sum_out).
template <typename T, int VEC>
__global__ void add_rmsnorm_slab_kernel(
    const float* __restrict__ parts, int nks, const T* __restrict__ resid,
    const T* __restrict__ w, T* __restrict__ out, T* __restrict__ sum_out,
    int rows, int H, float eps) {
  __shared__ float red[8];
  const long sstride = (long)rows * H;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    T* yr = out + (long)row * H;
    float ss = 0.f;
    for (int i = threadIdx.x * VEC; i < H; i += blockDim.x * VEC) {
      float f[VEC];
      const float* pr = parts + (long)row * H + i;
      #pragma unroll
      for (int j = 0; j < VEC; j += 4)
        *(float4v*)(f + j) = *(const float4v*)(pr + j);
      for (int s = 1; s < nks; s++) {
        #pragma unroll
        for (int j = 0; j < VEC; j += 4) {
          float4v p = *(const float4v*)(pr + (long)s * sstride + j);
          #pragma unroll
          for (int q = 0; q < 4; q++) f[j + q] += p[q];
        }
      }
      T r[VEC], v[VEC];
      *(float4v*)r = *(const float4v*)(resid + (long)row * H + i);
      #pragma unroll
      for (int j = 0; j < VEC; j++) {
        float fv = f[j] + to_f32<T>(r[j]);
        v[j] = from_f32<T>(fv);
        float fq = to_f32<T>(v[j]);
        ss += fq * fq;
      }
      *(float4v*)(sum_out + (long)row * H + i) = *(float4v*)v;
    }
    ss = wave_sum(ss);
    int wid = threadIdx.x / WAVE;
    int nw = blockDim.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) red[wid] = ss;
    __syncthreads();
    if (threadIdx.x < 8) {
      float v2 = (threadIdx.x < nw) ? red[threadIdx.x] : 0.f;
      for (int off = 4; off > 0; off >>= 1) v2 += __shfl_down(v2, off, 64);
      if (threadIdx.x == 0) red[0] = v2;
    }
    __syncthreads();
    float rs = rsqrtf(red[0] / H + eps);
    const T* src2 = sum_out + (long)row * H;
    for (int i = threadIdx.x * VEC; i < H; i += blockDim.x * VEC) {
      T v[VEC], wv[VEC], o[VEC];
      *(float4v*)v = *(const float4v*)(src2 + i);
      *(float4v*)wv = *(const float4v*)(w + i);
      #pragma unroll
      for (int j = 0; j < VEC; j++)
        o[j] = from_f32<T>(to_f32<T>(v[j]) * rs * to_f32<T>(wv[j]));
      *(float4v*)(yr + i) = *(float4v*)o;
    }
    __syncthreads();
  }
}

// backward: dx = rs * w * dy - rs^3/H * x * sum(dy * w * x)
// dw: per-thread register partials over this block's rows (each thread owns
// fixed columns), ONE atomicAdd per column per block at the end.
#define RMS_MAX_COLS 64  // supports H <= 256 threads * VEC * (64/VEC)
template <typename T, int VEC>
__global__ void rmsnorm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ rstd, T* __restrict__ dx,
    float* __restrict__ dw, int rows, int H) {
  __shared__ float red[8];
  float dwacc[RMS_MAX_COLS];
  #pragma unroll
  for (int j = 0; j < RMS_MAX_COLS; j++) dwacc[j] = 0.f;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + (long)row * H;
    const T* xr = x + (long)row * H;
    T* dxr = dx + (long)row * H;
    float rs = rstd[row];
    float dot = 0.f;
    for (int i = threadIdx.x * VEC; i < H; i += blockDim.x * VEC) {
      T a[VEC], b[VEC], c[VEC];
      *(float4v*)a = *(const float4v*)(dyr + i);
      *(float4v*)b = *(const float4v*)(w + i);
      *(float4v*)c = *(const float4v*)(xr + i);
      #pragma unroll
      for (int j = 0; j < VEC; j++)
        dot += to_f32<T>(a[j]) * to_f32<T>(b[j]) * to_f32<T>(c[j]);
    }
    dot = wave_sum(dot);
    int wid = threadIdx.x / WAVE, nw = blockDim.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) red[wid] = dot;
    __syncthreads();
    if (threadIdx.x < 8) {
      float v2 = (threadIdx.x < nw) ? red[threadIdx.x] : 0.f;
      for (int off = 4; off > 0; off >>= 1) v2 += __shfl_down(v2, off, 64);
      if (threadIdx.x == 0) red[0] = v2;
    }
    __syncthreads();
    float k = red[0] * rs * rs * rs / H;
    int slot = 0;
    for (int i = threadIdx.x * VEC; i < H; i += blockDim.x * VEC, slot += VEC) {
      T a[VEC], b[VEC], c[VEC], o[VEC];
      *(float4v*)a = *(const float4v*)(dyr + i);
      *(float4v*)b = *(const float4v*)(w + i);
      *(float4v*)c = *(const float4v*)(xr + i);
      #pragma unroll
      for (int j = 0; j < VEC; j++) {
        float g = to_f32<T>(a[j]);
        float xv = to_f32<T>(c[j]);
        o[j] = from_f32<T>(rs * to_f32<T>(b[j]) * g - k * xv);
        if (slot + j < RMS_MAX_COLS) dwacc[slot + j] += g * xv * rs;
      }
      *(float4v*)(dxr + i) = *(float4v*)o;
    }
    __syncthreads();
  }
  // one partial row per block — no atomics (2048 blocks contending on
  // 4096 floats serialized ~500x; partials + torch sum is ~30x faster)
  int slot = 0;
  float* dwrow = dw + (long)blockIdx.x * H;
  for (int i = threadIdx.x * VEC; i < H; i += blockDim.x * VEC, slot += VEC) {
    #pragma unroll
    for (int j = 0; j < VEC; j++)
      if (slot + j < RMS_MAX_COLS) dwrow[i + j] = dwacc[slot + j];
  }
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  int H = x.size(-1);
  long rows = x.numel() / H;
  auto out = torch::empty_like(x);
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat));
  int vec = 16 / x.element_size();
  TORCH_CHECK(H % vec == 0, "hidden dim must be divisible by ", vec);
  int grid = (int)std::min<long>(rows, 2048);
  DISPATCH_BF16_FP16_FP32(x.scalar_type(), "rmsnorm_fwd", [&] {
    if (x.element_size() == 2) {
      hipLaunchKernelGGL((rmsnorm_fwd_kernel<scalar_t, 8>), dim3(grid), dim3(256), 0,
        cur_stream(), (const scalar_t*)x.data_ptr(), (const scalar_t*)w.data_ptr(),
        (scalar_t*)out.data_ptr(), rstd.data_ptr<float>(), (int)rows, H, (float)eps);
    } else {
      hipLaunchKernelGGL((rmsnorm_fwd_kernel<scalar_t, 4>), dim3(grid), dim3(256), 0,
        cur_stream(), (const scalar_t*)x.data_ptr(), (const scalar_t*)w.data_ptr(),
        (scalar_t*)out.data_ptr(), rstd.data_ptr<float>(), (int)rows, H, (float)eps);
    }
  });
  CHECK_CUDA_OK();
  return {out, rstd};
}

std::vector<torch::Tensor> add_rmsnorm_fwd(torch::Tensor x, torch::Tensor resid,
                                           torch::Tensor w, double eps) {
  // out_norm = rmsnorm(x + resid); sum_out = x + resid (the new residual).
  // x may also be fp32 split-K partial slabs [nks, rows, H] from
  // skinny_gemm_nc — then x := sum over slabs (launch-boundary reduce).
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && resid.is_contiguous());
  int H = x.size(-1);
  long rows = resid.numel() / H;
  TORCH_CHECK(H % 8 == 0);
  auto out = torch::empty_like(resid);
  auto sum_out = torch::empty_like(resid);
  int grid = (int)std::min<long>(rows, 2048);
  if (x.dim() == 3 && x.scalar_type() == torch::kFloat) {
    TORCH_CHECK(resid.scalar_type() == torch::kBFloat16);
    TORCH_CHECK(x.size(1) * x.size(2) == rows * H);
    hipLaunchKernelGGL((add_rmsnorm_slab_kernel<bf16, 8>), dim3(grid),
      dim3(256), 0, cur_stream(), x.data_ptr<float>(), (int)x.size(0),
      (const bf16*)resid.data_ptr(), (const bf16*)w.data_ptr(),
      (bf16*)out.data_ptr(), (bf16*)sum_out.data_ptr(), (int)rows, H,
      (float)eps);
    CHECK_CUDA_OK();
    return {out, sum_out};
  }
  TORCH_CHECK(x.element_size() == 2);
  DISPATCH_BF16_FP16_FP32(x.scalar_type(), "add_rmsnorm", [&] {
    if constexpr (sizeof(scalar_t) == 2) {
      hipLaunchKernelGGL((rmsnorm_fwd_kernel<scalar_t, 8, true>), dim3(grid),
        dim3(256), 0, cur_stream(), (const scalar_t*)x.data_ptr(),
        (const scalar_t*)w.data_ptr(), (scalar_t*)out.data_ptr(),
        nullptr, (int)rows, H, (float)eps,
        (const scalar_t*)resid.data_ptr(), (scalar_t*)sum_out.data_ptr());
    }
  });
  CHECK_CUDA_OK();
  return {out, sum_out};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  int H = x.size(-1);
  TORCH_CHECK(H <= 256 * RMS_MAX_COLS, "rmsnorm_bwd supports hidden <= 16384");
  long rows = x.numel() / H;
  auto dx = torch::empty_like(x);
  int grid = (int)std::min<long>(rows, 512);
  auto dw32 = torch::zeros({grid, H}, x.options().dtype(torch::kFloat));
  DISPATCH_BF16_FP16_FP32(x.scalar_type(), "rmsnorm_bwd", [&] {
    if (x.element_size() == 2) {
      hipLaunchKernelGGL((rmsnorm_bwd_kernel<scalar_t, 8>), dim3(grid), dim3(256), 0,
        cur_stream(), (const scalar_t*)dy.data_ptr(), (const scalar_t*)x.data_ptr(),
        (const scalar_t*)w.data_ptr(), rstd.data_ptr<float>(),
        (scalar_t*)dx.data_ptr(), dw32.data_ptr<float>(), (int)rows, H);
    } else {
      hipLaunchKernelGGL((rmsnorm_bwd_kernel<scalar_t, 4>), dim3(grid), dim3(256), 0,
        cur_stream(), (const scalar_t*)dy.data_ptr(), (const scalar_t*)x.data_ptr(),
        (const scalar_t*)w.data_ptr(), rstd.data_ptr<float>(),
        (scalar_t*)dx.data_ptr(), dw32.data_ptr<float>(), (int)rows, H);
    }
  });
  CHECK_CUDA_OK();
  return {dx, dw32.sum(0).to(x.scalar_type())};
}
