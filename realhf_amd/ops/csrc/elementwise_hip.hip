#include "hip/hip_runtime.h"
// RoPE + SwiGLU + fused AdamW — memory-bound elementwise kernels for gfx950.
// All bf16/f16 I/O is 8-element vectorized (guide G13); trig tables are
// host-precomputed (guide Appendix B: no sinf/cosf on device).
#include "common.h"

// ---------------------------------------------------------------------------
// RoPE: x [total, nh, hd], cos/sin [maxlen, hd/2] fp32, positions [total].
// Non-interleaved (neox/llama): pairs (i, i + hd/2).
// backward = forward with sin negated (conj flag).
// One thread per (token, head) pair-quad: each thread handles 4 pairs.
// ---------------------------------------------------------------------------
template <typename T, bool CONJ>
__global__ void rope_kernel(
    const T* __restrict__ x, T* __restrict__ out,
    const float* __restrict__ cosb, const float* __restrict__ sinb,
    const long* __restrict__ pos, int total, int nh, int hd) {
  int hd2 = hd / 2;
  long n_quads = (long)total * nh * (hd2 / 4);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n_quads;
       i += (long)gridDim.x * blockDim.x) {
    int q = i % (hd2 / 4);
    long th = i / (hd2 / 4);
    int h = th % nh;
    long t = th / nh;
    int d0 = q * 4;
    const T* base = x + (t * nh + h) * (long)hd;
    T* obase = out + (t * nh + h) * (long)hd;
    long p = pos[t];
    float4v c = *(const float4v*)(cosb + p * hd2 + d0);
    float4v s = *(const float4v*)(sinb + p * hd2 + d0);
    short4v x1 = *(const short4v*)((const short*)base + d0);
    short4v x2 = *(const short4v*)((const short*)base + hd2 + d0);
    short o1[4], o2[4];
    #pragma unroll
    for (int j = 0; j < 4; j++) {
      float a = to_f32<T>(((const T*)&x1)[j]);
      float b = to_f32<T>(((const T*)&x2)[j]);
      float sj = CONJ ? -s[j] : s[j];
      ((T*)o1)[j] = from_f32<T>(a * c[j] - b * sj);
      ((T*)o2)[j] = from_f32<T>(b * c[j] + a * sj);
    }
    *(short4v*)((short*)obase + d0) = *(short4v*)o1;
    *(short4v*)((short*)obase + hd2 + d0) = *(short4v*)o2;
  }
}

torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cosb, torch::Tensor sinb,
                       torch::Tensor positions, bool interleaved, bool conj) {
  TORCH_CHECK(!interleaved, "interleaved rotary not yet in the HIP path");
  TORCH_CHECK(x.is_contiguous() && x.dim() == 3);
  TORCH_CHECK(x.element_size() == 2, "rope kernel: bf16/fp16 only");
  int total = x.size(0), nh = x.size(1), hd = x.size(2);
  TORCH_CHECK(hd % 8 == 0);
  auto out = torch::empty_like(x);
  auto pos = positions.to(torch::kLong).contiguous();
  long n = (long)total * nh * (hd / 8);
  int grid = (int)std::min<long>((n + 255) / 256, 8192);
  DISPATCH_BF16_FP16_FP32(x.scalar_type(), "rope", [&] {
    if constexpr (sizeof(scalar_t) == 2) {
      if (conj)
        hipLaunchKernelGGL((rope_kernel<scalar_t, true>), dim3(grid), dim3(256), 0,
          cur_stream(), (const scalar_t*)x.data_ptr(), (scalar_t*)out.data_ptr(),
          cosb.data_ptr<float>(), sinb.data_ptr<float>(), pos.data_ptr<long>(),
          total, nh, hd);
      else
        hipLaunchKernelGGL((rope_kernel<scalar_t, false>), dim3(grid), dim3(256), 0,
          cur_stream(), (const scalar_t*)x.data_ptr(), (scalar_t*)out.data_ptr(),
          cosb.data_ptr<float>(), sinb.data_ptr<float>(), pos.data_ptr<long>(),
          total, nh, hd);
    }
  });
  CHECK_CUDA_OK();
  return out;
}

// ---------------------------------------------------------------------------
// SwiGLU: gate_up [tokens, 2*I] -> out [tokens, I]; out = silu(g) * u
// ---------------------------------------------------------------------------
template <typename T>
__global__ void swiglu_fwd_kernel(const T* __restrict__ gu, T* __restrict__ out,
                                  long tokens, int I) {
  long n = tokens * (I / 8);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    long t = i / (I / 8);
    int d0 = (int)(i % (I / 8)) * 8;
    const T* g = gu + t * (2L * I) + d0;
    const T* u = g + I;
    short8 gv = *(const short8*)g;
    short8 uv = *(const short8*)u;
    short o[8];
    #pragma unroll
    for (int j = 0; j < 8; j++) {
      float gf = to_f32<T>(((const T*)&gv)[j]);
      float uf = to_f32<T>(((const T*)&uv)[j]);
      float s = gf / (1.f + __expf(-gf));
      ((T*)o)[j] = from_f32<T>(s * uf);
    }
    *(short8*)(out + t * (long)I + d0) = *(short8*)o;
  }
}

// slab variant: gate_up comes in as fp32 split-K partial slabs
// [nks, tokens, 2I] from skinny_gemm_nc (launch-boundary reduce)
template <typename T>
__global__ void swiglu_slab_kernel(const float* __restrict__ parts, int nks,
                                   T* __restrict__ out, long tokens, int I) {
  const long sstride = tokens * (2L * I);
  long n = tokens * (I / 8);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    long t = i / (I / 8);
    int d0 = (int)(i % (I / 8)) * 8;
    const float* g = parts + t * (2L * I) + d0;
    const float* u = g + I;
    float gf[8], uf[8];
    #pragma unroll
    for (int j = 0; j < 8; j += 4) {
      *(float4v*)(gf + j) = *(const float4v*)(g + j);
      *(float4v*)(uf + j) = *(const float4v*)(u + j);
    }
    for (int s = 1; s < nks; s++) {
      #pragma unroll
      for (int j = 0; j < 8; j += 4) {
        float4v pg = *(const float4v*)(g + (long)s * sstride + j);
        float4v pu = *(const float4v*)(u + (long)s * sstride + j);
        #pragma unroll
        for (int q = 0; q < 4; q++) { gf[j + q] += pg[q]; uf[j + q] += pu[q]; }
      }
    }
    short o[8];
    #pragma unroll
    for (int j = 0; j < 8; j++) {
      float sg = gf[j] / (1.f + __expf(-gf[j]));
      ((T*)o)[j] = from_f32<T>(sg * uf[j]);
    }
    *(short8*)(out + t * (long)I + d0) = *(short8*)o;
  }
}

template <typename T>
__global__ void swiglu_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ gu,
                                  T* __restrict__ dgu, long tokens, int I) {
  long n = tokens * (I / 8);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    long t = i / (I / 8);
    int d0 = (int)(i % (I / 8)) * 8;
    const T* g = gu + t * (2L * I) + d0;
    const T* u = g + I;
    short8 gv = *(const short8*)g;
    short8 uv = *(const short8*)u;
    short8 dv = *(const short8*)(dy + t * (long)I + d0);
    short dg[8], du[8];
    #pragma unroll
    for (int j = 0; j < 8; j++) {
      float gf = to_f32<T>(((const T*)&gv)[j]);
      float uf = to_f32<T>(((const T*)&uv)[j]);
      float d = to_f32<T>(((const T*)&dv)[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float s = gf * sig;
      ((T*)dg)[j] = from_f32<T>(d * uf * (sig + s * (1.f - sig)));
      ((T*)du)[j] = from_f32<T>(d * s);
    }
    *(short8*)(dgu + t * (2L * I) + d0) = *(short8*)dg;
    *(short8*)(dgu + t * (2L * I) + I + d0) = *(short8*)du;
  }
}

torch::Tensor swiglu_fwd(torch::Tensor gu) {
  TORCH_CHECK(gu.is_contiguous());
  int I2 = gu.size(-1);
  TORCH_CHECK(I2 % 16 == 0);
  if (gu.dim() == 3 && gu.scalar_type() == torch::kFloat) {
    // fp32 split-K partial slabs [nks, tokens, 2I] (launch-boundary reduce)
    long tokens = gu.size(1);
    int I = I2 / 2;
    auto out = torch::empty({tokens, (long)I},
                            gu.options().dtype(torch::kBFloat16));
    long n = tokens * (I / 8);
    int grid = (int)std::min<long>((n + 255) / 256, 8192);
    hipLaunchKernelGGL((swiglu_slab_kernel<bf16>), dim3(grid), dim3(256), 0,
      cur_stream(), gu.data_ptr<float>(), (int)gu.size(0),
      (bf16*)out.data_ptr(), tokens, I);
    CHECK_CUDA_OK();
    return out;
  }
  TORCH_CHECK(gu.element_size() == 2);
  long tokens = gu.numel() / I2;
  int I = I2 / 2;
  auto out = torch::empty({tokens, I}, gu.options());
  long n = tokens * (I / 8);
  int grid = (int)std::min<long>((n + 255) / 256, 8192);
  DISPATCH_BF16_FP16_FP32(gu.scalar_type(), "swiglu", [&] {
    if constexpr (sizeof(scalar_t) == 2)
      hipLaunchKernelGGL((swiglu_fwd_kernel<scalar_t>), dim3(grid), dim3(256), 0,
        cur_stream(), (const scalar_t*)gu.data_ptr(), (scalar_t*)out.data_ptr(),
        tokens, I);
  });
  CHECK_CUDA_OK();
  return out;
}

torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor gu) {
  int I2 = gu.size(-1);
  long tokens = gu.numel() / I2;
  int I = I2 / 2;
  auto dgu = torch::empty_like(gu);
  long n = tokens * (I / 8);
  int grid = (int)std::min<long>((n + 255) / 256, 8192);
  DISPATCH_BF16_FP16_FP32(gu.scalar_type(), "swiglu_bwd", [&] {
    if constexpr (sizeof(scalar_t) == 2)
      hipLaunchKernelGGL((swiglu_bwd_kernel<scalar_t>), dim3(grid), dim3(256), 0,
        cur_stream(), (const scalar_t*)dy.data_ptr(), (const scalar_t*)gu.data_ptr(),
        (scalar_t*)dgu.data_ptr(), tokens, I);
  });
  CHECK_CUDA_OK();
  return dgu;
}

// ---------------------------------------------------------------------------
// Fused AdamW on the flat fp32 master shard, optional bf16 write-back.
// Replaces apex fused adam (reference Megatron dep).
// ---------------------------------------------------------------------------
template <typename TG>
__global__ void fused_adamw_kernel(
    float* __restrict__ p, const TG* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v, bf16* __restrict__ out_bf16,
    long n, float lr, float b1, float b2, float eps, float wd,
    float bc1, float bc2, float gscale, bool write_bf16) {
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4; i < n;
       i += (long)gridDim.x * blockDim.x * 4) {
    float4v pv = *(float4v*)(p + i);
    TG gv4[4];
    *(short4v*)gv4 = *(const short4v*)((const short*)(g + i));  // 8B for bf16
    float4v mv = *(float4v*)(m + i);
    float4v vv = *(float4v*)(v + i);
    short ob[4];
    #pragma unroll
    for (int j = 0; j < 4; j++) {
      float gj = to_f32<TG>(gv4[j]) * gscale;
      mv[j] = b1 * mv[j] + (1.f - b1) * gj;
      vv[j] = b2 * vv[j] + (1.f - b2) * gj * gj;
      float denom = sqrtf(vv[j] / bc2) + eps;
      pv[j] = pv[j] * (1.f - lr * wd) - lr / bc1 * mv[j] / denom;
      ob[j] = f2bf(pv[j]);
    }
    *(float4v*)(p + i) = pv;
    *(float4v*)(m + i) = mv;
    *(float4v*)(v + i) = vv;
    if (write_bf16) *(short4v*)((short*)out_bf16 + i) = *(short4v*)ob;
  }
}

// fp32-grad specialization needs 16B loads
template <>
__global__ void fused_adamw_kernel<float>(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v, bf16* __restrict__ out_bf16,
    long n, float lr, float b1, float b2, float eps, float wd,
    float bc1, float bc2, float gscale, bool write_bf16) {
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4; i < n;
       i += (long)gridDim.x * blockDim.x * 4) {
    float4v pv = *(float4v*)(p + i);
    float4v gv = *(const float4v*)(g + i);
    float4v mv = *(float4v*)(m + i);
    float4v vv = *(float4v*)(v + i);
    short ob[4];
    #pragma unroll
    for (int j = 0; j < 4; j++) {
      float gj = gv[j] * gscale;
      mv[j] = b1 * mv[j] + (1.f - b1) * gj;
      vv[j] = b2 * vv[j] + (1.f - b2) * gj * gj;
      float denom = sqrtf(vv[j] / bc2) + eps;
      pv[j] = pv[j] * (1.f - lr * wd) - lr / bc1 * mv[j] / denom;
      ob[j] = f2bf(pv[j]);
    }
    *(float4v*)(p + i) = pv;
    *(float4v*)(m + i) = mv;
    *(float4v*)(v + i) = vv;
    if (write_bf16) *(short4v*)((short*)out_bf16 + i) = *(short4v*)ob;
  }
}

void fused_adamw(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                 torch::Tensor v, torch::Tensor out_bf16, double lr, double b1,
                 double b2, double eps, double wd, long step, double gscale,
                 bool write_bf16) {
  long n = p.numel();
  TORCH_CHECK(n % 4 == 0, "flat shard must be 4-aligned");
  TORCH_CHECK(p.is_cuda() && g.is_cuda());
  float bc1 = 1.f - powf((float)b1, (float)step);
  float bc2 = 1.f - powf((float)b2, (float)step);
  int grid = (int)std::min<long>((n / 4 + 255) / 256, 4096);
  if (g.scalar_type() == torch::kFloat) {
    hipLaunchKernelGGL(fused_adamw_kernel<float>, dim3(grid), dim3(256), 0,
      cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
      m.data_ptr<float>(), v.data_ptr<float>(),
      write_bf16 ? (bf16*)out_bf16.data_ptr() : nullptr,
      n, (float)lr, (float)b1, (float)b2, (float)eps, (float)wd, bc1, bc2,
      (float)gscale, write_bf16);
  } else {
    TORCH_CHECK(g.scalar_type() == torch::kBFloat16);
    hipLaunchKernelGGL(fused_adamw_kernel<bf16>, dim3(grid), dim3(256), 0,
      cur_stream(), p.data_ptr<float>(), (const bf16*)g.data_ptr(),
      m.data_ptr<float>(), v.data_ptr<float>(),
      write_bf16 ? (bf16*)out_bf16.data_ptr() : nullptr,
      n, (float)lr, (float)b1, (float)b2, (float)eps, (float)wd, bc1, bc2,
      (float)gscale, write_bf16);
  }
  CHECK_CUDA_OK();
}
