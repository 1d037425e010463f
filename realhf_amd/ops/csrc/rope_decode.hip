// Fused decode-path epilogue: QKV split + optional bias + RoPE(q,k) +
// KV-cache append, one kernel.
// Replaces ~8 small per-layer kernels in the captured decode graph
// (2x contiguous-copy, 2x rope, 2x index_put, arange, clamp) — each tiny
// kernel costs ~1.5-5us of dispatch floor (guide: boundary row), which at
// 32 layers x 512 tokens dominated the glue time.
//
// SLAB mode: qkv comes in as the skinny GEMM's fp32 split-K partial
// slabs [nks, bs, qkvd]; the per-element sum over nks happens in this
// kernel's prologue (launch-boundary reduce) instead of a separate
// sg_combine launch.
#include "common.h"

template <bool SLAB>
__global__ void rope_qkv_decode_kernel(
    const void* __restrict__ qkv_in,  // bf16 [bs, qkvd] | fp32 [nks, bs, qkvd]
    const bf16* __restrict__ bias,  // [(nq+2nkv)*hd] or null
    bf16* __restrict__ q_out,  // [bs, nq, hd]
    bf16* __restrict__ kcache,  // [bs, maxlen, nkv, hd]
    bf16* __restrict__ vcache,
    const int* __restrict__ cache_seqlens,  // [bs]
    const float* __restrict__ cosb, const float* __restrict__ sinb,  // [*, hd/2]
    int bs, int nq, int nkv, int hd, long maxlen, long qkv_stride,
    bool apply_rope, int nks) {
  const int nh = nq + 2 * nkv;
  const int hd2 = hd / 2;
  const long nwork = (long)bs * nh * (hd2 / 4);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nwork;
       i += (long)gridDim.x * blockDim.x) {
    const int quad = i % (hd2 / 4);
    const int h = (i / (hd2 / 4)) % nh;
    const int b = i / ((long)nh * (hd2 / 4));
    const int d0 = quad * 4;
    const int pos = max(cache_seqlens[b] - 1, 0);

    float4v f1, f2;
    if constexpr (SLAB) {
      const float* src =
          (const float*)qkv_in + (long)b * qkv_stride + (long)h * hd;
      const long sstride = (long)bs * qkv_stride;
      f1 = *(const float4v*)(src + d0);
      f2 = *(const float4v*)(src + hd2 + d0);
      for (int s = 1; s < nks; s++) {
        float4v p1 = *(const float4v*)(src + (long)s * sstride + d0);
        float4v p2 = *(const float4v*)(src + (long)s * sstride + hd2 + d0);
        #pragma unroll
        for (int j = 0; j < 4; j++) { f1[j] += p1[j]; f2[j] += p2[j]; }
      }
    } else {
      const bf16* src =
          (const bf16*)qkv_in + (long)b * qkv_stride + (long)h * hd;
      short4v x1 = *(const short4v*)((const short*)src + d0);
      short4v x2 = *(const short4v*)((const short*)src + hd2 + d0);
      #pragma unroll
      for (int j = 0; j < 4; j++) {
        f1[j] = __bfloat162float(((const bf16*)&x1)[j]);
        f2[j] = __bfloat162float(((const bf16*)&x2)[j]);
      }
    }
    if (bias) {
      short4v b1 = *(const short4v*)((const short*)bias + (long)h * hd + d0);
      short4v b2 = *(const short4v*)((const short*)bias + (long)h * hd + hd2 + d0);
      #pragma unroll
      for (int j = 0; j < 4; j++) {
        f1[j] += __bfloat162float(((const bf16*)&b1)[j]);
        f2[j] += __bfloat162float(((const bf16*)&b2)[j]);
      }
    }
    bf16* dst;
    bool rope = apply_rope;
    if (h < nq) {
      dst = q_out + ((long)b * nq + h) * hd;
    } else if (h < nq + nkv) {
      dst = kcache + (((long)b * maxlen + pos) * nkv + (h - nq)) * hd;
    } else {
      dst = vcache + (((long)b * maxlen + pos) * nkv + (h - nq - nkv)) * hd;
      rope = false;
    }
    short o1[4], o2[4];
    if (rope) {
      float4v c = *(const float4v*)(cosb + (long)pos * hd2 + d0);
      float4v s = *(const float4v*)(sinb + (long)pos * hd2 + d0);
      #pragma unroll
      for (int j = 0; j < 4; j++) {
        ((bf16*)o1)[j] = __float2bfloat16(f1[j] * c[j] - f2[j] * s[j]);
        ((bf16*)o2)[j] = __float2bfloat16(f2[j] * c[j] + f1[j] * s[j]);
      }
    } else {
      #pragma unroll
      for (int j = 0; j < 4; j++) {
        ((bf16*)o1)[j] = __float2bfloat16(f1[j]);
        ((bf16*)o2)[j] = __float2bfloat16(f2[j]);
      }
    }
    *(short4v*)((short*)dst + d0) = *(short4v*)o1;
    *(short4v*)((short*)dst + hd2 + d0) = *(short4v*)o2;
  }
}

torch::Tensor rope_qkv_decode(
    torch::Tensor qkv, c10::optional<torch::Tensor> bias, torch::Tensor kcache,
    torch::Tensor vcache, torch::Tensor cache_seqlens, torch::Tensor cosb,
    torch::Tensor sinb, long nq, bool apply_rope) {
  TORCH_CHECK(qkv.is_cuda());
  const bool slab = qkv.dim() == 3;  // fp32 split-K partials [nks, bs, qkvd]
  if (slab) {
    TORCH_CHECK(qkv.scalar_type() == torch::kFloat && qkv.is_contiguous());
  } else {
    TORCH_CHECK(qkv.scalar_type() == torch::kBFloat16 && qkv.dim() == 2 &&
                qkv.stride(1) == 1);
  }
  int nks = slab ? qkv.size(0) : 1;
  int bs = slab ? qkv.size(1) : qkv.size(0);
  int nkv = kcache.size(2);
  int hd = kcache.size(3);
  long maxlen = kcache.size(1);
  TORCH_CHECK(qkv.size(-1) == (nq + 2 * nkv) * hd);
  TORCH_CHECK(hd % 8 == 0);
  auto q_out = torch::empty({bs, (long)nq, (long)hd},
                            kcache.options().dtype(torch::kBFloat16));
  long nwork = (long)bs * (nq + 2 * nkv) * (hd / 8);
  int grid = (int)std::min<long>((nwork + 255) / 256, 4096);
  const bf16* bptr = nullptr;
  if (bias.has_value()) bptr = (const bf16*)bias->data_ptr();
  long qkv_stride = slab ? qkv.stride(1) : qkv.stride(0);
  if (slab) {
    hipLaunchKernelGGL((rope_qkv_decode_kernel<true>), dim3(grid), dim3(256), 0,
      cur_stream(), qkv.data_ptr(), bptr,
      (bf16*)q_out.data_ptr(), (bf16*)kcache.data_ptr(),
      (bf16*)vcache.data_ptr(), cache_seqlens.data_ptr<int>(),
      cosb.data_ptr<float>(), sinb.data_ptr<float>(), bs, (int)nq, nkv, hd,
      maxlen, qkv_stride, apply_rope, nks);
  } else {
    hipLaunchKernelGGL((rope_qkv_decode_kernel<false>), dim3(grid), dim3(256), 0,
      cur_stream(), qkv.data_ptr(), bptr,
      (bf16*)q_out.data_ptr(), (bf16*)kcache.data_ptr(),
      (bf16*)vcache.data_ptr(), cache_seqlens.data_ptr<int>(),
      cosb.data_ptr<float>(), sinb.data_ptr<float>(), bs, (int)nq, nkv, hd,
      maxlen, qkv_stride, apply_rope, 1);
  }
  CHECK_CUDA_OK();
  return q_out;
}
