// Fused decode-path epilogue: QKV split + optional bias + RoPE(q,k) +
// KV-cache append, one kernel.
// Replaces ~8 small per-layer kernels in the captured decode graph
// (2x contiguous-copy, 2x rope, 2x index_put, arange, clamp) — each tiny
// kernel costs ~1.5-5us of dispatch floor (guide: boundary row), which at
// 32 layers x 512 tokens dominated the glue time.
#include "common.h"

__global__ void rope_qkv_decode_kernel(
    const bf16* __restrict__ qkv,  // [bs, (nq+2nkv)*hd]
    const bf16* __restrict__ bias,  // [(nq+2nkv)*hd] or null
    bf16* __restrict__ q_out,  // [bs, nq, hd]
    bf16* __restrict__ kcache,  // [bs, maxlen, nkv, hd]
    bf16* __restrict__ vcache,
    const int* __restrict__ cache_seqlens,  // [bs]
    const float* __restrict__ cosb, const float* __restrict__ sinb,  // [*, hd/2]
    int bs, int nq, int nkv, int hd, long maxlen, long qkv_stride,
    bool apply_rope) {
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

    const bf16* src = qkv + (long)b * qkv_stride + (long)h * hd;
    short4v x1 = *(const short4v*)((const short*)src + d0);
    short4v x2 = *(const short4v*)((const short*)src + hd2 + d0);
    if (bias) {
      short4v b1 = *(const short4v*)((const short*)bias + (long)h * hd + d0);
      short4v b2 = *(const short4v*)((const short*)bias + (long)h * hd + hd2 + d0);
      #pragma unroll
      for (int j = 0; j < 4; j++) {
        ((bf16*)&x1)[j] = __float2bfloat16(
            __bfloat162float(((bf16*)&x1)[j]) + __bfloat162float(((bf16*)&b1)[j]));
        ((bf16*)&x2)[j] = __float2bfloat16(
            __bfloat162float(((bf16*)&x2)[j]) + __bfloat162float(((bf16*)&b2)[j]));
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
    if (rope) {
      float4v c = *(const float4v*)(cosb + (long)pos * hd2 + d0);
      float4v s = *(const float4v*)(sinb + (long)pos * hd2 + d0);
      short o1[4], o2[4];
      #pragma unroll
      for (int j = 0; j < 4; j++) {
        float a = __bfloat162float(((bf16*)&x1)[j]);
        float bb = __bfloat162float(((bf16*)&x2)[j]);
        ((bf16*)o1)[j] = __float2bfloat16(a * c[j] - bb * s[j]);
        ((bf16*)o2)[j] = __float2bfloat16(bb * c[j] + a * s[j]);
      }
      *(short4v*)((short*)dst + d0) = *(short4v*)o1;
      *(short4v*)((short*)dst + hd2 + d0) = *(short4v*)o2;
    } else {
      *(short4v*)((short*)dst + d0) = x1;
      *(short4v*)((short*)dst + hd2 + d0) = x2;
    }
  }
}

torch::Tensor rope_qkv_decode(
    torch::Tensor qkv, c10::optional<torch::Tensor> bias, torch::Tensor kcache,
    torch::Tensor vcache, torch::Tensor cache_seqlens, torch::Tensor cosb,
    torch::Tensor sinb, long nq, bool apply_rope) {
  TORCH_CHECK(qkv.is_cuda() && qkv.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(qkv.dim() == 2 && qkv.stride(1) == 1);
  int bs = qkv.size(0);
  int nkv = kcache.size(2);
  int hd = kcache.size(3);
  long maxlen = kcache.size(1);
  TORCH_CHECK(qkv.size(1) == (nq + 2 * nkv) * hd);
  TORCH_CHECK(hd % 8 == 0);
  auto q_out = torch::empty({bs, (long)nq, (long)hd}, qkv.options());
  long nwork = (long)bs * (nq + 2 * nkv) * (hd / 8);
  int grid = (int)std::min<long>((nwork + 255) / 256, 4096);
  const bf16* bptr = nullptr;
  if (bias.has_value()) bptr = (const bf16*)bias->data_ptr();
  hipLaunchKernelGGL(rope_qkv_decode_kernel, dim3(grid), dim3(256), 0,
    cur_stream(), (const bf16*)qkv.data_ptr(), bptr,
    (bf16*)q_out.data_ptr(), (bf16*)kcache.data_ptr(),
    (bf16*)vcache.data_ptr(), cache_seqlens.data_ptr<int>(),
    cosb.data_ptr<float>(), sinb.data_ptr<float>(), bs, (int)nq, nkv, hd,
    maxlen, qkv.stride(0), apply_rope);
  CHECK_CUDA_OK();
  return q_out;
}
