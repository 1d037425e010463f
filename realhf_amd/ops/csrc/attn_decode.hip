// Single-token GQA decode attention over a contiguous KV cache (gfx950).
// Replaces flash_attn_with_kvcache in the reference hot loop
// (realhf/impl/model/nn/real_llm_generate.py:?38 decode step).
//
// Shapes: q [bs, nq, hd], k/v_cache [bs, maxlen, nkv, hd], cache_seqlens[bs].
// Memory-bound: the job is to stream each sequence's KV exactly once at
// full HBM bandwidth.  One 256-thread workgroup (4 waves) per (batch,
// kv-head); the workgroup computes ALL rep = nq/nkv query heads of that
// kv head so the KV stream is read once regardless of GQA ratio.
//
// Per 64-key chunk (one key per lane):
//   lane = key: score_r = q_r . k  (16-byte vectorized k loads, 256 B/lane
//   contiguous -> full lines); online-softmax per wave; p -> LDS; then the
//   lane role flips to "2 output dims per lane" and V is accumulated with
//   coalesced 4-byte loads.  Wave partials (m, s, acc) combine through LDS.
#include "common.h"

template <int HD, int REP>
__global__ void attn_decode_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ kc,
    const bf16* __restrict__ vc, const int* __restrict__ cache_seqlens,
    bf16* __restrict__ out, int bs, int nq, int nkv, long maxlen, float scale) {
  constexpr int DPL = HD / WAVE;  // dims per lane (2 for hd=128)
  const int b = blockIdx.x / nkv;
  const int kvh = blockIdx.x % nkv;
  const int L = cache_seqlens[b];
  const int lane = threadIdx.x & (WAVE - 1);
  const int w = threadIdx.x / WAVE;

  __shared__ float q_s[REP][HD];
  __shared__ float m_s[4], s_s[4];
  __shared__ float acc_s[4][HD];

  // stage q (scaled) into LDS
  for (int i = threadIdx.x; i < REP * HD; i += blockDim.x) {
    int r = i / HD, d = i % HD;
    q_s[r][d] = __bfloat162float(q[((long)b * nq + kvh * REP + r) * HD + d]) * scale;
  }
  __syncthreads();

  float m_w[REP], s_w[REP], acc_w[REP][DPL];
  #pragma unroll
  for (int r = 0; r < REP; r++) {
    m_w[r] = -1e30f;
    s_w[r] = 0.f;
    #pragma unroll
    for (int d = 0; d < DPL; d++) acc_w[r][d] = 0.f;
  }

  const long kv_stride = (long)nkv * HD;  // per-key stride
  const bf16* kb = kc + (long)b * maxlen * kv_stride + (long)kvh * HD;
  const bf16* vb = vc + (long)b * maxlen * kv_stride + (long)kvh * HD;

  for (int base = w * WAVE; base < L; base += 4 * WAVE) {
    int l = base + lane;
    bool valid = l < L;
    // scores for this lane's key against all REP q heads
    float sc[REP];
    #pragma unroll
    for (int r = 0; r < REP; r++) sc[r] = 0.f;
    if (valid) {
      const bf16* krow = kb + (long)l * kv_stride;
      #pragma unroll
      for (int i = 0; i < HD / 8; i++) {
        short8 kv8 = *(const short8*)((const short*)krow + i * 8);
        float kf[8];
        #pragma unroll
        for (int j = 0; j < 8; j++) kf[j] = bf2f(kv8[j]);
        #pragma unroll
        for (int r = 0; r < REP; r++) {
          #pragma unroll
          for (int j = 0; j < 8; j++) sc[r] += q_s[r][i * 8 + j] * kf[j];
        }
      }
    }
    #pragma unroll
    for (int r = 0; r < REP; r++) {
      if (!valid) sc[r] = -1e30f;
      float cmax = wave_max(sc[r]);
      float nm = fmaxf(m_w[r], cmax);
      float f = __expf(m_w[r] - nm);
      m_w[r] = nm;
      s_w[r] *= f;
      #pragma unroll
      for (int d = 0; d < DPL; d++) acc_w[r][d] *= f;
      float p = valid ? __expf(sc[r] - nm) : 0.f;
      s_w[r] += wave_sum(p);
      // role flip: lane owns DPL output dims; broadcast p via shfl
      #pragma unroll 4
      for (int j = 0; j < WAVE; j++) {
        int lj = base + j;
        if (lj >= L) break;
        float pj = __shfl(p, j, 64);
        const bf16* vrow = vb + (long)lj * kv_stride;
        #pragma unroll
        for (int d = 0; d < DPL; d++)
          acc_w[r][d] += pj * __bfloat162float(vrow[lane * DPL + d]);
      }
    }
  }

  // combine the 4 waves' partials per rep head
  #pragma unroll
  for (int r = 0; r < REP; r++) {
    if (lane == 0) { m_s[w] = m_w[r]; s_s[w] = s_w[r]; }
    #pragma unroll
    for (int d = 0; d < DPL; d++) acc_s[w][lane * DPL + d] = acc_w[r][d];
    __syncthreads();
    if (w == 0) {
      float M = fmaxf(fmaxf(m_s[0], m_s[1]), fmaxf(m_s[2], m_s[3]));
      float S = 0.f;
      float o[DPL];
      #pragma unroll
      for (int d = 0; d < DPL; d++) o[d] = 0.f;
      #pragma unroll
      for (int ww = 0; ww < 4; ww++) {
        float f = __expf(m_s[ww] - M);
        S += f * s_s[ww];
        #pragma unroll
        for (int d = 0; d < DPL; d++) o[d] += f * acc_s[ww][lane * DPL + d];
      }
      float inv = (S > 0.f) ? 1.f / S : 0.f;
      #pragma unroll
      for (int d = 0; d < DPL; d++)
        out[((long)b * nq + kvh * REP + r) * HD + lane * DPL + d] =
            __float2bfloat16(o[d] * inv);
    }
    __syncthreads();
  }
}

torch::Tensor attn_decode(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                          torch::Tensor cache_seqlens, double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && kc.is_contiguous() && vc.is_contiguous());
  int bs = q.size(0), nq = q.size(1), hd = q.size(2);
  long maxlen = kc.size(1);
  int nkv = kc.size(2);
  int rep = nq / nkv;
  auto cs = cache_seqlens.to(torch::kInt);
  auto out = torch::empty_like(q);
  dim3 grid(bs * nkv);
  auto launch = [&](auto hd_c, auto rep_c) {
    hipLaunchKernelGGL((attn_decode_kernel<hd_c.value, rep_c.value>), grid,
      dim3(256), 0, cur_stream(), (const bf16*)q.data_ptr(),
      (const bf16*)kc.data_ptr(), (const bf16*)vc.data_ptr(),
      cs.data_ptr<int>(), (bf16*)out.data_ptr(), bs, nq, nkv, maxlen,
      (float)scale);
  };
  #define REP_SWITCH(HDV) \
    switch (rep) { \
      case 1: launch(std::integral_constant<int, HDV>{}, std::integral_constant<int, 1>{}); break; \
      case 2: launch(std::integral_constant<int, HDV>{}, std::integral_constant<int, 2>{}); break; \
      case 4: launch(std::integral_constant<int, HDV>{}, std::integral_constant<int, 4>{}); break; \
      case 8: launch(std::integral_constant<int, HDV>{}, std::integral_constant<int, 8>{}); break; \
      default: TORCH_CHECK(false, "unsupported GQA ratio ", rep); \
    }
  if (hd == 128) { REP_SWITCH(128) }
  else if (hd == 64) { REP_SWITCH(64) }
  else { TORCH_CHECK(false, "unsupported head_dim ", hd); }
  #undef REP_SWITCH
  CHECK_CUDA_OK();
  return out;
}
