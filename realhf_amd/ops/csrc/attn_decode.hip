// Single-token GQA decode attention over a contiguous KV cache (gfx950).
// Replaces flash_attn_with_kvcache in the reference hot loop
// (realhf/impl/model/nn/real_llm_generate.py decode step).
//
// Shapes: q [bs, nq, hd], k/v_cache [bs, maxlen, nkv, hd], cache_seqlens[bs].
// Memory-bound: the job is to stream each sequence's KV exactly once at
// full HBM bandwidth.  One 512-thread workgroup (8 waves) per (batch,
// kv-head); the workgroup computes ALL rep = nq/nkv query heads of that
// kv head so the KV stream is read once regardless of GQA ratio.
//
// Per 64-key chunk (wave w takes chunks w, w+8, ...):
//   lane = key: the lane streams its key's K row AND V row with 16-byte
//   vector loads (256 B contiguous per lane — full cache lines); V goes
//   to this wave's LDS tile.  Online softmax per wave (shfl reduces);
//   then the lane role flips to "DPL output dims per lane" and V is
//   accumulated from LDS (conflict-free b32 reads), p broadcast by shfl.
//   Wave partials (m, s, acc) combine through LDS at the end.
#include "common.h"

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2q;

// Wave count trade-off: the per-wave V tile is W*64*HD*2 bytes of LDS —
// at HD=128, W=8 needs 128 KB (ONE workgroup per CU: the PV phase has
// nothing co-resident to overlap with), W=4 needs 64 KB (TWO WGs per CU:
// one streams KV while the other runs its softmax/PV phase).  Default 4;
// REALHF_AMD_DEC_WAVES ∈ {2,4,8} for A/B.
// Split-K over the key range (flash-decoding): at decode batch sizes
// bs*nkv < 256 a one-WG-per-(batch, kv-head) grid leaves most of the
// 256 CUs idle; gridDim.y splits each sequence's keys into `nsplit`
// ranges whose unnormalized partials (acc, m, s) land in `ws` and are
// softmax-combined by attn_decode_combine_kernel.  nsplit is chosen
// from bs*nkv only (static for hipGraph capture); per-batch ranges are
// derived in-kernel from the dynamic cache_seqlens.
template <int HD, int REP, int W>
__global__ __launch_bounds__(64 * W) void attn_decode_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ kc,
    const bf16* __restrict__ vc, const int* __restrict__ cache_seqlens,
    bf16* __restrict__ out, float* __restrict__ ws, int bs, int nq, int nkv,
    long maxlen, float scale, int window) {
  constexpr int DPL = HD / WAVE;  // dims per lane (2 for hd=128)
  const int b = blockIdx.x / nkv;
  const int kvh = blockIdx.x % nkv;
  const int nsplit = gridDim.y;
  const int sp = blockIdx.y;
  const int L = cache_seqlens[b];
  const int lane = threadIdx.x & (WAVE - 1);
  const int w = threadIdx.x / WAVE;

  __shared__ __bf16 q_s[REP][HD];  // raw bf16: K-dot uses v_dot2 pairs
  __shared__ __bf16 v_s[W][WAVE][HD];  // per-wave V chunk tile
  __shared__ float m_s[W], s_s[W];
  __shared__ float acc_s[W][HD];

  for (int i = threadIdx.x; i < REP * HD; i += blockDim.x) {
    int r = i / HD, d = i % HD;
    q_s[r][d] = (__bf16)q[((long)b * nq + kvh * REP + r) * HD + d];
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

  // sliding window (mistral): only keys in [L - window, L) attend
  const int lo = (window > 0 && L > window) ? (L - window) : 0;
  const int origin = (lo / (W * WAVE)) * (W * WAVE);
  // this split's aligned key range [r0, r1)
  const int chunk = W * WAVE;
  const int nchunks = (L - origin + chunk - 1) / chunk;
  const int per_sp = (nchunks + nsplit - 1) / nsplit;
  const int r0 = origin + sp * per_sp * chunk;
  const int r1 = min((long)L, (long)origin + (long)(sp + 1) * per_sp * chunk);

  for (int base = r0 + w * WAVE; base < r1; base += W * WAVE) {
    const int l = base + lane;
    const bool valid = l < L && l >= lo;
    float sc[REP];
    #pragma unroll
    for (int r = 0; r < REP; r++) sc[r] = 0.f;
    if (!valid) {
      // windowed: front-invalid lanes are INSIDE the PV j-range, so
      // their V tile rows must be zeros (0 * uninitialized LDS = NaN)
      #pragma unroll
      for (int i = 0; i < HD / 8; i++)
        *(short8*)(&v_s[w][lane][i * 8]) = short8{};
    }
    if (valid) {
      const bf16* krow = kb + (long)l * kv_stride;
      const bf16* vrow = vb + (long)l * kv_stride;
      #pragma unroll
      for (int i = 0; i < HD / 8; i++) {
        short8 kv8 = *(const short8*)((const short*)krow + i * 8);
        short8 vv8 = *(const short8*)((const short*)vrow + i * 8);
        *(short8*)(&v_s[w][lane][i * 8]) = vv8;
        // packed bf16 dots: 4 v_dot2 per 8 elems instead of 8 cvt + 8 fma
        #pragma unroll
        for (int r = 0; r < REP; r++) {
          #pragma unroll
          for (int j = 0; j < 4; j++) {
            bf16x2q a = *(const bf16x2q*)(&q_s[r][i * 8 + j * 2]);
            bf16x2q kk = *(const bf16x2q*)((const short*)&kv8 + j * 2);
            sc[r] = __builtin_amdgcn_fdot2_f32_bf16(a, kk, sc[r], false);
          }
        }
      }
    }
    #pragma unroll
    for (int r = 0; r < REP; r++) sc[r] *= scale;  // q kept raw bf16
    const int nvalid = min(WAVE, r1 - base);
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
      // role flip: lane owns DPL output dims; V comes from LDS,
      // p broadcast by shfl (no cross-wave LDS -> in-wave visibility)
      for (int j = 0; j < nvalid; j++) {
        float pj = __shfl(p, j, 64);
        if constexpr (DPL == 2) {
          // one b32 LDS read for the lane's two dims
          unsigned u = *(const unsigned*)(&v_s[w][j][lane * 2]);
          acc_w[r][0] += pj * bf2f((short)(u & 0xffff));
          acc_w[r][1] += pj * bf2f((short)(u >> 16));
        } else {
          #pragma unroll
          for (int d = 0; d < DPL; d++)
            acc_w[r][d] += pj * (float)v_s[w][j][lane * DPL + d];
        }
      }
    }
  }

  // combine the wave partials per rep head
  #pragma unroll
  for (int r = 0; r < REP; r++) {
    if (lane == 0) { m_s[w] = m_w[r]; s_s[w] = s_w[r]; }
    #pragma unroll
    for (int d = 0; d < DPL; d++) acc_s[w][lane * DPL + d] = acc_w[r][d];
    __syncthreads();
    if (w == 0) {
      float M = -1e30f;
      #pragma unroll
      for (int ww = 0; ww < W; ww++) M = fmaxf(M, m_s[ww]);
      float S = 0.f;
      float o[DPL];
      #pragma unroll
      for (int d = 0; d < DPL; d++) o[d] = 0.f;
      #pragma unroll
      for (int ww = 0; ww < W; ww++) {
        float f = (s_s[ww] > 0.f) ? __expf(m_s[ww] - M) : 0.f;
        S += f * s_s[ww];
        #pragma unroll
        for (int d = 0; d < DPL; d++) o[d] += f * acc_s[ww][lane * DPL + d];
      }
      const long head = (long)b * nq + kvh * REP + r;
      if (ws == nullptr) {
        float inv = (S > 0.f) ? 1.f / S : 0.f;
        #pragma unroll
        for (int d = 0; d < DPL; d++)
          out[head * HD + lane * DPL + d] = __float2bfloat16(o[d] * inv);
      } else {
        // unnormalized split partial: [head][split][HD dims | m | s]
        float* row = ws + (head * nsplit + sp) * (HD + 2);
        #pragma unroll
        for (int d = 0; d < DPL; d++) row[lane * DPL + d] = o[d];
        if (lane == 0) { row[HD] = M; row[HD + 1] = S; }
      }
    }
    __syncthreads();
  }
}

template <int HD>
__global__ __launch_bounds__(WAVE) void attn_decode_combine_kernel(
    const float* __restrict__ ws, bf16* __restrict__ out, int nsplit) {
  constexpr int DPL = HD / WAVE;
  const long head = blockIdx.x;
  const int lane = threadIdx.x;
  const float* base = ws + head * (long)nsplit * (HD + 2);
  float M = -1e30f;
  for (int s = 0; s < nsplit; s++) M = fmaxf(M, base[s * (HD + 2) + HD]);
  float S = 0.f;
  float o[DPL];
  #pragma unroll
  for (int d = 0; d < DPL; d++) o[d] = 0.f;
  for (int s = 0; s < nsplit; s++) {
    const float* row = base + s * (HD + 2);
    float ss = row[HD + 1];
    float f = (ss > 0.f) ? __expf(row[HD] - M) : 0.f;
    S += f * ss;
    #pragma unroll
    for (int d = 0; d < DPL; d++) o[d] += f * row[lane * DPL + d];
  }
  float inv = (S > 0.f) ? 1.f / S : 0.f;
  #pragma unroll
  for (int d = 0; d < DPL; d++)
    out[head * HD + lane * DPL + d] = __float2bfloat16(o[d] * inv);
}

torch::Tensor attn_decode(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                          torch::Tensor cache_seqlens, double scale,
                          long window) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && kc.is_contiguous() && vc.is_contiguous());
  int bs = q.size(0), nq = q.size(1), hd = q.size(2);
  long maxlen = kc.size(1);
  int nkv = kc.size(2);
  int rep = nq / nkv;
  auto cs = cache_seqlens.to(torch::kInt);
  auto out = torch::empty_like(q);
  static int dec_waves = [] {
    const char* e = getenv("REALHF_AMD_DEC_WAVES");
    int v = e ? atoi(e) : 4;
    return (v == 2 || v == 4 || v == 8) ? v : 4;
  }();
  // split-K factor: fill the 256-CU chip when bs*nkv is small (decode
  // batches).  Static per shape -> hipGraph-safe; REALHF_AMD_DEC_SPLIT
  // overrides for A/B.
  static int dec_split_env = [] {
    const char* e = getenv("REALHF_AMD_DEC_SPLIT");
    return e ? atoi(e) : 0;
  }();
  int nsplit = dec_split_env > 0
      ? dec_split_env
      : std::min<long>(8, std::max<long>(1, (2 * 256) / std::max(1, bs * nkv)));
  torch::Tensor ws;
  float* ws_ptr = nullptr;
  if (nsplit > 1) {
    ws = torch::empty({(long)bs * nq * nsplit * (hd + 2)},
                      q.options().dtype(torch::kFloat));
    ws_ptr = ws.data_ptr<float>();
  }
  dim3 grid(bs * nkv, nsplit);
  auto launch = [&](auto hd_c, auto rep_c) {
    auto go = [&](auto w_c) {
      hipLaunchKernelGGL((attn_decode_kernel<hd_c.value, rep_c.value, w_c.value>),
        grid, dim3(64 * w_c.value), 0, cur_stream(), (const bf16*)q.data_ptr(),
        (const bf16*)kc.data_ptr(), (const bf16*)vc.data_ptr(),
        cs.data_ptr<int>(), (bf16*)out.data_ptr(), ws_ptr, bs, nq, nkv, maxlen,
        (float)scale, (int)window);
      if (ws_ptr) {
        if (hd_c.value == 128)
          hipLaunchKernelGGL((attn_decode_combine_kernel<128>),
            dim3((unsigned)(bs * nq)), dim3(WAVE), 0, cur_stream(),
            ws_ptr, (bf16*)out.data_ptr(), nsplit);
        else
          hipLaunchKernelGGL((attn_decode_combine_kernel<64>),
            dim3((unsigned)(bs * nq)), dim3(WAVE), 0, cur_stream(),
            ws_ptr, (bf16*)out.data_ptr(), nsplit);
      }
    };
    if (dec_waves == 2) go(std::integral_constant<int, 2>{});
    else if (dec_waves == 8) go(std::integral_constant<int, 8>{});
    else go(std::integral_constant<int, 4>{});
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
