// Flash-attention packed-varlen causal BACKWARD for gfx950 (MFMA).
// Replaces the batched-GEMM recompute path of _AttnVarlenFn.backward
// (reference counterpart: flash-attn 2's varlen bwd, an external CUDA
// dep of the reference — SURVEY.md §2.2).
//
// kv-stationary: grid = (kv_block, q_head); workgroup = 4 waves; wave w
// owns keys [k0 + 16w, k0 + 16w + 16) of a 64-key block and accumulates
// that slice's dK/dV in MFMA C fragments (16x128 fp32 = 64 VGPR/lane for
// both).  The workgroup loops over 32-query tiles >= the diagonal:
//   S^T = K Q^T               (A = K row frag, B = Q row frag)
//   P^T = exp(S^T*scale - lse[q])
//   dV += P^T dO              (A = P^T via LDS, B = dO via tr16)
//   dP^T = V dO^T             (A = V row frag, B = dO row frag)
//   dS^T = P^T o (dP^T - D[q]) * scale
//   dK += dS^T Q              (A = dS^T via LDS, B = Q via tr16)
//   dQ += dS K  -> fp32 atomics (A = dS via tr16 of the shared dS^T
//                 tile, B = K via tr16; hd chunks split across waves)
// dK/dV are written per Q-HEAD to fp32 buffers (no atomics); the host
// sums GQA groups.  D = rowsum(dO*O) and the final casts happen in
// torch (fp32).
//
// MFMA lane maps as in attn_varlen.hip (verified by mfma_probe):
//   A[i][k]: i = lane&15, k = (lane>>4)*8 + j
//   B[k][n]: n = lane&15, k = (lane>>4)*8 + j
//   C[i][n]: n = lane&15, i = (lane>>4)*4 + r
// tr16 hardware-transpose read (verified by tr16_probe): reading the
// 8-byte chunk at src[kbase + g*8 + (i16>>2) (+4)][nbase + (i16&3)*4]
// yields the B (or A — identical lane geometry) fragment with k = LDS
// row, n(or i) = LDS column.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4v;
typedef __attribute__((address_space(3))) bf16x4v lds_b64_t;

static __global__ void bw_build_blk_offsets(const int* __restrict__ cu,
                                             int bs, int blk,
                                             int* __restrict__ out) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    int acc = 0;
    for (int i = 0; i < bs; i++) {
      out[i] = acc;
      int L = cu[i + 1] - cu[i];
      acc += (L + blk - 1) / blk;
    }
    out[bs] = acc;
  }
}

DEVINL int bw_blk_lookup(const int* __restrict__ off, int bs, int blk_id) {
  int lo = 0, hi = bs;  // largest i with off[i] <= blk_id
  while (lo + 1 < hi) {
    int mid = (lo + hi) >> 1;
    if (off[mid] <= blk_id) lo = mid; else hi = mid;
  }
  return lo;
}

#define BW_WAVES 4
#define BW_KV 64      // keys per workgroup (16 per wave)
#define BW_QT 32      // queries per tile
#define BW_PAD 8
#define BW_TPAD 16    // pad for tr16-read tiles (bank spread, fwd-proven)

DEVINL bf16x8 tr16_bfrag(const __bf16* base, int pitch, int krow0, int col4) {
  // two transpose reads -> one 8-deep fragment (k = rows, n = cols)
  bf16x4v r0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_b64_t*)(base + (long)krow0 * pitch + col4));
  bf16x4v r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_b64_t*)(base + (long)(krow0 + 4) * pitch + col4));
  bf16x8 vb;
  #pragma unroll
  for (int j = 0; j < 4; j++) { vb[j] = r0[j]; vb[4 + j] = r1[j]; }
  return vb;
}

template <int HD>
__global__ __launch_bounds__(64 * BW_WAVES, 2) void attn_varlen_bwd_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ Dsum,
    const int* __restrict__ cu_seqlens, const int* __restrict__ blk_offsets,
    int n_seqs, float* __restrict__ dq32,
    float* __restrict__ dk32, float* __restrict__ dv32,
    int nq, int nkv, float scale, bool causal, int window) {
  constexpr int HDCH = HD / 32;
  const int blk = blockIdx.x;
  if (blk >= blk_offsets[n_seqs]) return;  // over-provisioned grid tail
  const int qh = blockIdx.y;
  const int kvh = qh / (nq / nkv);
  const int seq = bw_blk_lookup(blk_offsets, n_seqs, blk);
  const int k0 = (blk - blk_offsets[seq]) * BW_KV;
  const int s0 = cu_seqlens[seq], s1 = cu_seqlens[seq + 1];
  const int L = s1 - s0;
  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int i16 = lane & 15;
  const int g = lane >> 4;

  __shared__ __bf16 k_s[BW_KV][HD + BW_TPAD];   // row-major; tr16-read too
  __shared__ __bf16 v_s[BW_KV][HD + BW_PAD];
  __shared__ __bf16 q_s[BW_QT][HD + BW_TPAD];   // row-major; tr16-read too
  __shared__ __bf16 do_s[BW_QT][HD + BW_TPAD];
  __shared__ __bf16 pt_s[BW_WAVES][16][BW_QT + BW_PAD];  // P^T per wave
  __shared__ __bf16 ds_all[BW_KV][BW_QT + BW_TPAD];      // dS^T, all waves
  __shared__ float lse_s[BW_QT], d_s[BW_QT];

  // ---- stage the K/V block (whole kernel) ---------------------------
  for (int idx = threadIdx.x; idx < BW_KV * (HD / 8); idx += 64 * BW_WAVES) {
    int row = idx / (HD / 8);
    int col8 = (idx % (HD / 8)) * 8;
    bf16x8 kv = {}, vv = {};
    if (k0 + row < L) {
      kv = *(const bf16x8*)(k + ((long)(s0 + k0 + row) * nkv + kvh) * HD + col8);
      vv = *(const bf16x8*)(v + ((long)(s0 + k0 + row) * nkv + kvh) * HD + col8);
    }
    *(bf16x8*)(&k_s[row][col8]) = kv;
    *(bf16x8*)(&v_s[row][col8]) = vv;
  }
  __syncthreads();

  // wave-resident A fragments of K and V (rows = this wave's 16 keys)
  bf16x8 kfrag[HDCH], vfrag[HDCH];
  #pragma unroll
  for (int c = 0; c < HDCH; c++) {
    kfrag[c] = *(const bf16x8*)(&k_s[w * 16 + i16][c * 32 + g * 8]);
    vfrag[c] = *(const bf16x8*)(&v_s[w * 16 + i16][c * 32 + g * 8]);
  }

  // dK/dV accumulators: C fragments [key16][hd16 chunks]
  f32x4 dk_acc[HD / 16], dv_acc[HD / 16];
  #pragma unroll
  for (int t = 0; t < HD / 16; t++) {
    dk_acc[t] = {0.f, 0.f, 0.f, 0.f};
    dv_acc[t] = {0.f, 0.f, 0.f, 0.f};
  }

  const int q_begin = causal ? (k0 / BW_QT) * BW_QT : 0;
  // sliding window (mistral, fwd-matching semantics: query q sees keys
  // (q-window, q]): key kidx is consumed only by queries < kidx + window,
  // so this kv block's last relevant query is k0 + BW_KV - 1 + window - 1
  const int q_end = (window > 0) ? min(L, k0 + BW_KV + window - 1) : L;
  for (int q0 = q_begin; q0 < q_end; q0 += BW_QT) {
    const int qn = min(BW_QT, L - q0);
    // ---- stage Q/dO tile + lse + D --------------------------------
    __syncthreads();
    for (int idx = threadIdx.x; idx < BW_QT * (HD / 8); idx += 64 * BW_WAVES) {
      int row = idx / (HD / 8);
      int col8 = (idx % (HD / 8)) * 8;
      bf16x8 qv = {}, dv = {};
      if (row < qn) {
        qv = *(const bf16x8*)(q + ((long)(s0 + q0 + row) * nq + qh) * HD + col8);
        dv = *(const bf16x8*)(dout + ((long)(s0 + q0 + row) * nq + qh) * HD + col8);
      }
      *(bf16x8*)(&q_s[row][col8]) = qv;
      *(bf16x8*)(&do_s[row][col8]) = dv;
    }
    if (threadIdx.x < BW_QT) {
      int row = threadIdx.x;
      bool ok = row < qn;
      lse_s[row] = ok ? lse[(long)(s0 + q0 + row) * nq + qh] : 0.f;
      d_s[row] = ok ? Dsum[(long)(s0 + q0 + row) * nq + qh] : 0.f;
    }
    __syncthreads();

    // ---- S^T and dP^T over the two 16-q subtiles -------------------
    f32x4 st[2], dpt[2];
    #pragma unroll
    for (int qs = 0; qs < 2; qs++) {
      f32x4 a1 = {0.f, 0.f, 0.f, 0.f}, a2 = {0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int c = 0; c < HDCH; c++) {
        bf16x8 qb = *(const bf16x8*)(&q_s[qs * 16 + i16][c * 32 + g * 8]);
        bf16x8 db = *(const bf16x8*)(&do_s[qs * 16 + i16][c * 32 + g * 8]);
        a1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kfrag[c], qb, a1, 0, 0, 0);
        a2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag[c], db, a2, 0, 0, 0);
      }
      st[qs] = a1;
      dpt[qs] = a2;
    }

    // ---- P^T, dS^T (C layout: row = key w*16 + g*4 + r, col = q i16)
    #pragma unroll
    for (int qs = 0; qs < 2; qs++) {
      const int qidx = q0 + qs * 16 + i16;
      #pragma unroll
      for (int r = 0; r < 4; r++) {
        const int kidx = k0 + w * 16 + g * 4 + r;
        bool ok = (kidx < L) && (qidx < L) && (!causal || kidx <= qidx)
                  && (window <= 0 || kidx > qidx - window);
        float p = ok ? __expf(st[qs][r] * scale - lse_s[qs * 16 + i16]) : 0.f;
        float ds = ok ? p * (dpt[qs][r] - d_s[qs * 16 + i16]) * scale : 0.f;
        pt_s[w][g * 4 + r][qs * 16 + i16] = (__bf16)p;
        ds_all[w * 16 + g * 4 + r][qs * 16 + i16] = (__bf16)ds;
      }
    }
    __syncthreads();

    // ---- dV += P^T dO ; dK += dS^T Q (B fragments via tr16) --------
    {
      // A fragments: k = q dim is 32; the chunks g*8+j span [0,32) ✓
      bf16x8 pa = *(const bf16x8*)(&pt_s[w][i16][g * 8]);
      bf16x8 da = *(const bf16x8*)(&ds_all[w * 16 + i16][g * 8]);
      const int krow0 = g * 8 + (i16 >> 2);
      const int col4 = (i16 & 3) * 4;
      #pragma unroll
      for (int t = 0; t < HD / 16; t++) {
        bf16x8 dob = tr16_bfrag(&do_s[0][0], HD + BW_TPAD, krow0, t * 16 + col4);
        bf16x8 qb = tr16_bfrag(&q_s[0][0], HD + BW_TPAD, krow0, t * 16 + col4);
        dv_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, dob, dv_acc[t], 0, 0, 0);
        dk_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(da, qb, dk_acc[t], 0, 0, 0);
      }
    }

    // ---- dQ += dS K (cooperative over hd chunks; fp32 atomics) -----
    // A = dS[q][key] via tr16 of ds_all (k = LDS row = key, i = col = q,
    // one 16-q column window per qs2); B = K[key][hd] via tr16 of k_s.
    {
      const int col4 = (i16 & 3) * 4;
      #pragma unroll
      for (int qs2 = 0; qs2 < 2; qs2++) {
        #pragma unroll
        for (int t = 0; t < HD / 16; t += 1) {
          if ((t & (BW_WAVES - 1)) != w) continue;  // split chunks by wave
          f32x4 dq_acc = {0.f, 0.f, 0.f, 0.f};
          #pragma unroll
          for (int kc = 0; kc < 2; kc++) {  // two key chunks of 32
            const int krow0 = kc * 32 + g * 8 + (i16 >> 2);
            bf16x8 dsa = tr16_bfrag(&ds_all[0][0], BW_QT + BW_TPAD, krow0,
                                    qs2 * 16 + col4);
            bf16x8 kb = tr16_bfrag(&k_s[0][0], HD + BW_TPAD, krow0,
                                   t * 16 + col4);
            dq_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa, kb, dq_acc, 0, 0, 0);
          }
          #pragma unroll
          for (int r = 0; r < 4; r++) {
            const int qrow = q0 + qs2 * 16 + g * 4 + r;
            if (qrow < L)
              atomicAdd(&dq32[((long)(s0 + qrow) * nq + qh) * HD + t * 16 + i16],
                        dq_acc[r]);
          }
        }
      }
    }
  }

  // ---- write dK/dV per q-head (fp32, no atomics) --------------------
  __syncthreads();
  #pragma unroll
  for (int r = 0; r < 4; r++) {
    const int kidx = k0 + w * 16 + g * 4 + r;
    if (kidx >= L) continue;
    float* dkrow = dk32 + ((long)(s0 + kidx) * nq + qh) * HD;
    float* dvrow = dv32 + ((long)(s0 + kidx) * nq + qh) * HD;
    #pragma unroll
    for (int t = 0; t < HD / 16; t++) {
      dkrow[t * 16 + i16] = dk_acc[t][r];
      dvrow[t * 16 + i16] = dv_acc[t][r];
    }
  }
}

std::vector<torch::Tensor> attn_varlen_bwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, torch::Tensor dout,
    torch::Tensor lse, torch::Tensor Dsum, torch::Tensor cu_seqlens,
    bool causal, double scale, long window) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous() &&
              dout.is_contiguous());
  int total = q.size(0), nq = q.size(1), hd = q.size(2);
  int nkv = k.size(1);
  auto cu_dev = cu_seqlens.to(torch::kInt).to(q.device());
  int bs = cu_dev.numel() - 1;
  auto iopts = torch::TensorOptions().dtype(torch::kInt).device(q.device());
  auto blk_off = torch::empty({(long)bs + 1}, iopts);
  hipLaunchKernelGGL(bw_build_blk_offsets, dim3(1), dim3(64), 0,
    cur_stream(), cu_dev.data_ptr<int>(), bs, BW_KV, blk_off.data_ptr<int>());
  auto f32 = q.options().dtype(torch::kFloat);
  auto dq32 = torch::zeros({(long)total, (long)nq, (long)hd}, f32);
  auto dk32 = torch::empty({(long)total, (long)nq, (long)hd}, f32);
  auto dv32 = torch::empty({(long)total, (long)nq, (long)hd}, f32);
  dim3 grid((unsigned)(total / BW_KV + bs), nq);
  TORCH_CHECK(hd == 128 || hd == 64, "attn_varlen_bwd: hd 64/128 only");
  if (hd == 128) {
    hipLaunchKernelGGL((attn_varlen_bwd_kernel<128>), grid,
      dim3(64 * BW_WAVES), 0, cur_stream(), (const bf16*)q.data_ptr(),
      (const bf16*)k.data_ptr(), (const bf16*)v.data_ptr(),
      (const bf16*)dout.data_ptr(), lse.data_ptr<float>(),
      Dsum.data_ptr<float>(), cu_dev.data_ptr<int>(),
      blk_off.data_ptr<int>(), bs,
      dq32.data_ptr<float>(), dk32.data_ptr<float>(), dv32.data_ptr<float>(),
      nq, nkv, (float)scale, causal, (int)window);
  } else {
    hipLaunchKernelGGL((attn_varlen_bwd_kernel<64>), grid,
      dim3(64 * BW_WAVES), 0, cur_stream(), (const bf16*)q.data_ptr(),
      (const bf16*)k.data_ptr(), (const bf16*)v.data_ptr(),
      (const bf16*)dout.data_ptr(), lse.data_ptr<float>(),
      Dsum.data_ptr<float>(), cu_dev.data_ptr<int>(),
      blk_off.data_ptr<int>(), bs,
      dq32.data_ptr<float>(), dk32.data_ptr<float>(), dv32.data_ptr<float>(),
      nq, nkv, (float)scale, causal, (int)window);
  }
  CHECK_CUDA_OK();
  return {dq32, dk32, dv32};
}
