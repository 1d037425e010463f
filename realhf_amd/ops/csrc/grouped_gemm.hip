// Grouped GEMM for MoE experts (gfx950, MFMA).
// Replaces the reference's external grouped_gemm CUDA dep (SURVEY.md §2.2
// ext deps (c): grouped_gemm.ops.gmm, experts.py:194-207).
//
// Computes, for each expert e:  out[seg_e] = x[seg_e] @ W[e]^T
//   x:   [total, K] bf16 (tokens sorted by expert)
//   W:   [E, N, K] bf16 (row-major, K contiguous — the canonical layout)
//   out: [total, N] bf16
//   seg_e = rows [offs[e], offs[e+1])
//
// One workgroup per (tile_m 64 x tile_n 64) tile of one expert's segment;
// the tile list is host-built from the (already host-known) expert token
// counts.  4 waves; wave w owns 16 rows; K staged in 32-deep LDS tiles.
// MFMA v_mfma_f32_16x16x32_bf16 with the §3 lane maps (same as
// attn_varlen.hip, validated by mfma_probe).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;

#define GG_TM 64
#define GG_TN 64
#define GG_TK 32
#define GG_PAD 8

__global__ __launch_bounds__(256, 2) void grouped_gemm_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    bf16* __restrict__ out,
    const int* __restrict__ tile_expert, const int* __restrict__ tile_m0,
    const int* __restrict__ tile_n0, const long* __restrict__ seg_start,
    const int* __restrict__ seg_len, int K, int N, long w_estride) {
  const int e = tile_expert[blockIdx.x];
  const int m0 = tile_m0[blockIdx.x];
  const int n0 = tile_n0[blockIdx.x];
  const long s0 = seg_start[e];
  const int M = seg_len[e];
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int i16 = lane & 15;
  const int g = lane >> 4;

  __shared__ __bf16 a_s[GG_TM][GG_TK + GG_PAD];
  __shared__ __bf16 b_s[GG_TN][GG_TK + GG_PAD];

  // wave wv owns rows [wv*16, wv*16+16) of the tile; 4 n-subtiles
  f32x4 acc[4];
  #pragma unroll
  for (int t = 0; t < 4; t++) acc[t] = {0.f, 0.f, 0.f, 0.f};

  const long wbase = (long)e * w_estride;
  for (int k0 = 0; k0 < K; k0 += GG_TK) {
    __syncthreads();
    // stage A: 64 rows x 32 k (256 threads x 8 elems = 2048 = 64*32)
    {
      int idx = threadIdx.x;  // 256 threads, each 8 elems: 2048/8=256 slots
      int row = idx / 4;
      int col8 = (idx % 4) * 8;
      bf16x8v va = {};
      if (m0 + row < M)
        va = *(const bf16x8v*)(x + (s0 + m0 + row) * (long)K + k0 + col8);
      *(bf16x8v*)(&a_s[row][col8]) = va;
      // stage B: W[e][n0+row][k0+col8]
      bf16x8v vb = *(const bf16x8v*)(w + wbase + (long)(n0 + row) * K + k0 + col8);
      *(bf16x8v*)(&b_s[row][col8]) = vb;
    }
    __syncthreads();
    // A fragment: rows wv*16 + i16, k = g*8 + j
    bf16x8v afrag = *(const bf16x8v*)(&a_s[wv * 16 + i16][g * 8]);
    #pragma unroll
    for (int t = 0; t < 4; t++) {
      bf16x8v bfrag = *(const bf16x8v*)(&b_s[t * 16 + i16][g * 8]);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[t], 0, 0, 0);
    }
  }
  // epilogue: C rows = wv*16 + g*4 + r, col = t*16 + i16
  #pragma unroll
  for (int t = 0; t < 4; t++) {
    #pragma unroll
    for (int r = 0; r < 4; r++) {
      int row = m0 + wv * 16 + g * 4 + r;
      if (row < M)
        out[(s0 + row) * (long)N + n0 + t * 16 + i16] =
            __float2bfloat16(acc[t][r]);
    }
  }
}

torch::Tensor grouped_gemm(torch::Tensor x, torch::Tensor w,
                           torch::Tensor seg_lens_cpu) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  // w may be an expert-strided view of the flat buffer: stride(1) == K,
  // stride(2) == 1, stride(0) arbitrary
  TORCH_CHECK(w.dim() == 3 && x.is_contiguous());
  TORCH_CHECK(w.stride(2) == 1 && w.stride(1) == w.size(2));
  long total = x.size(0);
  int K = x.size(1);
  int E = w.size(0), N = w.size(1);
  TORCH_CHECK(w.size(2) == K);
  TORCH_CHECK(K % GG_TK == 0 && N % GG_TN == 0,
              "grouped_gemm needs K%32==0 and N%64==0, got ", K, " ", N);
  auto lens = seg_lens_cpu.to(torch::kInt).cpu();
  const int* lp = lens.data_ptr<int>();
  std::vector<long> starts(E);
  std::vector<int> te, tm, tn;
  long off = 0;
  for (int e = 0; e < E; e++) {
    starts[e] = off;
    for (int m0 = 0; m0 < lp[e]; m0 += GG_TM)
      for (int n0 = 0; n0 < N; n0 += GG_TN) {
        te.push_back(e);
        tm.push_back(m0);
        tn.push_back(n0);
      }
    off += lp[e];
  }
  TORCH_CHECK(off == total, off, " vs ", total);
  auto out = torch::empty({total, (long)N}, x.options());
  if (te.empty()) return out;
  auto opts = torch::TensorOptions().dtype(torch::kInt).device(x.device());
  auto te_d = torch::from_blob(te.data(), {(long)te.size()}, torch::kInt).to(x.device());
  auto tm_d = torch::from_blob(tm.data(), {(long)tm.size()}, torch::kInt).to(x.device());
  auto tn_d = torch::from_blob(tn.data(), {(long)tn.size()}, torch::kInt).to(x.device());
  auto ss_d = torch::from_blob(starts.data(), {(long)E}, torch::kLong).to(x.device());
  auto sl_d = lens.to(x.device());
  hipLaunchKernelGGL(grouped_gemm_kernel, dim3((unsigned)te.size()), dim3(256),
    0, cur_stream(), (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
    (bf16*)out.data_ptr(), te_d.data_ptr<int>(), tm_d.data_ptr<int>(),
    tn_d.data_ptr<int>(), ss_d.data_ptr<long>(), sl_d.data_ptr<int>(), K, N,
    (long)w.stride(0));
  CHECK_CUDA_OK();
  return out;
}
