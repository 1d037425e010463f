// Grouped GEMM BACKWARD for MoE expert training (gfx950, MFMA).
// Completes the native replacement of the reference's grouped_gemm CUDA
// dep: the reference trains through grouped_gemm.ops.gmm
// (realhf/impl/model/modules/moe/experts.py:194-207); forward lives in
// grouped_gemm.hip.  With these two kernels MoE training stays on the
// hand-written MFMA path instead of falling back to a per-expert GEMM
// loop.
//
// Forward was: out[seg_e] = x[seg_e] @ W[e]^T,  x:[total,K], W:[E,N,K].
// Backward:
//   dX[seg_e] = dOut[seg_e] @ W[e]        (reduce over n)   -> [total,K]
//   dW[e]     = dOut[seg_e]^T @ x[seg_e]  (reduce over m)    -> [E,N,K]
//
// Both reductions run over a dimension that is NOT contiguous in one of
// the operands, so the LDS staging TRANSPOSES that operand tile at store
// time (8 scalar ds_writes per loaded 16-byte vector); MFMA fragments
// then read 8 contiguous reduction elements per lane exactly as in the
// forward kernel (v_mfma_f32_16x16x32_bf16, §3 lane maps).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;

#define GB_TM 64   // output-row tile
#define GB_TC 64   // output-col tile
#define GB_TR 32   // reduction tile (MFMA k=32)
#define GB_PAD 8

// ---------------------------------------------------------------- dX
// dX[m, k] = sum_n dOut[m, n] * W[e][n, k]
// a_s: dOut tile [64 m][32 n] staged directly (n contiguous in memory).
// wt_s: W^T tile [64 k][32 n] staged transposed from W rows (k contig).
__global__ __launch_bounds__(256, 2) void grouped_gemm_dx_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ w,
    bf16* __restrict__ dx,
    const int* __restrict__ tile_expert, const int* __restrict__ tile_m0,
    const int* __restrict__ tile_k0, const long* __restrict__ seg_start,
    const int* __restrict__ seg_len, int K, int N, long w_estride) {
  const int e = tile_expert[blockIdx.x];
  const int m0 = tile_m0[blockIdx.x];
  const int k0 = tile_k0[blockIdx.x];
  const long s0 = seg_start[e];
  const int M = seg_len[e];
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int i16 = lane & 15;
  const int g = lane >> 4;

  __shared__ __bf16 a_s[GB_TM][GB_TR + GB_PAD];
  __shared__ __bf16 wt_s[GB_TC][GB_TR + GB_PAD];

  f32x4 acc[4];
  #pragma unroll
  for (int t = 0; t < 4; t++) acc[t] = {0.f, 0.f, 0.f, 0.f};

  const long wbase = (long)e * w_estride;
  for (int n0 = 0; n0 < N; n0 += GB_TR) {
    __syncthreads();
    {
      // A: 64 rows x 32 n, 8 elems/thread (n contiguous)
      int row = threadIdx.x >> 2;           // 0..63
      int c0 = (threadIdx.x & 3) * 8;       // 0,8,16,24
      bf16x8v va = {};
      if (m0 + row < M)
        va = *(const bf16x8v*)(dout + (s0 + m0 + row) * (long)N + n0 + c0);
      *(bf16x8v*)(&a_s[row][c0]) = va;
      // W^T: read 32 n-rows x 64 k (k contiguous), store transposed
      int nl = threadIdx.x >> 3;            // 0..31
      int kc0 = (threadIdx.x & 7) * 8;      // 0..56
      bf16x8v vw = *(const bf16x8v*)(w + wbase + (long)(n0 + nl) * K + k0 + kc0);
      #pragma unroll
      for (int j = 0; j < 8; j++) wt_s[kc0 + j][nl] = vw[j];
    }
    __syncthreads();
    bf16x8v afrag = *(const bf16x8v*)(&a_s[wv * 16 + i16][g * 8]);
    #pragma unroll
    for (int t = 0; t < 4; t++) {
      bf16x8v bfrag = *(const bf16x8v*)(&wt_s[t * 16 + i16][g * 8]);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[t], 0, 0, 0);
    }
  }
  #pragma unroll
  for (int t = 0; t < 4; t++) {
    #pragma unroll
    for (int r = 0; r < 4; r++) {
      int row = m0 + wv * 16 + g * 4 + r;
      if (row < M)
        dx[(s0 + row) * (long)K + k0 + t * 16 + i16] =
            __float2bfloat16(acc[t][r]);
    }
  }
}

// ---------------------------------------------------------------- dW
// dW[e][n, k] = sum_{m in seg_e} dOut[s0+m, n] * x[s0+m, k]
// Both operands are m-strided: stage both tiles transposed.
// Grid is REGULAR: blockIdx -> (e, n-tile, k-tile); the m loop walks the
// whole segment so each workgroup owns its output tile exclusively.
__global__ __launch_bounds__(256, 2) void grouped_gemm_dw_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ x,
    bf16* __restrict__ dw, const long* __restrict__ seg_start,
    const int* __restrict__ seg_len, int K, int N, int nt_n, int nt_k) {
  const int e = blockIdx.x / (nt_n * nt_k);
  const int rem = blockIdx.x % (nt_n * nt_k);
  const int n0 = (rem / nt_k) * GB_TM;
  const int k0 = (rem % nt_k) * GB_TC;
  const long s0 = seg_start[e];
  const int M = seg_len[e];
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int i16 = lane & 15;
  const int g = lane >> 4;

  __shared__ __bf16 dot_s[GB_TM][GB_TR + GB_PAD];
  __shared__ __bf16 xt_s[GB_TC][GB_TR + GB_PAD];

  f32x4 acc[4];
  #pragma unroll
  for (int t = 0; t < 4; t++) acc[t] = {0.f, 0.f, 0.f, 0.f};

  for (int m0 = 0; m0 < M; m0 += GB_TR) {
    __syncthreads();
    {
      // stage dOut^T tile [64 n][32 m] and x^T tile [64 k][32 m]:
      // read 32 m-rows (contiguous n / k), scatter-store transposed
      int ml = threadIdx.x >> 3;            // 0..31
      int c0 = (threadIdx.x & 7) * 8;       // 0..56
      bf16x8v vd = {}, vx = {};
      if (m0 + ml < M) {
        vd = *(const bf16x8v*)(dout + (s0 + m0 + ml) * (long)N + n0 + c0);
        vx = *(const bf16x8v*)(x + (s0 + m0 + ml) * (long)K + k0 + c0);
      }
      #pragma unroll
      for (int j = 0; j < 8; j++) {
        dot_s[c0 + j][ml] = vd[j];
        xt_s[c0 + j][ml] = vx[j];
      }
    }
    __syncthreads();
    bf16x8v afrag = *(const bf16x8v*)(&dot_s[wv * 16 + i16][g * 8]);
    #pragma unroll
    for (int t = 0; t < 4; t++) {
      bf16x8v bfrag = *(const bf16x8v*)(&xt_s[t * 16 + i16][g * 8]);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[t], 0, 0, 0);
    }
  }
  #pragma unroll
  for (int t = 0; t < 4; t++) {
    #pragma unroll
    for (int r = 0; r < 4; r++) {
      int n = n0 + wv * 16 + g * 4 + r;
      dw[(long)e * N * K + (long)n * K + k0 + t * 16 + i16] =
          __float2bfloat16(acc[t][r]);
    }
  }
}

// ---------------------------------------------------------------- hosts
torch::Tensor grouped_gemm_dx(torch::Tensor dout, torch::Tensor w,
                              torch::Tensor seg_lens_cpu) {
  TORCH_CHECK(dout.is_cuda() && dout.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.dim() == 3 && dout.is_contiguous());
  TORCH_CHECK(w.stride(2) == 1 && w.stride(1) == w.size(2));
  long total = dout.size(0);
  int N = dout.size(1);
  int E = w.size(0), K = w.size(2);
  TORCH_CHECK(w.size(1) == N);
  TORCH_CHECK(N % GB_TR == 0 && K % GB_TC == 0,
              "grouped_gemm_dx needs N%32==0 and K%64==0, got ", N, " ", K);
  auto lens = seg_lens_cpu.to(torch::kInt).cpu();
  const int* lp = lens.data_ptr<int>();
  std::vector<long> starts(E);
  std::vector<int> te, tm, tk;
  long off = 0;
  for (int e = 0; e < E; e++) {
    starts[e] = off;
    for (int m0 = 0; m0 < lp[e]; m0 += GB_TM)
      for (int k0 = 0; k0 < K; k0 += GB_TC) {
        te.push_back(e);
        tm.push_back(m0);
        tk.push_back(k0);
      }
    off += lp[e];
  }
  TORCH_CHECK(off == total, off, " vs ", total);
  auto dx = torch::empty({total, (long)K}, dout.options());
  if (te.empty()) return dx;
  auto te_d = torch::from_blob(te.data(), {(long)te.size()}, torch::kInt).to(dout.device());
  auto tm_d = torch::from_blob(tm.data(), {(long)tm.size()}, torch::kInt).to(dout.device());
  auto tk_d = torch::from_blob(tk.data(), {(long)tk.size()}, torch::kInt).to(dout.device());
  auto ss_d = torch::from_blob(starts.data(), {(long)E}, torch::kLong).to(dout.device());
  auto sl_d = lens.to(dout.device());
  hipLaunchKernelGGL(grouped_gemm_dx_kernel, dim3((unsigned)te.size()),
    dim3(256), 0, cur_stream(), (const bf16*)dout.data_ptr(),
    (const bf16*)w.data_ptr(), (bf16*)dx.data_ptr(), te_d.data_ptr<int>(),
    tm_d.data_ptr<int>(), tk_d.data_ptr<int>(), ss_d.data_ptr<long>(),
    sl_d.data_ptr<int>(), K, N, (long)w.stride(0));
  CHECK_CUDA_OK();
  return dx;
}

torch::Tensor grouped_gemm_dw(torch::Tensor dout, torch::Tensor x,
                              torch::Tensor seg_lens_cpu, long E) {
  TORCH_CHECK(dout.is_cuda() && dout.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(dout.is_contiguous() && x.is_contiguous());
  long total = dout.size(0);
  int N = dout.size(1), K = x.size(1);
  TORCH_CHECK(x.size(0) == total);
  TORCH_CHECK(N % GB_TM == 0 && K % GB_TC == 0,
              "grouped_gemm_dw needs N%64==0 and K%64==0, got ", N, " ", K);
  auto lens = seg_lens_cpu.to(torch::kInt).cpu();
  TORCH_CHECK(lens.numel() == E);
  const int* lp = lens.data_ptr<int>();
  std::vector<long> starts(E);
  long off = 0;
  for (long e = 0; e < E; e++) { starts[e] = off; off += lp[e]; }
  TORCH_CHECK(off == total, off, " vs ", total);
  auto dw = torch::empty({E, (long)N, (long)K}, dout.options());
  int nt_n = N / GB_TM, nt_k = K / GB_TC;
  auto ss_d = torch::from_blob(starts.data(), {E}, torch::kLong).to(dout.device());
  auto sl_d = lens.to(dout.device());
  hipLaunchKernelGGL(grouped_gemm_dw_kernel,
    dim3((unsigned)(E * nt_n * nt_k)), dim3(256), 0, cur_stream(),
    (const bf16*)dout.data_ptr(), (const bf16*)x.data_ptr(),
    (bf16*)dw.data_ptr(), ss_d.data_ptr<long>(), sl_d.data_ptr<int>(),
    K, N, nt_n, nt_k);
  CHECK_CUDA_OK();
  return dw;
}
