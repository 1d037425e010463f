// Weight-streaming skinny GEMM for the decode hot loop (gfx950).
//   out[M, N] = x[M, K] @ W[N, K]^T,  M <= 16 (decode batch)
//
// At M = 16 the GEMM is bound by streaming W once (guide §5 "GEMV / M<=16
// decode weights"); hipBLASLt's tilings reach ~55% of that roofline on
// these shapes (measured: ~120us vs 67us weight-read per 7B layer).
// Design: grid = (N/64 tiles) x SPLITK K-slices — enough workgroups to
// fill 256 CUs at every decode shape; x tile staged once in LDS; W
// streamed straight into MFMA B-fragments (16 B/lane loads, one pass,
// nothing cached); fp32 split-K partials combined with global atomics;
// a trailing kernel converts to bf16.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;

#define SG_TN 64  // n per workgroup (16 per wave)

// NB x 32-deep k body: NB 16-byte W loads in flight/lane.  WVS = waves
// per workgroup (n-tile = WVS*16): 4 (=SG_TN 64) default; 2 halves the
// tile for TAIL-BOUND small-N shapes (o-proj N=4096: 64 tiles left 88%
// of wave cycles parked in PMC — 128 tiles x half work fills the tail).
template <int NB, int WVS = 4>
__global__ __launch_bounds__(64 * WVS, 4) void skinny_gemm_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    float* __restrict__ out32, int M, int N, int K, int kslice) {
  const int ntile = blockIdx.x;
  const int ks = blockIdx.y;
  const int k_lo = ks * kslice;
  const int k_hi = min(K, k_lo + kslice);
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int i16 = lane & 15;
  const int g = lane >> 4;
  const int n0 = ntile * (WVS * 16) + wv * 16;

  extern __shared__ __bf16 x_s[];  // [16][kslice] rows (zero-padded m >= M)
  {
    const int kn = k_hi - k_lo;
    for (int idx = threadIdx.x * 8; idx < 16 * kslice; idx += 64 * WVS * 8) {
      int m = idx / kslice;
      int kk = idx % kslice;
      bf16x8v v = {};
      if (kk < kn && m < M)
        v = *(const bf16x8v*)(x + (long)m * K + k_lo + kk);
      *(bf16x8v*)(&x_s[(long)m * kslice + kk]) = v;
    }
  }
  __syncthreads();

  // 4 rotating accumulators relax the MFMA RAW chain (PMC: 35%
  // SQ_WAIT_INST_ANY with 2 accs at the qkv shape)
  f32x4 acc[4] = {};
  const bf16* wrow = w + (long)(n0 + i16) * K;
  int kk = k_lo;
  // NB*32-deep body: issue all NB W loads before the first MFMA so >= NB
  // 16B loads stay in flight per lane (one k-iter alone is latency-bound)
  // (nt loads measured 15-40% SLOWER here — keep plain loads)
  for (; kk + NB * 32 <= k_hi; kk += NB * 32) {
    bf16x8v b[NB], a[NB];
    #pragma unroll
    for (int t = 0; t < NB; t++)
      b[t] = *(const bf16x8v*)(wrow + kk + t * 32 + g * 8);
    const long xb = (long)i16 * kslice + (kk - k_lo) + g * 8;
    #pragma unroll
    for (int t = 0; t < NB; t++)
      a[t] = *(const bf16x8v*)(&x_s[xb + t * 32]);
    #pragma unroll
    for (int t = 0; t < NB; t++)
      acc[t & 3] =
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[t], b[t], acc[t & 3], 0, 0, 0);
  }
  for (; kk + 32 <= k_hi; kk += 32) {
    bf16x8v a0 = *(const bf16x8v*)(&x_s[(long)i16 * kslice + (kk - k_lo) + g * 8]);
    bf16x8v b0 = *(const bf16x8v*)(wrow + kk + g * 8);
    acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0], 0, 0, 0);
  }
  // non-atomic per-K-slice partial: out32[ks][m][n]
  #pragma unroll
  for (int r = 0; r < 4; r++) {
    int m = g * 4 + r;
    float vsum = acc[0][r] + acc[1][r] + acc[2][r] + acc[3][r];
    if (m < M)
      out32[((long)ks * M + m) * N + n0 + i16] = vsum;
  }
}

// pipeline depth: 4 (default) or 8 via REALHF_AMD_SKINNY_DEPTH=8
static int sg_depth() {
  static int d = [] {
    const char* e = getenv("REALHF_AMD_SKINNY_DEPTH");
    return (e && e[0] == '8') ? 8 : 4;
  }();
  return d;
}

// n-tile width threshold: shapes with N <= this use the 32-wide tile
// (2 waves).  Default 0 = DISABLED: the in-context A/B measured 3.55 vs
// 3.66 samples/s with it on at 4096 — doubling the workgroups also
// doubles the per-WG x-LDS stage and the tail moves, it does not
// shrink.  Kept env-gated (REALHF_AMD_SG_TN32_MAXN) for other shapes.
static int sg_tn32_maxn() {
  static int v = [] {
    const char* e = getenv("REALHF_AMD_SG_TN32_MAXN");
    return e ? atoi(e) : 0;
  }();
  return v;
}

template <typename F4, typename F8, typename F4w2, typename F8w2>
static void sg_dispatch(int N, F4 f4, F8 f8, F4w2 f4w2, F8w2 f8w2) {
  bool tn32 = N <= sg_tn32_maxn() && (N % 32) == 0;
  if (sg_depth() == 8) { if (tn32) f8w2(); else f8(); }
  else { if (tn32) f4w2(); else f4(); }
}

// combine the nks fp32 partial slabs -> bf16 (+ optional residual add)
__global__ void sg_combine_kernel(const float* __restrict__ parts,
                                  const bf16* __restrict__ resid,
                                  bf16* __restrict__ out, long n, int nks) {
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4; i < n;
       i += (long)gridDim.x * blockDim.x * 4) {
    float4v v = *(const float4v*)(parts + i);
    for (int s = 1; s < nks; s++) {
      float4v p = *(const float4v*)(parts + (long)s * n + i);
      #pragma unroll
      for (int j = 0; j < 4; j++) v[j] += p[j];
    }
    short o[4];
    if (resid) {
      short4v rv = *(const short4v*)((const short*)resid + i);
      #pragma unroll
      for (int j = 0; j < 4; j++)
        o[j] = f2bf(v[j] + __bfloat162float(((const bf16*)&rv)[j]));
    } else {
      #pragma unroll
      for (int j = 0; j < 4; j++) o[j] = f2bf(v[j]);
    }
    *(short4v*)((short*)out + i) = *(short4v*)o;
  }
}

// ---------------------------------------------------------------------------
// no-combine variant: run ONLY the split-K GEMM and hand the fp32 partial
// slabs [nks, M, N] to the NEXT kernel's prologue (launch-boundary
// reduce, guide split-K recipe: "combine in the NEXT kernel's prologue
// ... costs nothing extra when that kernel exists anyway").  Consumers:
// rope_qkv_decode / add_rmsnorm_fwd / swiglu_fwd slab modes.
torch::Tensor skinny_gemm_nc(torch::Tensor x, torch::Tensor w,
                             torch::Tensor out32_ws, long splitk) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.dim() == 2 && x.stride(1) == 1);
  TORCH_CHECK(w.dim() == 2 && w.stride(1) == 1);
  int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M <= 16 && K % 32 == 0 && N % SG_TN == 0);
  TORCH_CHECK(x.stride(0) == K, "x must be contiguous");
  int kslice = (K / (int)splitk + 63) / 64 * 64;
  int nks = (K + kslice - 1) / kslice;
  TORCH_CHECK(out32_ws.numel() >= (long)nks * M * N, "workspace too small");
  auto out32 = out32_ws.narrow(0, 0, (long)nks * M * N)
                   .view({(long)nks, (long)M, (long)N});
  size_t lds = (size_t)16 * kslice * sizeof(short);
  TORCH_CHECK(lds <= 160 * 1024, "kslice too large for LDS");
  auto launch = [&](auto nb, auto wvs) {
    dim3 grid(N / (16 * wvs.value), nks);
    hipLaunchKernelGGL((skinny_gemm_kernel<nb.value, wvs.value>), grid,
      dim3(64 * wvs.value), lds, cur_stream(),
      (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
      out32.data_ptr<float>(), M, N, K, kslice);
  };
  using c4 = std::integral_constant<int, 4>;
  using c8 = std::integral_constant<int, 8>;
  using c2 = std::integral_constant<int, 2>;
  sg_dispatch(N, [&]{ launch(c4{}, c4{}); }, [&]{ launch(c8{}, c4{}); },
              [&]{ launch(c4{}, c2{}); }, [&]{ launch(c8{}, c2{}); });
  CHECK_CUDA_OK();
  return out32;
}

// ---------------------------------------------------------------------------
// v2: split-K combine fused into the GEMM kernel via self-resetting
// semaphores — the LAST workgroup to finish an n-tile sums the fp32
// partial slabs, folds the residual, converts to bf16 and resets the
// tile's counter (so the sem buffer only needs zeroing once, at
// allocation).  Removes the separate sg_combine launch (~5 us x 4 per
// decode layer) and its extra fp32 pass.  256-deep k body keeps 8
// 16-byte W loads in flight per lane (vs 4 in v1).
__global__ __launch_bounds__(256, 4) void skinny_gemm2_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    float* __restrict__ out32, int* __restrict__ sem,
    const bf16* __restrict__ resid, bf16* __restrict__ out,
    int M, int N, int K, int kslice, int nks) {
  // 1-D grid; when the tile count divides by 8, keep all K-slices of one
  // n-tile on ONE XCD (the dispatcher places block b on XCD b%8) so the
  // reducer reads same-XCD slabs — a pure speed choice, correctness comes
  // from the agent-scope fences below (guide §6 G16 / split-K recipe).
  const int nt_total = N / SG_TN;
  int ntile, ks;
  {
    const int id = blockIdx.x;
    if ((nt_total & 7) == 0) {
      const int ix = id >> 3;
      ntile = (id & 7) * (nt_total >> 3) + ix / nks;
      ks = ix % nks;
    } else {
      ntile = id / nks;
      ks = id % nks;
    }
  }
  const int k_lo = ks * kslice;
  const int k_hi = min(K, k_lo + kslice);
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int i16 = lane & 15;
  const int g = lane >> 4;
  const int n0 = ntile * SG_TN + wv * 16;

  extern __shared__ __bf16 x_s[];  // [16][kslice] rows (zero-padded m >= M)
  {
    const int kn = k_hi - k_lo;
    for (int idx = threadIdx.x * 8; idx < 16 * kslice; idx += 256 * 8) {
      int m = idx / kslice;
      int kk = idx % kslice;
      bf16x8v v = {};
      if (kk < kn && m < M)
        v = *(const bf16x8v*)(x + (long)m * K + k_lo + kk);
      *(bf16x8v*)(&x_s[(long)m * kslice + kk]) = v;
    }
  }
  __syncthreads();

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = {0.f, 0.f, 0.f, 0.f};
  const bf16* wrow = w + (long)(n0 + i16) * K;
  int kk = k_lo;
  for (; kk + 256 <= k_hi; kk += 256) {
    bf16x8v b0 = *(const bf16x8v*)(wrow + kk + g * 8);
    bf16x8v b1 = *(const bf16x8v*)(wrow + kk + 32 + g * 8);
    bf16x8v b2 = *(const bf16x8v*)(wrow + kk + 64 + g * 8);
    bf16x8v b3 = *(const bf16x8v*)(wrow + kk + 96 + g * 8);
    bf16x8v b4 = *(const bf16x8v*)(wrow + kk + 128 + g * 8);
    bf16x8v b5 = *(const bf16x8v*)(wrow + kk + 160 + g * 8);
    bf16x8v b6 = *(const bf16x8v*)(wrow + kk + 192 + g * 8);
    bf16x8v b7 = *(const bf16x8v*)(wrow + kk + 224 + g * 8);
    const long xb = (long)i16 * kslice + (kk - k_lo) + g * 8;
    bf16x8v a0 = *(const bf16x8v*)(&x_s[xb]);
    bf16x8v a1 = *(const bf16x8v*)(&x_s[xb + 32]);
    bf16x8v a2 = *(const bf16x8v*)(&x_s[xb + 64]);
    bf16x8v a3 = *(const bf16x8v*)(&x_s[xb + 96]);
    bf16x8v a4 = *(const bf16x8v*)(&x_s[xb + 128]);
    bf16x8v a5 = *(const bf16x8v*)(&x_s[xb + 160]);
    bf16x8v a6 = *(const bf16x8v*)(&x_s[xb + 192]);
    bf16x8v a7 = *(const bf16x8v*)(&x_s[xb + 224]);
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc1, 0, 0, 0);
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a2, b2, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a3, b3, acc1, 0, 0, 0);
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a4, b4, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a5, b5, acc1, 0, 0, 0);
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a6, b6, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a7, b7, acc1, 0, 0, 0);
  }
  for (; kk + 32 <= k_hi; kk += 32) {
    bf16x8v a0 = *(const bf16x8v*)(&x_s[(long)i16 * kslice + (kk - k_lo) + g * 8]);
    bf16x8v b0 = *(const bf16x8v*)(wrow + kk + g * 8);
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc0, 0, 0, 0);
  }
  // publish the fp32 slab with plain stores
  #pragma unroll
  for (int r = 0; r < 4; r++) {
    int m = g * 4 + r;
    if (m < M)
      out32[((long)ks * M + m) * N + n0 + i16] = acc0[r] + acc1[r];
  }

  // --- in-launch split-K hand-off (guide split-K recipe, counter form):
  // every wave drains its stores; lane 0 issues ONE agent-scope release
  // (buffer_wbl2) with the post-fence wait restated (ROCm 7.2 drops it
  // otherwise), THEN takes a relaxed ticket.  NEVER __threadfence() per
  // block — that is a per-WG L2 writeback+invalidate, measured 4x slower
  // end-to-end in the decode graph.
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  volatile int* flag = (volatile int*)x_s;  // reuse the ONE shared array
  if (threadIdx.x == 0) {
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    int tick = __hip_atomic_fetch_add(&sem[ntile], 1, __ATOMIC_RELAXED,
                                      __HIP_MEMORY_SCOPE_AGENT);
    flag[0] = (tick == nks - 1) ? 1 : 0;
  }
  __syncthreads();
  if (!flag[0]) return;

  // --- last arriver reduces: one agent-scope acquire (drops this CU's
  // L1), then plain 16-B slab loads.  Thread t owns row t/16, columns
  // 4*(t%16)..+3 of the 16 x 64 tile.
  if (threadIdx.x == 0)
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  __syncthreads();
  const int base_n = ntile * SG_TN;
  const int m = threadIdx.x >> 4;
  const int c4 = (threadIdx.x & 15) * 4;
  if (m < M) {
    const long off = (long)m * N + base_n + c4;
    float4v vsum = *(const float4v*)(out32 + off);
    for (int s = 1; s < nks; s++) {
      float4v p = *(const float4v*)(out32 + (long)s * M * N + off);
      #pragma unroll
      for (int j = 0; j < 4; j++) vsum[j] += p[j];
    }
    short o4[4];
    if (resid) {
      short4v rv = *(const short4v*)((const short*)resid + off);
      #pragma unroll
      for (int j = 0; j < 4; j++)
        o4[j] = f2bf(vsum[j] + __bfloat162float(((const bf16*)&rv)[j]));
    } else {
      #pragma unroll
      for (int j = 0; j < 4; j++) o4[j] = f2bf(vsum[j]);
    }
    *(short4v*)((short*)out + off) = *(short4v*)o4;
  }
  // self-reset so the persistent sem buffer is zero for the next launch
  // (stream order makes this safe; the buffer is zeroed once at alloc)
  if (threadIdx.x == 0)
    __hip_atomic_store(&sem[ntile], 0, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
}

torch::Tensor skinny_gemm2(torch::Tensor x, torch::Tensor w,
                           torch::Tensor out32_ws, torch::Tensor sem,
                           long splitk, c10::optional<torch::Tensor> residual) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.dim() == 2 && x.stride(1) == 1);
  TORCH_CHECK(w.dim() == 2 && w.stride(1) == 1);
  int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M <= 16 && K % 32 == 0 && N % SG_TN == 0);
  TORCH_CHECK(x.stride(0) == K, "x must be contiguous");
  int kslice = (K / (int)splitk + 63) / 64 * 64;
  int nks = (K + kslice - 1) / kslice;
  TORCH_CHECK(out32_ws.numel() >= (long)nks * M * N, "workspace too small");
  TORCH_CHECK(sem.scalar_type() == torch::kInt32 && sem.numel() >= N / SG_TN,
              "semaphore buffer too small (must be zero-initialized once)");
  auto out = torch::empty({M, (long)N}, x.options());
  const bf16* rptr = nullptr;
  if (residual.has_value()) {
    TORCH_CHECK(residual->is_contiguous() &&
                residual->numel() == (long)M * N);
    rptr = (const bf16*)residual->data_ptr();
  }
  dim3 grid((N / SG_TN) * nks);  // 1-D: kernel derives (tile, slice)
  size_t lds = (size_t)16 * kslice * sizeof(short);
  TORCH_CHECK(lds <= 160 * 1024, "kslice too large for LDS");
  hipLaunchKernelGGL(skinny_gemm2_kernel, grid, dim3(256), lds, cur_stream(),
    (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
    out32_ws.data_ptr<float>(), sem.data_ptr<int>(), rptr,
    (bf16*)out.data_ptr(), M, N, K, kslice, nks);
  CHECK_CUDA_OK();
  return out;
}

torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w,
                          torch::Tensor out32_ws, long splitk,
                          c10::optional<torch::Tensor> residual) {
  // x [M, K] bf16; w [N, K] bf16 (row-major view, stride(1)==1);
  // out32_ws: fp32 workspace >= splitk * M * N (per-slice partials —
  // no zeroing, no atomics); residual (optional [M, N] bf16) is folded
  // into the combine kernel.
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.dim() == 2 && x.stride(1) == 1);
  TORCH_CHECK(w.dim() == 2 && w.stride(1) == 1);
  int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M <= 16 && K % 32 == 0 && N % SG_TN == 0);
  TORCH_CHECK(x.stride(0) == K, "x must be contiguous");
  int kslice = (K / (int)splitk + 63) / 64 * 64;
  int nks = (K + kslice - 1) / kslice;
  TORCH_CHECK(out32_ws.numel() >= (long)nks * M * N, "workspace too small");
  auto out32 = out32_ws.narrow(0, 0, (long)nks * M * N);
  size_t lds = (size_t)16 * kslice * sizeof(short);
  TORCH_CHECK(lds <= 160 * 1024, "kslice too large for LDS");
  auto launch = [&](auto nb, auto wvs) {
    dim3 grid(N / (16 * wvs.value), nks);
    hipLaunchKernelGGL((skinny_gemm_kernel<nb.value, wvs.value>), grid,
      dim3(64 * wvs.value), lds, cur_stream(),
      (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
      out32.data_ptr<float>(), M, N, K, kslice);
  };
  using c4 = std::integral_constant<int, 4>;
  using c8 = std::integral_constant<int, 8>;
  using c2 = std::integral_constant<int, 2>;
  sg_dispatch(N, [&]{ launch(c4{}, c4{}); }, [&]{ launch(c8{}, c4{}); },
              [&]{ launch(c4{}, c2{}); }, [&]{ launch(c8{}, c2{}); });
  auto out = torch::empty({M, (long)N}, x.options());
  long n = (long)M * N;
  int cgrid = (int)std::min<long>((n / 4 + 255) / 256, 2048);
  const bf16* rptr = nullptr;
  if (residual.has_value()) {
    TORCH_CHECK(residual->is_contiguous() && residual->numel() == n);
    rptr = (const bf16*)residual->data_ptr();
  }
  hipLaunchKernelGGL(sg_combine_kernel, dim3(cgrid), dim3(256), 0,
    cur_stream(), out32.data_ptr<float>(), rptr, (bf16*)out.data_ptr(), n,
    nks);
  CHECK_CUDA_OK();
  return out;
}
