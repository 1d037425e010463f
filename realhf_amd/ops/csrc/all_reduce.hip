// Custom intra-node all-reduce over xGMI peer mappings (gfx950).
//
// Reference counterpart: csrc/custom_all_reduce/custom_all_reduce.cuh
// (vLLM-style CUDA-IPC allreduce; declared legacy/unused by the reference
// — docs/source/arch.rst:83-85).  Rebuilt MI355X-native because all 8
// GPUs of an MI355X node are fully connected point-to-point by xGMI, so
// a 1-stage direct-read sum beats a ring for small TP messages (the ring
// pays 2(n-1) hops of latency; direct-read pays one).
//
// Design:
//  - each rank owns one IPC-shared DATA buffer (fine-grained, so peer
//    loads over xGMI are coherent without remote-L2 games) and one
//    SIGNAL buffer (uncached) holding [MAXBLOCKS][world] generation
//    flags
//  - barrier: block b of rank r writes the call's generation counter to
//    peer p's flags[b][r] (system-scope release), then spins (bounded!)
//    on its own flags[b][p]; generations increase monotonically so flags
//    are never reset
//  - 1-stage reduce: out[i] = sum_p data[p][i]; every block barriers
//    before reading (peers' copies done) and after (safe to overwrite)
//  - spins are BOUNDED; on timeout the kernel writes an error code into
//    the signal buffer's status word instead of hanging the device
#include "common.h"

#define XAR_MAX_WORLD 8
#define XAR_MAX_BLOCKS 64
#define XAR_SPIN_LIMIT (1ull << 33)  // ~3.5s at 2.4GHz — then bail

struct XgmiComm {
  int rank, world;
  long capacity;  // bytes per data buffer
  void* data[XAR_MAX_WORLD];
  unsigned* flags[XAR_MAX_WORLD];  // [XAR_MAX_BLOCKS][world] + status word
  unsigned gen;
  bool opened[XAR_MAX_WORLD];
};

#define XAR_STATUS_SLOT (XAR_MAX_BLOCKS * XAR_MAX_WORLD)

__device__ __forceinline__ void xar_store_sys(unsigned* p, unsigned v) {
  __hip_atomic_store(p, v, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}

__device__ __forceinline__ unsigned xar_load_sys(const unsigned* p) {
  return __hip_atomic_load(p, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
}

// Arrival-only barrier (no data hand-off): returns false on spin timeout
// (status word set); callers must exit.
__device__ bool xar_barrier(unsigned* const* flags, int rank, int world,
                            unsigned gen) {
  const int b = blockIdx.x;
  if (threadIdx.x < (unsigned)world) {
    const int p = threadIdx.x;
    // publish my arrival to peer p
    xar_store_sys(&flags[p][b * XAR_MAX_WORLD + rank], gen);
    // wait for peer p's arrival in MY flag page
    unsigned long long spins = 0;
    while (xar_load_sys(&flags[rank][b * XAR_MAX_WORLD + p]) < gen) {
      if (++spins > XAR_SPIN_LIMIT) {
        xar_store_sys(&flags[rank][XAR_STATUS_SLOT], 0xdead);
        return false;
      }
      __builtin_amdgcn_s_sleep(16);
    }
  }
  __syncthreads();
  return true;
}

// Data-publishing barrier (guide §6 G16 recipe): EVERY wave drains its
// stores (vmcnt is per-wave — lane 0's fence alone cannot see other
// waves' in-flight writes), lane 0 issues a system-scope release with
// the post-fence wait restated, THEN the arrival exchange; after the
// spin, lane 0 does the system-scope acquire so this block's plain
// loads see every peer's published data.
__device__ bool xar_barrier_publish(unsigned* const* flags, int rank,
                                    int world, unsigned gen) {
  __shared__ int bar_fail;
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    bar_fail = 0;
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __syncthreads();
  const int b = blockIdx.x;
  if (threadIdx.x < (unsigned)world) {
    const int p = threadIdx.x;
    xar_store_sys(&flags[p][b * XAR_MAX_WORLD + rank], gen);
    unsigned long long spins = 0;
    while (xar_load_sys(&flags[rank][b * XAR_MAX_WORLD + p]) < gen) {
      if (++spins > XAR_SPIN_LIMIT) {
        xar_store_sys(&flags[rank][XAR_STATUS_SLOT], 0xdead);
        bar_fail = 1;
        break;
      }
      __builtin_amdgcn_s_sleep(16);
    }
  }
  __syncthreads();
  if (threadIdx.x == 0)
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  __syncthreads();
  return bar_fail == 0;
}

template <typename T>
__global__ void xar_1stage_kernel(XgmiComm c, T* __restrict__ out, long n,
                                  unsigned gen) {
  // phase 1: everyone's copy-in is complete once the barrier passes
  if (!xar_barrier(c.flags, c.rank, c.world, gen)) return;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8; i < n;
       i += stride) {
    float acc[8] = {};
    for (int p = 0; p < c.world; p++) {
      const T* src = (const T*)c.data[p] + i;
      short8 v8;
      float4v f4a, f4b;
      if constexpr (sizeof(T) == 2) {
        v8 = *(const short8*)src;
        #pragma unroll
        for (int j = 0; j < 8; j++) acc[j] += to_f32<T>(((const T*)&v8)[j]);
      } else {
        f4a = *(const float4v*)((const float*)src);
        f4b = *(const float4v*)((const float*)src + 4);
        #pragma unroll
        for (int j = 0; j < 4; j++) { acc[j] += f4a[j]; acc[4 + j] += f4b[j]; }
      }
    }
    if constexpr (sizeof(T) == 2) {
      short o[8];
      #pragma unroll
      for (int j = 0; j < 8; j++) ((T*)o)[j] = from_f32<T>(acc[j]);
      *(short8*)(out + i) = *(short8*)o;
    } else {
      #pragma unroll
      for (int j = 0; j < 8; j++) ((float*)out)[i + j] = acc[j];
    }
  }
  // phase 2: done reading every peer's buffer — safe to overwrite next call
  xar_barrier(c.flags, c.rank, c.world, gen + 1);
}

// 2-stage (reduce-scatter + all-gather) for LARGE messages: rank r sums
// chunk r from every peer into its buffer's upper region, barrier, then
// every rank gathers the reduced chunks.  Traffic ~2n per rank instead
// of the 1-stage's world*n — wins once n is past the latency regime.
// Layout: region A = [0, cap/2) holds the input copy; region B =
// [cap/2, cap) holds this rank's reduced chunk.
template <typename T>
__global__ void xar_2stage_kernel(XgmiComm c, T* __restrict__ out, long n,
                                  long boff_elems, unsigned gen) {
  const long cn = n / c.world;  // chunk elems (host pads to world*8)
  const long my0 = (long)c.rank * cn;
  if (!xar_barrier(c.flags, c.rank, c.world, gen)) return;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  // stage 1: reduce my chunk from all peers -> my region B
  T* myB = (T*)c.data[c.rank] + boff_elems;
  for (long i = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8; i < cn;
       i += stride) {
    float acc[8] = {};
    for (int p = 0; p < c.world; p++) {
      const T* src = (const T*)c.data[p] + my0 + i;
      if constexpr (sizeof(T) == 2) {
        short8 v8 = *(const short8*)src;
        #pragma unroll
        for (int j = 0; j < 8; j++) acc[j] += to_f32<T>(((const T*)&v8)[j]);
      } else {
        #pragma unroll
        for (int j = 0; j < 8; j++) acc[j] += ((const float*)src)[j];
      }
    }
    if constexpr (sizeof(T) == 2) {
      short o[8];
      #pragma unroll
      for (int j = 0; j < 8; j++) ((T*)o)[j] = from_f32<T>(acc[j]);
      *(short8*)(myB + i) = *(short8*)o;
    } else {
      #pragma unroll
      for (int j = 0; j < 8; j++) ((float*)myB)[i + j] = acc[j];
    }
  }
  if (!xar_barrier_publish(c.flags, c.rank, c.world, gen + 1)) return;
  // stage 2: gather every rank's reduced chunk.  CRITICAL: the barrier is
  // pairwise per-BLOCK (block b syncs with block b of each peer), so
  // block b may only read intra-chunk offsets that block b itself wrote
  // in stage 1 — the loops below mirror stage 1's offset mapping exactly
  // (a full-chunk read pattern raced with peers' other blocks: measured
  // as zeroed stretches of the peer chunk under load).
  for (int p = 0; p < c.world; p++) {
    const T* srcB = (const T*)c.data[p] + boff_elems;
    T* outp = out + (long)p * cn;
    for (long o = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8; o < cn;
         o += stride) {
      if constexpr (sizeof(T) == 2) {
        *(short8*)(outp + o) = *(const short8*)(srcB + o);
      } else {
        #pragma unroll
        for (int j = 0; j < 8; j++)
          ((float*)outp)[o + j] = ((const float*)srcB)[o + j];
      }
    }
  }
  xar_barrier(c.flags, c.rank, c.world, gen + 2);
}

// ---------------------------------------------------------------------------
// host side
// ---------------------------------------------------------------------------
#include <hip/hip_runtime_api.h>
#include <vector>

static void xar_check(hipError_t e, const char* what) {
  TORCH_CHECK(e == hipSuccess, what, ": ", hipGetErrorString(e));
}

int64_t xgmi_create(int64_t rank, int64_t world, int64_t capacity) {
  TORCH_CHECK(world >= 1 && world <= XAR_MAX_WORLD);
  auto* c = new XgmiComm();
  memset(c, 0, sizeof(*c));
  c->rank = (int)rank;
  c->world = (int)world;
  c->capacity = capacity;
  c->gen = 0;
  // fine-grained data buffer: peer loads over xGMI are coherent
  xar_check(hipExtMallocWithFlags(&c->data[rank], capacity,
                                  hipDeviceMallocFinegrained),
            "alloc xgmi data");
  size_t sigbytes = (XAR_STATUS_SLOT + 1) * sizeof(unsigned);
  void* sig;
  xar_check(hipExtMallocWithFlags(&sig, sigbytes, hipDeviceMallocUncached),
            "alloc xgmi signals");
  xar_check(hipMemset(sig, 0, sigbytes), "zero signals");
  c->flags[rank] = (unsigned*)sig;
  return (int64_t)(intptr_t)c;
}

std::vector<py::bytes> xgmi_handles(int64_t h) {
  auto* c = (XgmiComm*)(intptr_t)h;
  hipIpcMemHandle_t hd, hs;
  xar_check(hipIpcGetMemHandle(&hd, c->data[c->rank]), "ipc data handle");
  xar_check(hipIpcGetMemHandle(&hs, c->flags[c->rank]), "ipc signal handle");
  return {py::bytes((const char*)&hd, sizeof(hd)),
          py::bytes((const char*)&hs, sizeof(hs))};
}

void xgmi_connect(int64_t h, const std::vector<std::string>& data_handles,
                  const std::vector<std::string>& sig_handles) {
  auto* c = (XgmiComm*)(intptr_t)h;
  TORCH_CHECK((int)data_handles.size() == c->world);
  for (int p = 0; p < c->world; p++) {
    if (p == c->rank) continue;
    hipIpcMemHandle_t hd, hs;
    TORCH_CHECK(data_handles[p].size() == sizeof(hd));
    memcpy(&hd, data_handles[p].data(), sizeof(hd));
    memcpy(&hs, sig_handles[p].data(), sizeof(hs));
    xar_check(hipIpcOpenMemHandle(&c->data[p], hd,
                                  hipIpcMemLazyEnablePeerAccess),
              "open peer data");
    void* sp;
    xar_check(hipIpcOpenMemHandle(&sp, hs, hipIpcMemLazyEnablePeerAccess),
              "open peer signals");
    c->flags[p] = (unsigned*)sp;
    c->opened[p] = true;
  }
}

torch::Tensor xgmi_all_reduce(int64_t h, torch::Tensor t) {
  auto* c = (XgmiComm*)(intptr_t)h;
  TORCH_CHECK(t.is_cuda() && t.is_contiguous());
  long bytes = t.numel() * t.element_size();
  TORCH_CHECK(t.numel() % 8 == 0, "numel must be a multiple of 8");
  static long two_stage_min = [] {
    const char* e = getenv("REALHF_AMD_XGMI_2STAGE_BYTES");
    return e ? atol(e) : (long)(512 << 10);
  }();
  // 2-stage needs region B (upper half) + a world-divisible chunking
  bool two_stage = c->world > 1 && bytes >= two_stage_min &&
                   bytes <= c->capacity / 2 &&
                   t.numel() % ((long)c->world * 8) == 0;
  TORCH_CHECK(two_stage || bytes <= c->capacity,
              "tensor larger than xgmi buffer");
  auto stream = cur_stream();
  xar_check(hipMemcpyAsync(c->data[c->rank], t.data_ptr(), bytes,
                           hipMemcpyDeviceToDevice, stream),
            "copy-in");
  auto out = torch::empty_like(t);
  unsigned gen = c->gen + 1;
  c->gen += two_stage ? 3 : 2;
  long vec = t.numel() / 8;
  int grid = (int)std::min<long>((vec + 255) / 256, XAR_MAX_BLOCKS);
  long boff_elems = (c->capacity / 2) / t.element_size();
  if (t.scalar_type() == torch::kBFloat16) {
    if (two_stage)
      hipLaunchKernelGGL((xar_2stage_kernel<bf16>), dim3(grid), dim3(256), 0,
                         stream, *c, (bf16*)out.data_ptr(), t.numel(),
                         boff_elems, gen);
    else
      hipLaunchKernelGGL((xar_1stage_kernel<bf16>), dim3(grid), dim3(256), 0,
                         stream, *c, (bf16*)out.data_ptr(), t.numel(), gen);
  } else if (t.scalar_type() == torch::kFloat) {
    if (two_stage)
      hipLaunchKernelGGL((xar_2stage_kernel<float>), dim3(grid), dim3(256), 0,
                         stream, *c, (float*)out.data_ptr(), t.numel(),
                         boff_elems, gen);
    else
      hipLaunchKernelGGL((xar_1stage_kernel<float>), dim3(grid), dim3(256), 0,
                         stream, *c, (float*)out.data_ptr(), t.numel(), gen);
  } else {
    TORCH_CHECK(false, "xgmi_all_reduce supports bf16/fp32");
  }
  CHECK_CUDA_OK();
  return out;
}

int64_t xgmi_status(int64_t h) {
  auto* c = (XgmiComm*)(intptr_t)h;
  unsigned st = 0;
  xar_check(hipMemcpy(&st, c->flags[c->rank] + XAR_STATUS_SLOT,
                      sizeof(unsigned), hipMemcpyDeviceToHost),
            "status read");
  return (int64_t)st;
}

void xgmi_destroy(int64_t h) {
  auto* c = (XgmiComm*)(intptr_t)h;
  for (int p = 0; p < c->world; p++) {
    if (c->opened[p]) {
      (void)hipIpcCloseMemHandle(c->data[p]);
      (void)hipIpcCloseMemHandle(c->flags[p]);
    }
  }
  (void)hipFree(c->data[c->rank]);
  (void)hipFree(c->flags[c->rank]);
  delete c;
}
