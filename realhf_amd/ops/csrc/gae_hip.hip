#include "hip/hip_runtime.h"
// GAE advantage/return scan over packed varlen sequences (gfx950).
// Reference semantics: csrc/cugae/gae.cu:10 gae_kernel_1d_nolp_misalign —
// one thread per sequence, sequential backward recursion
//   delta_t = r_t + gamma * V_{t+1} - V_t;  A_t = delta_t + gamma*lam*A_{t+1}
// values are per-token [total_r + bs] (one longer than rewards per seq:
// value slots of seq i live at [cu[i] + i, cu[i+1] + i + 1)).
#include "common.h"

__global__ void gae_1d_kernel(
    const float* __restrict__ rewards, const float* __restrict__ values,
    const int* __restrict__ cu_seqlens, const bool* __restrict__ bootstrap,
    float* __restrict__ adv, float* __restrict__ ret,
    int bs, float gamma, float lam) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= bs) return;
  int rs = cu_seqlens[i], re = cu_seqlens[i + 1];
  int vs = rs + i;
  int L = re - rs;
  float lastgae = 0.f;
  for (int t = L - 1; t >= 0; t--) {
    float nex = values[vs + t + 1];
    if (t == L - 1 && !bootstrap[i]) nex = 0.f;
    float delta = rewards[rs + t] + gamma * nex - values[vs + t];
    lastgae = delta + gamma * lam * lastgae;
    adv[rs + t] = lastgae;
    ret[rs + t] = lastgae + values[vs + t];
  }
}

std::vector<torch::Tensor> gae_1d(torch::Tensor rewards, torch::Tensor values,
                                  torch::Tensor cu_seqlens, torch::Tensor bootstrap,
                                  double gamma, double lam) {
  TORCH_CHECK(rewards.is_cuda() && rewards.scalar_type() == torch::kFloat);
  TORCH_CHECK(values.scalar_type() == torch::kFloat);
  int bs = cu_seqlens.numel() - 1;
  auto cu = cu_seqlens.to(torch::kInt).contiguous();
  auto bst = bootstrap.to(torch::kBool).contiguous();
  auto adv = torch::zeros_like(rewards);
  auto ret = torch::zeros_like(rewards);
  int grid = (bs + 255) / 256;
  hipLaunchKernelGGL(gae_1d_kernel, dim3(grid), dim3(256), 0, cur_stream(),
    rewards.data_ptr<float>(), values.data_ptr<float>(), cu.data_ptr<int>(),
    bst.data_ptr<bool>(), adv.data_ptr<float>(), ret.data_ptr<float>(),
    bs, (float)gamma, (float)lam);
  CHECK_CUDA_OK();
  return {adv, ret};
}
