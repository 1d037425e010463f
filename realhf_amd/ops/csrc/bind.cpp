// pybind bindings for realhf_amd._C — the hand-written gfx950 kernels.
#include <torch/extension.h>
#include <vector>

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd);
std::vector<torch::Tensor> add_rmsnorm_fwd(torch::Tensor x, torch::Tensor resid,
                                           torch::Tensor w, double eps);
torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cosb, torch::Tensor sinb,
                       torch::Tensor positions, bool interleaved, bool conj);
torch::Tensor swiglu_fwd(torch::Tensor gu);
torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor gu);
std::vector<torch::Tensor> gae_1d(torch::Tensor rewards, torch::Tensor values,
                                  torch::Tensor cu_seqlens, torch::Tensor bootstrap,
                                  double gamma, double lam);
torch::Tensor slice_intervals(torch::Tensor src, torch::Tensor intervals);
void set_intervals(torch::Tensor src, torch::Tensor dst, torch::Tensor intervals);
void fused_adamw(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                 torch::Tensor v, torch::Tensor out_bf16, double lr, double b1,
                 double b2, double eps, double wd, long step, double gscale,
                 bool write_bf16);
torch::Tensor attn_decode(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                          torch::Tensor cache_seqlens, double scale,
                          long window);
std::vector<torch::Tensor> attn_varlen_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor cu_seqlens, long max_seqlen, bool causal, double scale,
    long window);
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B);
torch::Tensor tr16_probe(torch::Tensor addr_elem);
torch::Tensor grouped_gemm(torch::Tensor x, torch::Tensor w,
                           torch::Tensor seg_lens_cpu);
torch::Tensor grouped_gemm_dx(torch::Tensor dout, torch::Tensor w,
                              torch::Tensor seg_lens_cpu);
torch::Tensor grouped_gemm_dw(torch::Tensor dout, torch::Tensor x,
                              torch::Tensor seg_lens_cpu, long E);
torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w,
                          torch::Tensor out32_ws, long splitk,
                          c10::optional<torch::Tensor> residual);
torch::Tensor skinny_gemm_nc(torch::Tensor x, torch::Tensor w,
                             torch::Tensor out32_ws, long splitk);
std::vector<torch::Tensor> attn_varlen_bwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, torch::Tensor dout,
    torch::Tensor lse, torch::Tensor Dsum, torch::Tensor cu_seqlens,
    bool causal, double scale, long window);
int64_t xgmi_create(int64_t rank, int64_t world, int64_t capacity);
std::vector<py::bytes> xgmi_handles(int64_t h);
void xgmi_connect(int64_t h, const std::vector<std::string>& data_handles,
                  const std::vector<std::string>& sig_handles);
torch::Tensor xgmi_all_reduce(int64_t h, torch::Tensor t);
int64_t xgmi_status(int64_t h);
void xgmi_destroy(int64_t h);
torch::Tensor skinny_gemm2(torch::Tensor x, torch::Tensor w,
                           torch::Tensor out32_ws, torch::Tensor sem,
                           long splitk, c10::optional<torch::Tensor> residual);
std::pair<std::vector<int64_t>, double> mcmc_search(
    int64_t n_gpus, std::vector<std::vector<std::vector<double>>> cand_rows,
    std::vector<std::vector<int64_t>> parents_in,
    std::vector<int64_t> role_in, std::vector<double> realloc_cost_in,
    int64_t n_strategies, double mem_cap_bytes, int64_t n_chains,
    int64_t n_steps, int64_t seed);
torch::Tensor rope_qkv_decode(
    torch::Tensor qkv, c10::optional<torch::Tensor> bias, torch::Tensor kcache,
    torch::Tensor vcache, torch::Tensor cache_seqlens, torch::Tensor cosb,
    torch::Tensor sinb, long nq, bool apply_rope);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("add_rmsnorm_fwd", &add_rmsnorm_fwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("gae_1d", &gae_1d);
  m.def("slice_intervals", &slice_intervals);
  m.def("set_intervals", &set_intervals);
  m.def("fused_adamw", &fused_adamw);
  m.def("attn_decode", &attn_decode);
  m.def("attn_varlen_fwd", &attn_varlen_fwd);
  m.def("mfma_probe", &mfma_probe);
  m.def("tr16_probe", &tr16_probe);
  m.def("rope_qkv_decode", &rope_qkv_decode);
  m.def("grouped_gemm", &grouped_gemm);
  m.def("grouped_gemm_dx", &grouped_gemm_dx);
  m.def("grouped_gemm_dw", &grouped_gemm_dw);
  m.def("skinny_gemm", &skinny_gemm);
  m.def("skinny_gemm2", &skinny_gemm2);
  m.def("skinny_gemm_nc", &skinny_gemm_nc);
  m.def("attn_varlen_bwd", &attn_varlen_bwd);
  m.def("xgmi_create", &xgmi_create);
  m.def("xgmi_handles", &xgmi_handles);
  m.def("xgmi_connect", &xgmi_connect);
  m.def("xgmi_all_reduce", &xgmi_all_reduce);
  m.def("xgmi_status", &xgmi_status);
  m.def("xgmi_destroy", &xgmi_destroy);
  m.def("mcmc_search", &mcmc_search);
}
