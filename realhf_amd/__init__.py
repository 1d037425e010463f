"""realhf_amd — an MI355X-native RLHF training framework.

A from-scratch system with the capabilities of openpsi-project/ReaLHF
(dataflow-graph RLHF with per-call parallelism + parameter reallocation),
re-designed for AMD MI355X (gfx950, CDNA4):

- PyTorch-ROCm compute path, RCCL (``torch.distributed`` backend "nccl")
  over xGMI for every collective, one process per GPU.
- Hand-written HIP/CDNA4 kernels (MFMA + LDS tiling) for the hot ops:
  flash-attention prefill/decode, RMSNorm, RoPE, SwiGLU, fused AdamW,
  GAE, flat-parameter interval gather/scatter, grouped GEMM for MoE.
- A flat-parameter transformer whose layout contract makes parameter
  reallocation (resharding between per-call 3D-parallel strategies) a
  pure interval-math + collective problem.
- An SPMD dataflow-graph executor: every rank runs the same deterministic
  plan; MFC-to-MFC data movement and weight resharding are collectives on
  the 8-GPU xGMI mesh (reference ReaLHF uses a ZMQ master/worker runtime,
  a multi-node design this single-node-first system does not need in the
  hot path; see realhf_amd/runtime/).

Reference feature map: SURVEY.md (repo root).
"""

__version__ = "0.1.0"
