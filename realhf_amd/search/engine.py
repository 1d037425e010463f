"""Allocation search: enumerate feasible per-MFC strategies and pick the
assignment minimizing the simulated DFG makespan via the C++ MCMC.

Reference semantics: realhf/search_engine/ (search.py:25
search_rpc_allocations, enumerate.py:15 enumerate_rpc_executions,
estimate.py cost models) + csrc/search multi_mcmc_search.

Cost model (analytic, MI355X-calibrated — the reference uses profiled
per-op tables; our estimates use measured end-to-end efficiencies from
bench runs on this hardware):
  train:     3 x fwd FLOPs at ~35% of bf16 dense peak + optimizer sweep
  inference: fwd FLOPs at ~55% of peak
  generate:  prefill (as inference) + per-token weight streaming at
             ~65% of HBM bandwidth + fixed per-token overhead
  realloc:   moved bytes / (n_links x 150 GB/s) + plan latency
Memory model mirrors parallel/ddp.py + generation KV sizing.
"""
import dataclasses
from typing import Dict, List, Tuple

from realhf_amd.api.config import ModelInterfaceType, ParallelismConfig
from realhf_amd.api.dfg import DFG
from realhf_amd.base import logging

logger = logging.getLogger("search")

BF16_PEAK_TF = 2495.0
HBM_GBPS = 6300.0
XGMI_LINK_GBPS = 150.0

TRAIN_EFF = 0.35
INF_EFF = 0.55
GEN_BW_EFF = 0.65
GEN_TOKEN_OVERHEAD_S = 2.0e-3 / 512  # residual per-token fixed cost


def load_cost_table(path: str):
    """Calibrate the analytic cost model from measured numbers
    (tools/profile_layers.py output; reference counterpart: the search
    engine's ProfileLayers-produced op-cost tables, search_engine/
    layers.py:56).  Recognized keys: bf16_tf, hbm_gbps, xgmi_link_gbps,
    train_eff, inf_eff, gen_bw_eff."""
    import json

    global BF16_PEAK_TF, HBM_GBPS, XGMI_LINK_GBPS
    global TRAIN_EFF, INF_EFF, GEN_BW_EFF
    with open(path) as f:
        t = json.load(f)
    BF16_PEAK_TF = t.get("bf16_tf", BF16_PEAK_TF)
    HBM_GBPS = t.get("hbm_gbps", HBM_GBPS)
    XGMI_LINK_GBPS = t.get("xgmi_link_gbps", XGMI_LINK_GBPS)
    TRAIN_EFF = t.get("train_eff", TRAIN_EFF)
    INF_EFF = t.get("inf_eff", INF_EFF)
    GEN_BW_EFF = t.get("gen_bw_eff", GEN_BW_EFF)
    logger.info("cost model calibrated from %s: tf=%.0f hbm=%.0f", path,
                BF16_PEAK_TF, HBM_GBPS)


import os as _os  # noqa: E402

if _os.environ.get("REALHF_AMD_COST_TABLE"):
    load_cost_table(_os.environ["REALHF_AMD_COST_TABLE"])
else:
    # measured-on-MI355X defaults shipped with the package
    _default = _os.path.join(_os.path.dirname(__file__), "..", "data",
                             "cost_table_gfx950.json")
    if _os.path.exists(_default):
        load_cost_table(_default)


@dataclasses.dataclass
class MFCSpec:
    name: str
    role: str
    interface_type: ModelInterfaceType
    # per-iteration work (whole job)
    n_seqs: int
    avg_seqlen: int
    gen_tokens: int = 0  # for GENERATE
    param_bytes: float = 0.0  # full model, bf16
    flops_per_token: float = 0.0  # fwd
    # memory-model inputs (reference: estimate_rpc_memory_cost,
    # estimate.py:387-450)
    n_layers: int = 32
    hidden_dim: int = 4096
    n_kv_heads: int = 8
    head_dim: int = 128
    gradient_checkpointing: bool = True
    offload_optimizer: bool = False
    n_minibatches: int = 1


def enumerate_strategies(n_gpus: int) -> List[ParallelismConfig]:
    out = []
    for tp in (1, 2, 4, 8):
        for pp in (1, 2, 4, 8):
            for dp in (1, 2, 4, 8):
                if dp * tp * pp <= n_gpus and tp <= 8:
                    out.append(ParallelismConfig(dp, tp, pp))
    return out


def estimate_time_s(m: MFCSpec, par: ParallelismConfig) -> float:
    world = par.world_size
    tokens = m.n_seqs * m.avg_seqlen
    fwd_flops = tokens * m.flops_per_token
    if m.interface_type == ModelInterfaceType.TRAIN_STEP:
        t = 3 * fwd_flops / (BF16_PEAK_TF * 1e12 * TRAIN_EFF) / world
        # optimizer sweep: ~7 fp32 passes over the sharded master
        t += 7 * (m.param_bytes * 2) / (HBM_GBPS * 1e9) / world
        # pp bubble
        t *= 1.0 + (par.pipeline_parallel_size - 1) / (2.0 * par.pipeline_parallel_size)
        return t
    if m.interface_type == ModelInterfaceType.INFERENCE:
        return fwd_flops / (BF16_PEAK_TF * 1e12 * INF_EFF) / world
    # GENERATE: prefill + decode weight-streaming
    prefill = fwd_flops / (BF16_PEAK_TF * 1e12 * INF_EFF) / world
    shard_bytes = m.param_bytes / (par.tensor_parallel_size * par.pipeline_parallel_size)
    per_tok = shard_bytes / (HBM_GBPS * 1e9 * GEN_BW_EFF) + GEN_TOKEN_OVERHEAD_S
    # decode is sequential in tokens; dp splits the batch without
    # shortening the token loop; pp pipelines it partially
    return prefill + m.gen_tokens * per_tok * (
        1.0 + 0.1 * (par.pipeline_parallel_size - 1)
    )


def estimate_mem_bytes(m: MFCSpec, par: ParallelismConfig, trainable: bool) -> float:
    """Peak device bytes of one MFC execution — params/grads/optimizer
    (ZeRO-1 over dp, optionally host-offloaded), activations (with the
    grad-checkpoint discount), and the decode KV cache for GENERATE
    (reference: estimate_rpc_memory_cost, estimate.py:387-450 — extended
    with the KV-cache/offload terms the reference marks TODO)."""
    tp, pp, dp = (par.tensor_parallel_size, par.pipeline_parallel_size,
                  par.data_parallel_size)
    shard = m.param_bytes / (tp * pp)
    seqs_per_dp = max(1, m.n_seqs // dp)
    mem = shard  # bf16 params
    if trainable:
        mem += shard  # bf16 grads
        if not m.offload_optimizer:
            mem += shard * 6 / dp  # fp32 master + m + v over ZeRO dp
        # activations of one microbatch: checkpointing keeps ~2 tensors
        # per layer boundary, else ~14 per layer (attn+mlp intermediates)
        mb_tokens = seqs_per_dp * m.avg_seqlen / max(1, m.n_minibatches)
        per_tok = m.hidden_dim * 2 * (2 * pp if m.gradient_checkpointing
                                      else 14 * m.n_layers / pp)
        mem += mb_tokens * per_tok / tp
    elif m.interface_type == ModelInterfaceType.GENERATE:
        # contiguous KV caches [bs/dp, prompt+gen, nkv/tp, hd] x 2 x layers
        cache_len = m.avg_seqlen + m.gen_tokens
        kv = (seqs_per_dp * cache_len * max(1, m.n_kv_heads // tp)
              * m.head_dim * 2 * 2 * m.n_layers / pp)
        mem += kv
        mem += seqs_per_dp * m.hidden_dim * 2 * 4  # decode activations
    else:
        mem += seqs_per_dp * m.avg_seqlen * m.hidden_dim * 2 * 4 / tp
    return mem


def search_allocations(
    graph: DFG,
    mfc_specs: Dict[str, MFCSpec],
    trainable_roles: List[str],
    n_gpus: int = 8,
    mem_cap_gib: float = 280.0,
    n_chains: int = 8,
    n_steps: int = 20000,
    seed: int = 1,
) -> Tuple[Dict[str, ParallelismConfig], float]:
    """Returns {mfc name: strategy} minimizing estimated step makespan."""
    import realhf_amd._C as C

    strategies = enumerate_strategies(n_gpus)
    sid = {i: s for i, s in enumerate(strategies)}

    order = [m.name for m in graph.topological_order()]
    idx = {n: i for i, n in enumerate(order)}
    roles = sorted({mfc_specs[n].role for n in order})
    role_id = {r: i for i, r in enumerate(roles)}

    cand_rows, parents, role_vec = [], [], []
    cand_sids = []
    for n in order:
        m = mfc_specs[n]
        trainable = m.role in trainable_roles
        row, sids = [], []
        for i, s in enumerate(strategies):
            t = estimate_time_s(m, s)
            mem = estimate_mem_bytes(m, s, trainable)
            # contiguous mesh of s.world_size GPUs starting at 0 (single
            # node, symmetric links: offset choice does not change cost)
            mesh = (1 << s.world_size) - 1
            row.append([float(mesh), t, mem, float(i)])
            sids.append(i)
        cand_rows.append(row)
        cand_sids.append(sids)
        g = graph.find(n)
        parents.append([idx[p.name] for p in g.parents])
        role_vec.append(role_id[m.role])

    # realloc cost between strategies (role-agnostic estimate; scaled by
    # the largest param_bytes of any role — conservative)
    pmax = max(m.param_bytes for m in mfc_specs.values())
    rc = []
    for a in strategies:
        for b in strategies:
            if a == b:
                rc.append(0.0)
            elif (
                a.tensor_parallel_size % b.tensor_parallel_size
                and b.tensor_parallel_size % a.tensor_parallel_size
            ):
                rc.append(-1.0)
            else:
                moved = pmax / max(a.tensor_parallel_size * a.pipeline_parallel_size,
                                   b.tensor_parallel_size * b.pipeline_parallel_size)
                links = min(7, max(a.world_size, b.world_size))
                rc.append(moved / (links * XGMI_LINK_GBPS * 1e9) + 2e-3)

    pick, cost = C.mcmc_search(
        n_gpus, cand_rows, parents, role_vec, rc, len(strategies),
        mem_cap_gib * 2**30, n_chains, n_steps, seed,
    )
    out = {n: sid[cand_sids[i][int(pick[i])]] for i, n in enumerate(order)}
    logger.info("search: estimated step time %.3fs, allocation %s",
                cost, {k: str(v) for k, v in out.items()})
    return out, cost
