"""Mixture-of-Experts layer: top-k router + dispatcher + experts.

Reference semantics: realhf/impl/model/modules/moe/ (TopKRouter router.py:24,
MoETokenDispatcher token_dispatcher.py:17, SequentialMLP/GroupedMLP
experts.py, aux loss utils/moe.py:13).

Beyond the reference: TRUE expert parallelism — the reference replicates
all experts and stubs the expert group to self (megatron.py:108, SURVEY.md
§2.3 row EP).  Here experts shard over an EP group (a block of DP ranks)
with token all-to-all over xGMI; EP degree = cfg.moe.expert_parallel_size.

Expert GEMMs run through the hand-written MFMA grouped-GEMM kernels for
BOTH inference and training (fwd: ops/csrc/grouped_gemm.hip; bwd dX/dW:
ops/csrc/grouped_gemm_bwd.hip, via the _GroupedGemm autograd function) —
matching the reference's grouped_gemm.ops.gmm training path
(experts.py:194-207).  The per-expert rocBLAS loop remains as the
CPU/odd-shape fallback (REALHF_AMD_MOE_LOOP=1 forces it, for tests).
"""
import contextlib
import math
from typing import Dict, List

import torch
import torch.distributed as dist
import torch.nn.functional as F

from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.base import constants
from realhf_amd.parallel import mappings

# Aux losses collected during the TRAINING forward only; the training
# interface's loss_fn drains this.  Collection is explicitly scoped (engine
# wraps the train forward in aux_loss_collection()) so that no-grad
# inference/generation and gradient-checkpoint RECOMPUTE (which re-runs the
# forward during backward) never pollute the list for the next minibatch.
_AUX_LOSSES: List[torch.Tensor] = []
_COLLECT_AUX = False


@contextlib.contextmanager
def aux_loss_collection():
    global _COLLECT_AUX, _AUX_LOSSES
    prev = _COLLECT_AUX
    _AUX_LOSSES = []
    _COLLECT_AUX = True
    try:
        yield
    finally:
        _COLLECT_AUX = prev


def _collecting() -> bool:
    return _COLLECT_AUX and torch.is_grad_enabled()


def pop_aux_losses() -> List[torch.Tensor]:
    global _AUX_LOSSES
    out, _AUX_LOSSES = _AUX_LOSSES, []
    return out


def switch_load_balancing_loss(
    probs: torch.Tensor, tokens_per_expert: torch.Tensor, topk: int
) -> torch.Tensor:
    """Standard switch-transformer aux loss (reference: utils/moe.py:13)."""
    n_exp = probs.shape[-1]
    n_tok = probs.shape[0]
    frac_tokens = tokens_per_expert.float() / max(1, n_tok * topk)
    frac_probs = probs.float().mean(dim=0)
    return n_exp * torch.sum(frac_tokens * frac_probs)


def sinkhorn(cost: torch.Tensor, tol: float = 1e-4, n_iter: int = 8):
    """Sinkhorn normalization for balanced routing (reference: utils/moe.py:69)."""
    cost = torch.exp(cost.float())
    d0 = torch.ones(cost.shape[0], device=cost.device)
    d1 = torch.ones(cost.shape[1], device=cost.device)
    for _ in range(n_iter):
        d0 = (1.0 / cost.shape[0]) / ((cost * d1.unsqueeze(0)).sum(1) + 1e-8)
        d1 = (1.0 / cost.shape[1]) / ((cost * d0.unsqueeze(1)).sum(0) + 1e-8)
    return d1.unsqueeze(0) * cost * d0.unsqueeze(1)


def apply_expert_capacity(
    scores: torch.Tensor,   # [tokens, k] routing weights
    idx: torch.Tensor,      # [tokens, k] global expert ids
    n_experts: int,
    capacity_factor: float,
    policy: str,            # "probs" | "position"
):
    """Limit each expert to cap = ceil(tokens*k/n_experts * factor)
    assignments (reference: utils/moe.py:310 topk_softmax_with_capacity).

    Over-capacity assignments are masked: "probs" keeps the cap
    highest-weight assignments per expert, "position" the cap earliest
    tokens.  Returns (masked scores, kept mask [tokens,k], cap).
    """
    ntok, k = scores.shape
    cap = int(math.ceil(ntok * k / n_experts * capacity_factor))
    sel = torch.zeros(ntok, n_experts, dtype=torch.bool, device=idx.device)
    sel.scatter_(1, idx, True)
    if policy == "position":
        rank = sel.long().cumsum(0) - 1          # arrival order per expert
    elif policy == "probs":
        gates = torch.zeros(ntok, n_experts, device=scores.device,
                            dtype=scores.dtype)
        gates.scatter_(1, idx, scores)
        order = torch.argsort(gates, dim=0, descending=True, stable=True)
        rank = torch.argsort(order, dim=0)       # weight rank per expert
    else:
        raise ValueError(f"token_drop_policy {policy!r}")
    kept_dense = sel & (rank < cap)
    kept = torch.gather(kept_dense, 1, idx)
    return scores * kept, kept, cap


class TopKRouter(torch.nn.Module):
    def __init__(self, cfg: ReaLModelConfig, weight: torch.Tensor):
        super().__init__()
        self.cfg = cfg
        self.moe = cfg.moe
        self.weight = weight  # [n_experts, hidden]

    def forward(self, h: torch.Tensor):
        moe = self.moe
        if moe.input_jitter_eps and self.training:
            noise = torch.empty_like(h).uniform_(
                1.0 - moe.input_jitter_eps, 1.0 + moe.input_jitter_eps
            )
            h = h * noise
        logits = F.linear(h.float(), self.weight.float())
        if moe.routing_type == "sinkhorn" and self.training:
            with torch.no_grad():
                norm = sinkhorn(logits)
                _, idx = torch.topk(norm, moe.top_k, dim=-1)
            scores = torch.gather(torch.sigmoid(logits), 1, idx)
        else:
            probs = torch.softmax(logits, dim=-1)
            scores, idx = torch.topk(probs, moe.top_k, dim=-1)
            if moe.norm_topk_prob:
                scores = scores / scores.sum(dim=-1, keepdim=True)
            if moe.routing_type == "aux_loss" and _collecting():
                counts = torch.bincount(
                    idx.flatten(), minlength=moe.num_experts
                )
                aux = switch_load_balancing_loss(probs, counts, moe.top_k)
                _AUX_LOSSES.append(moe.aux_loss_coef * aux)
            if moe.z_loss_coef > 0 and _collecting():
                z = torch.logsumexp(logits, dim=-1).square().mean()
                _AUX_LOSSES.append(moe.z_loss_coef * z)
        kept = None
        if moe.capacity_factor is not None:
            scores, kept, _ = apply_expert_capacity(
                scores, idx, moe.num_experts, moe.capacity_factor,
                moe.token_drop_policy,
            )
        return scores, idx, kept


class _GroupedGemm(torch.autograd.Function):
    """out[seg_e] = x[seg_e] @ W[e]^T through the MFMA grouped-GEMM
    kernels; backward stays native (dX = dOut @ W[e], dW = dOut^T x)."""

    @staticmethod
    def forward(ctx, x, w, counts_cpu):
        from realhf_amd import ops as _ops_pkg

        C = _ops_pkg.require_hip()
        ctx.save_for_backward(x, w, counts_cpu)
        return C.grouped_gemm(x.contiguous(), w, counts_cpu)

    @staticmethod
    def backward(ctx, dout):
        from realhf_amd import ops as _ops_pkg

        C = _ops_pkg.require_hip()
        x, w, counts_cpu = ctx.saved_tensors
        dout = dout.contiguous()
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = C.grouped_gemm_dx(dout, w, counts_cpu)
        if ctx.needs_input_grad[1]:
            dw = C.grouped_gemm_dw(dout, x.contiguous(), counts_cpu,
                                   w.shape[0])
        return dx, dw, None


class _AllToAll(torch.autograd.Function):
    """Autograd-wrapped all_to_all_single over the EP group: backward is
    the reverse exchange (xGMI all-pairs — the natural fit)."""

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.group = group
        ctx.out_splits, ctx.in_splits = out_splits, in_splits
        out = x.new_empty(sum(out_splits), *x.shape[1:])
        dist.all_to_all_single(
            out, x.contiguous(), output_split_sizes=out_splits,
            input_split_sizes=in_splits, group=group,
        )
        return out

    @staticmethod
    def backward(ctx, g):
        back = g.new_empty(sum(ctx.in_splits), *g.shape[1:])
        dist.all_to_all_single(
            back, g.contiguous(), output_split_sizes=ctx.in_splits,
            input_split_sizes=ctx.out_splits, group=ctx.group,
        )
        return back, None, None, None


class MoELayer(torch.nn.Module):
    """Dispatch + expert MLPs + combine; sort-based permutation, optional
    EP all-to-all, grouped expert GEMMs."""

    def __init__(self, cfg: ReaLModelConfig, layer_idx: int,
                 params: Dict[str, torch.Tensor], tp_size: int,
                 ep_rank: int = 0, ep_size: int = 1):
        super().__init__()
        self.cfg = cfg
        self.i = layer_idx
        self.p = params
        self.tp_size = tp_size
        self.ep_rank, self.ep_size = ep_rank, ep_size
        self.router = TopKRouter(cfg, params[f"{layer_idx}.mlp.router.weight"])
        self.n_experts = cfg.moe.num_experts
        assert self.n_experts % ep_size == 0
        self.n_local = self.n_experts // ep_size
        self.local_e0 = ep_rank * self.n_local

    # ---------------------------------------------------------------- mlp
    def _expert_ffn(self, e_local: int, x: torch.Tensor) -> torch.Tensor:
        i, e = self.i, self.local_e0 + e_local
        gate = F.linear(x, self.p[f"{i}.mlp.experts.{e}.gate.weight"])
        up = F.linear(x, self.p[f"{i}.mlp.experts.{e}.up.weight"])
        act = (F.silu(gate.float()) * up.float()).to(x.dtype)
        return F.linear(act, self.p[f"{i}.mlp.experts.{e}.down.weight"])

    def _expert_weight_stack(self, part: str):
        """[E_local, N, K] expert-strided view when the flat layout is
        uniform; None otherwise."""
        i = self.i
        ws = [
            self.p[f"{i}.mlp.experts.{self.local_e0 + e}.{part}.weight"]
            for e in range(self.n_local)
        ]
        base = ws[0]
        if len(ws) == 1:
            return base.unsqueeze(0)
        stride = ws[1].storage_offset() - base.storage_offset()
        for a, b in zip(ws, ws[1:]):
            if b.storage_offset() - a.storage_offset() != stride:
                return None
        return base.as_strided(
            (self.n_local, base.shape[0], base.shape[1]),
            (stride, base.shape[1], 1), base.storage_offset(),
        )

    def _experts_forward(self, x_sorted: torch.Tensor, counts_cpu: torch.Tensor):
        """x_sorted: tokens grouped by local expert; counts_cpu [n_local].

        Training AND inference run through the hand-written MFMA grouped
        GEMM (fwd: grouped_gemm.hip, bwd dX/dW: grouped_gemm_bwd.hip) —
        the reference trains through grouped_gemm.ops.gmm
        (experts.py:194-207); the per-expert rocBLAS loop below is only
        the CPU/odd-shape fallback."""
        import os

        # Size-adaptive dispatch (measured on MI355X, tools/bench_moe_train
        # sweep, fwd+bwd): at 64 routed tokens/expert the MFMA grouped
        # kernel wins 1.4x over the per-expert rocBLAS loop (14.5 vs
        # 20.6 ms); by 256/expert the loop edges ahead (19.7 vs 23.4) and
        # at 1024/expert it wins 2x (tuned big-GEMM kernels).  Crossover
        # ~128 tokens/expert; both paths are fully autograd.
        avg_seg = x_sorted.shape[0] / max(1, self.n_local)
        threshold = float(os.environ.get("REALHF_AMD_GG_THRESHOLD", 128))
        use_grouped = (
            x_sorted.is_cuda
            and x_sorted.dtype == torch.bfloat16
            and os.environ.get("REALHF_AMD_MOE_LOOP") != "1"
            and (avg_seg <= threshold
                 or os.environ.get("REALHF_AMD_MOE_GROUPED") == "1")
        )
        if use_grouped:
            wg = self._expert_weight_stack("gate")
            wu = self._expert_weight_stack("up")
            wd = self._expert_weight_stack("down")
            idim = wg.shape[1] if wg is not None else 0
            hid = x_sorted.shape[1]
            if (
                wg is not None and wu is not None and wd is not None
                # fwd needs K%32, N%64; bwd dX needs K%64, dW needs N%64 —
                # both dims multiple of 64 covers every direction
                and hid % 64 == 0 and idim % 64 == 0
            ):
                gate = _GroupedGemm.apply(x_sorted, wg, counts_cpu)
                up = _GroupedGemm.apply(x_sorted, wu, counts_cpu)
                act = (F.silu(gate.float()) * up.float()).to(x_sorted.dtype)
                return _GroupedGemm.apply(act, wd, counts_cpu)
        outs = torch.empty(
            x_sorted.shape[0],
            self.p[f"{self.i}.mlp.experts.{self.local_e0}.down.weight"].shape[0],
            dtype=x_sorted.dtype, device=x_sorted.device,
        )
        start = 0
        for e in range(self.n_local):
            n = int(counts_cpu[e])
            if n == 0:
                continue
            outs[start:start + n] = self._expert_ffn(e, x_sorted[start:start + n])
            start += n
        return outs

    # ------------------------------------------------------------ forward
    def forward(self, h: torch.Tensor) -> torch.Tensor:
        scores, idx, kept = self.router(h)  # [tokens, k]
        k = self.cfg.moe.top_k
        h_tp = mappings.copy_to_tp_region(h)
        if kept is not None:
            return self._capacity_forward(h, h_tp, scores, idx, kept)

        flat_idx = idx.flatten()  # [tokens*k] global expert ids
        sort_order = torch.argsort(flat_idx, stable=True)
        token_of = sort_order // k
        counts = torch.bincount(flat_idx, minlength=self.n_experts)
        permuted = h_tp[token_of]

        if self.ep_size > 1:
            out_sorted, inv = self._ep_exchange_and_compute(permuted, counts)
        else:
            counts_cpu = counts.cpu()
            out_sorted = self._experts_forward(permuted, counts_cpu)
            inv = None

        combined = torch.zeros_like(h_tp, dtype=torch.float32)
        w = scores.to(torch.float32).flatten()[sort_order]
        combined.index_add_(0, token_of, out_sorted.float() * w.unsqueeze(-1))
        combined = combined.to(h.dtype)
        return mappings.reduce_from_tp_region(combined)

    def _capacity_forward(self, h, h_tp, scores, idx, kept):
        """Dispatch with expert-capacity limiting (capacity_factor set).

        Dropped (token, slot) assignments leave the dispatch entirely —
        their tokens pass through with zero contribution from that slot,
        like the reference's drop mode (utils/moe.py:380-389).  With
        pad_to_capacity every expert processes exactly `cap` rows (zero
        rows for empty slots), so all dispatch shapes are static.
        """
        moe = self.cfg.moe
        ntok, k = scores.shape
        cap = int(math.ceil(ntok * k / self.n_experts * moe.capacity_factor))
        flat_keep = kept.flatten()
        flat_idx = idx.flatten()[flat_keep]
        flat_tok = torch.arange(
            ntok, device=idx.device
        ).repeat_interleave(k)[flat_keep]
        flat_w = scores.to(torch.float32).flatten()[flat_keep]

        if moe.pad_to_capacity:
            # static [n_experts*cap] dispatch: slot = e*cap + arrival rank
            order = torch.argsort(flat_idx, stable=True)
            e_sorted = flat_idx[order]
            seg_start = torch.searchsorted(
                e_sorted, torch.arange(self.n_experts, device=idx.device)
            )
            pos = (torch.arange(e_sorted.numel(), device=idx.device)
                   - seg_start[e_sorted])
            slots = e_sorted * cap + pos
            nslot = self.n_experts * cap
            permuted = torch.zeros(nslot, h_tp.shape[1], dtype=h_tp.dtype,
                                   device=h_tp.device)
            permuted[slots] = h_tp[flat_tok[order]]
            token_of = torch.zeros(nslot, dtype=torch.long, device=h_tp.device)
            token_of[slots] = flat_tok[order]
            w = torch.zeros(nslot, dtype=torch.float32, device=h_tp.device)
            w[slots] = flat_w[order]
            counts = torch.full((self.n_experts,), cap, dtype=torch.long,
                                device=h_tp.device)
        else:
            order = torch.argsort(flat_idx, stable=True)
            token_of = flat_tok[order]
            w = flat_w[order]
            counts = torch.bincount(flat_idx, minlength=self.n_experts)
            permuted = h_tp[token_of]

        if self.ep_size > 1:
            # padded layout is expert-sorted with static per-expert counts
            # — exactly the exchange's input contract
            out_sorted, _ = self._ep_exchange_and_compute(permuted, counts)
        elif moe.pad_to_capacity:
            # counts are compile-time constants: no device sync (the
            # counts.cpu() D2H below is what blocks hipGraph capture of
            # MoE decode)
            counts_cpu = torch.full((self.n_experts,), cap, dtype=torch.long)
            out_sorted = self._experts_forward(permuted, counts_cpu)
        else:
            counts_cpu = counts.cpu()
            out_sorted = self._experts_forward(permuted, counts_cpu)

        combined = torch.zeros_like(h_tp, dtype=torch.float32)
        combined.index_add_(0, token_of, out_sorted.float() * w.unsqueeze(-1))
        return mappings.reduce_from_tp_region(combined.to(h.dtype))

    def _ep_exchange_and_compute(self, permuted, counts):
        """All-to-all token exchange over the EP group, local expert
        compute, reverse exchange.  Returns outputs aligned with the
        (globally expert-sorted) `permuted` order."""
        g = constants.grid()
        group = g.ep_group()
        counts_cpu = counts.cpu()
        send_splits = [
            int(counts_cpu[r * self.n_local:(r + 1) * self.n_local].sum())
            for r in range(self.ep_size)
        ]
        # exchange counts so each rank knows what it receives per expert
        cmat = counts.to(torch.long)
        recv_cmat = torch.empty_like(cmat)
        dist.all_to_all_single(
            recv_cmat, cmat,
            output_split_sizes=[self.n_local] * self.ep_size,
            input_split_sizes=[self.n_local] * self.ep_size,
            group=group,
        )
        recv_cmat_cpu = recv_cmat.cpu().view(self.ep_size, self.n_local)
        recv_splits = [int(recv_cmat_cpu[r].sum()) for r in range(self.ep_size)]

        x_recv = _AllToAll.apply(permuted, recv_splits, send_splits, group)
        # re-sort received tokens (grouped by src rank, each sorted by
        # local expert) into a single local-expert-major order
        local_ids = torch.repeat_interleave(
            torch.arange(self.n_local, device=permuted.device).repeat(self.ep_size),
            recv_cmat.flatten().clamp(min=0),
        )
        order = torch.argsort(local_ids, stable=True)
        local_counts_cpu = recv_cmat_cpu.sum(0)
        y = self._experts_forward(x_recv[order], local_counts_cpu)
        y_unsorted = torch.empty_like(y)
        y_unsorted[order] = y
        out = _AllToAll.apply(y_unsorted, send_splits, recv_splits, group)
        return out, None
