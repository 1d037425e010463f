"""Mixture-of-Experts layer: top-k router + experts.

Reference semantics: realhf/impl/model/modules/moe/ (TopKRouter router.py:24,
MoETokenDispatcher token_dispatcher.py:17, SequentialMLP/GroupedMLP
experts.py, aux loss utils/moe.py:13).

Beyond the reference: true expert parallelism (EP) — experts sharded over
an EP group with all-to-all token exchange over xGMI (the reference
replicates all experts; SURVEY.md §2.3 row EP).  EP activates when the
model's grid has ep_size > 1 (see parallel/ep.py); otherwise experts are
local and dispatch is a pure permutation.

Grouped expert GEMMs run through the hand-written HIP grouped-GEMM kernel
on GPU (ops/csrc/grouped_gemm.hip) and a per-expert loop on CPU.
"""
from typing import Dict, List, Optional

import torch
import torch.nn.functional as F

from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.base import constants
from realhf_amd.ops import functional as ops
from realhf_amd.parallel import mappings

# aux losses collected during forward; training interfaces drain this.
_AUX_LOSSES: List[torch.Tensor] = []


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


class TopKRouter(torch.nn.Module):
    def __init__(self, cfg: ReaLModelConfig, weight: torch.Tensor):
        super().__init__()
        self.cfg = cfg
        self.moe = cfg.moe
        self.weight = weight  # [n_experts, hidden]

    def forward(self, h: torch.Tensor):
        logits = F.linear(h.float(), self.weight.float())
        moe = self.moe
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
            if moe.routing_type == "aux_loss" and self.training:
                counts = torch.bincount(
                    idx.flatten(), minlength=moe.num_experts
                )
                aux = switch_load_balancing_loss(probs, counts, moe.top_k)
                _AUX_LOSSES.append(moe.aux_loss_coef * aux)
            if moe.z_loss_coef > 0 and self.training:
                z = torch.logsumexp(logits, dim=-1).square().mean()
                _AUX_LOSSES.append(moe.z_loss_coef * z)
        return scores, idx


class MoELayer(torch.nn.Module):
    """Dispatch + expert MLPs + combine.  Token dispatch is a sort-based
    permutation; expert GEMMs are grouped."""

    def __init__(self, cfg: ReaLModelConfig, layer_idx: int, params: Dict[str, torch.Tensor], tp_size: int):
        super().__init__()
        self.cfg = cfg
        self.i = layer_idx
        self.p = params
        self.tp_size = tp_size
        self.router = TopKRouter(cfg, params[f"{layer_idx}.mlp.router.weight"])
        self.n_experts = cfg.moe.num_experts

    def _expert_ffn(self, e: int, x: torch.Tensor) -> torch.Tensor:
        i = self.i
        gate = F.linear(x, self.p[f"{i}.mlp.experts.{e}.gate.weight"])
        up = F.linear(x, self.p[f"{i}.mlp.experts.{e}.up.weight"])
        act = (F.silu(gate.float()) * up.float()).to(x.dtype)
        return F.linear(act, self.p[f"{i}.mlp.experts.{e}.down.weight"])

    def forward(self, h: torch.Tensor) -> torch.Tensor:
        scores, idx = self.router(h)  # [tokens, k]
        k = self.cfg.moe.top_k
        tokens = h.shape[0]
        h_tp = mappings.copy_to_tp_region(h)

        flat_idx = idx.flatten()  # [tokens*k]
        sort_order = torch.argsort(flat_idx, stable=True)
        token_of = sort_order // k  # source token of each dispatched slot
        counts = torch.bincount(flat_idx, minlength=self.n_experts)

        permuted = h_tp[token_of]
        outs = torch.empty_like(permuted)
        start = 0
        for e in range(self.n_experts):
            n = int(counts[e])
            if n == 0:
                continue
            outs[start : start + n] = self._expert_ffn(e, permuted[start : start + n])
            start += n
        # un-permute and combine with routing weights
        combined = torch.zeros_like(h_tp, dtype=torch.float32)
        w = scores.to(torch.float32).flatten()[sort_order]
        combined.index_add_(0, token_of, outs.float() * w.unsqueeze(-1))
        combined = combined.to(h.dtype)
        return mappings.reduce_from_tp_region(combined)
