"""Packed-sequence functional utilities.

Reference semantics: realhf/impl/model/utils/functional.py
(gather_packed_shifted_log_probs:165, build_shift_one_indices:87,
masked_normalization:227, compute_varlen_position_indices:429) and
utils/logits_warper.py (top_k_top_p_logits).
"""
from typing import Optional

import torch
import torch.distributed as dist


def build_shift_one_indices(total_len: int, cu_seqlens: torch.Tensor) -> torch.Tensor:
    """Indices selecting tokens 1..L-1 of each sequence from a packed
    [total_len] tensor — i.e. the predicted positions."""
    device = cu_seqlens.device
    idx = torch.arange(total_len, dtype=torch.long, device=device)
    starts = cu_seqlens[:-1]
    is_start = torch.zeros(total_len, dtype=torch.bool, device=device)
    is_start[starts.long()] = True
    return idx[~is_start]


def build_leave_one_indices(total_len: int, cu_seqlens: torch.Tensor) -> torch.Tensor:
    """Indices selecting tokens 0..L-2 of each sequence (the predicting
    positions)."""
    device = cu_seqlens.device
    idx = torch.arange(total_len, dtype=torch.long, device=device)
    ends = cu_seqlens[1:].long() - 1
    is_end = torch.zeros(total_len, dtype=torch.bool, device=device)
    is_end[ends] = True
    return idx[~is_end]


def gather_packed_shifted_log_probs(
    logits: torch.Tensor,  # [total, vocab] (local full vocab)
    cu_seqlens: torch.Tensor,  # [bs+1]
    packed_input_ids: torch.Tensor,  # [total]
) -> torch.Tensor:
    """Log-prob of each NEXT token: output length total - bs (reference:
    functional.py:165)."""
    total = packed_input_ids.shape[0]
    leave_one = build_leave_one_indices(total, cu_seqlens)
    shift_one = build_shift_one_indices(total, cu_seqlens)
    logits = logits[leave_one]  # predicting positions
    labels = packed_input_ids[shift_one]  # their targets
    logp = torch.log_softmax(logits.float(), dim=-1)
    return logp.gather(-1, labels.unsqueeze(-1)).squeeze(-1)


def masked_normalization(
    x: torch.Tensor,
    mask: Optional[torch.Tensor] = None,
    eps: float = 1e-5,
    high_precision: bool = True,
    all_reduce_group=None,
) -> torch.Tensor:
    """Normalize x to zero-mean unit-var over masked entries, optionally
    all-reduced over a DP group (reference: functional.py:227)."""
    dtype = torch.float64 if high_precision else torch.float32
    x = x.to(dtype)
    if mask is None:
        mask = torch.ones_like(x)
    else:
        mask = mask.to(dtype)
    x = x * mask
    factor = mask.sum()
    x_sum = x.sum()
    x_sq_sum = (x**2).sum()
    if all_reduce_group is not None and dist.is_initialized():
        stats = torch.stack([factor, x_sum, x_sq_sum])
        dist.all_reduce(stats, group=all_reduce_group)
        factor, x_sum, x_sq_sum = stats[0], stats[1], stats[2]
    mean = x_sum / factor
    var = x_sq_sum / factor - mean**2
    return ((x - mean) * mask * torch.rsqrt(var + eps)).float()


def compute_varlen_position_indices(
    total_len: int, cu_seqlens: torch.Tensor
) -> torch.Tensor:
    """Position index of each token within its sequence: [0..L0-1, 0..L1-1, ...]."""
    device = cu_seqlens.device
    idx = torch.arange(total_len, dtype=torch.long, device=device)
    seq_id = torch.bucketize(idx, cu_seqlens[1:].long(), right=True)
    return idx - cu_seqlens[:-1].long()[seq_id]


@torch.no_grad()
def top_k_top_p_logits(
    logits: torch.Tensor,  # [bs, vocab]
    top_k: int = 0,
    top_p: float = 1.0,
    inplace: bool = False,
) -> torch.Tensor:
    """Standard top-k / nucleus filtering (reference: logits_warper.py)."""
    if not inplace:
        logits = logits.clone()
    if top_k > 0 and top_k < logits.shape[-1]:
        kth = torch.topk(logits, top_k, dim=-1).values[..., -1, None]
        logits.masked_fill_(logits < kth, float("-inf"))
    if 0.0 < top_p < 1.0:
        sorted_logits, sorted_idx = torch.sort(logits, descending=True, dim=-1)
        probs = torch.softmax(sorted_logits, dim=-1)
        cum = probs.cumsum(dim=-1)
        remove = cum - probs > top_p  # keep tokens until cum prob exceeds p
        scatter_mask = remove.scatter(1, sorted_idx, remove)
        logits.masked_fill_(scatter_mask, float("-inf"))
    return logits


def apply_logits_mask(logits: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
    """Mask forbidden vocab entries to -inf (reference:
    utils/functional.py:214 apply_logits_mask — TP vocab-partition aware:
    `logits` may be this tp rank's vocab slice [..., V/tp] while `mask`
    (True = forbidden) covers the FULL vocab [..., V])."""
    from realhf_amd.base import constants

    v_local = logits.shape[-1]
    if mask.shape[-1] != v_local:
        tp = constants.tp_world_size() if constants.has_current() else 1
        assert mask.shape[-1] == v_local * tp, (mask.shape, logits.shape, tp)
        r = constants.tp_rank() if constants.has_current() else 0
        mask = mask[..., r * v_local:(r + 1) * v_local]
    return logits.masked_fill_(mask, float("-inf"))
