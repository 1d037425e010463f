"""Vocab-parallel log-probs / cross-entropy.

Reference semantics: realhf/impl/model/parallelism/model_parallel/modules.py
(_VocabParallelCrossEntropy:1050): logits stay vocab-partitioned across the
TP group; three small all-reduces (max, target-logit, sum-exp) replace
gathering the [tokens, vocab] matrix (which at 32k vocab would be the
largest activation in the step).
"""
import torch
import torch.distributed as dist

from realhf_amd.base import constants


class _VocabParallelLogProbs(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, labels: torch.Tensor):
        # logits: [n, v_local]; labels: [n] global vocab ids
        tp = constants.tp_world_size() if constants.has_current() else 1
        group = constants.tp_group() if tp > 1 else None
        v_local = logits.shape[-1]
        r = constants.tp_rank() if tp > 1 else 0
        lo = r * v_local

        logits = logits.float()
        vmax = logits.max(dim=-1).values
        if tp > 1:
            dist.all_reduce(vmax, op=dist.ReduceOp.MAX, group=group)
        shifted = logits - vmax.unsqueeze(-1)
        exp = shifted.exp()
        sum_exp = exp.sum(dim=-1)
        if tp > 1:
            dist.all_reduce(sum_exp, group=group)

        local_labels = labels - lo
        in_range = (local_labels >= 0) & (local_labels < v_local)
        safe = local_labels.clamp(0, v_local - 1)
        target = shifted.gather(-1, safe.unsqueeze(-1)).squeeze(-1)
        target = torch.where(in_range, target, torch.zeros_like(target))
        if tp > 1:
            dist.all_reduce(target, group=group)

        logp = target - sum_exp.log()
        ctx.save_for_backward(exp, sum_exp, safe, in_range)
        return logp

    @staticmethod
    def backward(ctx, grad_out):
        exp, sum_exp, safe, in_range = ctx.saved_tensors
        softmax = exp / sum_exp.unsqueeze(-1)
        g = -softmax * grad_out.unsqueeze(-1)
        add = torch.where(in_range, grad_out, torch.zeros_like(grad_out))
        g.scatter_add_(-1, safe.unsqueeze(-1), add.unsqueeze(-1))
        return g, None


def vocab_parallel_logprobs(logits: torch.Tensor, labels: torch.Tensor):
    """log p(label) per row, with logits vocab-partitioned over TP."""
    return _VocabParallelLogProbs.apply(logits, labels)


def packed_shifted_logprobs(
    logits: torch.Tensor, cu_seqlens: torch.Tensor, packed_input_ids: torch.Tensor
):
    """TP-aware gather_packed_shifted_log_probs: works on vocab-parallel
    logits; output length = total - bs."""
    from realhf_amd.utils.functional import (
        build_leave_one_indices,
        build_shift_one_indices,
    )

    total = packed_input_ids.shape[0]
    leave = build_leave_one_indices(total, cu_seqlens)
    shift = build_shift_one_indices(total, cu_seqlens)
    return vocab_parallel_logprobs(logits[leave], packed_input_ids[shift])
