"""Profiling, timing marks and FLOPs accounting.

Reference semantics: realhf/base/monitor.py (cuda_tmark:375,
caculate_llama_train_flops:277/296/318, kernelStatFromTrace:699) +
master_worker TFLOP/s logging (master_worker.py:1461-1488).

MI355X equivalents: hipEvents behind torch.cuda.Event for the timing
marks; rocprofv3 rocpd databases for kernel stats (tools/rocpd_summary.py);
torch.cuda memory stats for the per-MFC memory tables.
"""
import contextlib
import os
import pickle
import time
from collections import defaultdict
from typing import Dict

import torch

from realhf_amd.base import logging

logger = logging.getLogger("monitor")

# MI355X headline numbers for utilization reporting (MI355X_MICROARCH.md)
MI355X_BF16_DENSE_TFLOPS = 2495.0
MI355X_HBM_GBPS = 6300.0


# ---------------------------------------------------------------------------
# FLOPs accounting (hardware-independent; reference formulas monitor.py:277)
# ---------------------------------------------------------------------------
def dense_transformer_flops(
    n_layers: int, hidden: int, intermediate: int, vocab: int,
    n_heads: int, n_kv_heads: int, head_dim: int,
    total_tokens: int, sum_sq_seqlens: float,
    backward: bool = False, gated_mlp: bool = True,
) -> float:
    """FLOPs for one forward (x3 with backward) over a packed batch."""
    qd = n_heads * head_dim
    kvd = n_kv_heads * head_dim
    per_tok_layer = 2 * hidden * (qd + 2 * kvd)  # qkv
    per_tok_layer += 2 * qd * hidden  # o proj
    mlp_mult = 3 if gated_mlp else 2
    per_tok_layer += 2 * mlp_mult * hidden * intermediate
    gemm = total_tokens * per_tok_layer * n_layers
    attn = 4 * qd * sum_sq_seqlens * n_layers / 2  # causal halves it
    head = 2 * total_tokens * hidden * vocab
    fwd = gemm + attn + head
    return fwd * (3 if backward else 1)


def gen_flops(cfg_like, prompt_tokens: int, new_tokens_total: int,
              sum_ctx_tokens: float) -> float:
    """Prefill + per-token decode FLOPs."""
    prefill = dense_transformer_flops(
        cfg_like.n_layers, cfg_like.hidden_dim, cfg_like.intermediate_dim,
        cfg_like.vocab_size, cfg_like.n_heads, cfg_like.n_kv_heads,
        cfg_like.head_dim, prompt_tokens, 0.0,
    )
    per_tok = dense_transformer_flops(
        cfg_like.n_layers, cfg_like.hidden_dim, cfg_like.intermediate_dim,
        cfg_like.vocab_size, cfg_like.n_heads, cfg_like.n_kv_heads,
        cfg_like.head_dim, new_tokens_total, 0.0,
    )
    attn_kv = (
        4 * cfg_like.n_heads * cfg_like.head_dim * sum_ctx_tokens
        * cfg_like.n_layers
    )
    return prefill + per_tok + attn_kv


# ---------------------------------------------------------------------------
# timing marks (reference: cuda_tmark monitor.py:375)
# ---------------------------------------------------------------------------
class TimeMarks:
    """Named GPU-bracketed wall timers; dump per-process for analysis."""

    def __init__(self):
        self._acc: Dict[str, float] = defaultdict(float)
        self._cnt: Dict[str, int] = defaultdict(int)

    @contextlib.contextmanager
    def mark(self, name: str, sync: bool = True):
        if sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.monotonic()
        try:
            yield
        finally:
            if sync and torch.cuda.is_available():
                torch.cuda.synchronize()
            self._acc[name] += time.monotonic() - t0
            self._cnt[name] += 1

    def summary(self) -> Dict[str, Dict[str, float]]:
        return {
            k: {"total_s": v, "count": self._cnt[k], "mean_s": v / max(1, self._cnt[k])}
            for k, v in sorted(self._acc.items())
        }

    def dump(self, path: str):
        os.makedirs(os.path.dirname(path), exist_ok=True)
        with open(path, "wb") as f:
            pickle.dump(self.summary(), f)

    def clear(self):
        self._acc.clear()
        self._cnt.clear()


tmarks = TimeMarks()


def cuda_tmark(name: str):
    """Decorator form."""

    def deco(fn):
        def wrapped(*a, **kw):
            with tmarks.mark(name):
                return fn(*a, **kw)

        return wrapped

    return deco


# ---------------------------------------------------------------------------
# GPU stats (reference: model_worker __log_gpu_stats:999)
# ---------------------------------------------------------------------------
def gpu_memory_stats() -> Dict[str, float]:
    if not torch.cuda.is_available():
        return {}
    return {
        "allocated_gib": torch.cuda.memory_allocated() / 2**30,
        "reserved_gib": torch.cuda.memory_reserved() / 2**30,
        "peak_gib": torch.cuda.max_memory_allocated() / 2**30,
    }


def estimate_mfc_flops(itype, cfg, seqlens, out_seqlens=None) -> float:
    """Model FLOPs of one MFC execution over a packed shard (reference:
    InterfaceDataAmount + caculate_llama_*_flops, master_worker.py:233 /
    monitor.py:277).  `seqlens`: input per-seq lengths; `out_seqlens`:
    full prompt+generation lengths for GENERATE MFCs."""
    tokens = float(sum(seqlens))
    ssq = float(sum(l * l for l in seqlens))
    kw = dict(
        n_layers=cfg.n_layers, hidden=cfg.hidden_dim,
        intermediate=cfg.intermediate_dim, vocab=cfg.vocab_size,
        n_heads=cfg.n_heads, n_kv_heads=cfg.n_kv_heads,
        head_dim=cfg.head_dim,
    )
    name = getattr(itype, "name", str(itype)).upper()
    if name == "TRAIN_STEP":
        return dense_transformer_flops(total_tokens=tokens,
                                       sum_sq_seqlens=ssq, backward=True, **kw)
    if name == "GENERATE" and out_seqlens is not None:
        new_total = float(sum(o - i for o, i in zip(out_seqlens, seqlens)))
        # decode attention reads ctx t for each generated position t
        sum_ctx = float(sum((o * o - i * i) / 2.0
                            for o, i in zip(out_seqlens, seqlens)))

        class _C:
            pass

        c = _C()
        for k, v in (("n_layers", cfg.n_layers), ("hidden_dim", cfg.hidden_dim),
                     ("intermediate_dim", cfg.intermediate_dim),
                     ("vocab_size", cfg.vocab_size), ("n_heads", cfg.n_heads),
                     ("n_kv_heads", cfg.n_kv_heads), ("head_dim", cfg.head_dim)):
            setattr(c, k, v)
        prefill_flops = dense_transformer_flops(
            total_tokens=tokens, sum_sq_seqlens=ssq, **kw)
        decode_flops = dense_transformer_flops(
            total_tokens=new_total, sum_sq_seqlens=0.0, **kw)
        return prefill_flops + decode_flops + (
            4 * cfg.n_heads * cfg.head_dim * sum_ctx * cfg.n_layers)
    return dense_transformer_flops(total_tokens=tokens, sum_sq_seqlens=ssq,
                                   **kw)


def log_tflops(name: str, flops: float, seconds: float, n_gpus: int = 1):
    tf = flops / seconds / 1e12 / n_gpus
    util = tf / MI355X_BF16_DENSE_TFLOPS * 100
    logger.info(
        "%s: %.1f TFLOP/s per GPU (%.1f%% of MI355X bf16 dense peak)",
        name, tf, util,
    )
    return tf
