"""Hot ops: HIP kernels with torch reference fallbacks.

HIP kernels (realhf_amd/ops/csrc/, reference op inventory SURVEY.md §2.2):
  rms_norm           — fused RMSNorm fwd/bwd (reference used TransformerEngine)
  apply_rotary       — RoPE on packed varlen qk (reference: flash-attn fused rotary)
  swiglu             — silu(gate)*up fused fwd/bwd
  attn_varlen        — flash-attention packed prefill (reference: flash_attn_varlen_func)
  attn_decode        — single-token GQA decode over contiguous KV cache
                       (reference: flash_attn_with_kvcache)
  gae                — packed GAE scan (reference: realhf._C.cugae)
  slice_intervals /
  set_intervals      — flat-param interval gather/scatter (realhf._C.interval_op_cuda)
  fused_adamw        — flat-buffer AdamW (reference: apex fused adam via Megatron)

All torch reference paths are fp32-upcast where the reference math is fp32.
"""
import math
import os
from typing import List, Optional, Tuple

import torch

from realhf_amd import ops as _ops


# ---------------------------------------------------------------------------
# RMSNorm
# ---------------------------------------------------------------------------
def rms_norm_ref(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps)
    return (out * weight.float()).to(x.dtype)


def gemma_rms_norm_ref(x, weight, eps):
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps)
    return (out * (1.0 + weight.float())).to(x.dtype)


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        C = _ops.require_hip()
        out, rstd = C.rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, rstd)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        C = _ops.require_hip()
        x, weight, rstd = ctx.saved_tensors
        dx, dw = C.rmsnorm_bwd(grad_out.contiguous(), x, weight, rstd)
        return dx, dw, None


def rms_norm(x, weight, eps: float = 1e-5, gemma_style: bool = False):
    if _ops.use_hip(x) and not gemma_style:
        return _RMSNormFn.apply(x.contiguous(), weight, eps)
    if gemma_style:
        return gemma_rms_norm_ref(x, weight, eps)
    return rms_norm_ref(x, weight, eps)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------
class RotaryCache:
    """Host-precomputed cos/sin tables (Appendix B: never sinf/cosf on
    device).  Shapes: [max_len, head_dim/2]."""

    def __init__(self):
        self._cache = {}

    def get(
        self,
        head_dim: int,
        max_len: int,
        base: float,
        device,
        dtype=torch.float32,
        scaling: Optional[float] = None,
        scaling_type: Optional[str] = None,
        orig_max_pos: Optional[int] = None,
    ):
        """scaling_type: None/"linear" divide positions by `scaling`
        (position interpolation); "dynamic" rescales `base` NTK-style when
        max_len exceeds orig_max_pos (reference rotary.py:121
        `_update_cos_sin_cache` mirrors HF's dynamic-NTK rule).  Dynamic
        tables are keyed by their exact length — the base depends on it —
        so no doubling growth is applied there."""
        dynamic = scaling_type == "dynamic" and scaling is not None
        if dynamic:
            orig = int(orig_max_pos or max_len)
            max_len = max(int(max_len), orig)
            if max_len > orig:
                base = base * (
                    (scaling * max_len / orig) - (scaling - 1)
                ) ** (head_dim / (head_dim - 2))
            scaling = None  # dynamic rescales base; positions stay unscaled
        key = (head_dim, base, str(device), dtype, scaling)
        cos, sin, cached_len = self._cache.get(key, (None, None, 0))
        if cached_len < max_len:
            if not dynamic:
                max_len = max(max_len, 2 * cached_len, 2048)
            inv_freq = 1.0 / (
                base ** (torch.arange(0, head_dim, 2, dtype=torch.float64) / head_dim)
            )
            t = torch.arange(max_len, dtype=torch.float64)
            if scaling is not None:
                t = t / scaling
            freqs = torch.outer(t, inv_freq)
            cos = freqs.cos().to(dtype).to(device)
            sin = freqs.sin().to(dtype).to(device)
            self._cache[key] = (cos, sin, max_len)
        cos, sin, _ = self._cache[key]
        return cos, sin


rotary_cache = RotaryCache()


def apply_rotary_ref(
    x: torch.Tensor,  # [total, n_heads, head_dim]
    cos: torch.Tensor,  # [max_len, head_dim/2]
    sin: torch.Tensor,
    positions: torch.Tensor,  # [total] int32/64
    interleaved: bool = False,
) -> torch.Tensor:
    c = cos[positions].unsqueeze(1)  # [total, 1, hd/2]
    s = sin[positions].unsqueeze(1)
    xf = x.float()
    hd = x.shape[-1]
    if interleaved:
        x1, x2 = xf[..., 0::2], xf[..., 1::2]
        o1 = x1 * c - x2 * s
        o2 = x2 * c + x1 * s
        out = torch.stack([o1, o2], dim=-1).flatten(-2)
    else:
        x1, x2 = xf[..., : hd // 2], xf[..., hd // 2 :]
        o1 = x1 * c - x2 * s
        o2 = x2 * c + x1 * s
        out = torch.cat([o1, o2], dim=-1)
    return out.to(x.dtype)


class _RotaryFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin, positions, interleaved):
        C = _ops.require_hip()
        out = C.rope_fwd(x, cos, sin, positions, interleaved, False)
        ctx.save_for_backward(cos, sin, positions)
        ctx.interleaved = interleaved
        return out

    @staticmethod
    def backward(ctx, grad):
        C = _ops.require_hip()
        cos, sin, positions = ctx.saved_tensors
        dx = C.rope_fwd(grad.contiguous(), cos, sin, positions, ctx.interleaved, True)
        return dx, None, None, None, None


def apply_rotary(x, cos, sin, positions, interleaved=False):
    if _ops.use_hip(x):
        return _RotaryFn.apply(x.contiguous(), cos, sin, positions, interleaved)
    return apply_rotary_ref(x, cos, sin, positions, interleaved)


# ---------------------------------------------------------------------------
# SwiGLU epilogue: silu(gate) * up
# ---------------------------------------------------------------------------
def swiglu_ref(gate_up: torch.Tensor) -> torch.Tensor:
    """gate_up: [tokens, 2*idim] with gate = [:, :idim], up = [:, idim:]."""
    idim = gate_up.shape[-1] // 2
    gate, up = gate_up[..., :idim], gate_up[..., idim:]
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate_up.dtype)


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate_up):
        C = _ops.require_hip()
        out = C.swiglu_fwd(gate_up)
        ctx.save_for_backward(gate_up)
        return out

    @staticmethod
    def backward(ctx, grad):
        C = _ops.require_hip()
        (gate_up,) = ctx.saved_tensors
        return C.swiglu_bwd(grad.contiguous(), gate_up)


def swiglu(gate_up):
    if _ops.use_hip(gate_up):
        return _SwiGLUFn.apply(gate_up.contiguous())
    return swiglu_ref(gate_up)


# ---------------------------------------------------------------------------
# Attention — packed varlen (prefill/training)
# ---------------------------------------------------------------------------
def attn_varlen_ref(
    q: torch.Tensor,  # [total, nq, hd]
    k: torch.Tensor,  # [total, nkv, hd]
    v: torch.Tensor,  # [total, nkv, hd]
    cu_seqlens: torch.Tensor,  # [bs+1] int32
    causal: bool = True,
    softmax_scale: Optional[float] = None,
    window: Optional[int] = None,  # sliding window incl. self (mistral)
) -> torch.Tensor:
    """Per-sequence SDPA in fp32 — the numerics oracle."""
    nq, hd = q.shape[1], q.shape[2]
    nkv = k.shape[1]
    scale = softmax_scale or (1.0 / math.sqrt(hd))
    out = torch.empty_like(q)
    rep = nq // nkv
    cu = cu_seqlens.tolist()
    for i in range(len(cu) - 1):
        s, e = cu[i], cu[i + 1]
        qi = q[s:e].transpose(0, 1).float()  # [nq, L, hd]
        ki = k[s:e].transpose(0, 1).float()
        vi = v[s:e].transpose(0, 1).float()
        if rep > 1:
            ki = ki.repeat_interleave(rep, dim=0)
            vi = vi.repeat_interleave(rep, dim=0)
        scores = torch.matmul(qi, ki.transpose(-1, -2)) * scale
        L = e - s
        if causal and L > 1:
            mask = torch.triu(
                torch.ones(L, L, dtype=torch.bool, device=q.device), diagonal=1
            )
            scores = scores.masked_fill(mask, float("-inf"))
        if window is not None and L > window:
            # visible keys for query i: (i - window, i]  (HF mistral mask)
            wmask = torch.tril(
                torch.ones(L, L, dtype=torch.bool, device=q.device),
                diagonal=-window,
            )
            scores = scores.masked_fill(wmask, float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        o = torch.matmul(probs, vi)  # [nq, L, hd]
        out[s:e] = o.transpose(0, 1).to(q.dtype)
    return out


def _attn_varlen_blocked_torch(q, k, v, cu_seqlens, causal, scale, window=None):
    """Batched-padded torch implementation on GPU (used for backward
    recompute until the hand-written HIP backward lands): one set of
    batched rocBLAS GEMMs instead of a per-sequence python loop (the loop
    was CPU-launch-bound: bs x layers x ~15 kernels)."""
    total, nq, hd = q.shape
    nkv = k.shape[1]
    rep = nq // nkv
    lens = (cu_seqlens[1:] - cu_seqlens[:-1]).long()
    bs = lens.shape[0]
    Lmax = int(lens.max())
    device = q.device
    pos = torch.arange(total, device=device)
    seq_id = torch.bucketize(pos, cu_seqlens[1:].long(), right=True)
    local = pos - cu_seqlens[:-1].long()[seq_id]

    # bf16 MFMA matmuls (fp32 accumulation inside rocBLAS) with fp32
    # softmax — the same numerics class as a flash-attention backward;
    # fp32 matmuls here would run at 1/16 of the bf16 MFMA rate.
    mm_dtype = torch.bfloat16 if q.is_cuda else torch.float32

    def pad(x, nh):
        xp = x.new_zeros(bs, Lmax, nh, hd)
        xp[seq_id, local] = x
        return xp.permute(0, 2, 1, 3).to(mm_dtype)  # [bs, nh, Lmax, hd]

    qp = pad(q, nq)
    kp = pad(k, nkv)
    vp = pad(v, nkv)
    if rep > 1:
        kp = kp.repeat_interleave(rep, dim=1)
        vp = vp.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qp * scale, kp.transpose(-1, -2)).float()
    kpos = torch.arange(Lmax, device=device)
    valid = kpos.unsqueeze(0) < lens.unsqueeze(1)  # [bs, Lmax]
    mask = valid.unsqueeze(1).unsqueeze(2)  # [bs,1,1,L]
    if causal:
        cm = kpos.unsqueeze(0) <= kpos.unsqueeze(1)  # [Lq, Lk]
        if window is not None:
            cm = cm & (kpos.unsqueeze(0) > kpos.unsqueeze(1) - window)
        mask = mask & cm.unsqueeze(0).unsqueeze(0)
    scores = scores.masked_fill(~mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    probs = torch.nan_to_num(probs, nan=0.0)
    op = torch.matmul(probs.to(mm_dtype), vp)  # [bs, nh, Lmax, hd]
    op = op.permute(0, 2, 1, 3)  # [bs, Lmax, nh, hd]
    return op[seq_id, local].to(q.dtype)


class _AttnVarlenFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, cu_seqlens, max_seqlen, causal, scale,
                window=None):
        C = _ops.require_hip()
        out, lse = C.attn_varlen_fwd(q, k, v, cu_seqlens, int(max_seqlen),
                                     causal, scale, int(window or 0))
        ctx.save_for_backward(q, k, v, out, lse, cu_seqlens)
        ctx.causal, ctx.scale, ctx.max_seqlen = causal, scale, max_seqlen
        ctx.window = window
        return out

    @staticmethod
    def backward(ctx, grad_out):
        q, k, v, out, lse, cu_seqlens = ctx.saved_tensors
        if os.environ.get("REALHF_AMD_NO_HIP_ATTN_BWD") != "1":
            # hand-written MFMA backward (attn_bwd.hip): kv-stationary,
            # dQ via fp32 atomics, per-q-head dK/dV reduced here for GQA;
            # sliding-window via key-range restriction like the forward.
            # In-context A/B: 3.51 vs 3.19 samples/s over the padded
            # recompute fallback (REALHF_AMD_NO_HIP_ATTN_BWD=1).
            C = _ops.require_hip()
            dout = grad_out.contiguous()
            dsum = (dout.float() * out.float()).sum(-1)  # [total, nq]
            dq32, dk32, dv32 = C.attn_varlen_bwd(
                q, k, v, dout, lse, dsum, cu_seqlens, ctx.causal, ctx.scale,
                int(ctx.window or 0),
            )
            nq, nkv = q.shape[1], k.shape[1]
            rep = nq // nkv
            if rep > 1:
                t = dk32.shape[0]
                dk32 = dk32.view(t, nkv, rep, -1).sum(2)
                dv32 = dv32.view(t, nkv, rep, -1).sum(2)
            return (dq32.to(q.dtype), dk32.to(k.dtype), dv32.to(v.dtype),
                    None, None, None, None, None)
        # Fallback: blocked recompute with rocBLAS GEMMs (fp32 softmax).
        with torch.enable_grad():
            qg = q.detach().requires_grad_(True)
            kg = k.detach().requires_grad_(True)
            vg = v.detach().requires_grad_(True)
            ref = _attn_varlen_blocked_torch(
                qg, kg, vg, cu_seqlens, ctx.causal, ctx.scale,
                window=ctx.window,
            )
            dq, dk, dv = torch.autograd.grad(ref, (qg, kg, vg), grad_out)
        return dq, dk, dv, None, None, None, None, None


def attn_varlen(q, k, v, cu_seqlens, max_seqlen, causal=True, softmax_scale=None,
                window=None):
    scale = softmax_scale or (1.0 / math.sqrt(q.shape[-1]))
    if window is not None and window >= int(max_seqlen):
        window = None  # window wider than any sequence: plain causal
    if (
        _ops.use_hip(q)
        and q.dtype == torch.bfloat16
        and q.shape[-1] in (64, 128)
        and (window is None or causal)
    ):
        return _AttnVarlenFn.apply(
            q.contiguous(), k.contiguous(), v.contiguous(),
            cu_seqlens, max_seqlen, causal, scale, window,
        )
    if q.is_cuda:
        # odd head dims / dtypes / binding sliding window: batched rocBLAS
        # path (still GPU)
        return _attn_varlen_blocked_torch(q, k, v, cu_seqlens, causal, scale,
                                          window=window)
    return attn_varlen_ref(q, k, v, cu_seqlens, causal, scale, window=window)


# ---------------------------------------------------------------------------
# Attention — single-token decode over contiguous KV cache
# ---------------------------------------------------------------------------
def attn_decode_ref(
    q: torch.Tensor,  # [bs, nq, hd] — the new token's q
    k_cache: torch.Tensor,  # [bs, max_len, nkv, hd]
    v_cache: torch.Tensor,
    cache_seqlens: torch.Tensor,  # [bs] int32 — valid length INCLUDING new token
    softmax_scale: Optional[float] = None,
    window: Optional[int] = None,
) -> torch.Tensor:
    bs, nq, hd = q.shape
    nkv = k_cache.shape[2]
    rep = nq // nkv
    scale = softmax_scale or (1.0 / math.sqrt(hd))
    out = torch.empty_like(q)
    for b in range(bs):
        L = int(cache_seqlens[b])
        lo = max(0, L - window) if window is not None else 0
        kb = k_cache[b, lo:L].transpose(0, 1).float()  # [nkv, L-lo, hd]
        vb = v_cache[b, lo:L].transpose(0, 1).float()
        if rep > 1:
            kb = kb.repeat_interleave(rep, dim=0)
            vb = vb.repeat_interleave(rep, dim=0)
        qb = q[b].unsqueeze(1).float()  # [nq, 1, hd]
        scores = torch.matmul(qb, kb.transpose(-1, -2)) * scale
        probs = torch.softmax(scores, dim=-1)
        out[b] = torch.matmul(probs, vb).squeeze(1).to(q.dtype)
    return out


def attn_decode(q, k_cache, v_cache, cache_seqlens, softmax_scale=None,
                window=None):
    scale = softmax_scale or (1.0 / math.sqrt(q.shape[-1]))
    if window is not None and k_cache.shape[1] <= window:
        window = None  # cache can never exceed the window: plain causal
    if (
        _ops.use_hip(q)
        and q.dtype == torch.bfloat16
        and q.shape[-1] in (64, 128)
        and q.shape[1] // k_cache.shape[2] in (1, 2, 4, 8)
    ):
        C = _ops.require_hip()
        return C.attn_decode(q, k_cache, v_cache, cache_seqlens, scale,
                             int(window or 0))
    return attn_decode_ref(q, k_cache, v_cache, cache_seqlens, scale,
                           window=window)


# ---------------------------------------------------------------------------
# GAE over packed sequences
# ---------------------------------------------------------------------------
def gae_ref(
    rewards: torch.Tensor,  # [total] fp32 — per-token rewards (shifted: len-1 per seq)
    values: torch.Tensor,  # [total + bs] fp32 — values with one extra bootstrap per seq
    cu_seqlens: torch.Tensor,  # [bs+1] — over the REWARD lengths
    bootstrap: torch.Tensor,  # [bs] bool — whether the last value bootstraps
    gamma: float,
    lam: float,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """advantages [total], returns [total].  Mirrors cugae1d_nolp_misalign
    (reference csrc/cugae/gae.cu:10): values for sequence i live at
    [cu[i]+i, cu[i+1]+i+1) — one longer than rewards."""
    adv = torch.zeros_like(rewards)
    ret = torch.zeros_like(rewards)
    cu = cu_seqlens.tolist()
    bs = len(cu) - 1
    for i in range(bs):
        rs, re = cu[i], cu[i + 1]
        vs = rs + i
        lastgae = 0.0
        L = re - rs
        for t in range(L - 1, -1, -1):
            nex = values[vs + t + 1]
            if t == L - 1 and not bool(bootstrap[i]):
                nex = values.new_zeros(())
            delta = rewards[rs + t] + gamma * nex - values[vs + t]
            lastgae = delta + gamma * lam * lastgae
            adv[rs + t] = lastgae
            ret[rs + t] = lastgae + values[vs + t]
    return adv, ret


def gae(rewards, values, cu_seqlens, bootstrap, gamma, lam):
    if _ops.use_hip(rewards):
        C = _ops.require_hip()
        return C.gae_1d(rewards, values, cu_seqlens, bootstrap, float(gamma), float(lam))
    return gae_ref(rewards, values, cu_seqlens, bootstrap, gamma, lam)


# ---------------------------------------------------------------------------
# Flat-parameter interval gather/scatter
# ---------------------------------------------------------------------------
def slice_intervals_ref(src: torch.Tensor, intervals: torch.Tensor) -> torch.Tensor:
    outs = [src[s:e] for s, e in intervals.tolist()]
    return torch.cat(outs) if outs else src.new_empty(0)


def set_intervals_ref(src: torch.Tensor, dst: torch.Tensor, intervals: torch.Tensor):
    off = 0
    for s, e in intervals.tolist():
        n = e - s
        dst[s:e] = src[off : off + n]
        off += n
    return dst


def slice_intervals(src, intervals):
    if _ops.use_hip(src) and intervals.shape[0] >= 128:
        C = _ops.require_hip()
        return C.slice_intervals(src, intervals.to(src.device))
    return slice_intervals_ref(src, intervals)


def set_intervals(src, dst, intervals):
    if _ops.use_hip(dst) and intervals.shape[0] >= 128:
        C = _ops.require_hip()
        C.set_intervals(src, dst, intervals.to(dst.device))
        return dst
    return set_intervals_ref(src, dst, intervals)


def merge_intervals(intervals: List[Tuple[int, int]]) -> List[Tuple[int, int]]:
    """Coalesce adjacent/overlapping [a,b) pairs (reference:
    csrc/interval_op/interval_op.cpp merge_intervals)."""
    if not intervals:
        return []
    intervals = sorted(intervals)
    out = [list(intervals[0])]
    for s, e in intervals[1:]:
        if s <= out[-1][1]:
            out[-1][1] = max(out[-1][1], e)
        else:
            out.append([s, e])
    return [tuple(x) for x in out]


# ---------------------------------------------------------------------------
# Weight-streaming skinny GEMM (decode M <= 16); wins on square-ish
# projections (o/down: ~1.6-2x over hipBLASLt), loses on very wide N —
# callers pick per shape (tools/bench_skinny.py is the A/B harness).
# ---------------------------------------------------------------------------
def maybe_skinny_linear(x: torch.Tensor, w: torch.Tensor,
                        residual: Optional[torch.Tensor] = None) -> Optional[torch.Tensor]:
    if not (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and not torch.is_grad_enabled()
        and x.dim() == 2
        and x.shape[0] <= 16
        and x.stride(1) == 1
        and x.stride(0) == x.shape[1]
        and w.stride(1) == 1
        and w.shape[0] % 64 == 0
        and w.shape[1] % 32 == 0
        # in-context the weight-streaming kernel also edges out hipBLASLt
        # on the wide-N shapes (A/B: 3.09 vs 3.05 samples/s); opt out with
        # REALHF_AMD_NO_SKINNY_WIDE=1
        and (
            w.shape[0] <= 2 * w.shape[1]
            or os.environ.get("REALHF_AMD_NO_SKINNY_WIDE") != "1"
        )
        and _ops.hip_available()
    ):
        return None
    from realhf_amd.base.constants import get_global_memory_buffer

    C = _ops.require_hip()
    M, K = x.shape
    N = w.shape[0]
    splitk = 8
    while K // splitk > _sg_kslice_max():
        splitk *= 2
    ws = get_global_memory_buffer().get_tensor(
        (2 * splitk * 16 * N,), torch.float32, "skinny_gemm_ws"
    )
    if residual is not None:
        residual = residual.contiguous()
    if os.environ.get("REALHF_AMD_SKINNY_V2") == "1":
        # v2 (in-launch semaphore combine) measured SLOWER at every decode
        # shape: the per-block agent-scope release (buffer_wbl2) does not
        # amortize at 1.5k-2.7k blocks (qkv 67us vs 22.8us at splitk 8).
        # Kept for reference / small-grid experiments only.
        return C.skinny_gemm2(x, w, ws, _skinny_sem(x.device), splitk, residual)
    return C.skinny_gemm(x, w, ws, splitk, residual)


def _sg_kslice_max() -> int:
    """Max K elements per split-K slice (tunes workgroup count); A/B via
    REALHF_AMD_SG_KSLICE, default 1024."""
    global _SG_KSLICE
    if _SG_KSLICE is None:
        _SG_KSLICE = int(os.environ.get("REALHF_AMD_SG_KSLICE", 1024))
    return _SG_KSLICE


_SG_KSLICE = None


def skinny_linear_nc(x: torch.Tensor, w: torch.Tensor) -> Optional[torch.Tensor]:
    """Like maybe_skinny_linear but WITHOUT the split-K combine: returns
    the fp32 partial slabs [nks, M, N], to be summed in the CONSUMING
    kernel's prologue (launch-boundary reduce: rope_qkv_decode /
    add_rmsnorm_fwd / swiglu_fwd all accept slabs).  The slabs live in the
    shared skinny workspace — consume them before the next skinny call."""
    if not (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and not torch.is_grad_enabled()
        and x.dim() == 2
        and x.shape[0] <= 16
        and x.stride(1) == 1
        and x.stride(0) == x.shape[1]
        and w.stride(1) == 1
        and w.shape[0] % 64 == 0
        and w.shape[1] % 32 == 0
        and _ops.hip_available()
    ):
        return None
    from realhf_amd.base.constants import get_global_memory_buffer

    C = _ops.require_hip()
    K = x.shape[1]
    N = w.shape[0]
    splitk = 8
    while K // splitk > _sg_kslice_max():
        splitk *= 2
    ws = get_global_memory_buffer().get_tensor(
        (2 * splitk * 16 * N,), torch.float32, "skinny_gemm_ws"
    )
    return C.skinny_gemm_nc(x, w, ws, splitk)


_SKINNY_SEM: dict = {}


def _skinny_sem(device) -> torch.Tensor:
    """Per-device split-K semaphore buffer (one int32 per n-tile; 1024
    covers N up to 65536).  Zeroed ONCE at allocation; the kernel leaves
    every counter back at zero."""
    t = _SKINNY_SEM.get(device)
    if t is None:
        t = torch.zeros(1024, dtype=torch.int32, device=device)
        _SKINNY_SEM[device] = t
    return t


# ---------------------------------------------------------------------------
# Fused AdamW on flat buffers
# ---------------------------------------------------------------------------
def fused_adamw_ref(
    param_f32: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step: int,
    bf16_out: Optional[torch.Tensor] = None,
    grad_scale: float = 1.0,
):
    g = grad.float() * grad_scale
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1**step
    bc2 = 1 - beta2**step
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    param_f32.mul_(1 - lr * weight_decay)
    param_f32.addcdiv_(exp_avg, denom, value=-lr / bc1)
    if bf16_out is not None:
        bf16_out.copy_(param_f32)
    return param_f32


def fused_adamw(param_f32, grad, exp_avg, exp_avg_sq, lr, beta1, beta2, eps,
                weight_decay, step, bf16_out=None, grad_scale=1.0):
    if _ops.use_hip(param_f32):
        C = _ops.require_hip()
        C.fused_adamw(
            param_f32, grad, exp_avg, exp_avg_sq,
            bf16_out if bf16_out is not None else param_f32.new_empty(0).to(torch.bfloat16),
            float(lr), float(beta1), float(beta2), float(eps),
            float(weight_decay), int(step), float(grad_scale),
            bf16_out is not None,
        )
        return param_f32
    return fused_adamw_ref(
        param_f32, grad, exp_avg, exp_avg_sq, lr, beta1, beta2, eps,
        weight_decay, step, bf16_out, grad_scale,
    )
