"""Transformer layer modules over flat-buffer parameter views.

Reference semantics: realhf/impl/model/nn/real_llm_base.py (ReaLModelBlock:103,
VocabPositionEmbedding:260, OutputHead:346, SequenceParallelCriticHead:356,
ParallelActorHead:370) and modules/{attn,mlp}.py.

Every module receives its parameters as a dict of tensors that are VIEWS
into the owning ReaLModel's contiguous flat buffer; modules never allocate
weights.  TP behavior comes from the current model scope's grid
(realhf_amd.base.constants); with tp_size == 1 every collective is a no-op
so the same code runs single-process on CPU.

MI355X specifics: wq/wk/wv (and gate/up) flat regions are adjacent, so the
module runs ONE hipBLASLt GEMM over the concatenated view; RMSNorm / RoPE /
SwiGLU / attention are the hand-written HIP kernels in realhf_amd.ops.
"""
import math
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.base import constants
from realhf_amd.ops import functional as ops
from realhf_amd.parallel import mappings


def _maybe_merged(params: Dict[str, torch.Tensor], names, dim_out_total):
    """If the named 2-D params are contiguous consecutive views of the flat
    buffer, return a single merged [sum_out, in] view; else None."""
    ts = [params[n] for n in names]
    base = ts[0]
    if any(t.dtype != base.dtype or t.device != base.device for t in ts):
        return None
    try:
        storages = {t.untyped_storage().data_ptr() for t in ts}
    except RuntimeError:
        return None
    if len(storages) != 1:
        return None
    off = base.storage_offset()
    for t in ts:
        if t.storage_offset() != off or not t.is_contiguous():
            return None
        off += t.numel()
    in_dim = base.shape[1]
    merged = base.as_strided((dim_out_total, in_dim), (in_dim, 1), base.storage_offset())
    return merged


def _linear(x, w, b=None):
    return F.linear(x, w, b)


class VocabEmbedding(nn.Module):
    """Vocab-parallel token embedding (+ optional learned positions).

    TP: each rank holds rows [r*V/tp, (r+1)*V/tp); out-of-range ids embed
    to zero and the partial results all-reduce over the TP group
    (reference: model_parallel/modules.py:53 ParallelEmbedding)."""

    def __init__(self, cfg: ReaLModelConfig, params: Dict[str, torch.Tensor]):
        super().__init__()
        self.cfg = cfg
        self.wte = params["0.wte.weight"]
        self.wpe = params.get("0.wpe.weight")

    def forward(self, packed_input_ids: torch.Tensor, positions: torch.Tensor):
        tp = constants.tp_world_size() if constants.has_current() else 1
        if tp > 1:
            r = constants.tp_rank()
            n = self.wte.shape[0]
            lo, hi = r * n, (r + 1) * n
            mask = (packed_input_ids >= lo) & (packed_input_ids < hi)
            local_ids = (packed_input_ids - lo).clamp(0, n - 1)
            h = F.embedding(local_ids, self.wte)
            h = h * mask.unsqueeze(-1).to(h.dtype)
            h = mappings.reduce_from_tp_region(h)
        else:
            h = F.embedding(packed_input_ids, self.wte)
        if self.cfg.embedding_multiplier is not None:
            h = h * self.cfg.embedding_multiplier
        if self.wpe is not None:
            h = h + F.embedding(positions, self.wpe)
        if constants.has_current() and constants.sequence_parallel():
            h = mappings.scatter_to_sp_region(h)
        return h


def _norm(cfg: ReaLModelConfig, x, w, b):
    if cfg.norm_type == "rms":
        return ops.rms_norm(x, w, cfg.layer_norm_epsilon)
    if cfg.norm_type == "gemma_rms":
        return ops.rms_norm(x, w, cfg.layer_norm_epsilon, gemma_style=True)
    return F.layer_norm(
        x.float(), (x.shape[-1],), w.float(), b.float() if b is not None else None,
        cfg.layer_norm_epsilon,
    ).to(x.dtype)


class ReaLModelBlock(nn.Module):
    """Pre-LN transformer block on packed sequences (reference:
    real_llm_base.py:103, modules/attn.py:31, modules/mlp.py)."""

    def __init__(self, cfg: ReaLModelConfig, layer_idx: int, params: Dict[str, torch.Tensor], tp_size: int = 1, ep_rank: int = 0, ep_size: int = 1):
        super().__init__()
        self.cfg = cfg
        self.i = layer_idx
        self.p = params
        self.tp_size = tp_size
        self.nq = cfg.n_heads // tp_size
        self.nkv = max(cfg.n_kv_heads // tp_size, 1)
        assert cfg.n_kv_heads % tp_size == 0 or tp_size % cfg.n_kv_heads == 0
        self.hd = cfg.head_dim
        i = layer_idx
        self._qkv_names = [f"{i}.attn.wq.weight", f"{i}.attn.wk.weight", f"{i}.attn.wv.weight"]
        self._gu_names = (
            [f"{i}.mlp.gate.weight", f"{i}.mlp.up.weight"]
            if cfg.activation in ("silu", "geglu") and cfg.moe is None
            else None
        )
        self.moe = None
        if cfg.moe is not None:
            from realhf_amd.models.moe import MoELayer

            self.moe = MoELayer(cfg, layer_idx, params, tp_size,
                                ep_rank=ep_rank, ep_size=ep_size)
        # LoRA adapters (injected by ReaLModel.attach_lora): short key
        # ("wq"/"wk"/"wv"/"wo") -> (A [r, in], B [out, r]) views
        self.lora: Dict[str, tuple] = {}
        self.lora_scale = 0.0

    def _lora_delta(self, x, short: str):
        a, b = self.lora[short]
        return ((x @ a.t()) @ b.t()) * self.lora_scale

    # -- attention --------------------------------------------------------
    def _qkv(self, x, fused_sp=None):
        qkv_dim = (self.nq + 2 * self.nkv) * self.hd
        merged = _maybe_merged(self.p, self._qkv_names, qkv_dim)
        if fused_sp is not None and merged is not None:
            # x is PRE-mapping: the fused fn applies (copy|SP-gather) in
            # forward and overlaps the bwd grad collective with the
            # weight-grad GEMM (mappings._ColumnParallelLinear)
            qkv = mappings.column_parallel_linear(x, merged, fused_sp)
        elif fused_sp is not None:
            x = (mappings.gather_from_sp_region(x) if fused_sp
                 else mappings.copy_to_tp_region(x))
            qkv = torch.cat([_linear(x, self.p[n]) for n in self._qkv_names], dim=-1)
        elif merged is not None:
            qkv = _linear(x, merged)
        else:
            qkv = torch.cat([_linear(x, self.p[n]) for n in self._qkv_names], dim=-1)
        i = self.i
        if f"{i}.attn.wq.bias" in self.p:
            bias = torch.cat(
                [self.p[f"{i}.attn.w{c}.bias"] for c in "qkv"], dim=0
            )
            qkv = qkv + bias
        q, k, v = qkv.split(
            [self.nq * self.hd, self.nkv * self.hd, self.nkv * self.hd], dim=-1
        )
        if self.lora:
            if "wq" in self.lora:
                q = q + self._lora_delta(x, "wq")
            if "wk" in self.lora:
                k = k + self._lora_delta(x, "wk")
            if "wv" in self.lora:
                v = v + self._lora_delta(x, "wv")
        t = qkv.shape[0]  # with fused SP, x is the pre-gather shard
        return (
            q.reshape(t, self.nq, self.hd),
            k.reshape(t, self.nkv, self.hd),
            v.reshape(t, self.nkv, self.hd),
        )

    def forward(
        self,
        x: torch.Tensor,  # [total, h]
        cu_seqlens: torch.Tensor,
        max_seqlen: int,
        positions: torch.Tensor,  # [total]
        k_cache: Optional[torch.Tensor] = None,  # [bs, maxlen, nkv, hd]
        v_cache: Optional[torch.Tensor] = None,
        cache_seqlens: Optional[torch.Tensor] = None,  # [bs]
        decode: bool = False,
    ):
        i = self.i
        cfg = self.cfg
        h = _norm(cfg, x, self.p[f"{i}.attn.ln.weight"], self.p.get(f"{i}.attn.ln.bias"))
        sp = constants.has_current() and constants.sequence_parallel()
        # fused column linear (async bwd comm) takes the PRE-mapping h;
        # decode/no-grad/LoRA keep the explicit mapping route
        fused_col = (self.tp_size > 1 and torch.is_grad_enabled()
                     and not self.lora and not decode)
        if not fused_col:
            if sp:
                h = mappings.gather_from_sp_region(h)
            else:
                h = mappings.copy_to_tp_region(h)

        scale = 1.0 / math.sqrt(self.hd)
        if cfg.scale_attn_by_inverse_layer_idx:
            scale = scale / float(self.i)

        # ---- fused decode fast path: one kernel for qkv-split + bias +
        # RoPE + KV append (HIP only, bf16, no qk-layernorm)
        import os as _os

        if (
            decode
            and h.is_cuda
            and h.dtype == torch.bfloat16
            and not cfg.qk_layernorm
            and self.hd in (64, 128)
            and _os.environ.get("REALHF_AMD_NO_FUSED_DECODE") != "1"
            and not self.lora  # adapters applied in the standard path
        ):
            from realhf_amd import ops as _ops_pkg

            C = _ops_pkg.require_hip()
            qkv_dim = (self.nq + 2 * self.nkv) * self.hd
            merged = _maybe_merged(self.p, self._qkv_names, qkv_dim)
            if merged is not None:
                # launch-boundary reduce: hand the split-K fp32 slabs
                # straight to rope_qkv_decode's prologue (no combine pass)
                qkv_raw = ops.skinny_linear_nc(h, merged)
                if qkv_raw is None:
                    qkv_raw = ops.maybe_skinny_linear(h, merged)
                if qkv_raw is None:
                    qkv_raw = _linear(h, merged)
            else:
                qkv_raw = torch.cat(
                    [_linear(h, self.p[n]) for n in self._qkv_names], dim=-1
                )
            bias = None
            if f"{i}.attn.wq.bias" in self.p:
                bias = torch.cat(
                    [self.p[f"{i}.attn.w{c}.bias"] for c in "qkv"], dim=0
                ).to(qkv_raw.dtype)
            cos, sin = ops.rotary_cache.get(
                self.hd, int(k_cache.shape[1]), cfg.rotary_base, x.device,
                scaling=cfg.rotary_scaling,
                scaling_type=cfg.rotary_scaling_type,
                orig_max_pos=cfg.max_position_embeddings,
            )
            q = C.rope_qkv_decode(
                qkv_raw, bias, k_cache, v_cache,
                cache_seqlens, cos, sin, self.nq, cfg.apply_rotary,
            )
            attn_out = ops.attn_decode(q, k_cache, v_cache, cache_seqlens,
                                       scale, window=cfg.sliding_window)
            attn_out = attn_out.reshape(attn_out.shape[0], self.nq * self.hd)
            wo = self.p[f"{i}.attn.wo.weight"]
            if (
                self.tp_size == 1
                and f"{i}.attn.wo.bias" not in self.p
                and cfg.norm_type == "rms"
                and self.moe is None
            ):
                # launch-boundary reduce: o's split-K slabs are summed in
                # the fused add+rmsnorm prologue (no combine, no bf16 o)
                o_parts = ops.skinny_linear_nc(attn_out, wo)
                if o_parts is not None:
                    h2, x2 = C.add_rmsnorm_fwd(
                        o_parts, x, self.p[f"{i}.mlp.ln.weight"],
                        cfg.layer_norm_epsilon,
                    )
                    return self._mlp_body(h2, sp, residual=x2)
            o = ops.maybe_skinny_linear(attn_out, wo)
            if o is None:
                o = _linear(attn_out, wo)
            o = mappings.reduce_from_tp_region(o)
            if f"{i}.attn.wo.bias" in self.p:
                o = o + self.p[f"{i}.attn.wo.bias"]
            if cfg.norm_type == "rms" and self.moe is None:
                # fused (x + o) + mlp-norm, then the MLP body (the down
                # projection folds the x2 residual into its combine)
                h2, x2 = C.add_rmsnorm_fwd(
                    o, x, self.p[f"{i}.mlp.ln.weight"], cfg.layer_norm_epsilon
                )
                return self._mlp_body(h2, sp, residual=x2)
            x = x + o
            return self._mlp(x, sp)

        q, k, v = self._qkv(h, fused_sp=(sp if fused_col else None))
        if cfg.apply_rotary:
            # graph-capture-safe length bound: never read positions back
            if decode:
                rot_len = int(k_cache.shape[1])
            elif k_cache is not None:
                # generation prefill: use the SESSION's cache length so
                # dynamic-NTK picks one base for prefill AND decode (a
                # per-length base would rotate cached K with a different
                # basis than the decode queries reading it)
                rot_len = max(int(k_cache.shape[1]),
                              cfg.max_position_embeddings)
            else:
                ms = max_seqlen
                if ms is None:  # CP: token shard with explicit positions
                    from realhf_amd.parallel import cp as cp_mod

                    c = cp_mod.current()
                    ms = (c.info.full_max if c is not None
                          else cfg.max_position_embeddings)
                rot_len = max(int(ms), cfg.max_position_embeddings)
            cos, sin = ops.rotary_cache.get(
                self.hd,
                rot_len,
                cfg.rotary_base,
                x.device,
                scaling=cfg.rotary_scaling,
                scaling_type=cfg.rotary_scaling_type,
                orig_max_pos=cfg.max_position_embeddings,
            )
            q = ops.apply_rotary(q, cos, sin, positions, cfg.rotary_interleaved)
            k = ops.apply_rotary(k, cos, sin, positions, cfg.rotary_interleaved)

        if decode:
            # one new token per sequence; write into cache then attend
            bs = k_cache.shape[0]
            assert q.shape[0] == bs
            idx = (cache_seqlens.long() - 1).clamp(min=0)
            b_idx = torch.arange(bs, device=x.device)
            k_cache[b_idx, idx] = k.to(k_cache.dtype)
            v_cache[b_idx, idx] = v.to(v_cache.dtype)
            attn_out = ops.attn_decode(
                q, k_cache, v_cache, cache_seqlens, scale,
                window=cfg.sliding_window,
            )
        else:
            from realhf_amd.parallel import cp as cp_mod

            cpctx = cp_mod.current()
            if cpctx is not None:
                # Ulysses context parallelism: tokens are sharded over
                # the CP group; one all-to-all gives this rank ALL
                # tokens for 1/cp of the heads, the varlen kernel runs
                # on full sequences, and a second all-to-all restores
                # the (token-shard, all-heads) layout (parallel/cp.py).
                assert k_cache is None, "CP: no KV-cache prefill/generation"
                assert q.shape[1] % cpctx.size == 0, (
                    f"q heads {q.shape[1]} not divisible by cp {cpctx.size}")
                assert k.shape[1] % cpctx.size == 0, (
                    f"kv heads {k.shape[1]} not divisible by cp {cpctx.size}")
                q = cp_mod.seq_gather_head_scatter(q)
                k = cp_mod.seq_gather_head_scatter(k)
                v = cp_mod.seq_gather_head_scatter(v)
                attn_out = ops.attn_varlen(
                    q, k, v, cpctx.info.full_cu, cpctx.info.full_max,
                    causal=True, softmax_scale=scale,
                    window=cfg.sliding_window,
                )
                attn_out = cp_mod.head_gather_seq_scatter(attn_out)
            else:
                if k_cache is not None:
                    # prefill: write all tokens into the cache
                    bs = k_cache.shape[0]
                    seq_id = torch.bucketize(
                        torch.arange(x.shape[0], device=x.device),
                        cu_seqlens[1:].long(),
                        right=True,
                    )
                    k_cache[seq_id, positions] = k.detach().to(k_cache.dtype)
                    v_cache[seq_id, positions] = v.detach().to(v_cache.dtype)
                attn_out = ops.attn_varlen(
                    q, k, v, cu_seqlens, max_seqlen, causal=True,
                    softmax_scale=scale, window=cfg.sliding_window,
                )
        attn_out = attn_out.reshape(attn_out.shape[0], self.nq * self.hd)
        o = _linear(attn_out, self.p[f"{i}.attn.wo.weight"])
        if "wo" in self.lora:
            o = o + self._lora_delta(attn_out, "wo")
        if sp:
            o = mappings.reduce_scatter_to_sp_region(o)
        else:
            o = mappings.reduce_from_tp_region(o)
        if f"{i}.attn.wo.bias" in self.p:
            o = o + self.p[f"{i}.attn.wo.bias"]
        x = x + o
        return self._mlp(x, sp)

    def _mlp(self, x, sp):
        cfg = self.cfg
        i = self.i
        h = _norm(cfg, x, self.p[f"{i}.mlp.ln.weight"], self.p.get(f"{i}.mlp.ln.bias"))
        if self.moe is not None:
            return x + self.moe(h)
        return x + self._mlp_body(h, sp)  # residual not folded here

    def _mlp_body(self, h, sp, residual=None):
        """norm-output -> MLP delta; with `residual` (tp==1 decode path)
        the down-projection's combine kernel adds it in and the return
        value is the full residual-stream output."""
        cfg = self.cfg
        i = self.i
        fused_col = self.tp_size > 1 and torch.is_grad_enabled()
        if fused_col and cfg.activation in ("silu", "geglu"):
            idim_local = self.p[f"{i}.mlp.gate.weight"].shape[0]
            merged_f = _maybe_merged(self.p, self._gu_names, 2 * idim_local)
            if merged_f is None:
                fused_col = False
        elif fused_col:
            fused_col = False  # single-up MLP: keep the mapping route
        if not fused_col:
            if sp:
                h = mappings.gather_from_sp_region(h)
            else:
                h = mappings.copy_to_tp_region(h)
        if cfg.activation in ("silu", "geglu"):
            idim_local = self.p[f"{i}.mlp.gate.weight"].shape[0]
            merged = _maybe_merged(self.p, self._gu_names, 2 * idim_local)
            act = None
            if fused_col:
                gu = mappings.column_parallel_linear(h, merged, sp)
            elif merged is not None:
                if cfg.activation == "silu" and self.tp_size == 1:
                    # launch-boundary reduce: slabs summed in swiglu
                    gu_parts = ops.skinny_linear_nc(h, merged)
                    if gu_parts is not None:
                        from realhf_amd import ops as _ops_pkg

                        act = _ops_pkg.require_hip().swiglu_fwd(gu_parts)
                if act is None:
                    gu = ops.maybe_skinny_linear(h, merged)
                    if gu is None:
                        gu = _linear(h, merged)
            elif not fused_col:
                gu = torch.cat(
                    [_linear(h, self.p[n]) for n in self._gu_names], dim=-1
                )
            if act is not None:
                pass
            elif cfg.activation == "silu":
                act = ops.swiglu(gu)
            else:  # geglu (gemma)
                gate, up = gu.chunk(2, dim=-1)
                act = (
                    F.gelu(gate.float(), approximate="tanh") * up.float()
                ).to(gu.dtype)
        else:
            up = _linear(h, self.p[f"{i}.mlp.up.weight"], self.p.get(f"{i}.mlp.up.bias"))
            act = F.gelu(up, approximate="tanh")
        wd = self.p[f"{i}.mlp.down.weight"]
        use_resid_fold = residual is not None and self.tp_size == 1
        down = ops.maybe_skinny_linear(
            act, wd, residual=residual if use_resid_fold else None
        )
        folded = down is not None and use_resid_fold
        if down is None:
            down = _linear(act, wd)
        if folded:
            return down  # residual already added
        if residual is not None:
            # fold failed (shape/grad): plain add at the end
            pass
        if sp:
            down = mappings.reduce_scatter_to_sp_region(down)
        else:
            down = mappings.reduce_from_tp_region(down)
        if f"{i}.mlp.down.bias" in self.p:
            down = down + self.p[f"{i}.mlp.down.bias"]
        if residual is not None:
            return residual + down
        return down


class OutputHead(nn.Module):
    """Final norm + LM head (vocab-parallel) or critic head (replicated
    1-dim output) (reference: real_llm_base.py:346-392)."""

    def __init__(self, cfg: ReaLModelConfig, params: Dict[str, torch.Tensor],
                 tied_embedding_weight: Optional[torch.Tensor] = None):
        super().__init__()
        self.cfg = cfg
        self.p = params
        self.tied_w = tied_embedding_weight
        self.i = cfg.n_layers + 1

    def forward(self, x):
        cfg = self.cfg
        h = _norm(cfg, x, self.p[f"{self.i}.ln_f.weight"], self.p.get(f"{self.i}.ln_f.bias"))
        if constants.has_current() and constants.sequence_parallel():
            h = mappings.gather_from_sp_region(h)
        w = self.tied_w if self.tied_w is not None else self.p.get(f"{self.i}.head.weight")
        if cfg.is_critic:
            return _linear(h.float(), w.float())  # [total, 1] fp32
        tp = constants.tp_world_size() if constants.has_current() else 1
        if tp > 1 and torch.is_grad_enabled():
            # fused copy-to-TP + GEMM: bwd all-reduce overlaps dW
            return mappings.column_parallel_linear(h, w, False)
        h = mappings.copy_to_tp_region(h)
        if not torch.is_grad_enabled():
            # decode: the vocab head streams 262 MB of weights per token —
            # the weight-streaming skinny kernel beats hipBLASLt at M<=16
            o = ops.maybe_skinny_linear(h, w)
            if o is not None:
                return o
        return _linear(h, w)  # [total, vocab/tp] — vocab-parallel logits
