"""Packed-batch generation engine.

Reference semantics: realhf/impl/model/nn/real_llm_generate.py (generate:252,
genstep:26, prepare_generate_inputs:144, maybe_capture_cudagraph:214).

Design (MI355X-first):
- prefill runs the packed varlen forward once, writing the contiguous
  per-layer KV caches [bs, prompt_max + max_new, nkv_local, hd];
- decode is one fixed-shape forward per token (bs tokens), hipGraph-captured
  after the first step (launch-bound otherwise: ~32 small kernels/layer);
- sampling: temperature → top-k → top-p → Categorical, identical across TP
  ranks via a shared-seed torch.Generator (no sampling collective — the
  reference all-reduces sampled tokens instead, an extra hop per token that
  a shared RNG makes unnecessary);
- termination: all sequences hit EOS or max_new_tokens; min_new_tokens
  masks EOS early.
"""
import dataclasses
import os
from typing import Optional, Tuple

import torch

from realhf_amd.api.model import GenerationHyperparameters
from realhf_amd.base import logging
from realhf_amd.models.real_model import ReaLModel
from realhf_amd.utils.functional import top_k_top_p_logits  # noqa: F401 — re-exported for interface use

logger = logging.getLogger("generate")


@dataclasses.dataclass
class GenerationOutput:
    gen_tokens: torch.Tensor  # [bs, gen_len] (right-padded with pad_id)
    gen_logprobs: torch.Tensor  # [bs, gen_len] — logprob of each sampled token
    gen_lengths: torch.Tensor  # [bs] — actual generated length (incl. eos)
    no_eos_mask: torch.Tensor  # [bs] bool — True if never hit EOS
    prompt_logprobs: Optional[torch.Tensor] = None  # [total_prompt - bs] packed
    # [bs, gen_len, vocab] bool, True = vocab entry REMOVED by top-k/p at
    # that step (reference genstep logits_mask, real_llm_generate.py:131-136;
    # emitted only when gconfig.force_no_logits_mask is False)
    logits_mask: Optional[torch.Tensor] = None


def _sample_from_logits(
    logits: torch.Tensor,  # [bs, vocab] fp32
    gconfig: GenerationHyperparameters,
    generator: Optional[torch.Generator],
    return_mask: bool = False,
) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
    """Returns (tokens, logprob, mask).  mask (only when return_mask) is
    [bs, vocab] bool, True = entry removed by top-k/p; in that mode the
    logprob is under the FILTERED distribution (reference genstep samples
    from the warped logits, real_llm_generate.py:80-93) so behavior
    logprobs line up with masked re-forward logprobs."""
    if gconfig.temperature != 1.0 and not gconfig.greedy:
        logits = logits / max(gconfig.temperature, 1e-5)
    if gconfig.greedy:
        tokens = logits.argmax(dim=-1)
        logp = torch.log_softmax(logits, dim=-1)
        return tokens, logp.gather(-1, tokens.unsqueeze(-1)).squeeze(-1), None
    V = logits.shape[-1]
    k = gconfig.top_k if 0 < gconfig.top_k < V else V
    if k < V:
        # work in the compacted top-k space: selection + a k-long cumsum
        # instead of a full-vocab sort per token (the old path sorted all
        # 32k logits every decode step)
        vals, idx = torch.topk(logits, k, dim=-1)  # sorted descending
    else:
        vals, idx = torch.sort(logits, descending=True, dim=-1)
    probs = torch.softmax(vals, dim=-1)
    if 0.0 < gconfig.top_p < 1.0:
        cum = probs.cumsum(dim=-1)
        remove = cum - probs > gconfig.top_p
        vals = vals.masked_fill(remove, float("-inf"))
        probs = torch.softmax(vals, dim=-1)
    sel = torch.multinomial(probs, 1, generator=generator)
    tokens = idx.gather(-1, sel).squeeze(-1)
    if return_mask:
        mask = torch.ones_like(logits, dtype=torch.bool)
        mask.scatter_(1, idx, vals == float("-inf"))  # kept slots -> False
        logp_f = torch.log_softmax(vals, dim=-1)  # filtered distribution
        return tokens, logp_f.gather(-1, sel).squeeze(-1), mask
    logp_all = torch.log_softmax(logits, dim=-1)  # logprob under UNFILTERED dist
    return tokens, logp_all.gather(-1, tokens.unsqueeze(-1)).squeeze(-1), None


class DecodeGraph:
    """hipGraph capture of the decode-step forward.

    Static buffers: input token ids [bs], cache_seqlens [bs], output
    logits.  The KV caches are already static.  Replay = copy tokens in,
    bump cache_seqlens, one graph launch.
    """

    def __init__(self, model: ReaLModel, kv_caches, bs: int):
        self.model = model
        self.kv_caches = kv_caches
        self.bs = bs
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.in_tokens = None
        self.in_cache_seqlens = None
        self.out = None

    def _eager(self, tokens, cache_seqlens):
        return self.model(
            packed_input_ids=tokens,
            kv_caches=self.kv_caches,
            cache_seqlens=cache_seqlens,
            decode=True,
        )

    def capture(self, tokens, cache_seqlens):
        import os

        if os.environ.get("REALHF_AMD_FORCE_GRAPH_FAIL") == "1":
            # test hook: exercise the eager fallback path (a capture can
            # fail for real when the forward contains a graph-unsafe op,
            # e.g. an in-graph RCCL collective on some topologies)
            raise RuntimeError("forced capture failure (test hook)")
        self.in_tokens = tokens.clone()
        self.in_cache_seqlens = cache_seqlens.clone()
        torch.cuda.synchronize()
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._eager(self.in_tokens, self.in_cache_seqlens)
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.out = self._eager(self.in_tokens, self.in_cache_seqlens)
        logger.debug("decode hipGraph captured (bs=%d)", self.bs)

    def step(self, tokens, cache_seqlens):
        if self.graph is None:
            return self._eager(tokens, cache_seqlens)
        self.in_tokens.copy_(tokens)
        self.in_cache_seqlens.copy_(cache_seqlens)
        self.graph.replay()
        return self.out

    def destroy(self):
        self.graph = None


class GenerationSession:
    """Persistent KV caches + captured decode graph, reused across generate
    calls of the same shape (the capture costs ~100 eager layer launches;
    re-capturing per call was ~2 s/step at the 7B bench shape)."""

    def __init__(self, kv_caches, decoder):
        self.kv_caches = kv_caches
        self.decoder = decoder


def _get_session(model: ReaLModel, bs, cache_len, nkv_local, device):
    cfg = model.config
    key = (bs, cache_len, id(model.flat_param))
    sess = getattr(model, "_gen_session", None)
    if sess is not None and sess[0] == key:
        return sess[1]
    n_blocks = sum(1 for i in model.layer_indices if 1 <= i <= cfg.n_layers)
    kv_caches = [
        (
            torch.zeros(bs, cache_len, nkv_local, cfg.head_dim,
                        dtype=model.dtype, device=device),
            torch.zeros(bs, cache_len, nkv_local, cfg.head_dim,
                        dtype=model.dtype, device=device),
        )
        for _ in range(n_blocks)
    ]
    s = GenerationSession(kv_caches, DecodeGraph(model, kv_caches, bs))
    model._gen_session = (key, s)
    return s


@torch.no_grad()
def generate(
    model: ReaLModel,
    packed_prompts: torch.Tensor,  # [total_prompt] int64
    cu_seqlens: torch.Tensor,  # [bs+1] int32
    gconfig: GenerationHyperparameters,
    eos_token_id: Optional[int] = None,
    pad_token_id: int = 0,
    generator: Optional[torch.Generator] = None,
    return_prompt_logprobs: bool = False,
) -> GenerationOutput:
    assert model.pp_size == 1, "pp>1 generation goes through the pipe engine"
    from realhf_amd.base import constants as _c

    assert not (_c.has_current() and _c.sequence_parallel()), (
        "sequence parallelism is not supported during generation "
        "(reference: pipe_runner.py:856) — use a gen replica without SP"
    )
    cfg = model.config
    device = packed_prompts.device
    bs = cu_seqlens.shape[0] - 1
    prompt_lens = (cu_seqlens[1:] - cu_seqlens[:-1]).to(device)
    max_prompt = int(prompt_lens.max())
    max_new = gconfig.max_new_tokens
    # bucket the cache length so different max_prompt values reuse the
    # same session/graph
    cache_len = ((max_prompt + max_new + 127) // 128) * 128

    nkv_local = max(cfg.n_kv_heads // model.tp_size, 1)
    sess = _get_session(model, bs, cache_len, nkv_local, device)
    kv_caches = sess.kv_caches

    # ---- prefill --------------------------------------------------------
    logits = model(
        packed_input_ids=packed_prompts,
        cu_seqlens=cu_seqlens,
        max_seqlen=max_prompt,
        kv_caches=kv_caches,
    )
    from realhf_amd.parallel import mappings

    prompt_logprobs = None
    if return_prompt_logprobs:
        from realhf_amd.parallel.tp import packed_shifted_logprobs

        prompt_logprobs = packed_shifted_logprobs(logits, cu_seqlens, packed_prompts)

    last_idx = (cu_seqlens[1:].long() - 1)
    last_logits = logits[last_idx]  # [bs, vocab/tp]
    last_logits = mappings.gather_from_tp_region(last_logits).float()

    gen_tokens = torch.full((bs, max_new), pad_token_id, dtype=torch.long, device=device)
    gen_logprobs = torch.zeros(bs, max_new, dtype=torch.float32, device=device)
    done = torch.zeros(bs, dtype=torch.bool, device=device)
    gen_lengths = torch.zeros(bs, dtype=torch.long, device=device)

    cache_seqlens = prompt_lens.to(torch.int32).clone()

    decoder = sess.decoder
    # graph capture requires the fully HIP decode path (the torch
    # fallback for odd head dims reads tensors on the host)
    use_graph = (
        gconfig.use_hip_graph
        and device.type == "cuda"
        and model.dtype == torch.bfloat16
        and cfg.head_dim in (64, 128)
        # MoE dispatch reads expert counts on the host (counts.cpu());
        # a D2H sync inside stream capture DEADLOCKS rather than erroring.
        # pad_to_capacity makes counts compile-time constants (no sync) —
        # graph capture of MoE decode is then possible, but stays opt-in
        # (REALHF_AMD_MOE_GRAPH=1) until validated on hardware.
        and (cfg.moe is None
             or (cfg.moe.pad_to_capacity
                 and os.environ.get("REALHF_AMD_MOE_GRAPH") == "1"))
    )

    # logits-mask mode (force_no_logits_mask=False): record which vocab
    # entries top-k/p removed at each step so downstream logprob passes
    # (ref_inf, actor_train) can mask consistently.  [bs, vocab] bool per
    # step — memory-heavy by design, hence the reference's opt-out flag.
    want_mask = not gconfig.force_no_logits_mask and not gconfig.greedy
    step_masks = [] if want_mask else None

    cur_logits = last_logits
    for t in range(max_new):
        if eos_token_id is not None and t < gconfig.min_new_tokens:
            cur_logits[:, eos_token_id] = float("-inf")
        tokens, logp, smask = _sample_from_logits(
            cur_logits, gconfig, generator, return_mask=want_mask)
        if want_mask:
            step_masks.append(smask)
        tokens = torch.where(done, torch.full_like(tokens, pad_token_id), tokens)
        gen_tokens[:, t] = tokens
        gen_logprobs[:, t] = torch.where(done, torch.zeros_like(logp), logp)
        gen_lengths += (~done).long()
        if eos_token_id is not None:
            done = done | (tokens == eos_token_id)
            # early-exit check costs a host sync — amortize it
            if (t & 15) == 15 and bool(done.all()):
                break
        if t == max_new - 1:
            break

        cache_seqlens += 1  # the new token's slot
        if use_graph and decoder.graph is None and not getattr(decoder, "_capture_failed", False):
            try:
                decoder.capture(tokens, cache_seqlens)
            except Exception as e:  # graph-unsafe fallback path in use
                logger.warning("hipGraph capture failed (%s); eager decode", e)
                decoder.graph = None
                decoder._capture_failed = True
                torch.cuda.synchronize()
        logits_step = decoder.step(tokens, cache_seqlens)
        logits_step = mappings.gather_from_tp_region(logits_step).float()
        cur_logits = logits_step

    gmax = int(gen_lengths.max())
    logits_mask = None
    if want_mask and step_masks:
        logits_mask = torch.stack(step_masks[:gmax], dim=1)  # [bs, <=gmax, V]
        if logits_mask.shape[1] < gmax:  # early-exit break before step gmax
            pad_m = logits_mask.new_zeros(
                logits_mask.shape[0], gmax - logits_mask.shape[1],
                logits_mask.shape[2])
            logits_mask = torch.cat([logits_mask, pad_m], dim=1)
    return GenerationOutput(
        gen_tokens=gen_tokens[:, : int(gen_lengths.max())],
        gen_logprobs=gen_logprobs[:, : int(gen_lengths.max())],
        gen_lengths=gen_lengths,
        no_eos_mask=~done,
        prompt_logprobs=prompt_logprobs,
        logits_mask=logits_mask,
    )


def concat_prompt_to_generation_output(
    packed_prompts: torch.Tensor,
    prompt_cu_seqlens: torch.Tensor,
    out: GenerationOutput,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Pack prompt+generation into one packed batch.

    Returns (packed_input_ids, cu_seqlens, prompt_mask) — reference:
    real_llm_generate.py:451."""
    bs = prompt_cu_seqlens.shape[0] - 1
    seqs, masks = [], []
    for i in range(bs):
        s, e = int(prompt_cu_seqlens[i]), int(prompt_cu_seqlens[i + 1])
        gl = int(out.gen_lengths[i])
        seq = torch.cat([packed_prompts[s:e], out.gen_tokens[i, :gl]])
        seqs.append(seq)
        m = torch.zeros(seq.shape[0], dtype=torch.bool, device=seq.device)
        m[: e - s] = True
        masks.append(m)
    lens = [int(x.shape[0]) for x in seqs]
    cum = [0]
    for l in lens:
        cum.append(cum[-1] + l)
    cu = torch.tensor(cum, dtype=torch.int32, device=packed_prompts.device)
    return torch.cat(seqs), cu, torch.cat(masks)
