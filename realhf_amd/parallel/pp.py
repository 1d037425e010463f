"""Pipeline-parallel engine: fill-drain inference, 1F1B training, and
token-level pipelined generation.

Reference semantics: realhf/impl/model/parallelism/pipeline_parallel/
static_schedule.py (InferenceSchedule:155, TrainSchedule:319 — 1F1B,
GenerateSchedule:195) + backend/pipe_runner.py (instruction interpreter).

SPMD simplification: in this runtime every rank of an MFC's mesh holds
the SAME (metadata-complete) DP shard, so activation shapes are known on
every stage without the reference's cross-stage seqlen all-gather
(pipe_runner.py:70-88) — only hidden-state tensors move, over p2p on the
stage-adjacent global ranks (xGMI direct links).
"""
from typing import Callable, Dict, List, Optional

import torch
import torch.distributed as dist

from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import GenerationHyperparameters, PipelinableEngine
from realhf_amd.base import constants, logging, seeding
from realhf_amd.models import moe as moe_mod
from realhf_amd.runtime.engine import sample_to_packed

logger = logging.getLogger("pp")


_PENDING_SENDS: List = []


def _send(t: torch.Tensor, dst_global: int):
    """Non-blocking send (blocking sends deadlock 1F1B: a stage can be
    mid-send of fwd(i+1) while its peer is mid-send of grad(i)).  The
    buffer is kept alive until flushed."""
    buf = t.contiguous()
    _PENDING_SENDS.append((dist.isend(buf, dst=dst_global), buf))
    if len(_PENDING_SENDS) > 64:
        _flush_sends()


def _flush_sends():
    global _PENDING_SENDS
    for w, _ in _PENDING_SENDS:
        w.wait()
    _PENDING_SENDS = []


def _recv(shape, dtype, src_global: int, device):
    buf = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(buf, src=src_global)
    return buf


class PipelinedEngine(PipelinableEngine):
    """pp > 1 engine (train + forward + generate).  The module is a
    ReaLModel holding only this stage's layers."""

    def __init__(self, model, optimizer=None, n_mbs_train: Optional[int] = None):
        self.model = model
        self.optimizer = optimizer
        self.n_mbs_train = n_mbs_train

    @property
    def module(self):
        return self.model

    # ------------------------------------------------------------ helpers
    def _grid(self):
        return constants.grid()

    def _stage_fwd(self, mb: SequenceSample, hidden=None, kv_caches=None,
                   cache_seqlens=None, decode=False):
        if decode:
            ids = None
            cu = mx = None
            if self.model.is_first_stage:
                ids = mb  # mb is the token tensor in decode mode
            return self.model(
                packed_input_ids=ids, hidden_states=hidden,
                kv_caches=kv_caches, cache_seqlens=cache_seqlens, decode=True,
            )
        ids, cu, mx = sample_to_packed(mb)
        return self.model(
            packed_input_ids=ids if self.model.is_first_stage else None,
            cu_seqlens=cu, max_seqlen=mx, hidden_states=hidden,
            kv_caches=kv_caches, cache_seqlens=cache_seqlens,
        )

    # ------------------------------------------------------------ forward
    @torch.no_grad()
    def forward(self, input_: SequenceSample, n_mbs: Optional[int] = None,
                post_hook: Optional[Callable] = None,
                aggregate_fn: Callable = None):
        self.model.eval()
        g = self._grid()
        n_mbs = n_mbs or g.pp_size
        n_mbs = min(n_mbs, input_.bs)
        mbs = input_.split(n_mbs) if n_mbs > 1 else [input_]
        h_dim = self.model.config.hidden_dim
        dtype = self.model.dtype
        dev = self.model.flat_param.device
        outs = []
        # fill-drain: stage s processes mb i after receiving from s-1
        for mb in mbs:
            ids, cu, mx = sample_to_packed(mb)
            total = ids.shape[0]
            hidden = None
            if not self.model.is_first_stage:
                hidden = _recv((total, h_dim), dtype, g.pp_prev_global_rank(), dev)
            out = self._stage_fwd(mb, hidden)
            if not self.model.is_last_stage:
                _send(out, g.pp_next_global_rank())
            else:
                if post_hook is not None:
                    out = post_hook(out, mb)
                outs.append(out)
        _flush_sends()
        if not self.model.is_last_stage:
            return None
        agg = aggregate_fn or (lambda xs: torch.cat(xs, dim=0))
        return agg(outs)

    # -------------------------------------------------------------- train
    def train_batch(self, input_: SequenceSample, loss_fn: Callable,
                    version_steps: int = 0, n_mbs: Optional[int] = None) -> Dict:
        g = self._grid()
        self.model.train()
        self.optimizer.zero_grad()
        n_mbs = n_mbs or self.n_mbs_train or (2 * g.pp_size)
        n_mbs = max(min(n_mbs, input_.bs), 1)
        mbs = input_.split(n_mbs) if n_mbs > 1 else [input_]
        n_mbs = len(mbs)
        h_dim = self.model.config.hidden_dim
        dtype = self.model.dtype
        dev = self.model.flat_param.device
        stage, S = g.pp_rank, g.pp_size
        stats: Dict[str, float] = {}

        # 1F1B: warmup = S - stage - 1 forwards, then alternate
        warmup = min(S - stage - 1, n_mbs)
        fwd_state = {}  # mb idx -> (input_hidden, output)

        def do_fwd(i):
            mb = mbs[i]
            ids, _, _ = sample_to_packed(mb)
            total = ids.shape[0]
            hidden = None
            if not self.model.is_first_stage:
                hidden = _recv((total, h_dim), dtype, g.pp_prev_global_rank(), dev)
                hidden.requires_grad_(True)
            with torch.enable_grad(), moe_mod.aux_loss_collection():
                out = self._stage_fwd(mb, hidden)
                aux = moe_mod.pop_aux_losses()
            if not self.model.is_last_stage:
                _send(out.detach(), g.pp_next_global_rank())
            fwd_state[i] = (hidden, out, mb, aux)

        def do_bwd(i):
            hidden, out, mb, aux = fwd_state.pop(i)
            if self.model.is_last_stage:
                loss, st = loss_fn(out, mb)
                for a in aux:
                    loss = loss + a
                (loss / n_mbs).backward()
                for k, v in st.items():
                    stats[k] = stats.get(k, 0.0) + float(v) / n_mbs
            else:
                gout = _recv(tuple(out.shape), dtype, g.pp_next_global_rank(), dev)
                if aux:
                    # mid-stage MoE aux losses backprop locally alongside
                    # the received activation grads
                    s = torch.stack(aux).sum() / n_mbs
                    torch.autograd.backward([out, s], grad_tensors=[gout, None])
                else:
                    torch.autograd.backward(out, grad_tensors=gout)
            if hidden is not None:
                _send(hidden.grad, g.pp_prev_global_rank())

        fi = bi = 0
        for _ in range(warmup):
            do_fwd(fi)
            fi += 1
        while fi < n_mbs:
            do_fwd(fi)
            fi += 1
            do_bwd(bi)
            bi += 1
        while bi < n_mbs:
            do_bwd(bi)
            bi += 1

        _flush_sends()
        opt_stats = self.optimizer.step()
        stats.update(opt_stats)
        return stats

    @torch.no_grad()
    def eval_batch(self, input_: SequenceSample, loss_fn: Callable, n_mbs=None):
        raise NotImplementedError("pp eval_batch: use forward + loss on last stage")

    # ----------------------------------------------------------- generate
    @torch.no_grad()
    def generate(self, input_: SequenceSample, tokenizer=None,
                 gconfig: Optional[GenerationHyperparameters] = None,
                 n_mbs: Optional[int] = None, return_prompt_logprobs=False,
                 **kw):
        """Token-level pipelined generation: prefill passes through all
        stages; each decode step is a short pipeline pass, with the last
        stage sampling and sending next tokens to the first stage
        (reference: GenerateSchedule + Send/RecvNextTokens).
        Correctness-first: one microbatch (no token-interleaving yet)."""
        from realhf_amd.models import generation as genmod
        from realhf_amd.parallel import mappings

        self.model.eval()
        g = self._grid()
        gconfig = gconfig or GenerationHyperparameters()
        cfg = self.model.config
        dev = self.model.flat_param.device
        dtype = self.model.dtype
        eos = getattr(tokenizer, "eos_token_id", None) if tokenizer else None
        pad = getattr(tokenizer, "pad_token_id", 0) if tokenizer else 0
        if pad is None:
            pad = 0

        key = "packed_prompts" if "packed_prompts" in input_.keys else "packed_input_ids"
        prompts, cu, mx = sample_to_packed(input_, key)
        bs = cu.shape[0] - 1
        prompt_lens = (cu[1:] - cu[:-1]).to(dev)
        max_new = gconfig.max_new_tokens
        cache_len = int(prompt_lens.max()) + max_new
        nkv_local = max(cfg.n_kv_heads // self.model.tp_size, 1)
        n_blocks = sum(
            1 for i in self.model.layer_indices if 1 <= i <= cfg.n_layers
        )
        kv_caches = [
            (
                torch.zeros(bs, cache_len, nkv_local, cfg.head_dim, dtype=dtype, device=dev),
                torch.zeros(bs, cache_len, nkv_local, cfg.head_dim, dtype=dtype, device=dev),
            )
            for _ in range(n_blocks)
        ]
        gen = torch.Generator(device=dev)
        gen.manual_seed(seeding.base_seed() + 97 * g.dp_rank)

        # ---- prefill --------------------------------------------------
        hidden = None
        total = prompts.shape[0]
        if not self.model.is_first_stage:
            hidden = _recv((total, cfg.hidden_dim), dtype, g.pp_prev_global_rank(), dev)
        out = self.model(
            packed_input_ids=prompts if self.model.is_first_stage else None,
            cu_seqlens=cu, max_seqlen=mx, hidden_states=hidden,
            kv_caches=kv_caches,
        )
        prompt_logprobs = None
        if self.model.is_last_stage:
            if return_prompt_logprobs:
                from realhf_amd.parallel.tp import packed_shifted_logprobs

                prompt_logprobs = packed_shifted_logprobs(out, cu, prompts)
            last = out[(cu[1:].long() - 1)]
            cur_logits = mappings.gather_from_tp_region(last).float()
        else:
            _send(out, g.pp_next_global_rank())

        gen_tokens = torch.full((bs, max_new), pad, dtype=torch.long, device=dev)
        gen_logprobs = torch.zeros(bs, max_new, dtype=torch.float32, device=dev)
        done = torch.zeros(bs, dtype=torch.bool, device=dev)
        gen_lengths = torch.zeros(bs, dtype=torch.long, device=dev)
        cache_seqlens = prompt_lens.to(torch.int32).clone()
        first_rank = g.global_rank_of(0, g.dp_rank, g.tp_rank)
        last_rank = g.global_rank_of(g.pp_size - 1, g.dp_rank, g.tp_rank)

        tokens = None
        for t in range(max_new):
            # sampling on last stage; tokens broadcast last -> first
            if self.model.is_last_stage:
                if eos is not None and t < gconfig.min_new_tokens:
                    cur_logits[:, eos] = float("-inf")
                tokens, logp = genmod._sample_from_logits(cur_logits, gconfig, gen)
                tokens = torch.where(done, torch.full_like(tokens, pad), tokens)
                gen_tokens[:, t] = tokens
                gen_logprobs[:, t] = torch.where(done, torch.zeros_like(logp), logp)
                gen_lengths += (~done).long()
                if eos is not None:
                    done = done | (tokens == eos)
                if not self.model.is_first_stage and t < max_new - 1:
                    _send(tokens, first_rank)
                    _send(done.to(torch.uint8), first_rank)
            if t == max_new - 1:
                break
            if self.model.is_first_stage and not self.model.is_last_stage:
                tokens = _recv((bs,), torch.long, last_rank, dev)
                done_u8 = _recv((bs,), torch.uint8, last_rank, dev)
                if bool(done_u8.all()):
                    # termination: propagate via the normal flow below
                    pass
            cache_seqlens += 1
            hidden = None
            if not self.model.is_first_stage:
                hidden = _recv((bs, cfg.hidden_dim), dtype, g.pp_prev_global_rank(), dev)
            out = self.model(
                packed_input_ids=tokens if self.model.is_first_stage else None,
                hidden_states=hidden, kv_caches=kv_caches,
                cache_seqlens=cache_seqlens, decode=True,
            )
            if not self.model.is_last_stage:
                _send(out, g.pp_next_global_rank())
            else:
                cur_logits = mappings.gather_from_tp_region(out).float()

        _flush_sends()
        if not self.model.is_last_stage:
            return None
        max_len = int(gen_lengths.max())
        return [(
            genmod.GenerationOutput(
                gen_tokens=gen_tokens[:, :max_len],
                gen_logprobs=gen_logprobs[:, :max_len],
                gen_lengths=gen_lengths,
                no_eos_mask=~done,
                prompt_logprobs=prompt_logprobs,
            ),
            prompts,
            cu,
        )]
