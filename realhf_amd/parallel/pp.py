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


class _PPDecodeGraph:
    """Per-(stage, microbatch) hipGraph of the decode forward inside the
    pipeline (reference: per-microbatch CUDA graphs in PipeGenInstrSet,
    pipe_runner.py:422-475).  Static buffers: token ids (first stage) or
    received hidden states (later stages) + cache_seqlens; the KV caches
    are already static.  The p2p recv writes into the static input
    buffer, then one graph launch replaces ~32 kernel launches/layer."""

    def __init__(self, model, kv_caches, bs, is_first, device, hidden_dim,
                 dtype):
        self.model = model
        self.kv = kv_caches
        self.is_first = is_first
        self.graph = None
        self.failed = False
        self.in_tokens = torch.zeros(bs, dtype=torch.long, device=device)
        self.in_hidden = (None if is_first else
                          torch.zeros(bs, hidden_dim, dtype=dtype,
                                      device=device))
        self.in_cache_seqlens = torch.zeros(bs, dtype=torch.int32,
                                            device=device)
        self.out = None

    def _eager(self):
        return self.model(
            packed_input_ids=self.in_tokens if self.is_first else None,
            hidden_states=self.in_hidden,
            kv_caches=self.kv,
            cache_seqlens=self.in_cache_seqlens,
            decode=True,
        )

    def _capture(self):
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._eager()
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.out = self._eager()

    def step(self, tokens, hidden, cache_seqlens):
        if self.is_first:
            self.in_tokens.copy_(tokens)
        else:
            self.in_hidden.copy_(hidden)
        self.in_cache_seqlens.copy_(cache_seqlens)
        if self.graph is None and not self.failed:
            try:
                self._capture()
            except Exception as e:
                logger.warning("PP decode hipGraph capture failed (%s); "
                               "eager decode", e)
                self.failed = True
                torch.cuda.synchronize()
        if self.graph is None:
            return self._eager()
        self.graph.replay()
        return self.out


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
                    version_steps: int = 0, n_mbs: Optional[int] = None,
                    defer_allgather: bool = False) -> Dict:
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
            if i == n_mbs - 1 and hasattr(self.optimizer, "arm_overlap"):
                # ZeRO-1 overlap: grads are complete after the LAST
                # microbatch's backward — bucket reduce-scatters overlap it
                self.optimizer.arm_overlap()
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
            if hasattr(self.optimizer, "end_microbatch"):
                # ZeRO-2: this stage's residual grad buckets reduce-scatter
                # now, overlapped with the next microbatch's fwd/bwd
                self.optimizer.end_microbatch()

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
        opt_stats = self.optimizer.step(defer_allgather=defer_allgather)
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
        """Token-level PIPELINED generation (reference: GenerateSchedule
        static_schedule.py:195-306 + pipe_runner.py:179-256, 422-475).

        n_mbs microbatches (default = pp stages) are interleaved
        round-robin: while microbatch i's token is being sampled on the
        last stage, microbatch i+1 occupies the earlier stages, so the
        per-token pipeline bubble shrinks from (S-1)/S to ~0 once
        n_mbs >= S.  After sampling, the last stage broadcasts
        (tokens, done) over the PP group — one tiny collective that both
        feeds the first stage's next forward AND gives every stage the
        SAME termination view, replacing the reference's burn-out
        signal protocol: a microbatch whose sequences have all hit EOS
        (checked every 8 tokens to amortize the host sync) is retired by
        all stages at the same round, so short batches exit early.  Each
        (stage, microbatch) decode forward is hipGraph-captured with
        static recv/token buffers (_PPDecodeGraph)."""
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
        max_new = gconfig.max_new_tokens
        S = g.pp_size
        first, last = self.model.is_first_stage, self.model.is_last_stage
        pp_group = g.pp_group()

        key = "packed_prompts" if "packed_prompts" in input_.keys else "packed_input_ids"
        M = n_mbs or S
        M = max(1, min(M, input_.bs))
        mbs = input_.split(M) if M > 1 else [input_]
        M = len(mbs)
        gen = torch.Generator(device=dev)
        gen.manual_seed(seeding.base_seed() + 97 * g.dp_rank)
        nkv_local = max(cfg.n_kv_heads // self.model.tp_size, 1)
        n_blocks = sum(1 for i in self.model.layer_indices
                       if 1 <= i <= cfg.n_layers)

        class _MB:
            pass

        st: List[_MB] = []
        for mb in mbs:
            s = _MB()
            s.prompts, s.cu, s.mx = sample_to_packed(mb, key)
            s.bs = s.cu.shape[0] - 1
            s.prompt_lens = (s.cu[1:] - s.cu[:-1]).to(dev)
            cache_len = int(s.prompt_lens.max()) + max_new
            s.kv = [
                (torch.zeros(s.bs, cache_len, nkv_local, cfg.head_dim,
                             dtype=dtype, device=dev),
                 torch.zeros(s.bs, cache_len, nkv_local, cfg.head_dim,
                             dtype=dtype, device=dev))
                for _ in range(n_blocks)
            ]
            s.cache_seqlens = s.prompt_lens.to(torch.int32).clone()
            s.r = 0  # pipeline round: r produces token r (0 = prefill)
            s.active = True
            s.prompt_logprobs = None
            s.graph = None
            # broadcast payload: [tokens(int64) | done(int64)] as one row pair
            s.bc = torch.zeros(2, s.bs, dtype=torch.long, device=dev)
            s.done = torch.zeros(s.bs, dtype=torch.bool, device=dev)
            if last:
                s.gen_tokens = torch.full((s.bs, max_new), pad,
                                          dtype=torch.long, device=dev)
                s.gen_logprobs = torch.zeros(s.bs, max_new,
                                             dtype=torch.float32, device=dev)
                s.gen_lengths = torch.zeros(s.bs, dtype=torch.long, device=dev)
                s.step_masks = [] if (not gconfig.force_no_logits_mask
                                      and not gconfig.greedy) else None
            st.append(s)

        use_graph = (
            gconfig.use_hip_graph and dev.type == "cuda"
            and dtype == torch.bfloat16 and cfg.head_dim in (64, 128)
            and cfg.moe is None  # MoE counts.cpu() deadlocks under capture
        )

        last_rank = g.global_rank_of(S - 1, g.dp_rank, g.tp_rank)

        def sample_and_bcast(s, cur_logits):
            """LAST STAGE ONLY: sample token r, update bookkeeping,
            broadcast (tokens, done) to the pipeline.  Non-last stages
            receive this broadcast at the START of the microbatch's NEXT
            slot (not here) — otherwise an early stage would block on the
            sample of mb i before computing mb i+1 and the pipeline would
            serialize."""
            r = s.r
            if eos is not None and r < gconfig.min_new_tokens:
                cur_logits[:, eos] = float("-inf")
            tokens, logp, smask = genmod._sample_from_logits(
                cur_logits, gconfig, gen,
                return_mask=s.step_masks is not None)
            if s.step_masks is not None:
                s.step_masks.append(smask)
            tokens = torch.where(s.done, torch.full_like(tokens, pad), tokens)
            s.gen_tokens[:, r] = tokens
            s.gen_logprobs[:, r] = torch.where(
                s.done, torch.zeros_like(logp), logp)
            s.gen_lengths += (~s.done).long()
            if eos is not None:
                s.done = s.done | (tokens == eos)
            s.bc[0] = tokens
            s.bc[1] = s.done.long() if eos is not None else 0
            if S > 1:
                dist.broadcast(s.bc, src=last_rank, group=pp_group)

        def should_retire(s) -> bool:
            """Identical decision on every stage at slot r from the
            (mb, r-1) broadcast; host sync amortized over 8 tokens."""
            if s.r >= max_new:
                return True
            return (eos is not None and (s.r & 7) == 0
                    and bool(s.bc[1].all()))

        # ---- prefill (fill-drain over microbatches, round r = 0) ----------
        for s in st:
            hidden = None
            if not first:
                hidden = _recv((s.prompts.shape[0], cfg.hidden_dim), dtype,
                               g.pp_prev_global_rank(), dev)
            out = self.model(
                packed_input_ids=s.prompts if first else None,
                cu_seqlens=s.cu, max_seqlen=s.mx, hidden_states=hidden,
                kv_caches=s.kv,
            )
            if last:
                if return_prompt_logprobs:
                    from realhf_amd.parallel.tp import packed_shifted_logprobs

                    s.prompt_logprobs = packed_shifted_logprobs(
                        out, s.cu, s.prompts)
                cur_logits = mappings.gather_from_tp_region(
                    out[(s.cu[1:].long() - 1)]).float()
                sample_and_bcast(s, cur_logits)  # token 0
            else:
                _send(out, g.pp_next_global_rank())
            s.r = 1

        # ---- interleaved decode rounds (r >= 1) ---------------------------
        # All stages walk microbatches in the same (r-major, mb-minor)
        # order, so every pp-group collective matches by program order.
        self._gen_decode_slots = 0
        self._gen_trace = []  # per-rank event log: ("bc"|"fwd", mb, r)
        while any(s.active for s in st):
            for i, s in enumerate(st):
                if not s.active:
                    continue
                if S > 1 and not last:
                    # receive (tokens, done) of round r-1 sampled on the
                    # last stage while this stage was busy with other mbs
                    dist.broadcast(s.bc, src=last_rank, group=pp_group)
                    self._gen_trace.append(("bc", i, s.r - 1))
                if should_retire(s):
                    s.active = False
                    continue
                self._gen_decode_slots += 1
                self._gen_trace.append(("fwd", i, s.r))
                s.cache_seqlens += 1
                tokens = s.bc[0]
                if s.graph is None and use_graph:
                    s.graph = _PPDecodeGraph(self.model, s.kv, s.bs, first, dev,
                                             cfg.hidden_dim, dtype)
                if not first:
                    hidden = _recv((s.bs, cfg.hidden_dim), dtype,
                                   g.pp_prev_global_rank(), dev)
                else:
                    hidden = None
                if s.graph is not None:
                    out = s.graph.step(tokens, hidden, s.cache_seqlens)
                else:
                    out = self.model(
                        packed_input_ids=tokens if first else None,
                        hidden_states=hidden, kv_caches=s.kv,
                        cache_seqlens=s.cache_seqlens, decode=True,
                    )
                if last:
                    cur_logits = mappings.gather_from_tp_region(out).float()
                    sample_and_bcast(s, cur_logits)
                else:
                    _send(out, g.pp_next_global_rank())
                s.r += 1

        _flush_sends()
        if not last:
            return None
        outs = []
        for s, mb in zip(st, mbs):
            max_len = int(s.gen_lengths.max())
            outs.append((
                genmod.GenerationOutput(
                    gen_tokens=s.gen_tokens[:, :max_len],
                    gen_logprobs=s.gen_logprobs[:, :max_len],
                    gen_lengths=s.gen_lengths,
                    no_eos_mask=~s.done,
                    prompt_logprobs=s.prompt_logprobs,
                    logits_mask=(
                        torch.stack(s.step_masks[:max_len], dim=1)
                        if getattr(s, "step_masks", None) else None),
                ),
                s.prompts,
                s.cu,
            ))
        return outs
