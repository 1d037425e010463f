"""PipelinableEngine implementations (pp == 1 path).

Reference semantics: realhf/impl/model/backend/inference.py
(PipelinableInferenceEngine:21) and megatron.py (ReaLMegatronEngine:702).
The pp > 1 path goes through parallel/pp.py's schedule runner.

Backends registered here:
  "zero1"     — native ZeRO-1 trainer (parallel/ddp.py)
  "inference" — forward/generate only
"""
import dataclasses
from typing import Callable, Dict, List, Optional

import torch

from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import (
    FinetuneSpec,
    GenerationHyperparameters,
    Model,
    ModelBackend,
    PipelinableEngine,
    register_backend,
)
from realhf_amd.base import constants, logging, seeding
from realhf_amd.models import generation as genmod
from realhf_amd.models import moe as moe_mod
from realhf_amd.parallel.ddp import OptimizerConfig, ZeRO1Optimizer

logger = logging.getLogger("engine")


def sample_to_packed(sample: SequenceSample, key: str = "packed_input_ids"):
    """(tensor, cu_seqlens int32, max_seqlen) for a packed key.

    A sample may pack several sequences under one key (e.g. [pos, neg]
    pairs for RW/DPO); each inner seqlen is its own attention sequence."""
    lens = [l for x in sample.seqlens[key] for l in x]
    device = sample.data[key].device
    cu = torch.zeros(len(lens) + 1, dtype=torch.int32, device=device)
    cu[1:] = torch.cumsum(torch.tensor(lens, device=device), 0)
    return sample.data[key], cu, max(lens) if lens else 0


def _default_aggregate(outs: List[torch.Tensor]) -> torch.Tensor:
    return torch.cat(outs, dim=0)


class PipelinableTrainEngine(PipelinableEngine):
    """pp==1 engine: minibatched fwd/bwd + ZeRO-1 optimizer step."""

    def __init__(self, model, optimizer: ZeRO1Optimizer):
        self.model = model
        self.optimizer = optimizer

    @property
    def module(self):
        return self.model

    def train_batch(self, input_: SequenceSample, loss_fn: Callable,
                    version_steps: int = 0, n_mbs: Optional[int] = None,
                    defer_allgather: bool = False) -> Dict:
        self.model.train()
        self.optimizer.zero_grad()  # also waits any deferred all-gather
        mbs = input_.split(n_mbs) if n_mbs and n_mbs > 1 and input_.bs >= n_mbs else [input_]
        stats: Dict[str, float] = {}
        for i, mb in enumerate(mbs):
            ids, cu, mx = sample_to_packed(mb)
            # MoE aux losses are collected only inside this scope — never
            # during inference or gradient-checkpoint recompute in backward.
            with moe_mod.aux_loss_collection():
                out = self.model(packed_input_ids=ids, cu_seqlens=cu, max_seqlen=mx)
                loss, st = loss_fn(out, mb)
                # interfaces that know about MoE pop aux losses themselves
                # inside loss_fn; anything left over is added here so plain
                # interfaces (SFT/DPO/...) still train the router
                for aux in moe_mod.pop_aux_losses():
                    loss = loss + aux
            if i == len(mbs) - 1 and hasattr(self.optimizer, "arm_overlap"):
                # bucketed reduce-scatter overlaps this (last) backward
                self.optimizer.arm_overlap()
            (loss / len(mbs)).backward()
            if hasattr(self.optimizer, "end_microbatch"):
                # ZeRO-2: reduce-scatter this microbatch's residual buckets
                self.optimizer.end_microbatch()
            for k, v in st.items():
                stats[k] = stats.get(k, 0.0) + float(v) / len(mbs)
        opt_stats = self.optimizer.step(defer_allgather=defer_allgather)
        stats.update(opt_stats)
        return stats

    def _finish_ag(self):
        opt = getattr(self, "optimizer", None)
        if opt is not None and hasattr(opt, "finish_allgather"):
            opt.finish_allgather()

    @torch.no_grad()
    def eval_batch(self, input_: SequenceSample, loss_fn: Callable, n_mbs=None):
        self.model.eval()
        self._finish_ag()
        mbs = input_.split(n_mbs) if n_mbs and n_mbs > 1 else [input_]
        stats: Dict[str, float] = {}
        for mb in mbs:
            ids, cu, mx = sample_to_packed(mb)
            out = self.model(packed_input_ids=ids, cu_seqlens=cu, max_seqlen=mx)
            _, st = loss_fn(out, mb)
            for k, v in st.items():
                stats[k] = stats.get(k, 0.0) + float(v) / len(mbs)
        return stats

    @torch.no_grad()
    def forward(self, input_: SequenceSample, n_mbs: Optional[int] = None,
                post_hook: Optional[Callable] = None,
                aggregate_fn: Callable = _default_aggregate):
        self.model.eval()
        self._finish_ag()
        mbs = input_.split(n_mbs) if n_mbs and n_mbs > 1 and input_.bs >= n_mbs else [input_]
        outs = []
        for mb in mbs:
            ids, cu, mx = sample_to_packed(mb)
            out = self.model(packed_input_ids=ids, cu_seqlens=cu, max_seqlen=mx)
            if post_hook is not None:
                out = post_hook(out, mb)
            outs.append(out)
        return aggregate_fn(outs)

    @torch.no_grad()
    def generate(self, input_: SequenceSample, tokenizer=None,
                 gconfig: Optional[GenerationHyperparameters] = None,
                 n_mbs: Optional[int] = None, **gen_kw):
        self.model.eval()
        self._finish_ag()
        gconfig = gconfig or GenerationHyperparameters()
        key = "packed_prompts" if "packed_prompts" in input_.keys else "packed_input_ids"
        mbs = input_.split(n_mbs) if n_mbs and n_mbs > 1 and input_.bs >= n_mbs else [input_]
        outs = []
        eos = getattr(tokenizer, "eos_token_id", None) if tokenizer is not None else None
        pad = getattr(tokenizer, "pad_token_id", None) if tokenizer is not None else None
        if pad is None:
            pad = eos if eos is not None else 0
        gen = torch.Generator(device=self.model.flat_param.device)
        gen.manual_seed(seeding.base_seed() + 97 * (constants.dp_rank() if constants.has_current() else 0))
        for mb in mbs:
            ids, cu, mx = sample_to_packed(mb, key)
            outs.append(
                (
                    genmod.generate(
                        self.model, ids, cu, gconfig,
                        eos_token_id=eos, pad_token_id=pad, generator=gen,
                        **gen_kw,
                    ),
                    ids,
                    cu,
                )
            )
        return outs


class PipelinableInferenceEngine(PipelinableTrainEngine):
    def __init__(self, model):
        self.model = model
        self.optimizer = None

    def train_batch(self, *a, **kw):
        raise RuntimeError("inference engine cannot train")


# ---------------------------------------------------------------------------
@dataclasses.dataclass
class ZeRO1Backend(ModelBackend):
    optimizer: OptimizerConfig = dataclasses.field(default_factory=OptimizerConfig)
    bucket_size: int = 40_000_000

    def _initialize(self, model: Model, spec: FinetuneSpec) -> Model:
        if isinstance(self.optimizer, dict):
            self.optimizer = OptimizerConfig(**self.optimizer)
        target = model.module
        if getattr(model.module, "lora_flat", None) is not None:
            # LoRA: same ZeRO-1 machinery over the small adapter buffer;
            # the base flat_param stays frozen
            from realhf_amd.parallel.ddp import LoRAFacade

            target = LoRAFacade(model.module)
        opt = ZeRO1Optimizer(
            target, self.optimizer,
            total_train_steps=spec.total_train_steps,
            bucket_size=self.bucket_size,
        )
        if model.module.pp_size > 1:
            from realhf_amd.parallel.pp import PipelinedEngine

            model.module = PipelinedEngine(model.module, opt)
        else:
            model.module = PipelinableTrainEngine(model.module, opt)
        model.backend_name = "zero1"
        return model


@dataclasses.dataclass
class InferenceBackend(ModelBackend):
    def _initialize(self, model: Model, spec: FinetuneSpec) -> Model:
        if model.module.pp_size > 1:
            from realhf_amd.parallel.pp import PipelinedEngine

            model.module = PipelinedEngine(model.module, None)
        else:
            model.module = PipelinableInferenceEngine(model.module)
        model.backend_name = "inference"
        return model


register_backend("zero1", ZeRO1Backend)
register_backend("inference", InferenceBackend)
# reference name for "wrap without training machinery" (used by custom
# experiments like ppo_sentiment, system_api "null" backend) — here the
# inference engine IS that minimal wrap
register_backend("null", InferenceBackend)
