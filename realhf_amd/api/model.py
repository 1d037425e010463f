"""Model/backend/interface contracts + transformer config.

Reference semantics: realhf/api/core/model_api.py (GenerationHyperparameters:24,
ReaLModelConfig:143, ReaLMoEConfig:97, FinetuneSpec:288, PipelinableEngine:305,
Model:464, ModelBackend:513, ModelInterface:564, registries:635-738).
"""
import abc
import dataclasses
from typing import Any, Callable, Dict, List, Optional

import torch

from realhf_amd.api.config import ModelName
from realhf_amd.api.data import SequenceSample


@dataclasses.dataclass
class GenerationHyperparameters:
    max_new_tokens: int = 256
    min_new_tokens: int = 0
    greedy: bool = False
    top_k: int = 200
    top_p: float = 1.0
    temperature: float = 1.0
    use_hip_graph: bool = True  # hipGraph-captured decode step
    force_no_logits_mask: bool = True


@dataclasses.dataclass
class MoEConfig:
    num_experts: int = 8
    top_k: int = 2
    norm_topk_prob: bool = True  # renormalize top-k routing weights (mixtral)
    routing_type: str = "aux_loss"  # aux_loss | sinkhorn | none
    aux_loss_coef: float = 1e-2
    z_loss_coef: float = 0.0
    input_jitter_eps: Optional[float] = None
    capacity_factor: Optional[float] = None
    token_drop_policy: str = "probs"  # "probs" | "position" (model_api.py:120)
    pad_to_capacity: bool = False  # static [n_experts*cap] dispatch shapes
    use_grouped_gemm: bool = True
    expert_parallel_size: int = 1  # EP over xGMI (absent in reference)


@dataclasses.dataclass
class ReaLModelConfig:
    """Architecture config for the flat-parameter transformer."""

    n_layers: int
    hidden_dim: int
    n_heads: int
    head_dim: int
    intermediate_dim: int
    vocab_size: int
    n_kv_heads: Optional[int] = None
    max_position_embeddings: int = 4096
    activation: str = "silu"  # silu (SwiGLU) | gelu (GPT MLP)
    norm_type: str = "rms"  # rms | layer | gemma_rms
    layer_norm_epsilon: float = 1e-5
    use_attention_bias: bool = False
    use_mlp_bias: bool = False
    use_attn_proj_bias: bool = False
    # position embedding
    apply_rotary: bool = True
    rotary_base: float = 10000.0
    rotary_interleaved: bool = False
    rotary_scaling: Optional[float] = None
    rotary_scaling_type: Optional[str] = None
    abs_position_embedding_offset: int = 0  # gpt2-style learned positions if >0 or use_abs
    use_abs_position_embedding: bool = False
    # head
    is_critic: bool = False
    tied_embedding: bool = False
    # norm details
    scale_attn_by_inverse_layer_idx: bool = False
    qk_layernorm: bool = False
    # gemma-style sqrt(hidden) embedding scaling
    embedding_multiplier: Optional[float] = None
    # mistral-style sliding-window attention: each token attends to the
    # previous `sliding_window` tokens (inclusive of itself); None = full
    sliding_window: Optional[int] = None
    # dropout (0 for RLHF)
    attn_pdrop: float = 0.0
    resid_pdrop: float = 0.0
    embd_pdrop: float = 0.0
    # dtype
    dtype: str = "bfloat16"
    # MoE
    moe: Optional[MoEConfig] = None
    # bookkeeping
    base_model_path: Optional[str] = None
    family: Optional[str] = None

    def __post_init__(self):
        if self.n_kv_heads is None:
            self.n_kv_heads = self.n_heads
        assert self.n_heads % self.n_kv_heads == 0 or self.n_kv_heads % self.n_heads == 0

    @property
    def torch_dtype(self):
        return getattr(torch, self.dtype)

    def param_count(self) -> int:
        h, i, v = self.hidden_dim, self.intermediate_dim, self.vocab_size
        qkv = h * (self.n_heads + 2 * self.n_kv_heads) * self.head_dim
        attn = qkv + self.n_heads * self.head_dim * h
        if self.activation == "silu":
            mlp = 3 * h * i
        else:
            mlp = 2 * h * i
        if self.moe is not None:
            mlp = mlp * self.moe.num_experts + h * self.moe.num_experts
        per_layer = attn + mlp + 2 * h
        emb = v * h
        head = h if self.is_critic else (0 if self.tied_embedding else v * h)
        return emb + self.n_layers * per_layer + head + h


@dataclasses.dataclass
class ModelVersion:
    epoch: int = 0
    epoch_step: int = 0
    global_step: int = 0


@dataclasses.dataclass
class FinetuneSpec:
    total_train_epochs: int
    dataset_size: int
    train_batch_size: int

    @property
    def total_train_steps(self):
        steps_per_epoch = max(1, self.dataset_size // self.train_batch_size)
        return self.total_train_epochs * steps_per_epoch

    @property
    def steps_per_epoch(self):
        return max(1, self.dataset_size // self.train_batch_size)


class PipelinableEngine(abc.ABC):
    """What every backend produces: an engine that can run packed batches
    through the (possibly 3D-parallel) model (reference: model_api.py:305)."""

    @abc.abstractmethod
    def train_batch(
        self,
        input_: SequenceSample,
        loss_fn: Callable,
        version_steps: int,
        n_mbs: Optional[int] = None,
    ):
        ...

    @abc.abstractmethod
    def forward(
        self,
        input_: SequenceSample,
        n_mbs: Optional[int] = None,
        post_hook: Optional[Callable] = None,
        aggregate_fn: Callable = None,
    ):
        ...

    @abc.abstractmethod
    def generate(
        self,
        input_: SequenceSample,
        tokenizer=None,
        gconfig: Optional[GenerationHyperparameters] = None,
        n_mbs: Optional[int] = None,
    ):
        ...

    def eval_batch(self, input_: SequenceSample, loss_fn: Callable, n_mbs=None):
        raise NotImplementedError()


@dataclasses.dataclass
class Model:
    """A named model shard living on this process: the module + tokenizer +
    version counters (reference: model_api.py:464)."""

    name: ModelName
    module: Any  # ReaLModel or a PipelinableEngine wrapping it
    tokenizer: Any
    device: torch.device
    dtype: Optional[torch.dtype] = None
    version: ModelVersion = dataclasses.field(default_factory=ModelVersion)
    ft_spec: Optional[FinetuneSpec] = None
    backend_name: Optional[str] = None

    def __post_init__(self):
        try:
            self.module = self.module.to(self.device)
        except (AttributeError, RuntimeError, ValueError):
            pass

    def inc_version(self):
        self.version.global_step += 1
        self.version.epoch_step += 1


class ModelBackend(abc.ABC):
    """Wraps a Model's module into a PipelinableEngine (adds optimizer /
    parallel execution; reference: model_api.py:513)."""

    @abc.abstractmethod
    def _initialize(self, model: Model, spec: FinetuneSpec) -> Model:
        ...

    def initialize(self, model: Model, spec: FinetuneSpec) -> Model:
        model.ft_spec = spec
        return self._initialize(model, spec)

    def destroy(self, model: Model):
        pass


class ModelInterface(abc.ABC):
    """Algorithm operations on a model (reference: model_api.py:564).
    Subclasses implement some of generate/inference/train_step/evaluate."""

    def save(self, model: Model, save_dir: str):
        pass

    def evaluate(self, model: Model, eval_dataloader) -> Dict:
        return {}

    def inference(
        self, model: Model, input_: SequenceSample, n_mbs=None
    ) -> Optional[SequenceSample]:
        raise NotImplementedError()

    def generate(
        self, model: Model, input_: SequenceSample, n_mbs=None
    ) -> Optional[SequenceSample]:
        raise NotImplementedError()

    def train_step(self, model: Model, input_: SequenceSample, n_mbs=None) -> Dict:
        raise NotImplementedError()

    def mock(self, interface_type, model: Model, input_: SequenceSample):
        """Produce a fake output with correct metadata for profiling."""
        raise NotImplementedError()


# ---------------------------------------------------------------------------
# registries
# ---------------------------------------------------------------------------
_INTERFACES: Dict[str, Callable] = {}
_BACKENDS: Dict[str, Callable] = {}


def register_interface(name: str, cls: Callable):
    assert name not in _INTERFACES, name
    _INTERFACES[name] = cls


def make_interface(cfg) -> ModelInterface:
    from realhf_amd.api.config import Abstraction

    if isinstance(cfg, str):
        cfg = Abstraction(type_=cfg)
    return _INTERFACES[cfg.type_](**cfg.args)


def register_backend(name: str, cls: Callable):
    assert name not in _BACKENDS, name
    _BACKENDS[name] = cls


def make_backend(cfg) -> ModelBackend:
    from realhf_amd.api.config import Abstraction

    if isinstance(cfg, str):
        cfg = Abstraction(type_=cfg)
    return _BACKENDS[cfg.type_](**cfg.args)


# HF model family registry: family name -> converter hooks
_HF_FAMILIES: Dict[str, Any] = {}


def register_hf_family(name: str, registry):
    _HF_FAMILIES[name] = registry


def get_hf_family(name: str):
    return _HF_FAMILIES[name]


def hf_families() -> List[str]:
    return sorted(_HF_FAMILIES.keys())
