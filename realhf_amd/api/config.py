"""Core identity & abstraction configs.

Reference semantics: realhf/api/core/config.py (ModelName:52, ModelFamily:74,
ModelShardID:101, ModelInterfaceType:45, abstractions:120-189).
"""
import dataclasses
import enum
from typing import Any, Dict


@dataclasses.dataclass(frozen=True, order=True)
class ModelName:
    """role + replica id.  Replicas of one role share weights (possibly with
    different parallel strategies, synced by parameter reallocation)."""

    role: str
    replica_id: int = 0

    def __str__(self):
        return f"{self.role}@{self.replica_id}"

    @property
    def name(self):
        return str(self)


@dataclasses.dataclass(frozen=True)
class ModelFamily:
    """Architecture class, e.g. llama / gpt2 / qwen2 / gemma / mistral /
    mixtral, plus the critic flag (value head instead of LM head)."""

    _class: str
    size: int = 0
    is_critic: bool = False

    def __str__(self):
        return f"{self._class}-{self.size}{'-critic' if self.is_critic else ''}"


class ModelInterfaceType(enum.Enum):
    GENERATE = "generate"
    TRAIN_STEP = "train_step"
    EVALUATE = "evaluate"
    INFERENCE = "inference"


@dataclasses.dataclass(frozen=True)
class ModelShardID:
    """Identifies one shard of one model: (model_name, dp, tp, pp) →
    lives on exactly one global rank (reference: config.py:101)."""

    model_name: ModelName
    dp_rank: int
    tp_rank: int
    pp_rank: int
    topo_world_size: int

    def __str__(self):
        return (
            f"{self.model_name}/d{self.dp_rank}t{self.tp_rank}p{self.pp_rank}"
        )


@dataclasses.dataclass
class Abstraction:
    """Registry-style {type_, args} config used for datasets / interfaces /
    backends so experiment configs stay serializable."""

    type_: str
    args: Dict[str, Any] = dataclasses.field(default_factory=dict)


DatasetAbstraction = Abstraction
ModelInterfaceAbstraction = Abstraction
ModelBackendAbstraction = Abstraction


@dataclasses.dataclass
class ParallelismConfig:
    """Per-MFC 3D parallel strategy (reference: api/quickstart/model.py:15)."""

    data_parallel_size: int = 1
    tensor_parallel_size: int = 1
    pipeline_parallel_size: int = 1
    sequence_parallel: bool = False

    @property
    def world_size(self):
        return (
            self.data_parallel_size
            * self.tensor_parallel_size
            * self.pipeline_parallel_size
        )

    def __str__(self):
        return (
            f"d{self.data_parallel_size}t{self.tensor_parallel_size}"
            f"p{self.pipeline_parallel_size}"
            + ("s" if self.sequence_parallel else "")
        )


def parse_parallelism(s: str) -> ParallelismConfig:
    """Parse 'd4t2p1' / 'd4m2p1' style strings (reference allocation_mode)."""
    import re

    m = re.fullmatch(r"d(\d+)[tm](\d+)p(\d+)(s?)", s)
    if not m:
        raise ValueError(f"bad parallelism string {s!r}")
    return ParallelismConfig(
        data_parallel_size=int(m.group(1)),
        tensor_parallel_size=int(m.group(2)),
        pipeline_parallel_size=int(m.group(3)),
        sequence_parallel=bool(m.group(4)),
    )
