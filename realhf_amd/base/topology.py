"""Process topology & 3D-parallel grid.

Reference semantics: realhf/base/topology.py (ProcessTopology:65,
PipeModelDataParallelTopology:307, ParallelGrid:328, FakeGrid:593,
new_or_get_group:17).  Axis names follow the reference: "pipe", "data",
"tensor" (the reference calls the tensor axis "model").

MI355X-native notes: all groups are torch.distributed process groups over
RCCL (backend "nccl" on ROCm) or gloo for CPU tests.  Groups are cached
globally so that re-creating a grid for the same rank set reuses the
communicator (RCCL communicators are expensive and xGMI route setup is
per-communicator).
"""
import dataclasses
import itertools
from typing import Dict, List, Optional, Sequence, Tuple

import torch.distributed as dist

_GROUP_CACHE: Dict[Tuple[int, ...], object] = {}


def new_or_get_group(ranks: Sequence[int], backend: Optional[str] = None):
    """Create (or fetch cached) a process group for `ranks`.

    MUST be called by ALL ranks in the default group with identical
    arguments (torch.distributed requirement).
    """
    key = tuple(sorted(ranks)) + (backend or "",)
    if key not in _GROUP_CACHE:
        _GROUP_CACHE[key] = dist.new_group(sorted(ranks), backend=backend)
    return _GROUP_CACHE[key]


def clear_group_cache():
    _GROUP_CACHE.clear()


@dataclasses.dataclass(frozen=True)
class Coord:
    pipe: int
    data: int
    tensor: int

    def to_dict(self):
        return dataclasses.asdict(self)


class ProcessTopology:
    """Maps an N-D cartesian coordinate over named axes to a flat rank.

    Axis order determines rank layout: the LAST axis is fastest-varying.
    """

    def __init__(self, axes: List[str], dims: List[int]):
        assert len(axes) == len(dims)
        self.axes = list(axes)
        self.dims = list(dims)
        self._strides = [1] * len(dims)
        for i in range(len(dims) - 2, -1, -1):
            self._strides[i] = self._strides[i + 1] * dims[i + 1]

    def world_size(self) -> int:
        out = 1
        for d in self.dims:
            out *= d
        return out

    def get_dim(self, axis: str) -> int:
        return self.dims[self.axes.index(axis)]

    def get_rank(self, **coord) -> int:
        rank = 0
        for ax, st, d in zip(self.axes, self._strides, self.dims):
            c = coord[ax]
            assert 0 <= c < d, (ax, c, d)
            rank += c * st
        return rank

    def get_coord(self, rank: int):
        vals = {}
        for ax, st, d in zip(self.axes, self._strides, self.dims):
            vals[ax] = (rank // st) % d
        if set(self.axes) == {"pipe", "data", "tensor"}:
            return Coord(**vals)
        return vals

    def filter_match(self, **kw) -> List[int]:
        """All ranks whose coordinate matches the given axis values."""
        out = []
        for rank in range(self.world_size()):
            c = self.get_coord(rank)
            cd = c.to_dict() if isinstance(c, Coord) else c
            if all(cd[k] == v for k, v in kw.items()):
                out.append(rank)
        return out

    def get_axis_list(self, axis: str, axis_value: int) -> List[int]:
        return self.filter_match(**{axis: axis_value})

    def __repr__(self):
        return f"ProcessTopology({list(zip(self.axes, self.dims))})"


class PipeDataTensorTopology(ProcessTopology):
    """The canonical 3D topology: axes (pipe, data, tensor).

    Rank layout: tensor fastest (adjacent ranks = same TP group →
    TP collectives run on directly-connected xGMI peers), then data,
    then pipe.  Extra flags mirror the reference's topology metadata
    (sequence_parallel, gradient_checkpointing).
    """

    def __init__(
        self,
        num_pp: int,
        num_dp: int,
        num_tp: int,
        sequence_parallel: bool = False,
        gradient_checkpointing: bool = False,
        max_prompt_len: Optional[int] = None,
        ep_size: int = 1,
    ):
        super().__init__(["pipe", "data", "tensor"], [num_pp, num_dp, num_tp])
        self.sequence_parallel = sequence_parallel
        self.gradient_checkpointing = gradient_checkpointing
        self.max_prompt_len = max_prompt_len
        assert num_dp % ep_size == 0, (num_dp, ep_size)
        self.ep_size = ep_size

    @property
    def pp(self):
        return self.get_dim("pipe")

    @property
    def dp(self):
        return self.get_dim("data")

    @property
    def tp(self):
        return self.get_dim("tensor")


class ParallelGrid:
    """Builds/caches per-model process groups from a topology plus a
    mapping of topology ranks → global (default-PG) ranks.

    Groups built: tp group, dp group, pp group, the model group (all ranks
    of this model), embedding group (first+last pipe stage pairs for tied
    embedding grad sync), and a gloo DP group for CPU-side coordination.
    """

    def __init__(
        self,
        topology: PipeDataTensorTopology,
        rank_mapping: Optional[Dict[int, int]] = None,
        process_groups: bool = True,
    ):
        self.topo = topology
        n = topology.world_size()
        if rank_mapping is None:
            rank_mapping = {i: i for i in range(n)}
        assert len(rank_mapping) == n
        self.rank_mapping = dict(rank_mapping)  # topo rank -> global rank
        self._inv_mapping = {v: k for k, v in self.rank_mapping.items()}

        self._tp_group = None
        self._dp_group = None
        self._pp_group = None
        self._model_group = None
        self._dp_cpu_group = None
        self._embedding_group = None
        self._ep_group = None
        self._my_coord = None

        if process_groups:
            self._build_groups()

    # -- group construction ------------------------------------------------
    def _global_ranks(self, topo_ranks: List[int]) -> List[int]:
        return [self.rank_mapping[r] for r in topo_ranks]

    def _build_groups(self):
        my_rank = dist.get_rank()
        topo = self.topo
        in_model = my_rank in self._inv_mapping
        if in_model:
            self._my_coord = topo.get_coord(self._inv_mapping[my_rank])

        model_ranks = self._global_ranks(list(range(topo.world_size())))
        g = new_or_get_group(model_ranks)
        if in_model:
            self._model_group = g

        # TP groups: one per (pipe, data)
        for p, d in itertools.product(range(topo.pp), range(topo.dp)):
            ranks = self._global_ranks(topo.filter_match(pipe=p, data=d))
            g = new_or_get_group(ranks)
            if my_rank in ranks:
                self._tp_group = g
        # DP groups: one per (pipe, tensor); plus gloo twin
        for p, t in itertools.product(range(topo.pp), range(topo.tp)):
            ranks = self._global_ranks(topo.filter_match(pipe=p, tensor=t))
            g = new_or_get_group(ranks)
            g2 = new_or_get_group(ranks, backend="gloo")
            if my_rank in ranks:
                self._dp_group = g
                self._dp_cpu_group = g2
        # EP groups: blocks of `ep_size` consecutive dp ranks per (pipe, tensor)
        ep = getattr(topo, "ep_size", 1)
        if ep > 1:
            for p, t in itertools.product(range(topo.pp), range(topo.tp)):
                for b in range(topo.dp // ep):
                    ranks = [
                        self.rank_mapping[topo.get_rank(pipe=p, data=d, tensor=t)]
                        for d in range(b * ep, (b + 1) * ep)
                    ]
                    g = new_or_get_group(ranks)
                    if my_rank in ranks:
                        self._ep_group = g
        # PP groups: one per (data, tensor)
        for d, t in itertools.product(range(topo.dp), range(topo.tp)):
            ranks = self._global_ranks(topo.filter_match(data=d, tensor=t))
            g = new_or_get_group(ranks)
            if my_rank in ranks:
                self._pp_group = g
        # Embedding groups (tied embeddings): {first-stage, last-stage} per (d, t)
        if topo.pp > 1:
            for d, t in itertools.product(range(topo.dp), range(topo.tp)):
                r0 = self.rank_mapping[topo.get_rank(pipe=0, data=d, tensor=t)]
                r1 = self.rank_mapping[
                    topo.get_rank(pipe=topo.pp - 1, data=d, tensor=t)
                ]
                ranks = [r0, r1]
                g = new_or_get_group(ranks)
                if my_rank in ranks:
                    self._embedding_group = g

    # -- queries -----------------------------------------------------------
    @property
    def coord(self) -> Coord:
        assert self._my_coord is not None, "this rank is not in the grid"
        return self._my_coord

    def is_in_grid(self) -> bool:
        try:
            return dist.get_rank() in self._inv_mapping
        except (RuntimeError, ValueError):
            return False

    def topo_rank_of(self, global_rank: int) -> int:
        return self._inv_mapping[global_rank]

    def global_rank_of(self, pipe: int, data: int, tensor: int) -> int:
        return self.rank_mapping[
            self.topo.get_rank(pipe=pipe, data=data, tensor=tensor)
        ]

    # group getters
    def tp_group(self):
        return self._tp_group

    def dp_group(self):
        return self._dp_group

    def dp_cpu_group(self):
        return self._dp_cpu_group

    def pp_group(self):
        return self._pp_group

    def model_group(self):
        return self._model_group

    def embedding_group(self):
        return self._embedding_group

    def ep_group(self):
        return self._ep_group

    @property
    def ep_size(self):
        return getattr(self.topo, "ep_size", 1)

    @property
    def ep_rank(self):
        return self.coord.data % self.ep_size if self.ep_size > 1 else 0

    # convenience ranks/sizes
    @property
    def tp_rank(self):
        return self.coord.tensor

    @property
    def tp_size(self):
        return self.topo.tp

    @property
    def dp_rank(self):
        return self.coord.data

    @property
    def dp_size(self):
        return self.topo.dp

    @property
    def pp_rank(self):
        return self.coord.pipe

    @property
    def pp_size(self):
        return self.topo.pp

    def pp_prev_global_rank(self):
        c = self.coord
        return self.global_rank_of((c.pipe - 1) % self.topo.pp, c.data, c.tensor)

    def pp_next_global_rank(self):
        c = self.coord
        return self.global_rank_of((c.pipe + 1) % self.topo.pp, c.data, c.tensor)


class FakeGrid:
    """Grid math without process groups — for CPU tests, single-process
    runs and the allocation planner (reference: topology.py:593)."""

    def __init__(self, rank: int, topo: PipeDataTensorTopology):
        self.topo = topo
        self.rank = rank
        self._coord = topo.get_coord(rank)

    # group getters are no-ops (single process)
    def tp_group(self):
        return None

    def dp_group(self):
        return None

    def dp_cpu_group(self):
        return None

    def pp_group(self):
        return None

    def model_group(self):
        return None

    def embedding_group(self):
        return None

    @property
    def coord(self):
        return self._coord

    @property
    def tp_rank(self):
        return self._coord.tensor

    @property
    def tp_size(self):
        return self.topo.tp

    @property
    def dp_rank(self):
        return self._coord.data

    @property
    def dp_size(self):
        return self.topo.dp

    @property
    def pp_rank(self):
        return self._coord.pipe

    @property
    def pp_size(self):
        return self.topo.pp
