"""Dataflow graph of model function calls (MFCs).

Reference semantics: realhf/api/core/dfg.py (MFCDef:52, ParamReallocHook:25,
OffloadHook:20, build_graph:239).  An RLHF algorithm is a DAG of MFCs;
edges are resolved from input/output data keys.  E.g. PPO:

    actor_gen → {rew_inf, ref_inf, critic_inf} → {actor_train, critic_train}

Implemented without networkx (plain dict adjacency + toposort).
"""
import dataclasses
from typing import Any, Dict, List, Optional, Tuple

from realhf_amd.api.config import (
    Abstraction,
    ModelFamily,
    ModelInterfaceType,
    ModelName,
)


@dataclasses.dataclass
class OffloadHook:
    """Release model weights after the MFC: to pinned host memory
    (to="cpu", PCIe reload) or ZeRO-3-style sharded across the DP group
    (to="dp_shard", one xGMI all-gather to restore — faster and
    host-RAM-free; the 288 GB-native choice for frozen 70B ref/RM)."""

    to: str = "cpu"


@dataclasses.dataclass
class ParamReallocHook:
    """Reshard weights from/to another replica of the same role before or
    after the MFC.  eta < 1 EMA-merges into the destination (ref-EMA
    trick, reference examples/ppo_ref_ema.py)."""

    source: Optional[ModelName] = None
    target: Optional[ModelName] = None
    eta: float = 1.0


@dataclasses.dataclass
class MFCDef:
    """One model function call in the dataflow graph.

    `input_keys` are data keys consumed (from the dataset or produced by
    other MFCs); `output_keys` are produced.  `n_seqs` is the global batch
    size in sequences this MFC consumes per graph iteration.
    """

    name: str
    model_name: ModelName
    interface_type: ModelInterfaceType
    interface_impl: Abstraction
    model_type: Optional[ModelFamily] = None
    input_keys: Tuple[str, ...] = ()
    output_keys: Tuple[str, ...] = ()
    input_key_remap: Dict[str, str] = dataclasses.field(default_factory=dict)
    output_key_remap: Dict[str, str] = dataclasses.field(default_factory=dict)
    n_seqs: int = 1
    n_mbs: Optional[int] = None
    balanced_dp: bool = False
    log_return_value: bool = False
    # scheduling hint: lower runs earlier among ready MFCs — used to place
    # whole-mesh MFCs before disjoint-mesh concurrent groups so sub-mesh
    # ranks reach their own MFC without waiting (executor program order)
    priority: int = 0
    # hooks attached by the experiment planner
    pre_hooks: List[Any] = dataclasses.field(default_factory=list)
    post_hooks: List[Any] = dataclasses.field(default_factory=list)
    # filled by build_graph
    _parents: List["MFCDef"] = dataclasses.field(default_factory=list, repr=False)
    _children: List["MFCDef"] = dataclasses.field(default_factory=list, repr=False)

    @property
    def role(self):
        return self.model_name.role

    @property
    def parents(self) -> List["MFCDef"]:
        return self._parents

    @property
    def children(self) -> List["MFCDef"]:
        return self._children

    @property
    def is_src(self):
        return not self._parents

    @property
    def is_dst(self):
        return not self._children

    def __hash__(self):
        return hash(self.name)


@dataclasses.dataclass
class DFG:
    mfcs: List[MFCDef]
    data_producers: Dict[str, MFCDef]  # key -> producing MFC
    data_consumers: Dict[str, List[MFCDef]]

    def topological_order(self) -> List[MFCDef]:
        indeg = {m.name: 0 for m in self.mfcs}
        for m in self.mfcs:
            for c in m.children:
                indeg[c.name] += 1
        from collections import deque

        q = deque(sorted([m for m in self.mfcs if indeg[m.name] == 0],
                         key=lambda x: (x.priority, x.name)))
        out = []
        by_name = {m.name: m for m in self.mfcs}
        while q:
            m = q.popleft()
            out.append(m)
            for c in sorted(m.children, key=lambda x: (x.priority, x.name)):
                indeg[c.name] -= 1
                if indeg[c.name] == 0:
                    q.append(by_name[c.name])
        if len(out) != len(self.mfcs):
            raise ValueError("MFC graph has a cycle")
        return out

    @property
    def roles(self):
        return sorted({m.role for m in self.mfcs})

    def mfcs_of_role(self, role: str) -> List[MFCDef]:
        return [m for m in self.mfcs if m.role == role]

    def find(self, name: str) -> MFCDef:
        for m in self.mfcs:
            if m.name == name:
                return m
        raise KeyError(name)


def build_graph(mfcs: List[MFCDef], verbose: bool = False) -> DFG:
    """Resolve parent/child edges from data keys (reference: dfg.py:239).

    A key produced by no MFC must come from the dataset.  Each key has at
    most one producer.
    """
    names = [m.name for m in mfcs]
    assert len(set(names)) == len(names), f"duplicate MFC names: {names}"
    producers: Dict[str, MFCDef] = {}
    consumers: Dict[str, List[MFCDef]] = {}
    for m in mfcs:
        m._parents, m._children = [], []
        for k in m.output_keys:
            if k in producers:
                raise ValueError(
                    f"key {k} produced by both {producers[k].name} and {m.name}"
                )
            producers[k] = m
    for m in mfcs:
        for k in m.input_keys:
            consumers.setdefault(k, []).append(m)
            p = producers.get(k)
            if p is not None and p is not m:
                if p not in m._parents:
                    m._parents.append(p)
                if m not in p._children:
                    p._children.append(m)
    g = DFG(mfcs=mfcs, data_producers=producers, data_consumers=consumers)
    g.topological_order()  # raises on cycles
    return g
