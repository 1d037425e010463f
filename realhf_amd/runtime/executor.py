"""SPMD dataflow-graph executor with CONCURRENT mesh-scoped MFCs.

Replaces the reference's master/worker runtime (realhf/system/
master_worker.py + model_worker.py + request_reply_stream.py) for the
single-node MI355X deployment: instead of a ZMQ master coordinating
workers with a req->syn->ack total-order protocol, EVERY rank runs the
same deterministic program over the DFG — per-MFC participation, DP
splits, data merges and reallocation plans are all pure functions of
(config, step), so no control messages exist in the hot path at all.

Concurrency model (reference: one asyncio request+reply coroutine pair
per MFC, master_worker.py:455,602): independent MFCs allocated to
DISJOINT device meshes execute simultaneously in wall-clock because no
collective in the per-MFC path spans more than (producer mesh | consumer
mesh).  A rank simply skips MFCs it does not participate in and proceeds
to its next MFC; the static ownership plan (below) guarantees every
collective is entered by exactly its group's members in a single global
program order, so the schedule is deadlock-free by construction.

Data ownership: after an MFC completes, its output keys are held
(full-batch, DP-merged) by every rank of its mesh.  The `_static_plan`
computed at init simulates one step and records, per MFC, which keys
must be broadcast from a holder to the consumer mesh — the analogue of
the reference's data-transfer plan + send/recv caches
(impl/model/comm/data_transfer.py:123-323), but derived offline.

Per MFC and step:
  1. broadcast missing input keys to this MFC's mesh (plan-driven),
  2. run pre-hooks (parameter realloc in / reload from offload),
  3. participating ranks take their balanced DP shard of the input keys
     and call the interface (generate / inference / train_step),
  4. DP-head outputs are merged across the MESH (device-native gather),
  5. run post-hooks (realloc back / offload).
Train stats are merged once per step at the step boundary (the only
world-scoped communication in the loop).
"""
import dataclasses
import os
import time
from typing import Any, Dict, List, Optional, Set, Tuple

import torch
import torch.distributed as dist

from realhf_amd.api.config import ModelInterfaceType, ModelName
from realhf_amd.api.data import SequenceSample
from realhf_amd.api.dfg import DFG, MFCDef, OffloadHook, ParamReallocHook
from realhf_amd.api.model import Model, ModelInterface
from realhf_amd.base import constants, logging
from realhf_amd.base.topology import (
    ParallelGrid,
    PipeDataTensorTopology,
    new_or_get_group,
)
from realhf_amd.parallel.realloc import (
    ParallelStrategy,
    build_realloc_plan,
    execute_realloc,
)
from realhf_amd.runtime.data_transfer import (
    broadcast_sample,
    dp_shard,
    gather_across_dp,
)

logger = logging.getLogger("executor")


@dataclasses.dataclass
class MFCAllocation:
    strategy: ParallelStrategy
    sequence_parallel: bool = False
    gradient_checkpointing: bool = False
    n_mbs: Optional[int] = None

    @property
    def ranks(self):
        return sorted(r for _, r in self.strategy.rank_map)


@dataclasses.dataclass
class _TransferOp:
    key: str
    src: int  # global rank holding the key
    group_ranks: Tuple[int, ...]  # {src} | missing consumer ranks


@dataclasses.dataclass
class _MFCPlan:
    mesh: Tuple[int, ...]
    transfers: List[_TransferOp]
    scope: str  # constants scope name carrying the executing grid


class DFGExecutor:
    def __init__(
        self,
        graph: DFG,
        allocations: Dict[str, MFCAllocation],  # mfc name -> allocation
        models: Dict[ModelName, Model],  # this rank's model shards
        interfaces: Dict[str, ModelInterface],  # mfc name -> interface
        model_strategies: Dict[ModelName, ParallelStrategy],
    ):
        self.graph = graph
        self.allocations = allocations
        self.models = models
        self.interfaces = interfaces
        self.model_strategies = model_strategies
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        self.world = dist.get_world_size() if dist.is_initialized() else 1
        self._step = 0
        self._topo_order = self.graph.topological_order()
        self._register_mfc_grids()
        self._plan = self._static_plan()
        self._make_groups()
        self._mfc_wall: Dict[str, float] = {}
        self._mfc_flops: Dict[str, float] = {}

    # ----------------------------------------------------- static planning
    def _register_mfc_grids(self):
        """An MFC may execute on a SUB-MESH of its model's mesh with its
        own DP layout (valid whenever both layouts are pure-DP: every
        mesh rank holds full replica weights, so reallocation is FREE —
        the 288 GB HBM3E design point).  Register a grid per such MFC so
        dp_rank/dp_group inside the interface reflect the executing mesh."""
        self._mfc_scopes: Dict[str, str] = {}
        for mfc in self._topo_order:
            alloc = self.allocations[mfc.name]
            mstrat = self.model_strategies[mfc.model_name]
            if alloc.strategy == mstrat:
                self._mfc_scopes[mfc.name] = str(mfc.model_name)
                continue
            s = alloc.strategy
            assert s.tp == mstrat.tp and s.pp == mstrat.pp, (
                f"{mfc.name}: sub-mesh execution requires identical tp/pp "
                f"(got mfc {s.tp}x{s.pp} vs model {mstrat.tp}x{mstrat.pp}); "
                "differing shard layouts need a separate replica + realloc"
            )
            assert set(alloc.ranks) <= {r for _, r in mstrat.rank_map}, (
                f"{mfc.name}: executing mesh must be a subset of the model mesh"
            )
            scope = f"{mfc.model_name}@{mfc.name}"
            topo = PipeDataTensorTopology(
                num_pp=s.pp, num_dp=s.dp, num_tp=s.tp,
                sequence_parallel=alloc.sequence_parallel,
                gradient_checkpointing=alloc.gradient_checkpointing,
                ep_size=s.ep,
            )
            rank_mapping = {
                topo.get_rank(pipe=p, data=d, tensor=t): r
                for (p, d, t), r in s.rank_map
            }
            if dist.is_initialized():
                grid = ParallelGrid(topo, rank_mapping)
            else:
                from realhf_amd.base.topology import FakeGrid

                grid = FakeGrid(0, topo)
            constants.set_grid(scope, grid)
            self._mfc_scopes[mfc.name] = scope

    def _static_plan(self) -> Dict[str, _MFCPlan]:
        """Simulate one step to derive per-MFC input transfers.  Holder
        sets evolve deterministically, so the plan is identical on every
        rank and valid for all steps."""
        world_ranks = frozenset(range(self.world))
        holders: Dict[str, Set[int]] = {}
        plan: Dict[str, _MFCPlan] = {}
        for mfc in self._topo_order:
            alloc = self.allocations[mfc.name]
            mesh = tuple(alloc.ranks)
            ops: List[_TransferOp] = []
            for k in mfc.input_keys:
                h = holders.get(k)
                if h is None:
                    h = set(world_ranks)  # dataset key: replicated batch
                    holders[k] = h
                need = set(mesh) - h
                if need:
                    src = min(h)
                    ops.append(_TransferOp(
                        key=k, src=src,
                        group_ranks=tuple(sorted({src} | need)),
                    ))
                    h |= need
            if mfc.output_keys:
                # run_step only merges outputs of non-train MFCs across
                # the mesh (train/eval return stat dicts) — a train MFC
                # with output keys would leave stale holder state
                assert mfc.interface_type in (
                    ModelInterfaceType.GENERATE, ModelInterfaceType.INFERENCE,
                ), (f"{mfc.name}: {mfc.interface_type} MFCs cannot produce "
                    "data outputs (stats only)")
            for k in mfc.output_keys:
                holders[k] = set(mesh)
            plan[mfc.name] = _MFCPlan(
                mesh=mesh, transfers=ops, scope=self._mfc_scopes[mfc.name]
            )
        return plan

    def _make_groups(self):
        """Pre-create every process group the step will use (group
        creation is collective over the WORLD — it must happen here, in
        identical order on all ranks, never inside the concurrent loop)."""
        if not dist.is_initialized():
            return
        for mfc in self._topo_order:
            p = self._plan[mfc.name]
            if 1 < len(p.mesh) < self.world:
                new_or_get_group(list(p.mesh))
            for t in p.transfers:
                if len(t.group_ranks) < self.world:
                    new_or_get_group(list(t.group_ranks))

    def _mesh_group(self, mesh: Tuple[int, ...]):
        if not dist.is_initialized() or len(mesh) == self.world:
            return None  # default group
        return new_or_get_group(list(mesh))

    # ------------------------------------------------------- profiling
    def _maybe_profile(self, mfc: MFCDef):
        """Per-MFC profiler dumps (reference: __maybe_profile_rpc,
        model_worker.py:663/716): REALHF_AMD_DUMP_TRACE=1 writes a chrome
        trace per (rank, mfc, step) under LOG_ROOT/trace/;
        REALHF_AMD_DUMP_MEMORY=1 snapshots the allocator after the MFC."""
        import contextlib

        trace = os.environ.get("REALHF_AMD_DUMP_TRACE") == "1"
        memdump = (os.environ.get("REALHF_AMD_DUMP_MEMORY") == "1"
                   and torch.cuda.is_available())

        if not trace and not memdump:
            return contextlib.nullcontext()

        @contextlib.contextmanager
        def ctx():
            root = constants.LOG_ROOT(constants.experiment_name(),
                                      constants.trial_name())
            if memdump:
                torch.cuda.memory._record_memory_history(max_entries=100000)
            if trace:
                acts = [torch.profiler.ProfilerActivity.CPU]
                if torch.cuda.is_available():
                    acts.append(torch.profiler.ProfilerActivity.CUDA)
                with torch.profiler.profile(activities=acts) as prof:
                    yield
                d = os.path.join(root, "trace")
                os.makedirs(d, exist_ok=True)
                prof.export_chrome_trace(os.path.join(
                    d, f"{mfc.name}_r{self.rank}_s{self._step}.json"))
            else:
                yield
            if memdump:
                d = os.path.join(root, "memory")
                os.makedirs(d, exist_ok=True)
                torch.cuda.memory._dump_snapshot(os.path.join(
                    d, f"{mfc.name}_r{self.rank}_s{self._step}.pickle"))
                torch.cuda.memory._record_memory_history(enabled=None)

        return ctx()

    # ------------------------------------------------------------- hooks
    def _run_hook(self, hook, mfc: MFCDef, pre: bool):
        if isinstance(hook, OffloadHook):
            m = self.models.get(mfc.model_name)
            if m is not None:
                real = m.module.model if hasattr(m.module, "model") else m.module
                if hook.to == "dp_shard":
                    with constants.model_scope(str(mfc.model_name)):
                        real.shard_to_dp()
                else:
                    real.async_offload()
            return
        if isinstance(hook, ParamReallocHook):
            src_name = hook.source if hook.source is not None else mfc.model_name
            dst_name = hook.target if hook.target is not None else mfc.model_name
            src_strat = self.model_strategies[src_name]
            dst_strat = self.model_strategies[dst_name]
            src_m = self.models.get(src_name)
            dst_m = self.models.get(dst_name)
            involved_ranks = {r for _, r in src_strat.rank_map} | {
                r for _, r in dst_strat.rank_map
            }
            if self.rank not in involved_ranks:
                return
            src_real = dst_real = None
            if src_m is not None:
                src_real = (
                    src_m.module.model if hasattr(src_m.module, "model")
                    else src_m.module
                )
                if src_real._offloaded:
                    src_real.reload_from_offload()
                if getattr(src_real, "_dp_sharded", False):
                    with constants.model_scope(str(src_name)):
                        src_real.gather_from_dp()
            if dst_m is not None:
                dst_real = (
                    dst_m.module.model if hasattr(dst_m.module, "model")
                    else dst_m.module
                )
                if dst_real._offloaded:
                    dst_real.reload_from_offload()
                if getattr(dst_real, "_dp_sharded", False):
                    with constants.model_scope(str(dst_name)):
                        dst_real.gather_from_dp()
            cfg = (src_real or dst_real).config
            plan = build_realloc_plan(cfg, src_strat, dst_strat)
            execute_realloc(
                plan,
                src_real.flat_param if src_real is not None else None,
                dst_real.flat_param if dst_real is not None else None,
                eta=hook.eta,
            )
            return
        raise TypeError(hook)

    # -------------------------------------------------------------- step
    def run_step(self, batch: SequenceSample) -> Dict[str, Any]:
        """Run one full DFG iteration on `batch` (the replicated global
        batch).  Returns merged train stats (on every rank)."""
        # store: key -> single-key full-batch sample (this rank's copy;
        # present only if this rank holds the key)
        store: Dict[str, SequenceSample] = {
            k: batch.select_keys([k]) for k in batch.keys
        }
        local_stats: Dict[str, dict] = {}
        # transfer-path choice (device RCCL payloads vs gloo object
        # collectives) must be IDENTICAL on every rank of a broadcast
        # group — key off CUDA availability, not off this rank's model
        # placement (a rank hosting no model must still pick the same path)
        dev = (torch.device("cuda", torch.cuda.current_device())
               if torch.cuda.is_available() else None)
        for m in self.models.values():
            if m.device.type != "cuda":
                dev = None  # CPU-model runs (tests) use object collectives
            break

        for mfc in self._topo_order:
            plan = self._plan[mfc.name]
            alloc = self.allocations[mfc.name]
            mesh = plan.mesh
            on_mesh = self.rank in mesh

            # 1. plan-driven input transfers (only src/missing ranks engage)
            for t in plan.transfers:
                if self.rank not in t.group_ranks:
                    continue
                grp = (new_or_get_group(list(t.group_ranks))
                       if dist.is_initialized() and len(t.group_ranks) < self.world
                       else None)
                sample = store.get(t.key) if self.rank == t.src else None
                received = broadcast_sample(sample, t.src, grp, device=dev)
                if self.rank != t.src:
                    store[t.key] = received

            # 2. pre-hooks
            for h in mfc.pre_hooks:
                self._run_hook(h, mfc, pre=True)

            local_out = None
            if on_mesh:
                t0 = time.time()
                with constants.model_scope(plan.scope), \
                        self._maybe_profile(mfc):
                    local_out = self._exec_mfc(mfc, alloc, store, local_stats)
                if torch.cuda.is_available():
                    torch.cuda.synchronize()
                self._mfc_wall[mfc.name] = time.time() - t0

            # 4. merge outputs across this mesh's DP heads
            if mfc.interface_type != ModelInterfaceType.TRAIN_STEP and on_mesh:
                if len(mesh) == 1:
                    merged = local_out
                else:
                    grp = self._mesh_group(mesh)
                    merged = gather_across_dp(local_out, group=grp, device=dev)
                if merged is not None:
                    self._merge_outputs(mfc, merged, store)

            # 5. post-hooks
            for h in mfc.post_hooks:
                self._run_hook(h, mfc, pre=False)

        self._step += 1
        return self._merge_step_stats(local_stats)

    def _exec_mfc(self, mfc: MFCDef, alloc: MFCAllocation,
                  store: Dict[str, SequenceSample],
                  local_stats: Dict[str, dict]):
        g = constants.grid()
        model = self.models[mfc.model_name]
        real = (
            model.module.model
            if hasattr(model.module, "model") else model.module
        )
        if getattr(real, "_offloaded", False):
            real.reload_from_offload()
        if getattr(real, "_dp_sharded", False):
            with constants.model_scope(str(mfc.model_name)):
                real.gather_from_dp()
        inp = self._assemble_input(mfc, store)
        if mfc.input_key_remap:
            inp.remap_keys_(mfc.input_key_remap)
        shard = dp_shard(inp, g.dp_rank, g.dp_size)
        shard = shard.to_device(model.device)
        iface = self.interfaces[mfc.name]
        if mfc.interface_type == ModelInterfaceType.GENERATE:
            res = iface.generate(model, shard, n_mbs=alloc.n_mbs)
        elif mfc.interface_type == ModelInterfaceType.INFERENCE:
            res = iface.inference(model, shard, n_mbs=alloc.n_mbs)
        elif mfc.interface_type == ModelInterfaceType.TRAIN_STEP:
            res = iface.train_step(model, shard, n_mbs=alloc.n_mbs)
        else:
            res = iface.evaluate(model, [shard])
        if os.environ.get("REALHF_AMD_LOG_MEM") == "1":
            # per-MFC memory table (reference: model_worker __log_gpu_stats:999)
            from realhf_amd.base.monitor import gpu_memory_stats

            logger.info("%s mem: %s", mfc.name,
                        {k: round(v, 2) for k, v in gpu_memory_stats().items()})
        local_out = None
        if isinstance(res, SequenceSample):
            if mfc.output_key_remap:
                res.remap_keys_(mfc.output_key_remap)
            # one DP head per shard reports (tp 0, last pp stage)
            if g.tp_rank == 0 and g.pp_rank == g.pp_size - 1:
                local_out = res
        elif isinstance(res, dict):
            # every DP head contributes; step-end merge averages over them
            if g.tp_rank == 0 and g.pp_rank == g.pp_size - 1:
                local_stats[mfc.name] = res
        self._record_flops(mfc, getattr(real, "config", None), shard, res, g)
        return local_out

    def _record_flops(self, mfc, cfg, shard, res, g):
        """Per-MFC model-FLOPs estimate (reference: master_worker's
        TFLOP/s line, master_worker.py:1461-1488) — merged into the step
        stats as <mfc>/tflops_per_gpu at the step boundary."""
        if cfg is None:
            return
        try:
            from realhf_amd.base.monitor import estimate_mfc_flops

            key = ("packed_input_ids" if "packed_input_ids" in shard.keys
                   else next(iter(shard.keys)))
            seqlens = [sum(x) for x in shard.seqlens[key]]
            out_seqlens = None
            if (mfc.interface_type == ModelInterfaceType.GENERATE
                    and isinstance(res, SequenceSample)
                    and "packed_input_ids" in res.keys):
                out_seqlens = [sum(x) for x in
                               res.seqlens["packed_input_ids"]]
            fl = estimate_mfc_flops(mfc.interface_type, cfg, seqlens,
                                    out_seqlens)
            self._mfc_flops[mfc.name] = fl / max(1, g.tp_size * g.pp_size)
        except Exception:  # FLOPs logging must never break the step
            self._mfc_flops.pop(mfc.name, None)

    def _assemble_input(self, mfc: MFCDef,
                        store: Dict[str, SequenceSample]) -> SequenceSample:
        keys = [k for k in mfc.input_keys if k in store]
        assert keys, (mfc.name, mfc.input_keys, list(store))
        base = store[keys[0]]
        out = base
        for k in keys[1:]:
            s = store[k]
            if list(map(str, s.ids)) != list(map(str, out.ids)):
                order = _order_by_ids(s.ids, out.ids)
                assert len(order) == out.bs, (
                    f"{mfc.name}: id mismatch joining key {k}"
                )
                s = s.select_idx(order)
            if out is base:
                out = SequenceSample(
                    keys=tuple(base.keys), ids=list(base.ids),
                    seqlens=dict(base.seqlens), data=dict(base.data),
                )
            out.update_(s)
        return out

    def _merge_outputs(self, mfc: MFCDef, merged: SequenceSample,
                       store: Dict[str, SequenceSample]):
        ref_key = next(iter(store), None)
        same_batch = (
            ref_key is not None
            and sorted(map(str, merged.ids)) == sorted(map(str, store[ref_key].ids))
        )
        if same_batch:
            order = _order_by_ids(merged.ids, store[ref_key].ids)
            merged = merged.select_idx(order)
        else:
            # the producer re-keyed the batch (e.g. GRPO group expansion:
            # each prompt becomes group_size responses with fresh ids) —
            # its output id-space becomes the store's
            for k in list(store):
                del store[k]
        for k in merged.keys:
            store[k] = merged.select_keys([k])

    def _merge_step_stats(self, local_stats: Dict[str, dict]) -> Dict[str, Any]:
        """One world-scoped object gather per STEP (not per MFC) — the
        step boundary is a natural sync point; this also drains any rank
        skew before the next dataloader batch."""
        flat = {
            f"{mfc}/{k}": float(v)
            for mfc, st in local_stats.items()
            for k, v in st.items() if isinstance(v, (int, float))
        }
        for mfc, w in self._mfc_wall.items():
            flat[f"{mfc}/wall_s"] = w
            fl = self._mfc_flops.get(mfc)
            if fl and w > 0:
                flat[f"{mfc}/tflops_per_gpu"] = fl / w / 1e12
        if not dist.is_initialized():
            return flat
        bucket: List = [None] * self.world
        dist.all_gather_object(bucket, flat)
        vals: Dict[str, List[float]] = {}
        for b in bucket:
            for k, v in (b or {}).items():
                vals.setdefault(k, []).append(v)
        out = {k: sum(v) / len(v) for k, v in vals.items()}
        return out


def _order_by_ids(ids: List, target_ids: List) -> List[int]:
    pos = {str(i): j for j, i in enumerate(ids)}
    return [pos[str(t)] for t in target_ids if str(t) in pos]


def gather_across_dp_stats(local: Optional[dict]) -> Optional[dict]:
    """Legacy helper (kept for tests): world-mean of per-rank stat dicts."""
    if not dist.is_initialized():
        return local
    world = dist.get_world_size()
    bucket: List = [None] * world
    dist.all_gather_object(bucket, local)
    vals: Dict[str, List[float]] = {}
    for b in bucket:
        if b is None:
            continue
        for k, v in b.items():
            if isinstance(v, (int, float)):
                vals.setdefault(k, []).append(float(v))
    return {k: sum(v) / len(v) for k, v in vals.items()}
