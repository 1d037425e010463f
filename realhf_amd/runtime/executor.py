"""SPMD dataflow-graph executor.

Replaces the reference's master/worker runtime (realhf/system/
master_worker.py + model_worker.py + request_reply_stream.py) for the
single-node MI355X deployment: instead of a ZMQ master coordinating
workers with a req->syn->ack total-order protocol, EVERY rank runs the
same deterministic program over the DFG — per-MFC participation, DP
splits, data merges and reallocation plans are all pure functions of
(config, step), so no control messages exist in the hot path at all.
The req->syn->ack protocol's purpose (identical request order on every
worker, master_worker.py:74-95) is inherited by construction.

Per MFC and step:
  1. run pre-hooks (parameter realloc in / reload from offload),
  2. participating ranks take their balanced DP shard of the input keys
     and call the interface (generate / inference / train_step),
  3. DP-head outputs are merged into the replicated batch store
     (runtime/data_transfer.py),
  4. run post-hooks (realloc back / offload).
"""
import dataclasses
import os
import time
from typing import Any, Dict, List, Optional

import torch
import torch.distributed as dist

from realhf_amd.api.config import ModelInterfaceType, ModelName
from realhf_amd.api.data import SequenceSample
from realhf_amd.api.dfg import DFG, MFCDef, OffloadHook, ParamReallocHook
from realhf_amd.api.model import Model, ModelInterface
from realhf_amd.base import constants, logging
from realhf_amd.parallel.realloc import (
    ParallelStrategy,
    build_realloc_plan,
    execute_realloc,
)
from realhf_amd.runtime.data_transfer import dp_shard, gather_across_dp

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
        self._step = 0

    # ------------------------------------------------------------- hooks
    def _run_hook(self, hook, mfc: MFCDef, pre: bool):
        if isinstance(hook, OffloadHook):
            m = self.models.get(mfc.model_name)
            if m is not None:
                real = m.module.model if hasattr(m.module, "model") else m.module
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
            if dst_m is not None:
                dst_real = (
                    dst_m.module.model if hasattr(dst_m.module, "model")
                    else dst_m.module
                )
                if dst_real._offloaded:
                    dst_real.reload_from_offload()
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
        batch).  Returns merged train stats."""
        store = batch
        all_stats: Dict[str, Any] = {}
        for mfc in self.graph.topological_order():
            alloc = self.allocations[mfc.name]
            for h in mfc.pre_hooks:
                self._run_hook(h, mfc, pre=True)

            participating = self.rank in alloc.ranks
            local_out = None
            local_stats = None
            if participating:
                name = str(mfc.model_name)
                with constants.model_scope(name):
                    g = constants.grid()
                    model = self.models[mfc.model_name]
                    real = (
                        model.module.model
                        if hasattr(model.module, "model") else model.module
                    )
                    if getattr(real, "_offloaded", False):
                        real.reload_from_offload()
                    inp = store.select_keys(
                        [k for k in mfc.input_keys if k in store.keys]
                    )
                    if mfc.input_key_remap:
                        inp.remap_keys_(mfc.input_key_remap)
                    shard = dp_shard(inp, g.dp_rank, g.dp_size)
                    shard = shard.to_device(model.device)
                    iface = self.interfaces[mfc.name]
                    t0 = time.time()
                    if mfc.interface_type == ModelInterfaceType.GENERATE:
                        res = iface.generate(model, shard, n_mbs=alloc.n_mbs)
                    elif mfc.interface_type == ModelInterfaceType.INFERENCE:
                        res = iface.inference(model, shard, n_mbs=alloc.n_mbs)
                    elif mfc.interface_type == ModelInterfaceType.TRAIN_STEP:
                        res = iface.train_step(model, shard, n_mbs=alloc.n_mbs)
                    else:
                        res = iface.evaluate(model, [shard])
                    logger.debug(
                        "%s on rank %d took %.3fs", mfc.name, self.rank,
                        time.time() - t0,
                    )
                    if os.environ.get("REALHF_AMD_LOG_MEM") == "1":
                        # per-MFC memory table (reference: model_worker
                        # __log_gpu_stats:999)
                        from realhf_amd.base.monitor import gpu_memory_stats

                        logger.info("%s mem: %s", mfc.name,
                                    {k: round(v, 2) for k, v in
                                     gpu_memory_stats().items()})
                    if isinstance(res, SequenceSample):
                        if mfc.output_key_remap:
                            res.remap_keys_(mfc.output_key_remap)
                        # one DP head per shard reports (tp 0, last pp stage)
                        if g.tp_rank == 0 and g.pp_rank == g.pp_size - 1:
                            local_out = res
                    elif isinstance(res, dict):
                        if g.tp_rank == 0 and g.pp_rank == g.pp_size - 1:
                            local_stats = res

            if mfc.interface_type == ModelInterfaceType.TRAIN_STEP:
                merged_stats = gather_across_dp_stats(local_stats)
                if merged_stats:
                    for k, v in merged_stats.items():
                        all_stats[f"{mfc.name}/{k}"] = v
            else:
                merged = gather_across_dp(local_out)
                if merged is not None:
                    order = _order_by_ids(merged.ids, store.ids)
                    if len(order) == merged.bs == store.bs:
                        merged = merged.select_idx(order)
                        store.update_(merged)
                    else:
                        # the producer re-keyed the batch (e.g. GRPO group
                        # expansion: each prompt becomes group_size
                        # responses with fresh ids) — it becomes the store
                        store = merged

            for h in mfc.post_hooks:
                self._run_hook(h, mfc, pre=False)
        self._step += 1
        return all_stats


def _order_by_ids(ids: List, target_ids: List) -> List[int]:
    pos = {str(i): j for j, i in enumerate(ids)}
    return [pos[str(t)] for t in target_ids if str(t) in pos]


def gather_across_dp_stats(local: Optional[dict]) -> Optional[dict]:
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
