"""MFC-to-MFC data movement.

Reference semantics: realhf/impl/model/comm/data_transfer.py (derive plan
+ NCCL broadcasts from producer DP-heads to consumer shards, send/recv
caches) and master_worker's per-MFC data ownership.

MI355X-native design: the SPMD runtime tracks WHICH ranks hold each data
key (ownership is a pure function of (graph, allocations) — no metadata
coordination at runtime).  Collectives are scoped to the ranks involved:

  * producer DP shards merge with an all-gather over the PRODUCER MESH
    group only,
  * a consumer mesh that lacks a key receives it with a broadcast on a
    group of {one holder} | {missing ranks} only.

Nothing ever barriers the whole node, so MFCs on disjoint meshes
(rew_inf | ref_inf | critic_inf) run concurrently in wall-clock — the
reference's core throughput lever (master_worker.py:455-680), obtained
here without any asyncio: disjoint meshes simply never meet in a
collective until the step boundary.

Payload path: sample tensors are packed into ONE contiguous device
tensor per (collective, dtype-class) and moved with RCCL over xGMI;
metadata (ids, seqlens, keys — a few hundred bytes) goes through the
object collectives.  Falls back to pure object transport for non-CUDA
runs (gloo CPU tests).
"""
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from realhf_amd.api.data import SequenceSample
from realhf_amd.base import logging

logger = logging.getLogger("data_transfer")


def _pack_payload(sample: SequenceSample):
    """Split a sample into (meta, flat fp tensor parts).  Tensors are
    flattened and concatenated per dtype so one RCCL message moves all
    keys; meta carries shapes/dtypes for reassembly."""
    meta = {
        "ids": list(sample.ids),
        "keys": list(sample.keys),
        "seqlens": {k: [list(x) for x in v] for k, v in sample.seqlens.items()},
        "specs": [],
    }
    by_dtype: Dict[torch.dtype, List[torch.Tensor]] = {}
    for k in sample.keys:
        t = sample.data[k]
        if t is None:  # metadata-only key (SequenceSample.gather keeps them)
            meta["specs"].append((k, None, None))
            continue
        meta["specs"].append((k, tuple(t.shape), str(t.dtype)))
        by_dtype.setdefault(t.dtype, []).append(t.reshape(-1))
    flats = {str(dt): torch.cat(ts) if len(ts) > 1 else ts[0]
             for dt, ts in by_dtype.items()}
    meta["flat_sizes"] = {dt: int(t.numel()) for dt, t in flats.items()}
    return meta, flats


def _unpack_payload(meta, flats: Dict[str, torch.Tensor]) -> SequenceSample:
    offsets = {dt: 0 for dt in flats}
    data = {}
    for k, shape, dts in meta["specs"]:
        if shape is None:
            data[k] = None
            continue
        n = 1
        for s in shape:
            n *= s
        off = offsets[dts]
        data[k] = flats[dts][off:off + n].reshape(shape)
        offsets[dts] = off + n
    return SequenceSample(
        keys=tuple(meta["keys"]), ids=list(meta["ids"]),
        seqlens=meta["seqlens"], data=data,
    )


def _str_dtype(s: str) -> torch.dtype:
    return getattr(torch, s.replace("torch.", ""))


def gather_across_dp(
    local: Optional[SequenceSample],
    group=None,
    device: Optional[torch.device] = None,
) -> Optional[SequenceSample]:
    """All-gather per-DP-rank output shards into the full batch on every
    rank OF THE GROUP (callers must pass the producer-mesh group; only
    its members may call).  `local` is this rank's produced shard (None
    on non-DP-head ranks, e.g. tp_rank > 0 — identical replicas are
    deduplicated by passing None there)."""
    if not dist.is_initialized():
        return local
    world = dist.get_world_size(group)
    if world == 1:
        return local
    if device is not None and device.type == "cuda":
        return _gather_across_dp_device(local, group, device)
    obj = local.cpu() if local is not None else None
    bucket: List = [None] * world
    dist.all_gather_object(bucket, obj, group=group)
    shards = [s for s in bucket if s is not None]
    return _merge_unique(shards)


def _gather_across_dp_device(local, group, device):
    """Device-native gather: metadata via object collective (tiny), flat
    tensor payloads via per-rank RCCL broadcasts on xGMI."""
    world = dist.get_world_size(group)
    if local is not None:
        local = local.to_device(device)
        meta, flats = _pack_payload(local)
    else:
        meta, flats = None, {}
    metas: List = [None] * world
    dist.all_gather_object(metas, meta, group=group)
    group_ranks = (dist.get_process_group_ranks(group) if group is not None
                   else list(range(world)))
    my_rank = dist.get_rank()
    shards = []
    # one broadcast per contributing rank; messages are queued on the
    # communicator and pipeline back-to-back over xGMI
    for i, m in enumerate(metas):
        if m is None:
            continue
        src_global = group_ranks[i]
        bufs = {}
        for dts, n in m["flat_sizes"].items():
            if src_global == my_rank:
                bufs[dts] = flats[dts].contiguous()
            else:
                bufs[dts] = torch.empty(n, dtype=_str_dtype(dts), device=device)
            dist.broadcast(bufs[dts], src=src_global, group=group)
        shards.append(_unpack_payload(m, bufs))
    return _merge_unique(shards)


def _merge_unique(shards: List[SequenceSample]) -> Optional[SequenceSample]:
    if not shards:
        return None
    # dedupe identical replicas (same ids) — keep first occurrence
    seen = set()
    uniq = []
    for s in shards:
        key = tuple(map(str, s.ids))
        if key in seen:
            continue
        seen.add(key)
        uniq.append(s)
    return SequenceSample.gather(uniq) if len(uniq) > 1 else uniq[0]


def broadcast_sample(
    sample: Optional[SequenceSample],
    src: int,
    group,
    device: Optional[torch.device] = None,
) -> SequenceSample:
    """Broadcast a full SequenceSample from global rank `src` to every
    rank of `group` (a transfer to a consumer mesh that lacks the key).
    Device path: metadata object broadcast + one RCCL broadcast per dtype."""
    if not dist.is_initialized():
        return sample
    my_rank = dist.get_rank()
    if device is not None and device.type == "cuda":
        if my_rank == src:
            sample = sample.to_device(device)
            meta, flats = _pack_payload(sample)
        else:
            meta, flats = None, None
        box = [meta]
        dist.broadcast_object_list(box, src=src, group=group)
        meta = box[0]
        bufs = {}
        for dts, n in meta["flat_sizes"].items():
            if my_rank == src:
                bufs[dts] = flats[dts].contiguous()
            else:
                bufs[dts] = torch.empty(n, dtype=_str_dtype(dts), device=device)
            dist.broadcast(bufs[dts], src=src, group=group)
        return sample if my_rank == src else _unpack_payload(meta, bufs)
    box = [sample.cpu() if my_rank == src else None]
    dist.broadcast_object_list(box, src=src, group=group)
    return box[0]


def dp_shard(
    sample: SequenceSample, dp_rank: int, dp_size: int
) -> SequenceSample:
    """Deterministic balanced DP shard of the full batch."""
    if dp_size == 1:
        return sample
    return sample.split(dp_size)[dp_rank]
