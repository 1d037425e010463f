"""MFC-to-MFC data movement.

Reference semantics: realhf/impl/model/comm/data_transfer.py (derive plan
+ NCCL broadcasts from producer DP-heads to consumer shards).

MI355X-native design: sample payloads are tiny next to weights (a 128-seq
PPO rollout batch is a few MB vs 14 GB of actor weights), so the SPMD
runtime keeps the FULL batch replicated on every rank's host memory and
each MFC takes its balanced DP shard from it.  Producer outputs are
merged back with one all-gather over the node (RCCL all_gather_object on
xGMI); re-splits are deterministic so no metadata coordination is needed.
"""
from typing import List, Optional

import torch
import torch.distributed as dist

from realhf_amd.api.data import SequenceSample
from realhf_amd.base import logging

logger = logging.getLogger("data_transfer")


def gather_across_dp(
    local: Optional[SequenceSample], group=None, src_ranks: Optional[List[int]] = None
) -> Optional[SequenceSample]:
    """All-gather per-DP-rank output shards into the full batch on every
    rank.  `local` is this rank's produced shard (None on non-DP-head
    ranks, e.g. TP rank > 0 — identical replicas are deduplicated by
    passing None there)."""
    if not dist.is_initialized():
        return local
    world = dist.get_world_size(group)
    obj = None
    if local is not None:
        obj = local.cpu()
    bucket: List = [None] * world
    dist.all_gather_object(bucket, obj, group=group)
    shards = [s for s in bucket if s is not None]
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
    return SequenceSample.gather(uniq)


def dp_shard(
    sample: SequenceSample, dp_rank: int, dp_size: int
) -> SequenceSample:
    """Deterministic balanced DP shard of the full batch."""
    if dp_size == 1:
        return sample
    return sample.split(dp_size)[dp_rank]
