"""Parameter reallocation: reshard flat-parameter models between 3D
parallel strategies.

Reference semantics: realhf/impl/model/comm/param_realloc.py
(_derive_reparallelize_comm_plan:312, ReparallelizeSenderStep/ReceiverStep)
and realhf/impl/model/nn/real_llm_api.py (build_reparallelized_layers_async
:610, patch_reparallelization:762).

MI355X-native design: instead of the reference's one-sender-per-node
broadcast trees (a multi-node optimization), the plan is a set of
point-to-point interval transfers executed as ONE batched isend/irecv
round — on a single 8-GPU node every (src, dst) pair has a direct xGMI
link, so all pair messages move concurrently (SURVEY.md §2.4 MI355X
mapping note).  Interval gather/scatter on-device uses the HIP
slice_intervals/set_intervals kernels.

Every rank computes the SAME global plan deterministically (SPMD — no
coordinator), keyed by (config, src strategy, dst strategy).
"""
import dataclasses
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist

from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.base import logging
from realhf_amd.models import param_layout as PL
from realhf_amd.ops import functional as ops

logger = logging.getLogger("realloc")


@dataclasses.dataclass(frozen=True)
class ParallelStrategy:
    """Geometry + the global ranks hosting each (pp, dp, tp) coordinate.
    rank_map[(pp, dp, tp)] = global rank.  With expert parallelism,
    `ep` divides dp and dp-coordinate d holds expert block d % ep
    (matching the trainer's ep_rank = dp_rank % ep_size convention)."""

    pp: int
    dp: int
    tp: int
    rank_map: Tuple[Tuple[Tuple[int, int, int], int], ...]  # hashable
    ep: int = 1

    @classmethod
    def make(cls, pp: int, dp: int, tp: int, ranks: Optional[List[int]] = None,
             ep: int = 1):
        n = pp * dp * tp
        if ranks is None:
            ranks = list(range(n))
        assert len(ranks) == n
        assert dp % max(ep, 1) == 0
        rm = []
        i = 0
        for p in range(pp):
            for d in range(dp):
                for t in range(tp):
                    rm.append((((p, d, t)), ranks[i]))
                    i += 1
        return cls(pp=pp, dp=dp, tp=tp, rank_map=tuple(rm), ep=ep)

    def rank_of(self, p, d, t):
        return dict(self.rank_map)[(p, d, t)]

    @property
    def world(self):
        return self.pp * self.dp * self.tp


@dataclasses.dataclass
class PairTransfer:
    src_rank: int
    dst_rank: int
    # parallel arrays of [n, 2] interval matrices (element offsets)
    src_intervals: np.ndarray  # into src flat buffer
    dst_intervals: np.ndarray  # into dst flat buffer
    numel: int


@dataclasses.dataclass
class ReallocPlan:
    cfg: ReaLModelConfig
    src: ParallelStrategy
    dst: ParallelStrategy
    transfers: List[PairTransfer]  # deterministic global order


def _key_intervals_for_shard_pair(
    cfg: ReaLModelConfig,
    key: str,
    src_layout: PL.FlatLayout,
    dst_layout: PL.FlatLayout,
    src_t: int,
    src_tp: int,
    dst_t: int,
    dst_tp: int,
) -> Optional[Tuple[np.ndarray, np.ndarray]]:
    """Interval lists (src, dst) moving the overlap of key's TP shards."""
    full = PL.key_full_shape(cfg, key)
    kind = PL.key_kind(key)
    if kind == "head":
        kind = PL.REPLICATED if cfg.is_critic else PL.VOCAB
    sspec = src_layout.specs[key]
    dspec = dst_layout.specs[key]

    if kind in (PL.COLUMN, PL.VOCAB):
        rows = full[0]
        rowsize = int(np.prod(full[1:], dtype=np.int64)) if len(full) > 1 else 1
        sr0, sr1 = rows // src_tp * src_t, rows // src_tp * (src_t + 1)
        dr0, dr1 = rows // dst_tp * dst_t, rows // dst_tp * (dst_t + 1)
        a, b = max(sr0, dr0), min(sr1, dr1)
        if a >= b:
            return None
        si = np.array([[sspec.start + (a - sr0) * rowsize,
                        sspec.start + (b - sr0) * rowsize]], dtype=np.int64)
        di = np.array([[dspec.start + (a - dr0) * rowsize,
                        dspec.start + (b - dr0) * rowsize]], dtype=np.int64)
        return si, di
    if kind == PL.ROW:
        rows, cols = full[0], full[1]
        sc0, sc1 = cols // src_tp * src_t, cols // src_tp * (src_t + 1)
        dc0, dc1 = cols // dst_tp * dst_t, cols // dst_tp * (dst_t + 1)
        a, b = max(sc0, dc0), min(sc1, dc1)
        if a >= b:
            return None
        scols = sc1 - sc0
        dcols = dc1 - dc0
        r = np.arange(rows, dtype=np.int64)
        s_start = sspec.start + r * scols + (a - sc0)
        d_start = dspec.start + r * dcols + (a - dc0)
        si = np.stack([s_start, s_start + (b - a)], axis=1)
        di = np.stack([d_start, d_start + (b - a)], axis=1)
        return si, di
    # replicated: move whole once (only from matching src_t slot to avoid
    # duplicates — caller passes the chosen src_t)
    si = np.array([[sspec.start, sspec.end]], dtype=np.int64)
    di = np.array([[dspec.start, dspec.end]], dtype=np.int64)
    return si, di


def _key_src_ep_rank(cfg: ReaLModelConfig, key: str, ep: int) -> Optional[int]:
    """Which src ep block holds this key; None = present on every block."""
    if ep <= 1 or ".experts." not in key:
        return None
    e = int(key.split(".experts.")[1].split(".")[0])
    return e // (cfg.moe.num_experts // ep)


_PLAN_CACHE: Dict = {}


def _cfg_key(cfg: ReaLModelConfig):
    import json

    d = dataclasses.asdict(cfg)
    return json.dumps(d, sort_keys=True, default=str)


def build_realloc_plan(
    cfg: ReaLModelConfig, src: ParallelStrategy, dst: ParallelStrategy
) -> ReallocPlan:
    cache_key = (_cfg_key(cfg), src, dst)
    if cache_key in _PLAN_CACHE:
        return _PLAN_CACHE[cache_key]

    src_parts = PL.partition_pipeline_layers(cfg, src.pp)
    dst_parts = PL.partition_pipeline_layers(cfg, dst.pp)
    # layouts per (stage, tp shard, ep block) — same for every dp replica
    # within an ep block
    src_layouts = {
        (p, t, e): PL.build_flat_layout(
            cfg, list(range(*src_parts[p])), t, src.tp,
            ep_rank=e, ep_size=src.ep,
        )
        for p in range(src.pp)
        for t in range(src.tp)
        for e in range(src.ep)
    }
    dst_layouts = {
        (p, t, e): PL.build_flat_layout(
            cfg, list(range(*dst_parts[p])), t, dst.tp,
            ep_rank=e, ep_size=dst.ep,
        )
        for p in range(dst.pp)
        for t in range(dst.tp)
        for e in range(dst.ep)
    }
    layer_to_src_stage = {}
    for p, (lo, hi) in src_parts.items():
        for l in range(lo, hi):
            layer_to_src_stage[l] = p

    # accumulate intervals per (src_rank, dst_rank)
    pair_src: Dict[Tuple[int, int], List[np.ndarray]] = {}
    pair_dst: Dict[Tuple[int, int], List[np.ndarray]] = {}

    n_src_rep = src.dp // src.ep  # dp replicas per expert block
    for dp_ in range(dst.pp):
        d_lo, d_hi = dst_parts[dp_]
        for dd in range(dst.dp):
            for dt in range(dst.tp):
                dst_rank = dst.rank_of(dp_, dd, dt)
                dl = dst_layouts[(dp_, dt, dd % dst.ep)]
                for layer in range(d_lo, d_hi):
                    sp_ = layer_to_src_stage[layer]
                    for key in PL.keys_of_layer(cfg, layer):
                        if key not in dl.specs:
                            continue  # expert outside this dst ep block
                        # expert keys exist only on src dp ranks whose ep
                        # block owns the expert; others on every dp rank
                        src_e = _key_src_ep_rank(cfg, key, src.ep)
                        if src_e is None:
                            sd = (dd + layer) % src.dp
                        else:
                            sd = src_e + src.ep * ((dd + layer) % n_src_rep)
                        se = sd % src.ep
                        kind = PL.key_kind(key)
                        if kind == "head":
                            kind = PL.REPLICATED if cfg.is_critic else PL.VOCAB
                        if kind == PL.REPLICATED:
                            # one source shard suffices
                            src_ts = [dt % src.tp]
                        else:
                            src_ts = range(src.tp)
                        for st in src_ts:
                            out = _key_intervals_for_shard_pair(
                                cfg, key, src_layouts[(sp_, st, se)], dl,
                                st, src.tp, dt, dst.tp,
                            )
                            if out is None:
                                continue
                            si, di = out
                            src_rank = src.rank_of(sp_, sd, st)
                            pair_src.setdefault((src_rank, dst_rank), []).append(si)
                            pair_dst.setdefault((src_rank, dst_rank), []).append(di)

    transfers = []
    for pair in sorted(pair_src.keys()):
        si = np.concatenate(pair_src[pair], axis=0)
        di = np.concatenate(pair_dst[pair], axis=0)
        numel = int((si[:, 1] - si[:, 0]).sum())
        assert numel == int((di[:, 1] - di[:, 0]).sum())
        transfers.append(
            PairTransfer(
                src_rank=pair[0], dst_rank=pair[1],
                src_intervals=si, dst_intervals=di, numel=numel,
            )
        )
    plan = ReallocPlan(cfg=cfg, src=src, dst=dst, transfers=transfers)
    _PLAN_CACHE[cache_key] = plan
    return plan


@torch.no_grad()
def execute_realloc(
    plan: ReallocPlan,
    src_flat: Optional[torch.Tensor],  # this rank's src shard (or None)
    dst_flat: Optional[torch.Tensor],  # this rank's dst shard (or None)
    eta: float = 1.0,
):
    """Run the transfer plan.  Caller must hold torch.distributed
    initialized when any cross-rank transfer exists; all ranks call this
    with the same plan."""
    me = dist.get_rank() if dist.is_initialized() else 0
    dtype = (src_flat if src_flat is not None else dst_flat).dtype
    device = (src_flat if src_flat is not None else dst_flat).device

    recv_list = []  # (transfer, buf, work)
    sends = []
    # local copies first; post irecvs before isends
    for tr in plan.transfers:
        if tr.dst_rank == me and tr.src_rank == me:
            buf = ops.slice_intervals(
                src_flat, torch.from_numpy(tr.src_intervals)
            )
            _apply_recv(dst_flat, buf, tr.dst_intervals, eta)
        elif tr.dst_rank == me:
            buf = torch.empty(tr.numel, dtype=dtype, device=device)
            w = dist.irecv(buf, src=tr.src_rank)
            recv_list.append((tr, buf, w))
    for tr in plan.transfers:
        if tr.src_rank == me and tr.dst_rank != me:
            buf = ops.slice_intervals(
                src_flat, torch.from_numpy(tr.src_intervals)
            ).contiguous()
            w = dist.isend(buf, dst=tr.dst_rank)
            sends.append((w, buf))
    for tr, buf, w in recv_list:
        w.wait()
        _apply_recv(dst_flat, buf, tr.dst_intervals, eta)
    for w, buf in sends:
        w.wait()


def _apply_recv(dst_flat, buf, dst_intervals, eta):
    iv = torch.from_numpy(dst_intervals)
    if eta >= 1.0:
        ops.set_intervals(buf, dst_flat, iv)
    else:
        cur = ops.slice_intervals(dst_flat, iv)
        merged = cur.float().mul_(1.0 - eta).add_(buf.float(), alpha=eta)
        ops.set_intervals(merged.to(dst_flat.dtype), dst_flat, iv)


# ---------------------------------------------------------------------------
# high-level entry: reallocate a ReaLModel between two strategies
# ---------------------------------------------------------------------------
def reallocate_model(
    cfg: ReaLModelConfig,
    src_strategy: ParallelStrategy,
    dst_strategy: ParallelStrategy,
    src_model=None,  # ReaLModel on this rank holding the src shard (or None)
    dst_model=None,  # ReaLModel to fill (or None if this rank holds no dst shard)
    eta: float = 1.0,
):
    plan = build_realloc_plan(cfg, src_strategy, dst_strategy)
    execute_realloc(
        plan,
        src_model.flat_param if src_model is not None else None,
        dst_model.flat_param if dst_model is not None else None,
        eta=eta,
    )
    return dst_model
