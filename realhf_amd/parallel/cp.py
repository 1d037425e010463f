"""Ulysses-style context parallelism over xGMI (long-context training).

Capability BEYOND the reference: the reference has no context/sequence
parallelism across devices (context-parallel groups are stubbed to self,
megatron.py:74-76) and covers long context with packing + Megatron SP
only (which notes Ulysses all-to-all as the natural fit
for the xGMI all-pairs topology — every pair of GPUs on an MI355X node
has a direct link, exactly the traffic pattern of all-to-all).

Scheme (DeepSpeed-Ulysses): tokens of the packed batch are sharded over
the CP group; weights stay replicated.  Everything token-local
(embeddings, norms, MLPs, router) runs on the shard.  For attention,
ONE all-to-all turns the (token-shard, all-heads) layout into
(all-tokens, head-shard) so the existing varlen MFMA kernel sees full
sequences; a second all-to-all turns the output back.  Cost per layer =
2 all-to-alls of [T, H, hd]/cp each way vs ring-attention's P2P chain —
on xGMI the all-to-all moves every chunk concurrently over distinct
links.

Usage (see tests/test_cp.py):
    local_ids, positions, info = cp.shard_batch(ids, cu, rank, size)
    with cp.context(group, rank, size, info):
        out_local = model(packed_input_ids=local_ids, positions=positions,
                          cu_seqlens=None, max_seqlen=None)

Constraints: total tokens divisible by cp (shard_batch pads the batch
with a trailing self-contained pad sequence), attention heads (after
TP) divisible by cp, no KV-cache/generation under CP (training +
inference forward only).
"""
import contextlib
import dataclasses
from typing import Optional

import torch
import torch.distributed as dist


@dataclasses.dataclass
class CPInfo:
    full_cu: torch.Tensor  # [n_seqs+1] over the PADDED full batch
    full_max: int
    total: int  # padded total tokens
    orig_total: int  # tokens before padding


@dataclasses.dataclass
class _CPState:
    group: object
    rank: int
    size: int
    info: CPInfo


_STATE: Optional[_CPState] = None


def current() -> Optional[_CPState]:
    return _STATE


@contextlib.contextmanager
def context(group, rank: int, size: int, info: CPInfo):
    global _STATE
    prev = _STATE
    _STATE = _CPState(group, rank, size, info)
    try:
        yield
    finally:
        _STATE = prev


def shard_batch(packed_ids: torch.Tensor, cu_seqlens: torch.Tensor,
                cp_rank: int, cp_size: int, pad_id: int = 0):
    """Split a packed batch into equal token shards (padding the batch
    with one trailing pad 'sequence' when total % cp != 0).  Returns
    (local_ids, local_positions, CPInfo)."""
    total = packed_ids.shape[0]
    pad = (-total) % cp_size
    cu = cu_seqlens.to(torch.int64)
    if pad:
        packed_ids = torch.cat([
            packed_ids,
            torch.full((pad,), pad_id, dtype=packed_ids.dtype,
                       device=packed_ids.device),
        ])
        cu = torch.cat([cu, torch.tensor([total + pad], dtype=torch.int64,
                                         device=cu.device)])
    lens = (cu[1:] - cu[:-1])
    positions = torch.cat([
        torch.arange(int(l), device=packed_ids.device) for l in lens
    ])
    t_loc = (total + pad) // cp_size
    s0 = cp_rank * t_loc
    info = CPInfo(full_cu=cu.to(torch.int32), full_max=int(lens.max()),
                  total=total + pad, orig_total=total)
    return packed_ids[s0:s0 + t_loc], positions[s0:s0 + t_loc], info


class _SeqHeadAllToAll(torch.autograd.Function):
    """[t_local, H, hd] <-> [t_full, H/cp, hd] over the CP group.
    `gather_seq=True`: gather tokens, scatter heads (into attention);
    False: the inverse (out of attention).  Backward is the opposite
    direction — one collective each way."""

    @staticmethod
    def forward(ctx, x, gather_seq: bool):
        st = _STATE
        ctx.gather_seq = gather_seq
        ctx.st = st  # captured: backward may run after the context exits
        return _a2a(x, st, gather_seq)

    @staticmethod
    def backward(ctx, g):
        return _a2a(g.contiguous(), ctx.st, not ctx.gather_seq), None


def _a2a(x, st, gather_seq):
    cp = st.size
    if cp == 1:
        return x
    if gather_seq:
        t_loc, H, hd = x.shape
        hc = H // cp
        # [t_loc, cp, hc, hd] -> [cp, t_loc, hc, hd]; chunk r goes to rank r
        send = (x.view(t_loc, cp, hc, hd).permute(1, 0, 2, 3).contiguous())
        recv = torch.empty_like(send)
        dist.all_to_all_single(recv.view(cp, -1), send.view(cp, -1),
                               group=st.group)
        # recv[r] = rank r's token shard of MY head chunk; rank-major
        # concat = global token order
        return recv.view(cp * t_loc, hc, hd)
    else:
        t_full, hc, hd = x.shape
        t_loc = t_full // cp
        send = x.view(cp, t_loc, hc, hd).contiguous()
        recv = torch.empty_like(send)
        dist.all_to_all_single(recv.view(cp, -1), send.view(cp, -1),
                               group=st.group)
        # recv[r] = my token shard's head chunk r -> [t_loc, H, hd]
        return (recv.permute(1, 0, 2, 3).contiguous()
                .view(t_loc, cp * hc, hd))


def seq_gather_head_scatter(x: torch.Tensor) -> torch.Tensor:
    return _SeqHeadAllToAll.apply(x, True)


def head_gather_seq_scatter(x: torch.Tensor) -> torch.Tensor:
    return _SeqHeadAllToAll.apply(x, False)
