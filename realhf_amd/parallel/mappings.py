"""Autograd-wrapped TP/SP collectives.

Reference semantics: realhf/impl/model/parallelism/model_parallel/mappings.py
(_reduce:13, _gather_along_first_dim:97, _reduce_scatter_along_first_dim:118,
copy/scatter/gather/reduce region fns:270-296).

All collectives run on the CURRENT model scope's TP group (RCCL over xGMI;
every TP group is a subset of one node's 8 GPUs, so each all-reduce is a
ring over directly-connected xGMI links).
"""
import torch
import torch.distributed as dist

from realhf_amd.base import constants


def _tp_size():
    if not constants.has_current():
        return 1
    return constants.tp_world_size()


def _tp_group():
    return constants.tp_group()


def _all_reduce(x):
    if _tp_size() == 1:
        return x
    if x.is_cuda and not torch.is_grad_enabled():
        # opt-in xGMI direct-read allreduce for small decode-time TP
        # messages (REALHF_AMD_XGMI_AR=1); returns None when disabled
        from realhf_amd.parallel import xgmi

        ar = xgmi.maybe_init_xgmi(_tp_group())
        if ar is not None:
            return ar.all_reduce(x.contiguous())
    dist.all_reduce(x.contiguous(), group=_tp_group())
    return x


def _gather_along_last_dim(x):
    sz = _tp_size()
    if sz == 1:
        return x
    out = [torch.empty_like(x) for _ in range(sz)]
    dist.all_gather(out, x.contiguous(), group=_tp_group())
    return torch.cat(out, dim=-1)


def _split_along_last_dim(x):
    sz = _tp_size()
    if sz == 1:
        return x
    r = constants.tp_rank()
    n = x.shape[-1] // sz
    return x[..., r * n : (r + 1) * n].contiguous()


def _gather_along_first_dim(x):
    sz = _tp_size()
    if sz == 1:
        return x
    shape = list(x.shape)
    shape[0] *= sz
    out = torch.empty(shape, dtype=x.dtype, device=x.device)
    dist.all_gather_into_tensor(out, x.contiguous(), group=_tp_group())
    return out


def _split_along_first_dim(x):
    sz = _tp_size()
    if sz == 1:
        return x
    r = constants.tp_rank()
    n = x.shape[0] // sz
    return x[r * n : (r + 1) * n].contiguous()


def _reduce_scatter_along_first_dim(x):
    sz = _tp_size()
    if sz == 1:
        return x
    shape = list(x.shape)
    assert shape[0] % sz == 0
    shape[0] //= sz
    out = torch.empty(shape, dtype=x.dtype, device=x.device)
    dist.reduce_scatter_tensor(out, x.contiguous(), group=_tp_group())
    return out


class _CopyToTPRegion(torch.autograd.Function):
    """fwd: identity; bwd: all-reduce (input of column-parallel linear)."""

    @staticmethod
    def forward(ctx, x):
        return x

    @staticmethod
    def backward(ctx, g):
        return _all_reduce(g.contiguous())


class _ReduceFromTPRegion(torch.autograd.Function):
    """fwd: all-reduce (output of row-parallel linear); bwd: identity."""

    @staticmethod
    def forward(ctx, x):
        return _all_reduce(x)

    @staticmethod
    def backward(ctx, g):
        return g


class _GatherFromSPRegion(torch.autograd.Function):
    """SP → full: fwd all-gather along dim 0; bwd reduce-scatter."""

    @staticmethod
    def forward(ctx, x):
        return _gather_along_first_dim(x)

    @staticmethod
    def backward(ctx, g):
        return _reduce_scatter_along_first_dim(g.contiguous())


class _ScatterToSPRegion(torch.autograd.Function):
    """full → SP: fwd split along dim 0; bwd all-gather."""

    @staticmethod
    def forward(ctx, x):
        return _split_along_first_dim(x)

    @staticmethod
    def backward(ctx, g):
        return _gather_along_first_dim(g.contiguous())


class _ReduceScatterToSPRegion(torch.autograd.Function):
    """full(partial-sums) → SP: fwd reduce-scatter; bwd all-gather."""

    @staticmethod
    def forward(ctx, x):
        return _reduce_scatter_along_first_dim(x)

    @staticmethod
    def backward(ctx, g):
        return _gather_along_first_dim(g.contiguous())


class _GatherFromTPRegion(torch.autograd.Function):
    """Gather last dim (e.g. full logits); bwd: split."""

    @staticmethod
    def forward(ctx, x):
        return _gather_along_last_dim(x)

    @staticmethod
    def backward(ctx, g):
        return _split_along_last_dim(g)


def copy_to_tp_region(x):
    return _CopyToTPRegion.apply(x)


def reduce_from_tp_region(x):
    return _ReduceFromTPRegion.apply(x)


def gather_from_sp_region(x):
    return _GatherFromSPRegion.apply(x)


def scatter_to_sp_region(x):
    return _ScatterToSPRegion.apply(x)


def reduce_scatter_to_sp_region(x):
    return _ReduceScatterToSPRegion.apply(x)


def gather_from_tp_region(x):
    return _GatherFromTPRegion.apply(x)
