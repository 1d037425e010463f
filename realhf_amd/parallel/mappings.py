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


class _ColumnParallelLinear(torch.autograd.Function):
    """Fused (copy-to-TP | SP-gather) + GEMM with ASYNC backward comm
    (reference: LinearWithGradAccumulationAndAsyncCommunication,
    model_parallel/modules.py:232-340): the input-grad all-reduce (or
    SP reduce-scatter) is launched async and OVERLAPS the weight-grad
    GEMM, hiding the TP collective behind compute.  With SP, only the
    sequence shard of x is saved; the backward re-gathers it (memory
    for one extra all-gather, as the reference does)."""

    @staticmethod
    def forward(ctx, x, w, sp):
        ctx.sp = sp
        total_x = _gather_along_first_dim(x) if sp else x
        ctx.save_for_backward(x, w)
        return torch.nn.functional.linear(total_x, w)

    @staticmethod
    def backward(ctx, gy):
        x, w = ctx.saved_tensors
        sp = ctx.sp
        tp = _tp_size()
        gy2 = gy.reshape(-1, gy.shape[-1]).contiguous()
        total_x = _gather_along_first_dim(x) if sp else x
        gx_total = (gy2 @ w).view(*gy.shape[:-1], w.shape[1])
        handle = None
        if tp > 1 and sp:
            shape = list(gx_total.shape)
            shape[0] //= tp
            gx = torch.empty(shape, dtype=gx_total.dtype,
                             device=gx_total.device)
            handle = dist.reduce_scatter_tensor(
                gx, gx_total.contiguous(), group=_tp_group(), async_op=True)
        elif tp > 1:
            gx = gx_total.contiguous()
            handle = dist.all_reduce(gx, group=_tp_group(), async_op=True)
        else:
            gx = gx_total
        # the weight-grad GEMM runs while the collective is in flight
        gw = gy2.t() @ total_x.reshape(-1, total_x.shape[-1])
        if handle is not None:
            handle.wait()
        return gx, gw, None


def column_parallel_linear(x, w, sp: bool):
    """y = x @ w^T with the TP input mapping fused in; `x` must be the
    PRE-mapping tensor (SP shard when sp=True, full tokens otherwise)."""
    return _ColumnParallelLinear.apply(x, w, sp)


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
