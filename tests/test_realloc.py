"""Parameter reallocation round-trip tests (reference test:
tests/comm/test_param_realloc.py — the crown jewel).

Multi-process over gloo on CPU (world 4); the same code path runs RCCL
on GPU boxes."""
import numpy as np
import pytest
import torch

from realhf_amd.base.testing import LocalMultiProcessTest
from realhf_amd.models.hf.llama import make_test_config
from realhf_amd.models import param_layout as PL
from realhf_amd.parallel.realloc import (
    ParallelStrategy,
    build_realloc_plan,
    execute_realloc,
)


def _full_reference_sd(cfg, seed=3):
    torch.manual_seed(seed)
    sd = {}
    for k in PL.all_keys(cfg):
        sd[k] = torch.randn(PL.key_full_shape(cfg, k))
    return sd


def _fill_model_from_full(model, cfg, sd):
    with torch.no_grad():
        for k in model.layout.keys:
            shard = PL.tp_partition(cfg, k, sd[k], model.tp_rank, model.tp_size)
            model.param_view(k).copy_(shard)


def _check_model_vs_full(model, cfg, sd):
    for k in model.layout.keys:
        shard = PL.tp_partition(cfg, k, sd[k], model.tp_rank, model.tp_size)
        got = model.param_view(k)
        assert torch.equal(got, shard), (k, (got - shard).abs().max())


def test_plan_conservation():
    """Every dst element is written exactly once."""
    cfg = make_test_config(n_layers=4, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    for (sp, sd_, st), (dp, dd, dt) in [
        ((1, 2, 2), (2, 1, 2)),
        ((1, 1, 4), (4, 1, 1)),
        ((2, 2, 1), (1, 1, 4)),
    ]:
        src = ParallelStrategy.make(sp, sd_, st)
        dst = ParallelStrategy.make(dp, dd, dt,
                                    ranks=list(range(dp * dd * dt)))
        plan = build_realloc_plan(cfg, src, dst)
        # per dst rank: covered elements == union of key extents
        dst_parts = PL.partition_pipeline_layers(cfg, dp)
        for p in range(dp):
            for d in range(dd):
                for t in range(dt):
                    r = dst.rank_of(p, d, t)
                    layout = PL.build_flat_layout(
                        cfg, list(range(*dst_parts[p])), t, dt
                    )
                    expect = sum(
                        layout.specs[k].numel for k in layout.keys
                    )
                    ivs = []
                    for tr in plan.transfers:
                        if tr.dst_rank == r:
                            ivs.append(tr.dst_intervals)
                    ivs = np.concatenate(ivs, axis=0)
                    order = np.argsort(ivs[:, 0])
                    ivs = ivs[order]
                    total = int((ivs[:, 1] - ivs[:, 0]).sum())
                    assert total == expect, (p, d, t, total, expect)
                    # non-overlapping
                    assert (ivs[1:, 0] >= ivs[:-1, 1]).all()


def _realloc_worker(src_geom, dst_geom, eta=1.0):
    import torch.distributed as dist

    from realhf_amd.models.real_model import ReaLModel

    cfg = make_test_config(n_layers=4, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg)
    rank = dist.get_rank()
    sp, sdp, st = src_geom
    dp, ddp, dt = dst_geom
    src = ParallelStrategy.make(sp, sdp, st)
    dst = ParallelStrategy.make(dp, ddp, dt)

    src_model = None
    for (p, d, t), r in src.rank_map:
        if r == rank:
            src_model = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                                  tp_rank=t, tp_size=st, pp_rank=p, pp_size=sp)
            _fill_model_from_full(src_model, cfg, sd)
    dst_model = None
    for (p, d, t), r in dst.rank_map:
        if r == rank:
            dst_model = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                                  tp_rank=t, tp_size=dt, pp_rank=p, pp_size=dp)
            with torch.no_grad():
                dst_model.flat_param.zero_()
    plan = build_realloc_plan(cfg, src, dst)
    execute_realloc(
        plan,
        src_model.flat_param if src_model is not None else None,
        dst_model.flat_param if dst_model is not None else None,
        eta=eta,
    )
    if dst_model is not None and eta == 1.0:
        _check_model_vs_full(dst_model, cfg, sd)
    dist.barrier()
    # round trip back
    plan2 = build_realloc_plan(cfg, dst, src)
    if src_model is not None:
        with torch.no_grad():
            src_model.flat_param.zero_()
    execute_realloc(
        plan2,
        dst_model.flat_param if dst_model is not None else None,
        src_model.flat_param if src_model is not None else None,
    )
    if src_model is not None:
        _check_model_vs_full(src_model, cfg, sd)


@pytest.mark.distributed
@pytest.mark.parametrize(
    "src_geom,dst_geom",
    [
        ((1, 2, 2), (2, 2, 1)),
        ((1, 1, 4), (4, 1, 1)),
        ((2, 2, 1), (1, 1, 4)),
        ((1, 4, 1), (1, 1, 4)),
        ((2, 1, 2), (1, 2, 2)),
    ],
)
def test_realloc_roundtrip(src_geom, dst_geom):
    LocalMultiProcessTest(4, _realloc_worker, src_geom, dst_geom).launch()


def _ema_worker():
    import torch.distributed as dist

    from realhf_amd.models.real_model import ReaLModel

    cfg = make_test_config(n_layers=2, hidden_dim=32, n_heads=4,
                           n_kv_heads=2, vocab_size=64)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg)
    src = ParallelStrategy.make(1, 2, 1)
    dst = ParallelStrategy.make(1, 1, 2)
    rank = dist.get_rank()
    src_model = dst_model = None
    for (p, d, t), r in src.rank_map:
        if r == rank:
            src_model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
            _fill_model_from_full(src_model, cfg, sd)
    for (p, d, t), r in dst.rank_map:
        if r == rank:
            dst_model = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                                  tp_rank=t, tp_size=2)
            with torch.no_grad():
                dst_model.flat_param.fill_(1.0)
    plan = build_realloc_plan(cfg, src, dst)
    execute_realloc(
        plan,
        src_model.flat_param if src_model is not None else None,
        dst_model.flat_param if dst_model is not None else None,
        eta=0.5,
    )
    if dst_model is not None:
        k = "1.attn.wq.weight"
        shard = PL.tp_partition(cfg, k, sd[k], dst_model.tp_rank, 2)
        expect = 0.5 * shard + 0.5 * torch.ones_like(shard)
        torch.testing.assert_close(dst_model.param_view(k), expect)


@pytest.mark.distributed
def test_realloc_ema():
    """eta < 1 EMA-merges into the destination (ref-EMA)."""
    LocalMultiProcessTest(2, _ema_worker).launch()


def _moe_cfg():
    from realhf_amd.api.model import MoEConfig
    cfg = make_test_config(n_layers=2, hidden_dim=32, n_heads=4, n_kv_heads=2,
                           vocab_size=64)
    cfg.moe = MoEConfig(num_experts=4, top_k=2, expert_parallel_size=2)
    return cfg


def test_plan_conservation_ep():
    """EP: expert keys live only on their ep block; every dst element of
    every (tp, ep) shard is written exactly once."""
    cfg = _moe_cfg()
    for (s_geom, s_ep), (d_geom, d_ep) in [
        (((1, 2, 1), 2), ((1, 2, 1), 1)),   # ep2 -> replicated
        (((1, 2, 1), 1), ((1, 2, 1), 2)),   # replicated -> ep2
        (((1, 4, 1), 2), ((1, 2, 2), 2)),   # ep2 dp4 -> ep2 dp2 tp2
        (((1, 4, 1), 4), ((1, 2, 1), 2)),   # ep4 -> ep2
    ]:
        src = ParallelStrategy.make(*s_geom, ep=s_ep)
        dst = ParallelStrategy.make(*d_geom, ep=d_ep)
        plan = build_realloc_plan(cfg, src, dst)
        dst_parts = PL.partition_pipeline_layers(cfg, dst.pp)
        for p in range(dst.pp):
            for d in range(dst.dp):
                for t in range(dst.tp):
                    r = dst.rank_of(p, d, t)
                    layout = PL.build_flat_layout(
                        cfg, list(range(*dst_parts[p])), t, dst.tp,
                        ep_rank=d % dst.ep, ep_size=dst.ep,
                    )
                    expect = sum(layout.specs[k].numel for k in layout.keys)
                    ivs = np.concatenate(
                        [tr.dst_intervals for tr in plan.transfers
                         if tr.dst_rank == r], axis=0)
                    ivs = ivs[np.argsort(ivs[:, 0])]
                    total = int((ivs[:, 1] - ivs[:, 0]).sum())
                    assert total == expect, (s_geom, s_ep, d_geom, d_ep,
                                             p, d, t, total, expect)
                    assert (ivs[1:, 0] >= ivs[:-1, 1]).all()


def _fill_model_from_full_ep(model, cfg, sd):
    with torch.no_grad():
        for k in model.layout.keys:
            shard = PL.tp_partition(cfg, k, sd[k], model.tp_rank, model.tp_size)
            model.param_view(k).copy_(shard)


def _realloc_ep_worker(s_ep, d_ep):
    import torch.distributed as dist

    from realhf_amd.models.real_model import ReaLModel

    cfg = _moe_cfg()
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg)
    rank = dist.get_rank()
    src = ParallelStrategy.make(1, 4, 1, ep=s_ep)
    dst = ParallelStrategy.make(1, 2, 2, ep=d_ep)

    src_model = None
    for (p, d, t), r in src.rank_map:
        if r == rank:
            src_model = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                                  tp_rank=t, tp_size=1, pp_rank=p, pp_size=1,
                                  ep_rank=d % s_ep, ep_size=s_ep)
            _fill_model_from_full_ep(src_model, cfg, sd)
    dst_model = None
    for (p, d, t), r in dst.rank_map:
        if r == rank:
            dst_model = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                                  tp_rank=t, tp_size=2, pp_rank=p, pp_size=1,
                                  ep_rank=d % d_ep, ep_size=d_ep)
            with torch.no_grad():
                dst_model.flat_param.zero_()
    plan = build_realloc_plan(cfg, src, dst)
    execute_realloc(
        plan,
        src_model.flat_param if src_model is not None else None,
        dst_model.flat_param if dst_model is not None else None,
    )
    if dst_model is not None:
        _check_model_vs_full(dst_model, cfg, sd)
    dist.barrier()
    plan2 = build_realloc_plan(cfg, dst, src)
    if src_model is not None:
        with torch.no_grad():
            src_model.flat_param.zero_()
    execute_realloc(
        plan2,
        dst_model.flat_param if dst_model is not None else None,
        src_model.flat_param if src_model is not None else None,
    )
    if src_model is not None:
        _check_model_vs_full(src_model, cfg, sd)


@pytest.mark.distributed
@pytest.mark.parametrize("s_ep,d_ep", [(2, 1), (4, 2), (2, 2)])
def test_realloc_roundtrip_ep(s_ep, d_ep):
    LocalMultiProcessTest(4, _realloc_ep_worker, s_ep, d_ep).launch()


def _realloc_worker_ex(src_geom, dst_geom, src_ranks, dst_ranks, family,
                       is_critic):
    """Extended matrix: disjoint rank sets (the replica/asym flow), other
    model families, and critic head shapes (reference crown-jewel style:
    tests/comm/test_param_realloc.py parametrization)."""
    import torch.distributed as dist

    import realhf_amd.models.hf as hf_reg
    from realhf_amd.models.real_model import ReaLModel

    fam = hf_reg.get_family(family)
    kw = dict(n_layers=4, hidden_dim=64, n_heads=8, vocab_size=128)
    if family != "gpt2":  # gpt2 is MHA: kv heads fixed = n_heads
        kw["n_kv_heads"] = 4
    cfg = fam.make_test_config(**kw)
    cfg.is_critic = is_critic
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg, seed=101)
    rank = dist.get_rank()
    sp, sdp, st = src_geom
    dp, ddp, dt = dst_geom
    src = ParallelStrategy.make(sp, sdp, st, ranks=src_ranks)
    dst = ParallelStrategy.make(dp, ddp, dt, ranks=dst_ranks)

    src_model = None
    for (p, d, t), r in src.rank_map:
        if r == rank:
            src_model = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                                  tp_rank=t, tp_size=st, pp_rank=p, pp_size=sp)
            _fill_model_from_full(src_model, cfg, sd)
    dst_model = None
    for (p, d, t), r in dst.rank_map:
        if r == rank:
            dst_model = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                                  tp_rank=t, tp_size=dt, pp_rank=p, pp_size=dp)
            with torch.no_grad():
                dst_model.flat_param.zero_()
    plan = build_realloc_plan(cfg, src, dst)
    execute_realloc(
        plan,
        src_model.flat_param if src_model is not None else None,
        dst_model.flat_param if dst_model is not None else None,
    )
    if dst_model is not None:
        _check_model_vs_full(dst_model, cfg, sd)
    dist.barrier()


@pytest.mark.distributed
@pytest.mark.parametrize(
    "src_geom,dst_geom,src_ranks,dst_ranks,family,is_critic",
    [
        # DISJOINT rank sets: train mesh [0,1] -> gen mesh [2,3]
        ((1, 2, 1), (1, 2, 1), [0, 1], [2, 3], "llama", False),
        ((1, 1, 2), (2, 1, 1), [0, 1], [2, 3], "llama", False),
        # critic head (replicated 1-dim output) across tp<->pp remaps
        ((1, 2, 2), (2, 2, 1), None, None, "llama", True),
        # second family
        ((1, 1, 4), (4, 1, 1), None, None, "gpt2", False),
        ((2, 2, 1), (1, 1, 4), None, None, "gpt2", True),
    ],
)
def test_realloc_matrix_extended(src_geom, dst_geom, src_ranks, dst_ranks,
                                 family, is_critic):
    LocalMultiProcessTest(4, _realloc_worker_ex, src_geom, dst_geom,
                          src_ranks, dst_ranks, family, is_critic).launch()
