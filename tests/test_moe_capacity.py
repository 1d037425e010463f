"""Expert-capacity routing tests (reference: utils/moe.py:310
topk_softmax_with_capacity — capacity_factor / token_drop_policy /
pad_to_capacity on ReaLMoEConfig, model_api.py:112-123)."""
import numpy as np
import pytest
import torch

from realhf_amd.models.hf import mixtral
from realhf_amd.models.moe import apply_expert_capacity
from realhf_amd.models.real_model import ReaLModel
from tests.test_realloc import _fill_model_from_full, _full_reference_sd


def _make_models(capacity_factor=None, policy="probs", pad=False, seed=17):
    cfg = mixtral.make_test_config(n_layers=2, hidden_dim=64, n_heads=8,
                                   n_kv_heads=4, vocab_size=128)
    cfg.dtype = "float32"
    cfg.moe.capacity_factor = capacity_factor
    cfg.moe.token_drop_policy = policy
    cfg.moe.pad_to_capacity = pad
    sd = _full_reference_sd(cfg, seed=seed)
    m = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    _fill_model_from_full(m, cfg, sd)
    return cfg, m


def _batch(vocab=128, lens=(11, 13), seed=5):
    rng = np.random.RandomState(seed)
    packed = torch.from_numpy(rng.randint(0, vocab, size=sum(lens))).long()
    cu = torch.tensor([0] + list(np.cumsum(lens)), dtype=torch.int32)
    return packed, cu, max(lens)


def test_loose_capacity_is_noop():
    """cap >= tokens*k/n_experts * factor covers every assignment when the
    factor equals n_experts — output must equal the dropless model."""
    packed, cu, mx = _batch()
    _, dropless = _make_models(None)
    _, capped = _make_models(float(8))  # cap = tokens*k, nothing can drop
    dropless.eval(), capped.eval()
    with torch.no_grad():
        a = dropless(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
        b = capped(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    torch.testing.assert_close(a, b, atol=1e-5, rtol=1e-5)


@pytest.mark.parametrize("policy", ["probs", "position"])
def test_tight_capacity_drops_and_trains(policy):
    packed, cu, mx = _batch()
    _, dropless = _make_models(None)
    _, capped = _make_models(0.5, policy=policy)
    dropless.eval(), capped.eval()
    with torch.no_grad():
        a = dropless(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
        b = capped(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    assert not torch.allclose(a, b)  # the cap must actually bind
    # and the capped model still backprops
    capped.train()
    capped.allocate_grad_buffer()
    for k, p in capped._params.items():
        p.requires_grad_(True)
        p.grad = capped.grad_view(k)
    out = capped(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    out.float().square().mean().backward()
    assert torch.isfinite(capped.flat_grad).all()
    assert float(capped.flat_grad.abs().sum()) > 0


def test_pad_to_capacity_backward():
    """Gradients flow through the padded (static-shape) dispatch — the
    zero pad rows contribute nothing."""
    packed, cu, mx = _batch()
    _, pad = _make_models(0.5, pad=True)
    pad.train()
    pad.allocate_grad_buffer()
    for k, p in pad._params.items():
        p.requires_grad_(True)
        p.grad = pad.grad_view(k)
    out = pad(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    out.float().square().mean().backward()
    assert torch.isfinite(pad.flat_grad).all()
    assert float(pad.flat_grad.abs().sum()) > 0


@pytest.mark.parametrize("policy", ["probs", "position"])
def test_pad_to_capacity_matches_drop_mode(policy):
    """Padding only adds zero-weighted rows — logits must match the
    unpadded drop mode bit-for-bit in fp32 up to summation order."""
    packed, cu, mx = _batch()
    _, drop = _make_models(0.5, policy=policy, pad=False)
    _, pad = _make_models(0.5, policy=policy, pad=True)
    drop.eval(), pad.eval()
    with torch.no_grad():
        a = drop(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
        b = pad(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    torch.testing.assert_close(a, b, atol=1e-5, rtol=1e-5)


def test_apply_expert_capacity_unit():
    # 4 tokens, 2 experts, k=1: everyone picks expert 0 with descending
    # weights; cap = ceil(4*1/2 * 0.5) = 1
    scores = torch.tensor([[0.1], [0.4], [0.3], [0.2]])
    idx = torch.zeros(4, 1, dtype=torch.long)
    s, kept, cap = apply_expert_capacity(scores, idx, 2, 0.5, "probs")
    assert cap == 1
    assert kept.flatten().tolist() == [False, True, False, False]  # highest
    assert s.flatten().tolist() == [0.0, pytest.approx(0.4), 0.0, 0.0]
    s, kept, cap = apply_expert_capacity(scores, idx, 2, 0.5, "position")
    assert kept.flatten().tolist() == [True, False, False, False]  # earliest
    # per-expert kept counts never exceed cap under random top-k routing
    # (top-k picks distinct experts per token, like the real router)
    g = torch.Generator().manual_seed(0)
    logits = torch.rand(64, 8, generator=g)
    scores, idx = torch.topk(torch.softmax(logits, -1), 2, dim=-1)
    s, kept, cap = apply_expert_capacity(scores, idx, 8, 1.0, "probs")
    counts = torch.bincount(idx[kept], minlength=8)
    assert (counts <= cap).all()
    assert kept.sum() < idx.numel()  # something was dropped at factor 1.0


@pytest.mark.gpu
def test_capacity_forward_backward_gpu():
    """Capacity-limited + padded dispatch on the HIP grouped-GEMM path
    (bf16, kernel-eligible dims)."""
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    cfg = mixtral.make_test_config(n_layers=2, hidden_dim=64, n_heads=1,
                                   n_kv_heads=1, head_dim=64,
                                   intermediate_dim=128, vocab_size=128)
    cfg.dtype = "bfloat16"
    cfg.moe.capacity_factor = 1.0
    for pad in (False, True):
        cfg.moe.pad_to_capacity = pad
        m = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
        m.random_init()
        m.allocate_grad_buffer()
        for k, p in m._params.items():
            p.requires_grad_(True)
            p.grad = m.grad_view(k)
        rng = np.random.RandomState(11)
        ids = torch.from_numpy(rng.randint(0, 128, size=96)).long().cuda()
        cu = torch.tensor([0, 48, 96], dtype=torch.int32, device="cuda")
        out = m(packed_input_ids=ids, cu_seqlens=cu, max_seqlen=48)
        out.float().square().mean().backward()
        assert torch.isfinite(m.flat_grad.float()).all()
        assert float(m.flat_grad.float().abs().sum()) > 0


def _ep_capacity_worker():
    """Capacity dispatch under EP: the padded/filtered expert-sorted
    layout feeds the all-to-all exchange (counts per GLOBAL expert)."""
    import torch.distributed as dist

    from realhf_amd.base import constants
    from realhf_amd.base.topology import ParallelGrid, PipeDataTensorTopology
    from realhf_amd.models.real_model import ReaLModel

    for pad in (False, True):
        cfg = mixtral.make_test_config(n_layers=2, hidden_dim=64, n_heads=8,
                                       n_kv_heads=4, vocab_size=128)
        cfg.dtype = "float32"
        cfg.moe.capacity_factor = 0.5
        cfg.moe.pad_to_capacity = pad
        sd = _full_reference_sd(cfg, seed=77)
        topo = PipeDataTensorTopology(num_pp=1, num_dp=2, num_tp=1, ep_size=2)
        grid = ParallelGrid(topo)
        constants.set_grid(f"m{pad}", grid)
        cfg_ep = mixtral.make_test_config(n_layers=2, hidden_dim=64, n_heads=8,
                                          n_kv_heads=4, vocab_size=128)
        cfg_ep.dtype = "float32"
        cfg_ep.moe.capacity_factor = 0.5
        cfg_ep.moe.pad_to_capacity = pad
        cfg_ep.moe.expert_parallel_size = 2
        m = ReaLModel(cfg_ep, device="cpu", dtype=torch.float32,
                      ep_rank=grid.ep_rank, ep_size=2)
        _fill_model_from_full(m, cfg_ep, sd)
        m.eval()
        single = ReaLModel(cfg, device="cpu", dtype=torch.float32)
        _fill_model_from_full(single, cfg, sd)
        single.eval()
        rng = np.random.RandomState(8)
        lens = [9, 7]
        packed = torch.from_numpy(rng.randint(0, 128, size=16)).long()
        cu = torch.tensor([0, 9, 16], dtype=torch.int32)
        with torch.no_grad():
            ref = single(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=9)
            with constants.model_scope(f"m{pad}"):
                out = m(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=9)
        torch.testing.assert_close(out, ref, atol=1e-4, rtol=1e-4)
    import torch.distributed as dist
    dist.barrier()


@pytest.mark.distributed
def test_ep2_capacity_matches_replicated():
    from realhf_amd.base.testing import LocalMultiProcessTest

    LocalMultiProcessTest(2, _ep_capacity_worker).launch()
