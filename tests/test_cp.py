"""Ulysses context parallelism (capability beyond the reference —
SURVEY.md §5.7: the reference stubs context-parallel groups to self;
all-to-all CP is the natural fit for xGMI's all-pairs topology)."""
import numpy as np
import pytest
import torch

from realhf_amd.base.testing import LocalMultiProcessTest
from tests.test_realloc import _fill_model_from_full, _full_reference_sd
from realhf_amd.models.hf.llama import make_test_config


def _cp_worker():
    import torch.distributed as dist

    from realhf_amd.models.real_model import ReaLModel
    from realhf_amd.parallel import cp

    rank, world = dist.get_rank(), dist.get_world_size()
    cfg = make_test_config(n_layers=4, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg, seed=91)
    m = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    _fill_model_from_full(m, cfg, sd)

    rng = np.random.RandomState(23)
    lens = [9, 12, 6]  # total 27: NOT divisible by 2 -> exercises padding
    toks = torch.from_numpy(rng.randint(0, 128, size=sum(lens))).long()
    cu = torch.tensor([0] + list(np.cumsum(lens)), dtype=torch.int32)

    local_ids, local_pos, info = cp.shard_batch(toks, cu, rank, world)
    assert info.total % world == 0

    # ---- forward equivalence -----------------------------------------
    with torch.no_grad(), cp.context(None, rank, world, info):
        out_local = m(packed_input_ids=local_ids, positions=local_pos,
                      cu_seqlens=None, max_seqlen=None)
    single = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    _fill_model_from_full(single, cfg, sd)
    pad = info.total - info.orig_total
    padded_ids = torch.cat([toks, torch.zeros(pad, dtype=toks.dtype)])
    with torch.no_grad():
        ref = single(packed_input_ids=padded_ids,
                     cu_seqlens=info.full_cu, max_seqlen=info.full_max)
    t_loc = info.total // world
    ref_shard = ref[rank * t_loc:(rank + 1) * t_loc]
    torch.testing.assert_close(out_local, ref_shard, atol=1e-4, rtol=1e-4)

    # ---- backward: summed shard grads == single-process grads --------
    m.allocate_grad_buffer()
    for k, p in m._params.items():
        p.requires_grad_(True)
        p.grad = m.grad_view(k)
    with cp.context(None, rank, world, info):
        out = m(packed_input_ids=local_ids, positions=local_pos,
                cu_seqlens=None, max_seqlen=None)
    # backward OUTSIDE the context: the autograd fn must have captured
    # the cp state at forward time
    (out.float().square().sum() / info.total).backward()
    g = m.flat_grad.clone()
    dist.all_reduce(g)  # weights replicated: CP grads sum like DP

    single.allocate_grad_buffer()
    for k, p in single._params.items():
        p.requires_grad_(True)
        p.grad = single.grad_view(k)
    out_ref = single(packed_input_ids=padded_ids, cu_seqlens=info.full_cu,
                     max_seqlen=info.full_max)
    (out_ref.float().square().sum() / info.total).backward()
    torch.testing.assert_close(g, single.flat_grad, atol=5e-4, rtol=5e-4)
    dist.barrier()


@pytest.mark.distributed
def test_ulysses_cp2_forward_backward_matches_single():
    LocalMultiProcessTest(2, _cp_worker).launch()


def test_shard_batch_padding_math():
    from realhf_amd.parallel import cp

    toks = torch.arange(10)
    cu = torch.tensor([0, 4, 10], dtype=torch.int32)
    ids0, pos0, info = cp.shard_batch(toks, cu, 0, 4)
    assert info.total == 12 and info.orig_total == 10
    assert info.full_cu.tolist() == [0, 4, 10, 12]
    ids3, pos3, _ = cp.shard_batch(toks, cu, 3, 4)
    assert ids0.tolist() == [0, 1, 2]
    assert pos0.tolist() == [0, 1, 2]
    assert ids3.tolist() == [9, 0, 0]  # last real token + 2 pad
    assert pos3.tolist() == [5, 0, 1]


def _cp_sft_worker():
    """End-to-end CP TRAINING with a real next-token CE loss: labels are
    built globally (shift crosses shard boundaries) then sharded with the
    tokens; summed shard grads must match the single-process step."""
    import torch.distributed as dist
    import torch.nn.functional as F

    from realhf_amd.models.real_model import ReaLModel
    from realhf_amd.parallel import cp

    rank, world = dist.get_rank(), dist.get_world_size()
    cfg = make_test_config(n_layers=4, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg, seed=93)
    rng = np.random.RandomState(29)
    lens = [10, 14]
    toks = torch.from_numpy(rng.randint(0, 128, size=sum(lens))).long()
    cu = torch.tensor([0] + list(np.cumsum(lens)), dtype=torch.int32)

    # global labels: next token within each sequence, -100 at seq ends
    labels = torch.full((sum(lens),), -100, dtype=torch.long)
    for i in range(len(lens)):
        s, e = int(cu[i]), int(cu[i + 1])
        labels[s:e - 1] = toks[s + 1:e]

    local_ids, local_pos, info = cp.shard_batch(toks, cu, rank, world)
    pad = info.total - info.orig_total
    labels_p = torch.cat([labels, torch.full((pad,), -100, dtype=torch.long)])
    t_loc = info.total // world
    local_labels = labels_p[rank * t_loc:(rank + 1) * t_loc]

    def ce(logits, lab):
        return F.cross_entropy(logits.float(), lab, ignore_index=-100,
                               reduction="sum")

    m = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    _fill_model_from_full(m, cfg, sd)
    m.allocate_grad_buffer()
    for k, p in m._params.items():
        p.requires_grad_(True)
        p.grad = m.grad_view(k)
    n_pred = int((labels != -100).sum())
    with cp.context(None, rank, world, info):
        out = m(packed_input_ids=local_ids, positions=local_pos,
                cu_seqlens=None, max_seqlen=None)
    (ce(out, local_labels) / n_pred).backward()
    g = m.flat_grad.clone()
    dist.all_reduce(g)

    single = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    _fill_model_from_full(single, cfg, sd)
    single.allocate_grad_buffer()
    for k, p in single._params.items():
        p.requires_grad_(True)
        p.grad = single.grad_view(k)
    out_ref = single(packed_input_ids=toks, cu_seqlens=cu,
                     max_seqlen=max(lens))
    (ce(out_ref, labels) / n_pred).backward()
    torch.testing.assert_close(g, single.flat_grad, atol=5e-4, rtol=5e-4)
    dist.barrier()


@pytest.mark.distributed
def test_ulysses_cp2_sft_loss_matches_single():
    LocalMultiProcessTest(2, _cp_sft_worker).launch()
