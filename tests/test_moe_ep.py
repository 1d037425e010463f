"""Expert parallelism tests (capability beyond the reference — SURVEY.md
§2.3 row EP: the reference replicates experts)."""
import numpy as np
import pytest
import torch

from realhf_amd.base.testing import LocalMultiProcessTest
from realhf_amd.models.hf import mixtral
from tests.test_realloc import _fill_model_from_full, _full_reference_sd


def _ep_worker():
    import torch.distributed as dist

    from realhf_amd.base import constants
    from realhf_amd.base.topology import ParallelGrid, PipeDataTensorTopology
    from realhf_amd.models.real_model import ReaLModel

    cfg = mixtral.make_test_config(n_layers=2, hidden_dim=64, n_heads=8,
                                   n_kv_heads=4, vocab_size=128)
    cfg.dtype = "float32"
    cfg.moe.expert_parallel_size = 2
    sd = _full_reference_sd(cfg, seed=61)
    rank = dist.get_rank()
    topo = PipeDataTensorTopology(num_pp=1, num_dp=2, num_tp=1, ep_size=2)
    grid = ParallelGrid(topo)
    constants.set_grid("m", grid)
    m = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                  ep_rank=grid.ep_rank, ep_size=2)
    _fill_model_from_full(m, cfg, sd)
    m.eval()

    single = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    _fill_model_from_full(single, cfg, sd)
    single.eval()

    rng = np.random.RandomState(3)
    lens = [7, 9]
    packed = torch.from_numpy(rng.randint(0, 128, size=sum(lens))).long()
    cu = torch.tensor([0] + list(np.cumsum(lens)), dtype=torch.int32)
    with torch.no_grad():
        ref = single(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=9)
        with constants.model_scope("m"):
            out = m(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=9)
    torch.testing.assert_close(out, ref, atol=1e-4, rtol=1e-4)
    # EP halves the expert parameter footprint
    n_full = single.layout.total_numel
    n_ep = m.layout.total_numel
    assert n_ep < n_full
    dist.barrier()


@pytest.mark.distributed
def test_ep2_forward_matches_replicated():
    LocalMultiProcessTest(2, _ep_worker).launch()


def _ep_train_worker():
    """EP backward: gradients flow through the all-to-all."""
    import torch.distributed as dist

    from realhf_amd.base import constants
    from realhf_amd.base.topology import ParallelGrid, PipeDataTensorTopology
    from realhf_amd.models.real_model import ReaLModel
    from realhf_amd.parallel.tp import packed_shifted_logprobs

    cfg = mixtral.make_test_config(n_layers=2, hidden_dim=64, n_heads=8,
                                   n_kv_heads=4, vocab_size=128)
    cfg.dtype = "float32"
    cfg.moe.expert_parallel_size = 2
    sd = _full_reference_sd(cfg, seed=71)
    topo = PipeDataTensorTopology(num_pp=1, num_dp=2, num_tp=1, ep_size=2)
    grid = ParallelGrid(topo)
    constants.set_grid("m", grid)
    m = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                  ep_rank=grid.ep_rank, ep_size=2)
    _fill_model_from_full(m, cfg, sd)
    m.allocate_grad_buffer()
    for k, p in m._params.items():
        p.requires_grad_(True)
        p.grad = m.grad_view(k)

    rng = np.random.RandomState(4)
    lens = [8, 8]
    packed = torch.from_numpy(rng.randint(0, 128, size=16)).long()
    cu = torch.tensor([0, 8, 16], dtype=torch.int32)
    with constants.model_scope("m"):
        logits = m(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=8)
        logp = packed_shifted_logprobs(logits, cu, packed)
        loss = -logp.mean()
        loss.backward()
    assert torch.isfinite(m.flat_grad).all()
    # expert grads nonzero somewhere
    gsum = float(m.flat_grad.abs().sum())
    assert gsum > 0
    dist.barrier()


@pytest.mark.distributed
def test_ep2_backward_through_all_to_all():
    LocalMultiProcessTest(2, _ep_train_worker).launch()


def _grpo_mixtral_ep_worker(data, fileroot):
    """BASELINE config #5 shape: Mixtral GRPO with expert parallelism,
    scaled down to a tiny model on 2 gloo ranks."""
    import os

    from realhf_amd.api.experiment import GRPOConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    cfg = GRPOConfig(experiment_name="t-grpo-ep", trial_name="cpu", n_gpus=2)
    for mc in (cfg.actor, cfg.ref, cfg.rew):
        mc.dtype = "float32"
        mc.family = "mixtral"
    cfg.rew.family = "llama"  # reward model stays dense
    cfg.rew.is_critic = True
    cfg.group_size = 2
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 2
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 4
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 1
    # inject EP: tiny mixtral test config has 4 experts; shard over 2 ranks
    from realhf_amd.runtime import trainer as T

    orig = T.build_experiment

    def patched(c, world):
        built = orig(c, world)
        for name, rcfg in built.model_cfgs.items():
            if rcfg.moe is not None:
                rcfg.moe.expert_parallel_size = 2
        return built

    T.build_experiment = patched
    try:
        Trainer(cfg).run()
    finally:
        T.build_experiment = orig


@pytest.mark.distributed
def test_grpo_mixtral_ep_experiment(tmp_path):
    import json

    rng = np.random.RandomState(5)
    data = str(tmp_path / "p.jsonl")
    with open(data, "w") as f:
        for _ in range(8):
            f.write(json.dumps(
                {"input_ids": rng.randint(3, 60, size=6).tolist()}) + "\n")
    LocalMultiProcessTest(
        2, _grpo_mixtral_ep_worker, data, str(tmp_path / "root")
    ).launch()
