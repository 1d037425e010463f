"""Allocation search engine tests (reference: realhf/search_engine +
csrc/search MCMC; allocation_mode=search)."""
import pytest

from realhf_amd.api.config import ModelInterfaceType
from realhf_amd.search.engine import (
    MFCSpec,
    enumerate_strategies,
    estimate_time_s,
    search_allocations,
)
from tests.test_dfg import ppo_mfcs
from realhf_amd.api.dfg import build_graph


def _specs():
    T = ModelInterfaceType
    p7 = 6.7e9
    common = dict(n_seqs=128, avg_seqlen=640, param_bytes=p7 * 2,
                  flops_per_token=2 * p7)
    return {
        "actor_gen": MFCSpec("actor_gen", "actor", T.GENERATE,
                             gen_tokens=512, **{**common, "avg_seqlen": 128}),
        "rew_inf": MFCSpec("rew_inf", "reward", T.INFERENCE, **common),
        "ref_inf": MFCSpec("ref_inf", "ref", T.INFERENCE, **common),
        "critic_inf": MFCSpec("critic_inf", "critic", T.INFERENCE, **common),
        "actor_train": MFCSpec("actor_train", "actor", T.TRAIN_STEP, **common),
        "critic_train": MFCSpec("critic_train", "critic", T.TRAIN_STEP, **common),
    }


def test_enumerate():
    ss = enumerate_strategies(8)
    assert any(s.world_size == 8 for s in ss)
    assert all(s.world_size <= 8 for s in ss)


def test_time_model_monotonic():
    T = ModelInterfaceType
    m = _specs()["actor_train"]
    ss = enumerate_strategies(8)
    one = [s for s in ss if s.world_size == 1][0]
    eight = [s for s in ss if s.world_size == 8 and s.data_parallel_size == 8][0]
    assert estimate_time_s(m, eight) < estimate_time_s(m, one)


def test_mcmc_search_runs():
    g = build_graph(ppo_mfcs())
    alloc, cost = search_allocations(
        g, _specs(), trainable_roles=["actor", "critic"], n_gpus=8,
        n_chains=4, n_steps=4000,
    )
    assert set(alloc) == {m.name for m in g.mfcs}
    assert 0 < cost < 1e6
    # the searched plan should use the full node for training MFCs
    assert alloc["actor_train"].world_size >= 4


def test_search_allocation_mode_in_experiment():
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import build_experiment

    cfg = PPOConfig(allocation_mode="search", n_gpus=8)
    cfg.dataset.train_bs_n_seqs = 128
    cfg.dataset.max_prompt_len = 128
    cfg.ppo.gen.max_new_tokens = 512
    built = build_experiment(cfg, 8)
    assert cfg.allocation_mode == "manual"
    assert built.allocations["actor_train"].strategy.world >= 4


def test_cost_table_calibration(tmp_path):
    import json

    from realhf_amd.search import engine as E

    old = (E.BF16_PEAK_TF, E.HBM_GBPS)
    p = tmp_path / "ct.json"
    p.write_text(json.dumps({"bf16_tf": 1000.0, "hbm_gbps": 5000.0}))
    E.load_cost_table(str(p))
    assert E.BF16_PEAK_TF == 1000.0 and E.HBM_GBPS == 5000.0
    E.BF16_PEAK_TF, E.HBM_GBPS = old
