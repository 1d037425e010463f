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


def test_memory_model_terms():
    """KV cache, activation, and optimizer-offload terms of the memory
    model (reference: estimate_rpc_memory_cost, estimate.py:387-450)."""
    from realhf_amd.api.config import ModelInterfaceType, ParallelismConfig
    from realhf_amd.search.engine import MFCSpec, estimate_mem_bytes

    base = dict(name="g", role="actor", n_seqs=128, avg_seqlen=128,
                param_bytes=14e9, flops_per_token=14e9,
                n_layers=32, hidden_dim=4096, n_kv_heads=8, head_dim=128)
    par = ParallelismConfig(data_parallel_size=1)
    gen_short = MFCSpec(interface_type=ModelInterfaceType.GENERATE,
                        gen_tokens=128, **base)
    gen_long = MFCSpec(interface_type=ModelInterfaceType.GENERATE,
                       gen_tokens=1024, **base)
    m_s = estimate_mem_bytes(gen_short, par, trainable=False)
    m_l = estimate_mem_bytes(gen_long, par, trainable=False)
    assert m_l > m_s  # KV cache grows with generation length
    # exact KV delta: bs * dtok * nkv * hd * 2 * 2 * L
    expect = 128 * (1024 - 128) * 8 * 128 * 2 * 2 * 32
    assert abs((m_l - m_s) - expect) / expect < 1e-6

    tr = MFCSpec(interface_type=ModelInterfaceType.TRAIN_STEP, **base)
    tr_off = MFCSpec(interface_type=ModelInterfaceType.TRAIN_STEP,
                     offload_optimizer=True, **base)
    m_tr = estimate_mem_bytes(tr, par, trainable=True)
    m_off = estimate_mem_bytes(tr_off, par, trainable=True)
    assert m_tr - m_off == pytest.approx(14e9 * 6)  # fp32 states offloaded

    ckpt = MFCSpec(interface_type=ModelInterfaceType.TRAIN_STEP,
                   gradient_checkpointing=True, **base)
    nockpt = MFCSpec(interface_type=ModelInterfaceType.TRAIN_STEP,
                     gradient_checkpointing=False, **base)
    assert (estimate_mem_bytes(nockpt, par, True)
            > estimate_mem_bytes(ckpt, par, True))


def test_search_allocation_grpo(tmp_path):
    """allocation_mode=search now extends past PPO (round-1 gap)."""
    from realhf_amd.api.experiment import GRPOConfig
    from realhf_amd.runtime.trainer import _apply_search_allocation

    cfg = GRPOConfig(experiment_name="s-grpo", trial_name="t", n_gpus=8)
    cfg.allocation_mode = "search"
    cfg.dataset.train_bs_n_seqs = 32
    cfg.dataset.max_prompt_len = 64
    cfg.ppo.gen.max_new_tokens = 64
    _apply_search_allocation(cfg, 8)
    assert cfg.allocation_mode == "manual"
    assert cfg.actor.parallel.world_size <= 8
    assert cfg.ref.parallel.world_size <= 8
