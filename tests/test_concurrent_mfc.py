"""Concurrent mesh-scoped MFC execution (round-2 item #1).

Proves the executor's core concurrency property (reference counterpart:
per-MFC asyncio coroutines, master_worker.py:455-680): two inference
MFCs allocated to DISJOINT device meshes execute simultaneously in
wall-clock, because no collective in the per-MFC path spans more than
(producer mesh | consumer mesh).  Also covers the static ownership/
transfer plan and the asymmetric PPO heuristic end-to-end on 4 ranks.
"""
import dataclasses
import json
import os
import time

import numpy as np
import pytest
import torch

from realhf_amd.api.config import Abstraction, ModelInterfaceType, ModelName
from realhf_amd.api.data import SequenceSample
from realhf_amd.api.dfg import MFCDef, build_graph
from realhf_amd.api.model import Model, ModelInterface
from realhf_amd.base import constants
from realhf_amd.base.testing import LocalMultiProcessTest
from realhf_amd.base.topology import ParallelGrid, PipeDataTensorTopology
from realhf_amd.parallel.realloc import ParallelStrategy
from realhf_amd.runtime.executor import DFGExecutor, MFCAllocation


def _make_batch(bs=8, seqlen=5):
    toks = torch.arange(bs * seqlen, dtype=torch.long)
    return SequenceSample(
        keys=("x",),
        ids=[f"s{i}" for i in range(bs)],
        seqlens={"x": [[seqlen]] * bs},
        data={"x": toks},
    )


@dataclasses.dataclass
class _SleepInference(ModelInterface):
    out_key: str = "y"
    sleep_s: float = 0.0

    def inference(self, model, data, n_mbs=None):
        time.sleep(self.sleep_s)
        return SequenceSample(
            keys=(self.out_key,),
            ids=list(data.ids),
            seqlens={self.out_key: [[1]] * data.bs},
            data={self.out_key: torch.full((data.bs,), float(self.sleep_s))},
        )


@dataclasses.dataclass
class _ConsumeTrain(ModelInterface):
    need: tuple = ()

    def train_step(self, model, data, n_mbs=None):
        for k in self.need:
            assert k in data.keys, (k, data.keys)
        return {"ok": 1.0}


def _dummy_model(name):
    return Model(name=name, module=torch.nn.Identity(), tokenizer=None,
                 device=torch.device("cpu"), dtype=torch.float32)


def _register_grid(scope: str, ranks, dp):
    topo = PipeDataTensorTopology(num_pp=1, num_dp=dp, num_tp=1)
    rank_mapping = {i: r for i, r in enumerate(ranks)}
    constants.set_grid(scope, ParallelGrid(topo, rank_mapping))


def _overlap_worker(sleep_s, outfile):
    import torch.distributed as dist

    rank = dist.get_rank()
    ma, mb, mc = ModelName("a", 0), ModelName("b", 0), ModelName("c", 0)
    strat_a = ParallelStrategy.make(1, 2, 1, ranks=[0, 1])
    strat_b = ParallelStrategy.make(1, 2, 1, ranks=[2, 3])
    strat_c = ParallelStrategy.make(1, 4, 1, ranks=[0, 1, 2, 3])
    _register_grid("a@0", [0, 1], 2)
    _register_grid("b@0", [2, 3], 2)
    _register_grid("c@0", [0, 1, 2, 3], 4)
    mfcs = [
        MFCDef(name="inf_a", model_name=ma,
               interface_type=ModelInterfaceType.INFERENCE,
               interface_impl=Abstraction("x"), input_keys=("x",),
               output_keys=("ya",)),
        MFCDef(name="inf_b", model_name=mb,
               interface_type=ModelInterfaceType.INFERENCE,
               interface_impl=Abstraction("x"), input_keys=("x",),
               output_keys=("yb",)),
        MFCDef(name="train_c", model_name=mc,
               interface_type=ModelInterfaceType.TRAIN_STEP,
               interface_impl=Abstraction("x"),
               input_keys=("x", "ya", "yb"), output_keys=()),
    ]
    graph = build_graph(mfcs)
    allocations = {
        "inf_a": MFCAllocation(strategy=strat_a),
        "inf_b": MFCAllocation(strategy=strat_b),
        "train_c": MFCAllocation(strategy=strat_c),
    }
    models = {}
    if rank in (0, 1):
        models[ma] = _dummy_model(ma)
    if rank in (2, 3):
        models[mb] = _dummy_model(mb)
    models[mc] = _dummy_model(mc)
    interfaces = {
        "inf_a": _SleepInference(out_key="ya", sleep_s=sleep_s),
        "inf_b": _SleepInference(out_key="yb", sleep_s=sleep_s),
        "train_c": _ConsumeTrain(need=("x", "ya", "yb")),
    }
    ex = DFGExecutor(graph, allocations, models, interfaces,
                     {ma: strat_a, mb: strat_b, mc: strat_c})

    # the static plan must schedule broadcasts of ya -> {2,3} and yb -> {0,1}
    tplan = {t.key: t for t in ex._plan["train_c"].transfers}
    assert set(tplan) == {"ya", "yb"}
    assert set(tplan["ya"].group_ranks) >= {2, 3}
    assert tplan["ya"].src in (0, 1)
    assert set(tplan["yb"].group_ranks) >= {0, 1}
    assert tplan["yb"].src in (2, 3)

    dist.barrier()
    t0 = time.time()
    stats = ex.run_step(_make_batch())
    wall = time.time() - t0
    assert stats.get("train_c/ok") == 1.0, stats
    if rank == 0:
        with open(outfile, "w") as f:
            json.dump({"wall": wall}, f)


@pytest.mark.distributed
def test_disjoint_mesh_mfcs_overlap(tmp_path):
    """Two 0.6 s inference MFCs on disjoint 2-rank meshes: serial
    execution would take >= 1.2 s; the concurrent executor must finish
    the step well under that."""
    sleep_s = 1.0
    out = str(tmp_path / "wall.json")
    LocalMultiProcessTest(4, _overlap_worker, sleep_s, out).launch()
    wall = json.load(open(out))["wall"]
    # concurrent ~= sleep_s + transfer overhead; serial >= 2*sleep_s
    assert wall < 2 * sleep_s * 0.85, (
        f"step took {wall:.2f}s — disjoint-mesh MFCs did not overlap "
        f"(serial would be ~{2*sleep_s:.1f}s)"
    )


def test_static_plan_sub_mesh_ownership():
    """Plan math without torch.distributed: a consumer mesh missing a
    producer's key gets exactly one broadcast from a holder."""
    ma, mb = ModelName("a", 0), ModelName("b", 0)
    strat_a = ParallelStrategy.make(1, 2, 1, ranks=[0, 1])
    strat_b = ParallelStrategy.make(1, 2, 1, ranks=[2, 3])
    mfcs = [
        MFCDef(name="p", model_name=ma,
               interface_type=ModelInterfaceType.INFERENCE,
               interface_impl=Abstraction("x"), input_keys=("x",),
               output_keys=("y",)),
        MFCDef(name="q", model_name=mb,
               interface_type=ModelInterfaceType.INFERENCE,
               interface_impl=Abstraction("x"), input_keys=("y",),
               output_keys=("z",)),
    ]
    graph = build_graph(mfcs)
    allocations = {
        "p": MFCAllocation(strategy=strat_a),
        "q": MFCAllocation(strategy=strat_b),
    }
    ex = DFGExecutor(graph, allocations, {}, {},
                     {ma: strat_a, mb: strat_b})
    # single process: world == 1 -> every mesh collapses to {0}; instead
    # check the plan builder directly with a forced world of 4
    ex.world = 4
    plan = ex._static_plan()
    assert plan["p"].transfers == []
    (t,) = plan["q"].transfers
    assert t.key == "y"
    assert t.src in (0, 1)
    assert set(t.group_ranks) == {t.src, 2, 3}


def _write_prompt_data(path, n=16, vocab=64):
    rng = np.random.RandomState(1)
    with open(path, "w") as f:
        for _ in range(n):
            rec = {"input_ids": rng.randint(3, vocab - 3,
                                            size=rng.randint(4, 8)).tolist()}
            f.write(json.dumps(rec) + "\n")


def _ppo_heuristic4_worker(data, fileroot, logits_mask=False):
    """4-rank PPO with allocation_mode=heuristic: the asymmetric plan
    puts critic_inf on ranks {0,1} and rew_inf (and the reward model
    itself) on {2,3} — concurrent disjoint-mesh inference in a real
    experiment."""
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    cfg = PPOConfig(experiment_name="t-ppoh", trial_name="dist", n_gpus=4)
    for mc in (cfg.actor, cfg.critic, cfg.ref, cfg.rew):
        mc.dtype = "float32"
    cfg.allocation_mode = "heuristic"
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 8
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 5
    cfg.ppo.gen.use_hip_graph = False
    if logits_mask:
        cfg.ppo.gen.top_k = 4
        cfg.ppo.gen.force_no_logits_mask = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 1
    t = Trainer(cfg)
    # verify the asymmetric shape before running
    assert t.executor._plan["critic_inf"].mesh == (0, 1)
    assert t.executor._plan["rew_inf"].mesh == (2, 3)
    assert t.executor._plan["ref_inf"].mesh == (0, 1, 2, 3)
    rew_name = ModelName("rew", 0)
    import torch.distributed as dist

    if dist.get_rank() in (0, 1):
        assert rew_name not in t.models  # reward model only lives on {2,3}
    else:
        assert rew_name in t.models
    t.run()


@pytest.mark.distributed
def test_ppo_heuristic_asymmetric_four_ranks(tmp_path):
    data = str(tmp_path / "prompts.jsonl")
    _write_prompt_data(data, n=16)
    LocalMultiProcessTest(4, _ppo_heuristic4_worker, data,
                          str(tmp_path / "root")).launch()


@pytest.mark.distributed
def test_ppo_heuristic_four_ranks_logits_mask(tmp_path):
    """Asymmetric heuristic + logits-mask mode: the 2-D mask key rides
    the sub-mesh transfer plan (gen whole-node -> ref whole-node /
    actor_train whole-node, skipping critic/rew halves)."""
    data = str(tmp_path / "prompts.jsonl")
    _write_prompt_data(data, n=16)
    LocalMultiProcessTest(4, _ppo_heuristic4_worker, data,
                          str(tmp_path / "root2"), True).launch()


def test_ppo_heuristic8_static_plan():
    """World-8 asymmetric PPO plan (no processes): critic_inf on the
    first half, rew_inf on the second (where the reward model lives),
    trains whole-node; the train MFCs receive exactly the cross-half
    broadcasts of `values` and `rewards`."""
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import build_experiment
    from realhf_amd.runtime.executor import DFGExecutor

    cfg = PPOConfig(allocation_mode="heuristic")
    built = build_experiment(cfg, 8)
    ex = DFGExecutor(built.graph, built.allocations, {}, built.interfaces,
                     built.model_strategies)
    ex.world = 8
    plan = ex._static_plan()
    assert plan["critic_inf"].mesh == (0, 1, 2, 3)
    assert plan["rew_inf"].mesh == (4, 5, 6, 7)
    assert plan["ref_inf"].mesh == tuple(range(8))
    assert plan["actor_train"].mesh == tuple(range(8))
    # actor_train: values held by {0..3} -> broadcast to {4..7};
    # rewards held by {4..7} -> broadcast to {0..3}
    t = {x.key: x for x in plan["actor_train"].transfers}
    assert set(t) == {"values", "rewards"}
    assert set(t["values"].group_ranks) >= {4, 5, 6, 7}
    assert t["values"].src in (0, 1, 2, 3)
    assert set(t["rewards"].group_ranks) >= {0, 1, 2, 3}
    assert t["rewards"].src in (4, 5, 6, 7)
    # critic_train afterwards needs nothing new (actor_train spread both)
    assert plan["critic_train"].transfers == []


def test_train_mfc_with_outputs_rejected():
    """Train MFCs return stat dicts, not data — output keys on one must
    fail at PLAN time, not silently corrupt holder tracking."""
    ma = ModelName("a", 0)
    strat = ParallelStrategy.make(1, 2, 1, ranks=[0, 1])
    mfcs = [
        MFCDef(name="t", model_name=ma,
               interface_type=ModelInterfaceType.TRAIN_STEP,
               interface_impl=Abstraction("x"), input_keys=("x",),
               output_keys=("bad",)),
    ]
    graph = build_graph(mfcs)
    with pytest.raises(AssertionError, match="cannot produce"):
        ex = DFGExecutor(graph, {"t": MFCAllocation(strategy=strat)}, {}, {},
                         {ma: strat})
        ex.world = 2
        ex._static_plan()
