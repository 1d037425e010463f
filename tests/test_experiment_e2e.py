"""End-to-end experiment runs through the CLI config path on CPU
(reference milestone: config #1 'GPT-2 small SFT on CPU' + the quickstart
CLI, SURVEY.md §7 stage 4)."""
import json
import os

import numpy as np
import pytest
import torch

from realhf_amd.apps.quickstart import parse_cli
from realhf_amd.base.testing import LocalMultiProcessTest


def _write_sft_data(path, n=16, vocab=64):
    rng = np.random.RandomState(0)
    with open(path, "w") as f:
        for _ in range(n):
            rec = {
                "prompt_ids": rng.randint(0, vocab, size=rng.randint(3, 6)).tolist(),
                "answer_ids": rng.randint(0, vocab, size=rng.randint(4, 9)).tolist(),
            }
            f.write(json.dumps(rec) + "\n")


def _write_prompt_data(path, n=16, vocab=64):
    rng = np.random.RandomState(1)
    with open(path, "w") as f:
        for _ in range(n):
            rec = {"input_ids": rng.randint(3, vocab - 3, size=rng.randint(4, 8)).tolist()}
            f.write(json.dumps(rec) + "\n")


def _write_rw_data(path, n=16, vocab=64):
    rng = np.random.RandomState(2)
    with open(path, "w") as f:
        for _ in range(n):
            rec = {
                "pos_ids": rng.randint(0, vocab, size=rng.randint(5, 10)).tolist(),
                "neg_ids": rng.randint(0, vocab, size=rng.randint(5, 10)).tolist(),
            }
            f.write(json.dumps(rec) + "\n")


def test_cli_override_parsing():
    exp, cfg = parse_cli([
        "ppo", "experiment_name=x", "n_gpus=4",
        "ppo.gen.max_new_tokens=64", "ppo.kl_ctl=0.05",
        "actor.optimizer.lr=0.0001", "exp_ctrl.save_freq_steps=null",
        "dataset.train_bs_n_seqs=8", "actor.gradient_checkpointing=true",
    ])
    assert exp == "ppo"
    assert cfg.n_gpus == 4
    assert cfg.ppo.gen.max_new_tokens == 64
    assert cfg.ppo.kl_ctl == 0.05
    # late round-2 fields stay CLI-drivable
    _, c2 = parse_cli(["ppo", "actor.optimizer.zero_stage=2",
                       "ppo.gen.force_no_logits_mask=false",
                       "ppo.value_norm_type=ma"])
    assert c2.actor.optimizer.zero_stage == 2
    assert c2.ppo.gen.force_no_logits_mask is False
    assert c2.ppo.value_norm_type == "ma"
    assert cfg.actor.optimizer.lr == 1e-4
    assert cfg.exp_ctrl.save_freq_steps is None
    assert cfg.actor.gradient_checkpointing is True


def test_sft_experiment_single_process(tmp_path):
    from realhf_amd.api.experiment import SFTConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "sft.jsonl")
    _write_sft_data(data)
    cfg = SFTConfig(
        experiment_name="t-sft", trial_name="cpu", n_gpus=1,
    )
    cfg.model.dtype = "float32"
    cfg.dataset.type_ = "prompt_answer"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.exp_ctrl.total_train_epochs = 1
    cfg.exp_ctrl.benchmark_steps = 2
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()


def test_ppo_experiment_single_process(tmp_path):
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "prompts.jsonl")
    _write_prompt_data(data)
    cfg = PPOConfig(experiment_name="t-ppo", trial_name="cpu", n_gpus=1)
    for mc in (cfg.actor, cfg.critic, cfg.ref, cfg.rew):
        mc.dtype = "float32"
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 6
    cfg.ppo.gen.min_new_tokens = 2
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 1
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()


def test_rw_experiment_with_save(tmp_path):
    from realhf_amd.api.experiment import RWConfig
    from realhf_amd.base import constants
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "rw.jsonl")
    _write_rw_data(data)
    cfg = RWConfig(experiment_name="t-rw", trial_name="cpu", n_gpus=1)
    cfg.model.dtype = "float32"
    cfg.dataset.type_ = "rw_paired"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.exp_ctrl.benchmark_steps = 1
    cfg.exp_ctrl.save_freq_steps = 1
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()
    root = constants.MODEL_SAVE_ROOT("t-rw", "cpu")
    saved = [d for d, _, fs in os.walk(root) if any(f.endswith(".safetensors") for f in fs)]
    assert saved, f"no checkpoint written under {root}"


def _sft_dist_worker(data, fileroot):
    from realhf_amd.api.experiment import SFTConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    cfg = SFTConfig(experiment_name="t-sft2", trial_name="dist", n_gpus=2)
    cfg.model.dtype = "float32"
    cfg.allocation_mode = "global"
    cfg.dataset.type_ = "prompt_answer"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.exp_ctrl.benchmark_steps = 2
    Trainer(cfg).run()


@pytest.mark.distributed
def test_sft_experiment_two_ranks(tmp_path):
    data = str(tmp_path / "sft.jsonl")
    _write_sft_data(data)
    LocalMultiProcessTest(2, _sft_dist_worker, data, str(tmp_path / "root")).launch()


def _ppo_realloc_worker(data, fileroot):
    """PPO with a DIFFERENT gen strategy for the actor -> exercises the
    realloc hooks through the executor."""
    from realhf_amd.api.config import ParallelismConfig
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    cfg = PPOConfig(experiment_name="t-ppo2", trial_name="dist", n_gpus=2)
    for mc in (cfg.actor, cfg.critic, cfg.ref, cfg.rew):
        mc.dtype = "float32"
        mc.parallel = ParallelismConfig(data_parallel_size=2)
    cfg.allocation_mode = "manual"
    cfg.actor.gen_parallel = ParallelismConfig(
        data_parallel_size=1, tensor_parallel_size=2
    )
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 5
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 1
    Trainer(cfg).run()


@pytest.mark.distributed
def test_ppo_realloc_two_ranks(tmp_path):
    data = str(tmp_path / "prompts.jsonl")
    _write_prompt_data(data)
    LocalMultiProcessTest(2, _ppo_realloc_worker, data, str(tmp_path / "root")).launch()


def test_sft_recover_resumes_state(tmp_path):
    """Kill-and-resume: second Trainer picks up global_step AND the saved
    model+optimizer state (reference: recover_mode=auto resume path)."""
    from realhf_amd.api.experiment import SFTConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "sft.jsonl")
    _write_sft_data(data, n=16)
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")

    def mkcfg():
        cfg = SFTConfig(experiment_name="t-rec", trial_name="cpu", n_gpus=1)
        cfg.model.dtype = "float32"
        cfg.dataset.type_ = "prompt_answer"
        cfg.dataset.path = data
        cfg.dataset.train_bs_n_seqs = 4
        cfg.exp_ctrl.total_train_epochs = 1
        cfg.exp_ctrl.benchmark_steps = 2
        cfg.exp_ctrl.save_freq_steps = 1  # recover ckpt every step
        cfg.recover_mode = "auto"
        return cfg

    t1 = Trainer(mkcfg())
    t1.run()
    step1 = t1.global_step
    name = t1.built.trainable[0]
    flat_after = t1.models[name].module.module.flat_param.detach().clone()
    opt_step = t1.models[name].module.optimizer.step_count
    assert step1 >= 2 and opt_step >= 2

    t2 = Trainer(mkcfg())
    recov = t2._maybe_load_recover()
    assert recov is not None
    assert t2.global_step == step1
    m2 = t2.models[name].module.module
    torch.testing.assert_close(m2.flat_param, flat_after)
    assert t2.models[name].module.optimizer.step_count == opt_step


def test_sft_eval_split(tmp_path):
    from realhf_amd.api.experiment import SFTConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "sft.jsonl")
    valid = str(tmp_path / "valid.jsonl")
    _write_sft_data(data, n=16)
    _write_sft_data(valid, n=8)
    cfg = SFTConfig(experiment_name="t-eval", trial_name="cpu", n_gpus=1)
    cfg.model.dtype = "float32"
    cfg.dataset.type_ = "prompt_answer"
    cfg.dataset.path = data
    cfg.dataset.valid_path = valid
    cfg.dataset.train_bs_n_seqs = 4
    cfg.exp_ctrl.total_train_epochs = 1
    cfg.exp_ctrl.benchmark_steps = 2
    cfg.exp_ctrl.eval_freq_steps = 1
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    t = Trainer(cfg)
    t.run()
    stats = t.evaluate()
    assert "default/eval_loss" in stats
    assert stats["default/eval_loss"] > 0


def _ppo_mixed4_worker(data, fileroot):
    """4 ranks, mixed strategies per role: actor trains d2t2, generates
    d1t2p2-like? -> keep pp out: gen d4t1, ref d2t2, rew d4 — exercises
    realloc across tp<->dp remaps and partial-participation shards at a
    world size above 2 (insurance for the 8-GPU scale run)."""
    from realhf_amd.api.config import ParallelismConfig
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    cfg = PPOConfig(experiment_name="t-ppo4", trial_name="dist", n_gpus=4)
    for mc in (cfg.actor, cfg.critic, cfg.ref, cfg.rew):
        mc.dtype = "float32"
    cfg.allocation_mode = "manual"
    cfg.actor.parallel = ParallelismConfig(data_parallel_size=2,
                                           tensor_parallel_size=2)
    cfg.actor.gen_parallel = ParallelismConfig(data_parallel_size=4)
    cfg.critic.parallel = ParallelismConfig(data_parallel_size=4)
    cfg.ref.parallel = ParallelismConfig(data_parallel_size=2,
                                         tensor_parallel_size=2)
    cfg.rew.parallel = ParallelismConfig(data_parallel_size=4)
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 8
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 5
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 1
    Trainer(cfg).run()


@pytest.mark.distributed
def test_ppo_mixed_strategies_four_ranks(tmp_path):
    data = str(tmp_path / "prompts.jsonl")
    _write_prompt_data(data, n=16)
    LocalMultiProcessTest(4, _ppo_mixed4_worker, data,
                          str(tmp_path / "root")).launch()


def test_dpo_experiment_single_process(tmp_path):
    from realhf_amd.api.experiment import DPOConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "rw.jsonl")
    _write_rw_data(data)
    cfg = DPOConfig(experiment_name="t-dpo", trial_name="cpu", n_gpus=1)
    cfg.actor.dtype = "float32"
    cfg.ref.dtype = "float32"
    cfg.dataset.type_ = "rw_paired"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.exp_ctrl.benchmark_steps = 2
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()


def test_grpo_experiment_single_process(tmp_path):
    from realhf_amd.api.experiment import GRPOConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "prompts.jsonl")
    _write_prompt_data(data)
    cfg = GRPOConfig(experiment_name="t-grpo", trial_name="cpu", n_gpus=1)
    for mc in (cfg.actor, cfg.ref, cfg.rew):
        mc.dtype = "float32"
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.group_size = 2
    cfg.ppo.gen.max_new_tokens = 6
    cfg.ppo.gen.min_new_tokens = 2
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 1
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()


def test_gen_experiment_single_process(tmp_path):
    from realhf_amd.api.experiment import GenerationConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "prompts.jsonl")
    _write_prompt_data(data)
    cfg = GenerationConfig(experiment_name="t-gen", trial_name="cpu", n_gpus=1)
    cfg.model.dtype = "float32"
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.gen.max_new_tokens = 6
    cfg.gen.use_hip_graph = False
    cfg.exp_ctrl.benchmark_steps = 1
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()
    # output_file dump (reference gen_exp.py:49): one jsonl per dp rank
    from realhf_amd.base import constants

    root = constants.LOG_ROOT("t-gen", "cpu")
    out = os.path.join(root, "output.jsonl.rank0")
    assert os.path.exists(out), os.listdir(root)
    recs = [json.loads(l) for l in open(out)]
    assert len(recs) > 0 and len(recs) % 4 == 0  # multiples of the batch
    assert all("prompt_ids" in r and "answer_ids" in r for r in recs)


def test_sft_training_deterministic(tmp_path):
    """Two identical runs produce (near-)identical params.  What SPMD
    actually requires is identical CONTROL FLOW on every rank (ids,
    shapes, schedules) — floating-point accumulation order inside the
    parallel autograd engine is not bitwise-stable, so params are
    compared tightly rather than bitwise."""
    from realhf_amd.api.experiment import SFTConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "sft.jsonl")
    _write_sft_data(data, n=16)
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")

    def run(trial):
        cfg = SFTConfig(experiment_name="t-det", trial_name=trial, n_gpus=1)
        cfg.model.dtype = "float32"
        cfg.dataset.type_ = "prompt_answer"
        cfg.dataset.path = data
        cfg.dataset.train_bs_n_seqs = 4
        cfg.exp_ctrl.benchmark_steps = 2
        t = Trainer(cfg)
        t.run()
        name = t.built.trainable[0]
        return t.models[name].module.module.flat_param.clone()

    a, b = run("a"), run("b")
    torch.testing.assert_close(a, b, atol=1e-6, rtol=1e-6)


def _ppo_dpshard_worker(data, fileroot):
    """ZeRO-3-style frozen-weight sharding wired through the OffloadHook:
    rew/ref release to 1/dp shards after their MFC and restore with one
    all-gather at the next step (to="dp_shard")."""
    from realhf_amd.api.config import ModelName
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    cfg = PPOConfig(experiment_name="t-ppods", trial_name="dist", n_gpus=2)
    for mc in (cfg.actor, cfg.critic, cfg.ref, cfg.rew):
        mc.dtype = "float32"
    cfg.allocation_mode = "global"
    cfg.ref.offload = "dp_shard"
    cfg.rew.offload = "dp_shard"
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 5
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 2  # 2 steps: shard -> gather -> shard
    t = Trainer(cfg)
    t.run()
    # after the last rew_inf post-hook the frozen models are sharded
    rew = t.models[ModelName("rew", 0)].module.module
    assert rew._dp_sharded and rew.flat_param is None


@pytest.mark.distributed
def test_ppo_dp_shard_offload_two_ranks(tmp_path):
    data = str(tmp_path / "prompts.jsonl")
    _write_prompt_data(data)
    LocalMultiProcessTest(2, _ppo_dpshard_worker, data,
                          str(tmp_path / "root")).launch()


def _sft_zero2_worker(data, fileroot):
    """SFT at dp2 with optimizer.zero_stage=2 from the experiment config
    (ZeRO-2: no full-model grad buffer; engine drives end_microbatch)."""
    from realhf_amd.api.experiment import SFTConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    cfg = SFTConfig(experiment_name="t-sft-z2", trial_name="dist", n_gpus=2)
    cfg.model.dtype = "float32"
    cfg.model.optimizer.zero_stage = 2
    cfg.allocation_mode = "global"
    cfg.dataset.type_ = "prompt_answer"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.exp_ctrl.benchmark_steps = 2
    Trainer(cfg).run()


@pytest.mark.distributed
def test_sft_zero2_two_ranks(tmp_path):
    data = str(tmp_path / "sft.jsonl")
    _write_sft_data(data)
    LocalMultiProcessTest(2, _sft_zero2_worker, data,
                          str(tmp_path / "root")).launch()


def test_save_eval_freq_epochs_and_secs(tmp_path):
    """Epoch- and seconds-frequency save/eval triggers (reference:
    system_api.py:157 ExperimentSaveEvalControl + EpochStepTimeFreqCtl)."""
    from realhf_amd.api.experiment import SFTConfig
    from realhf_amd.base import constants
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "sft.jsonl")
    _write_sft_data(data)
    cfg = SFTConfig(experiment_name="t-sft-freq", trial_name="cpu", n_gpus=1)
    cfg.model.dtype = "float32"
    cfg.dataset.type_ = "prompt_answer"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 8
    cfg.exp_ctrl.total_train_epochs = 2
    cfg.exp_ctrl.save_freq_epochs = 1
    cfg.exp_ctrl.eval_freq_secs = 0.0  # fires every step
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()
    # epoch-frequency saves landed
    save_root = os.path.join(constants.MODEL_SAVE_ROOT("t-sft-freq", "cpu"))
    assert os.path.isdir(save_root) and os.listdir(save_root)


def _ppo_z2_mask_worker(data, fileroot):
    """PPO at dp2 composing the late round-2 features: ZeRO-2 actors,
    logits-mask mode, and the tp2 gen replica with param realloc."""
    from realhf_amd.api.config import ParallelismConfig
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    cfg = PPOConfig(experiment_name="t-ppo-z2m", trial_name="dist", n_gpus=2)
    for mc in (cfg.actor, cfg.critic, cfg.ref, cfg.rew):
        mc.dtype = "float32"
        mc.parallel = ParallelismConfig(data_parallel_size=2)
    cfg.actor.optimizer.zero_stage = 2
    cfg.critic.optimizer.zero_stage = 2
    cfg.allocation_mode = "manual"
    cfg.actor.gen_parallel = ParallelismConfig(
        data_parallel_size=1, tensor_parallel_size=2
    )
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 5
    cfg.ppo.gen.top_k = 4
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.gen.force_no_logits_mask = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 2
    Trainer(cfg).run()


@pytest.mark.distributed
def test_ppo_zero2_logits_mask_two_ranks(tmp_path):
    data = str(tmp_path / "prompts.jsonl")
    _write_prompt_data(data)
    LocalMultiProcessTest(2, _ppo_z2_mask_worker, data,
                          str(tmp_path / "root")).launch()


def _sft_dp2pp2_worker(data, fileroot):
    """4-rank SFT at dp2 x pp2 through the full trainer — the config
    class that exposed the unarmed-overlap crash (ZeRO-1 bucketed RS +
    PipelinedEngine); runs both zero stages."""
    from realhf_amd.api.config import ParallelismConfig
    from realhf_amd.api.experiment import SFTConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    for stage in (1, 2):
        cfg = SFTConfig(experiment_name=f"t-sft-dp2pp2-z{stage}",
                        trial_name="dist", n_gpus=4)
        cfg.model.dtype = "float32"
        cfg.model.parallel = ParallelismConfig(
            data_parallel_size=2, pipeline_parallel_size=2)
        cfg.model.optimizer.zero_stage = stage
        cfg.allocation_mode = "manual"
        cfg.dataset.type_ = "prompt_answer"
        cfg.dataset.path = data
        cfg.dataset.train_bs_n_seqs = 4
        cfg.exp_ctrl.benchmark_steps = 2
        Trainer(cfg).run()


@pytest.mark.distributed
def test_sft_dp2_pp2_two_stages(tmp_path):
    data = str(tmp_path / "sft.jsonl")
    _write_sft_data(data)
    LocalMultiProcessTest(4, _sft_dp2pp2_worker, data,
                          str(tmp_path / "root")).launch()
