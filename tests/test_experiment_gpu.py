"""Single-GPU end-to-end experiment runs (the executor + interfaces on
the HIP path)."""
import json
import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs GPU", allow_module_level=True)


def test_tiny_ppo_experiment_gpu(tmp_path):
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    rng = np.random.RandomState(1)
    data = str(tmp_path / "prompts.jsonl")
    with open(data, "w") as f:
        for _ in range(16):
            rec = {"input_ids": rng.randint(3, 60, size=rng.randint(6, 10)).tolist()}
            f.write(json.dumps(rec) + "\n")
    cfg = PPOConfig(experiment_name="t-ppo-gpu", trial_name="g", n_gpus=1)
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 8
    cfg.dataset.max_prompt_len = 10
    cfg.ppo.gen.max_new_tokens = 16
    cfg.ppo.gen.min_new_tokens = 4
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 2
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()


def test_tiny_sft_gpu(tmp_path):
    from realhf_amd.api.experiment import SFTConfig
    from realhf_amd.runtime.trainer import Trainer

    rng = np.random.RandomState(2)
    data = str(tmp_path / "sft.jsonl")
    with open(data, "w") as f:
        for _ in range(16):
            rec = {
                "prompt_ids": rng.randint(0, 60, size=4).tolist(),
                "answer_ids": rng.randint(0, 60, size=8).tolist(),
            }
            f.write(json.dumps(rec) + "\n")
    cfg = SFTConfig(experiment_name="t-sft-gpu", trial_name="g", n_gpus=1)
    cfg.dataset.type_ = "prompt_answer"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.exp_ctrl.benchmark_steps = 2
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()
