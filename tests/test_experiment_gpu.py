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
    # Kernel-eligible dims (hidden/inter % 64) so the MoE experts run
    # the MFMA grouped path end-to-end.  EVERY role must share the
    # actor's vocab: generated token ids feed the reward/ref embeddings,
    # and an out-of-vocab id is a device-side OOB gather (HW exception —
    # took a kernel-serialized hunt to find; tools/repro_grpo_e2e.py).
    import realhf_amd.models.hf as hf_reg
    from realhf_amd.runtime import trainer as T

    orig = T.build_experiment

    def patched(c, world):
        built = orig(c, world)
        for name, rcfg in built.model_cfgs.items():
            fam = hf_reg.get_family(rcfg.family or "llama")
            big = fam.make_test_config(
                n_layers=2, hidden_dim=64, n_heads=1, n_kv_heads=1,
                head_dim=64, intermediate_dim=128, vocab_size=128)
            big.is_critic = rcfg.is_critic
            big.dtype = rcfg.dtype
            big.family = rcfg.family
            built.model_cfgs[name] = big
        return built

    T.build_experiment = patched
    try:
        Trainer(cfg).run()
    finally:
        T.build_experiment = orig


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


def test_llama70b_fits_one_gpu_forward():
    """288 GB HBM sizing tier: the whole 70B fits one MI355X in bf16 for
    inference; forward 512 packed tokens + a short decode."""
    from realhf_amd.api.model import GenerationHyperparameters
    from realhf_amd.models.generation import generate
    from realhf_amd.models.hf.llama import llama70b_config
    from realhf_amd.models.real_model import ReaLModel

    cfg = llama70b_config()
    try:
        m = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
    except torch.OutOfMemoryError:
        pytest.skip("box lacks free HBM for the 70B tier test")
    m.random_init()
    toks = torch.randint(0, cfg.vocab_size, (512,), device="cuda")
    cu = torch.tensor([0, 256, 512], dtype=torch.int32, device="cuda")
    with torch.no_grad():
        logits = m(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=256)
    assert logits.shape == (512, cfg.vocab_size)
    out = generate(
        m, toks[:64], torch.tensor([0, 32, 64], dtype=torch.int32, device="cuda"),
        GenerationHyperparameters(max_new_tokens=8, greedy=True),
    )
    assert out.gen_tokens.shape[0] == 2
    del m
    torch.cuda.empty_cache()


def test_tiny_ppo_with_rccl_groups(tmp_path):
    """Single-rank RCCL process group: exercises grid construction, the
    executor's gathers and ZeRO collectives on the nccl(=RCCL) backend —
    the same code the 8-GPU scale run executes."""
    import torch.distributed as dist

    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29719")
    torch.cuda.set_device(0)
    created = False
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
        created = True
    try:
        rng = np.random.RandomState(3)
        data = str(tmp_path / "p.jsonl")
        with open(data, "w") as f:
            for _ in range(8):
                f.write(json.dumps(
                    {"input_ids": rng.randint(3, 60, size=8).tolist()}) + "\n")
        cfg = PPOConfig(experiment_name="t-ppo-rccl", trial_name="g", n_gpus=1)
        cfg.dataset.type_ = "prompt"
        cfg.dataset.path = data
        cfg.dataset.train_bs_n_seqs = 4
        cfg.dataset.max_prompt_len = 8
        cfg.ppo.gen.max_new_tokens = 8
        cfg.ppo.gen.use_hip_graph = False
        cfg.ppo.ppo_n_minibatches = 2
        cfg.exp_ctrl.benchmark_steps = 1
        os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
        Trainer(cfg).run()
    finally:
        if created:
            dist.destroy_process_group()
        from realhf_amd.base import constants
        from realhf_amd.base.topology import clear_group_cache

        constants.clear_grids()
        clear_group_cache()


def test_tiny_grpo_mixtral_gpu(tmp_path):
    """BASELINE config #5 end-to-end ON THE HIP PATH: Mixtral-style GRPO
    (bf16, grouped-GEMM experts, native attention/decode kernels) for
    two steps on one GPU."""
    from realhf_amd.api.experiment import GRPOConfig
    from realhf_amd.runtime.trainer import Trainer

    rng = np.random.RandomState(7)
    data = str(tmp_path / "p.jsonl")
    with open(data, "w") as f:
        for _ in range(8):
            f.write(json.dumps(
                {"input_ids": rng.randint(3, 60, size=8).tolist()}) + "\n")
    cfg = GRPOConfig(experiment_name="t-grpo-gpu", trial_name="g", n_gpus=1)
    for mc in (cfg.actor, cfg.ref):
        mc.family = "mixtral"
    cfg.rew.family = "llama"
    cfg.rew.is_critic = True
    cfg.group_size = 2
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 8
    cfg.ppo.gen.min_new_tokens = 2
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 2
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    # Kernel-eligible dims (hidden/inter % 64) so the MoE experts run
    # the MFMA grouped path end-to-end.  EVERY role must share the
    # actor's vocab: generated token ids feed the reward/ref embeddings,
    # and an out-of-vocab id is a device-side OOB gather (HW exception —
    # took a kernel-serialized hunt to find; tools/repro_grpo_e2e.py).
    import realhf_amd.models.hf as hf_reg
    from realhf_amd.runtime import trainer as T

    orig = T.build_experiment

    def patched(c, world):
        built = orig(c, world)
        for name, rcfg in built.model_cfgs.items():
            fam = hf_reg.get_family(rcfg.family or "llama")
            big = fam.make_test_config(
                n_layers=2, hidden_dim=64, n_heads=1, n_kv_heads=1,
                head_dim=64, intermediate_dim=128, vocab_size=128)
            big.is_critic = rcfg.is_critic
            big.dtype = rcfg.dtype
            big.family = rcfg.family
            built.model_cfgs[name] = big
        return built

    T.build_experiment = patched
    try:
        Trainer(cfg).run()
    finally:
        T.build_experiment = orig


def test_ppo_logits_mask_gpu(tmp_path):
    """PPO with gen.force_no_logits_mask=False on the HIP path: the
    sampler's top-k mask is recorded outside the decode hipGraph and
    flows to ref_inf/actor_train."""
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    rng = np.random.RandomState(4)
    data = str(tmp_path / "prompts.jsonl")
    with open(data, "w") as f:
        for _ in range(16):
            rec = {"input_ids": rng.randint(3, 60, size=rng.randint(6, 10)).tolist()}
            f.write(json.dumps(rec) + "\n")
    cfg = PPOConfig(experiment_name="t-ppo-lmask-gpu", trial_name="g", n_gpus=1)
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 8
    cfg.dataset.max_prompt_len = 10
    cfg.ppo.gen.max_new_tokens = 16
    cfg.ppo.gen.min_new_tokens = 4
    cfg.ppo.gen.top_k = 8  # binds at vocab 128 -> non-trivial mask
    cfg.ppo.gen.force_no_logits_mask = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 2
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    import realhf_amd.models.hf as hf_reg
    from realhf_amd.runtime import trainer as T

    orig = T.build_experiment

    def patched(c, world):
        built = orig(c, world)
        for name, rcfg in built.model_cfgs.items():
            fam = hf_reg.get_family(rcfg.family or "llama")
            big = fam.make_test_config(
                n_layers=2, hidden_dim=64, n_heads=1, n_kv_heads=1,
                head_dim=64, intermediate_dim=128, vocab_size=128)
            big.is_critic = rcfg.is_critic
            big.dtype = rcfg.dtype
            big.family = rcfg.family
            built.model_cfgs[name] = big
        return built

    T.build_experiment = patched
    try:
        Trainer(cfg).run()
    finally:
        T.build_experiment = orig
