"""End-to-end interface tests on CPU: SFT learns, PPO/RW/DPO/GRPO step
through all MFCs single-process (reference: config #1 'GPT-2 small SFT on
CPU' milestone, SURVEY.md §7 stage 4)."""
import numpy as np
import pytest
import torch

import realhf_amd.interfaces  # noqa: F401 — registers interfaces
import realhf_amd.models.hf as hf_reg
import realhf_amd.runtime.engine  # noqa: F401 — registers backends
from realhf_amd.api.config import Abstraction, ModelName
from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import (
    FinetuneSpec,
    GenerationHyperparameters,
    Model,
    make_backend,
    make_interface,
)
from realhf_amd.models.real_model import ReaLModel


def make_model(family="llama", is_critic=False, seed=0, **cfg_kw):
    fam = hf_reg.get_family(family)
    cfg = fam.make_test_config(is_critic=is_critic, **cfg_kw)
    cfg.dtype = "float32"
    cfg.family = family
    m = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    torch.manual_seed(seed)
    m.random_init()
    return Model(
        name=ModelName("test", 0), module=m, tokenizer=None,
        device=torch.device("cpu"), dtype=torch.float32,
    )


def init_train(model, lr=1e-3):
    backend = make_backend(
        Abstraction("zero1", {"optimizer": {"lr": lr, "warmup_steps_proportion": 0.0,
                                            "lr_scheduler_type": "constant"}})
    )
    return backend.initialize(model, FinetuneSpec(1, 64, 8))


def init_inf(model):
    return make_backend(Abstraction("inference")).initialize(
        model, FinetuneSpec(1, 64, 8)
    )


def sft_batch(vocab, bs=8, seed=0):
    rng = np.random.RandomState(seed)
    samples = []
    for i in range(bs):
        pl = rng.randint(3, 6)
        al = rng.randint(4, 10)
        toks = torch.from_numpy(rng.randint(0, vocab, size=pl + al)).long()
        pm = torch.zeros(pl + al, dtype=torch.bool)
        pm[:pl] = True
        samples.append(
            SequenceSample(
                keys=("packed_input_ids", "prompt_mask"),
                ids=[f"s{i}"],
                seqlens={"packed_input_ids": [[pl + al]], "prompt_mask": [[pl + al]]},
                data={"packed_input_ids": toks, "prompt_mask": pm},
            )
        )
    return SequenceSample.gather(samples)


def test_sft_learns():
    model = make_model("gpt2")
    model = init_train(model, lr=5e-3)
    iface = make_interface(Abstraction("sft"))
    batch = sft_batch(model.module.module.config.vocab_size)
    losses = []
    for _ in range(8):
        stats = iface.train_step(model, batch, n_mbs=2)
        losses.append(stats["loss"])
    assert losses[-1] < losses[0] * 0.8, losses


def prompt_batch(vocab, bs=4, seed=1):
    rng = np.random.RandomState(seed)
    samples = []
    for i in range(bs):
        l = rng.randint(4, 8)
        samples.append(
            SequenceSample(
                keys=("packed_prompts",),
                ids=[f"p{i}"],
                seqlens={"packed_prompts": [[l]]},
                data={"packed_prompts": torch.from_numpy(
                    rng.randint(3, vocab - 3, size=l)).long()},
            )
        )
    return SequenceSample.gather(samples)


def test_ppo_full_graph_cpu():
    """actor_gen -> {rew, ref, critic} -> {actor_train, critic_train}."""
    actor = init_train(make_model("llama", seed=1))
    ref = init_inf(make_model("llama", seed=1))
    critic = init_train(make_model("llama", is_critic=True, seed=2))
    rew = init_inf(make_model("llama", is_critic=True, seed=3))

    gconfig = GenerationHyperparameters(
        max_new_tokens=8, min_new_tokens=2, greedy=False, top_k=20,
        use_hip_graph=False,
    )
    actor_iface = make_interface(
        Abstraction("ppo_actor", {"n_minibatches": 2, "gconfig": gconfig.__dict__})
    )
    critic_iface = make_interface(Abstraction("ppo_critic", {"n_minibatches": 2}))
    rew_iface = make_interface(Abstraction("paired_rw"))

    vocab = actor.module.module.config.vocab_size
    batch = prompt_batch(vocab)

    rollout = actor_iface.generate(actor, batch)
    assert "packed_input_ids" in rollout.keys
    total = sum(s[0] for s in rollout.seqlens["packed_input_ids"])
    assert rollout.data["packed_input_ids"].shape[0] == total
    assert rollout.data["packed_logprobs"].shape[0] == total - rollout.bs

    sample = rollout
    rew_out = rew_iface.inference(rew, sample.select_keys(["packed_input_ids"]))
    sample.update_(rew_out)
    ref_out = actor_iface.inference(ref, sample.select_keys(["packed_input_ids"]))
    sample.update_(ref_out)
    critic_out = critic_iface.inference(
        critic, sample.select_keys(["packed_input_ids"])
    )
    sample.update_(critic_out)

    a_stats = actor_iface.train_step(actor, sample)
    c_stats = critic_iface.train_step(critic, sample)
    assert np.isfinite(a_stats["actor_loss"])
    assert np.isfinite(c_stats["critic_loss"])
    assert "kl" in a_stats


def test_rw_pair_learning():
    model = init_train(make_model("llama", is_critic=True, seed=5), lr=5e-3)
    iface = make_interface(Abstraction("paired_rw"))
    rng = np.random.RandomState(0)
    vocab = model.module.module.config.vocab_size
    samples = []
    for i in range(8):
        pos = rng.randint(0, 10, size=rng.randint(4, 8))  # pos: low token ids
        neg = rng.randint(vocab - 10, vocab, size=rng.randint(4, 8))
        toks = torch.tensor(list(pos) + list(neg), dtype=torch.long)
        samples.append(
            SequenceSample(
                keys=("packed_input_ids",),
                ids=[f"r{i}"],
                seqlens={"packed_input_ids": [[len(pos), len(neg)]]},
                data={"packed_input_ids": toks},
            )
        )
    batch = SequenceSample.gather(samples)
    accs = []
    for _ in range(10):
        st = iface.train_step(model, batch)
        accs.append(st["acc"])
    assert accs[-1] >= 0.8, accs


def test_dpo_step():
    actor = init_train(make_model("llama", seed=7))
    ref = init_inf(make_model("llama", seed=7))
    iface = make_interface(Abstraction("dpo"))
    rng = np.random.RandomState(2)
    vocab = actor.module.module.config.vocab_size
    samples = []
    for i in range(4):
        pl = rng.randint(2, 4)
        p = list(rng.randint(0, vocab, size=pl))
        pos = p + list(rng.randint(0, vocab, size=rng.randint(3, 6)))
        neg = p + list(rng.randint(0, vocab, size=rng.randint(3, 6)))
        pm = [True] * pl + [False] * (len(pos) - pl) + [True] * pl + [False] * (
            len(neg) - pl
        )
        samples.append(
            SequenceSample(
                keys=("packed_input_ids", "prompt_mask"),
                ids=[f"d{i}"],
                seqlens={
                    "packed_input_ids": [[len(pos), len(neg)]],
                    "prompt_mask": [[len(pos), len(neg)]],
                },
                data={
                    "packed_input_ids": torch.tensor(pos + neg, dtype=torch.long),
                    "prompt_mask": torch.tensor(pm, dtype=torch.bool),
                },
            )
        )
    batch = SequenceSample.gather(samples)
    ref_out = iface.inference(ref, batch)
    batch.update_(ref_out)
    st = iface.train_step(actor, batch)
    assert np.isfinite(st["loss"])


def test_grpo_step():
    actor = init_train(make_model("llama", seed=9))
    ref = init_inf(make_model("llama", seed=9))
    rew = init_inf(make_model("llama", is_critic=True, seed=10))
    gconfig = GenerationHyperparameters(
        max_new_tokens=6, greedy=False, top_k=20, use_hip_graph=False
    )
    iface = make_interface(
        Abstraction(
            "grpo",
            {"group_size": 2, "n_minibatches": 2, "gconfig": gconfig.__dict__,
             "kl_in_loss_coef": 0.1},
        )
    )
    rew_iface = make_interface(Abstraction("paired_rw"))
    actor_iface = make_interface(Abstraction("ppo_actor", {"gconfig": gconfig.__dict__}))

    vocab = actor.module.module.config.vocab_size
    batch = prompt_batch(vocab, bs=4, seed=11)
    rollout = iface.generate(actor, batch)
    assert rollout.bs == 8  # group_size * 4
    rew_out = rew_iface.inference(rew, rollout.select_keys(["packed_input_ids"]))
    rollout.update_(rew_out)
    ref_out = actor_iface.inference(ref, rollout.select_keys(["packed_input_ids"]))
    rollout.remap_keys_({})
    rollout.update_(ref_out)
    st = iface.train_step(actor, rollout)
    assert np.isfinite(st["actor_loss"])
    assert "kl_in_loss" in st


def test_grpo_group_norm_id_based():
    """Group advantages come from id prefixes, not positions: a shuffled
    or split batch (balanced DP split can cut groups across ranks) must
    still normalize within the right group; singletons get 0."""
    import torch

    from realhf_amd.api.data import SequenceSample
    from realhf_amd.interfaces.grpo import GRPOInterface

    iface = GRPOInterface(group_size=2, n_minibatches=1)
    # ids deliberately interleaved + one singleton (g1 of "b" elsewhere)
    ids = ["a@g0", "b@g0", "a@g1", "c@g0", "c@g1"]
    score = torch.tensor([1.0, 5.0, 3.0, 2.0, 4.0])
    lens = [4, 4, 4, 4, 4]
    toks = torch.randint(0, 32, (sum(lens),))
    pm = torch.zeros(sum(lens), dtype=torch.bool)
    data = SequenceSample(
        keys=("packed_input_ids", "packed_logprobs", "packed_ref_logprobs",
              "rewards", "prompt_mask"),
        ids=ids,
        seqlens={
            "packed_input_ids": [[l] for l in lens],
            "packed_logprobs": [[l - 1] for l in lens],
            "packed_ref_logprobs": [[l - 1] for l in lens],
            "rewards": [[1]] * 5,
            "prompt_mask": [[l] for l in lens],
        },
        data={
            "packed_input_ids": toks,
            "packed_logprobs": torch.zeros(sum(lens) - 5),
            "packed_ref_logprobs": torch.zeros(sum(lens) - 5),
            "rewards": score,
            "prompt_mask": pm,
        },
    )

    captured = {}

    class _FakeEngine:
        def train_batch(self, mb, loss_fn, **kw):
            captured["adv"] = mb.data["advantages"]
            captured["ids"] = list(mb.ids)
            return {}

    class _FakeModel:
        module = _FakeEngine()

        class version:
            global_step = 0

        @staticmethod
        def inc_version():
            pass

    iface.train_step(_FakeModel(), data)
    adv = captured["adv"]
    # per-token adv; group "a" = scores (1, 3): normalized to (-1, +1)/std
    # with ddof=1 std = sqrt(2): first seq tokens ~ -0.707
    seg = torch.repeat_interleave(torch.arange(5), torch.tensor([3, 3, 3, 3, 3]))
    per_seq = torch.stack([adv[seg == i][0] for i in range(5)])
    assert per_seq[1].abs() < 1e-4  # singleton "b@g0" -> 0
    assert torch.allclose(per_seq[0], -per_seq[2], atol=1e-5)  # group a pair
    assert torch.allclose(per_seq[3], -per_seq[4], atol=1e-5)  # group c pair
    assert per_seq[0] < 0 and per_seq[2] > 0


def test_value_norm_variants():
    """exp vs ma value normalization (reference: modules/rms.py:16/102,
    selected by value_norm_type, ppo_exp.py:62-68)."""
    from realhf_amd.interfaces import ppo_math

    x = torch.tensor([1.0, 2.0, 3.0, 4.0])
    ma = ppo_math.make_value_norm("ma")
    ma.update(x)
    m, s = ma.mean_std()
    assert abs(m - 2.5) < 1e-6
    # second update: cumulative average over ALL history, equal weights
    ma.update(torch.tensor([5.0, 6.0]))
    m2, _ = ma.mean_std()
    assert abs(m2 - (1 + 2 + 3 + 4 + 5 + 6) / 6) < 1e-6
    # round trip
    y = ma.normalize(x)
    torch.testing.assert_close(ma.denormalize(y), x, atol=1e-4, rtol=1e-4)
    # state dict round trip
    ma2 = ppo_math.make_value_norm("ma")
    ma2.load_state_dict(ma.state_dict())
    assert ma2.mean_std() == ma.mean_std()

    exp = ppo_math.make_value_norm("exp", beta=0.9)
    exp.update(x)
    m, s = exp.mean_std()
    assert abs(m - 2.5) < 1e-6  # debiased EMA after one update = batch mean

    # critic interface builds the right normalizer from config
    from realhf_amd.interfaces.ppo import PPOCriticInterface

    c = PPOCriticInterface(value_norm=True, value_norm_type="ma")
    assert isinstance(c._rms, ppo_math.MovingAverageRunningMeanStd)
    with pytest.raises(ValueError):
        PPOCriticInterface(value_norm=True, value_norm_type="bogus")


def test_reference_registry_name_aliases(tmp_path):
    """Reference configs use dataset "rw_pair" and backend "null" — both
    names must resolve here (drop-in config compatibility)."""
    import json

    import numpy as np

    import realhf_amd.api.datasets  # noqa: F401 (registers dataset names)
    from realhf_amd.api.data import make_dataset

    rng = np.random.RandomState(0)
    path = tmp_path / "rw.jsonl"
    with path.open("w") as f:
        for i in range(4):
            f.write(json.dumps({
                "prompt": "p",
                "pos_ids": rng.randint(0, 50, size=9).tolist(),
                "neg_ids": rng.randint(0, 50, size=9).tolist(),
            }) + "\n")
    ds = make_dataset(Abstraction("rw_pair", {"path": str(path),
                                              "max_seqlen": 32}),
                      seed=0, dp_rank=0, world_size=1)
    assert len(ds) == 4
    # "null" backend: minimal inference wrap
    model = make_model()
    wrapped = make_backend(Abstraction("null")).initialize(
        model, FinetuneSpec(1, 64, 8))
    assert wrapped.backend_name == "inference"
