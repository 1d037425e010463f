#!/usr/bin/env python3
"""REINFORCE-with-baseline as USER code (reference parity:
examples/new_algorithms/) — adds an algorithm WITHOUT touching the
framework: a custom ModelInterface + the standard engine/backends.

Runs on CPU in seconds:  python examples/new_algorithms/reinforce.py
"""
import dataclasses
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))
import torch

import realhf_amd.runtime.engine  # noqa: F401 — registers backends
import realhf_amd.models.hf as hf_reg
from realhf_amd.api.config import Abstraction, ModelName
from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import (
    FinetuneSpec,
    GenerationHyperparameters,
    Model,
    ModelInterface,
    make_backend,
    register_interface,
)
from realhf_amd.models.real_model import ReaLModel
from realhf_amd.utils.functional import gather_packed_shifted_log_probs


@dataclasses.dataclass
class ReinforceInterface(ModelInterface):
    """REINFORCE: maximize logp(sampled tokens) * (reward - baseline)."""

    baseline_ema: float = 0.9
    _baseline: float = 0.0

    def train_step(self, model: Model, data: SequenceSample, n_mbs=None):
        rewards = data.data["rewards"].float()
        adv = rewards - self._baseline
        self._baseline = (self.baseline_ema * self._baseline
                          + (1 - self.baseline_ema) * float(rewards.mean()))
        loose = {"adv": adv}

        def loss_fn(logits, mb):
            ids, cu = mb.data["packed_input_ids"], None
            import torch as T

            lens = [x for l in mb.seqlens["packed_input_ids"] for x in l]
            cu = T.zeros(len(lens) + 1, dtype=T.int32)
            cu[1:] = T.cumsum(T.tensor(lens), 0)
            logp = gather_packed_shifted_log_probs(logits, cu, ids)
            # per-sequence sum of logp, weighted by advantage
            seq_lp = []
            off = 0
            for L in lens:
                seq_lp.append(logp[off:off + L - 1].sum())
                off += L - 1
            seq_lp = T.stack(seq_lp)
            loss = -(seq_lp * loose["adv"][: len(lens)]).mean()
            return loss, {"loss": float(loss), "reward": float(rewards.mean())}

        return model.module.train_batch(data, loss_fn, n_mbs=n_mbs)


register_interface("reinforce", ReinforceInterface)


def main():
    torch.manual_seed(0)
    cfg = hf_reg.get_family("llama").make_test_config()
    cfg.dtype = "float32"
    m = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    m.random_init()
    model = Model(ModelName("actor", 0), m, None, torch.device("cpu"),
                  torch.float32)
    model = make_backend(
        Abstraction("zero1", {"optimizer": {"lr": 5e-3,
                                            "warmup_steps_proportion": 0.0,
                                            "lr_scheduler_type": "constant"}})
    ).initialize(model, FinetuneSpec(1, 64, 4))
    iface = ReinforceInterface()

    gconfig = GenerationHyperparameters(max_new_tokens=8, greedy=False,
                                        top_k=20, use_hip_graph=False)
    gen = torch.Generator().manual_seed(1)
    for step in range(5):
        # sample prompts, generate, reward = fraction of even tokens
        prompts = [torch.randint(0, cfg.vocab_size, (5,)) for _ in range(4)]
        cu = torch.tensor([0, 5, 10, 15, 20], dtype=torch.int32)
        from realhf_amd.models.generation import (
            concat_prompt_to_generation_output, generate)

        out = generate(m, torch.cat(prompts), cu, gconfig, eos_token_id=None,
                       generator=gen)
        seqs, cu_out, _ = concat_prompt_to_generation_output(
            torch.cat(prompts), cu, out)
        seqlens = (cu_out[1:] - cu_out[:-1]).tolist()
        rewards = torch.tensor(
            [float((s % 2 == 0).float().mean()) for s in
             torch.split(seqs, seqlens)])
        batch = SequenceSample(
            keys=("packed_input_ids", "rewards"),
            ids=[f"r{step}-{i}" for i in range(4)],
            seqlens={"packed_input_ids": [[l] for l in seqlens],
                     "rewards": [[1]] * 4},
            data={"packed_input_ids": seqs, "rewards": rewards},
        )
        stats = iface.train_step(model, batch)
        print(f"step {step}: loss={stats['loss']:.4f} "
              f"reward={stats['reward']:.3f}")


if __name__ == "__main__":
    main()
