#!/usr/bin/env python3
"""PPO with an EMA reference policy (reference parity:
examples/customized_exp/ppo_ref_ema.py).

The reference model tracks the actor by exponential moving average —
expressed as a ParamReallocHook with eta < 1 attached AFTER actor_train:
    ref_params <- eta * actor_params + (1 - eta) * ref_params
No framework changes needed: build the standard PPO experiment, then
attach the hook to the built graph.

Runs on CPU:  python examples/customized_exp/ppo_ref_ema.py
"""
import json
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))


def main():
    import torch

    from realhf_amd.api.dfg import ParamReallocHook
    from realhf_amd.api.config import ModelName
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    tmp = tempfile.mkdtemp()
    data = os.path.join(tmp, "prompts.jsonl")
    with open(data, "w") as f:
        for i in range(8):
            f.write(json.dumps(
                {"input_ids": [3 + i % 5, 7, 11, 2 + i % 3]}) + "\n")
    os.environ["REALHF_AMD_FILEROOT"] = os.path.join(tmp, "root")

    cfg = PPOConfig(experiment_name="ppo-ref-ema", trial_name="demo", n_gpus=1)
    for mc in (cfg.actor, cfg.critic, cfg.ref, cfg.rew):
        mc.dtype = "float32"
    cfg.ref.offload = False  # the EMA hook updates ref in place
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 6
    cfg.ppo.gen.min_new_tokens = 2
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 2

    trainer = Trainer(cfg)
    # attach the EMA hook: after actor_train, push actor params into ref
    # with eta = 0.1 (ref <- 0.1 * actor + 0.9 * ref)
    for mfc in trainer.built.graph.mfcs:
        if mfc.name == "actor_train":
            mfc.post_hooks.append(
                ParamReallocHook(source=ModelName("actor", 0),
                                 target=ModelName("ref", 0), eta=0.1)
            )
    ref_before = trainer.models[ModelName("ref", 0)] \
        .module.module.flat_param.clone()
    trainer.run()
    ref_after = trainer.models[ModelName("ref", 0)] \
        .module.module.flat_param
    drift = (ref_after - ref_before).abs().max()
    print(f"ref EMA drift after 2 steps: {float(drift):.6f} (nonzero = EMA active)")
    assert drift > 0


if __name__ == "__main__":
    main()
