#!/bin/bash
# Mixtral GRPO with expert parallelism over the node's xGMI mesh
# (BASELINE config #5; EP is a capability beyond the reference).
python3 -m realhf_amd.apps.quickstart grpo \
    experiment_name=quickstart-grpo trial_name=${TRIAL_NAME:-t0} \
    n_gpus=8 allocation_mode=global group_size=4 \
    actor.family=mixtral actor.path=${MIXTRAL_PATH:?HF mixtral ckpt} \
    ref.family=mixtral ref.path=$MIXTRAL_PATH \
    rew.family=llama rew.is_critic=true rew.path=${RW_MODEL_PATH:?} \
    dataset.path=${DATASET:?prompts jsonl} dataset.train_bs_n_seqs=64 \
    ppo.gen.max_new_tokens=512
# EP degree is cfg.moe.expert_parallel_size on the model config (set via
# a custom experiment or by editing the mixtral config); see
# tests/test_moe_ep.py::test_grpo_mixtral_ep_experiment for the wiring.
