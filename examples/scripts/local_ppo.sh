#!/bin/bash
# PPO on one MI355X node (mirror of the reference examples/scripts/local/ppo.sh).
# SFT_MODEL_PATH / RW_MODEL_PATH are HF-format checkpoints (e.g. saved by
# the sft / rw experiments below — realhf_amd saves HF format directly).
MODEL_FAMILY=llama
SFT_MODEL_PATH=${SFT_MODEL_PATH:?set to an HF checkpoint dir}
RW_MODEL_PATH=${RW_MODEL_PATH:?set to an HF checkpoint dir}

python3 -m realhf_amd.apps.quickstart ppo \
    experiment_name=quickstart-ppo trial_name=${TRIAL_NAME:-t0} \
    n_gpus=8 allocation_mode=heuristic \
    actor.family=$MODEL_FAMILY actor.path=$SFT_MODEL_PATH \
    critic.family=$MODEL_FAMILY critic.is_critic=true critic.path=$RW_MODEL_PATH \
    ref.family=$MODEL_FAMILY ref.path=$SFT_MODEL_PATH \
    rew.family=$MODEL_FAMILY rew.is_critic=true rew.path=$RW_MODEL_PATH \
    dataset.path=${DATASET:-.data/ppo_prompt.jsonl} \
    dataset.max_prompt_len=128 dataset.train_bs_n_seqs=128 \
    ppo.gen.max_new_tokens=512 ppo.gen.min_new_tokens=512 \
    ppo.gen.top_p=0.9 ppo.gen.top_k=1000 \
    ppo.ppo_n_minibatches=4 ppo.kl_ctl=0.1
