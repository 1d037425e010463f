#!/bin/bash
# 70B-tier PPO on one 8x MI355X node (BASELINE config #4): TP across the
# node for the trainable roles, gradient checkpointing, host-offloaded
# fp32 optimizer states (async chunked PCIe pipeline), and ZeRO-3-style
# DP sharding of the frozen ref/reward weights between their MFCs
# (restored with one xGMI all-gather — offload=dp_shard).
MODEL_FAMILY=llama
SFT_MODEL_PATH=${SFT_MODEL_PATH:?set to a 70B HF checkpoint dir}
RW_MODEL_PATH=${RW_MODEL_PATH:?set to a 70B HF checkpoint dir}

python3 -m realhf_amd.apps.quickstart ppo \
    experiment_name=quickstart-ppo-70b trial_name=${TRIAL_NAME:-t0} \
    n_gpus=8 allocation_mode=manual \
    actor.family=$MODEL_FAMILY actor.path=$SFT_MODEL_PATH \
    actor.parallel.tensor_parallel_size=4 actor.parallel.data_parallel_size=2 \
    actor.parallel.sequence_parallel=true actor.gradient_checkpointing=true \
    actor.optimizer.offload=true \
    actor.gen_parallel.data_parallel_size=8 \
    critic.family=$MODEL_FAMILY critic.is_critic=true critic.path=$RW_MODEL_PATH \
    critic.parallel.tensor_parallel_size=4 critic.parallel.data_parallel_size=2 \
    critic.parallel.sequence_parallel=true critic.gradient_checkpointing=true \
    critic.optimizer.offload=true \
    ref.family=$MODEL_FAMILY ref.path=$SFT_MODEL_PATH \
    ref.parallel.data_parallel_size=8 ref.offload=dp_shard \
    rew.family=$MODEL_FAMILY rew.is_critic=true rew.path=$RW_MODEL_PATH \
    rew.parallel.data_parallel_size=8 rew.offload=dp_shard \
    dataset.path=${DATASET:-.data/ppo_prompt.jsonl} \
    dataset.max_prompt_len=128 dataset.train_bs_n_seqs=64 \
    ppo.gen.max_new_tokens=512 ppo.gen.min_new_tokens=512 \
    ppo.ppo_n_minibatches=4 ppo.kl_ctl=0.1
