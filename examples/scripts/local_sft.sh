#!/bin/bash
# SFT on one node (reference: examples/scripts/local/sft.sh)
python3 -m realhf_amd.apps.quickstart sft \
    experiment_name=quickstart-sft trial_name=${TRIAL_NAME:-t0} \
    n_gpus=${N_GPUS:-8} \
    model.family=llama model.path=${MODEL_PATH:?HF checkpoint} \
    dataset.type_=prompt_answer dataset.path=${DATASET:?jsonl} \
    dataset.max_seqlen=1024 dataset.train_bs_n_seqs=256 \
    model.optimizer.lr=1e-5 exp_ctrl.total_train_epochs=2 \
    exp_ctrl.save_freq_steps=50
