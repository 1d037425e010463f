#!/bin/bash
# Strategy-sweep profiling (reference: examples/profiling/profile.sh +
# profile_exp.py).  Runs the registered interfaces over a grid of
# parallel strategies with mock data of the given shape, and writes
# per-(interface, strategy) timings to
# $REALHF_AMD_FILEROOT/logs/.../profile_result.json.
#
# REALHF_AMD_DUMP_TRACE=1 additionally dumps a chrome trace per
# (mfc, rank, step); REALHF_AMD_DUMP_MEMORY=1 dumps allocator snapshots.
MODEL_FAMILY=${MODEL_FAMILY:-llama}

MODEL_PATH_ARG=""
[ -n "${SFT_MODEL_PATH:-}" ] && MODEL_PATH_ARG="model.path=$SFT_MODEL_PATH"

python -m realhf_amd.apps.quickstart profile \
    experiment_name=profile-example trial_name=t0 n_gpus=8 \
    model.family=$MODEL_FAMILY $MODEL_PATH_ARG \
    interfaces=inference,train_step,generate \
    strategies="d8;d4t2;d2t4;d1t8;d4p2" \
    n_seqs=128 seq_len=512 gen_tokens=128
