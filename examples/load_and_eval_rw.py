#!/usr/bin/env python3
"""Load a reward model programmatically and score sequences
(reference parity: examples/load_and_eval_rw.py).

Usage:  python examples/load_and_eval_rw.py [hf_checkpoint_dir]
Without a checkpoint it builds a tiny random-init critic (CPU demo).
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch

import realhf_amd.models.hf as hf_reg
from realhf_amd.models.real_model import ReaLModel


def main():
    if len(sys.argv) > 1:
        path = sys.argv[1]
        cfg = hf_reg.config_from_hf_path("llama", path)
        cfg.is_critic = True
        model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
        hf_reg.load_from_hf(model, "llama", path)
    else:
        cfg = hf_reg.get_family("llama").make_test_config(is_critic=True)
        cfg.dtype = "float32"
        model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
        model.random_init()

    # two packed sequences
    lens = [9, 7]
    ids = torch.randint(0, cfg.vocab_size, (sum(lens),))
    cu = torch.tensor([0, 9, 16], dtype=torch.int32)
    with torch.no_grad():
        values = model(packed_input_ids=ids, cu_seqlens=cu, max_seqlen=9)
    # reward = value at the last token of each sequence
    scores = [float(values[int(cu[i + 1]) - 1, 0]) for i in range(2)]
    print("rewards:", scores)


if __name__ == "__main__":
    main()
