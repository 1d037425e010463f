#!/usr/bin/env python3
"""Print the MFC dataflow graph of an experiment (reference:
examples/visualize_dfg.py).  Usage: python examples/visualize_dfg.py ppo"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from realhf_amd.api.experiment import EXPERIMENT_TYPES
from realhf_amd.runtime.trainer import build_experiment


def main():
    exp = sys.argv[1] if len(sys.argv) > 1 else "ppo"
    cfg = EXPERIMENT_TYPES[exp]()
    built = build_experiment(cfg, world=8)
    g = built.graph
    print(f"=== {exp} dataflow graph ===")
    for m in g.topological_order():
        alloc = built.allocations[m.name]
        deps = ", ".join(p.name for p in m.parents) or "(dataset)"
        hooks = [type(h).__name__ for h in m.pre_hooks + m.post_hooks]
        print(f"{m.name:14} [{m.interface_type.value:10}] model={m.model_name} "
              f"strategy={alloc.strategy.pp}p{alloc.strategy.dp}d"
              f"{alloc.strategy.tp}t  <- {deps}"
              + (f"  hooks={hooks}" if hooks else ""))
    print("keys:", {k: p.name for k, p in g.data_producers.items()})


if __name__ == "__main__":
    main()
