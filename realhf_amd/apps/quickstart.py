"""CLI entry point (reference: realhf/apps/quickstart.py — the Hydra CLI
`python -m realhf.apps.quickstart {sft,rw,dpo,ppo,gen} a.b=c ...`).

Hydra is replaced by a dotted-override parser over the same structured
dataclasses; usage is otherwise identical:

    python -m realhf_amd.apps.quickstart ppo \
        experiment_name=quickstart-ppo trial_name=t0 n_gpus=8 \
        actor.path=/path/to/sft-ckpt critic.path=/path/to/rw-ckpt \
        dataset.path=prompts.jsonl dataset.train_bs_n_seqs=128 \
        ppo.gen.max_new_tokens=512 allocation_mode=heuristic

Launch modes: under torchrun (WORLD_SIZE set) runs this rank's worker;
standalone it spawns n_gpus local workers (scheduler/local.py).
"""
import dataclasses
import os
import sys
import typing

from realhf_amd.api.experiment import EXPERIMENT_TYPES
from realhf_amd.base import logging

logger = logging.getLogger("quickstart")


def _set_dotted(obj, dotted: str, raw: str):
    parts = dotted.split(".")
    for p in parts[:-1]:
        obj = getattr(obj, p)
    leaf = parts[-1]
    fields = {f.name: f for f in dataclasses.fields(obj)}
    if leaf not in fields:
        raise KeyError(f"unknown config field {dotted}")
    t = fields[leaf].type
    setattr(obj, leaf, _coerce(raw, t, getattr(obj, leaf)))


def _coerce(raw: str, t, current):
    if raw.lower() in ("null", "none"):
        return None
    origin = typing.get_origin(t)
    if origin is typing.Union:  # Optional[...]
        args = [a for a in typing.get_args(t) if a is not type(None)]
        t = args[0]
    if isinstance(t, str):  # string annotations
        t = {"int": int, "float": float, "bool": bool, "str": str}.get(t, str)
    if t is bool or isinstance(current, bool):
        if raw.lower() in ("1", "true", "yes"):
            return True
        if raw.lower() in ("0", "false", "no"):
            return False
        return raw  # bool|str fields, e.g. offload=dp_shard
    if t is int or isinstance(current, int) and not isinstance(current, bool):
        return int(raw)
    if t is float or isinstance(current, float):
        return float(raw)
    return raw


def parse_cli(argv):
    if not argv or argv[0] in ("-h", "--help"):
        print(__doc__)
        print("experiments:", ", ".join(EXPERIMENT_TYPES))
        sys.exit(0)
    exp_type = argv[0]
    if exp_type not in EXPERIMENT_TYPES:
        raise SystemExit(f"unknown experiment {exp_type!r}; "
                         f"choose from {list(EXPERIMENT_TYPES)}")
    cfg = EXPERIMENT_TYPES[exp_type]()
    for arg in argv[1:]:
        if "=" not in arg:
            raise SystemExit(f"override must be key=value, got {arg!r}")
        k, v = arg.split("=", 1)
        _set_dotted(cfg, k, v)
    return exp_type, cfg


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    exp_type, cfg = parse_cli(argv)

    if "WORLD_SIZE" in os.environ or cfg.n_gpus <= 1:
        # worker mode (torchrun / single process)
        import torch
        import torch.distributed as dist

        world = int(os.environ.get("WORLD_SIZE", "1"))
        if world > 1 and not dist.is_initialized():
            backend = "nccl" if torch.cuda.is_available() else "gloo"
            dist.init_process_group(backend)
            if torch.cuda.is_available():
                torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        from realhf_amd.api.experiment import ProfileConfig

        if isinstance(cfg, ProfileConfig):
            from realhf_amd.runtime.profiler import run_profile

            run_profile(cfg)
        else:
            from realhf_amd.runtime.trainer import Trainer

            Trainer(cfg).run()
        if dist.is_initialized():
            dist.barrier()
            dist.destroy_process_group()
        return

    # launcher mode: spawn workers via the configured scheduler
    from realhf_amd.base.testing import find_free_port

    cmd = [sys.executable, "-m", "realhf_amd.apps.quickstart", exp_type] + list(argv[1:])
    if cfg.mode == "slurm":
        from realhf_amd.scheduler.slurm import SlurmScheduler

        sc = cfg.slurm
        sched = SlurmScheduler(
            cfg.experiment_name, cfg.trial_name,
            partition=sc.partition, account=sc.account,
            time_limit=sc.time_limit, container_image=sc.container_image,
            container_mounts=sc.container_mounts,
            gpus_per_node=sc.gpus_per_node, mem_per_node=sc.mem_per_node,
        )
        sched.submit_array(cmd, cfg.n_gpus)
        sched.wait()
        return
    from realhf_amd.scheduler.local import LocalScheduler

    sched = LocalScheduler(cfg.experiment_name, cfg.trial_name,
                           max_restarts=cfg.max_restarts)
    sched.submit_array(cmd, cfg.n_gpus, master_port=find_free_port())
    sched.wait()


if __name__ == "__main__":
    main()
