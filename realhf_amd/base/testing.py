"""Multi-process test harness (reference: realhf/base/testing.py:112
LocalMultiProcessTest).

Spawns N processes on one machine doing a REAL torch.distributed
rendezvous with backend "gloo" when no GPU is present — this is how all
multi-rank logic (topologies, TP math, realloc plans, the DFG executor)
is tested without hardware.  On a GPU box the same harness runs with
RCCL ("nccl" backend on ROCm).
"""
import os
import traceback
from typing import Callable, Optional

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from realhf_amd.base import constants, seeding
from realhf_amd.base.topology import (
    ParallelGrid,
    PipeDataTensorTopology,
    clear_group_cache,
)

TESTING_MODEL_CONFIG = dict(
    n_layers=2,
    hidden_dim=32,
    n_heads=4,
    n_kv_heads=2,
    head_dim=8,
    intermediate_dim=64,
    vocab_size=64,
    max_position_embeddings=128,
)


def _proc_entry(rank, world_size, port, backend, fn, args, kwargs, errq):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
        seeding.set_random_seed(12345, rank_offset=0)
        fn(*args, **kwargs)
        dist.barrier()
        dist.destroy_process_group()
    except Exception:
        errq.put((rank, traceback.format_exc()))
        raise


class LocalMultiProcessTest:
    def __init__(
        self,
        world_size: int,
        fn: Callable,
        *args,
        backend: Optional[str] = None,
        timeout_secs: int = 300,
        **kwargs,
    ):
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        self.world_size = world_size
        self.backend = backend
        self.timeout = timeout_secs
        self._spawn = (fn, args, kwargs)
        self._make_procs()

    def _make_procs(self):
        fn, args, kwargs = self._spawn
        ctx = mp.get_context("spawn")
        self.errq = ctx.Queue()
        port = find_free_port()
        self.procs = [
            ctx.Process(
                target=_proc_entry,
                args=(r, self.world_size, port, self.backend, fn, args,
                      kwargs, self.errq),
            )
            for r in range(self.world_size)
        ]

    def launch(self):
        errs = self._launch_once()
        if errs and all(
            "rendezvous" in tb or "EADDRINUSE" in tb
            or "Address already in use" in tb or "Connection" in tb
            for _, tb in errs
        ):
            # one retry on a fresh port: rendezvous races with TIME_WAIT
            # sockets of the previous test are a known flake source
            self._make_procs()
            errs = self._launch_once()
        if errs:
            raise RuntimeError("\n".join(f"[rank {r}]\n{tb}" for r, tb in errs))

    def _launch_once(self):
        for p in self.procs:
            p.start()
        for p in self.procs:
            p.join(self.timeout)
        errs = []
        while not self.errq.empty():
            errs.append(self.errq.get())
        for p in self.procs:
            if p.is_alive():
                p.terminate()
                errs.append((p.name, "timeout"))
        return errs


def find_free_port() -> int:
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def init_global_constants(
    num_dp=1,
    num_tp=1,
    num_pp=1,
    model_name: str = "default",
    sequence_parallel: bool = False,
    gradient_checkpointing: bool = False,
    rank_mapping=None,
):
    """Build a grid + register it under `model_name` (reference:
    testing.py:175)."""
    topo = PipeDataTensorTopology(
        num_pp=num_pp,
        num_dp=num_dp,
        num_tp=num_tp,
        sequence_parallel=sequence_parallel,
        gradient_checkpointing=gradient_checkpointing,
    )
    grid = ParallelGrid(topo, rank_mapping=rank_mapping)
    constants.set_grid(model_name, grid)
    return grid


def clear_constants():
    constants.clear_grids()
    clear_group_cache()
