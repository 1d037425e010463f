"""Per-process parallelism state.

Reference semantics: realhf/base/constants.py (model_scope:170, set_grid:197,
grid():301, parallelism_group:315, sequence_parallel:248, path constants:63).

Each process hosts shards of possibly several models (actor/critic/ref/rew),
each with its own grid.  `model_scope(name)` selects which model's grid the
parallel layers & collectives below use.  All state is process-local.
"""
import contextlib
import getpass
import os
from typing import Dict, Optional

import torch

# ---------------------------------------------------------------------------
# paths
# ---------------------------------------------------------------------------
_fileroot = os.environ.get("REALHF_AMD_FILEROOT", os.path.expanduser("~/.realhf_amd"))


def get_cache_path(*parts) -> str:
    p = os.path.join(_fileroot, *parts)
    os.makedirs(os.path.dirname(p) if parts else p, exist_ok=True)
    return p


def MODEL_SAVE_ROOT(experiment, trial):
    return get_cache_path("checkpoints", getpass.getuser(), experiment, trial)


def LOG_ROOT(experiment, trial):
    return get_cache_path("logs", getpass.getuser(), experiment, trial)


def RECOVER_ROOT(experiment, trial):
    return get_cache_path("recover", getpass.getuser(), experiment, trial)


# ---------------------------------------------------------------------------
# model scope
# ---------------------------------------------------------------------------
_grids: Dict[str, object] = {}
_model_scope: Optional[str] = None
_experiment_name: str = "local"
_trial_name: str = "test"


def set_experiment_trial_names(experiment: str, trial: str):
    global _experiment_name, _trial_name
    _experiment_name, _trial_name = experiment, trial


def experiment_name():
    return _experiment_name


def trial_name():
    return _trial_name


def set_grid(model_name: str, grid):
    _grids[model_name] = grid


def has_model(model_name: str) -> bool:
    return model_name in _grids


@contextlib.contextmanager
def model_scope(model_name: str):
    global _model_scope
    assert model_name in _grids, f"no grid registered for model {model_name}"
    prev = _model_scope
    _model_scope = model_name
    try:
        yield
    finally:
        _model_scope = prev


def has_current() -> bool:
    """True when inside a model_scope (parallel state available)."""
    return _model_scope is not None


def current_model_name() -> str:
    assert _model_scope is not None, "not inside a model_scope"
    return _model_scope


def grid():
    assert _model_scope is not None, "not inside a model_scope"
    return _grids[_model_scope]


def grid_of(model_name: str):
    return _grids[model_name]


def clear_grids():
    global _model_scope
    _grids.clear()
    _model_scope = None


# -- convenience accessors over the current grid ---------------------------
def tp_group():
    return grid().tp_group()


def tp_rank() -> int:
    return grid().tp_rank


def tp_world_size() -> int:
    return grid().tp_size


def dp_group():
    return grid().dp_group()


def dp_rank() -> int:
    return grid().dp_rank


def dp_world_size() -> int:
    return grid().dp_size


def pp_group():
    return grid().pp_group()


def pp_rank() -> int:
    return grid().pp_rank


def pp_world_size() -> int:
    return grid().pp_size


def model_group():
    return grid().model_group()


def is_last_pipe_stage() -> bool:
    return pp_rank() == pp_world_size() - 1


def is_first_pipe_stage() -> bool:
    return pp_rank() == 0


def sequence_parallel() -> bool:
    g = grid()
    return bool(getattr(g.topo, "sequence_parallel", False))


def gradient_checkpointing() -> bool:
    g = grid()
    return bool(getattr(g.topo, "gradient_checkpointing", False))


def max_prompt_len():
    g = grid()
    return getattr(g.topo, "max_prompt_len", None)


# ---------------------------------------------------------------------------
# Reused workspace buffer for TP gathers (reference: GlobalMemoryBuffer
# constants.py:24).  On MI355X with 288 GB HBM we keep one buffer per
# (dtype, name) and grow geometrically; cleared per forward.
# ---------------------------------------------------------------------------
class GlobalMemoryBuffer:
    def __init__(self):
        self.buffers: Dict[tuple, torch.Tensor] = {}

    def get_tensor(self, shape, dtype, name: str) -> torch.Tensor:
        numel = 1
        for s in shape:
            numel *= s
        key = (name, dtype)
        buf = self.buffers.get(key)
        if buf is None or buf.numel() < numel:
            dev = "cuda" if torch.cuda.is_available() else "cpu"
            buf = torch.empty(numel, dtype=dtype, device=dev)
            self.buffers[key] = buf
        return buf[:numel].view(*shape)

    def clear(self):
        self.buffers.clear()


_global_memory_buffer = GlobalMemoryBuffer()


def get_global_memory_buffer() -> GlobalMemoryBuffer:
    return _global_memory_buffer
