"""Deterministic seeding across python/numpy/torch (reference: realhf/base/seeding.py)."""
import random

import numpy as np
import torch

_BASE_SEED = None


def set_random_seed(seed: int, rank_offset: int = 0):
    global _BASE_SEED
    _BASE_SEED = seed
    seed = seed + rank_offset
    random.seed(seed)
    np.random.seed(seed % (2**32))
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def base_seed() -> int:
    return _BASE_SEED if _BASE_SEED is not None else 0
