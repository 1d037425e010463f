"""Balanced partitioning of variable-length sequences.

Reference semantics: realhf/base/datapack.py (min_abs_diff_partition:76,
partition_balanced:13, reorder_to_balanced_batches:116, flat2d:8).
Used to split a packed batch into k contiguous data-parallel shards whose
total token counts are as equal as possible.
"""
from typing import List, Sequence

import numpy as np


def flat2d(xs: Sequence[Sequence]) -> list:
    return [x for sub in xs for x in sub]


def _feasible(lens: np.ndarray, k: int, cap: int) -> bool:
    """Can we split into <= k contiguous non-empty groups each of sum <= cap?"""
    groups = 1
    cur = 0
    for x in lens:
        if x > cap:
            return False
        if cur + x > cap:
            groups += 1
            cur = int(x)
            if groups > k:
                return False
        else:
            cur += int(x)
    return True


def min_abs_diff_partition(seqlens: Sequence[int], k: int) -> List[tuple]:
    """Partition `seqlens` into k CONTIGUOUS non-empty groups minimizing the
    max group sum (binary search over the cap, O(n log sum)).

    Returns list of (start, end) index pairs covering [0, len(seqlens)).
    """
    n = len(seqlens)
    if n < k:
        raise ValueError(f"cannot partition {n} sequences into {k} non-empty groups")
    lens = np.asarray(seqlens, dtype=np.int64)
    lo, hi = int(lens.max()), int(lens.sum())
    while lo < hi:
        mid = (lo + hi) // 2
        if _feasible(lens, k, mid):
            hi = mid
        else:
            lo = mid + 1
    cap = lo
    # Greedy construction honoring "each remaining group needs >= 1 item".
    bounds = []
    start = 0
    for g in range(k):
        remaining_groups = k - g - 1
        end = start + 1  # at least one item
        cur = int(lens[start])
        while end < n - remaining_groups and cur + int(lens[end]) <= cap:
            cur += int(lens[end])
            end += 1
        bounds.append((start, end))
        start = end
    assert start == n, (bounds, n, k, cap)
    return bounds


def partition_balanced(seqlens: Sequence[int], k: int) -> List[List[int]]:
    """Contiguous index groups from min_abs_diff_partition."""
    return [list(range(s, e)) for s, e in min_abs_diff_partition(seqlens, k)]


def reorder_to_balanced_batches(
    seqlens: Sequence[int], n_batches: int
) -> List[List[int]]:
    """Greedy longest-first assignment of indices into n_batches balanced
    (non-contiguous) batches; longest sequences placed first so OOM shows
    up on the first batch (reference: datapack.py:116)."""
    order = np.argsort(-np.asarray(seqlens))
    sums = np.zeros(n_batches, dtype=np.int64)
    out = [[] for _ in range(n_batches)]
    for idx in order:
        b = int(np.argmin(sums))
        out[b].append(int(idx))
        sums[b] += seqlens[idx]
    for b in out:
        b.sort()
    return out
