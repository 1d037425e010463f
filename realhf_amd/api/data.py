"""SequenceSample — the universal packed-sequence batch — plus the dataset
registry and packed dataloaders.

Reference semantics: realhf/api/core/data_api.py (SequenceSample:96,
gather:268, split:316, unpack:409, meta:428, update_:441, from_default:499,
key→seqlen rules:455-497, register_dataset:671, PackedDataLoader:761).

Everything is packed: a batch is a dict of 1-D (plus optional trailing
dims) tensors concatenated across samples, with per-key nested sequence
lengths and one unique id per sample.  No padding anywhere.
"""
import dataclasses
from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from realhf_amd.base import datapack, logging

logger = logging.getLogger("data")

_VALIDATE = True


def disable_validation():
    global _VALIDATE
    _VALIDATE = False


@dataclasses.dataclass
class SequenceSample:
    keys: Tuple[str, ...]
    ids: List[Any]
    seqlens: Dict[str, List[List[int]]]  # key -> per-sample list of seqlens
    data: Optional[Dict[str, Optional[torch.Tensor]]] = None
    dtypes: Dict[str, Optional[torch.dtype]] = dataclasses.field(default_factory=dict)
    trailing_shapes: Dict[str, Tuple[int, ...]] = dataclasses.field(
        default_factory=dict
    )
    metadata: Dict[str, Any] = dataclasses.field(default_factory=dict)

    # ------------------------------------------------------------------ init
    def __post_init__(self):
        self.keys = tuple(sorted(self.keys))
        if self.data is not None and not self.dtypes:
            self.dtypes = {
                k: (v.dtype if v is not None else None)
                for k, v in self.data.items()
            }
        if self.data is not None and not self.trailing_shapes:
            self.trailing_shapes = {
                k: (tuple(v.shape[1:]) if v is not None else ())
                for k, v in self.data.items()
            }
        if _VALIDATE:
            self._validate()

    def _validate(self):
        assert len(set(map(str, self.ids))) == len(self.ids), "duplicate ids"
        bs = len(self.ids)
        for k in self.keys:
            assert k in self.seqlens, f"missing seqlens for key {k}"
            assert len(self.seqlens[k]) == bs, (
                k,
                len(self.seqlens[k]),
                bs,
            )
        if self.data is not None:
            assert set(self.data.keys()) == set(self.keys), (
                set(self.data.keys()),
                set(self.keys),
            )
            for k, v in self.data.items():
                if v is None:
                    continue
                expected = sum(datapack.flat2d(self.seqlens[k]))
                assert v.shape[0] == expected, (
                    f"key {k}: tensor dim0 {v.shape[0]} != sum seqlens {expected}"
                )

    # ------------------------------------------------------------- factories
    @classmethod
    def from_default(
        cls,
        ids: List[Any],
        seqlens: List[int],
        data: Dict[str, torch.Tensor],
        metadata: Optional[Dict[str, Any]] = None,
    ) -> "SequenceSample":
        """Build from per-sample MAIN seqlens; per-key seqlens are inferred
        from the key name (reference rules at data_api.py:455-497)."""
        assert all(isinstance(s, (int, np.integer)) for s in seqlens)
        key_seqlens = {
            k: [[cls._default_key_seqlen(k, s)] for s in seqlens] for k in data
        }
        return cls(
            keys=tuple(data.keys()),
            ids=list(ids),
            seqlens=key_seqlens,
            data=dict(data),
            metadata=metadata or {},
        )

    @staticmethod
    def _default_key_seqlen(key: str, seqlen: int) -> int:
        if key in (
            "packed_input_ids",
            "packed_prompts",
            "prompt_mask",
            "packed_seq",
            "seq",
            "values",
            "loss_mask",
            "input_ids",
        ):
            return seqlen
        if key in (
            "packed_logprobs",
            "logprobs",
            "packed_ref_logprobs",
            "ref_logprobs",
            "old_logp",
            "ref_logp",
            "advantages",
            "ppo_loss_mask",
            "kl_rewards",
            "returns",
        ):
            return seqlen - 1
        if key in (
            "rewards",
            "greedy_rewards",
            "scores",
            "seq_no_eos_mask",
            "base_scores",
        ):
            return 1
        raise ValueError(
            f"cannot infer per-key seqlen for key {key!r}; construct the "
            f"SequenceSample explicitly with per-key seqlens"
        )

    @classmethod
    def gather(
        cls, samples: Sequence["SequenceSample"], keys: Optional[Sequence[str]] = None
    ) -> "SequenceSample":
        """Concatenate samples (reference: data_api.py:268)."""
        assert len(samples) > 0
        if keys is None:
            keys = samples[0].keys
        keys = tuple(sorted(keys))
        seqlens = {k: datapack.flat2d([s.seqlens[k] for s in samples]) for k in keys}
        ids = datapack.flat2d([s.ids for s in samples])
        if samples[0].data is not None:
            data = {}
            for k in keys:
                vs = [s.data[k] for s in samples]
                if any(v is None for v in vs):
                    data[k] = None
                else:
                    data[k] = torch.cat(vs, dim=0)
        else:
            data = None
        metadata = {}
        for s in samples:
            for mk, mv in s.metadata.items():
                metadata.setdefault(mk, []).extend(
                    mv if isinstance(mv, list) else [mv]
                )
        out = cls(
            keys=keys,
            ids=ids,
            seqlens=seqlens,
            data=data,
            dtypes=dict(samples[0].dtypes),
            trailing_shapes=dict(samples[0].trailing_shapes),
            metadata=metadata,
        )
        return out

    # ------------------------------------------------------------- splitting
    @property
    def bs(self) -> int:
        return len(self.ids)

    def _main_key(self) -> str:
        for cand in ("packed_input_ids", "packed_prompts", "packed_seq"):
            if cand in self.keys:
                return cand
        # fall back: the key with the largest total length
        return max(
            self.keys,
            key=lambda k: sum(datapack.flat2d(self.seqlens[k])),
        )

    def main_seqlens(self) -> List[int]:
        """Per-sample total length of the main key."""
        k = self._main_key()
        return [sum(x) for x in self.seqlens[k]]

    def get_split_spec(self, k: int) -> List[Tuple[int, int]]:
        """Balanced contiguous partition of samples into k parts by main
        seqlen (reference: data_api.py:316 via min_abs_diff_partition)."""
        return datapack.min_abs_diff_partition(self.main_seqlens(), k)

    def split_with_spec(self, spec: List[Tuple[int, int]]) -> List["SequenceSample"]:
        out = []
        offsets = {k: 0 for k in self.keys}
        # precompute per-sample flat lengths per key
        for start, end in spec:
            seqlens = {k: self.seqlens[k][start:end] for k in self.keys}
            ids = self.ids[start:end]
            data = None
            if self.data is not None:
                data = {}
                for k in self.keys:
                    v = self.data[k]
                    if v is None:
                        data[k] = None
                        continue
                    n = sum(datapack.flat2d(seqlens[k]))
                    data[k] = v[offsets[k] : offsets[k] + n]
                    offsets[k] += n
            metadata = {
                mk: (mv[start:end] if isinstance(mv, list) and len(mv) == self.bs else mv)
                for mk, mv in self.metadata.items()
            }
            out.append(
                SequenceSample(
                    keys=self.keys,
                    ids=ids,
                    seqlens=seqlens,
                    data=data,
                    dtypes=dict(self.dtypes),
                    trailing_shapes=dict(self.trailing_shapes),
                    metadata=metadata,
                )
            )
        return out

    def split(self, k: int) -> List["SequenceSample"]:
        return self.split_with_spec(self.get_split_spec(k))

    def unpack(self) -> List["SequenceSample"]:
        return self.split_with_spec([(i, i + 1) for i in range(self.bs)])

    def select_idx(self, indices: Sequence[int]) -> "SequenceSample":
        """Gather an arbitrary subset/order of samples (not necessarily
        contiguous)."""
        singles = self.unpack()
        return SequenceSample.gather([singles[i] for i in indices])

    def select_keys(self, keys: Sequence[str]) -> "SequenceSample":
        keys = tuple(sorted(keys))
        for k in keys:
            assert k in self.keys, (k, self.keys)
        return SequenceSample(
            keys=keys,
            ids=list(self.ids),
            seqlens={k: self.seqlens[k] for k in keys},
            data=None if self.data is None else {k: self.data[k] for k in keys},
            dtypes={k: self.dtypes.get(k) for k in keys},
            trailing_shapes={k: self.trailing_shapes.get(k, ()) for k in keys},
            metadata=dict(self.metadata),
        )

    # ----------------------------------------------------------- meta/update
    def meta(self) -> "SequenceSample":
        """Metadata-only view (no tensors) — what the planner passes
        around (reference: data_api.py:428)."""
        return SequenceSample(
            keys=self.keys,
            ids=list(self.ids),
            seqlens=dict(self.seqlens),
            data=None,
            dtypes=dict(self.dtypes),
            trailing_shapes=dict(self.trailing_shapes),
            metadata=dict(self.metadata),
        )

    def update_(self, other: "SequenceSample"):
        """Merge keys of `other` (same ids, same order) into self
        (reference: data_api.py:441)."""
        assert [str(i) for i in self.ids] == [str(i) for i in other.ids], (
            "id mismatch in update_",
            self.ids[:4],
            other.ids[:4],
        )
        self.keys = tuple(sorted(set(self.keys) | set(other.keys)))
        self.seqlens.update(other.seqlens)
        self.dtypes.update(other.dtypes)
        self.trailing_shapes.update(other.trailing_shapes)
        if self.data is None:
            self.data = {}
        if other.data is not None:
            self.data.update(other.data)
        for k in self.keys:
            self.data.setdefault(k, None)
        self.metadata.update(other.metadata)

    def remap_keys_(self, remap: Dict[str, str]):
        """Rename keys in place (reference: data_api.py:560)."""
        if not remap:
            return
        def rn(k):
            return remap.get(k, k)

        self.keys = tuple(sorted(rn(k) for k in self.keys))
        self.seqlens = {rn(k): v for k, v in self.seqlens.items()}
        self.dtypes = {rn(k): v for k, v in self.dtypes.items()}
        self.trailing_shapes = {rn(k): v for k, v in self.trailing_shapes.items()}
        if self.data is not None:
            self.data = {rn(k): v for k, v in self.data.items()}

    # --------------------------------------------------------------- devices
    def to_device(self, device) -> "SequenceSample":
        if self.data is None:
            return self
        self.data = {
            k: (v.to(device, non_blocking=True) if v is not None else None)
            for k, v in self.data.items()
        }
        return self

    def cuda(self):
        return self.to_device("cuda")

    def cpu(self):
        return self.to_device("cpu")

    def __repr__(self):
        return (
            f"SequenceSample(bs={self.bs}, keys={list(self.keys)}, "
            f"total_tokens={sum(self.main_seqlens())})"
        )


# ---------------------------------------------------------------------------
# Micro-batch splitting helper used by every engine call.
# ---------------------------------------------------------------------------
def split_minibatches(
    sample: SequenceSample, n_mbs: Optional[int], max_tokens: Optional[int] = None
) -> List[SequenceSample]:
    if n_mbs is None or n_mbs <= 1:
        return [sample]
    n_mbs = min(n_mbs, sample.bs)
    return sample.split(n_mbs)


# ---------------------------------------------------------------------------
# Dataset registry
# ---------------------------------------------------------------------------
_DATASETS: Dict[str, Callable] = {}


def register_dataset(name: str, cls: Callable):
    assert name not in _DATASETS, name
    _DATASETS[name] = cls


def make_dataset(cfg, seed: int, dp_rank: int, world_size: int, tokenizer=None):
    """cfg: Abstraction{type_, args}.  With REALHF_AMD_DATASET_CACHE set to
    a directory, constructed datasets are pickled there keyed by their
    full spec (reference: data_api.py:671 make_dataset's optional on-disk
    cache) — useful when tokenization dominates startup."""
    import hashlib
    import os
    import pickle

    from realhf_amd.api.config import Abstraction

    if isinstance(cfg, str):
        cfg = Abstraction(type_=cfg)
    cache_dir = os.environ.get("REALHF_AMD_DATASET_CACHE")
    cache_path = None
    if cache_dir:
        key = repr((cfg.type_, sorted(cfg.args.items()), seed, dp_rank,
                    world_size))
        h = hashlib.sha256(key.encode()).hexdigest()[:16]
        cache_path = os.path.join(cache_dir, f"dataset_{cfg.type_}_{h}.pkl")
        if os.path.exists(cache_path):
            with open(cache_path, "rb") as f:
                return pickle.load(f)
    cls = _DATASETS[cfg.type_]
    ds = cls(
        seed=seed, dp_rank=dp_rank, world_size=world_size, tokenizer=tokenizer,
        **cfg.args,
    )
    if cache_path:
        os.makedirs(cache_dir, exist_ok=True)
        try:
            with open(cache_path, "wb") as f:
                pickle.dump(ds, f)
        except Exception as e:  # tokenizer refs etc. may not pickle
            logger.warning("dataset cache write failed: %s", e)
    return ds


def load_shuffle_split_dataset(path: str, seed: int, dp_rank: int, world_size: int):
    """Load a json/jsonl list of dicts, shuffle with `seed`, return this DP
    rank's contiguous shard (reference: data_api.py:631)."""
    import json

    if path.endswith(".jsonl"):
        with open(path) as f:
            records = [json.loads(l) for l in f if l.strip()]
    elif path.endswith(".json"):
        with open(path) as f:
            records = json.load(f)
    else:
        raise ValueError(f"unsupported dataset file {path}")
    rng = np.random.RandomState(seed)
    perm = rng.permutation(len(records))
    records = [records[i] for i in perm]
    shard = np.array_split(np.arange(len(records)), world_size)[dp_rank]
    return [records[i] for i in shard]


class PackedDataLoader:
    """Iterates a torch Dataset of SequenceSamples, gathering `batch_n_seqs`
    samples per batch (collate = SequenceSample.gather; reference:
    data_api.py:761)."""

    def __init__(self, dataset, batch_n_seqs: int, shuffle: bool = True, seed: int = 0):
        self.dataset = dataset
        self.batch_n_seqs = batch_n_seqs
        self.shuffle = shuffle
        self.seed = seed
        self._epoch = 0

    def __len__(self):
        return (len(self.dataset) + self.batch_n_seqs - 1) // self.batch_n_seqs

    def __iter__(self):
        order = np.arange(len(self.dataset))
        if self.shuffle:
            rng = np.random.RandomState(self.seed + self._epoch)
            rng.shuffle(order)
        self._epoch += 1
        for i in range(0, len(order), self.batch_n_seqs):
            idxs = order[i : i + self.batch_n_seqs]
            samples = [self.dataset[int(j)] for j in idxs]
            yield SequenceSample.gather(samples)
