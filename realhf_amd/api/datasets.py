"""Dataset implementations (reference: realhf/impl/dataset/{prompt_dataset,
prompt_answer_dataset,rw_paired_dataset}.py) plus synthetic datasets for
benchmarking (no-network environment).

All datasets produce SequenceSamples with unique ids.
"""
import numpy as np
import torch

from realhf_amd.api.data import (
    SequenceSample,
    load_shuffle_split_dataset,
    register_dataset,
)


class PromptDataset(torch.utils.data.Dataset):
    """JSONL records {"prompt": str} (or {"input_ids": [...]}) -> packed
    prompts for PPO rollout (reference: prompt_dataset.py)."""

    def __init__(self, path, max_prompt_len, seed=0, dp_rank=0, world_size=1,
                 tokenizer=None, pad_to_max=False):
        self.records = load_shuffle_split_dataset(path, seed, dp_rank, world_size)
        self.max_prompt_len = max_prompt_len
        self.tokenizer = tokenizer
        # deterministic ids: every SPMD rank must assign identical ids
        self.ids = [f"prompt-{seed}-{dp_rank}-{i}" for i in range(len(self.records))]

    def __len__(self):
        return len(self.records)

    def __getitem__(self, i):
        r = self.records[i]
        if "input_ids" in r:
            toks = r["input_ids"][: self.max_prompt_len]
        else:
            toks = self.tokenizer(
                r["prompt"], truncation=True, max_length=self.max_prompt_len
            )["input_ids"]
        t = torch.tensor(toks, dtype=torch.long)
        return SequenceSample(
            keys=("packed_prompts",),
            ids=[self.ids[i]],
            seqlens={"packed_prompts": [[len(toks)]]},
            data={"packed_prompts": t},
        )


class PromptAnswerDataset(torch.utils.data.Dataset):
    """JSONL {"prompt": str, "answer": str} -> packed prompt+answer with
    prompt_mask for SFT (reference: prompt_answer_dataset.py)."""

    def __init__(self, path, max_seqlen, seed=0, dp_rank=0, world_size=1,
                 tokenizer=None):
        self.records = load_shuffle_split_dataset(path, seed, dp_rank, world_size)
        self.max_seqlen = max_seqlen
        self.tokenizer = tokenizer
        self.ids = [f"pa-{seed}-{dp_rank}-{i}" for i in range(len(self.records))]

    def __len__(self):
        return len(self.records)

    def __getitem__(self, i):
        r = self.records[i]
        if "prompt_ids" in r:
            p, a = r["prompt_ids"], r["answer_ids"]
        else:
            p = self.tokenizer(r["prompt"])["input_ids"]
            a = self.tokenizer(r["answer"])["input_ids"]
            if self.tokenizer.eos_token_id is not None:
                a = a + [self.tokenizer.eos_token_id]
        toks = (p + a)[: self.max_seqlen]
        pm = ([True] * len(p) + [False] * len(a))[: self.max_seqlen]
        return SequenceSample(
            keys=("packed_input_ids", "prompt_mask"),
            ids=[self.ids[i]],
            seqlens={
                "packed_input_ids": [[len(toks)]],
                "prompt_mask": [[len(toks)]],
            },
            data={
                "packed_input_ids": torch.tensor(toks, dtype=torch.long),
                "prompt_mask": torch.tensor(pm, dtype=torch.bool),
            },
        )


class RewardModelingPairedDataset(torch.utils.data.Dataset):
    """JSONL {"prompt", "pos_answers": [...], "neg_answers": [...]} ->
    [pos, neg] sequence pairs (reference: rw_paired_dataset.py)."""

    def __init__(self, path, max_seqlen, max_pairs_per_prompt=1, seed=0,
                 dp_rank=0, world_size=1, tokenizer=None):
        self.records = load_shuffle_split_dataset(path, seed, dp_rank, world_size)
        self.max_seqlen = max_seqlen
        self.max_pairs = max_pairs_per_prompt
        self.tokenizer = tokenizer
        self.ids = [f"rw-{seed}-{dp_rank}-{i}" for i in range(len(self.records))]
        self.rng = np.random.RandomState(seed)

    def __len__(self):
        return len(self.records)

    def _tok(self, s):
        return self.tokenizer(s, truncation=True, max_length=self.max_seqlen)[
            "input_ids"
        ]

    def __getitem__(self, i):
        r = self.records[i]
        if "pos_ids" in r:
            pos, neg = r["pos_ids"][: self.max_seqlen], r["neg_ids"][: self.max_seqlen]
        else:
            p = self.tokenizer(r["prompt"])["input_ids"]
            pos = (p + self._tok(r["pos_answers"][0]))[: self.max_seqlen]
            neg = (p + self._tok(r["neg_answers"][0]))[: self.max_seqlen]
        toks = torch.tensor(pos + neg, dtype=torch.long)
        return SequenceSample(
            keys=("packed_input_ids",),
            ids=[self.ids[i]],
            seqlens={"packed_input_ids": [[len(pos), len(neg)]]},
            data={"packed_input_ids": toks},
        )


class SyntheticPromptDataset(torch.utils.data.Dataset):
    """Random prompts of a given length distribution (bench path: BASELINE
    is measured on synthetic prompts, random-init weights)."""

    def __init__(self, n_prompts, prompt_len, vocab_size, seed=0, dp_rank=0,
                 world_size=1, tokenizer=None, fixed_len=True):
        rng = np.random.RandomState(seed + dp_rank)
        self.n = n_prompts
        lens = (
            np.full(n_prompts, prompt_len)
            if fixed_len
            else rng.randint(prompt_len // 2, prompt_len + 1, size=n_prompts)
        )
        self.prompts = [
            torch.from_numpy(rng.randint(10, vocab_size - 10, size=l)).long()
            for l in lens
        ]
        self.ids = [f"synth-{dp_rank}-{i}" for i in range(n_prompts)]

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        return SequenceSample(
            keys=("packed_prompts",),
            ids=[self.ids[i]],
            seqlens={"packed_prompts": [[self.prompts[i].shape[0]]]},
            data={"packed_prompts": self.prompts[i]},
        )


register_dataset("prompt", PromptDataset)
register_dataset("prompt_answer", PromptAnswerDataset)
register_dataset("rw_paired", RewardModelingPairedDataset)
# reference registers this dataset as "rw_pair" (rw_paired_dataset.py) —
# keep that name working for drop-in configs
register_dataset("rw_pair", RewardModelingPairedDataset)
register_dataset("synthetic_prompt", SyntheticPromptDataset)
