import numpy as np
import pytest
import torch

from realhf_amd.api.data import PackedDataLoader, SequenceSample


def make_sample(bs, seed=0, with_reward=True):
    rng = np.random.RandomState(seed)
    seqlens = rng.randint(5, 40, size=bs).tolist()
    total = sum(seqlens)
    data = {
        "packed_input_ids": torch.arange(total, dtype=torch.long),
        "packed_logprobs": torch.randn(total - bs),
    }
    if with_reward:
        data["rewards"] = torch.randn(bs)
    ids = [f"s{seed}_{i}" for i in range(bs)]
    return SequenceSample.from_default(ids=ids, seqlens=seqlens, data=data)


def test_from_default_key_rules():
    s = make_sample(4)
    assert s.seqlens["packed_logprobs"] == [[l[0] - 1] for l in s.seqlens["packed_input_ids"]]
    assert s.seqlens["rewards"] == [[1]] * 4


def test_gather_split_roundtrip():
    parts = [make_sample(3, seed=i) for i in range(4)]
    g = SequenceSample.gather(parts)
    assert g.bs == 12
    back = g.split_with_spec([(0, 3), (3, 6), (6, 9), (9, 12)])
    for orig, b in zip(parts, back):
        assert orig.ids == b.ids
        for k in orig.keys:
            assert torch.equal(orig.data[k], b.data[k])


@pytest.mark.parametrize("dp", [1, 2, 3, 4, 8])
def test_balanced_split(dp):
    s = make_sample(16, seed=dp)
    shards = s.split(dp)
    assert sum(x.bs for x in shards) == 16
    recon = SequenceSample.gather(shards)
    for k in s.keys:
        assert torch.equal(s.data[k], recon.data[k])
    # balance check
    tok = [sum(x.main_seqlens()) for x in shards]
    assert max(tok) - min(tok) <= max(s.main_seqlens())


def test_unpack_and_meta():
    s = make_sample(5)
    singles = s.unpack()
    assert len(singles) == 5
    assert all(x.bs == 1 for x in singles)
    m = s.meta()
    assert m.data is None
    assert m.seqlens == s.seqlens
    assert m.dtypes["packed_input_ids"] == torch.long


def test_update_and_remap():
    s = make_sample(4, with_reward=False)
    total = sum(s.main_seqlens())
    other = SequenceSample(
        keys=("values",),
        ids=list(s.ids),
        seqlens={"values": s.seqlens["packed_input_ids"]},
        data={"values": torch.randn(total)},
    )
    s.update_(other)
    assert "values" in s.keys
    s.remap_keys_({"values": "old_values"})
    assert "old_values" in s.keys and "values" not in s.keys


def test_select():
    s = make_sample(6)
    sub = s.select_idx([5, 0, 3])
    assert sub.ids == [s.ids[5], s.ids[0], s.ids[3]]
    k = s.select_keys(["packed_input_ids"])
    assert k.keys == ("packed_input_ids",)


def test_packed_dataloader():
    class DS(torch.utils.data.Dataset):
        def __len__(self):
            return 10

        def __getitem__(self, i):
            return make_sample(1, seed=100 + i)

    dl = PackedDataLoader(DS(), batch_n_seqs=4, shuffle=True, seed=1)
    batches = list(dl)
    assert len(batches) == 3
    assert batches[0].bs == 4
    all_ids = [i for b in batches for i in b.ids]
    assert len(set(all_ids)) == 10


def test_dataset_cache(tmp_path, monkeypatch):
    import json

    from realhf_amd.api.config import Abstraction
    from realhf_amd.api.data import make_dataset

    p = tmp_path / "d.jsonl"
    with open(p, "w") as f:
        for i in range(6):
            f.write(json.dumps({"input_ids": [3, 4, 5 + i]}) + "\n")
    monkeypatch.setenv("REALHF_AMD_DATASET_CACHE", str(tmp_path / "cache"))
    import realhf_amd.api.datasets  # noqa: F401

    cfg = Abstraction("prompt", {"path": str(p), "max_prompt_len": 8})
    d1 = make_dataset(cfg, seed=1, dp_rank=0, world_size=1)
    files = list((tmp_path / "cache").iterdir())
    assert len(files) == 1
    d2 = make_dataset(cfg, seed=1, dp_rank=0, world_size=1)  # from cache
    assert len(d1) == len(d2)
    assert d1[0].data["packed_prompts"].tolist() == \
        d2[0].data["packed_prompts"].tolist()


def test_dataloader_identical_across_replicas(tmp_path):
    """SPMD critical: every rank builds the same dataset/loader with the
    same seed and MUST iterate identical batches (ids and order)."""
    import json

    from realhf_amd.api.config import Abstraction
    from realhf_amd.api.data import PackedDataLoader, make_dataset
    import realhf_amd.api.datasets  # noqa: F401

    p = tmp_path / "d.jsonl"
    with open(p, "w") as f:
        for i in range(24):
            f.write(json.dumps({"input_ids": [3, 4, 5 + i % 7]}) + "\n")
    cfg = Abstraction("prompt", {"path": str(p), "max_prompt_len": 8})

    def batch_ids():
        ds = make_dataset(cfg, seed=3, dp_rank=0, world_size=1)
        dl = PackedDataLoader(ds, batch_n_seqs=6, shuffle=True, seed=3)
        return [tuple(map(str, b.ids)) for b in dl]

    a, b = batch_ids(), batch_ids()
    assert a == b and len(a) == 4


def test_flops_formula_sanity():
    from realhf_amd.base.monitor import dense_transformer_flops

    # llama-7b-ish: 2 * params * tokens dominates; check the right scale
    f = dense_transformer_flops(
        n_layers=32, hidden=4096, intermediate=11008, vocab=32000,
        n_heads=32, n_kv_heads=32, head_dim=128,
        total_tokens=1024, sum_sq_seqlens=0.0,
    )
    approx = 2 * 6.7e9 * 1024
    assert 0.8 * approx < f < 1.3 * approx
    assert dense_transformer_flops(
        32, 4096, 11008, 32000, 32, 32, 128, 1024, 0.0, backward=True
    ) == 3 * f


def test_payload_pack_unpack_roundtrip():
    """Device-transfer payload codec: mixed dtypes, a 0-d-free mix of
    shapes, and a None-valued (metadata-only) key must survive the
    pack -> flat-tensors -> unpack round trip bit-exactly."""
    from realhf_amd.runtime.data_transfer import _pack_payload, _unpack_payload

    s = SequenceSample(
        keys=("ids", "logp", "mask", "meta_only"),
        ids=["a", "b"],
        seqlens={"ids": [[4], [3]], "logp": [[3], [2]],
                 "mask": [[4], [3]], "meta_only": [[1], [1]]},
        data={
            "ids": torch.arange(7, dtype=torch.long),
            "logp": torch.randn(5, dtype=torch.float32),
            "mask": torch.tensor([1, 0, 1, 1, 0, 1, 1], dtype=torch.bool),
            "meta_only": None,
        },
    )
    meta, flats = _pack_payload(s)
    # one flat tensor per dtype
    assert set(flats) == {"torch.int64", "torch.float32", "torch.bool"}
    out = _unpack_payload(meta, flats)
    assert out.ids == s.ids and out.keys == s.keys
    assert out.data["meta_only"] is None
    for k in ("ids", "logp", "mask"):
        assert out.data[k].dtype == s.data[k].dtype
        assert torch.equal(out.data[k], s.data[k])


def test_payload_roundtrip_2d_bool_logits_mask():
    """packed_logits_mask is 2-D [rows, vocab] bool — the codec must
    restore its shape (per-key shape travels in the payload meta)."""
    from realhf_amd.runtime.data_transfer import _pack_payload, _unpack_payload

    lm = torch.rand(7, 16) > 0.5
    s = SequenceSample(
        keys=("packed_logits_mask",),
        ids=["a", "b"],
        seqlens={"packed_logits_mask": [[4], [3]]},
        data={"packed_logits_mask": lm},
    )
    meta, flats = _pack_payload(s)
    out = _unpack_payload(meta, flats)
    assert out.data["packed_logits_mask"].shape == (7, 16)
    assert torch.equal(out.data["packed_logits_mask"], lm)
    # and dim-0 split/gather keeps rows aligned
    a, b = s.split(2)
    assert a.data["packed_logits_mask"].shape == (4, 16)
    assert torch.equal(SequenceSample.gather([a, b]).data["packed_logits_mask"], lm)
