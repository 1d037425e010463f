"""Generation vs HuggingFace greedy decode on CPU (reference test:
tests/model/test_generate.py)."""
import numpy as np
import pytest
import torch

import realhf_amd.models.hf as hf_reg
from realhf_amd.api.model import GenerationHyperparameters
from realhf_amd.models.generation import (
    concat_prompt_to_generation_output,
    generate,
)
from realhf_amd.models.real_model import ReaLModel
from tests.test_model_cpu import make_hf_model, packed_batch


@pytest.mark.parametrize("family", ["llama", "gpt2"])
def test_greedy_generate_matches_hf(family):
    fam = hf_reg.get_family(family)
    cfg = fam.make_test_config()
    cfg.dtype = "float32"
    hf_model = make_hf_model(family, cfg)
    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    hf_reg.load_from_hf_state_dict(model, family, hf_model.state_dict())

    packed, cu, mx = packed_batch(cfg.vocab_size, bs=3, lo=4, hi=10, seed=7)
    gconfig = GenerationHyperparameters(
        max_new_tokens=8, greedy=True, use_hip_graph=False
    )
    out = generate(model, packed, cu, gconfig, eos_token_id=None)

    for i in range(3):
        s, e = int(cu[i]), int(cu[i + 1])
        hf_out = hf_model.generate(
            input_ids=packed[s:e].unsqueeze(0),
            max_new_tokens=8,
            do_sample=False,
            use_cache=True,
            pad_token_id=0,
        )[0, e - s :]
        got = out.gen_tokens[i, : int(out.gen_lengths[i])]
        assert got.tolist() == hf_out.tolist(), (i, got.tolist(), hf_out.tolist())


def test_generate_eos_termination():
    fam = hf_reg.get_family("llama")
    cfg = fam.make_test_config()
    cfg.dtype = "float32"
    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    torch.manual_seed(0)
    model.random_init()
    packed, cu, mx = packed_batch(cfg.vocab_size, bs=4, seed=3)
    gconfig = GenerationHyperparameters(
        max_new_tokens=12, min_new_tokens=2, greedy=False, top_k=5, top_p=0.9,
        use_hip_graph=False,
    )
    g = torch.Generator().manual_seed(123)
    out = generate(model, packed, cu, gconfig, eos_token_id=2, generator=g)
    assert out.gen_tokens.shape[0] == 4
    assert (out.gen_lengths >= 2).all()
    packed_full, cu_full, pmask = concat_prompt_to_generation_output(packed, cu, out)
    assert packed_full.shape[0] == int(cu_full[-1])
    # prompt mask marks exactly the prompt tokens
    total_prompt = int(cu[-1])
    assert int(pmask.sum()) == total_prompt


def test_generate_deterministic_with_generator():
    fam = hf_reg.get_family("llama")
    cfg = fam.make_test_config()
    cfg.dtype = "float32"
    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    torch.manual_seed(1)
    model.random_init()
    packed, cu, _ = packed_batch(cfg.vocab_size, bs=2, seed=5)
    gconfig = GenerationHyperparameters(
        max_new_tokens=6, greedy=False, top_k=10, use_hip_graph=False
    )
    o1 = generate(model, packed, cu, gconfig, generator=torch.Generator().manual_seed(9))
    o2 = generate(model, packed, cu, gconfig, generator=torch.Generator().manual_seed(9))
    assert torch.equal(o1.gen_tokens, o2.gen_tokens)


def test_mistral_sliding_window_generate_matches_hf():
    """Greedy decode with a binding sliding window (cache grows past the
    window) matches transformers mistral."""
    fam = hf_reg.get_family("mistral")
    cfg = fam.make_test_config()
    cfg.dtype = "float32"
    cfg.sliding_window = 6
    hf_model = make_hf_model("mistral", cfg)
    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    hf_reg.load_from_hf_state_dict(model, "mistral", hf_model.state_dict())

    packed, cu, mx = packed_batch(cfg.vocab_size, bs=3, lo=3, hi=6, seed=11)
    gconfig = GenerationHyperparameters(
        max_new_tokens=10, greedy=True, use_hip_graph=False
    )
    out = generate(model, packed, cu, gconfig, eos_token_id=None)
    for i in range(3):
        s, e = int(cu[i]), int(cu[i + 1])
        hf_out = hf_model.generate(
            input_ids=packed[s:e].unsqueeze(0),
            max_new_tokens=10,
            do_sample=False,
            use_cache=True,
            pad_token_id=0,
        )[0, e - s:]
        got = out.gen_tokens[i, : int(out.gen_lengths[i])]
        assert got.tolist() == hf_out.tolist(), (i, got.tolist(), hf_out.tolist())
