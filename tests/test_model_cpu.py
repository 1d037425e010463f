"""ReaLModel vs HuggingFace forward equivalence on CPU (reference test:
tests/model/test_cpu_inference.py)."""
import numpy as np
import pytest
import torch

import realhf_amd.models.hf as hf_reg
from realhf_amd.models.real_model import ReaLModel
from realhf_amd.utils.functional import gather_packed_shifted_log_probs

FAMILIES = ["llama", "gpt2", "qwen2", "gemma", "mistral", "mixtral"]


def make_hf_model(family, cfg):
    import transformers

    fam = hf_reg.get_family(family)
    hf_cfg_dict = dict(fam.config_to_hf(cfg))
    model_type = hf_cfg_dict.pop("model_type")
    hf_cfg = transformers.AutoConfig.for_model(model_type, **hf_cfg_dict)
    torch.manual_seed(42)
    model = transformers.AutoModelForCausalLM.from_config(hf_cfg)
    model = model.eval().float()
    return model


def packed_batch(vocab, bs=4, lo=5, hi=20, seed=0):
    rng = np.random.RandomState(seed)
    seqlens = rng.randint(lo, hi, size=bs)
    ids = [torch.from_numpy(rng.randint(0, vocab, size=l)).long() for l in seqlens]
    packed = torch.cat(ids)
    cu = torch.from_numpy(np.concatenate([[0], np.cumsum(seqlens)])).int()
    return packed, cu, int(seqlens.max())


@pytest.mark.parametrize("family", FAMILIES)
def test_forward_matches_hf(family):
    fam = hf_reg.get_family(family)
    cfg = fam.make_test_config()
    cfg.dtype = "float32"
    hf_model = make_hf_model(family, cfg)

    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    hf_reg.load_from_hf_state_dict(model, family, hf_model.state_dict())

    packed, cu, mx = packed_batch(cfg.vocab_size)
    with torch.no_grad():
        ours = model(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
        # HF forward per sequence (padding-free comparison)
        offs = cu.tolist()
        for i in range(len(offs) - 1):
            s, e = offs[i], offs[i + 1]
            ref = hf_model(input_ids=packed[s:e].unsqueeze(0)).logits[0]
            got = ours[s:e]
            torch.testing.assert_close(
                torch.log_softmax(got.float(), -1),
                torch.log_softmax(ref.float(), -1),
                atol=2e-4,
                rtol=2e-3,
            )


def test_mistral_sliding_window_matches_hf():
    """Binding sliding window (window < seqlen) vs transformers mistral."""
    fam = hf_reg.get_family("mistral")
    cfg = fam.make_test_config()
    cfg.dtype = "float32"
    cfg.sliding_window = 6
    hf_model = make_hf_model("mistral", cfg)
    assert hf_model.config.sliding_window == 6

    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    hf_reg.load_from_hf_state_dict(model, "mistral", hf_model.state_dict())

    packed, cu, mx = packed_batch(cfg.vocab_size, bs=3, lo=10, hi=20, seed=7)
    assert mx > cfg.sliding_window  # the window must actually bind
    with torch.no_grad():
        ours = model(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
        offs = cu.tolist()
        for i in range(len(offs) - 1):
            s, e = offs[i], offs[i + 1]
            ref = hf_model(input_ids=packed[s:e].unsqueeze(0)).logits[0]
            torch.testing.assert_close(
                torch.log_softmax(ours[s:e].float(), -1),
                torch.log_softmax(ref.float(), -1),
                atol=2e-4,
                rtol=2e-3,
            )
    # and the window must change the result vs full causal
    cfg2 = fam.make_test_config()
    cfg2.dtype = "float32"
    model2 = ReaLModel(cfg2, device="cpu", dtype=torch.float32)
    hf_reg.load_from_hf_state_dict(model2, "mistral", hf_model.state_dict())
    with torch.no_grad():
        full = model2(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    assert not torch.allclose(ours, full)


def test_logprob_gather_matches_hf_loss():
    fam = hf_reg.get_family("llama")
    cfg = fam.make_test_config()
    cfg.dtype = "float32"
    hf_model = make_hf_model("llama", cfg)
    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    hf_reg.load_from_hf_state_dict(model, "llama", hf_model.state_dict())
    packed, cu, mx = packed_batch(cfg.vocab_size, bs=3)
    with torch.no_grad():
        logits = model(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
        logp = gather_packed_shifted_log_probs(logits, cu, packed)
    assert logp.shape[0] == packed.shape[0] - (len(cu) - 1)
    # manual check on first sequence
    s, e = int(cu[0]), int(cu[1])
    ref = torch.log_softmax(logits[s : e - 1].float(), -1)
    ref = ref.gather(-1, packed[s + 1 : e].unsqueeze(-1)).squeeze(-1)
    torch.testing.assert_close(logp[: e - 1 - s], ref)


def test_critic_head():
    fam = hf_reg.get_family("llama")
    cfg = fam.make_test_config(is_critic=True)
    cfg.dtype = "float32"
    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    model.random_init()
    packed, cu, mx = packed_batch(cfg.vocab_size)
    with torch.no_grad():
        out = model(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    assert out.shape == (packed.shape[0], 1)
    assert out.dtype == torch.float32


def test_save_load_roundtrip(tmp_path):
    fam = hf_reg.get_family("llama")
    cfg = fam.make_test_config()
    cfg.dtype = "float32"
    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    model.random_init()
    hf_reg.save_to_hf(model, "llama", str(tmp_path))
    model2 = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    hf_reg.load_from_hf(model2, "llama", str(tmp_path))
    for k in model.layout.keys:
        torch.testing.assert_close(model.param_view(k), model2.param_view(k))
    # and transformers can open it
    import transformers

    m = transformers.AutoModelForCausalLM.from_pretrained(str(tmp_path))
    assert m.config.num_hidden_layers == cfg.n_layers


@pytest.mark.parametrize("scaling_type", ["linear", "dynamic"])
def test_rope_scaling_matches_hf(scaling_type):
    """Linear position-interpolation and dynamic-NTK rope scaling vs
    transformers (reference modules/rotary.py:121 supports both via
    `_update_cos_sin_cache`).  Dynamic only binds past
    max_position_embeddings, so sequences run to 2x that length; HF
    recomputes its table per forward from that forward's seq_len, so we
    compare per-sequence with matching max_seqlen."""
    fam = hf_reg.get_family("llama")
    cfg = fam.make_test_config(max_position_embeddings=16)
    cfg.dtype = "float32"
    cfg.rotary_scaling = 2.0
    cfg.rotary_scaling_type = scaling_type
    hf_model = make_hf_model("llama", cfg)
    assert hf_model.config.rope_scaling["factor"] == 2.0

    model = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    hf_reg.load_from_hf_state_dict(model, "llama", hf_model.state_dict())

    rng = np.random.RandomState(3)
    for L in (12, 32):  # under and over max_position_embeddings
        ids = torch.from_numpy(rng.randint(0, cfg.vocab_size, size=L)).long()
        cu = torch.tensor([0, L], dtype=torch.int32)
        with torch.no_grad():
            ours = model(packed_input_ids=ids, cu_seqlens=cu, max_seqlen=L)
            ref = hf_model(input_ids=ids.unsqueeze(0)).logits[0]
        torch.testing.assert_close(
            torch.log_softmax(ours.float(), -1),
            torch.log_softmax(ref.float(), -1),
            atol=2e-4,
            rtol=2e-3,
        )
    # the scaling must actually change the logits past the window
    cfg2 = fam.make_test_config(max_position_embeddings=16)
    cfg2.dtype = "float32"
    plain = ReaLModel(cfg2, device="cpu", dtype=torch.float32)
    hf_reg.load_from_hf_state_dict(plain, "llama", hf_model.state_dict())
    with torch.no_grad():
        base = plain(packed_input_ids=ids, cu_seqlens=cu, max_seqlen=32)
    assert not torch.allclose(ours, base)


def test_rope_scaling_decode_matches_prefill():
    """The decode path builds its rotary table from the KV-cache length
    (layers.py rot_len = cache_len), so dynamic-NTK must produce the
    same logits step-by-step as a full prefill at the same positions."""
    from realhf_amd.api.model import GenerationHyperparameters
    from realhf_amd.models.generation import generate

    fam = hf_reg.get_family("llama")
    cfg = fam.make_test_config(max_position_embeddings=16)
    cfg.dtype = "float32"
    cfg.rotary_scaling = 2.0
    cfg.rotary_scaling_type = "dynamic"
    from realhf_amd.models.real_model import ReaLModel

    m = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    m.random_init()
    rng = np.random.RandomState(2)
    prompt = torch.from_numpy(rng.randint(0, cfg.vocab_size, size=20)).long()
    cu = torch.tensor([0, 20], dtype=torch.int32)
    g = GenerationHyperparameters(max_new_tokens=4, greedy=True,
                                  use_hip_graph=False)
    out = generate(m, prompt, cu, g, eos_token_id=None, pad_token_id=0)
    # re-score the generated sequence with a fresh full forward: the
    # greedy choice at each step must be reproduced
    full = torch.cat([prompt, out.gen_tokens[0].cpu()])
    L = full.shape[0]
    cu2 = torch.tensor([0, L], dtype=torch.int32)
    with torch.no_grad():
        logits = m(packed_input_ids=full, cu_seqlens=cu2, max_seqlen=L)
    for t in range(4):
        pos = 20 + t - 1  # logits row predicting token 20+t
        assert int(logits[pos].argmax()) == int(out.gen_tokens[0, t])
