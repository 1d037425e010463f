"""GPT-2 HF converters (reference: realhf/api/from_hf/gpt2.py).

GPT-2 uses Conv1D weights (stored transposed vs nn.Linear), fused c_attn,
learned absolute positions, LayerNorm, gelu MLP, tied lm head.
"""
from typing import Dict, List

import torch

from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.models.hf import HFFamily, register_family


def _pfx(hf: str) -> str:
    return hf


def hf_deps(cfg: ReaLModelConfig, key: str) -> List[str]:
    layer, name = key.split(".", 1)
    li = int(layer)
    if name == "wte.weight":
        return ["transformer.wte.weight"]
    if name == "wpe.weight":
        return ["transformer.wpe.weight"]
    if name in ("ln_f.weight", "ln_f.bias"):
        return [f"transformer.{name}"]
    if name == "head.weight":
        return [] if cfg.is_critic else ["transformer.wte.weight"]
    l = li - 1
    if name.startswith("attn.ln."):
        return [f"transformer.h.{l}.ln_1.{name.split('.')[-1]}"]
    if name.startswith("mlp.ln."):
        return [f"transformer.h.{l}.ln_2.{name.split('.')[-1]}"]
    if name.startswith(("attn.wq.", "attn.wk.", "attn.wv.")):
        leaf = name.split(".")[-1]
        return [f"transformer.h.{l}.attn.c_attn.{leaf}"]
    if name.startswith("attn.wo."):
        return [f"transformer.h.{l}.attn.c_proj.{name.split('.')[-1]}"]
    if name.startswith("mlp.up."):
        return [f"transformer.h.{l}.mlp.c_fc.{name.split('.')[-1]}"]
    if name.startswith("mlp.down."):
        return [f"transformer.h.{l}.mlp.c_proj.{name.split('.')[-1]}"]
    raise KeyError(key)


def from_hf(cfg: ReaLModelConfig, key: str, deps: Dict[str, torch.Tensor]):
    (t,) = deps.values()
    name = key.split(".", 1)[1]
    h = cfg.hidden_dim
    if name.startswith(("attn.wq", "attn.wk", "attn.wv")):
        part = {"q": 0, "k": 1, "v": 2}[name.split(".")[1][1]]
        if name.endswith("weight"):
            # c_attn.weight is Conv1D [h, 3h] -> take column block, transpose
            return t[:, part * h : (part + 1) * h].t().contiguous()
        return t[part * h : (part + 1) * h]
    if name in ("attn.wo.weight", "mlp.up.weight", "mlp.down.weight"):
        return t.t().contiguous()  # Conv1D -> Linear
    return t


def to_hf(cfg: ReaLModelConfig, sd: Dict[str, torch.Tensor]):
    out = {}
    h = cfg.hidden_dim
    for l in range(cfg.n_layers):
        i = l + 1
        out[f"transformer.h.{l}.ln_1.weight"] = sd[f"{i}.attn.ln.weight"]
        out[f"transformer.h.{l}.ln_1.bias"] = sd[f"{i}.attn.ln.bias"]
        out[f"transformer.h.{l}.attn.c_attn.weight"] = torch.cat(
            [sd[f"{i}.attn.w{c}.weight"].t() for c in "qkv"], dim=1
        ).contiguous()
        out[f"transformer.h.{l}.attn.c_attn.bias"] = torch.cat(
            [sd[f"{i}.attn.w{c}.bias"] for c in "qkv"]
        )
        out[f"transformer.h.{l}.attn.c_proj.weight"] = sd[f"{i}.attn.wo.weight"].t().contiguous()
        out[f"transformer.h.{l}.attn.c_proj.bias"] = sd[f"{i}.attn.wo.bias"]
        out[f"transformer.h.{l}.ln_2.weight"] = sd[f"{i}.mlp.ln.weight"]
        out[f"transformer.h.{l}.ln_2.bias"] = sd[f"{i}.mlp.ln.bias"]
        out[f"transformer.h.{l}.mlp.c_fc.weight"] = sd[f"{i}.mlp.up.weight"].t().contiguous()
        out[f"transformer.h.{l}.mlp.c_fc.bias"] = sd[f"{i}.mlp.up.bias"]
        out[f"transformer.h.{l}.mlp.c_proj.weight"] = sd[f"{i}.mlp.down.weight"].t().contiguous()
        out[f"transformer.h.{l}.mlp.c_proj.bias"] = sd[f"{i}.mlp.down.bias"]
    out["transformer.wte.weight"] = sd["0.wte.weight"]
    out["transformer.wpe.weight"] = sd["0.wpe.weight"]
    L = cfg.n_layers + 1
    out["transformer.ln_f.weight"] = sd[f"{L}.ln_f.weight"]
    out["transformer.ln_f.bias"] = sd[f"{L}.ln_f.bias"]
    if cfg.is_critic and f"{L}.head.weight" in sd:
        out[f"score.{L}.head.weight"] = sd[f"{L}.head.weight"]
    return out


def config_from_hf(hf: dict) -> ReaLModelConfig:
    return ReaLModelConfig(
        n_layers=hf["n_layer"],
        hidden_dim=hf["n_embd"],
        n_heads=hf["n_head"],
        n_kv_heads=hf["n_head"],
        head_dim=hf["n_embd"] // hf["n_head"],
        intermediate_dim=hf.get("n_inner") or 4 * hf["n_embd"],
        vocab_size=hf["vocab_size"],
        max_position_embeddings=hf.get("n_positions", 1024),
        activation="gelu",
        norm_type="layer",
        layer_norm_epsilon=hf.get("layer_norm_epsilon", 1e-5),
        apply_rotary=False,
        use_abs_position_embedding=True,
        use_attention_bias=True,
        use_attn_proj_bias=True,
        use_mlp_bias=True,
        tied_embedding=True,
        scale_attn_by_inverse_layer_idx=hf.get("scale_attn_by_inverse_layer_idx", False),
    )


def config_to_hf(cfg: ReaLModelConfig) -> dict:
    return {
        "architectures": ["GPT2LMHeadModel"],
        "model_type": "gpt2",
        "n_layer": cfg.n_layers,
        "n_embd": cfg.hidden_dim,
        "n_head": cfg.n_heads,
        "n_inner": cfg.intermediate_dim,
        "n_positions": cfg.max_position_embeddings,
        "n_ctx": cfg.max_position_embeddings,
        "vocab_size": cfg.vocab_size,
        "layer_norm_epsilon": cfg.layer_norm_epsilon,
        "activation_function": "gelu_new",
        "scale_attn_by_inverse_layer_idx": cfg.scale_attn_by_inverse_layer_idx,
        "tie_word_embeddings": True,
        "bos_token_id": 0,
        "eos_token_id": 0,
    }


def make_test_config(n_layers=2, hidden_dim=32, n_heads=4, vocab_size=64, **kw):
    return ReaLModelConfig(
        n_layers=n_layers,
        hidden_dim=hidden_dim,
        n_heads=n_heads,
        n_kv_heads=n_heads,
        head_dim=hidden_dim // n_heads,
        intermediate_dim=hidden_dim * 4,
        vocab_size=vocab_size,
        max_position_embeddings=128,
        activation="gelu",
        norm_type="layer",
        apply_rotary=False,
        use_abs_position_embedding=True,
        use_attention_bias=True,
        use_attn_proj_bias=True,
        use_mlp_bias=True,
        tied_embedding=True,
        **kw,
    )


register_family(
    HFFamily(
        name="gpt2",
        hf_arch="GPT2LMHeadModel",
        hf_deps=hf_deps,
        from_hf=from_hf,
        to_hf=to_hf,
        config_from_hf=config_from_hf,
        config_to_hf=config_to_hf,
        make_test_config=make_test_config,
    )
)
