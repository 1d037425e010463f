"""LLaMA-family HF converters (reference: realhf/api/from_hf/llama.py).

Shared by mistral (same tensor names) and, with small deltas, qwen2/gemma.
"""
from typing import Dict, List

import torch

from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.models.hf import HFFamily, register_family


def _canonical_to_hf_key(cfg: ReaLModelConfig, key: str) -> List[str]:
    layer, name = key.split(".", 1)
    li = int(layer)
    if name == "wte.weight":
        return ["model.embed_tokens.weight"]
    if name == "ln_f.weight":
        return ["model.norm.weight"]
    if name == "head.weight":
        if cfg.is_critic:
            return []  # critic head has no HF counterpart
        if cfg.tied_embedding:
            return ["model.embed_tokens.weight"]
        return ["lm_head.weight"]
    l = li - 1
    m = {
        "attn.ln.weight": f"model.layers.{l}.input_layernorm.weight",
        "attn.wq.weight": f"model.layers.{l}.self_attn.q_proj.weight",
        "attn.wk.weight": f"model.layers.{l}.self_attn.k_proj.weight",
        "attn.wv.weight": f"model.layers.{l}.self_attn.v_proj.weight",
        "attn.wq.bias": f"model.layers.{l}.self_attn.q_proj.bias",
        "attn.wk.bias": f"model.layers.{l}.self_attn.k_proj.bias",
        "attn.wv.bias": f"model.layers.{l}.self_attn.v_proj.bias",
        "attn.wo.weight": f"model.layers.{l}.self_attn.o_proj.weight",
        "mlp.ln.weight": f"model.layers.{l}.post_attention_layernorm.weight",
        "mlp.gate.weight": f"model.layers.{l}.mlp.gate_proj.weight",
        "mlp.up.weight": f"model.layers.{l}.mlp.up_proj.weight",
        "mlp.down.weight": f"model.layers.{l}.mlp.down_proj.weight",
    }
    return [m[name]]


def hf_deps(cfg: ReaLModelConfig, key: str) -> List[str]:
    return _canonical_to_hf_key(cfg, key)


def from_hf(cfg, key, deps: Dict[str, torch.Tensor]) -> torch.Tensor:
    (t,) = deps.values()
    return t


def to_hf(cfg: ReaLModelConfig, sd: Dict[str, torch.Tensor]):
    out = {}
    for k, v in sd.items():
        hfks = _canonical_to_hf_key(cfg, k)
        if not hfks:
            out[f"score.{k}"] = v  # critic head saved under a scoring name
            continue
        out[hfks[0]] = v
    return out


def config_from_hf(hf: dict) -> ReaLModelConfig:
    nh = hf["num_attention_heads"]
    # HF rope_scaling {"type"|"rope_type": "linear"|"dynamic", "factor": f}
    # (reference rotary.py:121 supports both) -> rotary_scaling(_type)
    rs = hf.get("rope_scaling") or {}
    rs_type = rs.get("type") or rs.get("rope_type")
    rs_factor = rs.get("factor")
    if rs_type not in (None, "linear", "dynamic"):
        raise NotImplementedError(f"rope_scaling type {rs_type!r}")
    return ReaLModelConfig(
        n_layers=hf["num_hidden_layers"],
        hidden_dim=hf["hidden_size"],
        n_heads=nh,
        n_kv_heads=hf.get("num_key_value_heads", nh),
        head_dim=hf.get("head_dim") or hf["hidden_size"] // nh,
        intermediate_dim=hf["intermediate_size"],
        vocab_size=hf["vocab_size"],
        max_position_embeddings=hf.get("max_position_embeddings", 4096),
        activation="silu",
        norm_type="rms",
        layer_norm_epsilon=hf.get("rms_norm_eps", 1e-5),
        apply_rotary=True,
        rotary_base=hf.get("rope_theta", 10000.0),
        rotary_scaling=rs_factor,
        rotary_scaling_type=rs_type,
        tied_embedding=hf.get("tie_word_embeddings", False),
        use_attention_bias=hf.get("attention_bias", False),
    )


def config_to_hf(cfg: ReaLModelConfig) -> dict:
    return {
        "architectures": ["LlamaForCausalLM"],
        "model_type": "llama",
        "hidden_size": cfg.hidden_dim,
        "num_hidden_layers": cfg.n_layers,
        "num_attention_heads": cfg.n_heads,
        "num_key_value_heads": cfg.n_kv_heads,
        "head_dim": cfg.head_dim,
        "intermediate_size": cfg.intermediate_dim,
        "vocab_size": cfg.vocab_size,
        "max_position_embeddings": cfg.max_position_embeddings,
        "rms_norm_eps": cfg.layer_norm_epsilon,
        "rope_theta": cfg.rotary_base,
        **(
            {"rope_scaling": {"type": cfg.rotary_scaling_type,
                              "factor": cfg.rotary_scaling}}
            if cfg.rotary_scaling_type else {}
        ),
        "tie_word_embeddings": cfg.tied_embedding,
        "attention_bias": cfg.use_attention_bias,
        "hidden_act": "silu",
        "torch_dtype": cfg.dtype,
        "bos_token_id": 1,
        "eos_token_id": 2,
    }


def make_test_config(
    n_layers=2, hidden_dim=32, n_heads=4, n_kv_heads=2, vocab_size=64, **kw
) -> ReaLModelConfig:
    kw.setdefault("head_dim", hidden_dim // n_heads)
    kw.setdefault("intermediate_dim", hidden_dim * 2)
    kw.setdefault("max_position_embeddings", 128)
    kw.setdefault("activation", "silu")
    kw.setdefault("norm_type", "rms")
    return ReaLModelConfig(
        n_layers=n_layers,
        hidden_dim=hidden_dim,
        n_heads=n_heads,
        n_kv_heads=n_kv_heads,
        vocab_size=vocab_size,
        **kw,
    )


def llama7b_config(is_critic: bool = False) -> ReaLModelConfig:
    """LLaMA-2-7B architecture (the BASELINE.json bench model)."""
    return ReaLModelConfig(
        n_layers=32,
        hidden_dim=4096,
        n_heads=32,
        n_kv_heads=32,
        head_dim=128,
        intermediate_dim=11008,
        vocab_size=32000,
        max_position_embeddings=4096,
        activation="silu",
        norm_type="rms",
        is_critic=is_critic,
        family="llama",
    )


def llama70b_config(is_critic: bool = False) -> ReaLModelConfig:
    """LLaMA-2-70B architecture (the 288 GB HBM sizing tier: the whole
    model fits one MI355X in bf16; train tier uses tp + optimizer
    offload)."""
    return ReaLModelConfig(
        n_layers=80,
        hidden_dim=8192,
        n_heads=64,
        n_kv_heads=8,
        head_dim=128,
        intermediate_dim=28672,
        vocab_size=32000,
        max_position_embeddings=4096,
        activation="silu",
        norm_type="rms",
        is_critic=is_critic,
        family="llama",
    )


register_family(
    HFFamily(
        name="llama",
        hf_arch="LlamaForCausalLM",
        hf_deps=hf_deps,
        from_hf=from_hf,
        to_hf=to_hf,
        config_from_hf=config_from_hf,
        config_to_hf=config_to_hf,
        make_test_config=make_test_config,
    )
)
