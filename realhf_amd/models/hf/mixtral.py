"""Mixtral HF converters (reference: realhf/api/from_hf/mixtral.py).
Mixtral = mistral attention + block-sparse MoE MLP (8 experts, top-2)."""
from typing import Dict, List

import torch

from realhf_amd.api.model import MoEConfig, ReaLModelConfig
from realhf_amd.models.hf import HFFamily, register_family
from realhf_amd.models.hf import llama as L


def hf_deps(cfg: ReaLModelConfig, key: str) -> List[str]:
    """Uses the transformers>=5 fused expert format
    (mlp.experts.gate_up_proj [E, 2i, h] / down_proj [E, h, i]);
    the pre-5.x per-expert w1/w2/w3 layout is converted by transformers
    itself on load."""
    layer, name = key.split(".", 1)
    li = int(layer)
    l = li - 1
    if name == "mlp.router.weight":
        return [f"model.layers.{l}.mlp.gate.weight"]
    if ".experts." in name:
        part = name.split(".")[3]
        if part == "down":
            return [f"model.layers.{l}.mlp.experts.down_proj"]
        return [f"model.layers.{l}.mlp.experts.gate_up_proj"]
    if name == "mlp.ln.weight":
        return [f"model.layers.{l}.post_attention_layernorm.weight"]
    return L.hf_deps(cfg, key)


def from_hf(cfg, key, deps):
    name = key.split(".", 1)[1]
    if ".experts." in name:
        (t,) = deps.values()
        e = int(name.split(".")[2])
        part = name.split(".")[3]
        idim = cfg.intermediate_dim
        if part == "down":
            return t[e]  # [h, idim]
        if part == "gate":
            return t[e, :idim]
        return t[e, idim:]
    return L.from_hf(cfg, key, deps)


def to_hf(cfg: ReaLModelConfig, sd: Dict[str, torch.Tensor]):
    out = {}
    idim = cfg.intermediate_dim
    h = cfg.hidden_dim
    E = cfg.moe.num_experts
    for l in range(cfg.n_layers):
        i = l + 1
        out[f"model.layers.{l}.mlp.gate.weight"] = sd[f"{i}.mlp.router.weight"]
        gu = torch.empty(E, 2 * idim, h, dtype=sd[f"{i}.mlp.router.weight"].dtype)
        dn = torch.empty(E, h, idim, dtype=gu.dtype)
        for e in range(E):
            gu[e, :idim] = sd[f"{i}.mlp.experts.{e}.gate.weight"]
            gu[e, idim:] = sd[f"{i}.mlp.experts.{e}.up.weight"]
            dn[e] = sd[f"{i}.mlp.experts.{e}.down.weight"]
        out[f"model.layers.{l}.mlp.experts.gate_up_proj"] = gu
        out[f"model.layers.{l}.mlp.experts.down_proj"] = dn
    for k, v in sd.items():
        name = k.split(".", 1)[1]
        if name == "mlp.router.weight" or ".experts." in name:
            continue
        out.update(L.to_hf(cfg, {k: v}))
    return out


def config_from_hf(hf: dict) -> ReaLModelConfig:
    cfg = L.config_from_hf(hf)
    cfg.moe = MoEConfig(
        num_experts=hf.get("num_local_experts", 8),
        top_k=hf.get("num_experts_per_tok", 2),
        routing_type="aux_loss",
        aux_loss_coef=hf.get("router_aux_loss_coef", 0.02),
    )
    return cfg


def config_to_hf(cfg: ReaLModelConfig) -> dict:
    out = L.config_to_hf(cfg)
    out["architectures"] = ["MixtralForCausalLM"]
    out["model_type"] = "mixtral"
    out["num_local_experts"] = cfg.moe.num_experts
    out["num_experts_per_tok"] = cfg.moe.top_k
    out["router_aux_loss_coef"] = cfg.moe.aux_loss_coef
    return out


def make_test_config(**kw):
    cfg = L.make_test_config(**kw)
    cfg.moe = MoEConfig(num_experts=4, top_k=2)
    return cfg


register_family(
    HFFamily(
        name="mixtral",
        hf_arch="MixtralForCausalLM",
        hf_deps=hf_deps,
        from_hf=from_hf,
        to_hf=to_hf,
        config_from_hf=config_from_hf,
        config_to_hf=config_to_hf,
        make_test_config=make_test_config,
    )
)
