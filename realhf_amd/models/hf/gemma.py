"""Gemma HF converters (reference: realhf/api/from_hf/gemma.py).
Gemma = llama naming + gemma-RMSNorm ((1+w) scaling), GeGLU MLP, tied
embeddings, sqrt(hidden) embedding multiplier."""
from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.models.hf import HFFamily, register_family
from realhf_amd.models.hf import llama as L


def config_from_hf(hf: dict) -> ReaLModelConfig:
    cfg = L.config_from_hf(hf)
    cfg.norm_type = "gemma_rms"
    cfg.activation = "geglu"
    cfg.tied_embedding = True
    cfg.head_dim = hf.get("head_dim", cfg.hidden_dim // cfg.n_heads)
    cfg.embedding_multiplier = float(cfg.hidden_dim) ** 0.5
    return cfg


def config_to_hf(cfg: ReaLModelConfig) -> dict:
    out = L.config_to_hf(cfg)
    out["architectures"] = ["GemmaForCausalLM"]
    out["model_type"] = "gemma"
    out["hidden_act"] = "gelu_pytorch_tanh"
    out["tie_word_embeddings"] = True
    return out


def make_test_config(**kw):
    kw.setdefault("norm_type", "gemma_rms")
    kw.setdefault("tied_embedding", True)
    cfg = L.make_test_config(**kw)
    cfg.activation = "geglu"
    cfg.embedding_multiplier = float(cfg.hidden_dim) ** 0.5
    return cfg


register_family(
    HFFamily(
        name="gemma",
        hf_arch="GemmaForCausalLM",
        hf_deps=L.hf_deps,
        from_hf=L.from_hf,
        to_hf=L.to_hf,
        config_from_hf=config_from_hf,
        config_to_hf=config_to_hf,
        make_test_config=make_test_config,
    )
)
