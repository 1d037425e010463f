"""Qwen2 HF converters (reference: realhf/api/from_hf/qwen2.py).
Qwen2 = llama tensor naming + QKV biases."""
from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.models.hf import HFFamily, register_family
from realhf_amd.models.hf import llama as L


def config_from_hf(hf: dict) -> ReaLModelConfig:
    cfg = L.config_from_hf(hf)
    cfg.use_attention_bias = True
    return cfg


def config_to_hf(cfg: ReaLModelConfig) -> dict:
    out = L.config_to_hf(cfg)
    out["architectures"] = ["Qwen2ForCausalLM"]
    out["model_type"] = "qwen2"
    out["attention_bias"] = True
    return out


def make_test_config(**kw):
    kw.setdefault("use_attention_bias", True)
    return L.make_test_config(**kw)


register_family(
    HFFamily(
        name="qwen2",
        hf_arch="Qwen2ForCausalLM",
        hf_deps=L.hf_deps,
        from_hf=L.from_hf,
        to_hf=L.to_hf,
        config_from_hf=config_from_hf,
        config_to_hf=config_to_hf,
        make_test_config=make_test_config,
    )
)
