"""Mistral HF converters (reference: realhf/api/from_hf/mistral.py).
Same tensor naming as llama; sliding-window attention is honored when
`sliding_window` is set in the HF config (masking in ops.attn_varlen /
ops.attn_decode; the MFMA fast paths run whenever the window is at least
as wide as the sequence/cache, i.e. whenever it cannot bind)."""
from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.models.hf import HFFamily, register_family
from realhf_amd.models.hf import llama as L


def config_from_hf(hf: dict) -> ReaLModelConfig:
    cfg = L.config_from_hf(hf)
    cfg.sliding_window = hf.get("sliding_window")
    return cfg


def config_to_hf(cfg: ReaLModelConfig) -> dict:
    out = L.config_to_hf(cfg)
    out["architectures"] = ["MistralForCausalLM"]
    out["model_type"] = "mistral"
    out["sliding_window"] = cfg.sliding_window
    return out


register_family(
    HFFamily(
        name="mistral",
        hf_arch="MistralForCausalLM",
        hf_deps=L.hf_deps,
        from_hf=L.from_hf,
        to_hf=L.to_hf,
        config_from_hf=config_from_hf,
        config_to_hf=config_to_hf,
        make_test_config=L.make_test_config,
    )
)
