"""Mistral HF converters (reference: realhf/api/from_hf/mistral.py).
Same tensor naming as llama.  Sliding-window attention is NOT applied
(packing keeps sequences <= window in the supported configs; gap noted
in README parity table)."""
from realhf_amd.api.model import ReaLModelConfig
from realhf_amd.models.hf import HFFamily, register_family
from realhf_amd.models.hf import llama as L


def config_to_hf(cfg: ReaLModelConfig) -> dict:
    out = L.config_to_hf(cfg)
    out["architectures"] = ["MistralForCausalLM"]
    out["model_type"] = "mistral"
    out["sliding_window"] = None
    return out


register_family(
    HFFamily(
        name="mistral",
        hf_arch="MistralForCausalLM",
        hf_deps=L.hf_deps,
        from_hf=L.from_hf,
        to_hf=L.to_hf,
        config_from_hf=L.config_from_hf,
        config_to_hf=config_to_hf,
        make_test_config=L.make_test_config,
    )
)
