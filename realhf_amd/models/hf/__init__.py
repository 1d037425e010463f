"""HF checkpoint conversion registry.

Reference semantics: realhf/impl/model/conversion/hf_registry.py
(HFModelRegistry:24) + realhf/api/from_hf/{llama,gpt2,qwen2,gemma,mistral,
mixtral}.py.  Each family provides state-dict and config converters between
HF format and the canonical flat-layout key schema
(realhf_amd/models/param_layout.py).

Loading is shard-aware: only the tensors needed by this (pp stage, tp rank)
are read from safetensors files and TP-partitioned on the fly.  Saved
checkpoints are HF-format (shards + index json) and load directly in
transformers/vLLM.
"""
import dataclasses
import json
import os
from typing import Callable, Dict, List

import torch

from realhf_amd.api.model import ReaLModelConfig, register_hf_family
from realhf_amd.base import logging
from realhf_amd.models import param_layout as PL

logger = logging.getLogger("hf")

MAX_SHARD_BYTES = 10 * 1024**3


@dataclasses.dataclass
class HFFamily:
    name: str
    hf_arch: str
    # canonical key -> list of HF keys it is built from
    hf_deps: Callable[[ReaLModelConfig, str], List[str]]
    # build canonical tensor from {hf_key: tensor}
    from_hf: Callable[[ReaLModelConfig, str, Dict[str, torch.Tensor]], torch.Tensor]
    # {canonical full state dict} -> {hf state dict}
    to_hf: Callable[[ReaLModelConfig, Dict[str, torch.Tensor]], Dict[str, torch.Tensor]]
    config_from_hf: Callable[[dict], ReaLModelConfig]
    config_to_hf: Callable[[ReaLModelConfig], dict]
    make_test_config: Callable[..., ReaLModelConfig]


_FAMILIES: Dict[str, HFFamily] = {}


def register_family(fam: HFFamily):
    _FAMILIES[fam.name] = fam
    register_hf_family(fam.name, fam)


def get_family(name: str) -> HFFamily:
    return _FAMILIES[name]


def family_names():
    return sorted(_FAMILIES)


# ---------------------------------------------------------------------------
# loading
# ---------------------------------------------------------------------------
def _read_hf_index(path: str) -> Dict[str, str]:
    """hf key -> filename."""
    idx_json = os.path.join(path, "model.safetensors.index.json")
    if os.path.exists(idx_json):
        with open(idx_json) as f:
            return json.load(f)["weight_map"]
    single = os.path.join(path, "model.safetensors")
    if os.path.exists(single):
        from safetensors import safe_open

        with safe_open(single, framework="pt") as f:
            return {k: "model.safetensors" for k in f.keys()}
    pt = os.path.join(path, "pytorch_model.bin")
    if os.path.exists(pt):
        return {}
    raise FileNotFoundError(f"no HF checkpoint found under {path}")


def load_hf_config(path: str) -> dict:
    with open(os.path.join(path, "config.json")) as f:
        return json.load(f)


def config_from_hf_path(family: str, path: str) -> ReaLModelConfig:
    fam = get_family(family)
    cfg = fam.config_from_hf(load_hf_config(path))
    cfg.family = family
    cfg.base_model_path = path
    return cfg


def load_from_hf(model, family: str, path: str):
    """Fill `model`'s flat buffer from an HF checkpoint: reads only the
    tensors this shard needs, TP-partitions on the fly (reference:
    hf_registry.py:62)."""
    fam = get_family(family)
    cfg = model.config
    weight_map = _read_hf_index(path)

    needed: Dict[str, List[str]] = {}
    for k in model.layout.keys:
        needed[k] = fam.hf_deps(cfg, k)
    hf_keys = {h for deps in needed.values() for h in deps}

    hf_tensors: Dict[str, torch.Tensor] = {}
    if weight_map:
        by_file: Dict[str, List[str]] = {}
        for h in hf_keys:
            if h not in weight_map:
                continue
            by_file.setdefault(weight_map[h], []).append(h)
        from concurrent.futures import ThreadPoolExecutor

        from safetensors import safe_open

        def _load_file(fn_ks):
            fn, ks = fn_ks
            out = {}
            with safe_open(os.path.join(path, fn), framework="pt") as f:
                for h in ks:
                    out[h] = f.get_tensor(h)
            return out

        # thread-pool shard loading (reference: hf_registry.py:62 loads
        # shards concurrently — IO-bound, so threads suffice)
        if by_file:
            with ThreadPoolExecutor(max_workers=min(8, len(by_file))) as ex:
                for part in ex.map(_load_file, by_file.items()):
                    hf_tensors.update(part)
    else:
        full = torch.load(
            os.path.join(path, "pytorch_model.bin"), map_location="cpu",
            weights_only=True,
        )
        hf_tensors = {h: full[h] for h in hf_keys if h in full}

    return load_from_hf_state_dict(model, family, hf_tensors)


def load_from_hf_state_dict(model, family: str, hf_tensors: Dict[str, torch.Tensor]):
    """In-memory variant: fill `model` from an HF-format state dict."""
    fam = get_family(family)
    cfg = model.config
    with torch.no_grad():
        for k in model.layout.keys:
            deps_keys = fam.hf_deps(cfg, k)
            deps = {h: hf_tensors[h] for h in deps_keys if h in hf_tensors}
            if len(deps) < len(deps_keys):
                missing = set(deps_keys) - set(deps)
                if k.endswith("head.weight") and cfg.is_critic:
                    model.param_view(k).normal_(0.0, 1.0 / (cfg.hidden_dim**0.5))
                    continue
                raise KeyError(f"HF checkpoint missing {missing} for {k}")
            full_t = fam.from_hf(cfg, k, deps)
            shard = PL.tp_partition(cfg, k, full_t, model.tp_rank, model.tp_size)
            model.param_view(k).copy_(shard.to(model.dtype))
    return model


# ---------------------------------------------------------------------------
# saving
# ---------------------------------------------------------------------------
def save_to_hf(model, family: str, save_dir: str, tokenizer=None):
    """Save an HF-format checkpoint.  TP shards are merged via the model
    scope's TP group when tp_size > 1 (rank 0 of each (pp, dp=0) writes
    its stage's tensors); with pp > 1 each stage writes its own shard file
    and stage 0 writes the index."""
    import torch.distributed as dist

    from realhf_amd.base import constants

    fam = get_family(family)
    cfg = model.config
    os.makedirs(save_dir, exist_ok=True)

    # merge TP shards of this stage
    full_sd: Dict[str, torch.Tensor] = {}
    tp_size = model.tp_size
    for k in model.layout.keys:
        local = model.param_view(k)
        if tp_size == 1:
            full_sd[k] = local.detach().cpu()
        else:
            shards = [torch.empty_like(local) for _ in range(tp_size)]
            dist.all_gather(shards, local.contiguous(), group=constants.tp_group())
            full_sd[k] = PL.tp_merge(cfg, k, [s.cpu() for s in shards])

    write = True
    if constants.has_current():
        g = constants.grid()
        write = g.tp_rank == 0 and g.dp_rank == 0
    if not write:
        return

    hf_sd = fam.to_hf(cfg, full_sd)
    stage_tag = f"-p{model.pp_rank:02d}" if model.pp_size > 1 else ""
    from safetensors.torch import save_file

    # split into <=10GB shards
    shards, cur, cur_bytes = [], {}, 0
    for k, v in hf_sd.items():
        b = v.numel() * v.element_size()
        if cur and cur_bytes + b > MAX_SHARD_BYTES:
            shards.append(cur)
            cur, cur_bytes = {}, 0
        cur[k] = v.contiguous()
        cur_bytes += b
    if cur:
        shards.append(cur)

    weight_map = {}
    n = len(shards)
    for i, sd in enumerate(shards):
        fn = (
            "model.safetensors"
            if n == 1 and model.pp_size == 1
            else f"model{stage_tag}-{i + 1:05d}-of-{n:05d}.safetensors"
        )
        save_file(sd, os.path.join(save_dir, fn))
        for k in sd:
            weight_map[k] = fn

    if model.pp_size > 1:
        # each stage writes its partial map; stage writers then merge
        with open(
            os.path.join(save_dir, f"weight_map{stage_tag}.json"), "w"
        ) as f:
            json.dump(weight_map, f)
        if constants.has_current():
            dist.barrier(group=constants.model_group())
        if model.pp_rank == 0:
            merged = {}
            for p in range(model.pp_size):
                fn = os.path.join(save_dir, f"weight_map-p{p:02d}.json")
                with open(fn) as f:
                    merged.update(json.load(f))
                os.remove(fn)
            _write_index(save_dir, merged)
    elif n > 1:
        _write_index(save_dir, weight_map)

    if model.pp_rank == 0 or model.pp_size == 1:
        with open(os.path.join(save_dir, "config.json"), "w") as f:
            json.dump(fam.config_to_hf(cfg), f, indent=2)
        if tokenizer is not None:
            tokenizer.save_pretrained(save_dir)


def _write_index(save_dir, weight_map):
    with open(os.path.join(save_dir, "model.safetensors.index.json"), "w") as f:
        json.dump({"metadata": {}, "weight_map": weight_map}, f)


from realhf_amd.models.hf import gemma, gpt2, llama, mistral, mixtral, qwen2  # noqa: E402,F401
