"""The flat-parameter layout contract.

Reference semantics: realhf/impl/model/nn/real_llm_base.py:394
(ReaLModelParamKeys — the ordered key schema), flatten_param.py:213
(build_param_spec), real_llm_parallel.py (mp_partition_key:129,
partition_pipeline_layers:342, pipeline_repartition_strategy:378).

Layer indexing: 0 = embedding, 1..n_layers = transformer blocks,
n_layers+1 = head (final norm + lm/critic head).

Every model shard owns ONE contiguous flat buffer holding its TP shard of
its PP stage's layers, in the canonical key order below.  This ordering is
the contract that makes parameter reallocation pure interval math, and it
deliberately places {wq,wk,wv} and {gate,up} adjacently so their flat
regions concatenate into a single GEMM operand view (merged QKV / merged
gate-up GEMMs without a separate merged tensor).
"""
import dataclasses
from typing import Dict, List, Tuple

import torch

from realhf_amd.api.model import ReaLModelConfig

# TP partition kind per key suffix
COLUMN = "column"  # split dim 0 (output features)
ROW = "row"  # split dim 1 (input features)
REPLICATED = "replicated"
VOCAB = "vocab"  # split dim 0 = vocab (embedding & lm head)


def align(n: int, a: int = 64) -> int:
    return (n + a - 1) // a * a


# ---------------------------------------------------------------------------
# key schema
# ---------------------------------------------------------------------------
def embed_keys(cfg: ReaLModelConfig) -> List[str]:
    ks = ["0.wte.weight"]
    if cfg.use_abs_position_embedding:
        ks.append("0.wpe.weight")
    return ks


def tblock_keys(cfg: ReaLModelConfig, layer_idx: int) -> List[str]:
    """layer_idx is 1-based (0 is the embedding)."""
    i = layer_idx
    ks = [f"{i}.attn.ln.weight"]
    if cfg.norm_type == "layer":
        ks.append(f"{i}.attn.ln.bias")
    ks += [f"{i}.attn.wq.weight", f"{i}.attn.wk.weight", f"{i}.attn.wv.weight"]
    if cfg.use_attention_bias:
        ks += [f"{i}.attn.wq.bias", f"{i}.attn.wk.bias", f"{i}.attn.wv.bias"]
    ks.append(f"{i}.attn.wo.weight")
    if cfg.use_attn_proj_bias:
        ks.append(f"{i}.attn.wo.bias")
    ks.append(f"{i}.mlp.ln.weight")
    if cfg.norm_type == "layer":
        ks.append(f"{i}.mlp.ln.bias")
    if cfg.moe is not None:
        ks.append(f"{i}.mlp.router.weight")
        for e in range(cfg.moe.num_experts):
            ks += [
                f"{i}.mlp.experts.{e}.gate.weight",
                f"{i}.mlp.experts.{e}.up.weight",
                f"{i}.mlp.experts.{e}.down.weight",
            ]
    elif cfg.activation in ("silu", "geglu"):
        ks += [f"{i}.mlp.gate.weight", f"{i}.mlp.up.weight", f"{i}.mlp.down.weight"]
        if cfg.use_mlp_bias:
            ks += [f"{i}.mlp.gate.bias", f"{i}.mlp.up.bias", f"{i}.mlp.down.bias"]
    else:  # gelu (gpt2-style): fc (up) + proj (down)
        ks += [f"{i}.mlp.up.weight", f"{i}.mlp.down.weight"]
        if cfg.use_mlp_bias:
            ks += [f"{i}.mlp.up.bias", f"{i}.mlp.down.bias"]
    return ks


def head_keys(cfg: ReaLModelConfig) -> List[str]:
    i = cfg.n_layers + 1
    ks = [f"{i}.ln_f.weight"]
    if cfg.norm_type == "layer":
        ks.append(f"{i}.ln_f.bias")
    if not (cfg.tied_embedding and not cfg.is_critic):
        ks.append(f"{i}.head.weight")
    return ks


def keys_of_layer(cfg: ReaLModelConfig, layer_idx: int) -> List[str]:
    if layer_idx == 0:
        return embed_keys(cfg)
    if layer_idx == cfg.n_layers + 1:
        return head_keys(cfg)
    assert 1 <= layer_idx <= cfg.n_layers
    return tblock_keys(cfg, layer_idx)


def keys_of_layers(cfg: ReaLModelConfig, layer_indices: List[int]) -> List[str]:
    out = []
    for i in sorted(layer_indices):
        out += keys_of_layer(cfg, i)
    return out


def all_keys(cfg: ReaLModelConfig) -> List[str]:
    return keys_of_layers(cfg, list(range(cfg.n_layers + 2)))


# ---------------------------------------------------------------------------
# shapes & TP partitioning
# ---------------------------------------------------------------------------
def key_kind(key: str) -> str:
    name = key.split(".", 1)[1]  # strip layer idx
    if name in ("wte.weight",):
        return VOCAB
    if name == "head.weight":
        return "head"  # vocab-split unless critic (decided in key_shape)
    base = name.rsplit(".", 1)[0]
    leaf = name.rsplit(".", 1)[1]
    if base.endswith(("wq", "wk", "wv", "gate", "up")):
        return COLUMN  # both .weight and .bias split on dim 0
    if base.endswith(("wo", "down")):
        return ROW if leaf == "weight" else REPLICATED
    return REPLICATED


def key_full_shape(cfg: ReaLModelConfig, key: str) -> Tuple[int, ...]:
    h = cfg.hidden_dim
    name = key.split(".", 1)[1]
    qd = cfg.n_heads * cfg.head_dim
    kvd = cfg.n_kv_heads * cfg.head_dim
    idim = cfg.intermediate_dim
    table = {
        "wte.weight": (cfg.vocab_size, h),
        "wpe.weight": (cfg.max_position_embeddings, h),
        "attn.ln.weight": (h,),
        "attn.ln.bias": (h,),
        "attn.wq.weight": (qd, h),
        "attn.wk.weight": (kvd, h),
        "attn.wv.weight": (kvd, h),
        "attn.wq.bias": (qd,),
        "attn.wk.bias": (kvd,),
        "attn.wv.bias": (kvd,),
        "attn.wo.weight": (h, qd),
        "attn.wo.bias": (h,),
        "mlp.ln.weight": (h,),
        "mlp.ln.bias": (h,),
        "mlp.gate.weight": (idim, h),
        "mlp.up.weight": (idim, h) if cfg.activation == "silu" else (idim, h),
        "mlp.down.weight": (h, idim),
        "mlp.gate.bias": (idim,),
        "mlp.up.bias": (idim,),
        "mlp.down.bias": (h,),
        "ln_f.weight": (h,),
        "ln_f.bias": (h,),
        "head.weight": (1, h) if cfg.is_critic else (cfg.vocab_size, h),
    }
    if name in table:
        return table[name]
    if ".experts." in name:
        # mlp.experts.{e}.{gate|up|down}.weight
        part = name.split(".")[-2]
        if part == "down":
            return (h, idim)
        return (idim, h)
    if name == "mlp.router.weight":
        return (cfg.moe.num_experts, h)
    raise KeyError(key)


def key_local_shape(
    cfg: ReaLModelConfig, key: str, tp_rank: int, tp_size: int
) -> Tuple[int, ...]:
    full = key_full_shape(cfg, key)
    kind = key_kind(key)
    if kind == "head":
        kind = REPLICATED if cfg.is_critic else VOCAB
    if kind in (COLUMN, VOCAB):
        assert full[0] % tp_size == 0, (key, full, tp_size)
        return (full[0] // tp_size,) + tuple(full[1:])
    if kind == ROW:
        assert full[1] % tp_size == 0, (key, full, tp_size)
        return (full[0], full[1] // tp_size)
    return full


def tp_partition(
    cfg: ReaLModelConfig, key: str, tensor: torch.Tensor, tp_rank: int, tp_size: int
) -> torch.Tensor:
    """Slice the FULL tensor down to this TP rank's shard."""
    if tp_size == 1:
        return tensor
    kind = key_kind(key)
    if kind == "head":
        kind = REPLICATED if cfg.is_critic else VOCAB
    if kind in (COLUMN, VOCAB):
        n = tensor.shape[0] // tp_size
        return tensor[tp_rank * n : (tp_rank + 1) * n]
    if kind == ROW:
        n = tensor.shape[1] // tp_size
        return tensor[:, tp_rank * n : (tp_rank + 1) * n]
    return tensor


def tp_merge(
    cfg: ReaLModelConfig, key: str, shards: List[torch.Tensor]
) -> torch.Tensor:
    """Merge TP shards back into the full tensor."""
    if len(shards) == 1:
        return shards[0]
    kind = key_kind(key)
    if kind == "head":
        kind = REPLICATED if cfg.is_critic else VOCAB
    if kind in (COLUMN, VOCAB):
        return torch.cat(shards, dim=0)
    if kind == ROW:
        return torch.cat(shards, dim=1)
    return shards[0]


# ---------------------------------------------------------------------------
# pipeline layer partition
# ---------------------------------------------------------------------------
def layer_param_count(cfg: ReaLModelConfig, layer_idx: int) -> int:
    total = 0
    for k in keys_of_layer(cfg, layer_idx):
        shape = key_full_shape(cfg, k)
        n = 1
        for s in shape:
            n *= s
        total += n
    return total


def partition_pipeline_layers(
    cfg: ReaLModelConfig, pp_size: int
) -> Dict[int, Tuple[int, int]]:
    """Balanced-by-param-count contiguous partition of layers 0..n_layers+1
    over pp stages (reference: real_llm_parallel.py:342).
    Returns {stage: (start_layer_incl, end_layer_excl)}."""
    from realhf_amd.base.datapack import min_abs_diff_partition

    counts = [layer_param_count(cfg, i) for i in range(cfg.n_layers + 2)]
    bounds = min_abs_diff_partition(counts, pp_size)
    return {s: b for s, b in enumerate(bounds)}


def pipeline_repartition_strategy(
    src_map: Dict[int, Tuple[int, int]], dst_map: Dict[int, Tuple[int, int]]
) -> Dict[Tuple[int, int], List[int]]:
    """For realloc: {(src_stage, dst_stage): [layer indices moved]}
    (reference: real_llm_parallel.py:378)."""
    out = {}
    for si, (s0, s1) in src_map.items():
        for di, (d0, d1) in dst_map.items():
            inter = list(range(max(s0, d0), min(s1, d1)))
            if inter:
                out[(si, di)] = inter
    return out


# ---------------------------------------------------------------------------
# flat buffer spec
# ---------------------------------------------------------------------------
@dataclasses.dataclass
class ParamSpec:
    start: int
    end: int
    shape: Tuple[int, ...]

    @property
    def numel(self):
        n = 1
        for s in self.shape:
            n *= s
        return n


@dataclasses.dataclass
class FlatLayout:
    """Flat-buffer layout for one shard: ordered keys → [start, end) in the
    contiguous buffer.  Offsets are element counts, each param aligned to
    64 elements (128 B in bf16) for vectorized HIP access."""

    specs: Dict[str, ParamSpec]
    total_numel: int
    keys: List[str]

    def bounds(self, key: str) -> Tuple[int, int]:
        s = self.specs[key]
        return s.start, s.end


def filter_keys_for_ep(cfg: ReaLModelConfig, keys: List[str], ep_rank: int,
                       ep_size: int) -> List[str]:
    """With expert parallelism, each rank stores only its expert block."""
    if ep_size <= 1 or cfg.moe is None:
        return keys
    n_local = cfg.moe.num_experts // ep_size
    lo, hi = ep_rank * n_local, (ep_rank + 1) * n_local
    out = []
    for k in keys:
        if ".experts." in k:
            e = int(k.split(".experts.")[1].split(".")[0])
            if not (lo <= e < hi):
                continue
        out.append(k)
    return out


def build_flat_layout(
    cfg: ReaLModelConfig,
    layer_indices: List[int],
    tp_rank: int,
    tp_size: int,
    alignment: int = 64,
    ep_rank: int = 0,
    ep_size: int = 1,
) -> FlatLayout:
    keys = filter_keys_for_ep(cfg, keys_of_layers(cfg, layer_indices), ep_rank, ep_size)
    specs = {}
    off = 0
    for k in keys:
        shape = key_local_shape(cfg, k, tp_rank, tp_size)
        n = 1
        for s in shape:
            n *= s
        specs[k] = ParamSpec(start=off, end=off + n, shape=tuple(shape))
        off = align(off + n, alignment)
    return FlatLayout(specs=specs, total_numel=off, keys=keys)
