"""LoRA adapters on the flat-param model (reference: the "lora"
ModelWrapperAbstraction — api/quickstart/model.py:56 LoRAConfig, :204
wrapper over attention projections with the "squash" op)."""
import numpy as np
import torch

import realhf_amd.interfaces  # noqa: F401
import realhf_amd.models.hf as hf_reg
import realhf_amd.runtime.engine  # noqa: F401
from realhf_amd.api.config import Abstraction
from realhf_amd.api.model import FinetuneSpec, make_backend, make_interface
from tests.test_interfaces import make_model, sft_batch
from tests.test_model_cpu import packed_batch


def test_lora_attach_is_identity_at_init():
    """B starts at zero, so attaching LoRA must not change the forward."""
    model = make_model("llama").module
    packed, cu, mx = packed_batch(model.config.vocab_size, seed=3)
    with torch.no_grad():
        before = model(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    model.attach_lora(dim=4, scaling=8.0)
    assert model.lora_flat.numel() > 0
    with torch.no_grad():
        after = model(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    torch.testing.assert_close(before, after)


def test_lora_sft_learns_and_base_frozen():
    model = make_model("llama")
    model.module.attach_lora(dim=8, scaling=16.0)
    base_before = model.module.flat_param.clone()
    backend = make_backend(
        Abstraction("zero1", {"optimizer": {"lr": 1e-2,
                                            "warmup_steps_proportion": 0.0,
                                            "lr_scheduler_type": "constant"}})
    )
    model = backend.initialize(model, FinetuneSpec(1, 64, 8))
    iface = make_interface(Abstraction("sft"))
    batch = sft_batch(model.module.module.config.vocab_size)
    losses = []
    for _ in range(20):
        stats = iface.train_step(model, batch, n_mbs=2)
        losses.append(stats["loss"])
    # rank-8 adapters on a random tiny model learn slower than full FT;
    # require a clear monotone-ish improvement
    assert losses[-1] < losses[0] * 0.95, losses
    m = model.module.module
    torch.testing.assert_close(m.flat_param, base_before)  # base untouched
    assert float(m.lora_flat.abs().sum()) > 0  # adapters moved


def test_lora_squash_matches_adapter_forward():
    model = make_model("llama", seed=5).module
    model.attach_lora(dim=4, scaling=8.0)
    with torch.no_grad():  # give B a nonzero value so the delta is real
        for k, t in model.lora_params.items():
            if k.endswith("lora_B"):
                t.normal_(0.0, 0.2)
    packed, cu, mx = packed_batch(model.config.vocab_size, seed=9)
    with torch.no_grad():
        with_adapters = model(packed_input_ids=packed, cu_seqlens=cu,
                              max_seqlen=mx)
    model.squash_lora()
    assert model.lora_flat is None
    with torch.no_grad():
        squashed = model(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    torch.testing.assert_close(squashed, with_adapters, atol=2e-4, rtol=2e-3)


def test_lora_state_dict_roundtrip():
    m1 = make_model("llama", seed=7).module
    m1.attach_lora(dim=4)
    with torch.no_grad():
        for t in m1.lora_params.values():
            t.normal_(0, 0.1)
    sd = m1.lora_state_dict()
    m2 = make_model("llama", seed=7).module
    m2.attach_lora(dim=4)
    m2.load_lora_state_dict(sd)
    packed, cu, mx = packed_batch(m1.config.vocab_size, seed=2)
    with torch.no_grad():
        o1 = m1(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
        o2 = m2(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)
    torch.testing.assert_close(o1, o2)


def _lora_tp2_worker():
    """LoRA on a tp2 model == LoRA on the single model (A replicated via
    per-name seeding; B filled from a full tensor partitioned like the
    base weight: COLUMN for wq/wk/wv, ROW-ish for wo)."""
    import torch.distributed as dist

    from realhf_amd.base import constants
    from realhf_amd.base.testing import init_global_constants
    from realhf_amd.models.real_model import ReaLModel
    import realhf_amd.models.hf as hf_reg
    from realhf_amd.models import param_layout as PL
    from tests.test_model_cpu import packed_batch

    fam = hf_reg.get_family("llama")
    cfg = fam.make_test_config(n_layers=2, hidden_dim=64, n_heads=8,
                               n_kv_heads=4, vocab_size=128)
    cfg.dtype = "float32"
    torch.manual_seed(77)
    full_sd = {k: torch.randn(PL.key_full_shape(cfg, k))
               for k in PL.all_keys(cfg)}

    init_global_constants(num_dp=1, num_tp=2, num_pp=1, model_name="m")
    g = constants.grid_of("m")
    dim = 4

    def build(tp_rank, tp_size):
        m = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                      tp_rank=tp_rank, tp_size=tp_size)
        with torch.no_grad():
            for k in m.layout.keys:
                m.param_view(k).copy_(
                    PL.tp_partition(cfg, k, full_sd[k], tp_rank, tp_size))
        m.attach_lora(dim=dim, scaling=8.0)
        # deterministic nonzero B, partitioned like the base weight rows
        with torch.no_grad():
            for name, t in m.lora_params.items():
                if not name.endswith("lora_B"):
                    continue
                wname = name.rsplit(".lora_", 1)[0]
                import zlib
                gen = torch.Generator().manual_seed(zlib.crc32(name.encode()) % 2**31)
                out_full = PL.key_full_shape(cfg, wname)[0]
                b_full = torch.randn(out_full, dim, generator=gen) * 0.2
                if PL.key_kind(wname) == PL.COLUMN:  # wq/wk/wv: rows shard
                    sh = out_full // tp_size
                    t.copy_(b_full[tp_rank * sh:(tp_rank + 1) * sh])
                else:  # wo (ROW kind): B replicated, A's in-dim shards
                    t.copy_(b_full)
        return m

    # NOTE on wo: its A [r, in_local] shards are independent per-rank
    # draws under the per-name seeding, so tp2 wo-adapters are not a
    # partition of any tp1 adapter — zero them and compare on q/k/v.
    with constants.model_scope("m"):
        m = build(g.tp_rank, 2)
        with torch.no_grad():
            for name, t in m.lora_params.items():
                if ".attn.wo." in name:
                    t.zero_()
        packed, cu, mx = packed_batch(cfg.vocab_size, bs=3, seed=4)
        with torch.no_grad():
            out_tp = m(packed_input_ids=packed, cu_seqlens=cu, max_seqlen=mx)

    constants.clear_grids()
    single = build(0, 1)
    with torch.no_grad():
        for name, t in single.lora_params.items():
            if ".attn.wo." in name:
                t.zero_()
        out_single = single(packed_input_ids=packed, cu_seqlens=cu,
                            max_seqlen=mx)
    vshard = cfg.vocab_size // 2
    ref_shard = out_single[:, g.tp_rank * vshard:(g.tp_rank + 1) * vshard]
    torch.testing.assert_close(out_tp, ref_shard, atol=5e-4, rtol=5e-3)
    dist.barrier()


def test_lora_tp2_matches_single():
    from realhf_amd.base.testing import LocalMultiProcessTest

    LocalMultiProcessTest(2, _lora_tp2_worker).launch()
