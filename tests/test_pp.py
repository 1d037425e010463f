"""Pipeline-parallel correctness over gloo on CPU (reference tests:
pipeline schedules in tests/ + test_generate.py distributed cases)."""
import numpy as np
import pytest
import torch

from realhf_amd.base.testing import LocalMultiProcessTest
from realhf_amd.models.hf.llama import make_test_config
from tests.test_realloc import _fill_model_from_full, _full_reference_sd


def _make_pp_engine(cfg, sd, pp, trainable=False, lr=1e-3):
    import torch.distributed as dist

    from realhf_amd.api.config import Abstraction, ModelName
    from realhf_amd.api.model import FinetuneSpec, Model, make_backend
    import realhf_amd.runtime.engine  # noqa: F401
    from realhf_amd.base import constants
    from realhf_amd.base.testing import init_global_constants
    from realhf_amd.models.real_model import ReaLModel

    init_global_constants(num_dp=1, num_tp=1, num_pp=pp, model_name="m")
    g = constants.grid_of("m")
    m = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                  pp_rank=g.pp_rank, pp_size=pp)
    _fill_model_from_full(m, cfg, sd)
    model = Model(ModelName("m", 0), m, None, torch.device("cpu"), torch.float32)
    backend = make_backend(
        Abstraction("zero1", {"optimizer": {
            "lr": lr, "warmup_steps_proportion": 0.0,
            "lr_scheduler_type": "constant", "gradient_clipping": 0.0}})
        if trainable else Abstraction("inference")
    )
    model = backend.initialize(model, FinetuneSpec(1, 64, 4))
    return model.module, m


def _mk_batch(cfg, bs=4, seed=5):
    from realhf_amd.api.data import SequenceSample

    rng = np.random.RandomState(seed)
    lens = rng.randint(6, 14, size=bs).tolist()
    toks = torch.from_numpy(
        rng.randint(0, cfg.vocab_size, size=sum(lens))
    ).long()
    pm = torch.zeros(sum(lens), dtype=torch.bool)
    return SequenceSample(
        keys=("packed_input_ids", "prompt_mask"),
        ids=[f"b{seed}-{i}" for i in range(bs)],
        seqlens={"packed_input_ids": [[l] for l in lens],
                 "prompt_mask": [[l] for l in lens]},
        data={"packed_input_ids": toks, "prompt_mask": pm},
    )


def _pp_forward_worker(pp):
    import torch.distributed as dist

    from realhf_amd.base import constants
    from realhf_amd.models.real_model import ReaLModel

    cfg = make_test_config(n_layers=4, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg, seed=31)
    engine, m = _make_pp_engine(cfg, sd, pp)
    batch = _mk_batch(cfg)
    with constants.model_scope("m"):
        out = engine.forward(batch, n_mbs=2)
    if m.is_last_stage:
        single = ReaLModel(cfg, device="cpu", dtype=torch.float32)
        _fill_model_from_full(single, cfg, sd)
        from realhf_amd.runtime.engine import sample_to_packed

        ids, cu, mx = sample_to_packed(batch)
        with torch.no_grad():
            ref = single(packed_input_ids=ids, cu_seqlens=cu, max_seqlen=mx)
        torch.testing.assert_close(out, ref, atol=1e-4, rtol=1e-4)
    dist.barrier()


@pytest.mark.distributed
def test_pp2_forward():
    LocalMultiProcessTest(2, _pp_forward_worker, 2).launch()


@pytest.mark.distributed
def test_pp4_forward():
    LocalMultiProcessTest(4, _pp_forward_worker, 4).launch()


def _pp_train_worker():
    import torch.distributed as dist

    from realhf_amd.base import constants
    from realhf_amd.interfaces.sft import sft_loss_fn
    from realhf_amd.models.real_model import ReaLModel

    cfg = make_test_config(n_layers=4, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg, seed=41)
    engine, m = _make_pp_engine(cfg, sd, 2, trainable=True, lr=1e-2)
    batch = _mk_batch(cfg, seed=7)
    with constants.model_scope("m"):
        stats = engine.train_batch(batch, sft_loss_fn, n_mbs=2)

    # single-process equivalent
    from realhf_amd.api.config import Abstraction, ModelName
    from realhf_amd.api.model import FinetuneSpec, Model, make_backend

    constants.clear_grids()
    single = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    _fill_model_from_full(single, cfg, sd)
    smodel = Model(ModelName("s", 0), single, None, torch.device("cpu"),
                   torch.float32)
    backend = make_backend(
        Abstraction("zero1", {"optimizer": {
            "lr": 1e-2, "warmup_steps_proportion": 0.0,
            "lr_scheduler_type": "constant", "gradient_clipping": 0.0}})
    )
    seng = backend.initialize(smodel, FinetuneSpec(1, 64, 4)).module
    seng.train_batch(batch, sft_loss_fn, n_mbs=2)
    # compare this stage's params against the single model
    for k in m.layout.keys:
        torch.testing.assert_close(
            m.param_view(k), single._params[k], atol=1e-5, rtol=1e-4,
        )
    dist.barrier()


@pytest.mark.distributed
def test_pp2_1f1b_train_matches_single():
    LocalMultiProcessTest(2, _pp_train_worker).launch()


def _pp_gen_worker():
    import torch.distributed as dist

    from realhf_amd.api.data import SequenceSample
    from realhf_amd.api.model import GenerationHyperparameters
    from realhf_amd.base import constants
    from realhf_amd.models.generation import generate as gen_single
    from realhf_amd.models.real_model import ReaLModel

    cfg = make_test_config(n_layers=4, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg, seed=51)
    engine, m = _make_pp_engine(cfg, sd, 2)
    rng = np.random.RandomState(9)
    lens = [5, 8, 6]
    toks = torch.from_numpy(rng.randint(0, 128, size=sum(lens))).long()
    batch = SequenceSample(
        keys=("packed_prompts",),
        ids=["g0", "g1", "g2"],
        seqlens={"packed_prompts": [[l] for l in lens]},
        data={"packed_prompts": toks},
    )
    gconfig = GenerationHyperparameters(max_new_tokens=6, greedy=True,
                                        use_hip_graph=False)
    with constants.model_scope("m"):
        outs = engine.generate(batch, gconfig=gconfig)
    if m.is_last_stage:
        (gen_out, prompts, cu) = outs[0]
        single = ReaLModel(cfg, device="cpu", dtype=torch.float32)
        _fill_model_from_full(single, cfg, sd)
        cu_t = torch.tensor([0] + list(np.cumsum(lens)), dtype=torch.int32)
        ref = gen_single(single, toks, cu_t, gconfig)
        assert torch.equal(gen_out.gen_tokens, ref.gen_tokens), (
            gen_out.gen_tokens, ref.gen_tokens)
    dist.barrier()


@pytest.mark.distributed
def test_pp2_generate_matches_single():
    LocalMultiProcessTest(2, _pp_gen_worker).launch()


def _pp_tp_forward_worker():
    """Combined tp2 x pp2 on 4 ranks: forward matches single-process."""
    import torch.distributed as dist

    from realhf_amd.api.config import Abstraction, ModelName
    from realhf_amd.api.model import FinetuneSpec, Model, make_backend
    import realhf_amd.runtime.engine  # noqa: F401
    from realhf_amd.base import constants
    from realhf_amd.base.testing import init_global_constants
    from realhf_amd.models.real_model import ReaLModel
    from realhf_amd.runtime.engine import sample_to_packed

    cfg = make_test_config(n_layers=4, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg, seed=41)
    init_global_constants(num_dp=1, num_tp=2, num_pp=2, model_name="m")
    g = constants.grid_of("m")
    m = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                  tp_rank=g.tp_rank, tp_size=2,
                  pp_rank=g.pp_rank, pp_size=2)
    _fill_model_from_full(m, cfg, sd)
    model = Model(ModelName("m", 0), m, None, torch.device("cpu"),
                  torch.float32)
    model = make_backend(Abstraction("inference")).initialize(
        model, FinetuneSpec(1, 64, 4))
    batch = _mk_batch(cfg)
    with constants.model_scope("m"):
        out = model.module.forward(batch, n_mbs=2)
    if m.is_last_stage:
        single = ReaLModel(cfg, device="cpu", dtype=torch.float32)
        _fill_model_from_full(single, cfg, sd)
        ids, cu, mx = sample_to_packed(batch)
        with torch.no_grad():
            ref = single(packed_input_ids=ids, cu_seqlens=cu, max_seqlen=mx)
        # last stage returns this tp rank's vocab-parallel logit shard
        vshard = cfg.vocab_size // 2
        ref_shard = ref[:, g.tp_rank * vshard:(g.tp_rank + 1) * vshard]
        torch.testing.assert_close(out, ref_shard, atol=2e-4, rtol=2e-4)
    dist.barrier()


@pytest.mark.distributed
def test_pp2_tp2_forward():
    LocalMultiProcessTest(4, _pp_tp_forward_worker).launch()
