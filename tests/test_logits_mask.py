"""Logits-mask mode (gconfig.force_no_logits_mask=False): the sampler's
top-k/p removal mask travels actor_gen -> ref_inf/actor_train so every
logprob is computed under the distribution that was actually sampled
from (reference: real_llm_generate.py:131-136 genstep logits_mask,
ppo_interface.py:38-54/245-246)."""
import json
import os

import numpy as np
import pytest
import torch

from realhf_amd.api.model import GenerationHyperparameters
from realhf_amd.models.generation import _sample_from_logits


def test_sampler_mask_and_filtered_logp():
    g = GenerationHyperparameters(top_k=3, top_p=1.0, temperature=1.0,
                                  force_no_logits_mask=False)
    torch.manual_seed(0)
    logits = torch.randn(5, 16)
    gen = torch.Generator().manual_seed(1)
    tokens, logp, mask = _sample_from_logits(logits, g, gen, return_mask=True)
    assert mask.shape == (5, 16) and mask.dtype == torch.bool
    assert (mask.sum(-1) == 13).all()  # exactly V - top_k removed
    # the sampled token is never a removed one
    assert not mask.gather(1, tokens.unsqueeze(1)).any()
    # logp is under the FILTERED (renormalized top-k) distribution
    masked = logits.masked_fill(mask, float("-inf"))
    want = torch.log_softmax(masked, -1).gather(1, tokens.unsqueeze(1)).squeeze(1)
    torch.testing.assert_close(logp, want)


def test_sampler_mask_with_top_p():
    g = GenerationHyperparameters(top_k=0, top_p=0.5,
                                  force_no_logits_mask=False)
    torch.manual_seed(2)
    logits = torch.randn(4, 32) * 3  # spiky -> top_p binds
    gen = torch.Generator().manual_seed(3)
    tokens, logp, mask = _sample_from_logits(logits, g, gen, return_mask=True)
    assert mask.any()  # something was removed
    assert not mask.gather(1, tokens.unsqueeze(1)).any()
    masked = logits.masked_fill(mask, float("-inf"))
    want = torch.log_softmax(masked, -1).gather(1, tokens.unsqueeze(1)).squeeze(1)
    torch.testing.assert_close(logp, want)


def test_generation_emits_logits_mask():
    from realhf_amd.models.generation import generate
    from realhf_amd.models.hf import llama

    cfg = llama.make_test_config(vocab_size=64)
    cfg.dtype = "float32"
    from realhf_amd.models.real_model import ReaLModel

    m = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    m.random_init()
    g = GenerationHyperparameters(max_new_tokens=5, top_k=4,
                                  use_hip_graph=False,
                                  force_no_logits_mask=False)
    rng = np.random.RandomState(0)
    packed = torch.from_numpy(rng.randint(0, 64, size=14)).long()
    cu = torch.tensor([0, 6, 14], dtype=torch.int32)
    out = generate(m, packed, cu, g, eos_token_id=None, pad_token_id=0)
    lm = out.logits_mask
    assert lm is not None
    gmax = int(out.gen_lengths.max())
    assert lm.shape == (2, gmax, 64)
    assert (lm.sum(-1) == 60).all()  # V - top_k removed every step


def test_ppo_experiment_with_logits_mask(tmp_path):
    """Full 1-rank PPO step with the mask flowing actor_gen -> ref_inf ->
    actor_train through the DFG executor."""
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "prompts.jsonl")
    rng = np.random.RandomState(5)
    with open(data, "w") as f:
        for i in range(16):
            ids = rng.randint(0, 64, size=rng.randint(4, 8)).tolist()
            f.write(json.dumps({"prompt": "x", "input_ids": ids}) + "\n")
    cfg = PPOConfig(experiment_name="t-ppo-lmask", trial_name="cpu", n_gpus=1)
    for mc in (cfg.actor, cfg.critic, cfg.ref, cfg.rew):
        mc.dtype = "float32"
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 6
    cfg.ppo.gen.min_new_tokens = 2
    cfg.ppo.gen.top_k = 4  # binds: the mask is non-trivial
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.gen.force_no_logits_mask = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 1
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()


def _pp_mask_worker():
    """Token-interleaved PP generation records per-step sampler masks
    (last stage), so the mask mode works under pipeline parallelism."""
    import torch.distributed as dist

    from realhf_amd.api.data import SequenceSample
    from realhf_amd.base import constants
    from tests.test_pp import _make_pp_engine
    from tests.test_realloc import _full_reference_sd
    from realhf_amd.models.hf.llama import make_test_config

    cfg = make_test_config(n_layers=4, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg, seed=53)
    engine, m = _make_pp_engine(cfg, sd, 2)
    rng = np.random.RandomState(10)
    lens = [5, 8, 6]
    toks = torch.from_numpy(rng.randint(0, 128, size=sum(lens))).long()
    batch = SequenceSample(
        keys=("packed_prompts",),
        ids=["g0", "g1", "g2"],
        seqlens={"packed_prompts": [[l] for l in lens]},
        data={"packed_prompts": toks},
    )
    g = GenerationHyperparameters(max_new_tokens=6, top_k=4,
                                  use_hip_graph=False,
                                  force_no_logits_mask=False)
    with constants.model_scope("m"):
        outs = engine.generate(batch, gconfig=g, n_mbs=2)
    if m.is_last_stage:
        n = 0
        for gen_out, prompts, cu in outs:
            lm = gen_out.logits_mask
            assert lm is not None
            bs = cu.shape[0] - 1
            gmax = gen_out.gen_tokens.shape[1]
            assert lm.shape == (bs, gmax, 128)
            assert (lm.sum(-1) == 124).all()  # V - top_k removed per step
            # sampled tokens are never masked (rows where not yet done)
            for i in range(bs):
                gl = int(gen_out.gen_lengths[i])
                got = lm[i, torch.arange(gl), gen_out.gen_tokens[i, :gl]]
                assert not got.any()
            n += bs
        assert n == len(lens)
    dist.barrier()


@pytest.mark.distributed
def test_pp2_generation_logits_mask():
    from realhf_amd.base.testing import LocalMultiProcessTest

    LocalMultiProcessTest(2, _pp_mask_worker).launch()


def test_grpo_experiment_with_logits_mask(tmp_path):
    """GRPO (no critic) with the mask flowing through its DFG."""
    from realhf_amd.api.experiment import GRPOConfig
    from realhf_amd.runtime.trainer import Trainer

    data = str(tmp_path / "prompts.jsonl")
    rng = np.random.RandomState(9)
    with open(data, "w") as f:
        for i in range(8):
            ids = rng.randint(0, 64, size=rng.randint(4, 8)).tolist()
            f.write(json.dumps({"prompt": "x", "input_ids": ids}) + "\n")
    cfg = GRPOConfig(experiment_name="t-grpo-lmask", trial_name="cpu", n_gpus=1)
    for mc in (cfg.actor, cfg.ref, cfg.rew):
        mc.dtype = "float32"
    cfg.rew.is_critic = True
    cfg.group_size = 2
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 2
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 4
    cfg.ppo.gen.top_k = 4
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.gen.force_no_logits_mask = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 1
    os.environ["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    Trainer(cfg).run()


def _ppo_mask_dist_worker(data, fileroot):
    """2-rank PPO with the logits mask crossing mesh boundaries: a tp2
    gen replica produces it, dp2 ref_inf/actor_train consume it — the
    2-D bool key rides the executor's device-payload transfer plan."""
    from realhf_amd.api.config import ParallelismConfig
    from realhf_amd.api.experiment import PPOConfig
    from realhf_amd.runtime.trainer import Trainer

    os.environ["REALHF_AMD_FILEROOT"] = fileroot
    cfg = PPOConfig(experiment_name="t-ppo2-lmask", trial_name="dist",
                    n_gpus=2)
    for mc in (cfg.actor, cfg.critic, cfg.ref, cfg.rew):
        mc.dtype = "float32"
        mc.parallel = ParallelismConfig(data_parallel_size=2)
    cfg.allocation_mode = "manual"
    cfg.actor.gen_parallel = ParallelismConfig(
        data_parallel_size=1, tensor_parallel_size=2
    )
    cfg.dataset.type_ = "prompt"
    cfg.dataset.path = data
    cfg.dataset.train_bs_n_seqs = 4
    cfg.dataset.max_prompt_len = 8
    cfg.ppo.gen.max_new_tokens = 5
    cfg.ppo.gen.top_k = 4
    cfg.ppo.gen.use_hip_graph = False
    cfg.ppo.gen.force_no_logits_mask = False
    cfg.ppo.ppo_n_minibatches = 2
    cfg.exp_ctrl.benchmark_steps = 1
    Trainer(cfg).run()


@pytest.mark.distributed
def test_ppo_logits_mask_two_ranks(tmp_path):
    from realhf_amd.base.testing import LocalMultiProcessTest

    data = str(tmp_path / "prompts.jsonl")
    rng = np.random.RandomState(13)
    with open(data, "w") as f:
        for i in range(16):
            ids = rng.randint(0, 64, size=rng.randint(4, 8)).tolist()
            f.write(json.dumps({"prompt": "x", "input_ids": ids}) + "\n")
    LocalMultiProcessTest(2, _ppo_mask_dist_worker, data,
                          str(tmp_path / "root")).launch()


def test_mask_mode_reforward_matches_sampler_logp():
    """THE consistency property the mask exists for: re-forwarding the
    generated sequence with the mask + temperature applied must
    reproduce the sampler's behavior logprobs on generated tokens
    (importance ratio == 1 at step 0).  Without the mask, top-k
    renormalization makes them differ."""
    from realhf_amd.api.config import ModelName
    from realhf_amd.api.model import Model
    from realhf_amd.interfaces.ppo import PPOActorInterface
    from realhf_amd.models.hf import llama
    from realhf_amd.models.real_model import ReaLModel
    from realhf_amd.runtime.engine import PipelinableInferenceEngine

    cfg = llama.make_test_config(vocab_size=64)
    cfg.dtype = "float32"
    m = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    torch.manual_seed(3)
    m.random_init()
    model = Model(name=ModelName("actor", 0), module=PipelinableInferenceEngine(m),
                  tokenizer=None, device=torch.device("cpu"),
                  dtype=torch.float32)
    iface = PPOActorInterface(
        gconfig={"max_new_tokens": 6, "min_new_tokens": 6, "top_k": 4,
                 "use_hip_graph": False, "force_no_logits_mask": False},
    )
    rng = np.random.RandomState(1)
    lens = [5, 7]
    prompts = torch.from_numpy(rng.randint(0, 64, size=sum(lens))).long()
    sample = __import__("realhf_amd.api.data", fromlist=["SequenceSample"]).SequenceSample(
        keys=("packed_prompts",),
        ids=["a", "b"],
        seqlens={"packed_prompts": [[lens[0]], [lens[1]]]},
        data={"packed_prompts": prompts},
    )
    out = iface.generate(model, sample)
    assert "packed_logits_mask" in out.keys
    ref = iface.inference(model, out)  # same weights = "ref" model
    old = out.data["packed_logprobs"]
    new = ref.data["packed_ref_logprobs"]
    pm = out.data["prompt_mask"].bool()
    # shifted positions: row j of logprobs corresponds to token j+1
    from realhf_amd.utils.functional import build_shift_one_indices

    total = int(sum(sum(x) for x in out.seqlens["packed_input_ids"]))
    cu = torch.tensor(
        [0] + list(np.cumsum([sum(x) for x in out.seqlens["packed_input_ids"]])),
        dtype=torch.int32)
    gen_rows = ~pm[build_shift_one_indices(total, cu)]
    torch.testing.assert_close(new[gen_rows], old[gen_rows],
                               atol=1e-4, rtol=1e-4)
    # sanity: the filtered behavior logp differs from the unfiltered one
    # (top-k renormalization concentrates mass on the kept entries), so
    # the match above is non-trivial
    masked_rows = out.data["packed_logits_mask"][gen_rows]
    assert masked_rows.any()
    assert (old[gen_rows] > np.log(1.0 / 64) + 0.05).all()  # > uniform-ish
    # unfiltered re-forward of the SAME sequences (no mask applied)
    plain = PPOActorInterface(
        gconfig={"max_new_tokens": 6, "use_hip_graph": False},
    ).inference(model, out.select_keys(["packed_input_ids"]))
    assert not torch.allclose(
        plain.data["packed_ref_logprobs"][gen_rows], old[gen_rows],
        atol=1e-4)


def _tp2_masked_logprob_worker():
    """Masked + temperature-scaled logprobs under TP vocab sharding:
    apply_logits_mask slices the full-vocab mask per tp rank and the
    vocab-parallel logprob matches the single-process masked reference."""
    import torch.distributed as dist

    from realhf_amd.base import constants
    from realhf_amd.base.testing import init_global_constants
    from realhf_amd.interfaces.ppo import _warp_logits_like_sampler
    from realhf_amd.models.hf.llama import make_test_config
    from realhf_amd.models.real_model import ReaLModel
    from realhf_amd.parallel.tp import packed_shifted_logprobs
    from realhf_amd.api.data import SequenceSample
    from tests.test_realloc import _fill_model_from_full, _full_reference_sd

    cfg = make_test_config(n_layers=2, hidden_dim=64, n_heads=8, n_kv_heads=4,
                           vocab_size=128)
    cfg.dtype = "float32"
    sd = _full_reference_sd(cfg, seed=57)
    init_global_constants(num_dp=1, num_tp=2, num_pp=1, model_name="m")
    g = constants.grid_of("m")

    rng = np.random.RandomState(6)
    lens = [10, 8]
    toks = torch.from_numpy(rng.randint(0, 128, size=18)).long()
    cu = torch.tensor([0, 10, 18], dtype=torch.int32)
    # random full-vocab mask on the predicting rows, never masking the
    # next token itself (the sampled one is always kept)
    lm = torch.rand(16, 128) > 0.4
    from realhf_amd.utils.functional import build_shift_one_indices

    nxt = toks[build_shift_one_indices(18, cu)]
    lm[torch.arange(16), nxt] = False
    mb = SequenceSample(
        keys=("packed_input_ids", "packed_logits_mask"),
        ids=["a", "b"],
        seqlens={"packed_input_ids": [[10], [8]],
                 "packed_logits_mask": [[9], [7]]},
        data={"packed_input_ids": toks, "packed_logits_mask": lm},
    )

    with constants.model_scope("m"):
        m = ReaLModel(cfg, device="cpu", dtype=torch.float32,
                      tp_rank=g.tp_rank, tp_size=2)
        _fill_model_from_full(m, cfg, sd)
        with torch.no_grad():
            logits = m(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=10)
            _warp_logits_like_sampler(logits, cu, mb, temperature=0.7)
            lp = packed_shifted_logprobs(logits, cu, toks)

    single = ReaLModel(cfg, device="cpu", dtype=torch.float32)
    _fill_model_from_full(single, cfg, sd)
    constants.clear_grids()
    with torch.no_grad():
        full = single(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=10)
        full = full / 0.7
        rows = torch.arange(18)[~torch.isin(
            torch.arange(18), cu[1:].long() - 1)]  # leave-one rows
        full[rows] = full[rows].masked_fill(lm, float("-inf"))
        want = torch.log_softmax(full.float(), -1)
        want = want[rows].gather(1, nxt.unsqueeze(1)).squeeze(1)
    torch.testing.assert_close(lp, want, atol=1e-4, rtol=1e-4)
    dist.barrier()


@pytest.mark.distributed
def test_tp2_masked_logprobs_match_single():
    from realhf_amd.base.testing import LocalMultiProcessTest

    LocalMultiProcessTest(2, _tp2_masked_logprob_worker).launch()
