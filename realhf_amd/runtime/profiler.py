"""Profile experiment: time an interface across parallel strategies with
mock data (reference: ProfileConfig, experiments/benchmark/profile_exp.py:61
+ examples/profiling/ sweeps — workers are paused and reconfigured between
setups; here the SPMD ranks simply rebuild the model per strategy).

Run:  python -m realhf_amd.apps.quickstart profile \\
          model.family=llama n_gpus=1 \\
          strategies="d1;d1t1p1" interfaces=inference,train_step \\
          n_seqs=16 seq_len=512

Writes per-(strategy, interface) seconds + model TFLOP/s to
LOG_ROOT/profile_result.json and logs a table.
"""
import json
import os
import time
from typing import Dict

import torch
import torch.distributed as dist

import realhf_amd.runtime.engine  # noqa: F401 — register backends

from realhf_amd.api.config import (
    Abstraction,
    ModelName,
    parse_parallelism,
)
from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import (
    FinetuneSpec,
    GenerationHyperparameters,
    Model,
    make_backend,
)
from realhf_amd.base import constants, logging, seeding
from realhf_amd.base.monitor import estimate_mfc_flops
from realhf_amd.base.topology import ParallelGrid, PipeDataTensorTopology
from realhf_amd.models import hf as hf_reg
from realhf_amd.models.real_model import ReaLModel

logger = logging.getLogger("profiler")


def _mock_batch(n_seqs, seq_len, vocab, device, key="packed_input_ids"):
    toks = torch.randint(3, vocab - 3, (n_seqs * seq_len,), device=device)
    return SequenceSample(
        keys=(key,), ids=[f"p{i}" for i in range(n_seqs)],
        seqlens={key: [[seq_len]] * n_seqs},
        data={key: toks},
    )


def run_profile(cfg) -> Dict:
    rank = dist.get_rank() if dist.is_initialized() else 0
    world = dist.get_world_size() if dist.is_initialized() else 1
    seeding.set_random_seed(cfg.seed)
    constants.set_experiment_trial_names(cfg.experiment_name, cfg.trial_name)
    device = (torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
              if torch.cuda.is_available() else torch.device("cpu"))
    mc = cfg.model
    if mc.path:
        rcfg = hf_reg.config_from_hf_path(mc.family, mc.path)
    else:
        rcfg = hf_reg.get_family(mc.family).make_test_config()
        rcfg.family = mc.family
    rcfg.dtype = mc.dtype
    dtype = rcfg.torch_dtype if device.type == "cuda" else torch.float32

    ifaces = [s for s in cfg.interfaces.split(",") if s]
    results = {}
    for si, sspec in enumerate(s for s in cfg.strategies.split(";") if s):
        par = parse_parallelism(sspec)
        if par.world_size > world:
            logger.warning("skip %s: needs %d ranks (world %d)", sspec,
                           par.world_size, world)
            continue
        scope = f"prof{si}"
        topo = PipeDataTensorTopology(
            num_pp=par.pipeline_parallel_size, num_dp=par.data_parallel_size,
            num_tp=par.tensor_parallel_size,
            sequence_parallel=par.sequence_parallel,
            gradient_checkpointing=mc.gradient_checkpointing,
        )
        if dist.is_initialized():
            grid = ParallelGrid(topo)
        else:
            from realhf_amd.base.topology import FakeGrid

            grid = FakeGrid(0, topo)
        constants.set_grid(scope, grid)
        if rank >= par.world_size:
            continue
        with constants.model_scope(scope):
            g = constants.grid()
            m = ReaLModel(rcfg, device=device, dtype=dtype,
                          tp_rank=g.tp_rank, tp_size=g.tp_size,
                          pp_rank=g.pp_rank, pp_size=g.pp_size)
            if mc.path:
                hf_reg.load_from_hf(m, mc.family, mc.path)
            else:
                m.random_init()
            model = Model(ModelName("prof", si), m, None, device, dtype)
            backend = make_backend(
                Abstraction("zero1",
                            {"optimizer": {"warmup_steps_proportion": 0.0}})
                if "train_step" in ifaces else Abstraction("inference"))
            model = backend.initialize(model, FinetuneSpec(1, 1024, cfg.n_seqs))

            n_local = max(1, cfg.n_seqs // g.dp_size)
            batch = _mock_batch(n_local, cfg.seq_len, rcfg.vocab_size, device)

            def timed(fn, itype, seqlens, out_lens=None):
                for _ in range(cfg.warmup):
                    fn()
                if device.type == "cuda":
                    torch.cuda.synchronize()
                t0 = time.time()
                for _ in range(cfg.n_steps):
                    fn()
                if device.type == "cuda":
                    torch.cuda.synchronize()
                dt = (time.time() - t0) / cfg.n_steps
                fl = estimate_mfc_flops(itype, rcfg, seqlens, out_lens)
                return {"seconds": dt,
                        "tflops_per_gpu": fl / dt / 1e12
                        / (g.tp_size * g.pp_size)}

            res = {}
            lens = [cfg.seq_len] * n_local
            if "inference" in ifaces:
                res["inference"] = timed(
                    lambda: model.module.forward(batch), "INFERENCE", lens)
            if "train_step" in ifaces:
                def loss_fn(out, mb):
                    return out.float().square().mean(), {}

                res["train_step"] = timed(
                    lambda: model.module.train_batch(batch, loss_fn),
                    "TRAIN_STEP", lens)
            if "generate" in ifaces:
                gk = GenerationHyperparameters(
                    max_new_tokens=cfg.gen_tokens,
                    min_new_tokens=cfg.gen_tokens, greedy=True)
                res["generate"] = timed(
                    lambda: model.module.generate(batch, gconfig=gk),
                    "GENERATE", lens,
                    [l + cfg.gen_tokens for l in lens])
            results[sspec] = res
            if rank == 0:
                for k, v in res.items():
                    logger.info("%s %s: %.4fs (%.0f TFLOP/s per GPU)",
                                sspec, k, v["seconds"], v["tflops_per_gpu"])
            del model, m
            if device.type == "cuda":
                torch.cuda.empty_cache()
    if rank == 0:
        root = constants.LOG_ROOT(cfg.experiment_name, cfg.trial_name)
        os.makedirs(root, exist_ok=True)
        out = os.path.join(root, "profile_result.json")
        with open(out, "w") as f:
            json.dump(results, f, indent=1)
        logger.info("profile results -> %s", out)
    return results
