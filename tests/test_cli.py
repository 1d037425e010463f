"""Quickstart CLI + local scheduler end-to-end (reference:
apps/quickstart.py + scheduler/local/client.py)."""
import json
import os
import subprocess
import sys

import numpy as np
import pytest


def test_cli_launcher_spawns_workers(tmp_path):
    """Full CLI path: launcher spawns 2 CPU workers which rendezvous over
    gloo and run a 2-step SFT experiment."""
    rng = np.random.RandomState(0)
    data = tmp_path / "sft.jsonl"
    with open(data, "w") as f:
        for _ in range(8):
            rec = {
                "prompt_ids": rng.randint(0, 60, size=3).tolist(),
                "answer_ids": rng.randint(0, 60, size=6).tolist(),
            }
            f.write(json.dumps(rec) + "\n")
    env = dict(os.environ)
    env["REALHF_AMD_FILEROOT"] = str(tmp_path / "root")
    env.pop("WORLD_SIZE", None)
    env.pop("RANK", None)
    out = subprocess.run(
        [sys.executable, "-m", "realhf_amd.apps.quickstart", "sft",
         "experiment_name=cli-sft", "trial_name=t", "n_gpus=2",
         "model.dtype=float32",
         "dataset.type_=prompt_answer", f"dataset.path={data}",
         "dataset.train_bs_n_seqs=4", "exp_ctrl.benchmark_steps=2",
         "exp_ctrl.total_train_epochs=4"],
        env=env, capture_output=True, text=True, timeout=240,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "benchmark" in out.stderr


def test_cli_help():
    out = subprocess.run(
        [sys.executable, "-m", "realhf_amd.apps.quickstart", "--help"],
        capture_output=True, text=True, timeout=60,
    )
    assert out.returncode == 0
    assert "ppo" in out.stdout


def test_examples_run(tmp_path):
    """The example scripts are runnable user documentation — keep them
    working (reference parity: examples/{load_and_eval_rw,new_algorithms,
    customized_exp})."""
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for script in ("examples/load_and_eval_rw.py",
                   "examples/new_algorithms/reinforce.py",
                   "examples/customized_exp/ppo_ref_ema.py",
                   "examples/visualize_dfg.py"):
        r = subprocess.run([sys.executable, os.path.join(root, script)],
                           capture_output=True, text=True, timeout=300)
        assert r.returncode == 0, (script, r.stdout[-800:], r.stderr[-800:])


def test_scheduler_gang_restart(tmp_path):
    """A worker that crashes once is gang-restarted and the job completes
    (elastic recovery; pairs with recover_mode=auto checkpoints)."""
    import sys

    from realhf_amd.scheduler.local import LocalScheduler

    marker = tmp_path / "crashed_once"
    script = tmp_path / "w.py"
    script.write_text(
        "import os, sys\n"
        f"m = {str(marker)!r}\n"
        "if not os.path.exists(m):\n"
        "    open(m, 'w').close()\n"
        "    sys.exit(3)\n"  # first attempt: rank crashes
        "print('ok rank', os.environ.get('RANK'))\n"
    )
    sched = LocalScheduler("exp", "t0", max_restarts=2)
    sched.submit_array([sys.executable, str(script)], n_procs=2)
    assert sched.wait(timeout=60) == 0
    assert marker.exists()


def test_scheduler_restart_exhausted(tmp_path):
    import sys

    import pytest as _pytest

    from realhf_amd.scheduler.local import JobException, LocalScheduler

    script = tmp_path / "w.py"
    script.write_text("import sys; sys.exit(5)\n")
    sched = LocalScheduler("exp", "t1", max_restarts=1)
    sched.submit_array([sys.executable, str(script)], n_procs=2)
    with _pytest.raises(JobException):
        sched.wait(timeout=60)


def test_profile_experiment(tmp_path, monkeypatch):
    """Profile experiment: times interfaces across strategies with mock
    data (reference: ProfileConfig, profile_exp.py:61)."""
    import json
    import os

    from realhf_amd.apps.quickstart import parse_cli
    from realhf_amd.base import constants
    from realhf_amd.runtime.profiler import run_profile

    monkeypatch.setenv("REALHF_AMD_FILEROOT", str(tmp_path / "root"))
    exp, cfg = parse_cli([
        "profile", "experiment_name=t-prof", "trial_name=x",
        "model.family=llama", "model.dtype=float32", "n_gpus=1",
        "strategies=d1t1p1", "interfaces=inference,train_step,generate",
        "n_seqs=2", "seq_len=16", "gen_tokens=4", "n_steps=1", "warmup=0",
    ])
    assert exp == "profile"
    res = run_profile(cfg)
    assert "d1t1p1" in res
    for k in ("inference", "train_step", "generate"):
        assert res["d1t1p1"][k]["seconds"] > 0
    out = os.path.join(constants.LOG_ROOT("t-prof", "x"),
                       "profile_result.json")
    assert json.load(open(out))["d1t1p1"]


def test_sentiment_example_interface(tmp_path, monkeypatch):
    """Custom external-scorer reward interface (reference:
    examples/customized_exp/ppo_sentiment.py): a stub classifier stands
    in for the HF model; the interface must emit per-sequence rewards."""
    import sys
    import types

    import torch

    # stub transformers.AutoModelForSequenceClassification
    class _StubModel(torch.nn.Module):
        def forward(self, input_ids=None, attention_mask=None):
            bs = input_ids.shape[0]
            out = types.SimpleNamespace()
            out.logits = torch.stack(
                [torch.zeros(bs), input_ids.float().mean(dim=1)], dim=1)
            return out

        def parameters(self):
            yield torch.nn.Parameter(torch.zeros(1))

    import transformers

    monkeypatch.setattr(
        transformers.AutoModelForSequenceClassification, "from_pretrained",
        staticmethod(lambda path: _StubModel()))
    monkeypatch.setenv("SCORER", "stub")
    sys.path.insert(0, "examples/customized_exp")
    try:
        import ppo_sentiment

        iface = ppo_sentiment.SentimentScoringInterface()
    finally:
        sys.path.pop(0)

    from realhf_amd.api.data import SequenceSample

    toks = torch.arange(14)
    data = SequenceSample(
        keys=("packed_input_ids",), ids=["a", "b"],
        seqlens={"packed_input_ids": [[6], [8]]},
        data={"packed_input_ids": toks},
    )
    out = iface.inference(None, data)
    assert out.keys == ("rewards",) and out.data["rewards"].shape == (2,)
    # stub reward = mean token id of the PADDED row
    assert out.data["rewards"][1] > out.data["rewards"][0]
