"""SLURM scheduler client against stubbed sbatch/squeue/scancel binaries
(reference test surface: scheduler/slurm/client.py submit/wait/stop)."""
import os
import stat
import subprocess
import sys

import pytest

from realhf_amd.scheduler.slurm import JobException, SlurmScheduler


def _stub(tmp_path, name, body):
    p = tmp_path / name
    p.write_text("#!/bin/bash\n" + body)
    p.chmod(p.stat().st_mode | stat.S_IEXEC)
    return p


@pytest.fixture()
def fake_slurm(tmp_path, monkeypatch):
    state = tmp_path / "state"
    state.write_text("RUNNING")
    _stub(tmp_path, "sbatch", 'echo "Submitted batch job 4242"\n')
    _stub(
        tmp_path, "squeue",
        f'if [ "$1" = "-j" ]; then echo "$(cat {state})|node[01-02]"; '
        f'else echo "exp_t0|$(cat {state})|4242|node01"; fi\n',
    )
    _stub(tmp_path, "scancel", f'echo CANCELLED > {state}\n')
    _stub(tmp_path, "scontrol", 'echo node01\n')
    monkeypatch.setenv("PATH", f"{tmp_path}:{os.environ['PATH']}")
    return state


def test_script_rendering(tmp_path):
    s = SlurmScheduler("exp", "t0", partition="amd", time_limit="1:00:00",
                       container_image="rocm.sqsh",
                       container_mounts="/data:/data",
                       log_dir=str(tmp_path), gpus_per_node=8)
    script = s.render_script([sys.executable, "-m", "realhf_amd.apps.quickstart",
                              "ppo", "n_gpus=16"], n_procs=16)
    assert "#SBATCH --nodes=2" in script
    assert "#SBATCH --ntasks=16" in script
    assert "#SBATCH --ntasks-per-node=8" in script
    assert "#SBATCH --partition=amd" in script
    assert "#SBATCH --container-image=rocm.sqsh" in script
    assert "export RANK=$SLURM_PROCID" in script
    assert "export ROCR_VISIBLE_DEVICES=$SLURM_LOCALID" in script
    assert "export WORLD_SIZE=16" in script
    assert "MASTER_ADDR=$(scontrol show hostnames" in script


def test_submit_wait_completed(tmp_path, fake_slurm):
    s = SlurmScheduler("exp", "t0", log_dir=str(tmp_path))
    s.submit_array(["python", "train.py"], n_procs=8)
    assert s.job_id == "4242"
    assert s.find().state == "RUNNING"
    fake_slurm.write_text("COMPLETED")
    assert s.wait(timeout=5, poll_interval=0.01) == 0
    sbatch_file = tmp_path / "exp_t0.sbatch"
    assert sbatch_file.exists()


def test_wait_raises_on_failure(tmp_path, fake_slurm):
    s = SlurmScheduler("exp", "t0", log_dir=str(tmp_path))
    s.submit_array(["python", "train.py"], n_procs=8)
    fake_slurm.write_text("FAILED")
    with pytest.raises(JobException):
        s.wait(timeout=5, poll_interval=0.01)


def test_stop_all_and_find_all(tmp_path, fake_slurm):
    s = SlurmScheduler("exp", "t0", log_dir=str(tmp_path))
    s.submit_array(["python", "train.py"], n_procs=8)
    jobs = s.find_all("exp_.*")
    assert jobs and jobs[0].job_id == "4242"
    s.stop_all()
    assert fake_slurm.read_text().strip() == "CANCELLED"
