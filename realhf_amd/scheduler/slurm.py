"""SLURM scheduler client for multi-node launches.

Reference semantics: realhf/scheduler/slurm/client.py:25 (SlurmSchedulerClient
submit_array/stop/find/wait over sbatch+squeue+scancel) and
scheduler/slurm/utils.py:822 (SlurmLaunchInfo: sbatch script + srun
multiprog file generation, container mounts, squeue polling).

MI355X-native differences: one task per GPU with `torch.distributed`
rendezvous env (RANK/WORLD_SIZE/MASTER_ADDR) instead of the reference's
worker-type/ZMQ topology — every rank runs the same SPMD program, so the
sbatch script is a plain array of identical tasks.  GPU binding uses
ROCR_VISIBLE_DEVICES (the ROCm equivalent of the reference's
CUDA_VISIBLE_DEVICES isolation, gpu_utils.py:64).

All SLURM interaction goes through the `sbatch`/`squeue`/`scancel`
binaries on PATH, so tests can stub them (tests/test_slurm.py).
"""
import dataclasses
import os
import re
import subprocess
import time
from typing import Dict, List, Optional

from realhf_amd.base import logging

logger = logging.getLogger("slurm")


class JobException(Exception):
    pass


@dataclasses.dataclass
class JobInfo:
    name: str
    state: str  # PENDING | RUNNING | COMPLETED | FAILED | CANCELLED | ...
    job_id: Optional[str] = None
    host: Optional[str] = None


_ACTIVE = {"PENDING", "RUNNING", "CONFIGURING", "COMPLETING", "SUSPENDED"}
_FAILED = {"FAILED", "CANCELLED", "TIMEOUT", "NODE_FAIL", "OUT_OF_MEMORY",
           "PREEMPTED", "BOOT_FAIL", "DEADLINE"}


def _run(cmd: List[str]) -> str:
    out = subprocess.run(cmd, capture_output=True, text=True)
    if out.returncode != 0:
        raise JobException(f"{cmd[0]} failed: {out.stderr.strip()}")
    return out.stdout


class SlurmScheduler:
    """Submit the SPMD trainer as one sbatch job of nnodes x gpus_per_node
    tasks; poll squeue; scancel on stop."""

    def __init__(
        self,
        experiment: str,
        trial: str,
        partition: Optional[str] = None,
        container_image: Optional[str] = None,
        container_mounts: Optional[str] = None,
        account: Optional[str] = None,
        time_limit: Optional[str] = None,
        log_dir: str = "/tmp/realhf_amd/slurm",
        gpus_per_node: int = 8,
        mem_per_node: Optional[str] = None,
        env_vars: Optional[Dict[str, str]] = None,
    ):
        self.experiment = experiment
        self.trial = trial
        self.partition = partition
        self.container_image = container_image
        self.container_mounts = container_mounts
        self.account = account
        self.time_limit = time_limit
        self.log_dir = log_dir
        self.gpus_per_node = gpus_per_node
        self.mem_per_node = mem_per_node
        self.env_vars = dict(env_vars or {})
        self.job_name = f"{experiment}_{trial}"
        self.job_id: Optional[str] = None

    # ------------------------------------------------------------- script
    def render_script(self, cmd: List[str], n_procs: int,
                      master_port: int = 29501) -> str:
        """The sbatch script: srun launches n_procs tasks (one per GPU);
        each task derives RANK/LOCAL_RANK from SLURM_PROCID/SLURM_LOCALID
        and rendezvous at the first node."""
        nnodes = max(1, (n_procs + self.gpus_per_node - 1) // self.gpus_per_node)
        per_node = min(n_procs, self.gpus_per_node)
        lines = [
            "#!/bin/bash",
            f"#SBATCH --job-name={self.job_name}",
            f"#SBATCH --nodes={nnodes}",
            f"#SBATCH --ntasks={n_procs}",
            f"#SBATCH --ntasks-per-node={per_node}",
            f"#SBATCH --gpus-per-task=1",
            f"#SBATCH --output={self.log_dir}/{self.job_name}.%j.out",
        ]
        if self.partition:
            lines.append(f"#SBATCH --partition={self.partition}")
        if self.account:
            lines.append(f"#SBATCH --account={self.account}")
        if self.time_limit:
            lines.append(f"#SBATCH --time={self.time_limit}")
        if self.mem_per_node:
            lines.append(f"#SBATCH --mem={self.mem_per_node}")
        if self.container_image:
            # pyxis/enroot (reference utils.py: container_image/mounts)
            lines.append(f"#SBATCH --container-image={self.container_image}")
            if self.container_mounts:
                lines.append(
                    f"#SBATCH --container-mounts={self.container_mounts}")
        lines += [
            "",
            'MASTER_ADDR=$(scontrol show hostnames "$SLURM_JOB_NODELIST" | head -n1)',
            "export MASTER_ADDR",
            f"export MASTER_PORT={master_port}",
            f"export WORLD_SIZE={n_procs}",
            "export HSA_ENABLE_IPC_MODE_LEGACY=0",  # dmabuf IPC (RCCL)
        ]
        for k, v in self.env_vars.items():
            lines.append(f"export {k}={v}")
        quoted = " ".join(_shquote(c) for c in cmd)
        lines += [
            "srun --export=ALL bash -c '",
            "  export RANK=$SLURM_PROCID",
            "  export LOCAL_RANK=$SLURM_LOCALID",
            "  export ROCR_VISIBLE_DEVICES=$SLURM_LOCALID",
            f"  exec {quoted}",
            "'",
            "",
        ]
        return "\n".join(lines)

    # ------------------------------------------------------------- submit
    def submit_array(self, cmd: List[str], n_procs: int,
                     env_extra: Optional[Dict[str, str]] = None,
                     master_port: int = 29501):
        if env_extra:
            self.env_vars.update(env_extra)
        os.makedirs(self.log_dir, exist_ok=True)
        script = self.render_script(cmd, n_procs, master_port)
        path = os.path.join(self.log_dir, f"{self.job_name}.sbatch")
        with open(path, "w") as f:
            f.write(script)
        out = _run(["sbatch", path])
        m = re.search(r"Submitted batch job (\d+)", out)
        if not m:
            raise JobException(f"cannot parse sbatch output: {out!r}")
        self.job_id = m.group(1)
        logger.info("submitted %s as job %s", self.job_name, self.job_id)
        return self

    # -------------------------------------------------------------- query
    def find(self) -> JobInfo:
        assert self.job_id is not None, "not submitted"
        out = subprocess.run(
            ["squeue", "-j", self.job_id, "-h", "-o", "%T|%N"],
            capture_output=True, text=True,
        )
        line = out.stdout.strip().splitlines()
        if out.returncode != 0 or not line:
            # not in queue anymore: completed or failed; sacct if present
            acct = subprocess.run(
                ["sacct", "-j", self.job_id, "-n", "-X", "-o", "State"],
                capture_output=True, text=True,
            )
            state = (acct.stdout.strip().split() or ["COMPLETED"])[0]
            return JobInfo(self.job_name, state.rstrip("+"), self.job_id)
        state, host = (line[0].split("|") + [None])[:2]
        return JobInfo(self.job_name, state, self.job_id, host or None)

    def find_all(self, name_regex: str = ".*") -> List[JobInfo]:
        out = subprocess.run(
            ["squeue", "-h", "-o", "%j|%T|%i|%N"], capture_output=True,
            text=True,
        )
        jobs = []
        for ln in out.stdout.strip().splitlines():
            parts = ln.split("|")
            if len(parts) >= 3 and re.fullmatch(name_regex, parts[0]):
                jobs.append(JobInfo(parts[0], parts[1], parts[2],
                                    parts[3] if len(parts) > 3 else None))
        return jobs

    # --------------------------------------------------------------- wait
    def wait(self, timeout: Optional[float] = None,
             poll_interval: float = 10.0) -> int:
        t0 = time.time()
        while True:
            info = self.find()
            if info.state == "COMPLETED":
                return 0
            if info.state in _FAILED:
                raise JobException(f"job {self.job_id} {info.state}")
            if timeout and time.time() - t0 > timeout:
                self.stop_all()
                raise JobException("timeout")
            time.sleep(poll_interval)

    def stop_all(self, sig: str = "SIGKILL"):
        if self.job_id:
            subprocess.run(["scancel", "-s", sig, self.job_id],
                           capture_output=True)


def _shquote(s: str) -> str:
    if re.fullmatch(r"[A-Za-z0-9_./:=,@%+-]+", s):
        return s
    return '"' + s.replace('"', '\\"') + '"'
