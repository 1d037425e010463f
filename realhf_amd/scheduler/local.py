"""Local scheduler: spawn one worker process per GPU and babysit them.

Reference semantics: realhf/scheduler/client.py:44 (SchedulerClient) +
scheduler/local/client.py:66 (subprocess scheduler).  SLURM multi-node
scheduling is out of scope for the single-node target (gap vs reference
noted in README parity table)."""
import os
import signal
import subprocess
import time
from typing import Dict, List, Optional

from realhf_amd.base import logging

logger = logging.getLogger("scheduler")


class JobException(Exception):
    pass


class LocalScheduler:
    def __init__(self, experiment: str, trial: str):
        self.experiment = experiment
        self.trial = trial
        self.procs: List[subprocess.Popen] = []

    def submit_array(self, cmd: List[str], n_procs: int,
                     env_extra: Optional[Dict[str, str]] = None,
                     master_port: int = 29501):
        for rank in range(n_procs):
            env = dict(os.environ)
            env.update(env_extra or {})
            env.update(
                RANK=str(rank),
                LOCAL_RANK=str(rank),
                WORLD_SIZE=str(n_procs),
                MASTER_ADDR="127.0.0.1",
                MASTER_PORT=str(master_port),
            )
            p = subprocess.Popen(cmd, env=env)
            self.procs.append(p)
        return self

    def wait(self, timeout: Optional[float] = None) -> int:
        t0 = time.time()
        try:
            while True:
                codes = [p.poll() for p in self.procs]
                if any(c is not None and c != 0 for c in codes):
                    self.stop_all()
                    bad = [i for i, c in enumerate(codes) if c not in (None, 0)]
                    raise JobException(f"worker(s) {bad} failed: {codes}")
                if all(c == 0 for c in codes):
                    return 0
                if timeout and time.time() - t0 > timeout:
                    self.stop_all()
                    raise JobException("timeout")
                time.sleep(0.5)
        except KeyboardInterrupt:
            self.stop_all()
            raise

    def stop_all(self):
        for p in self.procs:
            if p.poll() is None:
                p.send_signal(signal.SIGTERM)
        t0 = time.time()
        for p in self.procs:
            while p.poll() is None and time.time() - t0 < 10:
                time.sleep(0.2)
            if p.poll() is None:
                p.kill()
