"""Local scheduler: spawn one worker process per GPU and babysit them.

Reference semantics: realhf/scheduler/client.py:44 (SchedulerClient) +
scheduler/local/client.py:66 (subprocess scheduler) + the controller's
worker-failure monitoring.  SPMD needs GANG restart: a dead rank hangs
every collective, so on failure the whole array is stopped and (with
max_restarts > 0) respawned — paired with recover_mode=auto the job
resumes from the last recover checkpoint (runtime/trainer.py).  SLURM
multi-node scheduling lives in scheduler/slurm.py."""
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
    def __init__(self, experiment: str, trial: str, max_restarts: int = 0):
        self.experiment = experiment
        self.trial = trial
        self.max_restarts = max_restarts
        self.procs: List[subprocess.Popen] = []
        self._spawn_args = None

    def submit_array(self, cmd: List[str], n_procs: int,
                     env_extra: Optional[Dict[str, str]] = None,
                     master_port: int = 29501):
        self._spawn_args = (list(cmd), n_procs, dict(env_extra or {}),
                            master_port)
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
        """Babysit the gang.  On a worker failure: stop everyone (a dead
        rank hangs every collective) and, if restarts remain, respawn the
        whole array on a fresh rendezvous port — workers launched with
        recover_mode=auto resume from the last recover checkpoint."""
        restarts = 0
        t0 = time.time()
        try:
            while True:
                codes = [p.poll() for p in self.procs]
                if any(c is not None and c != 0 for c in codes):
                    self.stop_all()
                    bad = [i for i, c in enumerate(codes) if c not in (None, 0)]
                    if restarts < self.max_restarts:
                        restarts += 1
                        logger.warning(
                            "worker(s) %s failed (%s); gang restart %d/%d",
                            bad, codes, restarts, self.max_restarts,
                        )
                        cmd, n, env_extra, port = self._spawn_args
                        self.procs = []
                        self.submit_array(cmd, n, env_extra,
                                          master_port=port + restarts)
                        continue
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
