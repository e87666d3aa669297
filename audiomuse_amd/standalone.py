"""Standalone supervisor: one command runs the whole stack on one box.

Reference analog: native-build/native_common/supervisor_common.py +
deployment/supervisord.conf — the packaged app starts the web process,
the queue workers (high + default), and a maintenance loop under one
supervisor with restart handling and a control-plane restart listener.
Here the same process tree runs from `python -m audiomuse_amd
standalone`: web in the supervisor process, one worker subprocess per
GPU (or --workers N), each pinned to its rank's device via
HIP_VISIBLE_DEVICES (the 1-process-per-GPU deployment shape, SURVEY
§2.2 P1). Workers that die restart with backoff; a control-plane
restart request recycles them cleanly (workers ack and exit on their
own — taskqueue/control.py — and the supervisor relaunches)."""

from __future__ import annotations

import logging
import os
import subprocess
import sys
import threading
import time
from typing import Callable, Dict, List, Optional

logger = logging.getLogger(__name__)


def default_worker_count() -> int:
    """One worker per visible GPU; 1 on CPU-only boxes."""
    try:
        import torch
        n = torch.cuda.device_count()
        return max(n, 1)
    except Exception:
        return 1


def _worker_cmd(db_url: Optional[str], queues: str) -> List[str]:
    cmd = [sys.executable, "-m", "audiomuse_amd", "worker",
           "--queues", queues]
    if db_url:
        cmd += ["--db", db_url]
    return cmd


class Supervisor:
    """Restart-with-backoff process tree (supervisor_common behavior).

    spawn_fn is injectable for tests; it must return an object with
    poll() -> Optional[int] and terminate()/wait().
    """

    def __init__(self, db_url: Optional[str] = None,
                 workers: Optional[int] = None,
                 queues: str = "high,default",
                 spawn_fn: Optional[Callable] = None,
                 backoff_seconds: float = 2.0,
                 max_restarts: int = 50):
        self.db_url = db_url
        self.n_workers = workers if workers is not None \
            else default_worker_count()
        self.queues = queues
        self.backoff = backoff_seconds
        self.max_restarts = max_restarts
        self._spawn = spawn_fn or self._spawn_subprocess
        self._procs: Dict[int, object] = {}
        self.restarts: Dict[int, int] = {}
        self._stop = threading.Event()

    def _spawn_subprocess(self, rank: int):
        env = dict(os.environ)
        # rank -> its own GPU: the deployment shape the queue was built
        # for (one worker process per MI355X)
        env["HIP_VISIBLE_DEVICES"] = str(rank)
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        return subprocess.Popen(_worker_cmd(self.db_url, self.queues),
                                env=env)

    def start(self) -> None:
        for rank in range(self.n_workers):
            self._procs[rank] = self._spawn(rank)
            self.restarts[rank] = 0
        logger.info("standalone supervisor: %d worker(s) up",
                    self.n_workers)

    def tick(self) -> int:
        """One supervision pass: restart dead workers with backoff.
        Returns how many were restarted."""
        n = 0
        for rank, proc in list(self._procs.items()):
            if proc.poll() is None:
                continue
            if self.restarts[rank] >= self.max_restarts:
                logger.error("worker %d exceeded max restarts", rank)
                continue
            self.restarts[rank] += 1
            logger.warning("worker %d exited rc=%s; restart #%d",
                           rank, proc.poll(), self.restarts[rank])
            time.sleep(min(self.backoff * self.restarts[rank], 30.0))
            self._procs[rank] = self._spawn(rank)
            n += 1
        return n

    def run_forever(self, poll_seconds: float = 2.0) -> None:
        self.start()
        try:
            while not self._stop.is_set():
                self.tick()
                self._stop.wait(poll_seconds)
        finally:
            self.shutdown()

    def stop(self) -> None:
        self._stop.set()

    def shutdown(self) -> None:
        for proc in self._procs.values():
            try:
                proc.terminate()
            except Exception:
                pass
        for proc in self._procs.values():
            try:
                proc.wait(timeout=10)
            except Exception:
                pass


def run_standalone(db_url: Optional[str] = None, host: str = "0.0.0.0",
                   port: int = 8000, workers: Optional[int] = None,
                   no_auth: bool = False) -> None:
    """Web in this process + supervised workers (the packaged-app
    entry; reference scripts/standalone/)."""
    sup = Supervisor(db_url=db_url, workers=workers)
    t = threading.Thread(target=sup.run_forever, daemon=True)
    t.start()
    from audiomuse_amd.web.app import create_app
    app = create_app(db_url, auth_disabled=no_auth)
    try:
        app.run(host=host, port=port)
    finally:
        sup.stop()
        sup.shutdown()
