"""Worker loop (reference: taskqueue/worker.py Worker.run_forever).

A worker claims jobs from its queues, runs the registered handler with a
heartbeat thread keeping the lease alive, and recycles after max_jobs.
Handlers receive a TaskContext giving cooperative-cancel checks and
progress reporting. One worker process per GPU rank is the deployment
shape; job fork-isolation is available via
run_in_subprocess for crash containment.
"""

from __future__ import annotations

import json
import logging
import multiprocessing as mp
import os
import sqlite3
import threading
import time
import traceback
import uuid
from typing import Callable, Dict, Optional, Sequence

from audiomuse_amd import config as C
from audiomuse_amd.db import connect
from audiomuse_amd.taskqueue import FAILURE, SUCCESS
from audiomuse_amd.taskqueue import sql as qsql

logger = logging.getLogger(__name__)

_REGISTRY: Dict[str, Callable] = {}


def task_handler(task_type: str):
    """Decorator registering a handler: fn(ctx, payload) -> dict|None."""
    def deco(fn: Callable) -> Callable:
        _REGISTRY[task_type] = fn
        return fn
    return deco


def get_handler(task_type: str) -> Optional[Callable]:
    return _REGISTRY.get(task_type)


def import_builtin_handlers() -> None:
    """Register the built-in task modules (idempotent)."""
    import audiomuse_amd.analysis.maintenance  # noqa: F401
    import audiomuse_amd.analysis.migration  # noqa: F401
    import audiomuse_amd.analysis.tasks  # noqa: F401
    import audiomuse_amd.cluster.tasks  # noqa: F401


class CancelledError(RuntimeError):
    pass


class TaskContext:
    def __init__(self, conn: sqlite3.Connection, row: sqlite3.Row,
                 worker_id: str):
        self.conn = conn
        self.task_id = row["task_id"]
        self.task_type = row["task_type"]
        self.parent_task_id = row["parent_task_id"]
        self.worker_id = worker_id
        self.payload = json.loads(row["payload"] or "{}")

    def cancelled(self) -> bool:
        return qsql.is_cancelled(self.conn, self.task_id)

    def check_cancelled(self) -> None:
        if self.cancelled():
            raise CancelledError(self.task_id)

    def report(self, progress: float, details: Optional[str] = None) -> None:
        qsql.update_progress(self.conn, self.task_id, progress, details)


class Worker:
    def __init__(self, db_url: Optional[str] = None,
                 queues: Sequence[str] = ("high", "default"),
                 worker_id: Optional[str] = None,
                 max_jobs: Optional[int] = None,
                 poll_seconds: Optional[float] = None):
        self.db_url = db_url or C.DATABASE_URL
        self.queues = tuple(queues)
        self.worker_id = worker_id or f"{os.getpid()}-{uuid.uuid4().hex[:8]}"
        self.max_jobs = max_jobs if max_jobs is not None else C.WORKER_MAX_JOBS
        self.poll_seconds = poll_seconds or C.QUEUE_POLL_SECONDS
        self._stop = threading.Event()
        self.jobs_done = 0

    def stop(self) -> None:
        self._stop.set()

    def _heartbeat_loop(self, conn_url: str, task_id: str,
                        stop: threading.Event) -> None:
        conn = connect(conn_url)
        try:
            while not stop.wait(C.QUEUE_HEARTBEAT_SECONDS):
                if not qsql.heartbeat(conn, task_id, self.worker_id):
                    return
        finally:
            conn.close()

    def run_one(self, conn: sqlite3.Connection) -> bool:
        """Claim and run a single job. Returns True if a job ran."""
        row = qsql.claim(conn, self.worker_id, self.queues)
        if row is None:
            return False
        ctx = TaskContext(conn, row, self.worker_id)
        hb_stop = threading.Event()
        hb = threading.Thread(target=self._heartbeat_loop,
                              args=(self.db_url, ctx.task_id, hb_stop),
                              daemon=True)
        hb.start()
        try:
            handler = get_handler(ctx.task_type)
            if handler is None:
                raise RuntimeError(f"no handler for task type {ctx.task_type!r}")
            if ctx.cancelled():
                raise CancelledError(ctx.task_id)
            result = handler(ctx, ctx.payload)
            qsql.finish(conn, ctx.task_id, self.worker_id, SUCCESS,
                        result=result if isinstance(result, dict) else None)
        except CancelledError:
            logger.info("task %s cancelled", ctx.task_id)
            # row already REVOKED (or ancestor); leave terminal status alone
        except Exception:
            logger.exception("task %s failed", ctx.task_id)
            qsql.finish(conn, ctx.task_id, self.worker_id, FAILURE,
                        result={"traceback": traceback.format_exc()[-2000:]})
        finally:
            hb_stop.set()
            hb.join(timeout=2.0)
        self.jobs_done += 1
        return True

    def run_forever(self, idle_timeout: Optional[float] = None) -> None:
        """Claim loop (reference: worker.py:404). Exits on stop(),
        max_jobs recycle, or idle_timeout with an empty queue."""
        conn = connect(self.db_url)
        from audiomuse_amd.db.schema import init_db
        init_db(conn)
        # PG: park on LISTEN channels instead of polling (reference:
        # worker.py:197 + listen.py); SQLite: bounded polling.
        from audiomuse_amd import db as dbmod

        dbmod.listen(conn, dbmod.CHAN_JOB)
        dbmod.listen(conn, dbmod.CHAN_CANCEL)
        dbmod.listen(conn, dbmod.CHAN_CONTROL)
        try:
            import_builtin_handlers()
        except Exception:
            logger.exception("builtin handler import failed")
        try:
            # hydrate persisted config overrides (reference: worker.py:698)
            from audiomuse_amd.db.store import get_app_config

            C.apply_db_overrides(get_app_config(conn))
        except Exception:
            logger.exception("config hydrate failed")
        try:
            # DB-stored plugins contribute hooks + task handlers here too
            from audiomuse_amd.plugin import plugin_manager

            plugin_manager.load_from_db(conn)
        except Exception:
            logger.exception("plugin load failed")
        idle_since = time.time()
        last_maintenance = 0.0
        try:
            while not self._stop.is_set():
                if self.max_jobs and self.jobs_done >= self.max_jobs:
                    return
                # control plane: stop/restart broadcast (reference:
                # control.py; reclaim stands down inside the window)
                from audiomuse_amd.taskqueue import control as qctl

                reqs = qctl.pending_requests(
                    conn, self.worker_id,
                    [qctl.ACTION_STOP_WORKERS, qctl.ACTION_RESTART])
                if reqs:
                    for r in reqs:
                        qctl.ack(conn, r["id"], self.worker_id)
                    logger.info("worker %s stopping on control request",
                                self.worker_id)
                    return
                now = time.time()
                if now - last_maintenance > C.QUEUE_LEASE_SECONDS:
                    if not qctl.control_window_active(conn):
                        qsql.reclaim_orphans(conn)
                    last_maintenance = now
                if self.run_one(conn):
                    idle_since = time.time()
                    continue
                if idle_timeout is not None and time.time() - idle_since > idle_timeout:
                    return
                if getattr(conn, "kind", "sqlite") == "postgres":
                    conn.wait_notify(self.poll_seconds)  # NOTIFY wakes us early
                else:
                    self._stop.wait(self.poll_seconds)
        finally:
            conn.close()


def run_in_subprocess(db_url: str, queues: Sequence[str],
                      max_jobs: int = 1) -> int:
    """Fork-isolated worker (reference: worker.py:497-553 fork-per-job)."""
    def _child() -> None:
        w = Worker(db_url=db_url, queues=queues, max_jobs=max_jobs)
        w.run_forever(idle_timeout=1.0)

    proc = mp.get_context("fork").Process(target=_child)
    proc.start()
    proc.join()
    return proc.exitcode or 0
