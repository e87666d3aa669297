"""Standalone supervisor (audiomuse_amd/standalone.py) — the packaged-
app process tree (reference: native_common/supervisor_common.py +
supervisord.conf): restart-with-backoff, per-rank GPU pinning, clean
shutdown, and a real end-to-end run with one live worker subprocess."""

import os
import subprocess
import sys
import time

import numpy as np
import pytest

from audiomuse_amd.standalone import (Supervisor, _worker_cmd,
                                      default_worker_count)


class FakeProc:
    def __init__(self, rank):
        self.rank = rank
        self.rc = None
        self.terminated = False

    def poll(self):
        return self.rc

    def terminate(self):
        self.terminated = True
        self.rc = -15

    def wait(self, timeout=None):
        return self.rc


def test_worker_cmd_shape():
    cmd = _worker_cmd("sqlite:///x.db", "high,default")
    assert cmd[:3] == [sys.executable, "-m", "audiomuse_amd"]
    assert "worker" in cmd and "sqlite:///x.db" in cmd


def test_default_worker_count_positive():
    assert default_worker_count() >= 1


def test_supervisor_restarts_dead_workers_with_cap():
    spawned = []

    def spawn(rank):
        p = FakeProc(rank)
        spawned.append(p)
        return p

    sup = Supervisor(workers=2, spawn_fn=spawn, backoff_seconds=0.0,
                     max_restarts=2)
    sup.start()
    assert len(spawned) == 2
    # worker 0 dies twice -> restarted twice, then capped
    spawned[0].rc = 1
    assert sup.tick() == 1
    sup._procs[0].rc = 1
    assert sup.tick() == 1
    sup._procs[0].rc = 1
    assert sup.tick() == 0          # max_restarts reached
    assert sup.restarts[0] == 2 and sup.restarts[1] == 0
    sup.shutdown()
    assert all(p.terminated or p.rc is not None for p in spawned)


def test_supervisor_rank_pinning(monkeypatch):
    seen = {}
    real_popen = subprocess.Popen

    def fake_popen(cmd, env=None):
        seen[env["HIP_VISIBLE_DEVICES"]] = cmd
        return FakeProc(0)

    monkeypatch.setattr(subprocess, "Popen", fake_popen)
    sup = Supervisor(db_url="sqlite:///t.db", workers=3)
    sup.start()
    assert set(seen) == {"0", "1", "2"}   # one GPU per worker rank
    sup.shutdown()


def test_standalone_worker_subprocess_drains_queue(tmp_sqlite_url):
    """End to end: a REAL supervised worker subprocess claims and
    finishes a queued job (one rank, synthetic catalogue task)."""
    from audiomuse_amd.db import connect
    from audiomuse_amd.db.schema import init_db
    from audiomuse_amd.taskqueue import SUCCESS, enqueue, task_row

    conn = connect(tmp_sqlite_url)
    init_db(conn)
    tid = enqueue(conn, "clean_orphans", {"delete": False})

    sup = Supervisor(db_url=tmp_sqlite_url, workers=1)
    sup.start()
    try:
        deadline = time.time() + 30
        while time.time() < deadline:
            row = task_row(conn, tid)
            if row["status"] == SUCCESS:
                break
            time.sleep(0.3)
        assert task_row(conn, tid)["status"] == SUCCESS
    finally:
        sup.shutdown()
    conn.close()


def test_reuseport_web_procs_share_one_port(tmp_sqlite_url):
    """Two SO_REUSEPORT web processes bind the SAME port and both serve
    (web/serve.py — the web tier's scale-out path; one process's qps
    ceiling is measured in profiles/r2_http_load16c.log)."""
    import json
    import subprocess
    import sys
    import time
    import urllib.request

    from audiomuse_amd.db import connect
    from audiomuse_amd.db.schema import init_db

    conn = connect(tmp_sqlite_url)
    init_db(conn)
    conn.close()
    import socket as s_mod
    if not hasattr(s_mod, "SO_REUSEPORT"):
        import pytest as _pytest
        _pytest.skip("no SO_REUSEPORT")
    # pick a free port
    probe = s_mod.socket()
    probe.bind(("127.0.0.1", 0))
    port = probe.getsockname()[1]
    probe.close()
    cmd = [sys.executable, "-m", "audiomuse_amd", "web",
           "--host", "127.0.0.1", "--port", str(port), "--reuse-port",
           "--no-auth", "--db", tmp_sqlite_url]
    procs = [subprocess.Popen(cmd) for _ in range(2)]
    try:
        deadline = time.time() + 30
        body = None
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}/health", timeout=2) as r:
                    body = json.loads(r.read())
                break
            except Exception:
                time.sleep(0.3)
        assert body and body["status"] == "ok"
        assert all(p.poll() is None for p in procs)  # both still serving
        for _ in range(6):                           # a few more round trips
            with urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/health", timeout=2) as r:
                assert json.loads(r.read())["status"] == "ok"
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            p.wait(timeout=10)
