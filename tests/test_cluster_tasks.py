"""Clustering task flow: parent dispatches batches, workers claim,
absorb picks the best, playlists persist."""

import json

import numpy as np
import pytest

from audiomuse_amd.db import connect
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.db.store import save_track_analysis_and_embedding
from audiomuse_amd.taskqueue import SUCCESS, enqueue, task_row
from audiomuse_amd.taskqueue.worker import Worker


@pytest.fixture
def seeded(tmp_db_url):
    conn = connect(tmp_db_url)
    init_db(conn)
    rng = np.random.default_rng(0)
    centers = rng.standard_normal((3, 200)) * 4
    for i in range(90):
        c = i % 3
        save_track_analysis_and_embedding(
            conn, f"t{i}", title=f"T{i}", author=f"artist{i % 9}",
            mood_vector={"rock": float(c == 0), "jazz": float(c == 1),
                         "chill": float(c == 2)},
            other_features={"happy": 0.5},
            embedding=(centers[c] + rng.standard_normal(200) * 0.3
                       ).astype(np.float32))
    yield conn, tmp_db_url
    conn.close()


@pytest.mark.slow
def test_clustering_end_to_end(seeded):
    conn, url = seeded
    tid = enqueue(conn, "run_clustering",
                  {"algorithm": "kmeans", "runs": 8,
                   "iterations_per_batch": 4, "top_n": 5,
                   "drain_timeout": 120.0}, queue="high")
    import threading

    workers = [Worker(db_url=url, max_jobs=10) for _ in range(2)]
    threads = [threading.Thread(target=lambda w=w: w.run_forever(idle_timeout=15.0))
               for w in workers]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=240)

    row = task_row(conn, tid)
    assert row["status"] == SUCCESS, row["result"]
    result = json.loads(row["result"])
    assert result["playlists"] >= 1
    assert result["failed_batches"] == 0

    pls = conn.execute(
        "SELECT name, item_ids FROM playlist WHERE kind='automatic'").fetchall()
    assert len(pls) == result["playlists"]
    for p in pls:
        ids = json.loads(p["item_ids"])
        assert len(ids) >= 1
        assert p["name"].endswith("_automatic")


def test_clustering_tolerates_failed_batches(seeded, monkeypatch):
    """Stall-valve semantics (reference: clustering.py:1426-1449 +
    absorb :1581): up to CLUSTERING_MAX_FAILED_BATCHES dead batches are
    absorbed; one more fails the parent run."""
    from audiomuse_amd import config as C
    from audiomuse_amd.taskqueue import FAILURE
    from audiomuse_amd.taskqueue.worker import _REGISTRY, import_builtin_handlers

    conn, url = seeded
    import_builtin_handlers()
    real = _REGISTRY["run_clustering_batch"]

    def flaky(ctx, payload):
        # keyed on the batch seed so retries of a doomed batch also die
        if payload["seed"] % 2 == 1:     # half the batches die
            raise RuntimeError("injected batch crash")
        return real(ctx, payload)

    monkeypatch.setitem(_REGISTRY, "run_clustering_batch", flaky)
    monkeypatch.setattr(C, "CLUSTERING_MAX_FAILED_BATCHES", 2)

    def run_workers():
        import threading

        ws = [Worker(db_url=url, max_jobs=30) for _ in range(2)]
        ts = [threading.Thread(target=lambda w=w: w.run_forever(
            idle_timeout=10.0)) for w in ws]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=240)

    tid = enqueue(conn, "run_clustering",
                  {"runs": 8, "iterations_per_batch": 2,
                   "algorithm": "kmeans", "drain_timeout": 120.0})
    run_workers()
    row = task_row(conn, tid)
    assert row["status"] == SUCCESS, row["details"]
    res = json.loads(row["result"])
    assert res["failed_batches"] == 2 and res["playlists"] >= 1

    # over the valve: 3 of 4 batches die -> parent FAILURE

    def flakier(ctx, payload):
        if payload["seed"] % 4 != 0:
            raise RuntimeError("injected batch crash")
        return real(ctx, payload)

    monkeypatch.setitem(_REGISTRY, "run_clustering_batch", flakier)
    tid2 = enqueue(conn, "run_clustering",
                   {"runs": 8, "iterations_per_batch": 2,
                    "algorithm": "kmeans", "drain_timeout": 120.0})
    run_workers()
    assert task_row(conn, tid2)["status"] == FAILURE
