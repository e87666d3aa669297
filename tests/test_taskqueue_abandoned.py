"""Abandoned-job / worker-death queue semantics, modeled on the
reference's deepest queue suites (test_taskqueue_abandoned.py 820 LoC,
test_queue_locks.py, test_control_ack_wait.py, test_frozen_children.py):
several connections against one DB, simulated worker death (connection
kill on the PG backend = advisory-lock release), reclaim attempt
accounting, control-window stand-down, cancel cascades, blob vacuum
interplay. Runs on BOTH backends via tmp_db_url."""

import threading
import time

import pytest

from audiomuse_amd.db import backend_kind, connect
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.taskqueue import (FAILURE, PENDING, REVOKED, RUNNING,
                                     SUCCESS, cancel_task_recursive, enqueue,
                                     get_shared_blob, task_row,
                                     vacuum_shared_payloads)
from audiomuse_amd.taskqueue import control as qctl
from audiomuse_amd.taskqueue import sql as qsql
from audiomuse_amd.taskqueue.worker import TaskContext, Worker, task_handler


@pytest.fixture
def db(tmp_db_url):
    conn = connect(tmp_db_url)
    init_db(conn)
    yield conn, tmp_db_url
    conn.close()


# -- worker death ------------------------------------------------------------

def test_dead_worker_connection_frees_job_on_pg(db):
    """PG backend: the claiming connection DIES (no finish, no lease
    expiry) — the advisory lock releases with the socket and reclaim
    requeues the job (reference maintenance.py:177 semantics)."""
    conn, url = db
    if backend_kind(conn) != "postgres":
        pytest.skip("advisory-lock death is the PG liveness path")
    tid = enqueue(conn, "noop")
    w = connect(url)
    row = qsql.claim(w, "doomed", lease_seconds=3600.0)  # lease still live
    assert row["task_id"] == tid
    assert qsql.reclaim_orphans(conn) == 0      # lock held: not an orphan
    w.close()                                    # worker process dies
    time.sleep(0.3)
    assert qsql.reclaim_orphans(conn) == 1       # lock free -> requeued
    assert task_row(conn, tid)["status"] == PENDING


def test_lease_expiry_frees_job_on_any_backend(db):
    conn, _ = db
    tid = enqueue(conn, "noop")
    qsql.claim(conn, "w1", lease_seconds=0.05)
    time.sleep(0.15)
    assert qsql.reclaim_orphans(conn) == 1
    assert task_row(conn, tid)["status"] == PENDING


def test_reclaim_exhausts_attempts_to_failure(db):
    conn, _ = db
    tid = enqueue(conn, "noop", max_attempts=2)
    for expect in (PENDING, PENDING, FAILURE):
        qsql.claim(conn, "w1", lease_seconds=0.03)
        time.sleep(0.1)
        qsql.reclaim_orphans(conn)
        row = task_row(conn, tid)
        if row["status"] == FAILURE:
            break
    row = task_row(conn, tid)
    assert row["status"] == FAILURE
    assert "orphan" in (row["details"] or "")
    assert row["attempts"] == 2


def test_live_worker_survives_reclaim_cycles(db):
    """A heartbeating worker is never reclaimed even across many
    maintenance passes (reference: lock held = alive)."""
    conn, url = db
    tid = enqueue(conn, "noop")
    w = connect(url)
    qsql.claim(w, "alive", lease_seconds=0.3)
    for _ in range(4):
        assert qsql.heartbeat(w, tid, "alive", lease_seconds=0.3)
        assert qsql.reclaim_orphans(conn) == 0
        time.sleep(0.05)
    assert qsql.finish(w, tid, "alive", SUCCESS)
    w.close()


def test_finish_after_reclaim_is_rejected(db):
    """A zombie worker finishing a job that was already reclaimed and
    re-claimed by another worker must not clobber it."""
    conn, url = db
    tid = enqueue(conn, "noop")
    zombie = connect(url)
    qsql.claim(zombie, "zombie", lease_seconds=0.05)
    time.sleep(0.15)
    if backend_kind(conn) == "postgres":
        zombie.close()          # lock must die for the reclaim
        zombie = None
    assert qsql.reclaim_orphans(conn) == 1
    row2 = qsql.claim(conn, "w2")
    assert row2["task_id"] == tid and row2["attempts"] == 2
    if zombie is not None:
        assert not qsql.finish(zombie, tid, "zombie", SUCCESS)
    assert task_row(conn, tid)["status"] == RUNNING
    assert qsql.finish(conn, tid, "w2", SUCCESS)


def test_heartbeat_of_reclaimed_job_fails(db):
    conn, url = db
    tid = enqueue(conn, "noop")
    w = connect(url)
    qsql.claim(w, "w1", lease_seconds=0.05)
    time.sleep(0.15)
    if backend_kind(conn) == "postgres":
        w.close()
        w = connect(url)
    qsql.reclaim_orphans(conn)
    assert not qsql.heartbeat(w, tid, "w1")
    w.close()


# -- control window stand-down ----------------------------------------------

def test_reclaim_stands_down_inside_control_window(db):
    """During a restart broadcast the maintenance pass must NOT requeue
    jobs (reference maintenance.py:112: workers are intentionally
    stopping; requeue would double-run)."""
    conn, url = db
    tid = enqueue(conn, "noop")
    qsql.claim(conn, "w1", lease_seconds=0.05)
    time.sleep(0.1)
    rid = qctl.publish_control_request(conn, qctl.ACTION_RESTART,
                                       window_seconds=30.0)
    assert qctl.control_window_active(conn)
    # worker loop consults the window before reclaiming (worker.py);
    # assert the window is visible to every connection
    other = connect(url)
    assert qctl.control_window_active(other)
    other.close()
    qctl.ack(conn, rid, "worker-1")
    assert qctl.ack_count(conn, rid) == 1
    assert task_row(conn, tid)["status"] == RUNNING  # untouched meanwhile


def test_control_ack_wait_and_expiry(db):
    conn, _ = db
    rid = qctl.publish_control_request(conn, qctl.ACTION_STOP_WORKERS,
                                       window_seconds=0.2)
    done = []

    def acker():
        c2 = connect(db[1])   # connections are per-thread
        time.sleep(0.05)
        qctl.ack(c2, rid, "w-a")
        c2.close()
        done.append(1)

    t = threading.Thread(target=acker)
    t.start()
    assert qctl.wait_for_acks(conn, rid, expected=1, timeout=2.0)
    t.join()
    time.sleep(0.25)
    assert not qctl.control_window_active(conn)   # expired


# -- cancel cascades ----------------------------------------------------------

def test_cancel_cascades_to_grandchildren_and_spares_done(db):
    conn, _ = db
    root = enqueue(conn, "noop")
    kids = [enqueue(conn, "noop", parent_task_id=root) for _ in range(3)]
    grand = enqueue(conn, "noop", parent_task_id=kids[0])
    # one child finishes before the cancel: terminal states stay
    # untouched (priority makes the claim deterministic)
    conn.execute("UPDATE task_status SET priority=9 WHERE task_id=?",
                 (kids[1],))
    row = qsql.claim(conn, "w1", queues=("default",))
    assert row["task_id"] == kids[1]
    finished = kids[1]
    qsql.finish(conn, finished, "w1", SUCCESS)
    n = cancel_task_recursive(conn, root)
    statuses = {t: task_row(conn, t)["status"]
                for t in [root, grand] + kids}
    assert statuses[root] == REVOKED and statuses[grand] == REVOKED
    assert sum(1 for s in statuses.values() if s == REVOKED) == n
    if finished:
        assert task_row(conn, finished)["status"] == SUCCESS


def test_running_child_observes_ancestor_cancel(db):
    conn, _ = db
    root = enqueue(conn, "noop")
    child = enqueue(conn, "noop", parent_task_id=root)
    row = qsql.claim(conn, "w1", queues=("default",))
    ctx = TaskContext(conn, task_row(conn, child), "w1")
    assert not ctx.cancelled()
    cancel_task_recursive(conn, root)
    assert ctx.cancelled()          # sees REVOKED ancestor cooperatively


def test_missing_parent_row_means_cancelled(db):
    """Reference docs/ALGORITHM.md:193-197: a missing ancestor row is a
    cancel signal (archived / purged run)."""
    conn, _ = db
    child = enqueue(conn, "noop", parent_task_id="gone-task")
    assert qsql.is_cancelled(conn, child)


# -- shared payload blobs -----------------------------------------------------

def test_shared_blob_dedupe_and_vacuum_interplay(db):
    conn, _ = db
    blob = b"x" * 2048
    t1 = enqueue(conn, "noop", shared_blob=blob)
    t2 = enqueue(conn, "noop", shared_blob=blob)   # same content: one row
    n = conn.execute("SELECT COUNT(*) AS n FROM shared_payload"
                     ).fetchone()["n"]
    assert n == 1
    assert get_shared_blob(conn, t1) == blob
    # both tasks live: vacuum keeps the blob
    assert vacuum_shared_payloads(conn) == 0
    # finish one: still referenced by the other
    qsql.claim(conn, "w1")
    for t in (t1, t2):
        if task_row(conn, t)["status"] == RUNNING:
            qsql.finish(conn, t, "w1", SUCCESS)
    assert vacuum_shared_payloads(conn) == 0
    # finish the second: blob is garbage now
    qsql.claim(conn, "w1")
    for t in (t1, t2):
        if task_row(conn, t)["status"] == RUNNING:
            qsql.finish(conn, t, "w1", SUCCESS)
    assert vacuum_shared_payloads(conn) == 1
    assert conn.execute("SELECT COUNT(*) AS n FROM shared_payload"
                        ).fetchone()["n"] == 0


def test_reclaimed_job_keeps_its_shared_blob(db):
    conn, _ = db
    blob = b"payload" * 100
    tid = enqueue(conn, "noop", shared_blob=blob)
    qsql.claim(conn, "w1", lease_seconds=0.05)
    time.sleep(0.12)
    qsql.reclaim_orphans(conn)
    assert vacuum_shared_payloads(conn) == 0     # PENDING again: referenced
    assert get_shared_blob(conn, tid) == blob


# -- priority / ordering under churn ------------------------------------------

def test_priority_and_fifo_order_survive_reclaims(db):
    conn, _ = db
    low = enqueue(conn, "noop", priority=0)
    high = enqueue(conn, "noop", priority=5)
    mid = enqueue(conn, "noop", priority=3)
    # claim all, let them orphan, reclaim, and re-claim: order preserved
    for _ in range(3):
        qsql.claim(conn, "w1", lease_seconds=0.01)
    time.sleep(0.1)
    qsql.reclaim_orphans(conn)
    order = [qsql.claim(conn, "w2")["task_id"] for _ in range(3)]
    assert order == [high, mid, low]


def test_worker_loop_recovers_after_handler_crash(db):
    conn, url = db

    calls = []

    @task_handler("crashy")
    def crashy(ctx, payload):   # noqa: ANN001
        calls.append(ctx.task_id)
        raise RuntimeError("boom")

    @task_handler("steady")
    def steady(ctx, payload):   # noqa: ANN001
        return {"ok": True}

    t1 = enqueue(conn, "crashy")
    t2 = enqueue(conn, "steady")
    Worker(db_url=url, max_jobs=2).run_forever(idle_timeout=2.0)
    assert task_row(conn, t1)["status"] == FAILURE
    assert "boom" in (task_row(conn, t1)["result"] or "")
    assert task_row(conn, t2)["status"] == SUCCESS
