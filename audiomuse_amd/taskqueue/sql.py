"""Queue SQL operations (reference: taskqueue/sql.py).

Liveness model by backend:

- PostgreSQL: the claim is ``FOR UPDATE SKIP LOCKED`` (reference
  sql.py:415-430) and the worker connection takes a per-task advisory
  lock (sql.py:452-462) that dies with the connection — the primary
  orphan signal. Lease expiry remains as a belt-and-braces fallback for
  a wedged-but-connected worker.
- SQLite: the claim is one exclusive write transaction; lease +
  heartbeat is the only liveness signal.
"""

from __future__ import annotations

import json
import sqlite3
import time
from typing import Optional, Sequence

from audiomuse_amd import config as C
from audiomuse_amd.db import backend_kind, write_txn
from audiomuse_amd.taskqueue import (FAILURE, PENDING, REVOKED, RUNNING,
                                     SUCCESS)

# advisory-lock key namespace: (class, hashtext(task_id)); class 1=task,
# 2=maintenance election, 3=admission gate, 4=cron minute
LOCK_CLASS_TASK = 1
LOCK_CLASS_MAINT = 2
LOCK_CLASS_ADMIT = 3
LOCK_CLASS_CRON = 4


def try_advisory_lock(conn, lock_class: int, key: str) -> bool:
    """PG: session advisory lock (dies with the connection). SQLite:
    vacuously True — liveness is lease-based there. Held keys are
    tracked on the connection because PG advisory locks are
    session-reentrant: a liveness probe must know which locks ITS OWN
    session already owns (those tasks are trivially alive)."""
    if backend_kind(conn) != "postgres":
        return True
    row = conn.execute(
        "SELECT pg_try_advisory_lock(?, hashtext(?)) AS got",
        (lock_class, key)).fetchone()
    if bool(row["got"]):
        held = getattr(conn, "_held_advisory", None)
        if held is None:
            held = conn._held_advisory = set()
        held.add((lock_class, key))
        return True
    return False


def holds_advisory_lock(conn, lock_class: int, key: str) -> bool:
    return (lock_class, key) in getattr(conn, "_held_advisory", ())


def advisory_unlock(conn, lock_class: int, key: str) -> None:
    if backend_kind(conn) != "postgres":
        return
    conn.execute("SELECT pg_advisory_unlock(?, hashtext(?))",
                 (lock_class, key))
    getattr(conn, "_held_advisory", set()).discard((lock_class, key))


def claim(conn, worker_id: str,
          queues: Sequence[str] = ("high", "default"),
          lease_seconds: Optional[float] = None) -> Optional[sqlite3.Row]:
    """Atomically claim the next PENDING job.

    PG: FOR UPDATE SKIP LOCKED subselect + UPDATE ... RETURNING, then the
    advisory liveness lock (reference: sql.py:430,452). SQLite: one
    exclusive write txn.
    """
    lease = lease_seconds if lease_seconds is not None else C.QUEUE_LEASE_SECONDS
    now = time.time()
    qmarks = ",".join("?" for _ in queues)
    if backend_kind(conn) == "postgres":
        with write_txn(conn):
            row = conn.execute(
                f"""UPDATE task_status SET status=?, worker_id=?,
                        lease_expires=?, started_at=?, attempts=attempts+1
                    WHERE task_id = (
                        SELECT task_id FROM task_status
                        WHERE status = ? AND queue IN ({qmarks})
                        ORDER BY priority DESC, created_at
                        FOR UPDATE SKIP LOCKED LIMIT 1)
                    RETURNING task_id""",
                (RUNNING, worker_id, now + lease, now, PENDING,
                 *queues)).fetchone()
        if row is None:
            return None
        try_advisory_lock(conn, LOCK_CLASS_TASK, row["task_id"])
    else:
        with write_txn(conn):
            row = conn.execute(
                f"""SELECT task_id FROM task_status
                    WHERE status = ? AND queue IN ({qmarks})
                    ORDER BY priority DESC, created_at LIMIT 1""",
                (PENDING, *queues)).fetchone()
            if row is None:
                return None
            conn.execute(
                """UPDATE task_status SET status=?, worker_id=?,
                       lease_expires=?, started_at=?, attempts=attempts+1
                   WHERE task_id=?""",
                (RUNNING, worker_id, now + lease, now, row["task_id"]))
    return conn.execute("SELECT * FROM task_status WHERE task_id=?",
                        (row["task_id"],)).fetchone()


def heartbeat(conn: sqlite3.Connection, task_id: str, worker_id: str,
              lease_seconds: Optional[float] = None) -> bool:
    """Extend the lease (the liveness signal; analog of the advisory lock
    staying held, sql.py:452). Returns False if the job was revoked or
    reclaimed from under us."""
    lease = lease_seconds if lease_seconds is not None else C.QUEUE_LEASE_SECONDS
    with write_txn(conn):
        cur = conn.execute(
            """UPDATE task_status SET lease_expires=?
               WHERE task_id=? AND worker_id=? AND status=?""",
            (time.time() + lease, task_id, worker_id, RUNNING))
    return cur.rowcount == 1


def finish(conn: sqlite3.Connection, task_id: str, worker_id: str,
           status: str, result: Optional[dict] = None,
           error_code: Optional[int] = None) -> bool:
    assert status in (SUCCESS, FAILURE)
    with write_txn(conn):
        cur = conn.execute(
            """UPDATE task_status SET status=?, result=?, error_code=?,
                   finished_at=(julianday('now') - 2440587.5) * 86400.0
               WHERE task_id=? AND worker_id=? AND status=?""",
            (status, json.dumps(result or {}), error_code, task_id, worker_id,
             RUNNING))
        if cur.rowcount == 1:
            # task history trail (reference: record_task_history :223)
            row = conn.execute(
                "SELECT task_type FROM task_status WHERE task_id=?",
                (task_id,)).fetchone()
            conn.execute(
                "INSERT INTO task_history (task_id, task_type, status, note) "
                "VALUES (?,?,?,?)",
                (task_id, row["task_type"] if row else "", status,
                 (json.dumps(result)[:500] if result else None)))
    if cur.rowcount == 1:
        advisory_unlock(conn, LOCK_CLASS_TASK, task_id)
    return cur.rowcount == 1


def update_progress(conn: sqlite3.Connection, task_id: str, progress: float,
                    details: Optional[str] = None) -> None:
    with write_txn(conn):
        if details is None:
            conn.execute("UPDATE task_status SET progress=? WHERE task_id=?",
                         (progress, task_id))
        else:
            conn.execute(
                "UPDATE task_status SET progress=?, details=? WHERE task_id=?",
                (progress, details, task_id))


def is_cancelled(conn: sqlite3.Connection, task_id: str) -> bool:
    """Cooperative-cancel check: own row or any ancestor REVOKED/missing
    (reference: docs/ALGORITHM.md:193-197)."""
    tid: Optional[str] = task_id
    seen = set()
    while tid and tid not in seen:
        seen.add(tid)
        row = conn.execute(
            "SELECT status, parent_task_id FROM task_status WHERE task_id=?",
            (tid,)).fetchone()
        if row is None or row["status"] == REVOKED:
            return True
        tid = row["parent_task_id"]
    return False


def _orphaned_running_rows(conn, now: float) -> list:
    """RUNNING rows whose worker is provably gone.

    SQLite: lease expired. PG: lease expired OR the per-task advisory
    lock is free (the owner's connection died — reference
    maintenance.py:177: orphan = RUNNING and lock acquirable). A lock
    probed free is released immediately; the row decides the outcome.
    """
    rows = conn.execute(
        """SELECT task_id, attempts, max_attempts, lease_expires
           FROM task_status WHERE status=?""", (RUNNING,)).fetchall()
    out = []
    is_pg = backend_kind(conn) == "postgres"
    for row in rows:
        expired = (row["lease_expires"] is not None
                   and row["lease_expires"] < now)
        if expired:
            out.append(row)
        elif (is_pg
              and not holds_advisory_lock(conn, LOCK_CLASS_TASK,
                                          row["task_id"])  # ours = alive
              and try_advisory_lock(conn, LOCK_CLASS_TASK,
                                    row["task_id"])):
            advisory_unlock(conn, LOCK_CLASS_TASK, row["task_id"])
            out.append(row)
    return out


def reclaim_orphans(conn) -> int:
    """Maintenance pass (reference: maintenance.py:177 + sql.py:579-641):
    orphaned RUNNING rows => re-PENDING while attempts remain, else
    FAILURE for good. Only one node runs this at a time on PG (the
    maintenance-election advisory lock, reference sql.py:472)."""
    now = time.time()
    n = 0
    if not try_advisory_lock(conn, LOCK_CLASS_MAINT, "maintenance"):
        return 0
    try:
        orphans = _orphaned_running_rows(conn, now)
    finally:
        advisory_unlock(conn, LOCK_CLASS_MAINT, "maintenance")
    with write_txn(conn):
        for row in orphans:
            if row["attempts"] >= row["max_attempts"]:
                conn.execute(
                    """UPDATE task_status SET status=?, worker_id=NULL,
                           finished_at=(julianday('now') - 2440587.5) * 86400.0,
                           details='exceeded max attempts (orphaned)'
                       WHERE task_id=? AND status=?""",
                    (FAILURE, row["task_id"], RUNNING))
            else:
                conn.execute(
                    """UPDATE task_status SET status=?, worker_id=NULL,
                           lease_expires=NULL WHERE task_id=? AND status=?""",
                    (PENDING, row["task_id"], RUNNING))
            n += 1
    return n


def counts_by_status(conn: sqlite3.Connection) -> dict:
    return {r["status"]: r["n"] for r in conn.execute(
        "SELECT status, COUNT(*) AS n FROM task_status GROUP BY status")}


def pending_children(conn: sqlite3.Connection, parent_task_id: str) -> int:
    row = conn.execute(
        "SELECT COUNT(*) AS n FROM task_status WHERE parent_task_id=? "
        "AND status IN (?, ?)", (parent_task_id, PENDING, RUNNING)).fetchone()
    return int(row["n"])
