"""Queue SQL operations (reference: taskqueue/sql.py)."""

from __future__ import annotations

import json
import sqlite3
import time
from typing import Optional, Sequence

from audiomuse_amd import config as C
from audiomuse_amd.db import write_txn
from audiomuse_amd.taskqueue import (FAILURE, PENDING, REVOKED, RUNNING,
                                     SUCCESS)


def claim(conn: sqlite3.Connection, worker_id: str,
          queues: Sequence[str] = ("high", "default"),
          lease_seconds: Optional[float] = None) -> Optional[sqlite3.Row]:
    """Atomically claim the next PENDING job (reference: sql.py:430,
    FOR UPDATE SKIP LOCKED; here one exclusive write txn)."""
    lease = lease_seconds if lease_seconds is not None else C.QUEUE_LEASE_SECONDS
    now = time.time()
    qmarks = ",".join("?" for _ in queues)
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


def reclaim_orphans(conn: sqlite3.Connection) -> int:
    """Maintenance pass (reference: maintenance.py:177 + sql.py:579-641):
    RUNNING + expired lease => re-PENDING while attempts remain, else
    FAILURE for good."""
    now = time.time()
    n = 0
    with write_txn(conn):
        rows = conn.execute(
            """SELECT task_id, attempts, max_attempts FROM task_status
               WHERE status=? AND lease_expires IS NOT NULL AND lease_expires < ?""",
            (RUNNING, now)).fetchall()
        for row in rows:
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
