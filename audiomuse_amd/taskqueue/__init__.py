"""Task queue over SQL storage.

Re-expression of the reference's Postgres queue
(/root/reference/taskqueue/: claim via FOR UPDATE SKIP LOCKED, advisory
-lock liveness, LISTEN/NOTIFY wakeups, orphan reclaim, cooperative
cancel) on the SQLite backend:

- claim: single atomic write transaction (sql.claim)
- liveness: worker heartbeats extend `lease_expires`; a dead worker's
  lease expires and maintenance requeues the job up to max_attempts
  (reference semantics: advisory lock freed by connection death,
  maintenance.py:177)
- wakeups: bounded polling at QUEUE_POLL_SECONDS (LISTEN/NOTIFY has no
  SQLite analog; the poll interval bounds added latency)
- cancel: recursive REVOKED marking; running tasks observe it via
  TaskContext.cancelled() polls (reference: app_helper.py:478 +
  docs/ALGORITHM.md:193-197)
"""

from __future__ import annotations

import hashlib
import json
import sqlite3
import uuid
from typing import Any, Dict, Optional

from audiomuse_amd import config as C
from audiomuse_amd.db import CHAN_CANCEL, CHAN_JOB, notify, write_txn

QUEUE_HIGH = "high"
QUEUE_DEFAULT = "default"

PENDING = "PENDING"
RUNNING = "RUNNING"
SUCCESS = "SUCCESS"
FAILURE = "FAILURE"
REVOKED = "REVOKED"

TERMINAL = (SUCCESS, FAILURE, REVOKED)


def enqueue(conn: sqlite3.Connection, task_type: str,
            payload: Optional[Dict[str, Any]] = None, *,
            queue: str = QUEUE_DEFAULT, parent_task_id: Optional[str] = None,
            priority: int = 0, task_id: Optional[str] = None,
            max_attempts: Optional[int] = None,
            shared_blob: Optional[bytes] = None) -> str:
    """Insert a PENDING job (reference: taskqueue/__init__.py:159).
    Large artifacts ride the deduplicated shared_payload table via a
    content token (reference: sql.py:652-705)."""
    task_id = task_id or uuid.uuid4().hex
    token = None
    with write_txn(conn):
        if shared_blob is not None:
            token = hashlib.sha256(shared_blob).hexdigest()[:32]
            conn.execute(
                """INSERT INTO shared_payload (token, payload, refcount)
                   VALUES (?,?,1)
                   ON CONFLICT(token)
                   DO UPDATE SET refcount = refcount + 1""",
                (token, shared_blob))
        conn.execute(
            """INSERT INTO task_status (task_id, task_type, parent_task_id,
                   queue, status, priority, payload, shared_token,
                   max_attempts)
               VALUES (?,?,?,?,?,?,?,?,?)""",
            (task_id, task_type, parent_task_id, queue, PENDING, priority,
             json.dumps(payload or {}), token,
             max_attempts if max_attempts is not None else C.QUEUE_MAX_ATTEMPTS))
    # wake idle workers (PG NOTIFY, reference sql.py:399; no-op on SQLite
    # where workers poll)
    notify(conn, CHAN_JOB, task_id)
    return task_id


def get_shared_blob(conn: sqlite3.Connection, task_id: str) -> Optional[bytes]:
    row = conn.execute(
        """SELECT p.payload FROM task_status t
           JOIN shared_payload p ON p.token = t.shared_token
           WHERE t.task_id = ?""", (task_id,)).fetchone()
    return bytes(row["payload"]) if row else None


def vacuum_shared_payloads(conn: sqlite3.Connection) -> int:
    """Drop blobs no live (non-terminal) task references (reference:
    maintenance.py blob VACUUM; test_taskqueue_blob_reclaim)."""
    with write_txn(conn):
        cur = conn.execute(
            """DELETE FROM shared_payload WHERE token NOT IN (
                   SELECT shared_token FROM task_status
                   WHERE shared_token IS NOT NULL
                     AND status IN (?, ?))""", (PENDING, RUNNING))
    return cur.rowcount


def task_row(conn: sqlite3.Connection, task_id: str) -> Optional[sqlite3.Row]:
    return conn.execute("SELECT * FROM task_status WHERE task_id=?",
                        (task_id,)).fetchone()


def cancel_task_recursive(conn: sqlite3.Connection, task_id: str) -> int:
    """Mark a task and all descendants REVOKED (reference:
    app_helper.py:478). Running tasks observe the status cooperatively."""
    n = 0
    frontier = [task_id]
    with write_txn(conn):
        while frontier:
            tid = frontier.pop()
            cur = conn.execute(
                "UPDATE task_status SET status=?, finished_at=(julianday('now') - 2440587.5) * 86400.0 "
                "WHERE task_id=? AND status NOT IN (?,?)",
                (REVOKED, tid, SUCCESS, FAILURE))
            n += cur.rowcount
            kids = conn.execute(
                "SELECT task_id FROM task_status WHERE parent_task_id=?",
                (tid,)).fetchall()
            frontier.extend(k["task_id"] for k in kids)
    notify(conn, CHAN_CANCEL, task_id)
    return n
