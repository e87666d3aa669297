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

import json
import sqlite3
import uuid
from typing import Any, Dict, Optional

from audiomuse_amd import config as C
from audiomuse_amd.db import write_txn

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
            max_attempts: Optional[int] = None) -> str:
    """Insert a PENDING job (reference: taskqueue/__init__.py:159)."""
    task_id = task_id or uuid.uuid4().hex
    with write_txn(conn):
        conn.execute(
            """INSERT INTO task_status (task_id, task_type, parent_task_id,
                   queue, status, priority, payload, max_attempts)
               VALUES (?,?,?,?,?,?,?,?)""",
            (task_id, task_type, parent_task_id, queue, PENDING, priority,
             json.dumps(payload or {}),
             max_attempts if max_attempts is not None else C.QUEUE_MAX_ATTEMPTS))
    return task_id


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
    return n
