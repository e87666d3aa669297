"""Control plane: restart/stop broadcast with per-listener acks.

Reference: /root/reference/taskqueue/control.py (470 LoC) — a restart or
stop request carries a request id; every listener (worker / web process)
acks it; the requester waits for acks inside a bounded action window,
during which orphan reclaim stands down (maintenance.py:112). Used by
the setup wizard's restart handshake and provider migration.
"""

from __future__ import annotations

import json
import sqlite3
import time
from typing import List, Optional

from audiomuse_amd.db import write_txn

ACTION_STOP_WORKERS = "stop_workers"
ACTION_RESTART = "restart"
def _default_window() -> float:
    from audiomuse_amd import config as C
    return C.CONTROL_WINDOW_SECONDS


DEFAULT_WINDOW_SECONDS = 60.0  # fallback when config is unavailable


def publish_control_request(conn: sqlite3.Connection, action: str,
                            payload: Optional[dict] = None,
                            window_seconds: Optional[float] = None
                            ) -> int:
    """reference: control.publish_control_request :115"""
    if window_seconds is None:
        window_seconds = _default_window()
    from audiomuse_amd.db import insert_returning_id
    with write_txn(conn):
        rid = insert_returning_id(
            conn,
            "INSERT INTO control_request (action, payload, expires_at) "
            "VALUES (?,?,?)",
            (action, json.dumps(payload or {}), time.time() + window_seconds))
    return rid


def pending_requests(conn: sqlite3.Connection, listener: str,
                     actions: Optional[List[str]] = None) -> List[sqlite3.Row]:
    """Unacked, unexpired requests for this listener."""
    rows = conn.execute(
        """SELECT r.* FROM control_request r
           LEFT JOIN control_ack a
               ON a.request_id = r.id AND a.listener = ?
           WHERE a.listener IS NULL AND r.expires_at > ?""",
        (listener, time.time())).fetchall()
    if actions is not None:
        rows = [r for r in rows if r["action"] in actions]
    return rows


def ack(conn: sqlite3.Connection, request_id: int, listener: str) -> None:
    with write_txn(conn):
        conn.execute(
            "INSERT OR IGNORE INTO control_ack (request_id, listener) "
            "VALUES (?,?)", (request_id, listener))


def ack_count(conn: sqlite3.Connection, request_id: int) -> int:
    return int(conn.execute(
        "SELECT COUNT(*) FROM control_ack WHERE request_id=?",
        (request_id,)).fetchone()[0])


def wait_for_acks(conn: sqlite3.Connection, request_id: int, expected: int,
                  timeout: float = 30.0, poll: float = 0.1) -> bool:
    """reference: bounded ack wait (control.py:115-248)."""
    deadline = time.time() + timeout
    while time.time() < deadline:
        if ack_count(conn, request_id) >= expected:
            return True
        time.sleep(poll)
    return False


def control_window_active(conn: sqlite3.Connection) -> bool:
    """True while any restart/stop window is open — orphan reclaim stands
    down during it (reference: maintenance.py:112)."""
    row = conn.execute(
        "SELECT 1 FROM control_request WHERE expires_at > ? LIMIT 1",
        (time.time(),)).fetchone()
    return row is not None
