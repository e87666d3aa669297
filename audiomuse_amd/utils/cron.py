"""Cron scheduler: 5-field expressions + minute-claimed dispatch.

Reference: /root/reference/app_cron.py — cron table CRUD plus a
scheduler loop where exactly one process claims each due minute
(_claim_cron_minute :452 via advisory lock; here an atomic
last_claimed_minute compare-and-set per row) and enqueues the job
(run_due_cron_jobs :879), with a retry list for failures.
"""

from __future__ import annotations

import json
import sqlite3
import time
from typing import List, Optional

from audiomuse_amd import config as C
from audiomuse_amd.db import write_txn
from audiomuse_amd.taskqueue import enqueue


def _field_matches(field: str, value: int, minv: int, maxv: int) -> bool:
    for part in field.split(","):
        part = part.strip()
        step = 1
        if "/" in part:
            part, step_s = part.split("/")
            step = int(step_s)
        if part in ("*", ""):
            lo, hi = minv, maxv
        elif "-" in part:
            lo_s, hi_s = part.split("-")
            lo, hi = int(lo_s), int(hi_s)
        else:
            lo = hi = int(part)
        if lo <= value <= hi and (value - lo) % step == 0:
            return True
    return False


def cron_matches(expr: str, t: Optional[float] = None) -> bool:
    """5-field cron (minute hour dom month dow) against local time."""
    parts = expr.split()
    if len(parts) != 5:
        return False
    lt = time.localtime(t if t is not None else time.time())
    minute, hour, dom, month, dow = parts
    cron_dow = (lt.tm_wday + 1) % 7      # tm: Mon=0..Sun=6; cron: Sun=0
    return (_field_matches(minute, lt.tm_min, 0, 59)
            and _field_matches(hour, lt.tm_hour, 0, 23)
            and _field_matches(dom, lt.tm_mday, 1, 31)
            and _field_matches(month, lt.tm_mon, 1, 12)
            and _field_matches(dow, cron_dow, 0, 6))


def validate_cron(expr: str) -> bool:
    """Reject malformed or can-never-fire expressions before they are
    stored enabled (reference: ALGORITHM.md 16.1 step 3)."""
    parts = expr.split()
    if len(parts) != 5:
        return False
    ranges = [(0, 59), (0, 23), (1, 31), (1, 12), (0, 6)]
    for field, (lo, hi) in zip(parts, ranges):
        try:
            if not any(_field_matches(field, v, lo, hi)
                       for v in range(lo, hi + 1)):
                return False
        except (ValueError, ZeroDivisionError):
            return False
    return True


def minute_key(t: Optional[float] = None) -> str:
    lt = time.localtime(t if t is not None else time.time())
    return time.strftime("%Y-%m-%dT%H:%M", lt)


# task types that must not stack on top of each other (reference:
# ALGORITHM.md 16.2 step 5 — the queue guard)
GUARDED_TYPES = ("run_analysis", "run_clustering", "sonic_fingerprint")


def _is_guarded(task_type: str) -> bool:
    return task_type in GUARDED_TYPES or task_type.startswith("plugin.")


def _guard_blocker(conn: sqlite3.Connection) -> Optional[str]:
    """Another guarded task queued or running -> its type, else None."""
    marks = list(GUARDED_TYPES)
    q = ("SELECT task_type FROM task_status WHERE status IN "
         "('PENDING', 'RUNNING') AND (task_type IN (%s) OR task_type "
         "LIKE 'plugin.%%') LIMIT 1" % ",".join("?" * len(marks)))
    row = conn.execute(q, marks).fetchone()
    return row["task_type"] if row else None


def run_due_cron_jobs(conn: sqlite3.Connection,
                      now: Optional[float] = None) -> List[str]:
    """Claim-and-enqueue every due cron row for the current minute.
    The UPDATE ... WHERE last_claimed_minute IS DISTINCT FROM ? is the
    single-winner claim (reference: advisory-lock minute claim).

    Queue guard + retry (reference: ALGORITHM.md 16.2 steps 5-6): a due
    guarded run that would stack on an active guarded task is parked in
    cron_retry and re-attempted each tick until
    CRON_RETRY_MAX_MINUTES' worth of attempts, then surfaced as a
    FAILURE task row instead of silently dropped."""
    from audiomuse_amd import config as C

    mk = minute_key(now)
    ts = now if now is not None else time.time()
    enqueued: List[str] = []
    rows = conn.execute(
        "SELECT id, schedule, task_type, payload FROM cron WHERE enabled=1"
    ).fetchall()
    due = []
    for row in rows:
        if not cron_matches(row["schedule"], now):
            continue
        with write_txn(conn):
            cur = conn.execute(
                """UPDATE cron SET last_claimed_minute=?
                   WHERE id=? AND (last_claimed_minute IS NULL
                                   OR last_claimed_minute != ?)""",
                (mk, row["id"], mk))
            if cur.rowcount == 1:
                due.append(row)

    # retries whose wait elapsed rejoin the due set
    interval = C.CRON_RETRY_INTERVAL_MINUTES * 60.0
    max_attempts = max(1, int(C.CRON_RETRY_MAX_MINUTES
                              / max(C.CRON_RETRY_INTERVAL_MINUTES, 1e-9)))
    retry_rows = conn.execute(
        "SELECT r.id AS retry_id, r.attempts, c.* FROM cron_retry r "
        "JOIN cron c ON c.id = r.cron_id WHERE r.due_at <= ?", (ts,)
    ).fetchall()
    # a pending retry already carries its schedule: a fresh due claim of
    # the same row must not park a second retry entry
    parked_ids = {r["cron_id"] for r in conn.execute(
        "SELECT cron_id FROM cron_retry").fetchall()}
    due = [r for r in due if r["id"] not in parked_ids]

    for row in list(retry_rows) + list(due):
        retry_id = row["retry_id"] if "retry_id" in row.keys() else None
        attempts = row["attempts"] if retry_id is not None else 0
        blocker = _guard_blocker(conn) if _is_guarded(row["task_type"]) else None
        if blocker is not None:
            with write_txn(conn):
                if attempts + 1 >= max_attempts:
                    # window expired: visible failed run, not a silent drop
                    import uuid as _uuid
                    conn.execute(
                        """INSERT INTO task_status (task_id, task_type,
                               status, details)
                           VALUES (?, ?, 'FAILURE', ?)""",
                        (_uuid.uuid4().hex, row["task_type"],
                         json.dumps({"error": "cron retry window expired",
                                     "blocked_by": blocker,
                                     "cron_id": row["id"]})))
                    if retry_id is not None:
                        conn.execute("DELETE FROM cron_retry WHERE id=?",
                                     (retry_id,))
                elif retry_id is not None:
                    conn.execute(
                        "UPDATE cron_retry SET due_at=?, attempts=attempts+1 "
                        "WHERE id=?", (ts + interval, retry_id))
                else:
                    conn.execute(
                        "INSERT INTO cron_retry (cron_id, due_at, attempts) "
                        "VALUES (?, ?, 1)", (row["id"], ts + interval))
            continue
        with write_txn(conn):
            if retry_id is not None:
                conn.execute("DELETE FROM cron_retry WHERE id=?", (retry_id,))
        payload = json.loads(row["payload"] or "{}")
        enqueued.append(enqueue(conn, row["task_type"], payload))
    return enqueued


def cron_loop(conn: sqlite3.Connection, stop_event,
              poll_seconds: float = 20.0) -> None:
    """Background scheduler thread (reference: app.py cron loop)."""
    while not stop_event.wait(poll_seconds):
        if not C.CRON_ENABLED:
            continue  # scheduler parked (reference CRON kill switch)
        try:
            run_due_cron_jobs(conn)
        except Exception:  # noqa: BLE001
            import logging

            logging.getLogger(__name__).exception("cron tick failed")
