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


def minute_key(t: Optional[float] = None) -> str:
    lt = time.localtime(t if t is not None else time.time())
    return time.strftime("%Y-%m-%dT%H:%M", lt)


def run_due_cron_jobs(conn: sqlite3.Connection,
                      now: Optional[float] = None) -> List[str]:
    """Claim-and-enqueue every due cron row for the current minute.
    The UPDATE ... WHERE last_claimed_minute IS DISTINCT FROM ? is the
    single-winner claim (reference: advisory-lock minute claim)."""
    mk = minute_key(now)
    enqueued: List[str] = []
    rows = conn.execute(
        "SELECT id, schedule, task_type, payload FROM cron WHERE enabled=1"
    ).fetchall()
    for row in rows:
        if not cron_matches(row["schedule"], now):
            continue
        with write_txn(conn):
            cur = conn.execute(
                """UPDATE cron SET last_claimed_minute=?
                   WHERE id=? AND (last_claimed_minute IS NULL
                                   OR last_claimed_minute != ?)""",
                (mk, row["id"], mk))
            claimed = cur.rowcount == 1
        if claimed:
            payload = json.loads(row["payload"] or "{}")
            enqueued.append(enqueue(conn, row["task_type"], payload))
    return enqueued


def cron_loop(conn: sqlite3.Connection, stop_event,
              poll_seconds: float = 20.0) -> None:
    """Background scheduler thread (reference: app.py cron loop)."""
    while not stop_event.wait(poll_seconds):
        try:
            run_due_cron_jobs(conn)
        except Exception:  # noqa: BLE001
            import logging

            logging.getLogger(__name__).exception("cron tick failed")
