"""SQL storage layer — SQLite and PostgreSQL backends.

The reference uses PostgreSQL as both datastore and queue
(/root/reference/database.py, taskqueue/sql.py). This build ships both:

- **PostgreSQL** (``postgresql://user:pass@host/db``) — the declared
  deployment contract. Spoken through the first-party wire-protocol
  driver in :mod:`audiomuse_amd.db.pgwire` (no psycopg wheel exists in
  this image — docs/POSTGRES.md records the attempted installs). The
  queue layer upgrades itself on this backend: ``FOR UPDATE SKIP
  LOCKED`` claims, advisory-lock liveness, and LISTEN/NOTIFY wake
  channels, matching the reference's semantics
  (taskqueue/sql.py:415-462, :48-52).
- **SQLite** (``sqlite:///path``) — the zero-dependency single-box
  fallback preserving the same schema and the queue's
  observable semantics: exclusive-write-txn claims, lease-heartbeat
  liveness, polling wake.

Both expose the same connection surface (execute -> cursor with
fetchone/fetchall/rowcount, executescript, in_transaction), so every
module above this one is backend-agnostic.
"""

from __future__ import annotations

import os
import sqlite3
import threading
import time
from contextlib import contextmanager
from typing import Iterator, List, Optional, Tuple

from audiomuse_amd import config as C

_LOCAL = threading.local()

# LISTEN/NOTIFY channels (reference: taskqueue/sql.py:48-52)
CHAN_JOB = "audiomuse_job"
CHAN_CANCEL = "audiomuse_cancel"
CHAN_EVENT = "audiomuse_event"
CHAN_CONTROL = "audiomuse_control"
CHAN_RECLAIM = "audiomuse_reclaim"


def backend_kind(url_or_conn) -> str:
    if isinstance(url_or_conn, str):
        return "postgres" if url_or_conn.startswith("postgres") else "sqlite"
    return getattr(url_or_conn, "kind", "sqlite")


def _sqlite_path(url: str) -> str:
    assert url.startswith("sqlite:///"), f"unsupported DATABASE_URL {url!r}"
    return url[len("sqlite:///"):]


def connect(url: Optional[str] = None):
    """New connection (per thread/process)."""
    url = url or C.DATABASE_URL
    if url.startswith("postgres"):
        from audiomuse_amd.db import pgwire
        return pgwire.connect_url(url)
    path = _sqlite_path(url)
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    conn = sqlite3.connect(path, timeout=30.0, isolation_level=None)
    conn.row_factory = sqlite3.Row
    conn.execute("PRAGMA journal_mode=WAL")
    conn.execute("PRAGMA synchronous=NORMAL")
    conn.execute("PRAGMA foreign_keys=ON")
    return conn


def get_db(url: Optional[str] = None):
    """Thread-cached connection (reference: database.get_db)."""
    url = url or C.DATABASE_URL
    cache = getattr(_LOCAL, "conns", None)
    if cache is None:
        cache = _LOCAL.conns = {}
    key = (url, os.getpid())
    conn = cache.get(key)
    if conn is None:
        conn = cache[key] = connect(url)
        from audiomuse_amd.db.schema import init_db
        init_db(conn)
    return conn


@contextmanager
def write_txn(conn) -> Iterator:
    """Exclusive write transaction — SQLite: BEGIN IMMEDIATE (the analog
    of the reference's row-locked claim sections, taskqueue/sql.py:
    415-430); PostgreSQL: a plain transaction whose row claims use FOR
    UPDATE SKIP LOCKED at the statement level. Reentrant: a nested call
    joins the enclosing transaction (the outermost owns
    commit/rollback), so callers can batch many save_* helpers into one
    fsync."""
    if conn.in_transaction:
        yield conn
        return
    conn.execute("BEGIN IMMEDIATE")  # pgwire translates to BEGIN
    try:
        yield conn
        conn.execute("COMMIT")
    except BaseException:
        conn.execute("ROLLBACK")
        raise


def insert_returning_id(conn, sql: str, params=()) -> int:
    """INSERT into a table with a serial ``id`` and return it portably
    (PG: RETURNING id; SQLite: lastrowid)."""
    if backend_kind(conn) == "postgres":
        row = conn.execute(sql + " RETURNING id", params).fetchone()
        return int(row["id"])
    return int(conn.execute(sql, params).lastrowid)


# -- pub/sub (PG: LISTEN/NOTIFY; SQLite: callers poll) -----------------------

def listen(conn, channel: str) -> None:
    if backend_kind(conn) == "postgres":
        conn.listen(channel)


def notify(conn, channel: str, payload: str = "") -> None:
    if backend_kind(conn) == "postgres":
        conn.notify(channel, payload)


def wait_notify(conn, timeout: float) -> List[Tuple[int, str, str]]:
    """Block up to ``timeout`` for notifications. On SQLite there is no
    pub-sub; sleep the interval so worker loops keep their cadence."""
    if backend_kind(conn) == "postgres":
        return conn.wait_notify(timeout)
    time.sleep(timeout)
    return []
