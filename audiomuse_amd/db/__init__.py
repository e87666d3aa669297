"""SQL storage layer.

The reference uses PostgreSQL as both datastore and queue
(/root/reference/database.py, taskqueue/sql.py). This build ships an
SQLite backend (in-image; psycopg2 is not available here) that preserves
the schema layout and the queue's observable semantics:

- atomic claim (BEGIN IMMEDIATE write transaction ~ FOR UPDATE SKIP LOCKED)
- liveness by lease + heartbeat (~ advisory lock dying with the session:
  worker death => lease expiry => orphan reclaim)
- cooperative cancel via task_status rows
- segmented blob storage for index artifacts

A postgresql:// DATABASE_URL raises until a psycopg backend is added; the
API surface is identical so it can drop in.
"""

from __future__ import annotations

import os
import sqlite3
import threading
from contextlib import contextmanager
from typing import Iterator, Optional

from audiomuse_amd import config as C

_LOCAL = threading.local()


def _sqlite_path(url: str) -> str:
    assert url.startswith("sqlite:///"), f"unsupported DATABASE_URL {url!r}"
    return url[len("sqlite:///"):]


def connect(url: Optional[str] = None) -> sqlite3.Connection:
    """New connection (per thread/process). WAL mode for multi-process use."""
    url = url or C.DATABASE_URL
    if url.startswith("postgresql"):
        raise NotImplementedError(
            "postgresql backend requires psycopg (not present in this image); "
            "use sqlite:///path")
    path = _sqlite_path(url)
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    conn = sqlite3.connect(path, timeout=30.0, isolation_level=None)
    conn.row_factory = sqlite3.Row
    conn.execute("PRAGMA journal_mode=WAL")
    conn.execute("PRAGMA synchronous=NORMAL")
    conn.execute("PRAGMA foreign_keys=ON")
    return conn


def get_db(url: Optional[str] = None) -> sqlite3.Connection:
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
def write_txn(conn: sqlite3.Connection) -> Iterator[sqlite3.Connection]:
    """Exclusive write transaction — the SQLite analog of the reference's
    row-locked claim sections (taskqueue/sql.py:415-430). Reentrant: a
    nested call joins the enclosing transaction (the outermost owns
    commit/rollback), so callers can batch many save_* helpers into one
    fsync."""
    if conn.in_transaction:
        yield conn
        return
    conn.execute("BEGIN IMMEDIATE")
    try:
        yield conn
        conn.execute("COMMIT")
    except BaseException:
        conn.execute("ROLLBACK")
        raise
