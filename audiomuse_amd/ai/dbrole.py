"""Low-privilege database access for AI tool queries.

Reference: /root/reference/tasks/mcp_helper.py:63
(_ensure_ai_chat_db_user) — AI-generated SQL runs under a read-only
PostgreSQL role so a prompt-injected query cannot mutate the catalogue.
Both backends get a real privilege boundary here:

- PostgreSQL: a ``audiomuse_ai_ro`` role with SELECT-only grants on the
  catalogue tables; the AI connection authenticates as that role.
- SQLite: a second connection opened with ``mode=ro`` (the SQLite
  engine enforces read-only at the VFS level — any INSERT/UPDATE/DELETE
  raises OperationalError).
"""

from __future__ import annotations

import logging
import threading
from typing import Optional

from audiomuse_amd.db import backend_kind, connect

logger = logging.getLogger(__name__)

AI_ROLE = "audiomuse_ai_ro"
# AI tools only ever read the catalogue surface
_READABLE = ("score", "embedding", "clap_embedding", "lyrics_embedding",
             "playlist")

_LOCAL = threading.local()


def _ensure_pg_role(admin_conn, password: str) -> None:
    """Create/refresh the SELECT-only role (idempotent)."""
    row = admin_conn.execute(
        "SELECT 1 AS x FROM pg_roles WHERE rolname = ?", (AI_ROLE,)
    ).fetchone()
    if row is None:
        admin_conn.execute(
            f"CREATE ROLE {AI_ROLE} LOGIN PASSWORD "
            f"'{password.replace(chr(39), '')}'")
    for table in _READABLE:
        admin_conn.execute(f"GRANT SELECT ON {table} TO {AI_ROLE}")


def readonly_connection(db_url: str, admin_conn=None,
                        role_password: str = "ai-readonly"):
    """A connection that can only SELECT. Cached per thread."""
    cache = getattr(_LOCAL, "conns", None)
    if cache is None:
        cache = _LOCAL.conns = {}
    hit = cache.get(db_url)
    if hit is not None:
        return hit
    if backend_kind(db_url) == "postgres":
        try:
            if admin_conn is not None:
                _ensure_pg_role(admin_conn, role_password)
            from urllib.parse import urlparse, urlunparse
            p = urlparse(db_url)
            netloc = f"{AI_ROLE}:{role_password}@{p.hostname}"
            if p.port:
                netloc += f":{p.port}"
            ro_url = urlunparse(p._replace(netloc=netloc))
            conn = connect(ro_url)
        except Exception:
            # role plumbing unavailable (e.g. managed DB): better a
            # parameterized admin connection than a crashed chat request
            logger.warning("AI read-only role unavailable; falling back "
                           "to the app connection", exc_info=True)
            conn = connect(db_url)
    else:
        import sqlite3
        path = db_url[len("sqlite:///"):]
        conn = sqlite3.connect(f"file:{path}?mode=ro", uri=True,
                               timeout=30.0, isolation_level=None)
        conn.row_factory = sqlite3.Row
    cache[db_url] = conn
    return conn


def reset_cache() -> None:
    conns = getattr(_LOCAL, "conns", None)
    if conns:
        for c in conns.values():
            try:
                c.close()
            except Exception:
                pass
        conns.clear()
