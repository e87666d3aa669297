"""Auth: password users, signed-cookie sessions, API token.

Reference: /root/reference/app_auth.py (1337 LoC) — Argon2 password
hashes, JWT cookie sessions, an API token for plugins, and a setup/auth
barrier on every request. argon2-cffi is not in this image, so password
hashing uses stdlib scrypt (same role); session tokens are HMAC-signed
(stdlib) with expiry.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import os
import secrets
import sqlite3
import time
from functools import wraps
from typing import Optional

from flask import current_app, g, jsonify, request

from audiomuse_amd import config as C
from audiomuse_amd.db import write_txn

_SCRYPT_N, _SCRYPT_R, _SCRYPT_P = 2 ** 14, 8, 1
SESSION_COOKIE = "audiomuse_session"
SESSION_TTL = 7 * 86400


def hash_password(password: str) -> str:
    salt = secrets.token_bytes(16)
    dk = hashlib.scrypt(password.encode(), salt=salt, n=_SCRYPT_N,
                        r=_SCRYPT_R, p=_SCRYPT_P, dklen=32)
    return f"scrypt${base64.b64encode(salt).decode()}${base64.b64encode(dk).decode()}"


def verify_password(password: str, stored: str) -> bool:
    try:
        _scheme, salt_b64, dk_b64 = stored.split("$")
        salt = base64.b64decode(salt_b64)
        expect = base64.b64decode(dk_b64)
        dk = hashlib.scrypt(password.encode(), salt=salt, n=_SCRYPT_N,
                            r=_SCRYPT_R, p=_SCRYPT_P, dklen=32)
        return hmac.compare_digest(dk, expect)
    except Exception:
        return False


def _secret() -> bytes:
    """Session-signing secret. Priority: env/config > cached > persisted.

    A generated secret is persisted to app_config on first boot (key
    ``_jwt_secret``, underscore = internal, never surfaced via
    /api/config) so sessions survive restarts and every WSGI worker
    signs/verifies with the same key (a per-process random
    secret broke multi-process deployments).
    """
    s = C.JWT_SECRET or current_app.config.get("JWT_SECRET", "")
    if not s:
        from audiomuse_amd.db.store import get_app_config
        conn = current_app.extensions["audiomuse"].conn()
        s = get_app_config(conn).get("_jwt_secret", "")
        if not s:
            with write_txn(conn):
                conn.execute(
                    "INSERT INTO app_config (key, value) VALUES (?,?) "
                    "ON CONFLICT(key) DO NOTHING",
                    ("_jwt_secret", secrets.token_hex(32)))
            s = get_app_config(conn)["_jwt_secret"]  # racing boots converge
            import logging
            logging.getLogger(__name__).warning(
                "AUDIOMUSE_JWT_SECRET unset; generated one and persisted "
                "it to app_config (set the env var to rotate)")
        current_app.config["JWT_SECRET"] = s
    return s.encode()


def make_session_token(username: str, role: str = "admin") -> str:
    payload = {"u": username, "r": role, "exp": time.time() + SESSION_TTL}
    body = base64.urlsafe_b64encode(json.dumps(payload).encode()).decode()
    sig = hmac.new(_secret(), body.encode(), hashlib.sha256).hexdigest()
    return f"{body}.{sig}"


def verify_session_token(token: str) -> Optional[dict]:
    try:
        body, sig = token.rsplit(".", 1)
        expect = hmac.new(_secret(), body.encode(), hashlib.sha256).hexdigest()
        if not hmac.compare_digest(sig, expect):
            return None
        payload = json.loads(base64.urlsafe_b64decode(body))
        if payload.get("exp", 0) < time.time():
            return None
        return payload
    except Exception:
        return None


def seed_admin_from_env(conn: sqlite3.Connection) -> None:
    """reference: app_auth.seed_admin_from_env :417 — reference env
    names AUDIOMUSE_USER/AUDIOMUSE_PASSWORD (PARAMETERS.md), with the
    ADMIN_-prefixed variants kept as aliases."""
    user = os.environ.get("AUDIOMUSE_ADMIN_USER") or C.AUDIOMUSE_USER
    pw = os.environ.get("AUDIOMUSE_ADMIN_PASSWORD") or C.AUDIOMUSE_PASSWORD
    if not user or not pw:
        return
    with write_txn(conn):
        conn.execute(
            """INSERT INTO audiomuse_users (username, password_hash, role)
               VALUES (?,?, 'admin')
               ON CONFLICT(username) DO NOTHING""",
            (user, hash_password(pw)))


def check_setup_needed(conn: sqlite3.Connection) -> bool:
    """reference: app_auth.check_setup_needed :547 — no users => setup."""
    row = conn.execute("SELECT COUNT(*) AS n FROM audiomuse_users").fetchone()
    return row["n"] == 0


def current_user() -> Optional[dict]:
    token = request.cookies.get(SESSION_COOKIE)
    if token:
        payload = verify_session_token(token)
        if payload:
            return payload
    api_token = request.headers.get("X-API-Token")
    if api_token and C.API_TOKEN and hmac.compare_digest(api_token, C.API_TOKEN):
        return {"u": "api", "r": "api"}
    return None


def require_auth(fn):
    @wraps(fn)
    def wrapper(*args, **kwargs):
        if current_app.config.get("AUTH_DISABLED"):
            g.user = {"u": "anonymous", "r": "admin"}
            return fn(*args, **kwargs)
        conn = current_app.extensions["audiomuse"].conn()
        if check_setup_needed(conn):
            return jsonify({"error": "setup required", "setup": True}), 403
        user = current_user()
        if user is None:
            return jsonify({"error": "authentication required"}), 401
        g.user = user
        return fn(*args, **kwargs)
    return wrapper
