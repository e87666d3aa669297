"""Auth + setup blueprint (reference: app_auth.py + app_setup.py routes)."""

from __future__ import annotations

from flask import Blueprint, current_app, jsonify, make_response, request

from audiomuse_amd.db import write_txn
from audiomuse_amd.web.auth import (SESSION_COOKIE, check_setup_needed,
                                    hash_password, make_session_token,
                                    require_auth, verify_password)

bp = Blueprint("auth", __name__)


def _state():
    return current_app.extensions["audiomuse"]


@bp.get("/api/setup/status")
def setup_status():
    return jsonify({"setup_needed": check_setup_needed(_state().conn())})


@bp.post("/api/setup/admin")
def setup_admin():
    """First-boot admin creation (reference: setup wizard step)."""
    conn = _state().conn()
    body = request.get_json(force=True, silent=True) or {}
    username = (body.get("username") or "").strip()
    password = body.get("password") or ""
    if not username or len(password) < 8:
        return jsonify({"error": "username and password (>=8 chars) required"}), 400
    # COUNT check and INSERT share one write transaction so two racing
    # first-boot requests cannot both create an admin (ADVICE r1 TOCTOU).
    with write_txn(conn):
        if not check_setup_needed(conn):
            return jsonify({"error": "already configured"}), 409
        conn.execute(
            "INSERT INTO audiomuse_users (username, password_hash, role) "
            "VALUES (?,?, 'admin')", (username, hash_password(password)))
    return jsonify({"created": username})


# -- Plex PIN link flow (reference: app_setup.py:926-1030) ------------------
# plex.tv sends no CORS headers, so the browser cannot call it directly;
# these two routes proxy the PIN create/poll server-side. The
# client_id must stay constant between the two calls.

PLEX_PIN_API_BASE = "https://plex.tv/api/v2/pins"


def _plex_headers(client_id: str) -> dict:
    return {"Accept": "application/json",
            "X-Plex-Product": "AudioMuse-AMD",
            "X-Plex-Client-Identifier": client_id or "audiomuse-amd"}


@bp.post("/api/setup/plex/pin")
def plex_pin_create():
    """Create a plex.tv link PIN: the user enters the returned code at
    plex.tv/link; poll the companion GET until authToken appears."""
    from audiomuse_amd.mediaserver.http import MediaHttp

    body = request.get_json(force=True, silent=True) or {}
    client_id = body.get("client_id", "audiomuse-amd")
    try:
        r = MediaHttp().post(PLEX_PIN_API_BASE, params={"strong": "true"},
                             headers=_plex_headers(client_id))
        data = r.json()
    except Exception as exc:  # no egress / plex.tv unreachable
        return jsonify({"error": f"plex.tv unreachable: {exc}"}), 502
    return jsonify({"id": data.get("id"), "code": data.get("code"),
                    "client_id": client_id})


@bp.get("/api/setup/plex/pin/<pin_id>")
def plex_pin_poll(pin_id: str):
    from audiomuse_amd.mediaserver.http import MediaHttp

    client_id = request.args.get("client_id", "audiomuse-amd")
    try:
        r = MediaHttp().get(f"{PLEX_PIN_API_BASE}/{pin_id}",
                            headers=_plex_headers(client_id))
        data = r.json()
    except Exception as exc:
        return jsonify({"error": f"plex.tv unreachable: {exc}"}), 502
    return jsonify({"id": data.get("id"),
                    "auth_token": data.get("authToken") or None,
                    "claimed": bool(data.get("authToken"))})


@bp.post("/api/setup/providers/libraries")
def setup_provider_libraries():
    """Library list for an UNSAVED provider config during the setup
    wizard (reference: /api/setup/providers/libraries)."""
    from audiomuse_amd.mediaserver import make_provider

    body = request.get_json(force=True, silent=True) or {}
    try:
        provider = make_provider(body.get("server_type", ""),
                                 **(body.get("server_config") or {}))
        libs = getattr(provider, "list_libraries", lambda: [])()
    except Exception as exc:
        return jsonify({"error": str(exc)}), 502
    return jsonify(libs)


@bp.post("/api/setup/lyrics-api/analyze")
@require_auth
def setup_lyrics_api_analyze():
    """Try the configured external lyrics APIs against one artist/title
    and return what came back (reference: /api/setup/lyrics-api/analyze
    — lets the wizard verify LYRICS_API_* settings)."""
    from audiomuse_amd.engines.lyrics import fetch_external_lyrics

    body = request.get_json(force=True, silent=True) or {}
    artist = body.get("artist", "")
    title = body.get("title", "")
    if not artist or not title:
        return jsonify({"error": "artist and title required"}), 400
    try:
        text = fetch_external_lyrics(title, artist)
    except Exception as exc:
        return jsonify({"error": str(exc)}), 502
    return jsonify({"found": bool(text),
                    "preview": (text or "")[:500]})


@bp.post("/api/login")
def login():
    conn = _state().conn()
    body = request.get_json(force=True, silent=True) or {}
    row = conn.execute(
        "SELECT username, password_hash, role FROM audiomuse_users "
        "WHERE username=?", (body.get("username", ""),)).fetchone()
    if row is None or not verify_password(body.get("password", ""),
                                          row["password_hash"]):
        return jsonify({"error": "invalid credentials"}), 401
    token = make_session_token(row["username"], row["role"])
    resp = make_response(jsonify({"ok": True, "user": row["username"]}))
    resp.set_cookie(SESSION_COOKIE, token, httponly=True, samesite="Lax")
    return resp


@bp.post("/api/logout")
def logout():
    resp = make_response(jsonify({"ok": True}))
    resp.delete_cookie(SESSION_COOKIE)
    return resp


@bp.get("/api/me")
@require_auth
def me():
    from flask import g
    return jsonify({"user": g.user["u"], "role": g.user["r"]})


@bp.get("/api/users")
@require_auth
def list_users():
    """reference: app_users.py"""
    rows = _state().conn().execute(
        "SELECT username, role, created_at FROM audiomuse_users").fetchall()
    return jsonify([dict(r) for r in rows])


@bp.get("/api/config")
@require_auth
def get_config():
    """Config snapshot + persisted overrides (reference: setup manager)."""
    from audiomuse_amd import config as C
    from audiomuse_amd.db.store import get_app_config

    overrides = {k: v for k, v in get_app_config(_state().conn()).items()
                 if not k.startswith("_")}  # underscore keys are internal
    safe = {k: getattr(C, k) for k in dir(C)
            if k.isupper() and isinstance(getattr(C, k), (int, float, str, bool))
            and "SECRET" not in k and "TOKEN" not in k and "PASSWORD" not in k}
    return jsonify({"config": safe, "overrides": overrides})


@bp.get("/api/config/defaults")
@require_auth
def get_config_defaults():
    """Pristine env-resolved defaults, before DB overrides (reference:
    /api/config/defaults — the setup UI's reset-to-default values)."""
    from audiomuse_amd import config as C

    safe = {k: v for k, v in C.defaults().items()
            if "SECRET" not in k and "TOKEN" not in k and "PASSWORD" not in k}
    return jsonify({"defaults": safe})


@bp.post("/api/config")
@require_auth
def set_config():
    from audiomuse_amd.db.store import set_app_config

    body = request.get_json(force=True, silent=True) or {}
    conn = _state().conn()
    for k, v in body.items():
        if not isinstance(k, str) or not k.isupper():
            return jsonify({"error": f"bad key {k!r}"}), 400
        set_app_config(conn, k, str(v))
    from audiomuse_amd import config as C
    from audiomuse_amd.db.store import get_app_config

    C.apply_db_overrides(get_app_config(conn))   # workers hydrate on loop
    return jsonify({"saved": len(body)})
