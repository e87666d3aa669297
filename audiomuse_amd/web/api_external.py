"""External integration API (reference: app_external.py, mounted at
/external): read-only endpoints for media-server plugins and scripts —
raw score row, embedding vector, and autocomplete search. Callers may
send THEIR provider's track id; it resolves through track_server_map to
the canonical catalogue id exactly like internal inputs
(app_external.py:38-50 / app_server_context.resolve_input_item_id).
"""

from __future__ import annotations

import numpy as np
from flask import Blueprint, jsonify, request

from audiomuse_amd.web.auth import require_auth

bp = Blueprint("external", __name__, url_prefix="/external")


def _state():
    from flask import current_app

    return current_app.extensions["audiomuse"]


def _resolve_external_id(conn, raw_id: str, server_id: str | None) -> str:
    """Provider id -> canonical id; canonical/unknown ids pass through
    unchanged (the shared-resolver guarantee, app_external.py:44-49)."""
    if server_id:
        row = conn.execute(
            "SELECT item_id FROM track_server_map WHERE provider_id = ? "
            "AND server_id = ?", (raw_id, server_id)).fetchone()
    else:
        row = conn.execute(
            "SELECT item_id FROM track_server_map WHERE provider_id = ?",
            (raw_id,)).fetchone()
    return row["item_id"] if row else raw_id


@bp.get("/get_score")
@require_auth
def get_score():
    """Full score row for a track (app_external.py:96)."""
    raw = request.args.get("id", "")
    if not raw:
        return jsonify({"error": "missing id parameter"}), 400
    conn = _state().conn()
    iid = _resolve_external_id(conn, raw, request.args.get("server"))
    row = conn.execute("SELECT * FROM score WHERE item_id = ?",
                       (iid,)).fetchone()
    if row is None:
        return jsonify({"error": f"score not found for {raw!r}"}), 404
    return jsonify(dict(row))


@bp.get("/get_embedding")
@require_auth
def get_embedding():
    """Embedding vector as a float list (app_external.py:127)."""
    raw = request.args.get("id", "")
    if not raw:
        return jsonify({"error": "missing id parameter"}), 400
    conn = _state().conn()
    iid = _resolve_external_id(conn, raw, request.args.get("server"))
    row = conn.execute("SELECT * FROM embedding WHERE item_id = ?",
                       (iid,)).fetchone()
    if row is None:
        return jsonify({"error": f"embedding not found for {raw!r}"}), 404
    d = dict(row)
    d["embedding"] = np.frombuffer(d["embedding"],
                                   dtype=np.float32).tolist()
    return jsonify(d)


@bp.get("/search")
@require_auth
def search():
    """Autocomplete over title/author (app_external.py:156): unified
    ?q= or legacy ?title= / ?artist= parameters."""
    q = request.args.get("q", "")
    title = request.args.get("title", "")
    artist = request.args.get("artist", "")
    n = min(int(request.args.get("n", 10)), 50)
    conn = _state().conn()
    if q:
        like = f"%{q}%"
        rows = conn.execute(
            "SELECT item_id, title, author, album FROM score "
            "WHERE title LIKE ? OR author LIKE ? LIMIT ?",
            (like, like, n)).fetchall()
    else:
        rows = conn.execute(
            "SELECT item_id, title, author, album FROM score "
            "WHERE title LIKE ? AND author LIKE ? LIMIT ?",
            (f"%{title}%", f"%{artist}%", n)).fetchall()
    return jsonify([dict(r) for r in rows])
