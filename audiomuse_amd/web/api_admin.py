"""Admin blueprints: backup/restore, dashboard, anchors/radios, plugins,
provider migration.

References: app_backup.py (pg_dump/restore + lock), app_dashboard.py,
alchemy anchors/radios (song_alchemy + radio_manager.py),
app_provider_migration.py (migration wizard -> task).
"""

from __future__ import annotations

import json
import os
import tempfile

import numpy as np
from flask import (Blueprint, current_app, jsonify, request, send_file)

from audiomuse_amd.analysis.maintenance import (backup_database,
                                                refresh_dashboard_stats)
from audiomuse_amd.db import write_txn
from audiomuse_amd.taskqueue import enqueue
from audiomuse_amd.web.auth import require_auth

bp = Blueprint("admin", __name__)


def _state():
    return current_app.extensions["audiomuse"]


@bp.get("/api/backup")
@require_auth
def backup():
    conn = _state().conn()
    fd, path = tempfile.mkstemp(suffix=".db", prefix="audiomuse-backup-")
    os.close(fd)
    backup_database(conn, path)
    return send_file(path, as_attachment=True,
                     download_name="audiomuse-backup.db")


@bp.post("/api/restore")
@require_auth
def restore():
    """Restore lock semantics (app_backup.py:607): refuse while tasks
    run; replace the live DB file; engine caches invalidate."""
    from audiomuse_amd.taskqueue import PENDING, RUNNING

    state = _state()
    conn = state.conn()
    live = conn.execute(
        "SELECT COUNT(*) AS n FROM task_status WHERE status IN (?, ?)",
        (PENDING, RUNNING)).fetchone()["n"]
    if live:
        return jsonify({"error": "tasks running; cancel them first"}), 409
    blob = request.get_data()
    if not blob.startswith(b"SQLite format 3"):
        return jsonify({"error": "not an SQLite backup"}), 400
    import sqlite3 as s3

    from audiomuse_amd.db import backend_kind
    if backend_kind(state.db_url) == "postgres":
        # logical reload through the live connection (pg_dump analog)
        fd, tmp = tempfile.mkstemp(suffix=".restore.db")
        os.close(fd)
        try:
            with open(tmp, "wb") as fh:
                fh.write(blob)
            check = s3.connect(tmp)
            check.execute("SELECT COUNT(*) FROM score")
            check.close()
            from audiomuse_amd.analysis.maintenance import restore_database
            restore_database(conn, tmp)
        except Exception as exc:  # noqa: BLE001
            return jsonify({"error": f"backup failed validation: {exc}"}), 400
        finally:
            os.unlink(tmp)
        state.invalidate()
        return jsonify({"restored": True})
    db_path = state.db_url[len("sqlite:///"):]
    tmp = db_path + ".restore"
    with open(tmp, "wb") as fh:
        fh.write(blob)
    try:
        check = s3.connect(tmp)
        check.execute("SELECT COUNT(*) FROM score")
        check.close()
    except Exception as exc:  # noqa: BLE001
        os.unlink(tmp)
        return jsonify({"error": f"backup failed validation: {exc}"}), 400
    os.replace(tmp, db_path)
    state._local.__dict__.clear()
    state.invalidate()
    return jsonify({"restored": True})


@bp.get("/api/dashboard")
@require_auth
def dashboard():
    return jsonify(refresh_dashboard_stats(_state().conn()))


@bp.get("/api/dashboard/browse")
@require_auth
def dashboard_browse():
    """Paged catalogue browser (reference: app_dashboard.py browse with
    DASHBOARD_BROWSE_PAGE_SIZE / DASHBOARD_BROWSE_MAX_OFFSET caps)."""
    from audiomuse_amd import config as C

    offset = min(max(int(request.args.get("offset", 0)), 0),
                 C.DASHBOARD_BROWSE_MAX_OFFSET)
    limit = min(int(request.args.get("limit",
                                     C.DASHBOARD_BROWSE_PAGE_SIZE)),
                C.DASHBOARD_BROWSE_PAGE_SIZE)
    q = (request.args.get("q") or "").strip().lower()
    conn = _state().conn()
    if q:
        rows = conn.execute(
            """SELECT item_id, title, author, album, tempo, energy
               FROM score WHERE LOWER(title) LIKE ? OR LOWER(author) LIKE ?
               ORDER BY item_id LIMIT ? OFFSET ?""",
            (f"%{q}%", f"%{q}%", limit, offset)).fetchall()
    else:
        rows = conn.execute(
            """SELECT item_id, title, author, album, tempo, energy
               FROM score ORDER BY item_id LIMIT ? OFFSET ?""",
            (limit, offset)).fetchall()
    total = conn.execute("SELECT COUNT(*) AS n FROM score").fetchone()["n"]
    return jsonify({"rows": [dict(r) for r in rows], "offset": offset,
                    "limit": limit, "total": total})


# -- alchemy anchors / radios ----------------------------------------------

@bp.get("/api/alchemy/anchors")
@require_auth
def list_anchors():
    rows = _state().conn().execute(
        "SELECT name FROM alchemy_anchors").fetchall()
    return jsonify([r["name"] for r in rows])


@bp.post("/api/alchemy/anchors")
@require_auth
def save_anchor():
    from audiomuse_amd.analysis import index as idx

    body = request.get_json(force=True, silent=True) or {}
    name = body.get("name", "").strip()
    ids = body.get("item_ids", [])
    if not name or not ids:
        return jsonify({"error": "name and item_ids required"}), 400
    eng = _state().engine(idx.AUDIO_INDEX)
    if eng is None:
        return jsonify({"error": "audio index not built"}), 503
    vecs = [eng.vector_for_id(i) for i in ids]
    vecs = [v.cpu().numpy() for v in vecs if v is not None]
    if not vecs:
        return jsonify({"error": "no known tracks"}), 404
    centroid = np.mean(np.stack(vecs), axis=0).astype(np.float32)
    conn = _state().conn()
    with write_txn(conn):
        conn.execute(
            """INSERT INTO alchemy_anchors (name, vector) VALUES (?,?)
               ON CONFLICT(name) DO UPDATE SET vector=excluded.vector""",
            (name, centroid.tobytes()))
    return jsonify({"saved": name, "dim": centroid.shape[0]})


@bp.delete("/api/alchemy/anchors/<name>")
@require_auth
def delete_anchor(name):
    conn = _state().conn()
    with write_txn(conn):
        cur = conn.execute("DELETE FROM alchemy_anchors WHERE name=?", (name,))
    return jsonify({"deleted": cur.rowcount})


@bp.get("/api/alchemy/radios")
@require_auth
def list_radios():
    rows = _state().conn().execute(
        "SELECT name, definition FROM alchemy_radios").fetchall()
    return jsonify([{"name": r["name"],
                     "definition": json.loads(r["definition"])} for r in rows])


@bp.post("/api/alchemy/radios")
@require_auth
def save_radio():
    body = request.get_json(force=True, silent=True) or {}
    name = body.get("name", "").strip()
    if not name:
        return jsonify({"error": "name required"}), 400
    conn = _state().conn()
    with write_txn(conn):
        conn.execute(
            """INSERT INTO alchemy_radios (name, definition) VALUES (?,?)
               ON CONFLICT(name) DO UPDATE SET definition=excluded.definition""",
            (name, json.dumps(body.get("definition", {}))))
    return jsonify({"saved": name})


# -- provider migration wizard -----------------------------------------------
# (reference: app_provider_migration.py:2828 — probe, library select,
# path-format detection, match preview, transactional rewrite, restart
# handshake). Two flows: the one-shot probe/preview/start wizard below,
# and the per-album review sessions (migration_session.py) after it.

@bp.post("/api/migration/probe")
@require_auth
def migration_probe():
    """Wizard step 1: can we reach the target; which libraries; what do
    its paths look like."""
    from audiomuse_amd.analysis.migration import probe_server

    body = request.get_json(force=True, silent=True) or {}
    try:
        out = probe_server(body.get("server_type", "synthetic"),
                           body.get("server_config", {}))
    except Exception as exc:  # noqa: BLE001 — wizard shows the reason
        return jsonify({"reachable": False, "error": str(exc)}), 502
    return jsonify(out)


@bp.post("/api/migration/preview")
@require_auth
def migration_preview():
    """Wizard steps 2-3: path rule + tiered match preview. Read-only —
    nothing is written until /api/migration/start with apply=true."""
    from audiomuse_amd.analysis.migration import (build_match_preview,
                                                  propose_path_rule)
    from audiomuse_amd.mediaserver import make_provider

    body = request.get_json(force=True, silent=True) or {}
    conn = _state().conn()
    provider = make_provider(body.get("server_type", "synthetic"),
                             **body.get("server_config", {}))
    tracks = provider.get_all_songs()
    source = body.get("source_server_id", "default")
    src_paths = [r["file_path"] for r in conn.execute(
        "SELECT file_path FROM track_server_map WHERE server_id=? "
        "AND file_path != ''", (source,)).fetchall()]
    rule = propose_path_rule([t.file_path for t in tracks], src_paths)
    preview = build_match_preview(conn, tracks, source, path_rule=rule)
    return jsonify({"path_rule": rule, "tiers": preview["tiers"],
                    "total": preview["total"],
                    "matched": preview["matched"],
                    "match_ratio": round(preview["match_ratio"], 4),
                    "unmatched": preview["unmatched"][:50]})


@bp.post("/api/migration/start")
@require_auth
def migration_start():
    """Wizard step 4: queue the full wizard task. apply=false stops at
    the preview; apply=true performs the transactional rewrite and the
    restart handshake."""
    body = request.get_json(force=True, silent=True) or {}
    tid = enqueue(_state().conn(), "provider_migration", {
        "server_type": body.get("server_type", "synthetic"),
        "server_config": body.get("server_config", {}),
        "source_server_id": body.get("source_server_id", "default"),
        "target_server_id": body.get("target_server_id", "migrated"),
        "apply": bool(body.get("apply", False)),
        "remove_source": bool(body.get("remove_source", False)),
        "min_match_ratio": float(body.get("min_match_ratio", 0.5)),
    }, queue="high")
    return jsonify({"task_id": tid}), 202


# -- per-album review sessions (reference: migration_session +
# matched-albums / match-album / skip-album / dry-run routes,
# app_provider_migration.py:678-2502) --------------------------------------

@bp.post("/api/migration/session/start")
@require_auth
def migration_session_start():
    from audiomuse_amd.analysis import migration_session as ms

    body = request.get_json(force=True, silent=True) or {}
    try:
        out = ms.create_session(
            _state().conn(), body.get("server_type", "synthetic"),
            body.get("server_config", {}),
            body.get("source_server_id", "default"))
    except Exception as exc:  # noqa: BLE001 — wizard shows the reason
        return jsonify({"error": str(exc)}), 502
    return (jsonify(out), 502) if "error" in out else (jsonify(out), 201)


@bp.get("/api/migration/session/<int:sid>")
@require_auth
def migration_session_get(sid: int):
    from audiomuse_amd.analysis import migration_session as ms

    out = ms.get_session(_state().conn(), sid)
    if out is None:
        return jsonify({"error": "unknown session"}), 404
    return jsonify(out)


@bp.delete("/api/migration/session/<int:sid>")
@require_auth
def migration_session_discard(sid: int):
    from audiomuse_amd.analysis import migration_session as ms

    ok = ms.discard_session(_state().conn(), sid)
    return jsonify({"discarded": bool(ok)}), 200 if ok else 409


@bp.post("/api/migration/dry-run")
@require_auth
def migration_dry_run():
    from audiomuse_amd.analysis import migration_session as ms

    body = request.get_json(force=True, silent=True) or {}
    out = ms.run_dry_run(_state().conn(), int(body.get("session_id", 0)))
    if out is None:
        return jsonify({"error": "unknown or closed session"}), 404
    return jsonify(out)


@bp.get("/api/migration/dry-run-report/<int:sid>")
@require_auth
def migration_dry_run_report(sid: int):
    from audiomuse_amd.analysis import migration_session as ms

    out = ms.dry_run_report(_state().conn(), sid)
    if out is None:
        return jsonify({"error": "no report for session"}), 404
    return jsonify(out)


@bp.get("/api/migration/matched-albums/<int:sid>")
@require_auth
def migration_matched_albums(sid: int):
    from audiomuse_amd.analysis import migration_session as ms

    out = ms.matched_albums(_state().conn(), sid)
    if out is None:
        return jsonify({"error": "no report for session"}), 404
    return jsonify(out)


@bp.post("/api/migration/match-album")
@require_auth
def migration_match_album():
    from audiomuse_amd.analysis import migration_session as ms

    body = request.get_json(force=True, silent=True) or {}
    out = ms.set_decision(
        _state().conn(), int(body.get("session_id", 0)),
        body.get("album", ""), "map",
        target_album=body.get("target_album"))
    if out is None:
        return jsonify({"error": "unknown or closed session"}), 404
    return jsonify({"decisions": out})


@bp.post("/api/migration/skip-album")
@require_auth
def migration_skip_album():
    from audiomuse_amd.analysis import migration_session as ms

    body = request.get_json(force=True, silent=True) or {}
    action = "auto" if body.get("undo") else "skip"
    out = ms.set_decision(_state().conn(), int(body.get("session_id", 0)),
                          body.get("album", ""), action)
    if out is None:
        return jsonify({"error": "unknown or closed session"}), 404
    return jsonify({"decisions": out})


@bp.post("/api/migration/search-albums")
@require_auth
def migration_search_albums():
    from audiomuse_amd.analysis import migration_session as ms

    body = request.get_json(force=True, silent=True) or {}
    return jsonify(ms.search_albums(
        _state().conn(), int(body.get("session_id", 0)),
        body.get("q", "")))


@bp.post("/api/migration/finalize-dry-run")
@require_auth
def migration_finalize():
    from audiomuse_amd.analysis import migration_session as ms

    body = request.get_json(force=True, silent=True) or {}
    out = ms.finalize(_state().conn(), int(body.get("session_id", 0)))
    if out is None:
        return jsonify({"error": "session has no current dry run"}), 409
    return jsonify(out)


@bp.post("/api/migration/execute")
@require_auth
def migration_execute():
    from audiomuse_amd.analysis import migration_session as ms

    body = request.get_json(force=True, silent=True) or {}
    out = ms.execute_session(
        _state().conn(), int(body.get("session_id", 0)),
        body.get("target_server_id", "migrated"),
        remove_source=bool(body.get("remove_source", False)),
        min_match_ratio=float(body.get("min_match_ratio", 0.5)))
    if out is None:
        return jsonify({"error": "session is not finalized"}), 409
    return jsonify(out)


@bp.get("/api/migration/status/<task_id>")
@require_auth
def migration_status(task_id):
    from audiomuse_amd.taskqueue import task_row

    row = task_row(_state().conn(), task_id)
    if row is None:
        return jsonify({"error": "unknown task"}), 404
    result = row["result"]
    return jsonify({"status": row["status"], "progress": row["progress"],
                    "details": row["details"],
                    "result": json.loads(result) if result else None})


@bp.get("/api/plugins")
@require_auth
def list_plugins():
    rows = _state().conn().execute(
        "SELECT name, enabled, uploaded_at FROM plugin ORDER BY name"
    ).fetchall()
    return jsonify([dict(r) for r in rows])


@bp.post("/api/plugins")
@require_auth
def upload_plugin():
    """Store + load a plugin zip (reference: plugin/blueprint.py upload).
    Body: raw zip bytes with ?name=..., or JSON {name, zip_base64}."""
    import base64

    from audiomuse_amd.db import write_txn
    from audiomuse_amd.plugin import plugin_manager

    if request.is_json:
        body = request.get_json(silent=True) or {}
        name = body.get("name", "")
        try:
            blob = base64.b64decode(body.get("zip_base64", ""))
        except Exception:
            return jsonify({"error": "invalid zip_base64"}), 400
    else:
        name = request.args.get("name", "")
        blob = request.get_data()
    if not name or not name.replace("_", "").replace("-", "").isalnum():
        return jsonify({"error": "plugin name must be alphanumeric"}), 400
    if not blob:
        return jsonify({"error": "empty plugin zip"}), 400
    from audiomuse_amd import config as C
    if not C.PLUGINS_ENABLED:
        return jsonify({"error": "plugins disabled (PLUGINS_ENABLED=0)"}), 403
    if len(blob) > C.PLUGIN_MAX_DOWNLOAD_MB * 1024 * 1024:
        return jsonify({"error": f"plugin exceeds PLUGIN_MAX_DOWNLOAD_MB "
                                 f"({C.PLUGIN_MAX_DOWNLOAD_MB} MB)"}), 413
    try:
        plugin_manager.load_zip(name, blob)   # validate before persisting
    except Exception as exc:  # noqa: BLE001 — surface the load error
        return jsonify({"error": f"plugin failed to load: {exc}"}), 400
    conn = _state().conn()
    with write_txn(conn):
        conn.execute(
            "INSERT INTO plugin (name, blob, enabled) VALUES (?, ?, 1) "
            "ON CONFLICT(name) DO UPDATE SET blob=excluded.blob, enabled=1",
            (name, blob))
    plugin_manager.sync_cron(conn)
    return jsonify({"loaded": name}), 201


@bp.delete("/api/plugins/<name>")
@require_auth
def delete_plugin(name):
    from audiomuse_amd.db import write_txn
    from audiomuse_amd.plugin import plugin_manager

    conn = _state().conn()
    with write_txn(conn):
        cur = conn.execute("DELETE FROM plugin WHERE name = ?", (name,))
    plugin_manager.loaded.pop(name, None)
    plugin_manager.sync_cron(conn)
    if cur.rowcount == 0:
        return jsonify({"error": "unknown plugin"}), 404
    return jsonify({"deleted": name})
