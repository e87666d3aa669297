"""Task-control blueprints.

Reference: task control routes (app.py:415-768), queue-state blueprints
(app_analysis.py / app_clustering.py / app_cron.py), admission gate
(app_helper.admit_and_enqueue_main_task :95), recursive cancel (:478).
"""

from __future__ import annotations

import json

from flask import Blueprint, current_app, jsonify, request

from audiomuse_amd import config as C
from audiomuse_amd.taskqueue import (PENDING, RUNNING, cancel_task_recursive,
                                     enqueue, task_row)
from audiomuse_amd.taskqueue import sql as qsql
from audiomuse_amd.web.auth import require_auth

bp = Blueprint("tasks", __name__)

MAIN_TASK_TYPES = ("run_analysis", "run_clustering")


def _state():
    return current_app.extensions["audiomuse"]


def _admit_and_enqueue(conn, task_type: str, payload: dict) -> tuple:
    """Admission gate: one live main task of a type at a time
    (app_helper.py:95-131)."""
    live = conn.execute(
        "SELECT task_id FROM task_status WHERE task_type=? AND status IN (?,?)",
        (task_type, PENDING, RUNNING)).fetchone()
    if live is not None:
        return None, live["task_id"]
    return enqueue(conn, task_type, payload, queue="high"), None


@bp.post("/api/analysis/start")
@require_auth
def analysis_start():
    body = request.get_json(force=True, silent=True) or {}
    conn = _state().conn()
    tid, existing = _admit_and_enqueue(conn, "run_analysis", {
        "server_type": body.get("server_type", C.MEDIASERVER_TYPE),
        "server_config": body.get("server_config", {}),
        "server_id": body.get("server_id", "default"),
        "album_limit": int(body.get("album_limit", 0)),
    })
    if tid is None:
        return jsonify({"error": "analysis already running",
                        "task_id": existing}), 409
    return jsonify({"task_id": tid}), 202


@bp.post("/api/clustering/start")
@require_auth
def clustering_start():
    body = request.get_json(force=True, silent=True) or {}
    conn = _state().conn()
    tid, existing = _admit_and_enqueue(conn, "run_clustering", {
        "algorithm": body.get("algorithm", C.CLUSTER_ALGORITHM),
        "runs": int(body.get("runs", C.CLUSTERING_RUNS)),
        "server_id": body.get("server_id", "default"),
    })
    if tid is None:
        return jsonify({"error": "clustering already running",
                        "task_id": existing}), 409
    return jsonify({"task_id": tid}), 202


@bp.post("/api/index/rebuild")
@require_auth
def index_rebuild():
    conn = _state().conn()
    tid = enqueue(conn, "rebuild_indexes", {}, queue="high")
    return jsonify({"task_id": tid}), 202


@bp.post("/api/index/refresh")
@require_auth
def index_refresh():
    """Incremental IVF splice (analysis.index.refresh_ivf_index):
    new/removed tracks folded into the stored packed indexes without a
    full rebuild — cheap enough to run after every analysis batch."""
    conn = _state().conn()
    tid = enqueue(conn, "refresh_indexes", {}, queue="high")
    return jsonify({"task_id": tid}), 202


@bp.get("/api/task/<task_id>")
@require_auth
def task_status(task_id: str):
    row = task_row(_state().conn(), task_id)
    if row is None:
        return jsonify({"error": "unknown task"}), 404
    d = dict(row)
    for k in ("payload", "result"):
        try:
            d[k] = json.loads(d[k]) if d[k] else None
        except Exception:
            pass
    return jsonify(d)


@bp.post("/api/task/<task_id>/cancel")
@require_auth
def task_cancel(task_id: str):
    n = cancel_task_recursive(_state().conn(), task_id)
    return jsonify({"cancelled": n})


@bp.get("/api/active_tasks")
@require_auth
def active_tasks():
    """reference: app.py:768"""
    rows = _state().conn().execute(
        """SELECT task_id, task_type, status, progress, details, queue,
               parent_task_id, created_at FROM task_status
           WHERE status IN (?, ?) ORDER BY created_at DESC LIMIT 200""",
        (PENDING, RUNNING)).fetchall()
    return jsonify([dict(r) for r in rows])


@bp.get("/api/queue/stats")
@require_auth
def queue_stats():
    return jsonify(qsql.counts_by_status(_state().conn()))


@bp.post("/api/create_playlist")
@require_auth
def create_playlist():
    """Create a playlist ON the media server (reference: ALGORITHM.md
    §6.2 playlist creation): canonical ids translate back to the
    selected server's provider ids via track_server_map, tracks the
    server does not have are dropped, and the response reports how many
    were unavailable."""
    from audiomuse_amd.mediaserver import make_provider

    body = request.get_json(force=True, silent=True) or {}
    name = body.get("name", "")
    item_ids = body.get("item_ids", [])
    if not name or not item_ids:
        return jsonify({"error": "name and item_ids required"}), 400
    conn = _state().conn()
    server_id = body.get("server_id")
    row = conn.execute(
        "SELECT * FROM music_servers WHERE enabled = 1"
        + (" AND server_id = ?" if server_id else "") + " LIMIT 1",
        (server_id,) if server_id else ()).fetchone()
    if row is None:
        return jsonify({"error": "no configured media server"}), 404
    # per-artist cap for created playlists (reference
    # MAX_SONGS_PER_ARTIST_PLAYLIST; 0 = uncapped)
    cap = int(body.get("max_per_artist", C.MAX_SONGS_PER_ARTIST_PLAYLIST))
    per_artist = {}
    provider_ids, missing = [], 0
    for iid in item_ids:
        if cap:
            meta = _state().meta_fn(iid) or {}
            a = (meta.get("author") or "").strip().lower()
            if a and per_artist.get(a, 0) >= cap:
                continue
            if a:
                per_artist[a] = per_artist.get(a, 0) + 1
        m = conn.execute(
            "SELECT provider_id FROM track_server_map WHERE item_id = ? "
            "AND server_id = ?", (iid, row["server_id"])).fetchone()
        if m is None:
            missing += 1
        else:
            provider_ids.append(m["provider_id"])
    if not provider_ids:
        return jsonify({"error": "no tracks available on this server",
                        "missing": missing}), 404
    cfg = json.loads(row["config"] or "{}")
    provider = make_provider(row["server_type"], base_url=row["base_url"],
                             username=row["username"],
                             credential=row["credential"], **cfg)
    pid = provider.create_playlist(name, provider_ids)
    return jsonify({"playlist_id": pid, "created": len(provider_ids),
                    "missing": missing}), 201


# -- music server registry (reference: app_music_servers.py) ---------------

@bp.get("/api/servers")
@require_auth
def list_servers():
    rows = _state().conn().execute(
        "SELECT server_id, server_type, base_url, username, enabled "
        "FROM music_servers").fetchall()
    return jsonify([dict(r) for r in rows])


@bp.post("/api/servers")
@require_auth
def add_server():
    from audiomuse_amd.db import write_txn
    from audiomuse_amd.mediaserver import provider_types

    body = request.get_json(force=True, silent=True) or {}
    stype = body.get("server_type", "")
    if stype not in provider_types():
        return jsonify({"error": f"unsupported type {stype!r}",
                        "supported": provider_types()}), 400
    conn = _state().conn()
    with write_txn(conn):
        conn.execute(
            """INSERT INTO music_servers (server_id, server_type, base_url,
                   username, credential, config)
               VALUES (?,?,?,?,?,?)
               ON CONFLICT(server_id) DO UPDATE SET
                   server_type=excluded.server_type,
                   base_url=excluded.base_url, username=excluded.username,
                   credential=excluded.credential, config=excluded.config""",
            (body.get("server_id", "default"), stype,
             body.get("base_url", ""), body.get("username", ""),
             body.get("credential", ""),
             json.dumps(body.get("config", {}))))
    return jsonify({"ok": True})


@bp.delete("/api/servers/<server_id>")
@require_auth
def delete_server(server_id: str):
    from audiomuse_amd.db import write_txn

    conn = _state().conn()
    with write_txn(conn):
        cur = conn.execute("DELETE FROM music_servers WHERE server_id=?",
                           (server_id,))
    return jsonify({"deleted": cur.rowcount})


def _server_row(conn, server_id: str):
    return conn.execute("SELECT * FROM music_servers WHERE server_id = ?",
                        (server_id,)).fetchone()


def _provider_from_row(row):
    from audiomuse_amd.mediaserver import make_provider

    cfg = json.loads(row["config"] or "{}")
    return make_provider(row["server_type"], base_url=row["base_url"],
                         username=row["username"],
                         credential=row["credential"], **cfg)


@bp.post("/api/servers/test")
@require_auth
def server_test():
    """Connectivity probe for a server config BEFORE saving it
    (reference: app_music_servers.py /api/servers/test)."""
    from audiomuse_amd.analysis.migration import probe_server

    body = request.get_json(force=True, silent=True) or {}
    try:
        out = probe_server(body.get("server_type", ""),
                           body.get("server_config", body))
    except Exception as exc:  # unsupported type / connection refused
        out = {"reachable": False, "error": str(exc)}
    out["ok"] = bool(out.get("reachable"))
    return jsonify(out), 200 if out["ok"] else 502


@bp.get("/api/servers/<server_id>/libraries")
@require_auth
def server_libraries(server_id: str):
    """Music libraries/folders of a configured server (reference:
    /api/servers/libraries — scoping for MUSIC_LIBRARIES)."""
    conn = _state().conn()
    row = _server_row(conn, server_id)
    if row is None:
        return jsonify({"error": f"unknown server {server_id!r}"}), 404
    provider = _provider_from_row(row)
    libs = getattr(provider, "list_libraries", lambda: [])()
    return jsonify(libs)


@bp.post("/api/servers/<server_id>/sweep")
@require_auth
def server_sweep(server_id: str):
    """Metadata-only alignment sweep of ONE server (reference:
    /api/servers/<id>/sweep -> multiserver_sync for that server)."""
    conn = _state().conn()
    row = _server_row(conn, server_id)
    if row is None:
        return jsonify({"error": f"unknown server {server_id!r}"}), 404
    cfg = json.loads(row["config"] or "{}")
    tid, existing = _admit_and_enqueue(conn, "multiserver_sync", {
        "server_type": row["server_type"], "server_id": server_id,
        "server_config": {"base_url": row["base_url"],
                          "username": row["username"],
                          "credential": row["credential"], **cfg}})
    if tid is None:
        return jsonify({"error": "sweep already running",
                        "task_id": existing}), 409
    return jsonify({"task_id": tid}), 202


@bp.post("/api/sync")
@require_auth
def sync_all():
    """Sweep every enabled server (reference: app_sync.py /api/sync)."""
    conn = _state().conn()
    rows = conn.execute(
        "SELECT server_id FROM music_servers WHERE enabled = 1").fetchall()
    if not rows:
        return jsonify({"error": "no configured media server"}), 404
    tids = []
    for r in rows:
        row = _server_row(conn, r["server_id"])
        cfg = json.loads(row["config"] or "{}")
        tid = enqueue(conn, "multiserver_sync", {
            "server_type": row["server_type"],
            "server_id": row["server_id"],
            "server_config": {"base_url": row["base_url"],
                              "username": row["username"],
                              "credential": row["credential"], **cfg}},
            queue="high")
        tids.append(tid)
    return jsonify({"task_ids": tids}), 202


@bp.post("/api/cleaning/start")
@require_auth
def cleaning_start():
    """Orphan report/delete task (reference: app.py /api/cleaning/start
    -> cleaning.py:48). delete=false previews only."""
    body = request.get_json(force=True, silent=True) or {}
    conn = _state().conn()
    tid, existing = _admit_and_enqueue(conn, "clean_orphans", {
        "delete": bool(body.get("delete", False))})
    if tid is None:
        return jsonify({"error": "cleaning already running",
                        "task_id": existing}), 409
    return jsonify({"task_id": tid}), 202


@bp.post("/api/cancel_all/<prefix>")
@require_auth
def cancel_all(prefix: str):
    """Cancel every live task whose type starts with prefix (reference:
    app.py /api/cancel_all/<task_type_prefix>; recursive per task)."""
    conn = _state().conn()
    rows = conn.execute(
        "SELECT task_id FROM task_status WHERE task_type LIKE ? "
        "AND status IN (?,?)", (prefix + "%", PENDING, RUNNING)).fetchall()
    for r in rows:
        cancel_task_recursive(conn, r["task_id"])
    return jsonify({"cancelled": len(rows)})


@bp.get("/api/last_task")
@require_auth
def last_task():
    """Most recent task of a type (reference: app.py /api/last_task)."""
    ttype = request.args.get("task_type")
    conn = _state().conn()
    row = conn.execute(
        "SELECT task_id, task_type, status, progress, created_at "
        "FROM task_status"
        + (" WHERE task_type = ?" if ttype else "")
        + " ORDER BY created_at DESC LIMIT 1",
        (ttype,) if ttype else ()).fetchone()
    if row is None:
        return jsonify({}), 404
    return jsonify(dict(row))


@bp.get("/api/playlists")
@require_auth
def playlists():
    """Stored playlists (reference: /api/playlists)."""
    conn = _state().conn()
    rows = conn.execute(
        "SELECT id, name, server_id, kind, item_ids, created_at "
        "FROM playlist ORDER BY created_at DESC LIMIT ?",
        (int(request.args.get("n", 100)),)).fetchall()
    out = []
    for r in rows:
        d = dict(r)
        ids = json.loads(d.pop("item_ids") or "[]")
        d["n_tracks"] = len(ids)
        if request.args.get("include_tracks") in ("1", "true"):
            d["item_ids"] = ids
        out.append(d)
    return jsonify(out)


@bp.get("/api/search_playlists")
@require_auth
def search_playlists():
    """Name search over stored playlists (reference: /api/search_playlists)."""
    q = (request.args.get("q") or "").strip().lower()
    conn = _state().conn()
    rows = conn.execute(
        "SELECT id, name, kind, created_at FROM playlist "
        "WHERE LOWER(name) LIKE ? ORDER BY created_at DESC LIMIT ?",
        (f"%{q}%", int(request.args.get("n", 50)))).fetchall()
    return jsonify([dict(r) for r in rows])


# -- cron (reference: app_cron.py minute-claimed scheduler) -----------------

@bp.get("/api/cron")
@require_auth
def cron_list():
    """Rows plus retry-pending state (reference: ALGORITHM.md 16.2 step
    6 — the Scheduled Tasks page shows a waiting schedule as waiting)."""
    conn = _state().conn()
    rows = conn.execute(
        "SELECT id, name, schedule, task_type, payload, enabled FROM cron"
    ).fetchall()
    retries = {r["cron_id"]: r for r in conn.execute(
        "SELECT cron_id, attempts, due_at FROM cron_retry").fetchall()}
    out = []
    for r in rows:
        d = dict(r)
        ret = retries.get(r["id"])
        d["retry_pending"] = ret is not None
        if ret is not None:
            d["retry_attempts"] = ret["attempts"]
            d["retry_due_at"] = ret["due_at"]
        out.append(d)
    return jsonify(out)


@bp.post("/api/cron")
@require_auth
def cron_add():
    from audiomuse_amd.db import write_txn
    from audiomuse_amd.utils.cron import validate_cron

    body = request.get_json(force=True, silent=True) or {}
    schedule = body.get("schedule", "0 3 * * *")
    if not validate_cron(schedule):
        return jsonify({"error": f"invalid cron expression {schedule!r}"}), 400
    conn = _state().conn()
    from audiomuse_amd.db import insert_returning_id
    with write_txn(conn):
        rid = insert_returning_id(
            conn,
            "INSERT INTO cron (name, schedule, task_type, payload, enabled) "
            "VALUES (?,?,?,?,1)",
            (body.get("name", ""), schedule,
             body.get("task_type", "rebuild_indexes"),
             json.dumps(body.get("payload", {}))))
    return jsonify({"id": rid})


@bp.delete("/api/cron/<int:cron_id>")
@require_auth
def cron_delete(cron_id: int):
    from audiomuse_amd.db import write_txn

    conn = _state().conn()
    with write_txn(conn):
        cur = conn.execute("DELETE FROM cron WHERE id=?", (cron_id,))
    return jsonify({"deleted": cur.rowcount})
