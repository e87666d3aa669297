"""Flask application factory.

Reference: /root/reference/app.py (1409 LoC) + 26 app_*.py blueprints —
blueprint wiring, health, task status/cancel, startup threads
(index-reload listener, cron loop). Here: create_app() wires the
blueprints over the SQLite storage; the index cache reloads engines
when ivf_dir.updated_at changes (the LISTEN/NOTIFY analog on SQLite).
"""

from __future__ import annotations

import json
import threading
import time
from typing import Dict, Optional

from flask import Flask, jsonify

from audiomuse_amd import config as C
from audiomuse_amd.analysis import index as idx
from audiomuse_amd.db import connect
from audiomuse_amd.db.schema import init_db


class AppState:
    """Per-app storage + cached engines (reference: the module-level index
    caches reloaded by listen_for_index_reloads, app.py:971-1076)."""

    def __init__(self, db_url: str, device: str = "cpu"):
        self.db_url = db_url
        self.device = device
        self._local = threading.local()
        self._engines: Dict[str, object] = {}
        self._stamps: Dict[str, float] = {}
        self._lock = threading.Lock()

    def conn(self):
        conn = getattr(self._local, "conn", None)
        if conn is None:
            conn = self._local.conn = connect(self.db_url)
            init_db(conn)
        return conn

    def meta_fn(self, item_id: str) -> Optional[dict]:
        row = self.conn().execute(
            "SELECT title, author, mood_vector, other_features, tempo, energy,"
            " key, scale, duration FROM score WHERE item_id=?",
            (item_id,)).fetchone()
        if row is None:
            return None
        d = dict(row)
        for k in ("mood_vector", "other_features"):
            try:
                d[k] = json.loads(d[k]) if d[k] else {}
            except Exception:
                d[k] = {}
        return d

    def _stamp(self, name: str) -> Optional[float]:
        row = self.conn().execute(
            "SELECT updated_at FROM ivf_dir WHERE index_name=?",
            (name,)).fetchone()
        return float(row["updated_at"]) if row else None

    def engine(self, name: str):
        """Cached engine for an index blob; reloads when the stored stamp
        changes."""
        stamp = self._stamp(name)
        if stamp is None:
            return None
        with self._lock:
            if self._stamps.get(name) == stamp and name in self._engines:
                return self._engines[name]
        if name == idx.ARTIST_INDEX:
            eng = idx.load_artist_similarity(self.conn())
        elif name in (idx.SONG_MAP, idx.ARTIST_MAP):
            got = None
            from audiomuse_amd.db.store import load_index_blob
            got = load_index_blob(self.conn(), name)
            if got is None:
                return None
            import io
            import torch
            eng = torch.load(io.BytesIO(got[0]), map_location="cpu",
                             weights_only=True)
        else:
            eng = idx.load_ivf_engine(self.conn(), name, device=self.device,
                                      meta_fn=self.meta_fn)
        with self._lock:
            self._engines[name] = eng
            self._stamps[name] = stamp
        return eng

    def invalidate(self) -> None:
        with self._lock:
            self._engines.clear()
            self._stamps.clear()


class ProxyPrefixMiddleware:
    """Honor X-Forwarded-Prefix from a reverse proxy so url_for and
    redirects carry the mount prefix (reference: proxy_prefix.py:31).

    Opt-in via AUDIOMUSE_BEHIND_PROXY: without a proxy in front, these
    headers arrive attacker-controlled and could rewrite SCRIPT_NAME /
    url scheme in generated URLs."""

    def __init__(self, wsgi_app):
        self.wsgi_app = wsgi_app

    def __call__(self, environ, start_response):
        prefix = environ.get("HTTP_X_FORWARDED_PREFIX", "").rstrip("/")
        if prefix and prefix.startswith("/"):
            environ["SCRIPT_NAME"] = prefix
            path = environ.get("PATH_INFO", "")
            if path.startswith(prefix):
                environ["PATH_INFO"] = path[len(prefix):] or "/"
        scheme = environ.get("HTTP_X_FORWARDED_PROTO", "")
        if scheme in ("http", "https"):
            environ["wsgi.url_scheme"] = scheme
        return self.wsgi_app(environ, start_response)


def create_app(db_url: Optional[str] = None, device: Optional[str] = None,
               auth_disabled: Optional[bool] = None) -> Flask:
    import os as _os

    if device is None:
        # serve from the GPU when one is present: the index engines
        # deserialize onto this device and every /api/similar_tracks
        # scan runs there (a CPU default on an MI355X host measured
        # ~80x slower under load — profiles/r2_http_load*.log)
        import torch as _torch

        device = "cuda" if _torch.cuda.is_available() else "cpu"

    app = Flask("audiomuse_amd",
                static_folder=_os.path.join(_os.path.dirname(
                    _os.path.abspath(__file__)), "static"),
                static_url_path="/static")
    if getattr(C, "BEHIND_PROXY", False):
        app.wsgi_app = ProxyPrefixMiddleware(app.wsgi_app)
    state = AppState(db_url or C.DATABASE_URL, device=device)
    app.extensions["audiomuse"] = state
    # AUTH_ENABLED (reference PARAMETERS.md): config-level kill switch
    # for the auth layer; the explicit argument wins when given
    if auth_disabled is None:
        auth_disabled = not C.AUTH_ENABLED
    app.config["AUTH_DISABLED"] = auth_disabled

    from audiomuse_amd.web.api_auth import bp as auth_bp
    from audiomuse_amd.web.api_queries import bp as queries_bp
    from audiomuse_amd.web.api_tasks import bp as tasks_bp
    from audiomuse_amd.web.api_admin import bp as admin_bp
    from audiomuse_amd.web.api_chat import bp as chat_bp

    app.register_blueprint(auth_bp)
    app.register_blueprint(queries_bp)
    app.register_blueprint(tasks_bp)
    app.register_blueprint(chat_bp)
    app.register_blueprint(admin_bp)
    from audiomuse_amd.web.api_external import bp as external_bp

    app.register_blueprint(external_bp)

    from audiomuse_amd.web.auth import seed_admin_from_env

    # persisted config overrides (reference: config._apply_db_overrides)
    from audiomuse_amd import config as _C
    from audiomuse_amd.db.store import get_app_config as _gac

    _C.apply_db_overrides(_gac(state.conn()))

    with app.app_context():
        seed_admin_from_env(state.conn())
        # inline boot migrations (reference boot sequence)
        try:
            from audiomuse_amd.analysis.canonicalize import run_startup_migrations

            run_startup_migrations(state.conn())
        except Exception:  # noqa: BLE001 — boot must not die on migration
            import logging

            logging.getLogger(__name__).exception("startup migration failed")
        try:
            from audiomuse_amd.plugin import plugin_manager

            plugin_manager.load_from_db(state.conn())
        except Exception:  # noqa: BLE001 — plugin failures never block boot
            import logging

            logging.getLogger(__name__).exception("plugin boot load failed")

    # request-level logging (reference: note_request_start/end, app.py:194)
    import logging as _logging
    import time as _time

    from flask import g as _g
    from flask import request as _request

    _req_log = _logging.getLogger("audiomuse.requests")

    @app.before_request
    def _note_request_start():
        _g._t0 = _time.monotonic()

    @app.after_request
    def _note_request_end(resp):
        try:
            dt = (_time.monotonic() - getattr(_g, "_t0", _time.monotonic()))
            if _request.path.startswith(("/api/", "/chat/")):
                _req_log.info("%s %s -> %d in %.1f ms", _request.method,
                              _request.path, resp.status_code, dt * 1000)
        except Exception:
            pass
        return resp

    # background engine warm (reference: startup index load + map cache
    # build, app.py:1244): the first query should not pay the blob
    # deserialize + device upload (~8 s at 1M tracks on GPU)
    def _warm_engines():
        for name in (idx.AUDIO_INDEX, idx.CLAP_INDEX, idx.SONG_MAP):
            try:
                state.engine(name)
            except Exception:  # noqa: BLE001 — warm-up must never crash boot
                pass

    threading.Thread(target=_warm_engines, daemon=True,
                     name="audiomuse-engine-warm").start()

    # cron scheduler thread (reference: app.py cron loop)
    import threading as _threading

    if not app.config.get("TESTING"):
        _stop = _threading.Event()
        app.extensions["cron_stop"] = _stop

        def _cron_thread():
            from audiomuse_amd.db import connect as _connect
            from audiomuse_amd.utils.cron import cron_loop

            conn = _connect(state.db_url)
            cron_loop(conn, _stop)

        _threading.Thread(target=_cron_thread, daemon=True).start()

        # dashboard snapshot refresher (reference: app.py:1363-1394;
        # cadence DASHBOARD_REFRESH_SECONDS)
        def _dash_thread():
            from audiomuse_amd.analysis.maintenance import \
                refresh_dashboard_stats
            from audiomuse_amd.db import connect as _connect

            conn = _connect(state.db_url)
            while not _stop.wait(C.DASHBOARD_REFRESH_SECONDS):
                try:
                    refresh_dashboard_stats(conn)
                except Exception:  # noqa: BLE001 — stats must never kill it
                    pass

        _threading.Thread(target=_dash_thread, daemon=True).start()

    @app.get("/")
    def ui_index():  # L7: minimal first-party UI over the API
        return app.send_static_file("index.html")

    @app.get("/health")
    def health():  # reference: app.py:227
        try:
            state.conn().execute("SELECT 1")
            return jsonify({"status": "ok"})
        except Exception as exc:  # noqa: BLE001
            return jsonify({"status": "error", "detail": str(exc)}), 500

    @app.get("/api/spec")
    def api_spec():  # reference: Swagger at /apidocs (flasgger)
        out = []
        for rule in app.url_map.iter_rules():
            if rule.endpoint == "static":
                continue
            fn = app.view_functions[rule.endpoint]
            out.append({"path": str(rule),
                        "methods": sorted(m for m in rule.methods
                                          if m not in ("HEAD", "OPTIONS")),
                        "doc": (fn.__doc__ or "").strip().split("\n")[0]})
        return jsonify(sorted(out, key=lambda r: r["path"]))

    @app.get("/api/index_profile")
    def index_profile():  # reference: _log_startup_index_profile app.py:1244
        conn = state.conn()
        rows = conn.execute(
            "SELECT index_name, meta, n_parts, updated_at FROM ivf_dir"
        ).fetchall()
        out = []
        for r in rows:
            size = conn.execute(
                "SELECT COALESCE(SUM(LENGTH(blob)), 0) AS b FROM ivf_cell "
                "WHERE index_name=?", (r["index_name"],)).fetchone()
            out.append({"name": r["index_name"],
                        "meta": json.loads(r["meta"]),
                        "bytes": int(size["b"]),
                        "updated_at": r["updated_at"]})
        return jsonify(out)

    return app
