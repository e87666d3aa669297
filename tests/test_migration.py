"""Provider-migration wizard (analysis/migration.py + /api/migration/*):
probe, path-format detection, tiered preview, transactional rewrite,
restart handshake — end-to-end between two synthetic providers
(reference: app_provider_migration.py + provider_migration_tasks.py)."""

import json

import pytest

from audiomuse_amd.analysis.migration import (build_match_preview,
                                              detect_path_format,
                                              execute_migration,
                                              probe_server, propose_path_rule,
                                              rewrite_path)
from audiomuse_amd.db import connect, write_txn
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.db.store import save_track_analysis_and_embedding
from audiomuse_amd.mediaserver import make_provider

import numpy as np

SRC_CFG = {"n_albums": 3, "tracks_per_album": 4, "path_prefix": "/music"}
# same library, other server: new ids + new mount point
DST_CFG = {"n_albums": 3, "tracks_per_album": 4,
           "path_prefix": "/srv/media/library", "id_prefix": "x"}


def _seed_source(conn, server_id="default"):
    """Analyze-equivalent seeding: catalogue rows + source mappings."""
    provider = make_provider("synthetic", **SRC_CFG)
    rng = np.random.default_rng(0)
    tracks = provider.get_all_songs()
    for i, t in enumerate(tracks):
        item_id = f"fp_4{i:047x}"
        save_track_analysis_and_embedding(
            conn, item_id, title=t.title, author=t.author, album=t.album,
            tempo=120.0, key="C", scale="major",
            mood_vector={"rock": 0.8}, other_features={}, energy=0.5,
            duration=t.duration,
            embedding=rng.standard_normal(8).astype(np.float32))
        with write_txn(conn):
            conn.execute(
                """INSERT INTO track_server_map (provider_id, server_id,
                       item_id, title, author, album, file_path)
                   VALUES (?,?,?,?,?,?,?)""",
                (t.provider_id, server_id, item_id, t.title, t.author,
                 t.album, t.file_path))
    return tracks


@pytest.fixture
def db(tmp_db_url):
    # both backends: the session table round-trips zlib BLOBs and the
    # decisions JSON through sqlite AND the PG wire driver
    conn = connect(tmp_db_url)
    init_db(conn)
    yield conn, tmp_db_url
    conn.close()


def test_probe_reports_libraries_and_path_format():
    out = probe_server("synthetic", DST_CFG)
    assert out["reachable"] and out["libraries"][0]["id"] == "lib1"
    assert out["path_format"]["prefix"].startswith("/srv/media/library")
    assert out["sample_paths"]


def test_detect_path_format():
    fmt = detect_path_format([
        "/srv/media/library/A/Al 1/1.wav",
        "/srv/media/library/A/Al 1/2.wav",
        "/srv/media/library/B/Al 2/1.wav"])
    assert fmt["prefix"].startswith("/srv/media/library")
    assert fmt["separator"] == "/"
    win = detect_path_format([r"C:\Music\A\1.mp3", r"C:\Music\B\2.mp3"])
    assert win["separator"] == "\\" and win["prefix"].startswith("C:")


def test_path_rule_and_rewrite():
    rule = propose_path_rule(
        ["/srv/media/library/A/x.wav", "/srv/media/library/B/y.wav"],
        ["/music/A/x.wav", "/music/B/y.wav"])
    out = rewrite_path("/srv/media/library/A/x.wav", rule)
    assert out == "/music/A/x.wav"


def test_preview_tiers_and_no_writes(db):
    conn, _ = db
    _seed_source(conn)
    dst = make_provider("synthetic", **DST_CFG)
    tracks = dst.get_all_songs()
    rule = propose_path_rule(
        [t.file_path for t in tracks],
        [r["file_path"] for r in conn.execute(
            "SELECT file_path FROM track_server_map")])
    before = conn.execute(
        "SELECT COUNT(*) AS n FROM track_server_map").fetchone()["n"]
    preview = build_match_preview(conn, tracks, "default", path_rule=rule)
    assert preview["matched"] == len(tracks)           # everything matches
    assert preview["tiers"].get("path", 0) > 0         # via the path rule
    after = conn.execute(
        "SELECT COUNT(*) AS n FROM track_server_map").fetchone()["n"]
    assert after == before                              # preview is read-only


def test_execute_rewrites_transactionally_and_requests_restart(db):
    conn, _ = db
    _seed_source(conn)
    dst = make_provider("synthetic", **DST_CFG)
    tracks = dst.get_all_songs()
    rule = propose_path_rule(
        [t.file_path for t in tracks],
        [r["file_path"] for r in conn.execute(
            "SELECT file_path FROM track_server_map")])
    preview = build_match_preview(conn, tracks, "default", path_rule=rule)
    result = execute_migration(conn, preview["matches"], "new-server",
                               source_server_id="default",
                               remove_source=True, min_match_ratio=0.5,
                               preview=preview)
    assert result["applied"] and result["written"] == len(tracks)
    assert result["removed"] > 0
    # target mappings resolve to the SAME canonical ids
    rows = conn.execute(
        "SELECT provider_id, item_id FROM track_server_map "
        "WHERE server_id='new-server'").fetchall()
    assert len(rows) == len(tracks)
    assert all(r["provider_id"].startswith("x") for r in rows)
    # catalogue untouched
    n_cat = conn.execute("SELECT COUNT(*) FROM score").fetchone()[0]
    assert n_cat == len(tracks)
    # restart handshake published with a live window
    req = conn.execute("SELECT * FROM control_request").fetchone()
    assert req["action"] == "restart"


def test_execute_refuses_below_match_ratio(db):
    conn, _ = db
    _seed_source(conn)
    preview = {"match_ratio": 0.2}
    out = execute_migration(conn, [], "t", min_match_ratio=0.5,
                            preview=preview)
    assert not out["applied"] and "below" in out["reason"]


def test_wizard_task_preview_then_apply(db):
    conn, url = db
    _seed_source(conn)
    from audiomuse_amd.taskqueue import SUCCESS, enqueue, task_row
    from audiomuse_amd.taskqueue.worker import Worker

    # preview run (apply=false)
    tid = enqueue(conn, "provider_migration", {
        "server_type": "synthetic", "server_config": DST_CFG,
        "apply": False})
    Worker(db_url=url, max_jobs=1).run_forever(idle_timeout=3.0)
    row = task_row(conn, tid)
    assert row["status"] == SUCCESS
    result = json.loads(row["result"])
    assert result["stage"] == "preview" and result["match_ratio"] == 1.0
    assert conn.execute("SELECT COUNT(*) AS n FROM track_server_map "
                        "WHERE server_id='migrated'").fetchone()["n"] == 0

    # apply run
    tid = enqueue(conn, "provider_migration", {
        "server_type": "synthetic", "server_config": DST_CFG,
        "apply": True, "remove_source": False})
    Worker(db_url=url, max_jobs=1).run_forever(idle_timeout=3.0)
    result = json.loads(task_row(conn, tid)["result"])
    assert result["stage"] == "done" and result["applied"]
    n = conn.execute("SELECT COUNT(*) AS n FROM track_server_map "
                     "WHERE server_id='migrated'").fetchone()["n"]
    assert n == result["matched"] > 0


def test_wizard_api_flow(db):
    conn, url = db
    _seed_source(conn)
    from audiomuse_amd.web.app import create_app

    app = create_app(url, auth_disabled=True)
    app.testing = True
    client = app.test_client()

    r = client.post("/api/migration/probe",
                    json={"server_type": "synthetic",
                          "server_config": DST_CFG})
    assert r.status_code == 200 and r.json["reachable"]

    r = client.post("/api/migration/preview",
                    json={"server_type": "synthetic",
                          "server_config": DST_CFG})
    assert r.status_code == 200
    assert r.json["match_ratio"] == 1.0 and r.json["unmatched"] == []

    r = client.post("/api/migration/start",
                    json={"server_type": "synthetic",
                          "server_config": DST_CFG, "apply": True})
    assert r.status_code == 202
    tid = r.json["task_id"]
    from audiomuse_amd.taskqueue.worker import Worker
    Worker(db_url=url, max_jobs=1).run_forever(idle_timeout=3.0)
    r = client.get(f"/api/migration/status/{tid}")
    assert r.status_code == 200 and r.json["status"] == "SUCCESS"
    assert r.json["result"]["stage"] == "done"


# -- per-album review sessions (analysis/migration_session.py) --------------

def test_session_lifecycle_with_decisions(db):
    """start -> dry-run -> matched-albums -> skip + manual map ->
    re-dry-run -> finalize -> execute; decisions change the report."""
    from audiomuse_amd.analysis import migration_session as ms

    conn, _ = db
    _seed_source(conn)
    out = ms.create_session(conn, "synthetic", DST_CFG)
    sid = out["session_id"]
    assert out["target_tracks"] == 12 and out["libraries"]

    rep = ms.run_dry_run(conn, sid)
    assert rep["match_ratio"] > 0.9
    albums = ms.matched_albums(conn, sid)
    assert albums and all(a["complete"] for a in albums)
    first_album = albums[0]["album"]

    # skip one album -> its matches disappear from the next dry run
    ms.set_decision(conn, sid, first_album, "skip")
    rep2 = ms.run_dry_run(conn, sid)
    assert rep2["matched"] < rep["matched"]
    albums2 = ms.matched_albums(conn, sid)
    skipped = [a for a in albums2 if a["album"] == first_album][0]
    assert skipped["decision"] == "skip" and skipped["matched"] == 0

    # manual map: re-attach the album by explicit target choice
    hits = ms.search_albums(conn, sid, first_album[:4])
    assert any(h["album"] == first_album for h in hits)
    ms.set_decision(conn, sid, first_album, "map",
                    target_album=first_album)
    rep3 = ms.run_dry_run(conn, sid)
    assert rep3["matched"] == rep["matched"]
    assert rep3["tiers"].get("manual", 0) > 0

    full = ms.dry_run_report(conn, sid)
    assert any(m["tier"] == "manual" for m in full["matches"])

    fin = ms.finalize(conn, sid)
    assert fin["status"] == "finalized"
    res = ms.execute_session(conn, sid, "migrated")
    assert res["applied"] and res["written"] == rep["matched"]
    n = conn.execute("SELECT COUNT(*) AS n FROM track_server_map "
                     "WHERE server_id='migrated'").fetchone()["n"]
    assert n == rep["matched"]
    assert ms.get_session(conn, sid)["status"] == "executed"


def test_session_gates(db):
    from audiomuse_amd.analysis import migration_session as ms

    conn, _ = db
    _seed_source(conn)
    sid = ms.create_session(conn, "synthetic", DST_CFG)["session_id"]
    # execute before finalize refuses
    assert ms.execute_session(conn, sid, "m2") is None
    # finalize before dry run refuses
    assert ms.finalize(conn, sid) is None
    # a decision after dry-run reopens the session (stale report gate)
    ms.run_dry_run(conn, sid)
    ms.set_decision(conn, sid, "whatever", "skip")
    assert ms.get_session(conn, sid)["status"] == "open"
    assert ms.finalize(conn, sid) is None
    # discard
    assert ms.discard_session(conn, sid)
    assert ms.run_dry_run(conn, sid) is None


def test_session_endpoints(db, monkeypatch):
    from audiomuse_amd.web.app import create_app

    conn, url = db
    _seed_source(conn)
    app = create_app(url, auth_disabled=True)
    app.testing = True
    with app.test_client() as client:
        r = client.post("/api/migration/session/start",
                        json={"server_type": "synthetic",
                              "server_config": DST_CFG})
        assert r.status_code == 201
        sid = r.json["session_id"]
        assert client.post("/api/migration/dry-run",
                           json={"session_id": sid}).status_code == 200
        r = client.get(f"/api/migration/matched-albums/{sid}")
        assert r.status_code == 200 and len(r.json) >= 3
        album = r.json[0]["album"]
        assert client.post("/api/migration/skip-album",
                           json={"session_id": sid,
                                 "album": album}).status_code == 200
        assert client.post("/api/migration/search-albums",
                           json={"session_id": sid, "q": ""}).json
        client.post("/api/migration/dry-run", json={"session_id": sid})
        assert client.post("/api/migration/finalize-dry-run",
                           json={"session_id": sid}).status_code == 200
        r = client.post("/api/migration/execute",
                        json={"session_id": sid,
                              "target_server_id": "migrated"})
        assert r.status_code == 200 and r.json["applied"]
        r = client.get(f"/api/migration/session/{sid}")
        assert r.json["status"] == "executed"
        assert client.delete(
            f"/api/migration/session/{sid}").status_code == 409
