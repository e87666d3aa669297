"""Web API tests: auth barrier, task control, query endpoints over a
seeded catalogue (Flask test client)."""

import json

import numpy as np
import pytest

from audiomuse_amd.analysis.index import run_all_index_builds
from audiomuse_amd.db import connect
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.db.store import (save_clap_embedding,
                                    save_lyrics_embedding,
                                    save_track_analysis_and_embedding)
from audiomuse_amd.web.app import create_app


def _seed(conn, n=40):
    rng = np.random.default_rng(0)
    ids = []
    for i in range(n):
        iid = f"fp_4{'%050x' % i}"
        ids.append(iid)
        save_track_analysis_and_embedding(
            conn, iid, title=f"Song {i}", author=f"Artist {i % 6}",
            album=f"Album {i % 8}", tempo=90 + i, energy=(i % 10) / 10,
            key="C", scale="major", duration=180.0,
            mood_vector={"rock": (i % 3) / 2, "jazz": ((i + 1) % 3) / 2},
            other_features={"happy": 0.5},
            embedding=rng.standard_normal(200).astype(np.float32))
        save_clap_embedding(conn, iid,
                            rng.standard_normal(512).astype(np.float32))
        if i % 2 == 0:
            save_lyrics_embedding(conn, iid,
                                  rng.standard_normal(768).astype(np.float32),
                                  axis_scores={"love": 0.4})
    run_all_index_builds(conn)
    return ids


@pytest.fixture(scope="module")
def client_ids(tmp_path_factory):
    url = "sqlite:///" + str(tmp_path_factory.mktemp("web") / "web.db")
    conn = connect(url)
    init_db(conn)
    ids = _seed(conn)
    app = create_app(url, auth_disabled=True)
    app.testing = True
    with app.test_client() as client:
        yield client, ids
    conn.close()


def test_health(client_ids):
    client, _ = client_ids
    r = client.get("/health")
    assert r.status_code == 200 and r.json["status"] == "ok"


def test_similar_tracks(client_ids):
    client, ids = client_ids
    r = client.get(f"/api/similar_tracks?item_id={ids[0]}&n=5")
    assert r.status_code == 200
    body = r.json
    assert len(body) == 5
    assert all(b["item_id"] != ids[0] for b in body)
    assert all("title" in b and "author" in b for b in body)
    # unknown id -> 404
    assert client.get("/api/similar_tracks?item_id=nope").status_code == 404


def test_search_tracks(client_ids):
    client, _ = client_ids
    r = client.get("/api/search_tracks?q=song 1")
    assert r.status_code == 200 and len(r.json) >= 1


def test_path(client_ids):
    client, ids = client_ids
    r = client.get(f"/api/path?start={ids[0]}&end={ids[30]}&length=6")
    assert r.status_code == 200
    body = r.json
    assert body[0]["item_id"] == ids[0] and body[-1]["item_id"] == ids[30]


def test_alchemy(client_ids):
    client, ids = client_ids
    r = client.post("/api/alchemy", json={"add": [ids[1]], "n": 6})
    assert r.status_code == 200 and len(r.json) >= 1


def test_artist_similarity(client_ids):
    client, _ = client_ids
    r = client.get("/api/artist_similarity?artist=Artist 1&n=3")
    assert r.status_code == 200
    assert len(r.json) == 3
    assert client.get("/api/artist_similarity?artist=Unknown").status_code == 404


def test_text_searches(client_ids):
    client, _ = client_ids
    r = client.get("/api/clap_search?q=upbeat dance music&n=4")
    assert r.status_code == 200 and len(r.json) == 4
    r = client.get("/api/lyrics_search?q=love and heartbreak&n=4")
    assert r.status_code == 200 and len(r.json) == 4


def test_semgrove_and_hyperbolic(client_ids):
    client, ids = client_ids
    r = client.get(f"/api/semgrove?item_id={ids[0]}&n=4")
    assert r.status_code == 200 and len(r.json) >= 1
    r = client.get(f"/api/hyperbolic_similar?item_id={ids[0]}&n=4")
    assert r.status_code == 200 and len(r.json) == 4


def test_sonic_fingerprint_and_map(client_ids):
    client, ids = client_ids
    r = client.get(f"/api/sonic_fingerprint?item_id={ids[0]}&item_id={ids[1]}&n=5")
    assert r.status_code == 200 and len(r.json) == 5
    r = client.get("/api/map?percent=25")
    assert r.status_code == 200 and len(r.json) == 10  # 25% of 40
    assert {"item_id", "x", "y"} <= set(r.json[0])


def test_order_playlist_endpoint(client_ids):
    client, ids = client_ids
    r = client.post("/api/order_playlist", json={"item_ids": ids[:5]})
    assert r.status_code == 200 and sorted(r.json) == sorted(ids[:5])


def test_task_control_flow(client_ids):
    client, _ = client_ids
    r = client.post("/api/analysis/start", json={
        "server_type": "synthetic",
        "server_config": {"n_albums": 1, "tracks_per_album": 1}})
    assert r.status_code == 202
    tid = r.json["task_id"]
    # admission gate: second start conflicts
    r2 = client.post("/api/analysis/start", json={"server_type": "synthetic"})
    assert r2.status_code == 409
    # status + active list
    assert client.get(f"/api/task/{tid}").json["status"] == "PENDING"
    active = client.get("/api/active_tasks").json
    assert any(t["task_id"] == tid for t in active)
    # cancel recursively
    r3 = client.post(f"/api/task/{tid}/cancel")
    assert r3.json["cancelled"] == 1
    assert client.get(f"/api/task/{tid}").json["status"] == "REVOKED"


def test_chat_playlist_offline_planner(client_ids):
    client, _ = client_ids
    r = client.post("/chat/api/chatPlaylist",
                    json={"prompt": "15 songs of happy rock by Artist 1"})
    assert r.status_code == 200
    body = r.json
    assert body["hints"]["n"] == 15
    assert "rock" in body["hints"]["moods"]
    assert body["plan"] and all(c["tool"] in
                                ("seed_search", "text_match",
                                 "search_database", "knowledge_lookup")
                                for c in body["plan"])
    assert body["tracks"] and len(body["tracks"]) <= 15


def test_server_registry_crud(client_ids):
    client, _ = client_ids
    r = client.post("/api/servers", json={"server_id": "s1",
                                          "server_type": "synthetic"})
    assert r.status_code == 200
    assert any(s["server_id"] == "s1" for s in client.get("/api/servers").json)
    assert client.post("/api/servers",
                       json={"server_type": "bogus"}).status_code == 400
    assert client.delete("/api/servers/s1").json["deleted"] == 1


def test_auth_barrier():
    """Without auth_disabled: setup barrier, then login flow."""
    import tempfile

    with tempfile.TemporaryDirectory() as td:
        url = f"sqlite:///{td}/auth.db"
        conn = connect(url)
        init_db(conn)
        app = create_app(url, auth_disabled=False)
        app.testing = True
        with app.test_client() as client:
            # setup needed -> 403 on protected endpoints
            assert client.get("/api/active_tasks").status_code == 403
            assert client.get("/api/setup/status").json["setup_needed"]
            r = client.post("/api/setup/admin",
                            json={"username": "admin",
                                  "password": "longenough1"})
            assert r.status_code == 200
            # wrong password
            assert client.post("/api/login",
                               json={"username": "admin",
                                     "password": "bad"}).status_code == 401
            # login sets cookie; protected endpoint works
            r = client.post("/api/login", json={"username": "admin",
                                                "password": "longenough1"})
            assert r.status_code == 200
            assert client.get("/api/active_tasks").status_code == 200
            assert client.get("/api/me").json["user"] == "admin"
        conn.close()


def test_lyrics_axes_endpoint(client_ids):
    client, ids = client_ids
    r = client.get("/api/lyrics_axes?axis=love&n=5")
    assert r.status_code == 200 and len(r.json) == 5
    assert client.get("/api/lyrics_axes?axis=bogus").status_code == 400


def test_hyperbolic_tree_endpoints(client_ids):
    client, _ = client_ids
    # the seeding index rebuild persisted the tree already (the
    # run_all_index_builds hook); the inline endpoint rebuilds it
    r = client.post("/api/hyperbolic_tree/build")
    assert r.status_code == 200 and r.json["tracks"] > 0
    # root serves from the SKELETON: no full-tree warm happened
    status = client.get("/api/hyperbolic_tree/status").json
    assert status["skeleton_loaded"] and not status["full_loaded"]
    root = client.get("/api/hyperbolic_tree").json
    assert root["id"] == "root" and root["children_count"] >= 1
    mood_id = root["items"][0]["id"]
    folder = client.get(f"/api/hyperbolic_tree/node/{mood_id}").json
    assert folder["type"] == "folder" and not folder["leaf"]
    assert client.get("/api/hyperbolic_tree/status"
                      ).json["full_loaded"] is False  # still skeleton-only
    # a LEAF node lazily warms the full tree
    leaf_id = folder["items"][0]["id"]
    leaf = client.get(f"/api/hyperbolic_tree/node/{leaf_id}").json
    assert leaf["leaf"] and leaf["items"]
    assert all("item_id" in t and "radius" in t for t in leaf["items"])
    status = client.get("/api/hyperbolic_tree/status").json
    assert status["full_loaded"] and status["warm_seconds_left"] > 0
    assert client.get("/api/hyperbolic_tree/node/bogus").status_code == 404


def test_users_endpoint(client_ids):
    client, _ = client_ids
    r = client.get("/api/users")
    assert r.status_code == 200


def test_ui_served(client_ids):
    client, _ = client_ids
    r = client.get("/")
    assert r.status_code == 200
    assert b"AudioMuse-AMD" in r.data and b"app.js" in r.data


def test_config_override_round_trip(client_ids):
    from audiomuse_amd import config as C

    client, _ = client_ids
    original = C.IVF_RERANK_OVERFETCH
    try:
        r = client.post("/api/config", json={"IVF_RERANK_OVERFETCH": "7"})
        assert r.status_code == 200 and r.json["saved"] == 1
        assert C.IVF_RERANK_OVERFETCH == 7
        body = client.get("/api/config").json
        assert body["overrides"]["IVF_RERANK_OVERFETCH"] == "7"
        assert client.post("/api/config",
                           json={"lowercase": "x"}).status_code == 400
    finally:
        C.IVF_RERANK_OVERFETCH = original


def test_api_spec_lists_routes(client_ids):
    client, _ = client_ids
    spec = client.get("/api/spec").json
    paths = {r["path"] for r in spec}
    assert "/api/similar_tracks" in paths and "/chat/api/chatPlaylist" in paths
    assert all("methods" in r for r in spec)


def test_engine_cache_reloads_on_rebuild(tmp_path):
    """The web engine cache must pick up a rebuilt index (reference:
    listen_for_index_reloads, app.py:971)."""
    import numpy as np

    from audiomuse_amd.analysis.index import AUDIO_INDEX, build_audio_index
    from audiomuse_amd.web.app import AppState

    url = f"sqlite:///{tmp_path}/reload.db"
    conn = connect(url)
    init_db(conn)
    rng = np.random.default_rng(0)
    for i in range(10):
        save_track_analysis_and_embedding(
            conn, f"fp_4{'%050x' % i}", title=f"T{i}", author="A",
            embedding=rng.standard_normal(200).astype(np.float32))
    build_audio_index(conn)
    state = AppState(url)
    eng1 = state.engine(AUDIO_INDEX)
    assert eng1 is not None and eng1.index.n == 10
    assert state.engine(AUDIO_INDEX) is eng1          # cached
    # grow + rebuild -> stamp changes -> new engine object
    import time as _t

    _t.sleep(0.01)
    for i in range(10, 15):
        save_track_analysis_and_embedding(
            conn, f"fp_4{'%050x' % i}", title=f"T{i}", author="A",
            embedding=rng.standard_normal(200).astype(np.float32))
    build_audio_index(conn)
    eng2 = state.engine(AUDIO_INDEX)
    assert eng2 is not eng1 and eng2.index.n == 15
    conn.close()


def test_chat_stream_sse(client_ids):
    client, _ = client_ids
    r = client.post("/chat/api/chatPlaylistStream",
                    json={"prompt": "8 songs of jazz"})
    assert r.status_code == 200
    assert r.mimetype == "text/event-stream"
    body = r.get_data(as_text=True)
    assert "event: plan" in body and "event: playlist" in body


def test_suggested_queries(client_ids):
    client, _ = client_ids
    a = client.get("/api/clap_search/suggestions?n=5&seed=3").json
    b = client.get("/api/clap_search/suggestions?n=5&seed=3").json
    assert a == b and len(a) == 5 and len(set(a)) == 5


def test_index_refresh_endpoint(client_ids):
    client, _ = client_ids
    r = client.post("/api/index/refresh")
    assert r.status_code == 202 and "task_id" in r.json


def test_artist_map(client_ids):
    client, _ = client_ids
    r = client.get("/api/map?kind=artist")
    assert r.status_code == 200
    body = r.json
    assert len(body) == 6  # 6 artists seeded
    assert all("artist" in p and "x" in p and "y" in p for p in body)


def test_lyrics_axes_similar(client_ids):
    client, ids = client_ids
    # even-index tracks have axis profiles
    r = client.get(f"/api/lyrics_axes_similar?item_id={ids[0]}&n=5")
    assert r.status_code == 200 and len(r.json) >= 1
    assert all(b["item_id"] != ids[0] for b in r.json)
    # odd-index track has no profile -> 404
    assert client.get(
        f"/api/lyrics_axes_similar?item_id={ids[1]}").status_code == 404


def test_proxy_prefix_middleware(tmp_path, monkeypatch):
    from audiomuse_amd import config as C
    from audiomuse_amd.web.app import create_app

    monkeypatch.setattr(C, "BEHIND_PROXY", True)
    app = create_app(db_url=f"sqlite:///{tmp_path}/proxy.db",
                     auth_disabled=True)
    client = app.test_client()
    r = client.get("/am/health", headers={"X-Forwarded-Prefix": "/am"})
    assert r.status_code == 200 and r.json["status"] == "ok"
    # without the header the prefixed path does not exist
    assert client.get("/am/health").status_code == 404


def test_proxy_prefix_headers_ignored_by_default(client_ids):
    # Not behind a proxy (default): forwarded headers are untrusted and
    # must not rewrite SCRIPT_NAME/scheme.
    client, _ = client_ids
    assert client.get("/am/health",
                      headers={"X-Forwarded-Prefix": "/am"}).status_code == 404
    r = client.get("/health", headers={"X-Forwarded-Prefix": "/am"})
    assert r.status_code == 200


def test_external_api(client_ids):
    client, ids = client_ids
    # canonical id passes through
    r = client.get(f"/external/get_score?id={ids[3]}")
    assert r.status_code == 200 and r.json["title"] == "Song 3"
    r = client.get(f"/external/get_embedding?id={ids[3]}")
    assert r.status_code == 200 and len(r.json["embedding"]) == 200
    assert isinstance(r.json["embedding"][0], float)
    # unknown id -> 404; missing -> 400
    assert client.get("/external/get_score?id=zzz").status_code == 404
    assert client.get("/external/get_score").status_code == 400
    # unified autocomplete
    r = client.get("/external/search?q=Song 1")
    assert r.status_code == 200 and len(r.json) >= 1
    # legacy title/artist params
    r = client.get("/external/search?title=Song&artist=Artist 2")
    assert r.status_code == 200
    assert all(b["author"] == "Artist 2" for b in r.json)


def test_external_provider_id_resolution(client_ids):
    client, ids = client_ids
    from audiomuse_amd.db import write_txn

    state = None
    # map a provider id onto ids[5] through the app's own connection
    import flask

    app = client.application
    conn = app.extensions["audiomuse"].conn()
    with write_txn(conn):
        conn.execute(
            "INSERT OR REPLACE INTO track_server_map "
            "(provider_id, server_id, item_id, title, author) "
            "VALUES ('prov-42', 'srv1', ?, 'Song 5', 'Artist 5')", (ids[5],))
    r = client.get("/external/get_score?id=prov-42&server=srv1")
    assert r.status_code == 200 and r.json["item_id"] == ids[5]
    r = client.get("/external/get_score?id=prov-42")  # default: any server
    assert r.status_code == 200 and r.json["item_id"] == ids[5]


def test_map_percent_buckets_and_gzip(client_ids):
    client, _ = client_ids
    full = client.get("/api/map?percent=100").json
    half = client.get("/api/map?percent=50").json
    assert len(half) == len(full) // 2
    assert {"item_id", "x", "y", "title", "author", "mood"} <= set(full[0])
    # deterministic sample
    again = client.get("/api/map?percent=50").json
    assert again == half
    assert client.get("/api/map?percent=33").status_code == 400
    r = client.get("/api/map", headers={"Accept-Encoding": "gzip"})
    assert r.headers.get("Content-Encoding") == "gzip"
    import gzip as _gz
    import json as _json
    assert _json.loads(_gz.decompress(r.data)) == full


def test_create_playlist_on_server(client_ids):
    client, ids = client_ids
    from audiomuse_amd.db import write_txn

    # no server registered yet -> 404
    r = client.post("/api/create_playlist",
                    json={"name": "Mix", "item_ids": ids[:3]})
    assert r.status_code == 404
    # register a synthetic server + map 2 of 3 ids onto it
    assert client.post("/api/servers", json={
        "server_id": "syn1", "server_type": "synthetic"}).status_code == 200
    conn = client.application.extensions["audiomuse"].conn()
    with write_txn(conn):
        for k, iid in enumerate(ids[:2]):
            conn.execute(
                "INSERT OR REPLACE INTO track_server_map "
                "(provider_id, server_id, item_id) VALUES (?, 'syn1', ?)",
                (f"a0-t{k}", iid))
    r = client.post("/api/create_playlist",
                    json={"name": "Mix", "item_ids": ids[:3],
                          "server_id": "syn1"})
    assert r.status_code == 201, r.json
    assert r.json["created"] == 2 and r.json["missing"] == 1
    assert r.json["playlist_id"]
    # missing args -> 400
    assert client.post("/api/create_playlist",
                       json={"name": "x"}).status_code == 400


def test_clap_warmup_lifecycle(client_ids, monkeypatch):
    client, _ = client_ids
    r = client.get("/api/clap/warmup/status")
    assert r.status_code == 200
    r = client.post("/api/clap/warmup")
    assert r.status_code == 200 and r.json["loaded"]
    assert r.json["seconds"] > 0
    st = client.get("/api/clap/warmup/status").json
    assert st["loaded"] and st["seconds"] > 0
    # countdown expiry unloads the model
    lc = client.application.extensions["clap_text_lc"]
    monkeypatch.setattr(lc, "_last", lc._time() - 10_000)
    st = client.get("/api/clap/warmup/status").json
    assert not st["loaded"] and st["seconds"] == 0  # property
    # a search reloads it transparently
    assert client.get("/api/clap_search?q=rainy night").status_code == 200
    assert client.get("/api/clap/warmup/status").json["loaded"]


def test_server_availability_mask(client_ids):
    """reference ALGORITHM.md 4.2: ?server= drops unmapped tracks and
    attaches provider ids."""
    client, ids = client_ids
    from audiomuse_amd.db import write_txn

    conn = client.application.extensions["audiomuse"].conn()
    with write_txn(conn):
        for k, iid in enumerate(ids[2:8]):
            conn.execute(
                "INSERT OR REPLACE INTO track_server_map "
                "(provider_id, server_id, item_id) VALUES (?, 'navi', ?)",
                (f"nv-{k}", iid))
    r = client.get(f"/api/similar_tracks?item_id={ids[0]}&n=4&server=navi")
    assert r.status_code == 200
    assert 0 < len(r.json) <= 4
    mapped = set(ids[2:8])
    for b in r.json:
        assert b["item_id"] in mapped and b["provider_id"].startswith("nv-")
    # without the scope, unmapped tracks appear
    r2 = client.get(f"/api/similar_tracks?item_id={ids[0]}&n=10")
    assert any(b["item_id"] not in mapped for b in r2.json)


def test_map_per_server_bucket(client_ids):
    client, ids = client_ids
    from audiomuse_amd.db import write_txn

    conn = client.application.extensions["audiomuse"].conn()
    with write_txn(conn):
        for k, iid in enumerate(ids[10:16]):
            conn.execute(
                "INSERT OR REPLACE INTO track_server_map "
                "(provider_id, server_id, item_id) VALUES (?, 'mapsrv', ?)",
                (f"ms-{k}", iid))
    pts = client.get("/api/map?server=mapsrv").json
    assert 0 < len(pts) <= 6
    assert all("provider_id" in p for p in pts)
    mapped = set(ids[10:16])
    assert all(p["item_id"] in mapped for p in pts)


# -- round-2 parity endpoints (reference route sweep) -----------------------

def test_search_artists_and_tracks(client_ids):
    client, _ = client_ids
    r = client.get("/api/search_artists?q=artist 1")
    assert r.status_code == 200 and len(r.json) >= 1
    assert r.json[0]["n_tracks"] >= 1
    r = client.get("/api/artist_tracks?artist=Artist 1")
    assert r.status_code == 200 and len(r.json) >= 1
    assert client.get("/api/artist_tracks").status_code == 400


def test_track_detail(client_ids):
    client, ids = client_ids
    r = client.get(f"/api/track?item_id={ids[3]}")
    assert r.status_code == 200
    assert r.json["title"] == "Song 3"
    assert isinstance(r.json["mood_vector"], dict)
    assert isinstance(r.json["servers"], list)
    assert client.get("/api/track?item_id=nope").status_code == 404


def test_max_distance(client_ids):
    client, ids = client_ids
    r = client.get(f"/api/max_distance?item_id={ids[0]}")
    assert r.status_code == 200
    assert r.json["max_distance"] > 0
    assert r.json["farthest_item_id"] in ids
    assert client.get("/api/max_distance?item_id=nope").status_code == 404


def test_mood_centroids(client_ids):
    client, _ = client_ids
    r = client.get("/api/mood_centroids")
    assert r.status_code == 200
    moods = {m["mood"] for m in r.json}
    assert moods <= {"rock", "jazz"} and len(r.json) >= 1
    assert all(m["count"] > 0 for m in r.json)


def test_playlists_listing_and_search(client_ids):
    client, ids = client_ids
    from audiomuse_amd.db import write_txn
    state = client.application.extensions["audiomuse"]
    conn = state.conn()
    with write_txn(conn):
        conn.execute(
            "INSERT INTO playlist (name, item_ids, kind) VALUES (?,?,?)",
            ("Morning Mix", json.dumps(ids[:5]), "clustering"))
    r = client.get("/api/playlists")
    assert r.status_code == 200 and len(r.json) >= 1
    assert r.json[0]["n_tracks"] == 5 and "item_ids" not in r.json[0]
    r = client.get("/api/playlists?include_tracks=1")
    assert r.json[0]["item_ids"] == ids[:5]
    r = client.get("/api/search_playlists?q=morning")
    assert len(r.json) == 1 and r.json[0]["name"] == "Morning Mix"
    assert client.get("/api/search_playlists?q=zzz").json == []


def test_cleaning_start_and_last_task(client_ids):
    client, _ = client_ids
    r = client.post("/api/cleaning/start", json={"delete": False})
    assert r.status_code == 202
    # admission gate: a second start while pending conflicts
    assert client.post("/api/cleaning/start", json={}).status_code == 409
    r = client.get("/api/last_task?task_type=clean_orphans")
    assert r.status_code == 200 and r.json["task_type"] == "clean_orphans"
    # bulk cancel by prefix clears it
    r = client.post("/api/cancel_all/clean")
    assert r.status_code == 200 and r.json["cancelled"] >= 1


def test_server_test_probe_and_sweep(client_ids):
    client, _ = client_ids
    r = client.post("/api/servers/test",
                    json={"server_type": "synthetic",
                          "server_config": {"n_albums": 2}})
    assert r.status_code == 200 and r.json["ok"]
    r = client.post("/api/servers/test", json={"server_type": "bogus"})
    assert r.status_code == 502
    # register a synthetic server, then sweep it and sync all
    assert client.post("/api/servers", json={
        "server_id": "s1", "server_type": "synthetic",
        "config": {"n_albums": 2}}).status_code == 200
    r = client.get("/api/servers/s1/libraries")
    assert r.status_code == 200 and isinstance(r.json, list)
    r = client.post("/api/servers/s1/sweep")
    assert r.status_code == 202
    assert client.post("/api/servers/s1/sweep").status_code == 409
    client.post("/api/cancel_all/multiserver_sync")
    r = client.post("/api/sync")
    assert r.status_code == 202 and len(r.json["task_ids"]) >= 1
    client.post("/api/cancel_all/multiserver_sync")
    assert client.get("/api/servers/nope/libraries").status_code == 404


def test_config_defaults_and_family_stats(client_ids):
    client, _ = client_ids
    r = client.get("/api/config/defaults")
    assert r.status_code == 200 and "CLAP_GPU_BATCH" in r.json["defaults"]
    assert not any("PASSWORD" in k for k in r.json["defaults"])
    for fam in ("clap", "lyrics", "semgrove"):
        r = client.get(f"/api/{fam}/stats")
        assert r.status_code == 200, fam
        assert r.json["rows"] > 0 and r.json["index_loaded"], fam
        assert r.json["indexed"] > 0


def test_lyrics_warmup_and_map_cache(client_ids):
    client, _ = client_ids
    r = client.post("/api/lyrics/warmup")
    assert r.status_code == 200 and r.json["loaded"]
    assert client.get("/api/lyrics/warmup/status").json["loaded"]
    client.get("/api/map")                      # populate a bucket
    r = client.get("/api/map_cache_status")
    assert r.status_code == 200 and r.json["song_map_built"]
    assert r.json["buckets_cached"] >= 1
    r = client.post("/api/rebuild_map_cache")
    assert r.status_code == 202 and r.json["task_id"]
    assert client.get("/api/map_cache_status").json["buckets_cached"] == 0
    client.post("/api/cancel_all/rebuild_indexes")


def test_plex_pin_flow(client_ids, monkeypatch):
    """Plex PIN create/poll proxy (reference app_setup.py:926-1030) —
    plex.tv stubbed at the MediaHttp layer."""
    client, _ = client_ids
    from audiomuse_amd.mediaserver import http as mhttp

    class FakeResp:
        def __init__(self, body):
            self._body = body

        def json(self):
            return self._body

    calls = {}

    def fake_request(self, method, url, **kw):
        calls["last"] = (method, url,
                         kw.get("headers", {}).get("X-Plex-Client-Identifier"))
        if method == "POST":
            return FakeResp({"id": 777, "code": "ABCD"})
        return FakeResp({"id": 777, "authToken": "tok-xyz"})

    monkeypatch.setattr(mhttp.MediaHttp, "request", fake_request)
    r = client.post("/api/setup/plex/pin", json={"client_id": "cid-1"})
    assert r.status_code == 200
    assert r.json["code"] == "ABCD" and r.json["id"] == 777
    assert calls["last"][2] == "cid-1"
    r = client.get("/api/setup/plex/pin/777?client_id=cid-1")
    assert r.status_code == 200
    assert r.json["claimed"] and r.json["auth_token"] == "tok-xyz"


def test_family_cache_refresh_routes(client_ids):
    client, _ = client_ids
    for fam in ("clap", "lyrics", "semgrove"):
        r = client.post(f"/api/{fam}/cache/refresh")
        assert r.status_code == 202 and r.json["task_id"], fam
    client.post("/api/cancel_all/refresh_indexes")


def test_setup_provider_libraries_and_lyrics_api(client_ids, monkeypatch):
    client, _ = client_ids
    r = client.post("/api/setup/providers/libraries",
                    json={"server_type": "synthetic",
                          "server_config": {"n_albums": 1}})
    assert r.status_code == 200 and isinstance(r.json, list)
    assert client.post("/api/setup/providers/libraries",
                       json={"server_type": "bogus"}).status_code == 502
    # lyrics-api analyze with a stubbed fetch
    from audiomuse_amd import config as C2
    monkeypatch.setattr(C2, "LYRICS_API_ENABLE", True)
    monkeypatch.setattr(C2, "LYRICS_API_1_URL_TEMPLATE",
                        "http://lyrics.example/{artist}/{title}")
    import audiomuse_amd.engines.lyrics as lyr

    def fake_fetch(title, artist, http_get=None):
        return f"la la {title} by {artist}"

    monkeypatch.setattr(lyr, "fetch_external_lyrics", fake_fetch)
    r = client.post("/api/setup/lyrics-api/analyze",
                    json={"artist": "A", "title": "T"})
    assert r.status_code == 200 and r.json["found"]
    assert "la la T" in r.json["preview"]
    assert client.post("/api/setup/lyrics-api/analyze",
                       json={}).status_code == 400
