"""Chromaprint analog, cron scheduler, plugins, admin endpoints."""

import io
import json
import time
import zipfile

import numpy as np
import pytest
import torch

from audiomuse_amd.engines import chromaprint
from audiomuse_amd.ops.audio_io import synthetic_track
from audiomuse_amd.plugin.manager import HookRegistry, PluginManager
from audiomuse_amd.utils.cron import cron_matches, minute_key, run_due_cron_jobs


def test_chromaprint_self_and_noise_agree():
    a = synthetic_track(1, seconds=8.0, sr=22050)
    fp_a = chromaprint.compute(a, 22050)
    assert fp_a
    assert chromaprint.bit_match_ratio(fp_a, fp_a) > 0.999
    noisy = a + torch.randn_like(a) * 0.005
    fp_n = chromaprint.compute(noisy, 22050)
    assert chromaprint.bit_match_ratio(fp_a, fp_n) > 0.8


def test_chromaprint_different_tracks_disagree():
    fa = chromaprint.compute(synthetic_track(1, 8.0, 22050), 22050)
    fb = chromaprint.compute(synthetic_track(99, 8.0, 22050), 22050)
    assert chromaprint.bit_match_ratio(fa, fb) < 0.8
    assert not chromaprint.chromaprints_agree(fa, fb)
    assert chromaprint.bit_match_ratio(fa, b"") == 0.0


def test_cron_matching():
    # Wed 2026-01-07 03:05 local
    t = time.mktime((2026, 1, 7, 3, 5, 0, 0, 0, -1))
    assert cron_matches("5 3 * * *", t)
    assert cron_matches("*/5 * * * *", t)
    assert not cron_matches("6 3 * * *", t)
    assert cron_matches("5 3 7 1 *", t)
    assert cron_matches("5 3 * * 3", t)        # Wednesday = 3
    assert not cron_matches("5 3 * * 0", t)
    assert not cron_matches("bogus", t)


def test_cron_minute_claim_single_winner(tmp_db_url):
    from audiomuse_amd.db import connect, write_txn
    from audiomuse_amd.db.schema import init_db

    conn = connect(tmp_db_url)
    init_db(conn)
    with write_txn(conn):
        conn.execute(
            "INSERT INTO cron (name, schedule, task_type, payload, enabled) "
            "VALUES ('x', '* * * * *', 'rebuild_indexes', '{}', 1)")
    first = run_due_cron_jobs(conn)
    second = run_due_cron_jobs(conn)           # same minute: no double claim
    assert len(first) == 1 and second == []
    conn.close()


def _plugin_zip(body: str) -> bytes:
    buf = io.BytesIO()
    with zipfile.ZipFile(buf, "w") as zf:
        zf.writestr("plugin.py", body)
    return buf.getvalue()


def test_plugin_load_and_hook():
    pm = PluginManager(HookRegistry())
    calls = []
    blob = _plugin_zip(
        "def register(api):\n"
        "    api.on_song_analyzed(lambda item_id, a: a.setdefault('tag', item_id))\n"
        "    api.add_cron_task('0 3 * * *', 'rebuild_indexes')\n")
    api = pm.load_zip("demo", blob)
    assert api.cron_tasks[0]["task_type"] == "rebuild_indexes"
    analysis = {}
    pm.fire_song_analyzed("song1", analysis)
    assert analysis["tag"] == "song1"


def test_plugin_zip_traversal_rejected():
    buf = io.BytesIO()
    with zipfile.ZipFile(buf, "w") as zf:
        zf.writestr("../evil.py", "x = 1")
    with pytest.raises(ValueError):
        PluginManager(HookRegistry()).load_zip("bad", buf.getvalue())


def test_plugin_hook_errors_contained():
    pm = PluginManager(HookRegistry())
    blob = _plugin_zip(
        "def register(api):\n"
        "    api.on_song_analyzed(lambda *a: 1/0)\n")
    pm.load_zip("crashy", blob)
    pm.fire_song_analyzed("s", {})   # must not raise


@pytest.fixture
def admin_client(tmp_path):
    from audiomuse_amd.db import connect
    from audiomuse_amd.db.schema import init_db
    from audiomuse_amd.db.store import save_track_analysis_and_embedding
    from audiomuse_amd.analysis.index import run_all_index_builds
    from audiomuse_amd.web.app import create_app

    url = f"sqlite:///{tmp_path}/admin.db"
    conn = connect(url)
    init_db(conn)
    rng = np.random.default_rng(0)
    ids = [f"fp_4{'%050x' % i}" for i in range(20)]
    for i, iid in enumerate(ids):
        save_track_analysis_and_embedding(
            conn, iid, title=f"T{i}", author=f"A{i % 3}",
            embedding=rng.standard_normal(200).astype(np.float32))
    run_all_index_builds(conn)
    app = create_app(url, auth_disabled=True)
    app.testing = True
    with app.test_client() as client:
        yield client, ids
    conn.close()


def test_dashboard_endpoint(admin_client):
    client, _ = admin_client
    r = client.get("/api/dashboard")
    assert r.status_code == 200 and r.json["tracks"] == 20


def test_anchor_crud(admin_client):
    client, ids = admin_client
    r = client.post("/api/alchemy/anchors",
                    json={"name": "favs", "item_ids": ids[:3]})
    assert r.status_code == 200 and r.json["dim"] == 200
    assert "favs" in client.get("/api/alchemy/anchors").json
    assert client.delete("/api/alchemy/anchors/favs").json["deleted"] == 1


def test_radio_crud(admin_client):
    client, ids = admin_client
    r = client.post("/api/alchemy/radios",
                    json={"name": "morning",
                          "definition": {"add": ids[:2], "temperature": 0.3}})
    assert r.status_code == 200
    radios = client.get("/api/alchemy/radios").json
    assert radios[0]["definition"]["temperature"] == 0.3


def test_backup_endpoint(admin_client):
    client, _ = admin_client
    r = client.get("/api/backup")
    assert r.status_code == 200
    assert r.data.startswith(b"SQLite format 3")


def test_restore_rejects_garbage(admin_client):
    client, _ = admin_client
    r = client.post("/api/restore", data=b"not a database")
    assert r.status_code == 400


def test_anchor_in_alchemy_mix(admin_client):
    client, ids = admin_client
    client.post("/api/alchemy/anchors", json={"name": "mix1",
                                              "item_ids": ids[:4]})
    r = client.post("/api/alchemy", json={"add": ["anchor:mix1"], "n": 5})
    assert r.status_code == 200 and len(r.json) == 5


def test_radio_play(admin_client):
    client, ids = admin_client
    client.post("/api/alchemy/radios",
                json={"name": "r1", "definition": {"add": [ids[0]], "n": 4}})
    r = client.post("/api/alchemy/radios/r1/play")
    assert r.status_code == 200 and len(r.json) == 4
    assert client.post("/api/alchemy/radios/nope/play").status_code == 404


def test_plugin_task_handler_and_cron_sync(tmp_db_url):
    """Plugin-contributed task types run through the real queue, and the
    plugin's cron tasks land in (and prune from) the cron table."""
    from audiomuse_amd.db import connect
    from audiomuse_amd.db.schema import init_db
    from audiomuse_amd.taskqueue import SUCCESS, enqueue, task_row
    from audiomuse_amd.taskqueue.worker import Worker, get_handler

    pm = PluginManager(HookRegistry())
    blob = _plugin_zip(
        "def register(api):\n"
        "    api.add_task_handler('echo',\n"
        "        lambda ctx, payload: {'echoed': payload['v']})\n"
        "    api.add_cron_task('*/5 * * * *', 'plugin.demo.echo',\n"
        "                      {'v': 'cron'})\n")
    pm.load_zip("demo", blob)
    assert get_handler("plugin.demo.echo") is not None
    assert get_handler("echo") is None  # namespaced, no shadowing

    conn = connect(tmp_db_url)
    init_db(conn)
    assert pm.sync_cron(conn) == 1
    row = conn.execute("SELECT * FROM cron WHERE name LIKE 'plugin:%'"
                       ).fetchone()
    assert row["task_type"] == "plugin.demo.echo"
    assert row["schedule"] == "*/5 * * * *"

    tid = enqueue(conn, "plugin.demo.echo", {"v": "hi"})
    Worker(db_url=tmp_db_url, max_jobs=1).run_forever(idle_timeout=3.0)
    t = task_row(conn, tid)
    assert t["status"] == SUCCESS and '"echoed": "hi"' in t["result"]

    # plugin unloaded -> cron rows pruned
    pm.loaded.clear()
    assert pm.sync_cron(conn) == 0
    assert conn.execute("SELECT COUNT(*) FROM cron WHERE name LIKE "
                        "'plugin:%'").fetchone()[0] == 0
    conn.close()


def test_plugin_upload_endpoint_and_worker_boot(tmp_path):
    """Upload via admin API -> persisted -> fresh worker boots it and
    runs its contributed task type."""
    import base64

    from audiomuse_amd.db import connect
    from audiomuse_amd.db.schema import init_db
    from audiomuse_amd.plugin import plugin_manager
    from audiomuse_amd.taskqueue import SUCCESS, enqueue, task_row
    from audiomuse_amd.taskqueue.worker import Worker
    from audiomuse_amd.web.app import create_app

    url = f"sqlite:///{tmp_path}/plug.db"
    conn = connect(url)
    init_db(conn)
    app = create_app(url, auth_disabled=True)
    app.testing = True
    blob = _plugin_zip(
        "def register(api):\n"
        "    api.add_task_handler('stamp',\n"
        "        lambda ctx, payload: {'ok': payload['x'] * 2})\n")
    try:
        with app.test_client() as client:
            r = client.post("/api/plugins",
                            json={"name": "stamper",
                                  "zip_base64":
                                      base64.b64encode(blob).decode()})
            assert r.status_code == 201, r.json
            assert client.get("/api/plugins").json[0]["name"] == "stamper"
            # reject garbage
            assert client.post("/api/plugins",
                               json={"name": "bad!name",
                                     "zip_base64": ""}).status_code == 400

        # a fresh process' worker would start empty: simulate by clearing
        plugin_manager.loaded.clear()
        from audiomuse_amd.taskqueue.worker import _REGISTRY
        _REGISTRY.pop("plugin.stamper.stamp", None)
        tid = enqueue(conn, "plugin.stamper.stamp", {"x": 21})
        Worker(db_url=url, max_jobs=1).run_forever(idle_timeout=3.0)
        t = task_row(conn, tid)
        assert t["status"] == SUCCESS and '"ok": 42' in t["result"]

        with app.test_client() as client:
            assert client.delete("/api/plugins/stamper").status_code == 200
            assert client.delete("/api/plugins/stamper").status_code == 404
    finally:
        plugin_manager.loaded.clear()
        conn.close()


def test_cron_validate():
    from audiomuse_amd.utils.cron import validate_cron

    assert validate_cron("0 3 * * *")
    assert validate_cron("*/5 1-4 * * 0-5")
    assert not validate_cron("99 * * * *")      # can never fire
    assert not validate_cron("* * * *")          # wrong arity
    assert not validate_cron("a b c d e")        # garbage


def test_cron_queue_guard_and_retry(tmp_db_url, monkeypatch):
    """A due guarded cron run is parked while another guarded task is
    active, re-attempted, and surfaced as FAILURE when the retry window
    expires (reference: ALGORITHM.md 16.2 steps 5-6)."""
    import json as _json

    from audiomuse_amd import config as C
    from audiomuse_amd.db import connect, write_txn
    from audiomuse_amd.db.schema import init_db
    from audiomuse_amd.taskqueue import enqueue
    from audiomuse_amd.utils.cron import run_due_cron_jobs

    conn = connect(tmp_db_url)
    init_db(conn)
    monkeypatch.setattr(C, "CRON_RETRY_INTERVAL_MINUTES", 1.0)
    monkeypatch.setattr(C, "CRON_RETRY_MAX_MINUTES", 2.0)  # 2 attempts
    with write_txn(conn):
        conn.execute(
            "INSERT INTO cron (name, schedule, task_type, payload, enabled) "
            "VALUES ('nightly', '* * * * *', 'run_analysis', '{}', 1)")
    blocker = enqueue(conn, "run_clustering", {})   # active guarded task

    t0 = 1_700_000_000.0
    assert run_due_cron_jobs(conn, now=t0) == []    # parked, not enqueued
    row = conn.execute("SELECT * FROM cron_retry").fetchone()
    assert row is not None and row["attempts"] == 1

    # retry due, still blocked -> window expires into a visible FAILURE
    assert run_due_cron_jobs(conn, now=t0 + 61) == []
    fail = conn.execute(
        "SELECT * FROM task_status WHERE status='FAILURE' "
        "AND task_type='run_analysis'").fetchone()
    assert fail is not None
    assert "cron retry window expired" in fail["details"]
    assert conn.execute("SELECT COUNT(*) FROM cron_retry").fetchone()[0] == 0

    # guard clears -> a due row enqueues normally
    with write_txn(conn):
        conn.execute("UPDATE task_status SET status='SUCCESS' "
                     "WHERE task_id=?", (blocker,))
    got = run_due_cron_jobs(conn, now=t0 + 120)
    assert len(got) == 1
    t = conn.execute("SELECT task_type, status FROM task_status "
                     "WHERE task_id=?", (got[0],)).fetchone()
    assert t["task_type"] == "run_analysis" and t["status"] == "PENDING"
    conn.close()
