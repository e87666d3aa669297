"""End-to-end analysis slice on CPU: synthetic provider -> queue worker
-> embeddings + catalogue ids in DB -> index builds -> similarity query.
(Reference analog: test_analysis_integration.py against an ephemeral PG.)
CLAP runs on the GPU path (marked test); the CPU e2e disables it to keep
runtime bounded."""

import numpy as np
import pytest
import torch

import audiomuse_amd.analysis.tasks as atasks
from audiomuse_amd import config as C
from audiomuse_amd.analysis.index import (AUDIO_INDEX, load_artist_similarity,
                                          load_ivf_engine,
                                          run_all_index_builds)
from audiomuse_amd.db import connect
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.engines.simhash import is_signature_id
from audiomuse_amd.taskqueue import SUCCESS, enqueue, task_row
from audiomuse_amd.taskqueue.worker import Worker


@pytest.fixture
def db(tmp_db_url, monkeypatch):
    monkeypatch.setattr(C, "CLAP_ENABLED", False)
    atasks._RUNTIME = None
    conn = connect(tmp_db_url)
    init_db(conn)
    yield conn, tmp_db_url
    atasks._RUNTIME = None
    conn.close()


@pytest.mark.slow
def test_full_analysis_pipeline(db):
    conn, url = db
    payload = {"server_type": "synthetic", "server_id": "srv1",
               "server_config": {"n_albums": 2, "tracks_per_album": 2,
                                 "seconds": 6.0, "sr": 22050},
               "drain_timeout": 120.0}
    parent = enqueue(conn, "run_analysis", payload, queue="high")

    # one worker drains parent + children sequentially:
    # the parent's drain loop needs siblings, so run parent in a thread
    import threading

    w1 = Worker(db_url=url, max_jobs=10)
    w2 = Worker(db_url=url, max_jobs=10)
    t1 = threading.Thread(target=lambda: w1.run_forever(idle_timeout=20.0))
    t2 = threading.Thread(target=lambda: w2.run_forever(idle_timeout=20.0))
    t1.start(); t2.start()
    t1.join(timeout=300); t2.join(timeout=300)

    row = task_row(conn, parent)
    assert row["status"] == SUCCESS, row["result"]

    # catalogue rows with signature ids; random-init embeddings of similar
    # synthetic tracks may legitimately dedupe to one canonical recording,
    # so scores counts DISTINCT recordings while track_server_map covers
    # every provider track
    scores = conn.execute("SELECT item_id, author, tempo, energy FROM score").fetchall()
    assert 1 <= len(scores) <= 4
    for r in scores:
        assert is_signature_id(r["item_id"]) or r["item_id"].startswith("fp_0")
        assert 0.0 <= r["energy"] <= 1.0
    embs = conn.execute("SELECT COUNT(*) AS n FROM embedding").fetchone()
    assert embs["n"] == len(scores)
    maps = conn.execute("SELECT COUNT(*) AS n FROM track_server_map").fetchone()
    assert maps["n"] == 4
    mapped_ids = {r["item_id"] for r in conn.execute(
        "SELECT item_id FROM track_server_map")}
    assert mapped_ids == {r["item_id"] for r in scores}

    # indexes built by the parent task's final stage
    eng = load_ivf_engine(conn, AUDIO_INDEX)
    assert eng is not None and eng.index.n == len(scores)
    if eng.index.n >= 2:
        some_id = scores[0]["item_id"]
        res = eng.find_similar_by_id(some_id, 2, nprobe=64)
        assert res and all(r["item_id"] != some_id for r in res)

    art = load_artist_similarity(conn)
    assert art is not None and len(art.models) >= 1


@pytest.mark.slow
def test_reanalysis_skips_done_tracks(db):
    conn, url = db
    payload = {"server_type": "synthetic", "server_id": "srv1",
               "server_config": {"n_albums": 1, "tracks_per_album": 2,
                                 "seconds": 6.0, "sr": 22050}}
    tid = enqueue(conn, "analyze_album",
                  {**payload, "album_id": "a0"})
    Worker(db_url=url, max_jobs=1).run_forever(idle_timeout=5.0)
    row = task_row(conn, tid)
    assert row["status"] == SUCCESS
    assert '"analyzed": 2' in row["result"]

    tid2 = enqueue(conn, "analyze_album", {**payload, "album_id": "a0"})
    Worker(db_url=url, max_jobs=1).run_forever(idle_timeout=5.0)
    row2 = task_row(conn, tid2)
    assert '"analyzed": 0' in row2["result"]
    assert '"skipped": 2' in row2["result"]


def test_index_builds_on_synthetic_embeddings(db):
    conn, _url = db
    from audiomuse_amd.db.store import (save_clap_embedding,
                                        save_lyrics_embedding,
                                        save_track_analysis_and_embedding)

    rng = np.random.default_rng(0)
    for i in range(30):
        iid = f"fp_4{'%050x' % i}"
        save_track_analysis_and_embedding(
            conn, iid, title=f"T{i}", author=f"artist{i % 5}",
            tempo=120.0, energy=0.5, duration=60.0,
            embedding=rng.standard_normal(200).astype(np.float32))
        save_clap_embedding(conn, iid,
                            rng.standard_normal(512).astype(np.float32))
        if i % 2 == 0:
            save_lyrics_embedding(conn, iid,
                                  rng.standard_normal(768).astype(np.float32),
                                  axis_scores={"love": 0.5})
    built = run_all_index_builds(conn)
    assert built["audio"] == 30
    assert built["clap"] == 30
    assert built["lyrics"] == 15
    assert built["lyrics_axes"] == 15
    assert built["semgrove"] == 15
    assert built["artist"] == 5
    assert built["song_map"] == 30
    assert built["artist_map"] == 5


def test_incremental_index_refresh(db):
    conn, _url = db
    import torch

    from audiomuse_amd.analysis.index import (AUDIO_INDEX, build_audio_index,
                                              load_ivf_engine,
                                              refresh_ivf_index)
    from audiomuse_amd.db import write_txn
    from audiomuse_amd.db.store import save_track_analysis_and_embedding

    rng = np.random.default_rng(1)
    vecs = {}
    for i in range(40):
        vecs[f"t{i}"] = rng.standard_normal(64).astype(np.float32)
        save_track_analysis_and_embedding(conn, f"t{i}", title=f"T{i}",
                                          embedding=vecs[f"t{i}"])
    assert build_audio_index(conn) == 40

    # no-op refresh
    out = refresh_ivf_index(conn, AUDIO_INDEX)
    assert out == {"added": 0, "removed": 0, "total": 40, "rebuilt": 0}

    # +5 tracks, -3 tracks -> splice, not rebuild
    for i in range(40, 45):
        vecs[f"t{i}"] = rng.standard_normal(64).astype(np.float32)
        save_track_analysis_and_embedding(conn, f"t{i}", title=f"T{i}",
                                          embedding=vecs[f"t{i}"])
    with write_txn(conn):
        conn.execute("DELETE FROM embedding WHERE item_id IN ('t0','t1','t2')")
    out = refresh_ivf_index(conn, AUDIO_INDEX)
    assert out["added"] == 5 and out["removed"] == 3 and out["rebuilt"] == 0
    assert out["total"] == 42

    eng = load_ivf_engine(conn, AUDIO_INDEX)
    assert "t42" in eng.pos and "t0" not in [
        eng.item_ids[int(r)] for r in eng.index.ids.tolist()]
    # each spliced-in vector finds itself
    for iid in ("t40", "t44"):
        _, rows = eng.index.query(torch.from_numpy(vecs[iid]), k=1,
                                  nprobe=eng.index.nlist)
        assert eng.item_ids[int(rows[0])] == iid

    # large drift -> full rebuild
    with write_txn(conn):
        conn.execute("DELETE FROM embedding WHERE item_id IN ("
                     + ",".join(f"'t{i}'" for i in range(3, 25)) + ")")
    out = refresh_ivf_index(conn, AUDIO_INDEX)
    assert out["rebuilt"] == 1 and out["total"] == 20


def test_refresh_indexes_task_runs(db):
    conn, url = db
    from audiomuse_amd.db.store import save_track_analysis_and_embedding

    rng = np.random.default_rng(2)
    for i in range(10):
        save_track_analysis_and_embedding(
            conn, f"r{i}", title=f"R{i}",
            embedding=rng.standard_normal(32).astype(np.float32))
    tid = enqueue(conn, "refresh_indexes", {})
    Worker(db_url=url, max_jobs=1).run_forever(idle_timeout=3.0)
    row = task_row(conn, tid)
    assert row["status"] == SUCCESS
    assert '"rebuilt": 1' in row["result"]  # first run builds audio index
