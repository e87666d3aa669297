"""Canonicalize-integration depth (modeled on the reference's
test_canonicalize_integration.py, 746 LoC): multi-table key rewrites,
merge-into-existing semantics, idempotency, FK integrity, mixed
catalogues, duplicate repair with chromaprint-confirmed splits, and the
boot-sequence ordering. Runs on BOTH backends via tmp_db_url."""

import json

import numpy as np
import pytest

from audiomuse_amd.analysis.canonicalize import (canonicalize_legacy_ids,
                                                 repair_duplicate_track_maps,
                                                 run_startup_migrations)
from audiomuse_amd.db import connect, write_txn
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.db.store import (save_clap_embedding,
                                    save_lyrics_embedding,
                                    save_track_analysis_and_embedding)
from audiomuse_amd.engines import simhash


@pytest.fixture
def db(tmp_db_url):
    conn = connect(tmp_db_url)
    init_db(conn)
    yield conn
    conn.close()


def _vec(seed, dim=200):
    return np.random.default_rng(seed).standard_normal(dim).astype(np.float32)


def _add_track(conn, item_id, vec, title="T", duration=100.0,
               with_clap=True, with_lyrics=False, server="srv1",
               provider_id=None):
    save_track_analysis_and_embedding(
        conn, item_id, title=title, author="A", album="Al",
        tempo=100, key="C", scale="major", mood_vector={"rock": 0.5},
        other_features={}, energy=0.4, duration=duration, embedding=vec)
    if with_clap:
        save_clap_embedding(conn, item_id, _vec(hash(item_id) % 999, 512))
    if with_lyrics:
        save_lyrics_embedding(conn, item_id, _vec(1, 768),
                              axis_scores={"love": 0.2})
    with write_txn(conn):
        conn.execute(
            "INSERT INTO track_server_map (provider_id, server_id, item_id) "
            "VALUES (?,?,?)",
            (provider_id or f"prov-{item_id}", server, item_id))


def _count(conn, table, item_id):
    return conn.execute(
        f"SELECT COUNT(*) AS n FROM {table} WHERE item_id=?",
        (item_id,)).fetchone()["n"]


def test_rewrite_moves_every_table_together(db):
    """The reference's critical invariant: one transaction moves the id
    in score, embedding, clap, lyrics, chromaprint and mappings."""
    v = _vec(1)
    _add_track(db, "legacy-1", v, with_lyrics=True)
    with write_txn(db):
        db.execute("INSERT INTO chromaprint (item_id, fingerprint, duration)"
                   " VALUES ('legacy-1', ?, 100.0)", (b"fp-blob",))
    out = canonicalize_legacy_ids(db)
    assert out["relabeled"] == 1 and out["merged"] == 0
    sig_id = db.execute("SELECT item_id FROM score").fetchone()["item_id"]
    assert simhash.is_signature_id(sig_id)
    for table in ("score", "embedding", "clap_embedding",
                  "lyrics_embedding", "chromaprint", "track_server_map"):
        assert _count(db, table, "legacy-1") == 0, table
        assert _count(db, table, sig_id) == 1, table


def test_same_recording_on_two_servers_merges(db):
    """Two providers exposing one recording: after canonicalize both
    mappings point at ONE catalogue row (the cross-server dedupe that
    motivates the whole id scheme)."""
    v = _vec(2)
    _add_track(db, "nav-123", v, server="srv-nav")
    _add_track(db, "jf-999", v + 1e-4 * _vec(3), server="srv-jf")
    out = canonicalize_legacy_ids(db)
    assert out["relabeled"] == 2 and out["merged"] >= 1
    assert db.execute("SELECT COUNT(*) AS n FROM score").fetchone()["n"] == 1
    maps = db.execute(
        "SELECT DISTINCT item_id FROM track_server_map").fetchall()
    assert len(maps) == 1 and simhash.is_signature_id(maps[0]["item_id"])
    assert db.execute("SELECT COUNT(*) AS n FROM track_server_map"
                      ).fetchone()["n"] == 2


def test_different_recordings_stay_apart(db):
    _add_track(db, "a-1", _vec(10))
    _add_track(db, "a-2", _vec(20))
    out = canonicalize_legacy_ids(db)
    assert out["relabeled"] == 2 and out["merged"] == 0
    assert db.execute("SELECT COUNT(*) AS n FROM score").fetchone()["n"] == 2


def test_duration_gate_blocks_merge(db):
    """Same embedding but wildly different durations: the confirm gate
    (SIMHASH_CONFIRM_DURATION_SECONDS) refuses the merge."""
    v = _vec(4)
    _add_track(db, "x-1", v, duration=100.0)
    _add_track(db, "x-2", v, duration=250.0)
    out = canonicalize_legacy_ids(db)
    assert out["merged"] == 0
    assert db.execute("SELECT COUNT(*) AS n FROM score").fetchone()["n"] == 2


def test_canonicalize_idempotent(db):
    _add_track(db, "leg-1", _vec(5))
    first = canonicalize_legacy_ids(db)
    assert first["relabeled"] == 1
    again = canonicalize_legacy_ids(db)
    assert again["relabeled"] == 0 and again["merged"] == 0


def test_legacy_merges_into_preexisting_signature_id(db):
    v = _vec(6)
    sig = simhash.embedding_signature(v)
    sig_id = simhash.mint_canonical_id(sig, set())
    _add_track(db, sig_id, v)                     # already-canonical row
    _add_track(db, "old-7", v + 1e-4 * _vec(7))   # same recording, legacy
    out = canonicalize_legacy_ids(db)
    assert out["merged"] == 1
    rows = db.execute("SELECT item_id FROM score").fetchall()
    assert [r["item_id"] for r in rows] == [sig_id]
    # both mappings now resolve to the canonical id
    assert _count(db, "track_server_map", sig_id) == 2


def test_duplicate_repair_collapses_and_keeps_older_id(db):
    v = _vec(8)
    ids = []
    for i in range(2):
        sig_id = simhash.mint_canonical_id(
            simhash.embedding_signature(v + 1e-5 * i), set(ids))
        ids.append(sig_id)
        _add_track(db, sig_id, v + 1e-5 * i)
    repaired = repair_duplicate_track_maps(db)
    assert repaired == 1
    left = [r["item_id"] for r in db.execute("SELECT item_id FROM score")]
    assert left == [min(ids)]                    # older (sorted-first) kept
    assert _count(db, "track_server_map", min(ids)) == 2


def test_mixed_catalogue_end_to_end_boot(db):
    """run_startup_migrations on a mixed catalogue: legacy + canonical +
    duplicates; the pass leaves only signature ids with no orphans."""
    v1, v2 = _vec(30), _vec(31)
    _add_track(db, "legacy-a", v1, server="s1")
    _add_track(db, "legacy-b", v1 + 1e-4 * _vec(32), server="s2")  # dup of a
    _add_track(db, "legacy-c", v2, server="s1")
    out = run_startup_migrations(db)
    assert out["relabeled"] == 3
    ids = [r["item_id"] for r in db.execute("SELECT item_id FROM score")]
    assert len(ids) == 2 and all(simhash.is_signature_id(i) for i in ids)
    # no mapping left behind pointing at a dead id
    orphan = db.execute(
        """SELECT COUNT(*) AS n FROM track_server_map m
           LEFT JOIN score s ON s.item_id = m.item_id
           WHERE s.item_id IS NULL""").fetchone()["n"]
    assert orphan == 0


def test_signature_ids_are_stable_across_processes(db):
    """Bit-exact signatures (reference simhash.py:127): the same
    embedding always mints the same fp_4 id — the property cross-server
    dedupe depends on."""
    v = _vec(9)
    a = simhash.embedding_signature(v)
    b = simhash.embedding_signature(v.copy())
    assert a == b
    i1 = simhash.mint_canonical_id(a, set())
    i2 = simhash.mint_canonical_id(b, set())
    assert i1 == i2 and i1.startswith("fp_4") and len(i1) == 54
