"""ID canonicalization migration tests (reference:
test_canonicalize_integration.py behavior)."""

import numpy as np
import pytest

from audiomuse_amd.analysis.canonicalize import (canonicalize_legacy_ids,
                                                 repair_duplicate_track_maps,
                                                 run_startup_migrations)
from audiomuse_amd.db import connect, write_txn
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.db.store import save_track_analysis_and_embedding
from audiomuse_amd.engines import simhash


@pytest.fixture
def db(tmp_db_url):
    conn = connect(tmp_db_url)
    init_db(conn)
    yield conn
    conn.close()


def _add_track(conn, item_id, emb, provider_id=None, duration=120.0):
    save_track_analysis_and_embedding(conn, item_id, title=item_id,
                                      author="A", duration=duration,
                                      embedding=emb)
    if provider_id:
        with write_txn(conn):
            conn.execute(
                """INSERT INTO track_server_map (provider_id, server_id,
                       item_id) VALUES (?, 'srv', ?)""",
                (provider_id, item_id))


def test_legacy_ids_relabeled_to_signatures(db):
    rng = np.random.default_rng(0)
    emb = rng.standard_normal(200).astype(np.float32)
    _add_track(db, "navidrome-123", emb, provider_id="p1")
    out = canonicalize_legacy_ids(db)
    assert out["relabeled"] == 1
    row = db.execute("SELECT item_id FROM score").fetchone()
    assert simhash.is_signature_id(row["item_id"])
    # the mapping row followed the rewrite
    m = db.execute("SELECT item_id FROM track_server_map").fetchone()
    assert m["item_id"] == row["item_id"]
    # idempotent
    assert canonicalize_legacy_ids(db)["relabeled"] == 0


def test_legacy_merge_into_existing_signature_id(db):
    rng = np.random.default_rng(1)
    emb = rng.standard_normal(200).astype(np.float32)
    sig_id = simhash.mint_canonical_id(simhash.embedding_signature(emb), set())
    _add_track(db, sig_id, emb, provider_id="p-sig")
    _add_track(db, "legacy-9", emb + 1e-5, provider_id="p-legacy")
    out = canonicalize_legacy_ids(db)
    assert out["merged"] == 1
    # one catalogue row; both mappings point at it
    assert db.execute("SELECT COUNT(*) FROM score").fetchone()[0] == 1
    maps = db.execute("SELECT DISTINCT item_id FROM track_server_map").fetchall()
    assert len(maps) == 1 and maps[0]["item_id"] == sig_id


def test_duplicate_repair_collapses_same_recording(db):
    rng = np.random.default_rng(2)
    emb = rng.standard_normal(200).astype(np.float32)
    id_a = simhash.mint_canonical_id(simhash.embedding_signature(emb), set())
    # same recording minted under a stepped id (string collision case)
    id_b = simhash.mint_canonical_id(simhash.embedding_signature(emb), {id_a})
    _add_track(db, id_a, emb, provider_id="pa")
    _add_track(db, id_b, emb + 1e-6, provider_id="pb")
    repaired = repair_duplicate_track_maps(db)
    assert repaired == 1
    assert db.execute("SELECT COUNT(*) FROM score").fetchone()[0] == 1
    maps = db.execute("SELECT DISTINCT item_id FROM track_server_map").fetchall()
    assert len(maps) == 1


def test_run_startup_migrations_smoke(db):
    out = run_startup_migrations(db)
    assert out == {"relabeled": 0, "merged": 0, "total": 0, "repaired": 0}
