"""Catalogue-id migrations: canonicalize + duplicate repair.

Reference: /root/reference/tasks/fingerprint_canonicalize.py (989 LoC;
canonicalize_fingerprinted_ids :849) and tasks/duplicate_repair.py (607;
repair_duplicate_track_maps :434, split_chromaprint_false_merges :598) —
startup steps run inline on the web process: relabel legacy provider ids
to signature ids with a transactional key rewrite across every table,
then confirm/split merged recordings.
"""

from __future__ import annotations

import logging
import sqlite3
from typing import Dict

import numpy as np

from audiomuse_amd.db import write_txn
from audiomuse_amd.engines import simhash

logger = logging.getLogger(__name__)

_ID_TABLES = [
    ("score", "item_id"),
    ("embedding", "item_id"),
    ("clap_embedding", "item_id"),
    ("lyrics_embedding", "item_id"),
    ("chromaprint", "item_id"),
    ("track_server_map", "item_id"),
]


def _rewrite_id(conn: sqlite3.Connection, old: str, new: str) -> None:
    """Transactional key rewrite across every id-bearing table (the
    reference's critical invariant: all tables move together)."""
    for table, col in _ID_TABLES:
        if table == "track_server_map":
            conn.execute(f"UPDATE {table} SET {col}=? WHERE {col}=?",
                         (new, old))
            continue
        exists = conn.execute(
            f"SELECT 1 FROM {table} WHERE {col}=?", (new,)).fetchone()
        if exists:
            # target row already present (merge): drop the old duplicate
            conn.execute(f"DELETE FROM {table} WHERE {col}=?", (old,))
        else:
            conn.execute(f"UPDATE {table} SET {col}=? WHERE {col}=?",
                         (new, old))


def canonicalize_legacy_ids(conn: sqlite3.Connection) -> Dict[str, int]:
    """Relabel every non-signature item_id that has an embedding to its
    fp_4 signature id (canonicalize_fingerprinted_ids :849). Safe to run
    repeatedly (idempotent: signature ids pass through)."""
    rows = conn.execute(
        """SELECT e.item_id, e.embedding, s.duration FROM embedding e
           LEFT JOIN score s ON s.item_id = e.item_id""").fetchall()
    relabeled = 0
    merged = 0
    resolver = simhash.CatalogResolver()
    # register existing signature ids first so legacy rows match into them
    for r in rows:
        if simhash.is_signature_id(r["item_id"]):
            emb = np.frombuffer(r["embedding"], dtype=np.float32)
            resolver.register_existing(r["item_id"], emb, r["duration"] or 0.0)
    # key rewrites touch parent+child tables together; suspend FK checks
    # for the migration transaction (pragma must sit outside a txn)
    conn.execute("PRAGMA foreign_keys=OFF")
    try:
        with write_txn(conn):
            for r in rows:
                old = r["item_id"]
                if simhash.is_signature_id(old) or old.startswith("fp_0"):
                    continue
                emb = np.frombuffer(r["embedding"], dtype=np.float32)
                new, matched = resolver.resolve(emb, r["duration"] or 0.0,
                                                "legacy", old)
                if new == old:
                    continue
                _rewrite_id(conn, old, new)
                relabeled += 1
                if matched:
                    merged += 1
    finally:
        conn.execute("PRAGMA foreign_keys=ON")
    return {"relabeled": relabeled, "merged": merged,
            "total": len(rows)}


def repair_duplicate_track_maps(conn: sqlite3.Connection) -> int:
    """Collapse track_server_map rows pointing at recordings whose
    embeddings confirm as the same recording (repair path of
    duplicate_repair.py:434)."""
    rows = conn.execute(
        """SELECT e.item_id, e.embedding, s.duration FROM embedding e
           LEFT JOIN score s ON s.item_id = e.item_id
           WHERE e.item_id LIKE 'fp\\_4%' ESCAPE '\\'""").fetchall()
    index = simhash.SignatureIndex()
    vecs: Dict[str, np.ndarray] = {}
    for r in rows:
        emb = np.frombuffer(r["embedding"], dtype=np.float32)
        sig = simhash.embedding_signature(emb)
        if sig is None:
            continue
        index.add(r["item_id"], sig, r["duration"] or 0.0)
        vecs[r["item_id"]] = emb
    repaired = 0
    seen: set = set()
    conn.execute("PRAGMA foreign_keys=OFF")
    with write_txn(conn):
        for item_id, emb in vecs.items():
            if item_id in seen:
                continue
            sig = simhash.embedding_signature(emb)
            for other, ham in index.lookup(sig):
                if other == item_id or other in seen:
                    continue
                from audiomuse_amd import config as C

                if simhash.cosine_distance(emb, vecs[other]) < C.SIMHASH_CONFIRM_COSINE:
                    # same recording under two ids: remap the younger id
                    keep, drop = sorted([item_id, other])
                    _rewrite_id(conn, drop, keep)
                    seen.add(drop)
                    repaired += 1
    conn.execute("PRAGMA foreign_keys=ON")
    return repaired


def run_startup_migrations(conn: sqlite3.Connection) -> Dict[str, int]:
    """Inline boot steps (reference: app.py boot sequence §3.5)."""
    out = canonicalize_legacy_ids(conn)
    out["repaired"] = repair_duplicate_track_maps(conn)
    return out
