"""Persistence helpers (reference: database.py save_* functions and
index_build_helpers.py blob plumbing)."""

from __future__ import annotations

import json
import sqlite3
from typing import Dict, Iterator, List, Optional, Tuple

import numpy as np

from audiomuse_amd import config as C
from audiomuse_amd.db import write_txn


def save_track_analysis_and_embedding(
        conn: sqlite3.Connection, item_id: str, *, title: str = "",
        author: str = "", album: str = "", tempo: float = 0.0, key: str = "",
        scale: str = "", mood_vector: Optional[Dict[str, float]] = None,
        other_features: Optional[Dict[str, float]] = None, energy: float = 0.0,
        duration: float = 0.0, embedding: Optional[np.ndarray] = None) -> None:
    """reference: database.py:776"""
    with write_txn(conn):
        conn.execute(
            """INSERT INTO score (item_id, title, author, album, tempo, key,
                   scale, mood_vector, other_features, energy, duration)
               VALUES (?,?,?,?,?,?,?,?,?,?,?)
               ON CONFLICT(item_id) DO UPDATE SET title=excluded.title,
                   author=excluded.author, album=excluded.album,
                   tempo=excluded.tempo, key=excluded.key, scale=excluded.scale,
                   mood_vector=excluded.mood_vector,
                   other_features=excluded.other_features,
                   energy=excluded.energy, duration=excluded.duration""",
            (item_id, title, author, album, tempo, key, scale,
             json.dumps(mood_vector or {}), json.dumps(other_features or {}),
             energy, duration))
        if embedding is not None:
            blob = np.asarray(embedding, dtype=np.float32).tobytes()
            conn.execute(
                """INSERT INTO embedding (item_id, embedding) VALUES (?, ?)
                   ON CONFLICT(item_id) DO UPDATE SET embedding=excluded.embedding""",
                (item_id, blob))


def save_clap_embedding(conn: sqlite3.Connection, item_id: str,
                        embedding: np.ndarray) -> None:
    """reference: database.py:866"""
    blob = np.asarray(embedding, dtype=np.float32).tobytes()
    with write_txn(conn):
        conn.execute(
            """INSERT INTO clap_embedding (item_id, embedding) VALUES (?, ?)
               ON CONFLICT(item_id) DO UPDATE SET embedding=excluded.embedding""",
            (item_id, blob))


def save_lyrics_embedding(conn: sqlite3.Connection, item_id: str,
                          embedding: Optional[np.ndarray],
                          axis_scores: Optional[Dict[str, float]] = None,
                          lyrics_text: str = "", language: str = "",
                          instrumental: bool = False) -> None:
    """reference: database.py:1091"""
    blob = (None if embedding is None
            else np.asarray(embedding, dtype=np.float32).tobytes())
    with write_txn(conn):
        conn.execute(
            """INSERT INTO lyrics_embedding
                   (item_id, embedding, axis_scores, lyrics_text, language,
                    instrumental)
               VALUES (?,?,?,?,?,?)
               ON CONFLICT(item_id) DO UPDATE SET embedding=excluded.embedding,
                   axis_scores=excluded.axis_scores,
                   lyrics_text=excluded.lyrics_text,
                   language=excluded.language,
                   instrumental=excluded.instrumental""",
            (item_id, blob, json.dumps(axis_scores or {}), lyrics_text,
             language, int(instrumental)))


def iter_embeddings(conn: sqlite3.Connection, table: str = "embedding",
                    batch: int = 10000) -> Iterator[Tuple[List[str], np.ndarray]]:
    """Streaming read (reference: index_build_helpers.iter_embedding_batches)."""
    assert table in ("embedding", "clap_embedding", "lyrics_embedding")
    cur = conn.execute(f"SELECT item_id, embedding FROM {table} "
                       "WHERE embedding IS NOT NULL ORDER BY item_id")
    while True:
        rows = cur.fetchmany(batch)
        if not rows:
            return
        ids = [r["item_id"] for r in rows]
        mat = np.stack([np.frombuffer(r["embedding"], dtype=np.float32)
                        for r in rows])
        yield ids, mat


def load_all_embeddings(conn: sqlite3.Connection, table: str = "embedding"
                        ) -> Tuple[List[str], np.ndarray]:
    ids: List[str] = []
    mats: List[np.ndarray] = []
    for bids, bmat in iter_embeddings(conn, table):
        ids.extend(bids)
        mats.append(bmat)
    if not mats:
        return [], np.zeros((0, 0), dtype=np.float32)
    return ids, np.concatenate(mats, axis=0)


# -- segmented blob store (reference: index_build_helpers.py:399-603) -------

def store_index_blob(conn: sqlite3.Connection, name: str, blob: bytes,
                     meta: Optional[dict] = None) -> None:
    part_size = C.IVF_MAX_PART_SIZE_MB * 1024 * 1024
    parts = [blob[i : i + part_size] for i in range(0, len(blob), part_size)] or [b""]
    with write_txn(conn):
        conn.execute("DELETE FROM ivf_cell WHERE index_name = ?", (name,))
        for i, part in enumerate(parts):
            conn.execute(
                "INSERT INTO ivf_cell (index_name, part, blob) VALUES (?,?,?)",
                (name, i, part))
        conn.execute(
            """INSERT INTO ivf_dir (index_name, meta, n_parts,
                   updated_at) VALUES (?,?,?,(julianday('now') - 2440587.5) * 86400.0)
               ON CONFLICT(index_name) DO UPDATE SET meta=excluded.meta,
                   n_parts=excluded.n_parts, updated_at=excluded.updated_at""",
            (name, json.dumps(meta or {}), len(parts)))


def load_index_blob(conn: sqlite3.Connection, name: str
                    ) -> Optional[Tuple[bytes, dict]]:
    row = conn.execute("SELECT meta, n_parts FROM ivf_dir WHERE index_name=?",
                       (name,)).fetchone()
    if row is None:
        return None
    parts = conn.execute(
        "SELECT blob FROM ivf_cell WHERE index_name=? ORDER BY part",
        (name,)).fetchall()
    if len(parts) != row["n_parts"]:
        return None
    return b"".join(p["blob"] for p in parts), json.loads(row["meta"])


# -- app config overrides (reference: config._apply_db_overrides) -----------

def get_app_config(conn: sqlite3.Connection) -> Dict[str, str]:
    return {r["key"]: r["value"]
            for r in conn.execute("SELECT key, value FROM app_config")}


def set_app_config(conn: sqlite3.Connection, key: str, value: str) -> None:
    with write_txn(conn):
        conn.execute(
            """INSERT INTO app_config (key, value) VALUES (?,?)
               ON CONFLICT(key) DO UPDATE SET value=excluded.value""",
            (key, value))
