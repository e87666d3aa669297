"""Index rebuild orchestration.

Reference: /root/reference/tasks/analysis/index.py (_run_all_index_builds
:47): the 8 builds in order — audio IVF (fatal on failure), CLAP,
lyrics, lyrics-axes, SemGrove, artist, song map, artist map — then an
index-reload notification. Here each index serializes into the
segmented blob store (db.store.store_index_blob) and the web layer
reloads by watching ivf_dir.updated_at (the LISTEN/NOTIFY analog).
"""

from __future__ import annotations

import io
import logging
import sqlite3
from typing import Dict, List, Optional

import numpy as np
import torch

from audiomuse_amd import config as C
from audiomuse_amd.cluster.algorithms import pca_fit_transform
from audiomuse_amd.db.store import (load_all_embeddings, load_index_blob,
                                    store_index_blob)
from audiomuse_amd.engines.artist_gmm import ArtistSimilarity
from audiomuse_amd.engines.misc import SemGroveMerger
from audiomuse_amd.engines.similarity import SimilarityEngine
from audiomuse_amd.index.ivf import IVFIndex

logger = logging.getLogger(__name__)

AUDIO_INDEX = "audio_ivf"
CLAP_INDEX = "clap_ivf"
LYRICS_INDEX = "lyrics_ivf"
SEMGROVE_INDEX = "semgrove_ivf"
ARTIST_INDEX = "artist_models"
SONG_MAP = "song_map_projection"


def _store_ivf(conn: sqlite3.Connection, name: str, index: IVFIndex,
               item_ids: List[str]) -> None:
    payload = {"index": index.serialize(), "item_ids": item_ids}
    buf = io.BytesIO()
    torch.save(payload, buf)
    store_index_blob(conn, name, buf.getvalue(),
                     meta={"n": len(item_ids), "dim": index.dim,
                           "metric": index.metric, "storage": index.storage})


def load_ivf_engine(conn: sqlite3.Connection, name: str,
                    device: str = "cpu",
                    meta_fn=None) -> Optional[SimilarityEngine]:
    got = load_index_blob(conn, name)
    if got is None:
        return None
    blob, _meta = got
    payload = torch.load(io.BytesIO(blob), map_location="cpu",
                         weights_only=True)
    index = IVFIndex.deserialize(payload["index"], device=device)
    return SimilarityEngine(index, payload["item_ids"], meta_fn=meta_fn)


def build_audio_index(conn: sqlite3.Connection, device: str = "cpu") -> int:
    ids, mat = load_all_embeddings(conn, "embedding")
    if not ids:
        return 0
    idx = IVFIndex.build(torch.from_numpy(mat), metric="angular",
                         device=device)
    _store_ivf(conn, AUDIO_INDEX, idx, ids)
    return len(ids)


def build_clap_index(conn: sqlite3.Connection, device: str = "cpu") -> int:
    ids, mat = load_all_embeddings(conn, "clap_embedding")
    if not ids:
        return 0
    idx = IVFIndex.build(torch.from_numpy(mat), metric="angular",
                         device=device)
    _store_ivf(conn, CLAP_INDEX, idx, ids)
    return len(ids)


def build_lyrics_index(conn: sqlite3.Connection, device: str = "cpu") -> int:
    ids, mat = load_all_embeddings(conn, "lyrics_embedding")
    if not ids:
        return 0
    idx = IVFIndex.build(torch.from_numpy(mat), metric="angular",
                         device=device)
    _store_ivf(conn, LYRICS_INDEX, idx, ids)
    return len(ids)


def build_semgrove_index(conn: sqlite3.Connection, device: str = "cpu") -> int:
    """Fused lyrics+audio space (sem_grove_manager.py:65-108)."""
    lids, lmat = load_all_embeddings(conn, "lyrics_embedding")
    aids, amat = load_all_embeddings(conn, "clap_embedding")
    if not lids or not aids:
        return 0
    apos = {s: i for i, s in enumerate(aids)}
    pairs = [(i, apos[s]) for i, s in enumerate(lids) if s in apos]
    if not pairs:
        return 0
    li, ai = zip(*pairs)
    lyr = lmat[list(li)]
    aud = amat[list(ai)]
    merger = SemGroveMerger()
    merger.fit(lyr, aud)
    merged = merger.merge(lyr, aud)
    ids = [lids[i] for i in li]
    idx = IVFIndex.build(torch.from_numpy(merged), metric="angular",
                         device=device)
    _store_ivf(conn, SEMGROVE_INDEX, idx, ids)
    # persist whitening stats for query-time merging
    stats = io.BytesIO()
    torch.save({"lyr_mean": merger.lyr_mean, "lyr_std": merger.lyr_std,
                "aud_mean": merger.aud_mean, "aud_std": merger.aud_std}, stats)
    store_index_blob(conn, SEMGROVE_INDEX + "_stats", stats.getvalue())
    return len(ids)


def build_artist_index(conn: sqlite3.Connection, seed: int = 0) -> int:
    """Per-artist GMMs (artist_gmm_manager.build_and_store_artist_index)."""
    rows = conn.execute(
        """SELECT s.author, e.embedding FROM score s
           JOIN embedding e ON e.item_id = s.item_id
           WHERE s.author IS NOT NULL AND s.author != ''""").fetchall()
    per_artist: Dict[str, List[np.ndarray]] = {}
    for r in rows:
        per_artist.setdefault(r["author"], []).append(
            np.frombuffer(r["embedding"], dtype=np.float32))
    per_artist = {a: np.stack(v) for a, v in per_artist.items() if len(v) >= 1}
    if not per_artist:
        return 0
    sim = ArtistSimilarity()
    sim.fit_catalogue(per_artist, seed=seed)
    payload = {name: {"means": m.means, "weights": m.weights,
                      "n_tracks": m.n_tracks}
               for name, m in sim.models.items()}
    buf = io.BytesIO()
    torch.save(payload, buf)
    store_index_blob(conn, ARTIST_INDEX, buf.getvalue(),
                     meta={"n_artists": len(payload)})
    return len(payload)


def load_artist_similarity(conn: sqlite3.Connection) -> Optional[ArtistSimilarity]:
    got = load_index_blob(conn, ARTIST_INDEX)
    if got is None:
        return None
    payload = torch.load(io.BytesIO(got[0]), map_location="cpu",
                         weights_only=False)
    from audiomuse_amd.engines.artist_gmm import ArtistModel

    sim = ArtistSimilarity()
    for name, d in payload.items():
        sim.models[name] = ArtistModel(name=name, means=d["means"],
                                       weights=d["weights"],
                                       n_tracks=d["n_tracks"])
    sim._names = list(sim.models)
    cents = [m.means.mean(axis=0) for m in sim.models.values()]
    sim._centroids = np.stack(cents).astype(np.float32) if cents else None
    return sim


def build_song_map(conn: sqlite3.Connection, device: str = "cpu") -> int:
    """2-D map projection (reference: app_helper.build_and_store_map_
    projection :340 — UMAP with PCA fallback; engines/projection.py
    provides the UMAP-style layout)."""
    ids, mat = load_all_embeddings(conn, "embedding")
    if not ids:
        return 0
    x = torch.from_numpy(mat)
    try:
        from audiomuse_amd.engines.projection import umap_project

        coords = umap_project(x.to(device), seed=0).cpu().numpy()
    except Exception:
        proj, _, _ = pca_fit_transform(x.to(device), 2)
        coords = proj.cpu().numpy()
    buf = io.BytesIO()
    torch.save({"item_ids": ids,
                "coords": torch.from_numpy(coords.astype(np.float32))}, buf)
    store_index_blob(conn, SONG_MAP, buf.getvalue(), meta={"n": len(ids)})
    return len(ids)


def run_all_index_builds(conn: sqlite3.Connection, device: str = "cpu",
                         progress_cb=None) -> Dict[str, int]:
    """The ordered build set (_run_all_index_builds, index.py:47).
    Audio IVF failure is fatal; the rest log and continue."""
    results: Dict[str, int] = {}
    results["audio"] = build_audio_index(conn, device)   # fatal on raise
    for name, fn in [("clap", build_clap_index),
                     ("lyrics", build_lyrics_index),
                     ("semgrove", build_semgrove_index),
                     ("artist", lambda c, **k: build_artist_index(c)),
                     ("song_map", build_song_map)]:
        try:
            results[name] = fn(conn, device=device) if name != "artist" \
                else build_artist_index(conn)
        except Exception:
            logger.exception("index build %s failed", name)
            results[name] = -1
        if progress_cb is not None:
            progress_cb(name, results[name])
    return results
