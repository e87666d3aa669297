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
LYRICS_AXES_INDEX = "lyrics_axes_ivf"
SEMGROVE_INDEX = "semgrove_ivf"
ARTIST_INDEX = "artist_models"
SONG_MAP = "song_map_projection"
ARTIST_MAP = "artist_map_projection"


def _store_ivf(conn: sqlite3.Connection, name: str, index: IVFIndex,
               item_ids: List[str]) -> None:
    # the serialized index rides as a uint8 TENSOR: tensors serialize
    # through torch storages, while a raw bytes field would be pickled
    # as one string and overflow pickle's 4 GiB string cap (hit by the
    # semgrove build at 10^6 tracks)
    raw = torch.frombuffer(bytearray(index.serialize()), dtype=torch.uint8)
    payload = {"index": raw, "item_ids": item_ids}
    buf = io.BytesIO()
    torch.save(payload, buf)
    store_index_blob(conn, name, buf.getvalue(),
                     meta={"n": len(item_ids), "dim": index.dim,
                           "metric": index.metric, "storage": index.storage})


def _index_bytes(payload) -> bytes:
    raw = payload["index"]
    return raw.numpy().tobytes() if torch.is_tensor(raw) else raw


def load_ivf_engine(conn: sqlite3.Connection, name: str,
                    device: str = "cpu",
                    meta_fn=None) -> Optional[SimilarityEngine]:
    got = load_index_blob(conn, name)
    if got is None:
        return None
    blob, _meta = got
    payload = torch.load(io.BytesIO(blob), map_location="cpu",
                         weights_only=True)
    index = IVFIndex.deserialize(_index_bytes(payload), device=device)
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


def build_lyrics_axes_index(conn: sqlite3.Connection,
                            device: str = "cpu") -> int:
    """27-axis thematic-profile index (reference: the lyrics-axes build in
    _run_all_index_builds, index.py:47, backed by lyrics_manager's axis
    store): each track's 27 axis scores (engines/lyrics.score_axes) form a
    vector; angular IVF over those enables 'similar thematic profile'
    queries (/api/lyrics_axes_similar) beyond the single-axis ranking."""
    import json as _json

    rows = conn.execute(
        "SELECT item_id, axis_scores FROM lyrics_embedding "
        "WHERE axis_scores IS NOT NULL AND axis_scores != '{}'").fetchall()
    ids, vecs = [], []
    for r in rows:
        try:
            scores = _json.loads(r["axis_scores"])
        except Exception:
            continue
        if not scores:
            continue
        ids.append(r["item_id"])
        vecs.append([float(scores.get(a, 0.0)) for a in C.LYRICS_AXES])
    if not ids:
        return 0
    mat = torch.tensor(vecs, dtype=torch.float32)
    # 27-d profiles: f32 storage (i8 would quantize away the soft scores)
    idx = IVFIndex.build(mat, metric="angular", storage="f32", device=device)
    _store_ivf(conn, LYRICS_AXES_INDEX, idx, ids)
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
    payload = {name: {"means": torch.from_numpy(np.ascontiguousarray(m.means)),
                      "weights": torch.from_numpy(np.ascontiguousarray(m.weights)),
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
                         weights_only=True)
    from audiomuse_amd.engines.artist_gmm import ArtistModel

    sim = ArtistSimilarity()
    for name, d in payload.items():
        sim.models[name] = ArtistModel(name=name,
                                       means=d["means"].numpy(),
                                       weights=d["weights"].numpy(),
                                       n_tracks=d["n_tracks"])
    sim._names = list(sim.models)
    cents = [m.means.mean(axis=0) for m in sim.models.values()]
    sim._centroids = np.stack(cents).astype(np.float32) if cents else None
    return sim


def build_hyperbolic_tree_cache(conn: sqlite3.Connection,
                                device: str = "cpu") -> int:
    """Precompute + persist the explorer tree (full + skeleton blobs;
    reference: hyperbolic_manager.build_hyperbolic_tree_cache :613)."""
    import json as _json

    from audiomuse_amd.engines.hyperbolic_tree import build_tree, persist_tree

    ids, mat = load_all_embeddings(conn, "embedding")
    if not ids:
        return 0
    meta_rows = {r["item_id"]: r for r in conn.execute(
        "SELECT item_id, title, author, mood_vector FROM score")}

    def meta_fn(item_id):
        row = meta_rows.get(item_id)
        if row is None:
            return None
        moods = row["mood_vector"]
        return {"title": row["title"], "author": row["author"],
                "mood_vector": _json.loads(moods) if isinstance(moods, str)
                and moods else (moods or {})}

    tree = build_tree(torch.from_numpy(mat).to(device), ids, meta_fn)
    persist_tree(conn, tree)
    return tree["track_count"]


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


def build_artist_map(conn: sqlite3.Connection, device: str = "cpu") -> int:
    """2-D artist map (reference: the artist-map build in
    _run_all_index_builds, index.py:47): one point per artist at the mean
    of their track embeddings, projected with the same UMAP-style layout
    as the song map (PCA fallback)."""
    rows = conn.execute(
        """SELECT s.author, e.embedding FROM score s
           JOIN embedding e ON e.item_id = s.item_id
           WHERE s.author IS NOT NULL AND s.author != ''""").fetchall()
    per_artist: Dict[str, list] = {}
    for r in rows:
        per_artist.setdefault(r["author"], []).append(
            np.frombuffer(r["embedding"], dtype=np.float32))
    if not per_artist:
        return 0
    names = sorted(per_artist)
    cents = torch.from_numpy(
        np.stack([np.stack(per_artist[a]).mean(axis=0) for a in names]))
    if len(names) < 4:   # too few points for any layout: place on a line
        coords = np.stack([np.arange(len(names), dtype=np.float32),
                           np.zeros(len(names), dtype=np.float32)], axis=1)
    else:
        try:
            from audiomuse_amd.engines.projection import umap_project

            coords = umap_project(cents.to(device), seed=0).cpu().numpy()
        except Exception:
            proj, _, _ = pca_fit_transform(cents.to(device), 2)
            coords = proj.cpu().numpy()
    buf = io.BytesIO()
    torch.save({"item_ids": names,
                "coords": torch.from_numpy(coords.astype(np.float32)),
                "n_tracks": [len(per_artist[a]) for a in names]}, buf)
    store_index_blob(conn, ARTIST_MAP, buf.getvalue(), meta={"n": len(names)})
    return len(names)


_REFRESHABLE = {AUDIO_INDEX: "embedding", CLAP_INDEX: "clap_embedding",
                LYRICS_INDEX: "lyrics_embedding"}


def refresh_ivf_index(conn: sqlite3.Connection, name: str,
                      device: str = "cpu",
                      max_drift: Optional[float] = None) -> Dict[str, int]:
    """Incremental IVF refresh: splice new/removed tracks into the stored
    packed index (IVFIndex.add/remove) instead of rebuilding.

    The reference rebuilds wholesale on every index task; with the index
    HBM-resident the splice is cheap, so between full rebuilds queries can
    see fresh tracks immediately. Falls back to a full rebuild when more
    than `max_drift` of the corpus changed (the coarse quantizer is only
    trained on the old distribution) or when no index exists yet.

    Returns {"added": a, "removed": r, "total": n, "rebuilt": 0|1}.
    """
    if max_drift is None:
        max_drift = C.IVF_REFRESH_MAX_DRIFT
    column = _REFRESHABLE[name]
    ids, mat = load_all_embeddings(conn, column)
    got = load_index_blob(conn, name)
    if got is None or not ids:
        n = {AUDIO_INDEX: build_audio_index, CLAP_INDEX: build_clap_index,
             LYRICS_INDEX: build_lyrics_index}[name](conn, device)
        return {"added": n, "removed": 0, "total": n, "rebuilt": 1}

    payload = torch.load(io.BytesIO(got[0]), map_location="cpu",
                         weights_only=True)
    index = IVFIndex.deserialize(_index_bytes(payload), device=device)
    old_ids: List[str] = payload["item_ids"]
    old_pos = {s: i for i, s in enumerate(old_ids)}
    new_pos = {s: i for i, s in enumerate(ids)}
    added = [s for s in ids if s not in old_pos]
    removed = [s for s in old_ids if s not in new_pos]
    changed = len(added) + len(removed)
    if changed == 0:
        return {"added": 0, "removed": 0, "total": len(ids), "rebuilt": 0}
    # Past max_drift the OLD coarse quantizer no longer matches the
    # distribution: splice the membership as usual, then RETRAIN the
    # quantizer in place from the resident rows (IVFIndex.retrain). The
    # corpus was already read once for the diff above — the former
    # full-rebuild path read it a second time inside build_*_index.
    heavy = changed > max_drift * max(len(ids), 1)

    # The engine's item_ids list maps packed int rows -> string ids; keep
    # the integer key stable per string id across the splice.
    if removed:
        index.remove(torch.tensor([old_pos[s] for s in removed]))
    item_ids = list(old_ids)
    if added:
        removed_set = set(removed)
        free = [i for i, s in enumerate(item_ids) if s in removed_set]
        rows = []
        for s in added:
            if free:
                slot = free.pop()
                item_ids[slot] = s
            else:
                slot = len(item_ids)
                item_ids.append(s)
            rows.append(slot)
        vecs = torch.from_numpy(mat[[new_pos[s] for s in added]])
        index.add(vecs, torch.tensor(rows, dtype=torch.int64))
    if heavy:
        index.retrain()
    _store_ivf(conn, name, index, item_ids)
    return {"added": len(added), "removed": len(removed),
            "total": index.n, "rebuilt": 1 if heavy else 0}


def run_all_index_builds(conn: sqlite3.Connection, device: str = "cpu",
                         progress_cb=None) -> Dict[str, int]:
    """The ordered build set (_run_all_index_builds, index.py:47).
    Audio IVF failure is fatal; the rest log and continue."""
    results: Dict[str, int] = {}
    results["audio"] = build_audio_index(conn, device)   # fatal on raise
    for name, fn in [("clap", build_clap_index),
                     ("lyrics", build_lyrics_index),
                     ("lyrics_axes", build_lyrics_axes_index),
                     ("semgrove", build_semgrove_index),
                     ("artist", lambda c, **k: build_artist_index(c)),
                     ("song_map", build_song_map),
                     ("artist_map", build_artist_map),
                     ("hyperbolic_tree", build_hyperbolic_tree_cache)]:
        try:
            results[name] = fn(conn, device=device) if name != "artist" \
                else build_artist_index(conn)
        except Exception:
            logger.exception("index build %s failed", name)
            results[name] = -1
        if progress_cb is not None:
            progress_cb(name, results[name])
    return results
