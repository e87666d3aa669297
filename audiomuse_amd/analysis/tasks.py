"""Analysis task handlers (queue entry points).

Reference call stack: run_analysis_task
(/root/reference/tasks/analysis/main.py:777) -> per-server phases,
work map vs track_server_map, child analyze_album_task dispatch with
back-pressure <= MAX_QUEUED_ANALYSIS_JOBS, drain/monitor with
cooperative cancel, periodic + final rebuild_all_indexes_task.
analyze_album_task (album.py:393) stages per track: download ->
analyze (MusiCNN + features + CLAP batched on GPU) -> identity
(simhash resolve) -> persist.
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Dict, List, Optional

import numpy as np
import torch

from audiomuse_amd import config as C
from audiomuse_amd.analysis.index import run_all_index_builds
from audiomuse_amd.analysis.pipeline import AnalysisRuntime
from audiomuse_amd.db import write_txn
from audiomuse_amd.db.store import (save_clap_embedding,
                                    save_lyrics_embedding,
                                    save_track_analysis_and_embedding)
from audiomuse_amd.engines.simhash import CatalogResolver
from audiomuse_amd.mediaserver import make_provider
from audiomuse_amd.taskqueue import enqueue
from audiomuse_amd.taskqueue import sql as qsql
from audiomuse_amd.taskqueue.worker import TaskContext, task_handler

logger = logging.getLogger(__name__)

_RUNTIME: Optional[AnalysisRuntime] = None
_RUNTIME_LOCK = threading.Lock()


def get_runtime() -> AnalysisRuntime:
    """One model runtime per worker process (one process per GPU rank).
    Lock guards the threaded-workers deployment (several Worker threads
    sharing one GPU — scripts/e2e_soak.py --workers)."""
    global _RUNTIME
    if _RUNTIME is None:
        with _RUNTIME_LOCK:
            if _RUNTIME is None:
                device = "cuda" if torch.cuda.is_available() else "cpu"
                _RUNTIME = AnalysisRuntime(device=device)
    return _RUNTIME


_RESOLVER: Optional[CatalogResolver] = None
_RESOLVER_EXPECT_N = -1


def _resolver_from_db(conn) -> CatalogResolver:
    """Per-worker cached resolver. A fresh load is O(catalogue); reloading
    per album would be O(N^2) across a library scan. The cache tracks the
    ids it minted itself; if the DB grew beyond that, a sibling worker
    wrote rows we have not seen -> rebuild. A short blind window between
    siblings can mint duplicate canonical ids; the duplicate-repair
    migration collapses those (reference tolerates the same race)."""
    global _RESOLVER, _RESOLVER_EXPECT_N
    n = conn.execute("SELECT COUNT(*) FROM embedding").fetchone()[0]
    if _RESOLVER is not None and n <= _RESOLVER_EXPECT_N:
        return _RESOLVER
    resolver = CatalogResolver()
    rows = conn.execute(
        """SELECT e.item_id, e.embedding, s.duration FROM embedding e
           LEFT JOIN score s ON s.item_id = e.item_id""").fetchall()
    for r in rows:
        emb = np.frombuffer(r["embedding"], dtype=np.float32)
        resolver.register_existing(r["item_id"], emb, r["duration"] or 0.0)
    _RESOLVER = resolver
    _RESOLVER_EXPECT_N = n
    return resolver


def _analyzed_provider_ids(conn, server_id: str) -> set:
    return {r["provider_id"] for r in conn.execute(
        "SELECT provider_id FROM track_server_map WHERE server_id=?",
        (server_id,))}


@task_handler("analyze_album")
def analyze_album_task(ctx: TaskContext, payload: Dict) -> Dict:
    """Child task: one album end-to-end (album.py:393)."""
    conn = ctx.conn
    server_id = payload["server_id"]
    provider = make_provider(payload["server_type"],
                             **payload.get("server_config", {}))
    runtime = get_runtime()
    resolver = _resolver_from_db(conn)
    done = _analyzed_provider_ids(conn, server_id)

    tracks = provider.get_tracks_from_album(payload["album_id"])
    todo = [t for t in tracks if t.provider_id not in done]
    ctx.check_cancelled()
    # parallel downloads (reference downloads serially, album.py:290;
    # provider IO and WAV decode release the GIL, so a small pool
    # overlaps network/decode with the GPU batch of the previous album)
    if len(todo) > 1:
        from concurrent.futures import ThreadPoolExecutor

        from audiomuse_amd.utils.resources import usable_cpu_count

        workers = min(4, max(1, usable_cpu_count() - 1), len(todo))
        with ThreadPoolExecutor(max_workers=workers) as pool:
            blobs = list(pool.map(
                lambda t: provider.download_track(t.provider_id), todo))
    else:
        blobs = [provider.download_track(t.provider_id) for t in todo]
    ctx.check_cancelled()
    valid = [(t, b) for t, b in zip(todo, blobs) if b is not None]
    if not valid:
        return {"analyzed": 0, "skipped": len(tracks) - len(todo)}

    results = runtime.analyze_album_batch([b for _, b in valid])
    runtime.recycle_models()        # PER_SONG_MODEL_RELOAD compat switch
    n = 0
    # one write transaction per album (write_txn is reentrant: the
    # nested save_* helpers join it) — one fsync instead of ~3/track.
    # The txn's write lock also serializes resolver.resolve() across
    # threaded sibling workers (resolve only runs inside this section).
    with write_txn(conn):
        n = _persist_album(ctx, conn, valid, results, resolver, server_id,
                           runtime, provider)
    return {"analyzed": n, "skipped": len(tracks) - len(todo)}


def _persist_album(ctx, conn, valid, results, resolver, server_id,
                   runtime, provider) -> int:
    n = 0
    for (track, _blob), res in zip(valid, results):
        ctx.check_cancelled()
        if res is None:
            continue

        def _fp_confirm(cand_id: str) -> bool:
            """Chromaprint confirm gate (reference
            CHROMAPRINT_GATE_ENABLED + MATCH_THRESHOLD): merging two
            provider tracks into one recording requires the acoustic
            fingerprints to agree when both exist."""
            if not C.CHROMAPRINT_GATE_ENABLED or res.chromaprint is None:
                return True
            row = conn.execute(
                "SELECT fingerprint FROM chromaprint WHERE item_id=?",
                (cand_id,)).fetchone()
            if row is None or row["fingerprint"] is None:
                return True
            from audiomuse_amd.engines.chromaprint import bit_match_ratio
            return bit_match_ratio(res.chromaprint, row["fingerprint"]) \
                >= C.CHROMAPRINT_MATCH_THRESHOLD

        item_id, _matched = resolver.resolve(
            res.embedding, res.duration, server_id, track.provider_id,
            confirm_fn=_fp_confirm)
        if res.chromaprint is not None:
            conn.execute(
                """INSERT INTO chromaprint (item_id, fingerprint, duration)
                   VALUES (?,?,?)
                   ON CONFLICT(item_id) DO UPDATE SET
                       fingerprint=excluded.fingerprint,
                       duration=excluded.duration""",
                (item_id, res.chromaprint, res.duration))
        global _RESOLVER_EXPECT_N
        if not _matched:
            _RESOLVER_EXPECT_N += 1     # we will insert one embedding row
        save_track_analysis_and_embedding(
            conn, item_id, title=track.title, author=track.author,
            album=track.album, tempo=res.tempo, key=res.key, scale=res.scale,
            mood_vector=res.moods, other_features=res.other_features,
            energy=res.energy, duration=res.duration, embedding=res.embedding)
        if res.clap_embedding is not None:
            save_clap_embedding(conn, item_id, res.clap_embedding)
        if C.LYRICS_ENABLED:
            # stage 8 (album.py:276): provided lyrics win; ASR optional
            lyr = runtime.lyrics_pipeline().analyze(
                provided_lyrics=provider.get_lyrics(track.provider_id),
                title=track.title, artist=track.author)
            save_lyrics_embedding(conn, item_id, lyr.embedding,
                                  axis_scores=lyr.axis_scores,
                                  lyrics_text=lyr.text, language=lyr.language,
                                  instrumental=lyr.instrumental)
        # plugin hook (reference: song.py:101)
        from audiomuse_amd.plugin import hook_registry

        hook_registry.fire("song_analyzed", item_id,
                           {"tempo": res.tempo, "energy": res.energy,
                            "moods": res.moods})
        with write_txn(conn):
            conn.execute(
                """INSERT INTO track_server_map
                       (provider_id, server_id, item_id, title, author, album,
                        file_path)
                   VALUES (?,?,?,?,?,?,?)
                   ON CONFLICT(provider_id, server_id)
                   DO UPDATE SET item_id=excluded.item_id""",
                (track.provider_id, server_id, item_id, track.title,
                 track.author, track.album, track.file_path))
        n += 1
    return n


@task_handler("run_analysis")
def run_analysis_task(ctx: TaskContext, payload: Dict) -> Dict:
    """Parent task (main.py:777): preflight, work map, child dispatch with
    back-pressure, drain, final index rebuild."""
    conn = ctx.conn
    server_id = payload.get("server_id", "default")
    provider = make_provider(payload["server_type"],
                             **payload.get("server_config", {}))
    if not provider.test_connection():        # main.py:294 preflight
        raise RuntimeError(f"media server {server_id!r} unreachable")

    # NUM_RECENT_ALBUMS (reference): 0 = whole library, else only the
    # newest N albums are scanned (the incremental-analysis mode).
    # Standalone/loose tracks ride provider.get_recent_music_items
    # (Emby exposes them as single-track pseudo-albums).
    limit = int(payload.get("album_limit", C.NUM_RECENT_ALBUMS))
    albums = provider.get_recent_music_items(limit=limit)
    ctx.report(5.0, f"{len(albums)} albums to scan")

    child_ids: List[str] = []
    dispatched = 0
    rebuild_every = max(int(C.REBUILD_INDEX_BATCH_SIZE), 1)
    for album in albums:
        ctx.check_cancelled()
        # back-pressure (main.py: <= MAX_QUEUED_ANALYSIS_JOBS live children)
        while qsql.pending_children(conn, ctx.task_id) >= C.MAX_QUEUED_ANALYSIS_JOBS:
            ctx.check_cancelled()
            time.sleep(C.QUEUE_POLL_SECONDS)
        child_ids.append(enqueue(
            conn, "analyze_album",
            {"server_type": payload["server_type"],
             "server_config": payload.get("server_config", {}),
             "server_id": server_id, "album_id": album.provider_id},
            parent_task_id=ctx.task_id))
        dispatched += 1
        # mid-run index rebuild every REBUILD_INDEX_BATCH_SIZE albums
        # (main.py: features go live before the full scan finishes)
        if dispatched % rebuild_every == 0:
            enqueue(conn, "rebuild_indexes", {},
                    parent_task_id=ctx.task_id)
        ctx.report(5.0 + 60.0 * dispatched / max(len(albums), 1),
                   f"dispatched {dispatched}/{len(albums)}")

    # drain loop (main.py stage 4; cadence ANALYSIS_MONITOR_DB_INTERVAL)
    deadline = time.time() + payload.get("drain_timeout", 3600.0)
    while time.time() < deadline:
        ctx.check_cancelled()
        left = qsql.pending_children(conn, ctx.task_id)
        if left == 0:
            break
        ctx.report(65.0 + 25.0 * (1 - left / max(dispatched, 1)),
                   f"{left} album jobs outstanding")
        time.sleep(min(C.ANALYSIS_MONITOR_DB_INTERVAL,
                       C.QUEUE_POLL_SECONDS * 8))

    ctx.report(92.0, "rebuilding indexes")
    device = "cuda" if torch.cuda.is_available() else "cpu"
    built = run_all_index_builds(conn, device=device)
    ctx.report(100.0, "done")
    return {"albums": dispatched, "indexes": built}


@task_handler("chromaprint_backfill")
def chromaprint_backfill_task(ctx: TaskContext, payload: Dict) -> Dict:
    """Backfill fingerprints for tracks analyzed before collection was
    enabled (reference CHROMAPRINT_BACKFILL_ALBUMS_PER_RUN: bounded
    batches per run so the cron job never monopolizes a worker)."""
    from audiomuse_amd.engines import chromaprint as cp
    from audiomuse_amd.ops.audio_io import load_audio

    conn = ctx.conn
    server_id = payload.get("server_id", "default")
    provider = make_provider(payload.get("server_type", C.MEDIASERVER_TYPE),
                             **payload.get("server_config", {}))
    limit = int(payload.get("albums_per_run",
                            C.CHROMAPRINT_BACKFILL_ALBUMS_PER_RUN))
    rows = conn.execute(
        """SELECT m.provider_id, m.item_id FROM track_server_map m
           LEFT JOIN chromaprint c ON c.item_id = m.item_id
           WHERE m.server_id = ? AND c.item_id IS NULL
           LIMIT ?""", (server_id, limit * 10)).fetchall()
    done = 0
    for r in rows:
        ctx.check_cancelled()
        blob = provider.download_track(r["provider_id"])
        if blob is None:
            continue
        audio, sr = load_audio(blob)
        if audio is None:
            continue
        try:
            fp = cp.compute(audio, sr)
        except Exception:  # noqa: BLE001
            continue
        with write_txn(conn):
            conn.execute(
                """INSERT INTO chromaprint (item_id, fingerprint, duration)
                   VALUES (?,?,?) ON CONFLICT(item_id) DO NOTHING""",
                (r["item_id"], fp, audio.shape[-1] / sr))
        done += 1
        ctx.report(100.0 * done / max(len(rows), 1), f"{done} fingerprints")
    return {"backfilled": done}


@task_handler("rebuild_indexes")
def rebuild_indexes_task(ctx: TaskContext, payload: Dict) -> Dict:
    device = "cuda" if torch.cuda.is_available() else "cpu"
    built = run_all_index_builds(
        ctx.conn, device=device,
        progress_cb=lambda name, n: ctx.report(50.0, f"{name}: {n}"))
    return {"indexes": built}


@task_handler("refresh_indexes")
def refresh_indexes_task(ctx: TaskContext, payload: Dict) -> Dict:
    """Incremental IVF refresh (index.refresh_ivf_index): splice
    new/removed tracks into the stored packed indexes between full
    rebuilds. MI355X-native extra over the reference's wholesale
    rebuild path."""
    from audiomuse_amd.analysis import index as idx_mod

    device = "cuda" if torch.cuda.is_available() else "cpu"
    out: Dict[str, Dict] = {}
    only = payload.get("only")          # per-family refresh (cache/refresh
    names = [n for n in idx_mod._REFRESHABLE   # routes) or all families
             if not only or n == only]
    for name in names:
        try:
            out[name] = idx_mod.refresh_ivf_index(ctx.conn, name,
                                                  device=device)
        except Exception:
            logger.exception("incremental refresh %s failed", name)
            out[name] = {"error": 1}
        ctx.report(50.0, f"{name}: {out[name]}")
    return {"indexes": out}
