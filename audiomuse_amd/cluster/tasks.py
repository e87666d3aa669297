"""Clustering task handlers (queue entry points).

Reference: run_clustering_task (/root/reference/tasks/clustering.py:383)
-> per-server loop, batch jobs of ITERATIONS_PER_BATCH_JOB claimed by
sibling workers (run_clustering_batch_task :153), elite-pool absorb
(:1581), stall valve, winner post-processing + playlist creation.

The batch jobs carry their iteration seeds; results return through
task_status.result rows and the parent absorbs them (the same
Postgres-mediated reduction shape as the reference, on SQLite here).
"""

from __future__ import annotations

import json
import time
from typing import Dict, List

import numpy as np
import torch

from audiomuse_amd import config as C
from audiomuse_amd.cluster.evolve import (IterationResult, TrackRow,
                                          diverse_top_n, evolutionary_search)
from audiomuse_amd.db import write_txn
from audiomuse_amd.taskqueue import SUCCESS, enqueue, task_row
from audiomuse_amd.taskqueue import sql as qsql
from audiomuse_amd.taskqueue.worker import TaskContext, task_handler


def _load_catalogue(conn):
    """Clustering inputs: the 200-d embeddings when
    ENABLE_CLUSTERING_EMBEDDINGS (reference default), else the score
    feature vector (moods + other features + tempo/energy — the
    reference's pre-embedding clustering mode)."""
    rows = conn.execute(
        """SELECT s.item_id, s.title, s.author, s.mood_vector,
               s.other_features, s.tempo, s.energy, e.embedding
           FROM score s JOIN embedding e ON e.item_id = s.item_id""").fetchall()
    tracks: List[TrackRow] = []
    vecs: List[np.ndarray] = []
    use_emb = C.ENABLE_CLUSTERING_EMBEDDINGS
    for r in rows:
        moods = json.loads(r["mood_vector"] or "{}")
        others = json.loads(r["other_features"] or "{}")
        tracks.append(TrackRow(
            item_id=r["item_id"], title=r["title"] or "",
            author=r["author"] or "",
            mood_vector=moods, other_features=others))
        if use_emb:
            vecs.append(np.frombuffer(r["embedding"], dtype=np.float32))
        else:
            feat = ([moods.get(m, 0.0) for m in C.MOOD_LABELS]
                    + [others.get(m, 0.0) for m in C.OTHER_FEATURE_LABELS]
                    + [(r["tempo"] or 0.0) / 200.0, r["energy"] or 0.0])
            vecs.append(np.asarray(feat, dtype=np.float32))
    if not vecs:
        return [], torch.zeros(0, 0)
    return tracks, torch.from_numpy(np.stack(vecs))


def _calibrate(x, tracks, algorithm: str, device: str) -> Dict:
    """Calibration probes (reference _calibrate_cluster_params :775):
    up to CLUSTERING_CALIBRATION_MAX_TRIES quick single iterations; the
    first parameter set that yields >= 2 clusters seeds the batches."""
    import random as _random

    if not C.CLUSTERING_AUTO_CALIBRATION or x.numel() == 0:
        return {}
    from audiomuse_amd.cluster.evolve import _param_space, run_iteration
    rng = _random.Random(7)
    sub = min(512, x.shape[0])
    for _ in range(max(1, C.CLUSTERING_CALIBRATION_MAX_TRIES)):
        params = _param_space(algorithm, sub, rng)
        params["seed"] = rng.randint(0, 2**31 - 1)
        try:
            pick = rng.sample(range(x.shape[0]), sub)
            res = run_iteration(x[pick].to(device),
                                [tracks[i] for i in pick],
                                algorithm, dict(params))
            if len(res.playlists) >= 2:
                return params
        except Exception:
            continue
    return {}


@task_handler("run_clustering_batch")
def run_clustering_batch_task(ctx: TaskContext, payload: Dict) -> Dict:
    """ITERATIONS_PER_BATCH_JOB iterations; returns the batch's best
    (clustering.py:153)."""
    conn = ctx.conn
    tracks, x = _load_catalogue(conn)
    if x.numel() == 0:
        return {"best": None}
    device = ("cuda" if torch.cuda.is_available()
              and C.USE_GPU_CLUSTERING else "cpu")
    elites = evolutionary_search(
        x.to(device), tracks, payload.get("algorithm", C.CLUSTER_ALGORITHM),
        runs=payload.get("iterations", C.ITERATIONS_PER_BATCH_JOB),
        seed=payload.get("seed", 0), stall_limit=10**9,
        subset=payload.get("subset"),
        seed_params=payload.get("seed_params") or None,
        max_songs_per_cluster=payload.get("max_songs_per_cluster", 0),
        progress_cb=lambda i, n, s: ctx.report(100.0 * i / n, f"best {s:.4f}"))
    if not elites:
        return {"best": None}
    best = elites[0]
    return {"best": {"params": best.params, "fitness": best.fitness,
                     "playlists": best.playlists,
                     "centroids": best.centroids}}


@task_handler("run_clustering")
def run_clustering_task(ctx: TaskContext, payload: Dict) -> Dict:
    """Parent: dispatch batches, absorb, finalize winner
    (clustering.py:383/938/1581)."""
    conn = ctx.conn
    runs = int(payload.get("runs", C.CLUSTERING_RUNS))
    per_batch = int(payload.get("iterations_per_batch",
                                C.ITERATIONS_PER_BATCH_JOB))
    n_batches = max(1, (runs + per_batch - 1) // per_batch)
    algorithm = payload.get("algorithm", C.CLUSTER_ALGORITHM)
    max_songs = int(payload.get("max_songs_per_cluster",
                                C.CLUSTERING_MAX_PLAYLIST_SONGS))

    # calibration probes seed the search (reference :775)
    tracks, x = _load_catalogue(conn)
    device = ("cuda" if torch.cuda.is_available()
              and C.USE_GPU_CLUSTERING else "cpu")
    seed_params = _calibrate(x, tracks, algorithm, device)
    if seed_params:
        ctx.report(5.0, f"calibrated {seed_params}")

    # dispatch with back-pressure, absorbing incrementally; no
    # improvement across CLUSTERING_EARLY_STOP_BATCHES absorbed batches
    # stops dispatching (reference early-stop), and the stall valve
    # bounds total wall time (CLUSTERING_STALL_TIMEOUT_MINUTES,
    # reference :1426-1449)
    child_ids: List[str] = []
    absorbed: set = set()
    best = None
    failed = 0
    no_improve = 0
    stall_deadline = time.time() + payload.get(
        "drain_timeout", C.CLUSTERING_STALL_TIMEOUT_MINUTES * 60.0)

    def _absorb() -> None:
        nonlocal best, failed, no_improve
        for tid in child_ids:
            if tid in absorbed:
                continue
            row = task_row(conn, tid)
            if row is None:
                absorbed.add(tid)
                failed += 1
                continue
            if row["status"] not in ("SUCCESS", "FAILURE", "REVOKED"):
                continue
            absorbed.add(tid)
            if row["status"] != SUCCESS:
                failed += 1
                continue
            result = json.loads(row["result"] or "{}").get("best")
            if not result:
                no_improve += 1
                continue
            score = result["fitness"].get("fitness_score", -1.0)
            if best is None or score > best["fitness"].get(
                    "fitness_score", -1.0):
                best = result
                no_improve = 0
            else:
                no_improve += 1

    for b in range(n_batches):
        while qsql.pending_children(conn, ctx.task_id) >= C.MAX_CONCURRENT_BATCH_JOBS:
            ctx.check_cancelled()
            _absorb()
            time.sleep(C.QUEUE_POLL_SECONDS)
        _absorb()
        if best is not None and no_improve >= C.CLUSTERING_EARLY_STOP_BATCHES:
            ctx.report(50.0, f"early stop after {len(absorbed)} batches")
            break
        if time.time() > stall_deadline:
            ctx.report(50.0, "stall valve: forcing completion")
            break
        child_ids.append(enqueue(
            conn, "run_clustering_batch",
            {"algorithm": algorithm, "iterations": per_batch,
             "seed": 1000 + b, "seed_params": seed_params,
             "max_songs_per_cluster": max_songs},
            parent_task_id=ctx.task_id))
        ctx.report(10.0 + 40.0 * (b + 1) / n_batches,
                   f"batch {b + 1}/{n_batches} dispatched")

    while time.time() < stall_deadline:
        ctx.check_cancelled()
        if qsql.pending_children(conn, ctx.task_id) == 0:
            break
        time.sleep(C.QUEUE_POLL_SECONDS)
    _absorb()

    if failed > C.CLUSTERING_MAX_FAILED_BATCHES:
        raise RuntimeError(f"{failed} clustering batches failed")
    if best is None:
        return {"playlists": 0, "failed_batches": failed}

    winner = IterationResult(params=best["params"], fitness=best["fitness"],
                             playlists=best["playlists"],
                             centroids=best["centroids"])
    if C.CLUSTERING_CLEANING:
        top = diverse_top_n(winner, n=payload.get("top_n", C.TOP_N_PLAYLISTS))
    else:
        top = dict(winner.playlists)   # cleaning off: keep the raw winner

    # playlist-name history: avoid reusing names from the last
    # PLAYLIST_NAME_HISTORY_ROUNDS finalizations (reference
    # CLUSTER_NAMING_AI_HISTORY + playlist_name_history table)
    if C.CLUSTER_NAMING_AI_HISTORY:
        hist_n = C.PLAYLIST_NAME_HISTORY_ROUNDS * max(len(top), 1)
        recent = {r["name"] for r in conn.execute(
            "SELECT name FROM playlist_name_history "
            "ORDER BY id DESC LIMIT ?", (hist_n,)).fetchall()}
        renamed = {}
        for name, ids in top.items():
            new, i = name, 2
            while new in recent or new in renamed:
                base = name[:-len("_automatic")] \
                    if name.endswith("_automatic") else name
                new = f"{base} ({i})_automatic"
                i += 1
            renamed[new] = ids
        top = renamed

    # persist playlists (reference also pushes to the media server and
    # deletes old _automatic ones, mediaserver/__init__.py:321)
    with write_txn(conn):
        conn.execute("DELETE FROM playlist WHERE kind='automatic'")
        for name, ids in top.items():
            conn.execute(
                "INSERT INTO playlist (name, item_ids, kind) VALUES (?,?,?)",
                (name, json.dumps(ids), "automatic"))
            conn.execute(
                "INSERT INTO playlist_name_history (name) VALUES (?)", (name,))
    if payload.get("server_type"):
        try:
            from audiomuse_amd.mediaserver import make_provider

            provider = make_provider(payload["server_type"],
                                     **payload.get("server_config", {}))
            provider.delete_automatic_playlists()
            for name, ids in top.items():
                provider.create_or_replace_playlist(name, ids)
        except Exception:
            pass
    ctx.report(100.0, f"{len(top)} playlists")
    return {"playlists": len(top), "failed_batches": failed,
            "best_score": best["fitness"].get("fitness_score")}
