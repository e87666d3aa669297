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
    rows = conn.execute(
        """SELECT s.item_id, s.title, s.author, s.mood_vector,
               s.other_features, e.embedding
           FROM score s JOIN embedding e ON e.item_id = s.item_id""").fetchall()
    tracks: List[TrackRow] = []
    vecs: List[np.ndarray] = []
    for r in rows:
        tracks.append(TrackRow(
            item_id=r["item_id"], title=r["title"] or "",
            author=r["author"] or "",
            mood_vector=json.loads(r["mood_vector"] or "{}"),
            other_features=json.loads(r["other_features"] or "{}")))
        vecs.append(np.frombuffer(r["embedding"], dtype=np.float32))
    if not vecs:
        return [], torch.zeros(0, 0)
    return tracks, torch.from_numpy(np.stack(vecs))


@task_handler("run_clustering_batch")
def run_clustering_batch_task(ctx: TaskContext, payload: Dict) -> Dict:
    """ITERATIONS_PER_BATCH_JOB iterations; returns the batch's best
    (clustering.py:153)."""
    conn = ctx.conn
    tracks, x = _load_catalogue(conn)
    if x.numel() == 0:
        return {"best": None}
    device = "cuda" if torch.cuda.is_available() else "cpu"
    elites = evolutionary_search(
        x.to(device), tracks, payload.get("algorithm", C.CLUSTER_ALGORITHM),
        runs=payload.get("iterations", C.ITERATIONS_PER_BATCH_JOB),
        seed=payload.get("seed", 0), stall_limit=10**9,
        subset=payload.get("subset"),
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

    child_ids: List[str] = []
    for b in range(n_batches):
        while qsql.pending_children(conn, ctx.task_id) >= C.MAX_CONCURRENT_BATCH_JOBS:
            ctx.check_cancelled()
            time.sleep(C.QUEUE_POLL_SECONDS)
        child_ids.append(enqueue(
            conn, "run_clustering_batch",
            {"algorithm": algorithm, "iterations": per_batch,
             "seed": 1000 + b,
             "max_songs_per_cluster": payload.get("max_songs_per_cluster", 0)},
            parent_task_id=ctx.task_id))
        ctx.report(10.0 + 40.0 * (b + 1) / n_batches,
                   f"batch {b + 1}/{n_batches} dispatched")

    deadline = time.time() + payload.get("drain_timeout", 3600.0)
    while time.time() < deadline:
        ctx.check_cancelled()
        if qsql.pending_children(conn, ctx.task_id) == 0:
            break
        time.sleep(C.QUEUE_POLL_SECONDS)

    # absorb (clustering.py:1581): fold batch-best results; tolerate
    # CLUSTERING_MAX_FAILED_BATCHES dead batches
    best = None
    failed = 0
    for tid in child_ids:
        row = task_row(conn, tid)
        if row is None or row["status"] != SUCCESS:
            failed += 1
            continue
        result = json.loads(row["result"] or "{}").get("best")
        if not result:
            continue
        score = result["fitness"].get("fitness_score", -1.0)
        if best is None or score > best["fitness"].get("fitness_score", -1.0):
            best = result
    if failed > C.CLUSTERING_MAX_FAILED_BATCHES:
        raise RuntimeError(f"{failed} clustering batches failed")
    if best is None:
        return {"playlists": 0, "failed_batches": failed}

    winner = IterationResult(params=best["params"], fitness=best["fitness"],
                             playlists=best["playlists"],
                             centroids=best["centroids"])
    top = diverse_top_n(winner, n=payload.get("top_n", C.TOP_N_PLAYLISTS))

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
