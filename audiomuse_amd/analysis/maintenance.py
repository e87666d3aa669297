"""Maintenance tasks: multi-server sweep, cleaning, backup, dashboard.

References:
- multi-server sweep: /root/reference/tasks/multiserver_sync.py (750 LoC)
  — metadata-only alignment of a server's tracks onto existing canonical
  recordings by tiers (exact path / path tail / exact title+artist /
  noise-normalized title+artist), prune guarded by
  SWEEP_PRUNE_MIN_FETCH_RATIO.
- cleaning: tasks/cleaning.py (:48) — per-server stale-mapping prune and
  orphan album report/delete behind a triple guard; the catalogue itself
  is never deleted (test_catalogue_is_never_deleted invariant).
- backup: app_backup.py (pg_dump/restore + lock) — here the SQLite
  backup API with the same restore-lock semantics.
- dashboard: app_dashboard.py snapshot stats.
"""

from __future__ import annotations

import json
import re
import sqlite3
import time
from typing import Dict, List

from audiomuse_amd import config as C
from audiomuse_amd.db import connect, write_txn
from audiomuse_amd.mediaserver import make_provider
from audiomuse_amd.taskqueue.worker import TaskContext, task_handler

_NOISE = re.compile(r"\s*[\(\[].*?[\)\]]\s*|\s*(feat\.|ft\.)\s.*$|[^\w\s]",
                    re.IGNORECASE)


def normalize_title(s: str) -> str:
    """Noise-normalized matching key (remaster tags, feat. credits,
    punctuation stripped). casefold(), not lower(): unicode caseless
    matching must be stable through case round trips (µ -> Μ -> μ) —
    found by the property test."""
    return re.sub(r"\s+", " ",
                  _NOISE.sub(" ", (s or "").casefold())).strip()


def align_server_tracks(conn: sqlite3.Connection, server_id: str,
                        tracks: List) -> Dict[str, int]:
    """Tiered metadata alignment (multiserver_sync.enqueue_server_
    alignment :69): map provider tracks onto existing canonical ids
    without re-analysis."""
    existing = conn.execute(
        """SELECT m.item_id, m.file_path, s.title, s.author
           FROM track_server_map m JOIN score s ON s.item_id = m.item_id"""
    ).fetchall()
    by_path = {r["file_path"]: r["item_id"] for r in existing if r["file_path"]}
    by_tail = {r["file_path"].rsplit("/", 1)[-1]: r["item_id"]
               for r in existing if r["file_path"]}
    by_exact = {(r["title"] or "", r["author"] or ""): r["item_id"]
                for r in existing}
    by_norm = {(normalize_title(r["title"]), normalize_title(r["author"])):
               r["item_id"] for r in existing}
    tiers = {"path": 0, "tail": 0, "exact": 0, "normalized": 0, "unmatched": 0}
    with write_txn(conn):
        for t in tracks:
            item_id = None
            if t.file_path and t.file_path in by_path:
                item_id, tier = by_path[t.file_path], "path"
            elif t.file_path and t.file_path.rsplit("/", 1)[-1] in by_tail:
                item_id, tier = by_tail[t.file_path.rsplit("/", 1)[-1]], "tail"
            elif (t.title, t.author) in by_exact:
                item_id, tier = by_exact[(t.title, t.author)], "exact"
            elif (normalize_title(t.title), normalize_title(t.author)) in by_norm:
                item_id = by_norm[(normalize_title(t.title),
                                   normalize_title(t.author))]
                tier = "normalized"
            else:
                tiers["unmatched"] += 1
                continue
            tiers[tier] += 1
            conn.execute(
                """INSERT INTO track_server_map (provider_id, server_id,
                       item_id, title, author, album, file_path)
                   VALUES (?,?,?,?,?,?,?)
                   ON CONFLICT(provider_id, server_id)
                   DO UPDATE SET item_id=excluded.item_id""",
                (t.provider_id, server_id, item_id, t.title, t.author,
                 t.album, t.file_path))
    return tiers


@task_handler("multiserver_sync")
def multiserver_sync_task(ctx: TaskContext, payload: Dict) -> Dict:
    provider = make_provider(payload["server_type"],
                             **payload.get("server_config", {}))
    server_id = payload.get("server_id", "default")
    tracks = provider.get_all_songs()
    tiers = align_server_tracks(ctx.conn, server_id, tracks)

    # prune mappings whose provider track vanished — guarded: skip when
    # the fetch looks partial (SWEEP_PRUNE_MIN_FETCH_RATIO)
    conn = ctx.conn
    mapped = conn.execute(
        "SELECT COUNT(*) AS n FROM track_server_map WHERE server_id=?",
        (server_id,)).fetchone()["n"]
    pruned = 0
    if mapped and len(tracks) / mapped >= C.SWEEP_PRUNE_MIN_FETCH_RATIO:
        live = {t.provider_id for t in tracks}
        rows = conn.execute(
            "SELECT provider_id FROM track_server_map WHERE server_id=?",
            (server_id,)).fetchall()
        stale = [r["provider_id"] for r in rows if r["provider_id"] not in live]
        with write_txn(conn):
            for pid in stale:
                conn.execute(
                    "DELETE FROM track_server_map WHERE provider_id=? AND "
                    "server_id=?", (pid, server_id))
                pruned += 1
    return {"tiers": tiers, "pruned": pruned, "fetched": len(tracks)}


@task_handler("clean_orphans")
def clean_orphans_task(ctx: TaskContext, payload: Dict) -> Dict:
    """Orphan report/delete with the triple guard (cleaning.py:48):
    only mappings are removed, never catalogue rows (`score`/`embedding`
    are append-only — the reference's hardest invariant)."""
    conn = ctx.conn
    do_delete = bool(payload.get("delete", False))
    orphans = conn.execute(
        """SELECT m.provider_id, m.server_id FROM track_server_map m
           LEFT JOIN score s ON s.item_id = m.item_id
           WHERE s.item_id IS NULL""").fetchall()
    deleted = 0
    if do_delete and orphans:
        # triple guard: explicit flag + bounded fraction + re-check
        total = conn.execute(
            "SELECT COUNT(*) AS n FROM track_server_map").fetchone()["n"]
        # triple guard part 2: bounded fraction AND absolute safety cap
        # (reference CLEANING_SAFETY_LIMIT)
        if (total and len(orphans) / total <= payload.get("max_fraction", 0.2)
                and len(orphans) <= C.CLEANING_SAFETY_LIMIT):
            with write_txn(conn):
                for r in orphans:
                    conn.execute(
                        "DELETE FROM track_server_map WHERE provider_id=? "
                        "AND server_id=?",
                        (r["provider_id"], r["server_id"]))
                    deleted += 1
    # CLEANING_CATALOGUE (opt-in, default OFF — the catalogue is
    # append-only otherwise; reference keeps the same explicit switch):
    # catalogue rows with NO mapping on ANY server may be purged.
    purged = 0
    if do_delete and C.CLEANING_CATALOGUE:
        rows = conn.execute(
            """SELECT s.item_id FROM score s
               LEFT JOIN track_server_map m ON m.item_id = s.item_id
               WHERE m.item_id IS NULL LIMIT ?""",
            (C.CLEANING_SAFETY_LIMIT,)).fetchall()
        with write_txn(conn):
            for r in rows:
                for table in ("embedding", "clap_embedding",
                              "lyrics_embedding", "chromaprint", "score"):
                    conn.execute(f"DELETE FROM {table} WHERE item_id=?",
                                 (r["item_id"],))
                purged += 1
    return {"orphans": len(orphans), "deleted": deleted,
            "catalogue_purged": purged}


@task_handler("sonic_fingerprint")
def sonic_fingerprint_task(ctx: TaskContext, payload: Dict) -> Dict:
    """Cron-able fingerprint playlist (reference:
    sonic_fingerprint_manager cron task :33): top-played -> recency
    weights -> taste vector -> IVF expansion -> stored playlist."""
    import json as _json

    import numpy as np
    import torch

    from audiomuse_amd.analysis.index import AUDIO_INDEX, load_ivf_engine
    from audiomuse_amd.engines.misc import sonic_fingerprint

    conn = ctx.conn
    provider = make_provider(payload.get("server_type", "synthetic"),
                             **payload.get("server_config", {}))
    server_id = payload.get("server_id", "default")
    eng = load_ivf_engine(conn, AUDIO_INDEX)
    if eng is None:
        return {"error": "audio index not built"}
    top = provider.get_top_played_songs(
        payload.get("top_n", C.SONIC_FINGERPRINT_TOP_PLAYED))
    mapped = {r["provider_id"]: r["item_id"] for r in conn.execute(
        "SELECT provider_id, item_id FROM track_server_map WHERE server_id=?",
        (server_id,))}
    vecs, times = [], []
    per_album: Dict[str, int] = {}
    cap_album = int(payload.get("max_per_album",
                                C.SONIC_FINGERPRINT_MAX_SONGS_PER_ALBUM))
    for t in top:
        # per-album cap on the SEEDS (reference
        # SONIC_FINGERPRINT_MAX_SONGS_PER_ALBUM: one heavy-rotation
        # album must not dominate the taste vector)
        if cap_album and t.album:
            if per_album.get(t.album, 0) >= cap_album:
                continue
            per_album[t.album] = per_album.get(t.album, 0) + 1
        iid = mapped.get(t.provider_id)
        v = eng.vector_for_id(iid) if iid else None
        if v is None:
            continue
        vecs.append(v.cpu().numpy())
        times.append(provider.get_last_played_time(t.provider_id) or 0.0)
    if not vecs:
        return {"tracks": 0}
    fp = sonic_fingerprint(np.stack(vecs), times)
    # result size + per-seed neighborhood expansion (reference
    # SONIC_FINGERPRINT_TOP_N_SONGS / SONIC_FINGERPRINT_NEIGHBORS: the
    # taste-vector hits are widened with each seed's own neighbors)
    n_out = int(payload.get("n", C.SONIC_FINGERPRINT_TOP_N_SONGS))
    res = eng.find_similar_by_vector(torch.from_numpy(fp), n_out)
    if C.SONIC_FINGERPRINT_NEIGHBORS > 0:
        seen = {r["item_id"] for r in res}
        extras = []
        for v in vecs[: max(n_out // 5, 3)]:
            for r in eng.find_similar_by_vector(
                    torch.from_numpy(v), C.SONIC_FINGERPRINT_NEIGHBORS):
                if r["item_id"] not in seen:
                    seen.add(r["item_id"])
                    extras.append(r)
        res = (res + sorted(extras, key=lambda r: r["distance"]))[:n_out]
    name = payload.get("name", C.SONIC_FINGERPRINT_CRON_PLAYLIST_NAME)
    with write_txn(conn):
        conn.execute("DELETE FROM playlist WHERE name=?", (name,))
        conn.execute(
            "INSERT INTO playlist (name, server_id, item_ids, kind) "
            "VALUES (?,?,?, 'fingerprint')",
            (name, server_id, _json.dumps([r["item_id"] for r in res])))
        conn.execute("INSERT INTO playlist_name_history (name) VALUES (?)",
                     (name,))
    return {"tracks": len(res)}


# -- backup / restore (app_backup.py analog) --------------------------------

def _schema_tables() -> list:
    """Table names from the canonical DDL (one catalogue, both backends)."""
    import re

    from audiomuse_amd.db.schema import DDL
    return re.findall(r"CREATE TABLE IF NOT EXISTS (\w+)", DDL)


def backup_database(conn, dest_path: str) -> None:
    """Portable logical backup: the artifact is always an SQLite file.

    SQLite backend: the online-backup API snapshots consistently on its
    own. PostgreSQL backend: every table is streamed into a
    canonical-schema SQLite file inside one REPEATABLE READ snapshot
    (the stand-in for the reference's pg_dump, app_backup.py:607 — no
    pg_dump binary ships in this image)."""
    from audiomuse_amd.db import backend_kind
    if backend_kind(conn) == "sqlite":
        dest = sqlite3.connect(dest_path)
        try:
            conn.backup(dest)
        finally:
            dest.close()
        return
    from audiomuse_amd.db.schema import init_db
    dest = sqlite3.connect(dest_path)
    try:
        init_db(dest)
        conn.execute("BEGIN ISOLATION LEVEL REPEATABLE READ")
        try:
            for table in _schema_tables():
                rows = conn.execute(f"SELECT * FROM {table}").fetchall()
                if not rows:
                    continue
                cols = rows[0].keys()
                marks = ",".join("?" for _ in cols)
                dest.executemany(
                    f"INSERT INTO {table} ({','.join(cols)}) "
                    f"VALUES ({marks})",
                    [tuple(r) for r in rows])
        finally:
            conn.execute("COMMIT")
        dest.commit()
    finally:
        dest.close()


def restore_database(conn, src_sqlite_path: str) -> None:
    """Inverse of backup_database for the PostgreSQL backend: truncate
    and reload every table from the SQLite backup artifact in one
    transaction."""
    src = sqlite3.connect(src_sqlite_path)
    src.row_factory = sqlite3.Row
    try:
        with write_txn(conn):
            for table in reversed(_schema_tables()):  # FK-safe delete order
                conn.execute(f"DELETE FROM {table}")
            for table in _schema_tables():
                rows = src.execute(f"SELECT * FROM {table}").fetchall()
                if not rows:
                    continue
                cols = rows[0].keys()
                marks = ",".join("?" for _ in cols)
                for r in rows:
                    conn.execute(
                        f"INSERT INTO {table} ({','.join(cols)}) "
                        f"VALUES ({marks})", tuple(r))
    finally:
        src.close()


def refresh_dashboard_stats(conn: sqlite3.Connection) -> Dict[str, int]:
    """Snapshot stats (app_dashboard.py; dashboard_stats table)."""
    stats = {}
    for key, q in [
        ("tracks", "SELECT COUNT(*) FROM score"),
        ("embeddings", "SELECT COUNT(*) FROM embedding"),
        ("clap_embeddings", "SELECT COUNT(*) FROM clap_embedding"),
        ("lyrics", "SELECT COUNT(*) FROM lyrics_embedding"),
        ("mappings", "SELECT COUNT(*) FROM track_server_map"),
        ("playlists", "SELECT COUNT(*) FROM playlist"),
        ("artists", "SELECT COUNT(DISTINCT author) FROM score"),
    ]:
        stats[key] = int(conn.execute(q).fetchone()[0])
    with write_txn(conn):
        for k, v in stats.items():
            conn.execute(
                """INSERT INTO dashboard_stats (key, value, updated_at)
                   VALUES (?,?, (julianday('now') - 2440587.5) * 86400.0)
                   ON CONFLICT(key) DO UPDATE SET value=excluded.value,
                       updated_at=excluded.updated_at""",
                (k, str(v)))
    return stats
