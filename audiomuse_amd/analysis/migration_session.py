"""Per-album migration review sessions.

The one-shot wizard (migration.py: probe -> preview -> apply) covers
the fast path; this adds the reference's session workflow
(app_provider_migration.py:678-2502: migration_session +
migration_target_meta tables, dry-run reports, per-album manual
match / skip decisions, finalize gate) so a user can review and
correct matches album by album before anything is written.

Design (not a translation): one `migration_session` row holds the
target config, the zlib-json cached target track list, the decision
map {album: {"action": "skip"} | {"action": "map", "target_album": X}}
and the latest dry-run report. All reads are against the cached
target meta — probing the target server happens exactly once, at
session start.
"""

from __future__ import annotations

import json
import zlib
from types import SimpleNamespace
from typing import Dict, List, Optional

from audiomuse_amd.analysis.migration import (build_match_preview,
                                              execute_migration,
                                              normalize_title,
                                              probe_server,
                                              propose_path_rule)
from audiomuse_amd.db import insert_returning_id, write_txn
from audiomuse_amd.mediaserver import make_provider


def _pack(obj) -> bytes:
    return zlib.compress(json.dumps(obj).encode())


def _unpack(blob) -> object:
    return json.loads(zlib.decompress(bytes(blob)).decode())


def create_session(conn, server_type: str, server_config: Optional[Dict],
                   source_server_id: str = "default") -> Dict:
    """Probe the target ONCE, cache its full track list, open a session."""
    probe = probe_server(server_type, server_config or {})
    if not probe.get("reachable"):
        return {"error": "target server unreachable", "probe": probe}
    provider = make_provider(server_type, **(server_config or {}))
    tracks = [{"provider_id": t.provider_id, "title": t.title,
               "author": t.author, "album": t.album,
               "file_path": t.file_path}
              for t in provider.get_all_songs()]
    with write_txn(conn):
        sid = insert_returning_id(
            conn,
            """INSERT INTO migration_session
                   (status, server_type, server_config, source_server_id,
                    target_meta)
               VALUES ('open', ?, ?, ?, ?)""",
            (server_type, json.dumps(server_config or {}),
             source_server_id, _pack(tracks)))
    return {"session_id": sid, "target_tracks": len(tracks),
            "libraries": probe.get("libraries", []),
            "path_format": probe.get("path_format", {})}


def _row(conn, sid: int):
    return conn.execute("SELECT * FROM migration_session WHERE id=?",
                        (sid,)).fetchone()


def get_session(conn, sid: int) -> Optional[Dict]:
    r = _row(conn, sid)
    if r is None:
        return None
    decisions = json.loads(r["decisions"] or "{}")
    out = {"session_id": sid, "status": r["status"],
           "server_type": r["server_type"],
           "source_server_id": r["source_server_id"],
           "decisions": decisions,
           "n_skipped_albums": sum(1 for d in decisions.values()
                                   if d.get("action") == "skip"),
           "n_manual_albums": sum(1 for d in decisions.values()
                                  if d.get("action") == "map"),
           "created_at": r["created_at"]}
    if r["report"] is not None:
        rep = _unpack(r["report"])
        out["report_summary"] = {k: rep[k] for k in
                                 ("tiers", "total", "matched", "match_ratio")}
    return out


def discard_session(conn, sid: int) -> bool:
    with write_txn(conn):
        cur = conn.execute(
            "UPDATE migration_session SET status='discarded' "
            "WHERE id=? AND status NOT IN ('executed')", (sid,))
    return cur.rowcount > 0


def set_decision(conn, sid: int, album: str, action: str,
                 target_album: Optional[str] = None) -> Optional[Dict]:
    """Record a per-album override: 'skip', 'map' (to target_album), or
    'auto' (clear the override)."""
    r = _row(conn, sid)
    if r is None or r["status"] in ("executed", "discarded"):
        return None
    decisions = json.loads(r["decisions"] or "{}")
    if action == "auto":
        decisions.pop(album, None)
    elif action == "skip":
        decisions[album] = {"action": "skip"}
    elif action == "map":
        decisions[album] = {"action": "map", "target_album": target_album}
    else:
        raise ValueError(f"unknown action {action!r}")
    with write_txn(conn):
        conn.execute(
            "UPDATE migration_session SET decisions=?, status='open' "
            "WHERE id=?", (json.dumps(decisions), sid))
    return decisions


def search_albums(conn, sid: int, q: str, limit: int = 50) -> List[Dict]:
    """Albums of the CACHED target list matching q (manual-map picker)."""
    r = _row(conn, sid)
    if r is None:
        return []
    ql = (q or "").lower()
    seen: Dict[str, int] = {}
    for t in _unpack(r["target_meta"]):
        if ql in (t.get("album") or "").lower():
            seen[t["album"]] = seen.get(t["album"], 0) + 1
    return [{"album": a, "n_tracks": n}
            for a, n in sorted(seen.items())][:limit]


def _source_albums(conn, source_server_id: str) -> Dict[str, List]:
    rows = conn.execute(
        """SELECT m.provider_id, m.item_id, s.title, s.author, s.album,
                  m.file_path
           FROM track_server_map m JOIN score s ON s.item_id = m.item_id
           WHERE m.server_id = ?""", (source_server_id,)).fetchall()
    by_album: Dict[str, List] = {}
    for r in rows:
        by_album.setdefault(r["album"] or "", []).append(r)
    return by_album


def run_dry_run(conn, sid: int) -> Optional[Dict]:
    """Auto-match the cached target list (decisions applied), store the
    report on the session. Read-only outside the session row."""
    r = _row(conn, sid)
    if r is None or r["status"] in ("executed", "discarded"):
        return None
    decisions = json.loads(r["decisions"] or "{}")
    meta = _unpack(r["target_meta"])
    skipped_targets = {d.get("target_album") for a, d in decisions.items()
                       if d.get("action") == "map"}
    source = r["source_server_id"]

    tracks = [SimpleNamespace(**t) for t in meta
              if t.get("album") not in skipped_targets]
    src_paths = [x["file_path"] for x in conn.execute(
        "SELECT file_path FROM track_server_map WHERE server_id=? "
        "AND file_path != ''", (source,)).fetchall()]
    rule = propose_path_rule([t.file_path for t in tracks], src_paths)
    preview = build_match_preview(conn, tracks, source, path_rule=rule)

    # manual album maps: pair target tracks of target_album with the
    # source album's catalogue rows, by normalized title then by order
    by_album = _source_albums(conn, source)
    manual_matches: List[Dict] = []
    for album, d in decisions.items():
        if d.get("action") != "map":
            continue
        src_rows = by_album.get(album, [])
        tgt = [t for t in meta if t.get("album") == d.get("target_album")]
        by_title = {normalize_title(x["title"]): x for x in src_rows}
        used = set()
        pairs = []
        rest_t, rest_s = [], [x for x in src_rows]
        for t in tgt:
            hit = by_title.get(normalize_title(t["title"]))
            if hit is not None and id(hit) not in used:
                used.add(id(hit))
                pairs.append((t, hit))
            else:
                rest_t.append(t)
        rest_s = [x for x in src_rows if id(x) not in used]
        pairs.extend(zip(rest_t, rest_s))         # positional fallback
        for t, hit in pairs:
            manual_matches.append({
                "provider_id": t["provider_id"], "item_id": hit["item_id"],
                "tier": "manual", "title": t["title"],
                "author": t["author"], "file_path": t["file_path"],
                "album": t["album"]})

    manual_pids = {m["provider_id"] for m in manual_matches}
    matches = [m for m in preview["matches"]
               if m["provider_id"] not in manual_pids] + manual_matches
    # skip decisions drop every match for that SOURCE album
    skip_albums = {a for a, d in decisions.items()
                   if d.get("action") == "skip"}
    if skip_albums:
        skip_items = {row["item_id"] for a in skip_albums
                      for row in by_album.get(a, [])}
        matches = [m for m in matches if m["item_id"] not in skip_items]
    total = preview["total"] + len(manual_matches)
    tiers = dict(preview["tiers"])
    if manual_matches:
        tiers["manual"] = len(manual_matches)
    report = {"matches": matches, "unmatched": preview["unmatched"],
              "tiers": tiers, "total": total, "matched": len(matches),
              "match_ratio": len(matches) / total if total else 0.0,
              "path_rule": rule}
    with write_txn(conn):
        conn.execute(
            "UPDATE migration_session SET report=?, status='dry_run' "
            "WHERE id=?", (_pack(report), sid))
    return {k: report[k] for k in ("tiers", "total", "matched",
                                   "match_ratio", "path_rule")}


def dry_run_report(conn, sid: int) -> Optional[Dict]:
    r = _row(conn, sid)
    if r is None or r["report"] is None:
        return None
    rep = _unpack(r["report"])
    rep["unmatched"] = rep["unmatched"][:200]
    rep["matches"] = rep["matches"][:500]
    return rep


def matched_albums(conn, sid: int) -> Optional[List[Dict]]:
    """Album-level aggregation of the stored report (the review list)."""
    r = _row(conn, sid)
    if r is None or r["report"] is None:
        return None
    rep = _unpack(r["report"])
    decisions = json.loads(r["decisions"] or "{}")
    meta = _unpack(r["target_meta"])
    album_of = {t["provider_id"]: (t.get("album") or "") for t in meta}
    agg: Dict[str, Dict] = {}
    for t in meta:
        a = t.get("album") or ""
        e = agg.setdefault(a, {"album": a, "total": 0, "matched": 0,
                               "tiers": {}})
        e["total"] += 1
    for m in rep["matches"]:
        a = album_of.get(m["provider_id"], "")
        e = agg.setdefault(a, {"album": a, "total": 0, "matched": 0,
                               "tiers": {}})
        e["matched"] += 1
        e["tiers"][m["tier"]] = e["tiers"].get(m["tier"], 0) + 1
    out = []
    for a, e in sorted(agg.items()):
        e["decision"] = decisions.get(a, {}).get("action", "auto")
        e["complete"] = e["matched"] >= e["total"]
        out.append(e)
    return out


def finalize(conn, sid: int) -> Optional[Dict]:
    """Gate before execute: requires a current dry-run report."""
    r = _row(conn, sid)
    if r is None or r["report"] is None or r["status"] != "dry_run":
        return None
    with write_txn(conn):
        conn.execute("UPDATE migration_session SET status='finalized' "
                     "WHERE id=?", (sid,))
    rep = _unpack(r["report"])
    return {"session_id": sid, "status": "finalized",
            "matched": rep["matched"], "match_ratio": rep["match_ratio"]}


def execute_session(conn, sid: int, target_server_id: str,
                    remove_source: bool = False,
                    min_match_ratio: float = 0.5) -> Optional[Dict]:
    """Apply the FINALIZED session's matches (execute_migration
    semantics: one transaction + restart handshake)."""
    r = _row(conn, sid)
    if r is None or r["status"] != "finalized" or r["report"] is None:
        return None
    rep = _unpack(r["report"])
    result = execute_migration(
        conn, rep["matches"], target_server_id,
        source_server_id=r["source_server_id"],
        remove_source=remove_source,
        min_match_ratio=min_match_ratio, preview=rep)
    if result.get("applied"):
        with write_txn(conn):
            conn.execute("UPDATE migration_session SET status='executed' "
                         "WHERE id=?", (sid,))
    return result
