"""Provider-migration wizard: probe -> library select -> path-format
detection -> match preview -> transactional rewrite -> restart handshake.

Reference: /root/reference/app_provider_migration.py (2828 LoC) +
tasks/provider_migration_tasks.py (1420) + provider_migration_matcher.py
(349). The wizard moves a library's canonical mappings from one media
server to another WITHOUT re-analysis: the catalogue (score/embedding
rows keyed by fp_4 ids) is server-independent; only track_server_map
rows need new provider ids. Stages:

1. probe      — reachability, libraries, a path sample of the target
2. path       — detect the target's mount prefix/separator and propose
                the source->target path rewrite rule
3. preview    — tiered matching (rewritten-path / path-tail /
                exact title+artist / normalized) with per-tier counts
                and the unmatched list; NOTHING is written
4. execute    — one transaction rewrites the mappings (optionally
                removing the source server's rows), bumps the ivf_dir
                stamp so engines reload, and publishes a restart
                control request that workers ack (taskqueue/control.py)
"""

from __future__ import annotations

import json
from collections import Counter
from typing import Dict, List, Optional, Sequence

from audiomuse_amd.analysis.maintenance import normalize_title
from audiomuse_amd.db import write_txn
from audiomuse_amd.mediaserver import make_provider
from audiomuse_amd.taskqueue.worker import TaskContext, task_handler


# -- stage 1: probe ----------------------------------------------------------

def probe_server(server_type: str, server_config: Optional[Dict] = None,
                 sample: int = 50) -> Dict:
    """Reachability + libraries + a small track sample (reference:
    wizard probe step)."""
    cfg = {k: v for k, v in (server_config or {}).items()
           if k != "server_type"}  # callers may echo the type back
    provider = make_provider(server_type, **cfg)
    if not provider.test_connection():
        return {"reachable": False}
    libraries = provider.list_libraries()
    tracks = []
    for album in provider.get_recent_albums(limit=10):
        tracks.extend(provider.get_tracks_from_album(album.provider_id))
        if len(tracks) >= sample:
            break
    return {
        "reachable": True,
        "libraries": libraries,
        "sample_count": len(tracks),
        "sample_paths": [t.file_path for t in tracks[:10] if t.file_path],
        "path_format": detect_path_format(
            [t.file_path for t in tracks if t.file_path]),
    }


# -- stage 2: path-format detection -----------------------------------------

def detect_path_format(paths: Sequence[str]) -> Dict:
    """Mount prefix + separator inference (reference: path-format
    detection in the wizard). The prefix is the longest directory chain
    shared by a majority of paths."""
    paths = [p for p in paths if p]
    if not paths:
        return {"prefix": "", "separator": "/", "n_paths": 0}
    sep = "\\" if sum("\\" in p for p in paths) > len(paths) / 2 else "/"
    split = [p.split(sep) for p in paths]
    prefix_parts: List[str] = []
    for depth in range(min(len(s) for s in split) - 1):  # never the file
        counts = Counter(s[depth] for s in split)
        part, n = counts.most_common(1)[0]
        if n < len(paths) * 0.8:
            break
        prefix_parts.append(part)
    prefix = sep.join(prefix_parts)
    return {"prefix": prefix, "separator": sep, "n_paths": len(paths)}


def propose_path_rule(source_paths: Sequence[str],
                      target_paths: Sequence[str]) -> Dict:
    """source prefix -> target prefix rewrite rule."""
    src = detect_path_format(source_paths)
    dst = detect_path_format(target_paths)
    return {"from_prefix": src["prefix"], "to_prefix": dst["prefix"],
            "from_separator": src["separator"],
            "to_separator": dst["separator"]}


def rewrite_path(path: str, rule: Dict) -> str:
    if not path:
        return path
    out = path
    fp = rule.get("from_prefix") or ""
    if fp and out.startswith(fp):
        out = (rule.get("to_prefix") or "") + out[len(fp):]
    fs, ts = rule.get("from_separator", "/"), rule.get("to_separator", "/")
    if fs != ts:
        out = out.replace(fs, ts)
    return out


# -- stage 3: match preview --------------------------------------------------

def build_match_preview(conn, target_tracks: Sequence,
                        source_server_id: str,
                        path_rule: Optional[Dict] = None) -> Dict:
    """Tiered matching of the TARGET server's tracks onto the catalogue
    via the SOURCE server's mappings (reference:
    provider_migration_matcher.py). Read-only."""
    rows = conn.execute(
        """SELECT m.provider_id, m.item_id, m.file_path, s.title, s.author
           FROM track_server_map m JOIN score s ON s.item_id = m.item_id
           WHERE m.server_id = ?""", (source_server_id,)).fetchall()
    by_path = {r["file_path"]: r for r in rows if r["file_path"]}
    by_tail = {r["file_path"].rsplit("/", 1)[-1]: r
               for r in rows if r["file_path"]}
    by_exact = {(r["title"] or "", r["author"] or ""): r for r in rows}
    by_norm = {(normalize_title(r["title"]), normalize_title(r["author"])):
               r for r in rows}

    matches: List[Dict] = []
    unmatched: List[Dict] = []
    tiers = Counter()
    for t in target_tracks:
        rewritten = rewrite_path(t.file_path, path_rule) if path_rule \
            else t.file_path
        hit, tier = None, None
        if rewritten and rewritten in by_path:
            hit, tier = by_path[rewritten], "path"
        elif t.file_path and \
                t.file_path.rsplit("/", 1)[-1].rsplit("\\", 1)[-1] in by_tail:
            hit, tier = by_tail[
                t.file_path.rsplit("/", 1)[-1].rsplit("\\", 1)[-1]], "tail"
        elif (t.title, t.author) in by_exact:
            hit, tier = by_exact[(t.title, t.author)], "exact"
        elif (normalize_title(t.title), normalize_title(t.author)) in by_norm:
            hit, tier = by_norm[(normalize_title(t.title),
                                 normalize_title(t.author))], "normalized"
        if hit is None:
            tiers["unmatched"] += 1
            unmatched.append({"provider_id": t.provider_id,
                              "title": t.title, "author": t.author})
        else:
            tiers[tier] += 1
            matches.append({"provider_id": t.provider_id,
                            "item_id": hit["item_id"], "tier": tier,
                            "title": t.title, "author": t.author,
                            "file_path": t.file_path})
    total = len(matches) + len(unmatched)
    return {"matches": matches, "unmatched": unmatched[:200],
            "tiers": dict(tiers), "total": total,
            "matched": len(matches),
            "match_ratio": len(matches) / total if total else 0.0,
            "source_mappings": len(rows)}


# -- stage 4: transactional rewrite + restart handshake ----------------------

def execute_migration(conn, matches: Sequence[Dict], target_server_id: str,
                      source_server_id: Optional[str] = None,
                      remove_source: bool = False,
                      min_match_ratio: float = 0.0,
                      preview: Optional[Dict] = None) -> Dict:
    """One transaction: write the target server's mappings (and
    optionally retire the source's). Refuses when the preview's match
    ratio is below ``min_match_ratio`` — the reference wizard's
    dont-half-migrate guard. Publishes the restart control request
    afterwards (workers drain + re-hydrate; control.py handshake)."""
    if preview is not None and preview.get("match_ratio", 1.0) < min_match_ratio:
        return {"applied": False,
                "reason": f"match ratio {preview.get('match_ratio'):.2f} "
                          f"below required {min_match_ratio:.2f}"}
    written = 0
    removed = 0
    with write_txn(conn):
        for m in matches:
            conn.execute(
                """INSERT INTO track_server_map (provider_id, server_id,
                       item_id, title, author, file_path)
                   VALUES (?,?,?,?,?,?)
                   ON CONFLICT(provider_id, server_id)
                   DO UPDATE SET item_id=excluded.item_id,
                       file_path=excluded.file_path""",
                (m["provider_id"], target_server_id, m["item_id"],
                 m.get("title", ""), m.get("author", ""),
                 m.get("file_path", "")))
            written += 1
        if remove_source and source_server_id:
            cur = conn.execute(
                "DELETE FROM track_server_map WHERE server_id=?",
                (source_server_id,))
            removed = cur.rowcount
    from audiomuse_amd.taskqueue import control as qctl
    request_id = qctl.publish_control_request(
        conn, qctl.ACTION_RESTART,
        payload={"reason": "provider migration", "target": target_server_id})
    return {"applied": True, "written": written, "removed": removed,
            "restart_request_id": request_id}


# -- queued wizard task -------------------------------------------------------

@task_handler("provider_migration")
def provider_migration_task(ctx: TaskContext, payload: Dict) -> Dict:
    """End-to-end wizard run as one cancellable task: probe -> path rule
    -> preview -> (if apply) execute. ``apply: false`` stops after the
    preview, storing it in the task result — the wizard UI's preview
    step (reference MIGRATION_PLANNER_TASK_TYPE)."""
    conn = ctx.conn
    server_type = payload.get("server_type", "synthetic")
    server_config = payload.get("server_config", {})
    source_server_id = payload.get("source_server_id", "default")
    target_server_id = payload.get("target_server_id", "migrated")

    ctx.report(0.1, "probing target server")
    probe = probe_server(server_type, server_config)
    if not probe.get("reachable"):
        return {"stage": "probe", "error": "target server unreachable"}
    ctx.check_cancelled()

    provider = make_provider(server_type, **server_config)
    tracks = provider.get_all_songs()
    src_paths = [r["file_path"] for r in conn.execute(
        "SELECT file_path FROM track_server_map WHERE server_id=? "
        "AND file_path != ''", (source_server_id,)).fetchall()]
    rule = propose_path_rule(src_paths, [t.file_path for t in tracks])
    # the rewrite maps TARGET paths into the SOURCE's namespace
    rule = {"from_prefix": rule["to_prefix"],
            "to_prefix": rule["from_prefix"],
            "from_separator": rule["to_separator"],
            "to_separator": rule["from_separator"]}

    ctx.report(0.4, "matching against catalogue")
    preview = build_match_preview(conn, tracks, source_server_id,
                                  path_rule=rule)
    ctx.check_cancelled()
    summary = {"stage": "preview", "path_rule": rule,
               "tiers": preview["tiers"], "total": preview["total"],
               "matched": preview["matched"],
               "match_ratio": round(preview["match_ratio"], 4),
               "unmatched_sample": preview["unmatched"][:20]}
    if not payload.get("apply"):
        return summary

    ctx.report(0.7, "rewriting mappings")
    result = execute_migration(
        conn, preview["matches"], target_server_id,
        source_server_id=source_server_id,
        remove_source=bool(payload.get("remove_source")),
        min_match_ratio=float(payload.get("min_match_ratio", 0.5)),
        preview=preview)
    summary.update(result)
    summary["stage"] = "done" if result.get("applied") else "refused"
    ctx.report(1.0, json.dumps({"stage": summary["stage"]}))
    return summary
