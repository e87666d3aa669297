"""Instant playlist (AI chat) blueprint.

Reference: /root/reference/app_chat.py + tasks/ai/ (planner.py:1231
plan_and_execute_once): regex hint pre-extraction, ONE tool-calling LLM
request producing <= 4 tool calls (seed_search / text_match /
knowledge_lookup / search_database), hallucination stripping, tiered
re-rank with a relax loop, optional ordering.

Provider clients speak the OpenAI-compatible chat/completions HTTP API
(works for openai/mistral/self-hosted; gemini adapter maps the same
call shape). With AI_PROVIDER=none the planner is a deterministic
heuristic over the extracted hints, so the endpoint works offline —
the tool layer and re-rank are identical either way.
"""

from __future__ import annotations

import json
import re
from typing import Dict, List, Optional

from flask import Blueprint, current_app, jsonify, request

from audiomuse_amd import config as C
from audiomuse_amd.analysis import index as idx
from audiomuse_amd.web.auth import require_auth

bp = Blueprint("chat", __name__)

_MOOD_WORDS = set(w.lower() for w in C.MOOD_LABELS)
_HINT_COUNT = re.compile(r"\b(\d{1,3})\s*(?:songs|tracks)\b", re.I)
_HINT_ARTIST = re.compile(r"\b(?:by|like|similar to)\s+([A-Z][\w&' ]{2,40})")


def _state():
    return current_app.extensions["audiomuse"]


def extract_hints(prompt: str) -> Dict:
    """Regex pre-extraction (planner.extract_hints :218)."""
    hints: Dict = {"n": 20, "moods": [], "artists": [], "text": prompt}
    m = _HINT_COUNT.search(prompt)
    if m:
        hints["n"] = max(1, min(int(m.group(1)), 100))
    words = set(re.findall(r"[a-z']+", prompt.lower()))
    hints["moods"] = sorted(words & _MOOD_WORDS)
    hints["artists"] = [a.strip() for a in _HINT_ARTIST.findall(prompt)]
    return hints


# -- tools (tasks/ai/tool_impl.py equivalents) ------------------------------

def tool_seed_search(args: Dict) -> List[Dict]:
    """Similar tracks from seed titles/artists (tool_impl seed_search)."""
    eng = _state().engine(idx.AUDIO_INDEX)
    if eng is None:
        return []
    conn = _state().conn()
    seeds = []
    for name in args.get("seeds", []):
        row = conn.execute(
            "SELECT item_id FROM score WHERE LOWER(title) LIKE ? "
            "OR LOWER(author) LIKE ? LIMIT 1",
            (f"%{name.lower()}%", f"%{name.lower()}%")).fetchone()
        if row:
            seeds.append(row["item_id"])
    out: List[Dict] = []
    for sid in seeds[:4]:
        out.extend(eng.find_similar_by_id(sid, int(args.get("n", 20))))
    return out


def tool_text_match(args: Dict) -> List[Dict]:
    """CLAP text search (tool_impl text_match)."""
    eng = _state().engine(idx.CLAP_INDEX)
    if eng is None:
        return []
    from audiomuse_amd.web.api_queries import _clap_text_lifecycle

    emb = _clap_text_lifecycle().get()
    vec = emb.embed([args.get("query", "")])[0]
    return eng.find_similar_by_vector(vec, int(args.get("n", 20)))


def tool_search_database(args: Dict) -> List[Dict]:
    """Read-only metadata filters under the low-privilege connection
    (reference: tool_impl search_database via mcp_helper._ensure_ai_
    chat_db_user :63 — PG role with SELECT-only grants; SQLite mode=ro
    connection). AI-shaped queries can never write."""
    from audiomuse_amd.ai.dbrole import readonly_connection

    state = _state()
    conn = readonly_connection(state.db_url, admin_conn=state.conn())
    clauses, params = [], []
    if args.get("mood"):
        clauses.append("mood_vector LIKE ?")
        params.append(f"%\"{args['mood']}\"%")
    if args.get("artist"):
        clauses.append("LOWER(author) LIKE ?")
        params.append(f"%{args['artist'].lower()}%")
    if args.get("min_tempo"):
        clauses.append("tempo >= ?")
        params.append(float(args["min_tempo"]))
    if args.get("max_tempo"):
        clauses.append("tempo <= ?")
        params.append(float(args["max_tempo"]))
    where = (" WHERE " + " AND ".join(clauses)) if clauses else ""
    rows = conn.execute(
        f"SELECT item_id FROM score{where} LIMIT ?",
        (*params, int(args.get("n", 50)))).fetchall()
    return [{"item_id": r["item_id"], "distance": 0.5} for r in rows]


def tool_knowledge_lookup(args: Dict) -> List[Dict]:
    """Without network knowledge, fall back to text match."""
    return tool_text_match({"query": args.get("query", ""),
                            "n": args.get("n", 20)})


TOOLS = {"seed_search": tool_seed_search, "text_match": tool_text_match,
         "search_database": tool_search_database,
         "knowledge_lookup": tool_knowledge_lookup}


def validate_and_normalize_plan(plan: List[Dict]) -> List[Dict]:
    """Strip hallucinated tools, dedupe, cap at AI_MAX_TOOL_CALLS
    (planner.validate_and_normalize_plan :946)."""
    seen = set()
    out = []
    for call in plan:
        name = call.get("tool")
        if name not in TOOLS:
            continue
        key = json.dumps(call, sort_keys=True)
        if key in seen:
            continue
        seen.add(key)
        out.append(call)
        if len(out) >= C.AI_MAX_TOOL_CALLS:
            break
    return out


def heuristic_plan(hints: Dict) -> List[Dict]:
    """Deterministic offline planner (AI_PROVIDER=none)."""
    plan: List[Dict] = []
    if hints["artists"]:
        plan.append({"tool": "seed_search",
                     "args": {"seeds": hints["artists"], "n": hints["n"]}})
    for mood in hints["moods"][:2]:
        plan.append({"tool": "search_database",
                     "args": {"mood": mood, "n": hints["n"] * 2}})
    plan.append({"tool": "text_match",
                 "args": {"query": hints["text"], "n": hints["n"]}})
    return plan


def llm_plan(prompt: str, hints: Dict) -> Optional[List[Dict]]:
    """One tool-calling request to the configured vendor adapter
    (ai/providers.py: openai / mistral / gemini). None on any failure
    -> heuristic plan."""
    from audiomuse_amd.ai import plan_with_llm

    return plan_with_llm(prompt, {name: None for name in TOOLS})


_FILLER_TITLE = re.compile(r"\b(intro|outro|skit|interlude|spoken)\b", re.I)


def rerank(results_per_tool: List[List[Dict]], n: int,
           exclude_artists: Optional[List[str]] = None) -> List[str]:
    """Tiered re-rank (tasks/ai/rerank.py): intersections boost, primary
    similarity orders within a tier, intro/skit/interlude titles are
    pushed down, exclude_artists is the one hard cut."""
    state = _state()
    scores: Dict[str, float] = {}
    hits: Dict[str, int] = {}
    for results in results_per_tool:
        for r in results:
            iid = r["item_id"]
            scores[iid] = min(scores.get(iid, 10.0), r.get("distance", 1.0))
            hits[iid] = hits.get(iid, 0) + 1
    excl = {a.lower() for a in (exclude_artists or [])}
    filler: Dict[str, bool] = {}
    for iid in list(scores):
        meta = state.meta_fn(iid) or {}
        if excl and (meta.get("author") or "").lower() in excl:
            scores.pop(iid)      # hard cut
            continue
        filler[iid] = bool(_FILLER_TITLE.search(meta.get("title") or ""))
    ranked = sorted(scores,
                    key=lambda i: (filler[i], -hits[i], scores[i]))
    return ranked[:n]


def run_plan(prompt: str, hints: Dict):
    """Plan -> tools -> one replan on an empty pool (reference:
    planner 'one replan' rule) -> (plan, per-tool results)."""
    plan = llm_plan(prompt, hints) or heuristic_plan(hints)
    plan = validate_and_normalize_plan(plan)
    # per-tool candidate pools are capped at MAX_SONGS_IN_AI_PROMPT
    # (reference: the planner context/candidate budget)
    cap = max(int(C.MAX_SONGS_IN_AI_PROMPT), 1)
    results = [TOOLS[c["tool"]](c.get("args", {}))[:cap] for c in plan]
    if not any(results):
        replan = validate_and_normalize_plan(heuristic_plan(hints))
        if replan != plan:
            plan = replan
            results = [TOOLS[c["tool"]](c.get("args", {}))[:cap]
                       for c in plan]
    return plan, results


@bp.post("/chat/api/chatPlaylistStream")
@require_auth
def chat_playlist_stream():
    """SSE variant (reference: app_chat.py:292): streams plan, then each
    tool's results, then the final playlist."""
    import flask

    body = request.get_json(force=True, silent=True) or {}
    prompt = body.get("prompt", "")
    if not prompt:
        return jsonify({"error": "prompt required"}), 400
    hints = extract_hints(prompt)
    plan, _pre = run_plan(prompt, hints)
    state = _state()

    def generate():
        yield f"event: plan\ndata: {json.dumps(plan)}\n\n"
        results = []
        for call in plan:
            res = TOOLS[call["tool"]](call.get("args", {}))
            results.append(res)
            yield (f"event: tool\ndata: "
                   f"{json.dumps({'tool': call['tool'], 'n': len(res)})}\n\n")
        ids = rerank(results, hints["n"])
        tracks = []
        for i in ids:
            meta = state.meta_fn(i) or {}
            tracks.append({"item_id": i, "title": meta.get("title"),
                           "author": meta.get("author")})
        yield f"event: playlist\ndata: {json.dumps(tracks)}\n\n"

    return flask.Response(flask.stream_with_context(generate()),
                          mimetype="text/event-stream")


@bp.post("/chat/api/chatPlaylist")
@require_auth
def chat_playlist():
    """reference: app_chat.py:144,264"""
    body = request.get_json(force=True, silent=True) or {}
    prompt = body.get("prompt", "")
    if not prompt:
        return jsonify({"error": "prompt required"}), 400
    hints = extract_hints(prompt)
    plan, results = run_plan(prompt, hints)
    ids = rerank(results, hints["n"],
                 exclude_artists=body.get("exclude_artists"))
    state = _state()
    tracks = []
    for i in ids:
        meta = state.meta_fn(i) or {}
        tracks.append({"item_id": i, "title": meta.get("title"),
                       "author": meta.get("author")})
    if body.get("order"):
        rows = [{"item_id": t["item_id"],
                 **{k: (state.meta_fn(t["item_id"]) or {}).get(k)
                    for k in ("tempo", "energy", "key", "scale")}}
                for t in tracks]
        from audiomuse_amd.engines.misc import order_playlist
        ordered = {r["item_id"]: i for i, r in enumerate(
            order_playlist(rows, energy_arc=len(rows) >= 10))}
        tracks.sort(key=lambda t: ordered.get(t["item_id"], 0))
    return jsonify({"plan": plan, "tracks": tracks, "hints": hints})
