"""Similarity-feature blueprints.

Reference endpoints (SURVEY.md §2.1 query blueprints): /api/similar_tracks
(app_ivf.py:313), song path (app_path.py), alchemy (app_alchemy.py),
music map (app_map.py), artist similarity (app_artist_similarity.py),
CLAP text search (app_clap_search.py), lyrics search (app_lyrics.py),
SemGrove (app_sem_grove.py), sonic fingerprint
(app_sonic_fingerprint.py), hyperbolic explorer (app_hyperbolic.py).
"""

from __future__ import annotations

import json
from typing import List

import numpy as np
import torch
from flask import Blueprint, current_app, jsonify, request

from audiomuse_amd import config as C
from audiomuse_amd.analysis import index as idx
from audiomuse_amd.engines.alchemy import alchemy_query
from audiomuse_amd.engines.misc import order_playlist, sonic_fingerprint
from audiomuse_amd.engines.path import find_path
from audiomuse_amd.web.auth import require_auth

bp = Blueprint("queries", __name__)


def _state():
    return current_app.extensions["audiomuse"]


def _with_meta(results: List[dict]) -> List[dict]:
    """Attach title/author with ONE IN-clause query (a per-result
    meta_fn lookup was ~10 SQL round trips per response — measured in
    the http_load serving profile)."""
    if not results:
        return results
    ids = [r["item_id"] for r in results]
    rows = _state().conn().execute(
        "SELECT item_id, title, author FROM score WHERE item_id IN ("
        + ",".join("?" * len(ids)) + ")", ids).fetchall()
    meta = {m["item_id"]: m for m in rows}
    out = []
    for r in results:
        m = meta.get(r["item_id"])
        out.append({**r, "title": m["title"] if m else None,
                    "author": m["author"] if m else None})
    return out


def _server_scope(results: List[dict], n: int) -> List[dict]:
    """Availability mask (reference: ALGORITHM.md 4.2): when the request
    names a server, drop tracks that server does not have and attach the
    server's own provider id so every returned track can actually play."""
    server_id = request.args.get("server")
    if not server_id:
        return results[:n] if n else results
    conn = _state().conn()
    out = []
    for r in results:
        m = conn.execute(
            "SELECT provider_id FROM track_server_map WHERE item_id = ? "
            "AND server_id = ?", (r["item_id"], server_id)).fetchone()
        if m is not None:
            out.append({**r, "provider_id": m["provider_id"]})
            if n and len(out) >= n:
                break
    return out


@bp.get("/api/similar_tracks")
@require_auth
def similar_tracks():
    """reference: app_ivf.py:313"""
    eng = _state().engine(idx.AUDIO_INDEX)
    if eng is None:
        return jsonify({"error": "audio index not built"}), 503
    item_id = request.args.get("item_id", "")
    n = int(request.args.get("n", 10))
    radius_default = "1" if C.SIMILARITY_RADIUS_DEFAULT else "0"
    radius = request.args.get("radius_similarity",
                              request.args.get("radius", radius_default)
                              ) in ("1", "true")
    mood = request.args.get("mood_filter") or None
    cap = request.args.get("max_per_artist")
    # over-fetch when a server scope will drop unmapped tracks
    fetch = n * 2 if request.args.get("server") else n
    res = eng.find_similar_by_id(
        item_id, fetch, radius=radius, mood_filter=mood,
        eliminate_duplicates=request.args.get("eliminate_duplicates", "1")
        in ("1", "true"),
        max_per_artist=int(cap) if cap else None)
    if not res and eng.vector_for_id(item_id) is None:
        return jsonify({"error": f"unknown item_id {item_id!r}"}), 404
    return jsonify(_with_meta(_server_scope(res, n)))


@bp.get("/api/search_tracks")
@require_auth
def search_tracks():
    """Unified metadata search (ivf_manager unified track search)."""
    q = (request.args.get("q") or "").strip().lower()
    if not q:
        return jsonify([])
    conn = _state().conn()
    rows = conn.execute(
        """SELECT item_id, title, author, album FROM score
           WHERE LOWER(title) LIKE ? OR LOWER(author) LIKE ?
           LIMIT ?""",
        (f"%{q}%", f"%{q}%", int(request.args.get("n", 25)))).fetchall()
    return jsonify([dict(r) for r in rows])


@bp.get("/api/search_artists")
@require_auth
def search_artists():
    """Artist name search (reference: app_music_search /api/search_artists)."""
    q = (request.args.get("q") or "").strip().lower()
    if not q:
        return jsonify([])
    conn = _state().conn()
    rows = conn.execute(
        """SELECT author, COUNT(*) AS n_tracks FROM score
           WHERE LOWER(author) LIKE ? GROUP BY author
           ORDER BY n_tracks DESC LIMIT ?""",
        (f"%{q}%", int(request.args.get("n", 25)))).fetchall()
    return jsonify([{"artist": r["author"], "n_tracks": r["n_tracks"]}
                    for r in rows])


@bp.get("/api/artist_tracks")
@require_auth
def artist_tracks():
    """All catalogue tracks of one artist (reference: /api/artist_tracks)."""
    artist = (request.args.get("artist") or "").strip()
    if not artist:
        return jsonify({"error": "artist required"}), 400
    conn = _state().conn()
    rows = conn.execute(
        """SELECT item_id, title, album, tempo, energy FROM score
           WHERE author = ? ORDER BY album, title LIMIT ?""",
        (artist, int(request.args.get("n", 200)))).fetchall()
    return jsonify([dict(r) for r in rows])


@bp.get("/api/track")
@require_auth
def track_detail():
    """Full catalogue row for one track (reference: /api/track,
    /get_score + /get_embedding companions exist separately)."""
    item_id = request.args.get("item_id", "")
    conn = _state().conn()
    row = conn.execute("SELECT * FROM score WHERE item_id = ?",
                       (item_id,)).fetchone()
    if row is None:
        return jsonify({"error": f"unknown item_id {item_id!r}"}), 404
    d = dict(row)
    for k in ("mood_vector", "other_features"):
        if d.get(k):
            try:
                d[k] = json.loads(d[k])
            except (TypeError, ValueError):
                pass
    servers = conn.execute(
        "SELECT server_id, provider_id FROM track_server_map "
        "WHERE item_id = ?", (item_id,)).fetchall()
    d["servers"] = [dict(s) for s in servers]
    return jsonify(d)


@bp.get("/api/max_distance")
@require_auth
def max_distance():
    """Per-item distance ceiling for UI sliders (reference:
    app_ivf.py:515 -> get_max_distance_for_id)."""
    eng = _state().engine(idx.AUDIO_INDEX)
    if eng is None:
        return jsonify({"error": "audio index not built"}), 503
    item_id = request.args.get("item_id", "")
    out = eng.max_distance_for_id(item_id)
    if out is None:
        return jsonify({"error": f"unknown item_id {item_id!r}"}), 404
    return jsonify({**out, "metric": C.IVF_METRIC})


@bp.get("/api/mood_centroids")
@require_auth
def mood_centroids():
    """Per-mood centroid positions on the 2-D song map (reference:
    app_map.py /api/mood_centroids — map overlay labels)."""
    entry = _map_bucket("song", 100, "")
    if entry is None:
        return jsonify({"error": "map not built"}), 503
    pts = json.loads(entry[0])
    acc: dict = {}
    for p in pts:
        mood = p.get("mood") or ""
        if not mood:
            continue
        a = acc.setdefault(mood, [0.0, 0.0, 0])
        a[0] += p["x"]
        a[1] += p["y"]
        a[2] += 1
    return jsonify([{"mood": m, "x": a[0] / a[2], "y": a[1] / a[2],
                     "count": a[2]} for m, a in sorted(acc.items())])


@bp.get("/api/path")
@require_auth
def song_path():
    eng = _state().engine(idx.AUDIO_INDEX)
    if eng is None:
        return jsonify({"error": "audio index not built"}), 503
    a = request.args.get("start", "")
    b = request.args.get("end", "")
    length = int(request.args.get("length", 12))
    res = find_path(eng, a, b, length=length,
                    max_per_artist=C.MAX_SONGS_PER_ARTIST or None)
    if not res:
        return jsonify({"error": "unknown endpoints"}), 404
    return jsonify(_with_meta(res))


@bp.post("/api/alchemy")
@require_auth
def alchemy():
    eng = _state().engine(idx.AUDIO_INDEX)
    if eng is None:
        return jsonify({"error": "audio index not built"}), 503
    body = request.get_json(force=True, silent=True) or {}
    add_ids = body.get("add", [])
    sub_ids = body.get("subtract", [])

    def _vecs(ids):
        out = []
        conn = _state().conn()
        for i in ids:
            if isinstance(i, str) and i.startswith("anchor:"):
                # saved anchors join the mix (reference: alchemy_anchors)
                row = conn.execute(
                    "SELECT vector FROM alchemy_anchors WHERE name=?",
                    (i[len("anchor:"):],)).fetchone()
                if row is not None:
                    out.append(np.frombuffer(row["vector"], dtype=np.float32))
                continue
            v = eng.vector_for_id(i)
            if v is not None:
                out.append(v.cpu().numpy())
        return out

    res = alchemy_query(
        eng, _vecs(add_ids), _vecs(sub_ids),
        n=int(body["n"]) if "n" in body else None,      # ALCHEMY_DEFAULT/MAX
        subtract_radius=(float(body["subtract_radius"])
                         if "subtract_radius" in body else None),
        temperature=(float(body["temperature"])
                     if "temperature" in body else None),
        exclude=tuple(add_ids), seed=body.get("seed"))
    return jsonify(_with_meta(res))


@bp.post("/api/alchemy/radios/<name>/play")
@require_auth
def play_radio(name):
    """Run a saved radio definition through alchemy
    (reference: radio_manager.py)."""
    conn = _state().conn()
    row = conn.execute("SELECT definition FROM alchemy_radios WHERE name=?",
                       (name,)).fetchone()
    if row is None:
        return jsonify({"error": f"unknown radio {name!r}"}), 404
    definition = json.loads(row["definition"])
    with current_app.test_request_context(json=definition):
        return alchemy()


@bp.get("/api/artist_similarity")
@require_auth
def artist_similarity():
    sim = _state().engine(idx.ARTIST_INDEX)
    if sim is None:
        return jsonify({"error": "artist index not built"}), 503
    artist = request.args.get("artist", "")
    res = sim.find_similar_artists(artist, n=int(request.args.get("n", 10)))
    if not res and artist not in sim.models:
        return jsonify({"error": f"unknown artist {artist!r}"}), 404
    return jsonify([{"artist": a, "distance": d} for a, d in res])


@bp.get("/api/clap_search")
@require_auth
def clap_text_search():
    """Text -> CLAP space -> IVF (app_clap_search.py; clap_text_search.py:171)."""
    eng = _state().engine(idx.CLAP_INDEX)
    if eng is None:
        return jsonify({"error": "clap index not built"}), 503
    q = request.args.get("q", "")
    if not q:
        return jsonify([])
    emb = _clap_text_lifecycle().get()      # load + reset the countdown
    vec = emb.embed([q])[0]
    n = int(request.args.get("n", 20))
    fetch = n * 2 if request.args.get("server") else n
    res = eng.find_similar_by_vector(vec, fetch)
    return jsonify(_with_meta(_server_scope(res, n)))


def _clap_text_lifecycle():
    """Warm/idle lifecycle for the (large) text model: loaded on demand
    in the web process only, unloaded after the warm-up countdown
    expires; every search resets it (reference:
    clap_text_search.warmup_text_search_model :99)."""
    from audiomuse_amd.utils.resources import ModelLifecycle

    lc = current_app.extensions.get("clap_text_lc")
    if lc is None:
        def factory():
            from audiomuse_amd.models.text import TextEmbedder, clap_text_config
            return TextEmbedder(clap_text_config(), device=_state().device)

        lc = ModelLifecycle(factory,
                            idle_seconds=C.CLAP_TEXT_SEARCH_WARMUP_DURATION)
        current_app.extensions["clap_text_lc"] = lc
    lc.maybe_unload()
    return lc


@bp.post("/api/clap/warmup")
@require_auth
def clap_warmup():
    """Pre-load the text model and start the countdown."""
    lc = _clap_text_lifecycle()
    lc.get()
    return jsonify({"loaded": True, "seconds": lc.remaining()})


@bp.get("/api/clap/warmup/status")
@require_auth
def clap_warmup_status():
    lc = _clap_text_lifecycle()
    return jsonify({"loaded": lc.loaded, "seconds": lc.remaining()})


@bp.post("/api/lyrics/warmup")
@require_auth
def lyrics_warmup():
    """Pre-load the lyrics text embedder (reference: /api/lyrics/warmup)."""
    lc = _gte_lifecycle()
    lc.get()
    return jsonify({"loaded": True, "seconds": lc.remaining()})


@bp.get("/api/lyrics/warmup/status")
@require_auth
def lyrics_warmup_status():
    lc = _gte_lifecycle()
    return jsonify({"loaded": lc.loaded, "seconds": lc.remaining()})


def _family_stats(index_name: str, table: str):
    """Row count + loaded-index shape for one index family (reference:
    /api/clap/stats, /api/lyrics/stats, /api/sem_grove/stats)."""
    state = _state()
    n_rows = state.conn().execute(
        f"SELECT COUNT(*) AS n FROM {table}").fetchone()["n"]
    eng = state.engine(index_name)
    out = {"rows": n_rows, "index_loaded": eng is not None}
    if eng is not None and hasattr(eng, "index"):
        out.update({"indexed": eng.index.n, "nlist": eng.index.nlist,
                    "dim": eng.index.dim, "storage": eng.index.storage})
    return jsonify(out)


@bp.get("/api/clap/stats")
@require_auth
def clap_stats():
    return _family_stats(idx.CLAP_INDEX, "clap_embedding")


@bp.get("/api/lyrics/stats")
@require_auth
def lyrics_stats():
    return _family_stats(idx.LYRICS_INDEX, "lyrics_embedding")


@bp.get("/api/semgrove/stats")
@require_auth
def semgrove_stats():
    return _family_stats(idx.SEMGROVE_INDEX, "clap_embedding")


def _family_refresh(index_name: str):
    """Per-family incremental refresh trigger (reference:
    /api/clap/cache/refresh etc. — family-scoped, not a full rebuild)."""
    from audiomuse_amd.taskqueue import enqueue

    tid = enqueue(_state().conn(), "refresh_indexes", {"only": index_name},
                  queue="high")
    return jsonify({"task_id": tid, "index": index_name}), 202


@bp.post("/api/clap/cache/refresh")
@require_auth
def clap_cache_refresh():
    return _family_refresh(idx.CLAP_INDEX)


@bp.post("/api/lyrics/cache/refresh")
@require_auth
def lyrics_cache_refresh():
    return _family_refresh(idx.LYRICS_INDEX)


@bp.post("/api/semgrove/cache/refresh")
@require_auth
def semgrove_cache_refresh():
    return _family_refresh(idx.SEMGROVE_INDEX)


@bp.get("/api/lyrics_search")
@require_auth
def lyrics_search():
    """Text -> GTE space -> lyrics IVF (app_lyrics.py)."""
    eng = _state().engine(idx.LYRICS_INDEX)
    if eng is None:
        return jsonify({"error": "lyrics index not built"}), 503
    q = request.args.get("q", "")
    if not q:
        return jsonify([])
    emb = _gte_lifecycle().get()    # warm/unload countdown
    vec = emb.embed([q])[0]
    res = eng.find_similar_by_vector(vec, int(request.args.get("n", 20)))
    return jsonify(_with_meta(res))


def _gte_lifecycle():
    """GTE embedder warm/unload cycle (reference: tasks.gte_warm_cache,
    same shape as the CLAP text lifecycle; countdown
    LYRICS_GTE_WARMUP_DURATION)."""
    from audiomuse_amd.utils.resources import ModelLifecycle

    lc = current_app.extensions.get("gte_text_lc")
    if lc is None:
        def factory():
            from audiomuse_amd.models.text import TextEmbedder, gte_config
            return TextEmbedder(gte_config(), device=_state().device)

        lc = ModelLifecycle(factory,
                            idle_seconds=C.LYRICS_GTE_WARMUP_DURATION)
        current_app.extensions["gte_text_lc"] = lc
    lc.maybe_unload()
    return lc


# category-weighted suggested queries (reference: tasks/query.json used by
# clap_text_search.py suggested-queries generator)
_SUGGESTED_QUERIES = {
    "mood": (3, ["uplifting summer anthems", "melancholic rainy day songs",
                 "high energy workout tracks", "calm focus music"]),
    "genre": (3, ["classic soul grooves", "90s alternative rock",
                  "smooth jazz evenings", "underground hip-hop"]),
    "instrument": (2, ["acoustic guitar ballads", "piano-driven pieces",
                       "heavy synth textures"]),
    "scene": (2, ["late night driving", "sunday morning coffee",
                  "beach party at sunset"]),
}


@bp.get("/api/clap_search/suggestions")
@require_auth
def clap_search_suggestions():
    import random

    n = int(request.args.get("n", 6))
    seed = request.args.get("seed")
    rng = random.Random(int(seed) if seed else None)
    pool = []
    for _cat, (weight, queries) in _SUGGESTED_QUERIES.items():
        pool.extend((weight, q) for q in queries)
    picks = []
    while pool and len(picks) < n:
        total = sum(w for w, _ in pool)
        r = rng.uniform(0, total)
        acc = 0.0
        for i, (w, q) in enumerate(pool):
            acc += w
            if r <= acc:
                picks.append(q)
                pool.pop(i)
                break
    return jsonify(picks)


@bp.get("/api/semgrove")
@require_auth
def semgrove():
    """Seed-song search in the fused space (app_sem_grove.py)."""
    eng = _state().engine(idx.SEMGROVE_INDEX)
    if eng is None:
        return jsonify({"error": "semgrove index not built"}), 503
    item_id = request.args.get("item_id", "")
    res = eng.find_similar_by_id(
        item_id, int(request.args.get("n", 15)),
        radius=request.args.get("radius", "0") in ("1", "true"))
    if not res and eng.vector_for_id(item_id) is None:
        return jsonify({"error": f"unknown item_id {item_id!r}"}), 404
    return jsonify(_with_meta(res))


@bp.get("/api/lyrics_axes")
@require_auth
def lyrics_axes():
    """Axis-score search (reference: lyrics-axes index) — rank tracks by
    one of the 27 thematic axes."""
    axis = request.args.get("axis", "")
    if axis not in C.LYRICS_AXES:
        return jsonify({"error": f"unknown axis {axis!r}",
                        "axes": C.LYRICS_AXES}), 400
    conn = _state().conn()
    rows = conn.execute(
        "SELECT item_id, axis_scores FROM lyrics_embedding "
        "WHERE axis_scores IS NOT NULL").fetchall()
    scored = []
    for r in rows:
        try:
            score = json.loads(r["axis_scores"]).get(axis)
        except Exception:
            continue
        if score is not None:
            scored.append((r["item_id"], float(score)))
    scored.sort(key=lambda t: -t[1])
    n = int(request.args.get("n", 20))
    return jsonify(_with_meta([
        {"item_id": i, "distance": 1.0 - s} for i, s in scored[:n]]))


@bp.get("/api/lyrics_axes_similar")
@require_auth
def lyrics_axes_similar():
    """Similar thematic profile: nearest neighbors in the 27-axis space
    (reference: lyrics-axes index queries)."""
    item_id = request.args.get("item_id", "")
    eng = _state().engine(idx.LYRICS_AXES_INDEX)
    if eng is None:
        return jsonify({"error": "lyrics-axes index not built"}), 503
    n = int(request.args.get("n", 20))
    if eng.vector_for_id(item_id) is None:
        return jsonify({"error": f"no axis profile for {item_id!r}"}), 404
    res = eng.find_similar_by_id(item_id, n=n)
    return jsonify(_with_meta(res))


def _tree_cache():
    """Process-wide TreeCache (skeleton resident, full tree lazy-warm;
    reference hyperbolic_manager.py:832-897)."""
    from audiomuse_amd.engines.hyperbolic_tree import TreeCache

    cache = current_app.extensions.get("hyperbolic_tree")
    if cache is None:
        cache = current_app.extensions["hyperbolic_tree"] = TreeCache()
        cache.load_skeleton(_state().conn())
    return cache


@bp.get("/api/hyperbolic_tree")
@require_auth
def hyperbolic_tree():
    """Explorer tree root (reference: hyperbolic_manager tree cache
    :613): served from the persisted SKELETON — opening the explorer
    never forces the full tree into memory."""
    cache = _tree_cache()
    root = cache.node(_state().conn(), "root")
    if root is None:
        return jsonify({"error": "hyperbolic tree not built; run "
                                 "analysis/index rebuild"}), 503
    return jsonify(root)


@bp.get("/api/hyperbolic_tree/node/<path:node_id>")
@require_auth
def hyperbolic_tree_node(node_id):
    """One tree node. Folder nodes come from the skeleton; leaf nodes
    lazily warm the full tree (warm timer unloads it again after
    HYPERBOLIC_TREE_WARMUP_DURATION)."""
    cache = _tree_cache()
    node = cache.node(_state().conn(), node_id)
    if node is None:
        return jsonify({"error": f"unknown node {node_id!r}"}), 404
    return jsonify(node)


@bp.get("/api/hyperbolic_tree/status")
@require_auth
def hyperbolic_tree_status():
    return jsonify(_tree_cache().status())


@bp.post("/api/hyperbolic_tree/build")
@require_auth
def hyperbolic_tree_build():
    """Inline (re)build — normally the analysis run's index-build phase
    does this (analysis/index.py run_all_index_builds)."""
    from audiomuse_amd.analysis.index import build_hyperbolic_tree_cache

    state = _state()
    n = build_hyperbolic_tree_cache(state.conn(), device=state.device)
    cache = _tree_cache()
    cache.load_skeleton(state.conn())
    cache._unload_full()
    return jsonify({"tracks": n})


@bp.get("/api/sonic_fingerprint")
@require_auth
def sonic_fp():
    """Taste vector from listen history -> expansion (app_sonic_fingerprint)."""
    eng = _state().engine(idx.AUDIO_INDEX)
    if eng is None:
        return jsonify({"error": "audio index not built"}), 503
    body_ids = request.args.getlist("item_id")
    played = request.args.getlist("played_at", type=float)
    if not body_ids:
        return jsonify({"error": "item_id params required"}), 400
    vecs, times = [], []
    for i, pid in enumerate(body_ids):
        v = eng.vector_for_id(pid)
        if v is not None:
            vecs.append(v.cpu().numpy())
            import time as _t
            times.append(played[i] if i < len(played) else _t.time())
    if not vecs:
        return jsonify({"error": "no known tracks"}), 404
    fp = sonic_fingerprint(np.stack(vecs), times)
    res = eng.find_similar_by_vector(torch.from_numpy(fp),
                                     int(request.args.get("n", 20)),
                                     exclude=tuple(body_ids))
    return jsonify(_with_meta(res))


@bp.get("/api/hyperbolic_similar")
@require_auth
def hyperbolic_similar():
    """Poincare-space neighbors (app_hyperbolic.py:450)."""
    from audiomuse_amd.engines.hyperbolic import HyperbolicSpace

    eng = _state().engine(idx.AUDIO_INDEX)
    if eng is None:
        return jsonify({"error": "audio index not built"}), 503
    item_id = request.args.get("item_id", "")
    pos = eng.pos.get(item_id)
    if pos is None:
        return jsonify({"error": f"unknown item_id {item_id!r}"}), 404
    space = current_app.extensions.get("hyperbolic")
    if space is None or current_app.extensions.get("hyperbolic_n") != eng.index.n:
        space = HyperbolicSpace(eng.index.vectors_f32)
        current_app.extensions["hyperbolic"] = space
        current_app.extensions["hyperbolic_n"] = eng.index.n
    # defaults/caps + candidate overfetch before meta filtering
    # (reference HYPERBOLIC_DEFAULT_LIMIT / MAX_LIMIT /
    # CANDIDATE_OVERFETCH)
    n = min(int(request.args.get("n", C.HYPERBOLIC_DEFAULT_LIMIT)),
            C.HYPERBOLIC_MAX_LIMIT)
    fetch = min(n * max(C.HYPERBOLIC_CANDIDATE_OVERFETCH, 1),
                eng.index.n - 1)
    d, indices = space.similar(pos, fetch)
    res = []
    for dd, i in zip(d, indices):
        iid = eng.item_ids[int(i)]
        if _state().meta_fn(iid) is None:
            continue   # dropped catalogue rows never surface
        res.append({"item_id": iid, "distance": float(dd)})
        if len(res) >= n:
            break
    return jsonify(_with_meta(res))


_MAP_PERCENTS = (100, 75, 50, 25)
_map_cache: dict = {}


def _map_bucket(kind: str, percent: int, server_id: str = ""):
    """Percent-sampled map bucket, JSON + gzip serialized once
    (reference: app_map.build_map_cache :156 — deterministic samples at
    100/75/50/25%, pre-gzipped, served with zero further work). A
    per-server bucket (reference: the multi-server second cache layer)
    holds that server's own provider ids and drops unmapped tracks."""
    import gzip
    import random

    state = _state()
    name = idx.ARTIST_MAP if kind == "artist" else idx.SONG_MAP
    stamp = state._stamp(name)
    key = (kind, percent, server_id, stamp)
    hit = _map_cache.get(key)
    if hit is not None:
        return hit
    data = state.engine(name)
    if data is None:
        return None
    coords = data["coords"]
    ids = data["item_ids"]
    pid_of = {}
    if server_id and kind == "song":
        pid_of = {r["item_id"]: r["provider_id"] for r in state.conn().execute(
            "SELECT item_id, provider_id FROM track_server_map "
            "WHERE server_id = ?", (server_id,)).fetchall()}
    recs = []
    id_key = "artist" if kind == "artist" else "item_id"
    for i in range(len(ids)):
        rec = {id_key: ids[i], "x": float(coords[i][0]),
               "y": float(coords[i][1])}
        if kind == "song":
            if server_id:
                pid = pid_of.get(ids[i])
                if pid is None:
                    continue           # availability mask
                rec["provider_id"] = pid
            meta = state.meta_fn(ids[i]) or {}
            rec["title"] = meta.get("title", "")
            rec["author"] = meta.get("author", "")
            mv = meta.get("mood_vector") or {}
            rec["mood"] = max(mv, key=mv.get) if mv else ""
        recs.append(rec)
    if percent < 100:
        rng = random.Random(1234)           # deterministic sample
        recs = rng.sample(recs, max(1, len(recs) * percent // 100))
    raw = json.dumps(recs).encode()
    entry = (raw, gzip.compress(raw))
    if len(_map_cache) > 16:                # stamp change invalidates
        _map_cache.clear()
    _map_cache[key] = entry
    return entry


@bp.get("/api/map")
@require_auth
def music_map():
    """2-D map coordinates (app_map.py buckets). kind=song (default) or
    kind=artist; percent in {100, 75, 50, 25} selects the deterministic
    sample; responses are served from a pre-gzipped cache."""
    kind = request.args.get("kind", "song")
    percent = int(request.args.get("percent", 100))
    if percent not in _MAP_PERCENTS:
        return jsonify({"error": f"percent must be one of {_MAP_PERCENTS}"}), 400
    entry = _map_bucket(kind, percent, request.args.get("server", ""))
    if entry is None:
        return jsonify({"error": "map not built"}), 503
    raw, gz = entry
    from flask import Response

    if "gzip" in (request.headers.get("Accept-Encoding") or ""):
        return Response(gz, mimetype="application/json",
                        headers={"Content-Encoding": "gzip",
                                 "Cache-Control": "no-store"})
    return Response(raw, mimetype="application/json",
                    headers={"Cache-Control": "no-store"})


@bp.get("/api/map_cache_status")
@require_auth
def map_cache_status():
    """Pre-gzipped map bucket cache state (reference:
    app_map.py /api/map_cache_status)."""
    state = _state()
    return jsonify({
        "buckets_cached": len(_map_cache),
        "song_map_built": state.engine(idx.SONG_MAP) is not None,
        "artist_map_built": state.engine(idx.ARTIST_MAP) is not None,
        "percents": list(_MAP_PERCENTS)})


@bp.post("/api/rebuild_map_cache")
@require_auth
def rebuild_map_cache():
    """Drop the gzip bucket cache and enqueue an index refresh so the
    projections rebuild (reference: app_map.py /api/rebuild_map_cache)."""
    from audiomuse_amd.taskqueue import enqueue

    _map_cache.clear()
    tid = enqueue(_state().conn(), "rebuild_indexes", {}, queue="high")
    return jsonify({"task_id": tid}), 202


@bp.post("/api/order_playlist")
@require_auth
def order():
    body = request.get_json(force=True, silent=True) or {}
    ids = body.get("item_ids", [])
    state = _state()
    tracks = []
    for i in ids:
        meta = state.meta_fn(i) or {}
        tracks.append({"item_id": i, **{k: meta.get(k) for k in
                                        ("tempo", "energy", "key", "scale")}})
    arc = (bool(body["energy_arc"]) if "energy_arc" in body
           else C.PLAYLIST_ENERGY_ARC)
    ordered = order_playlist(tracks, energy_arc=arc)
    return jsonify([t["item_id"] for t in ordered])
