"""Feature-engine tests: similarity, alchemy, path, hyperbolic, artist
GMM, SemGrove, fingerprint, ordering."""

import math
import time

import numpy as np
import pytest
import torch

from audiomuse_amd import config as C
from audiomuse_amd.engines import misc
from audiomuse_amd.engines.alchemy import (alchemy_query, combine_vectors,
                                           temperature_sample)
from audiomuse_amd.engines.artist_gmm import (ArtistSimilarity, fit_artist,
                                              soft_chamfer_distance)
from audiomuse_amd.engines.hyperbolic import (HyperbolicSpace,
                                              poincare_distance, project)
from audiomuse_amd.engines.path import find_path, interpolate
from audiomuse_amd.engines.similarity import build_engine_from_matrix


def _catalogue(n=400, d=32, seed=0, n_artists=20):
    rng = np.random.default_rng(seed)
    x = rng.standard_normal((n, d)).astype(np.float32)
    ids = [f"s{i}" for i in range(n)]
    meta = {f"s{i}": {"title": f"T{i}", "author": f"artist{i % n_artists}",
                      "mood_vector": {"rock": float(i % 2)}}
            for i in range(n)}
    return x, ids, meta


def _engine(**kw):
    x, ids, meta = _catalogue(**kw)
    eng = build_engine_from_matrix(x, ids, meta_fn=lambda i: meta.get(i),
                                   nlist=16)
    return eng, x, ids, meta


def test_similar_by_id_excludes_self_and_sorted():
    eng, x, ids, meta = _engine()
    res = eng.find_similar_by_id("s10", 5, nprobe=16)
    assert len(res) == 5
    assert all(r["item_id"] != "s10" for r in res)
    dists = [r["distance"] for r in res]
    assert dists == sorted(dists)


def test_similarity_cache_hits():
    eng, *_ = _engine()
    a = eng.find_similar_by_id("s1", 3)
    b = eng.find_similar_by_id("s1", 3)
    assert a is b  # cached object


def test_artist_cap():
    eng, *_ = _engine(n_artists=2)
    res = eng.find_similar_by_id("s0", 10, max_per_artist=2, nprobe=16)
    authors = [eng.meta_fn(r["item_id"])["author"] for r in res]
    for a in set(authors):
        assert authors.count(a) <= 2


def test_duplicate_filter_drops_near_copies():
    x, ids, meta = _catalogue(n=50)
    x[1] = x[0] + 1e-6  # near-exact duplicate of s0
    eng = build_engine_from_matrix(x, ids, meta_fn=lambda i: meta.get(i), nlist=4)
    res = eng.find_similar_by_vector(torch.from_numpy(x[0]), 10,
                                     eliminate_duplicates=True, nprobe=4)
    got = [r["item_id"] for r in res]
    assert not ("s0" in got and "s1" in got)


def test_mood_filter():
    eng, *_ = _engine()
    res = eng.find_similar_by_id("s0", 8, mood_filter="rock", nprobe=16)
    for r in res:
        assert eng.meta_fn(r["item_id"])["mood_vector"]["rock"] > 0


def test_radius_walk_mode_returns_n():
    eng, *_ = _engine()
    res = eng.find_similar_by_id("s5", 12, radius=True, nprobe=16)
    assert 0 < len(res) <= 12
    assert len({r["item_id"] for r in res}) == len(res)


def test_multi_query_union():
    eng, x, *_ = _engine()
    res = eng.multi_query([torch.from_numpy(x[0]), torch.from_numpy(x[100])], 10)
    assert len(res) == 10
    ids = {r["item_id"] for r in res}
    assert "s0" in ids and "s100" in ids


# -- alchemy ---------------------------------------------------------------

def test_combine_vectors_add_subtract():
    a = np.array([1.0, 0.0]); b = np.array([0.0, 1.0])
    v = combine_vectors([a], [b], subtract_weight=0.5)
    assert np.allclose(np.linalg.norm(v), 1.0)
    assert v[0] > 0 > v[1] or (v[0] > 0 and v[1] < 0.01)
    assert combine_vectors([]) is None


def test_temperature_sample_seeded_and_capped():
    results = [{"item_id": f"x{i}", "distance": i * 0.1} for i in range(20)]
    a = temperature_sample(results, 5, 0.5, seed=7)
    b = temperature_sample(results, 5, 0.5, seed=7)
    assert a == b and len(a) == 5
    det = temperature_sample(results, 5, 0.0)
    assert det == results[:5]


def test_alchemy_query_end_to_end():
    eng, x, *_ = _engine()
    res = alchemy_query(eng, add=[x[3]], n=7, nprobe=16)
    assert res and res[0]["item_id"] == "s3"
    res2 = alchemy_query(eng, add=[x[3]], subtract=[x[3]], subtract_radius=0.5,
                         n=7, nprobe=16)
    assert all(r["item_id"] != "s3" for r in res2)


# -- path ------------------------------------------------------------------

def test_interpolate_slerp_unit_norm_and_monotone():
    a = np.array([1.0, 0, 0]); b = np.array([0, 1.0, 0])
    pts = interpolate(a, b, 5)
    assert pts.shape == (5, 3)
    np.testing.assert_allclose(np.linalg.norm(pts, axis=1), 1.0, atol=1e-5)
    angles = [math.atan2(p[1], p[0]) for p in pts]
    assert all(x < y for x, y in zip(angles, angles[1:]))


def test_find_path_endpoints_and_length():
    eng, *_ = _engine()
    path = find_path(eng, "s0", "s100", length=8)
    assert path[0]["item_id"] == "s0" and path[-1]["item_id"] == "s100"
    assert len(path) <= 8
    ids = [p["item_id"] for p in path]
    assert len(set(ids)) == len(ids)


# -- hyperbolic ------------------------------------------------------------

def test_projection_inside_ball_and_distance_props():
    x = torch.randn(100, 16) * 3
    space = HyperbolicSpace(x)
    assert float(space.points.norm(dim=1).max()) < 1.0
    u, v = space.points[0], space.points[1]
    d_uv = poincare_distance(u, v)
    d_vu = poincare_distance(v, u)
    torch.testing.assert_close(d_uv, d_vu)
    assert float(poincare_distance(u, u)) < 1e-3


def test_hyperbolic_similar_excludes_self():
    x = torch.randn(50, 8)
    space = HyperbolicSpace(x)
    d, idx = space.similar(3, 5)
    assert 3 not in idx.tolist()
    assert (d[:-1] <= d[1:]).all()


# -- artist GMM ------------------------------------------------------------

def test_chamfer_zero_for_identical_artist():
    rng = np.random.default_rng(0)
    embs = rng.standard_normal((30, 16)).astype(np.float32)
    m = fit_artist("a", embs, seed=0)
    assert soft_chamfer_distance(m, m) < 1e-6


def test_find_similar_artists_prefers_same_distribution():
    rng = np.random.default_rng(1)
    base = rng.standard_normal(16).astype(np.float32) * 3
    far = -base
    per_artist = {
        "a1": base + rng.standard_normal((25, 16)).astype(np.float32) * 0.2,
        "a2": base + rng.standard_normal((25, 16)).astype(np.float32) * 0.2,
        "b": far + rng.standard_normal((25, 16)).astype(np.float32) * 0.2,
    }
    sim = ArtistSimilarity()
    sim.fit_catalogue(per_artist, seed=0)
    res = sim.find_similar_artists("a1", n=2)
    assert res[0][0] == "a2"
    assert res[0][1] < res[1][1]


# -- misc engines ----------------------------------------------------------

def test_semgrove_merge_weights_and_shape():
    rng = np.random.default_rng(2)
    lyr = rng.standard_normal((40, 12)).astype(np.float32)
    aud = rng.standard_normal((40, 8)).astype(np.float32)
    m = misc.SemGroveMerger()
    m.fit(lyr, aud)
    merged = m.merge(lyr, aud)
    assert merged.shape == (40, 20)
    # sqrt-weight scaling: lyrics part norm ~ sqrt(0.75), audio ~ sqrt(0.25)
    ln = np.linalg.norm(merged[:, :12], axis=1).mean()
    an = np.linalg.norm(merged[:, 12:], axis=1).mean()
    assert abs(ln - math.sqrt(C.SEM_GROVE_LYRICS_WEIGHT)) < 0.05
    assert abs(an - math.sqrt(C.SEM_GROVE_AUDIO_WEIGHT)) < 0.05


def test_sonic_fingerprint_recency_weighting():
    now = time.time()
    v = np.stack([np.array([1.0, 0.0]), np.array([0.0, 1.0])]).astype(np.float32)
    # first played now, second played 90 days ago -> first dominates
    fp = misc.sonic_fingerprint(v, [now, now - 90 * 86400], now=now)
    assert fp[0] > 0.9
    assert misc.sonic_fingerprint(np.zeros((0, 2)), []) is None


def test_order_playlist_greedy_walk():
    tracks = [
        {"item_id": "a", "tempo": 80, "energy": 0.2, "key": "C", "scale": "major"},
        {"item_id": "b", "tempo": 85, "energy": 0.3, "key": "C", "scale": "major"},
        {"item_id": "c", "tempo": 160, "energy": 0.9, "key": "F#", "scale": "minor"},
        {"item_id": "d", "tempo": 82, "energy": 0.25, "key": "G", "scale": "major"},
    ]
    out = misc.order_playlist(tracks)
    assert {t["item_id"] for t in out} == {"a", "b", "c", "d"}
    assert out[0]["item_id"] == "a"      # starts lowest-energy
    assert out[-1]["item_id"] == "c"     # outlier lands last


def test_batched_catalogue_fit_matches_sequential():
    """_fit_catalogue_batched (one masked EM per k across all artists —
    the GPU path) produces models equivalent to the per-artist fits:
    same similarity ordering and close chamfer distances."""
    rng = np.random.default_rng(7)
    base = rng.standard_normal(16).astype(np.float32) * 3
    per_artist = {
        "a1": base + rng.standard_normal((30, 16)).astype(np.float32) * 0.2,
        "a2": base + rng.standard_normal((22, 16)).astype(np.float32) * 0.2,
        "b": -base + rng.standard_normal((40, 16)).astype(np.float32) * 0.2,
        "tiny": rng.standard_normal((1, 16)).astype(np.float32) * 3,
        "duo": rng.standard_normal((2, 16)).astype(np.float32) * 3,
    }
    seq = ArtistSimilarity()
    seq.fit_catalogue(per_artist, seed=0)
    bat = ArtistSimilarity()
    bat._fit_catalogue_batched(list(per_artist.items()), seed=0,
                               device="cpu")
    bat._names = list(bat.models)
    cents = [m.means.mean(axis=0) for m in bat.models.values()]
    bat._centroids = np.stack(cents).astype(np.float32)
    assert set(bat.models) == set(per_artist)
    for name in ("a1", "a2", "b"):
        d = soft_chamfer_distance(seq.models[name], bat.models[name])
        assert d < 0.05, (name, d)
    res = bat.find_similar_artists("a1", n=2)
    assert res[0][0] == "a2"


@pytest.mark.gpu
def test_batched_catalogue_fit_on_device():
    """fit_catalogue dispatches to the batched masked-EM path on GPU
    (>= 8 artists) and yields a queryable similarity catalogue."""
    rng = np.random.default_rng(3)
    base = rng.standard_normal(32).astype(np.float32) * 3
    per_artist = {f"c{i}": base * (1 if i < 5 else -1)
                  + rng.standard_normal((20 + i, 32)).astype(np.float32) * 0.2
                  for i in range(10)}
    sim = ArtistSimilarity()
    sim.fit_catalogue(per_artist, seed=0)
    assert len(sim.models) == 10
    res = sim.find_similar_artists("c0", n=3)
    assert all(name.startswith("c") for name, _ in res)
    assert {name for name, _ in res} <= {f"c{i}" for i in range(1, 5)}
