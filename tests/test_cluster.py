"""Clustering engine tests: algorithms vs sklearn, fitness metrics,
evolutionary driver."""

import numpy as np
import pytest
import torch
from sklearn.cluster import DBSCAN as SkDBSCAN
from sklearn.cluster import KMeans as SkKMeans
from sklearn.metrics import (calinski_harabasz_score, davies_bouldin_score,
                             silhouette_score as sk_silhouette)
from sklearn.mixture import GaussianMixture as SkGMM

from audiomuse_amd.cluster import algorithms as alg
from audiomuse_amd.cluster import fitness as fit
from audiomuse_amd.cluster.evolve import (TrackRow, diverse_top_n,
                                          evolutionary_search, run_iteration)


def _blobs(n=600, k=4, d=8, seed=0, spread=0.3):
    g = torch.Generator().manual_seed(seed)
    centers = torch.randn(k, d, generator=g) * 4
    assign = torch.randint(0, k, (n,), generator=g)
    return centers[assign] + torch.randn(n, d, generator=g) * spread, assign


def _agreement(a, b):
    """Cluster agreement via best-match relabeling (permutation-invariant)."""
    a = np.asarray(a); b = np.asarray(b)
    match = 0
    for c in np.unique(a):
        vals, counts = np.unique(b[a == c], return_counts=True)
        match += counts.max()
    return match / len(a)


def test_kmeans_recovers_blobs():
    x, truth = _blobs()
    r = alg.kmeans_fit(x, 4, seed=0)
    assert _agreement(truth.numpy(), r.labels.numpy()) > 0.97
    sk = SkKMeans(n_clusters=4, n_init=5, random_state=0).fit(x.numpy())
    assert abs(r.inertia - sk.inertia_) / sk.inertia_ < 0.25


def test_dbscan_matches_sklearn_on_blobs():
    x, _ = _blobs(300, 3, 5, spread=0.2)
    ours = alg.dbscan_fit(x, eps=0.8, min_samples=4).numpy()
    theirs = SkDBSCAN(eps=0.8, min_samples=4).fit_predict(x.numpy())
    # identical core-point clustering up to label permutation
    assert (ours == -1).sum() == (theirs == -1).sum()
    mask = theirs >= 0
    assert _agreement(theirs[mask], ours[mask]) > 0.99


def test_gmm_diag_fits_blobs():
    x, truth = _blobs(500, 3, 6, spread=0.25, seed=1)
    r = alg.gmm_fit(x, 3, seed=1)
    assert _agreement(truth.numpy(), r.labels.numpy()) > 0.95
    sk = SkGMM(n_components=3, covariance_type="diag", random_state=1,
               max_iter=60).fit(x.numpy())
    # log-likelihood in the same ballpark
    ours = r.log_likelihood / x.shape[0]
    theirs = sk.score(x.numpy())
    assert ours > theirs - 1.0


def test_gmm_bic_prefers_true_k():
    x, _ = _blobs(600, 3, 4, spread=0.2, seed=2)
    bics = {k: alg.gmm_fit(x, k, seed=2).bic(x.shape[0]) for k in (1, 2, 3, 5)}
    assert min(bics, key=bics.get) == 3


def test_pca_matches_svd_reconstruction():
    x, _ = _blobs(200, 2, 10)
    proj, comps, mean = alg.pca_fit_transform(x, 3)
    assert proj.shape == (200, 3) and comps.shape == (3, 10)
    recon = proj @ comps + mean
    resid = (x - recon).norm() / x.norm()
    from sklearn.decomposition import PCA
    sk = PCA(n_components=3).fit(x.numpy())
    sk_recon = sk.inverse_transform(sk.transform(x.numpy()))
    sk_resid = np.linalg.norm(x.numpy() - sk_recon) / np.linalg.norm(x.numpy())
    assert abs(resid - sk_resid) < 1e-3


def test_spectral_recovers_blobs():
    x, truth = _blobs(300, 3, 5, spread=0.2, seed=3)
    labels = alg.spectral_fit(x, 3, seed=3)
    assert _agreement(truth.numpy(), labels.numpy()) > 0.9


def test_fitness_metrics_match_sklearn():
    x, truth = _blobs(400, 4, 6, seed=4)
    labels = truth
    sil = fit.silhouette_score(x, labels, max_points=400)
    db = fit.davies_bouldin(x, labels)
    ch = fit.calinski_harabasz(x, labels)
    np.testing.assert_allclose(sil, sk_silhouette(x.numpy(), labels.numpy()),
                               atol=2e-3)
    np.testing.assert_allclose(db, davies_bouldin_score(x.numpy(), labels.numpy()),
                               rtol=1e-4)
    np.testing.assert_allclose(ch, calinski_harabasz_score(x.numpy(), labels.numpy()),
                               rtol=1e-4)


def test_fitness_good_beats_bad():
    x, truth = _blobs(300, 3, 5, seed=5)
    good = fit.fitness(x, truth)
    bad = fit.fitness(x, torch.randint(0, 3, (300,)))
    assert good["fitness_score"] > bad["fitness_score"]


def _rows(n, moods=("rock", "jazz", "pop")):
    rng = np.random.default_rng(0)
    rows = []
    for i in range(n):
        m = {lbl: float(v) for lbl, v in
             zip(moods, rng.dirichlet(np.ones(len(moods))))}
        rows.append(TrackRow(item_id=f"t{i}", title=f"T{i}",
                             author=f"artist{i % 17}", mood_vector=m,
                             other_features={"happy": 0.5}))
    return rows


def test_run_iteration_produces_named_playlists():
    x, _ = _blobs(200, 3, 6, seed=6)
    rows = _rows(200)
    res = run_iteration(x, rows, "kmeans", {"n_clusters": 3, "seed": 0},
                        max_songs_per_cluster=20)
    assert res.playlists and all(name.endswith("_automatic")
                                 for name in res.playlists)
    assert all(len(v) <= 20 for v in res.playlists.values())
    assert res.score > -1.0


def test_artist_cap_enforced(monkeypatch):
    from audiomuse_amd import config as C
    monkeypatch.setattr(C, "MAX_SONGS_PER_ARTIST", 2)
    x, _ = _blobs(100, 2, 4, seed=7)
    rows = _rows(100)
    res = run_iteration(x, rows, "kmeans", {"n_clusters": 2, "seed": 0})
    for ids in res.playlists.values():
        authors = [rows[int(i[1:])].author for i in ids]
        for a in set(authors):
            assert authors.count(a) <= 2


def test_evolutionary_search_improves_and_stops():
    x, _ = _blobs(300, 4, 6, seed=8)
    rows = _rows(300)
    calls = []
    elites = evolutionary_search(x, rows, "kmeans", runs=25, stall_limit=8,
                                 seed=0, subset=300,
                                 progress_cb=lambda i, n, s: calls.append(s))
    assert elites and elites[0].score >= elites[-1].score
    assert len(calls) >= 8


def test_diverse_top_n_selection():
    x, _ = _blobs(400, 8, 6, seed=9)
    rows = _rows(400)
    res = run_iteration(x, rows, "kmeans", {"n_clusters": 8, "seed": 1})
    top = diverse_top_n(res, n=4, min_size=2)
    assert len(top) <= 4
    assert all(len(v) >= 2 for v in top.values())


def test_gmm_fit_many_matches_single_fits():
    """Batched masked EM (gmm_fit_many — SURVEY P3 artist batching) on
    ragged datasets recovers the same mixtures as per-dataset gmm_fit:
    component means match the true centers and per-dataset BIC is in
    the same ballpark."""
    torch.manual_seed(0)
    xs, truths = [], []
    for i, n in enumerate((40, 90, 140)):
        x, _ = _blobs(n, 2, 12, seed=10 + i, spread=0.2)
        xs.append(x)
        truths.append(x)
    means, weights, bic = alg.gmm_fit_many(xs, 2, seed=3)
    assert means.shape == (3, 2, 12) and weights.shape == (3, 2)
    assert torch.allclose(weights.sum(dim=1), torch.ones(3), atol=1e-4)
    for i, x in enumerate(xs):
        single = alg.gmm_fit(x, 2, seed=3)
        # match batched components to single components greedily
        d = torch.cdist(means[i], single.means)
        assert float(d.min(dim=1).values.max()) < 0.5, i
        assert float(bic[i]) < single.bic(x.shape[0]) + 200.0, i


def test_gmm_fit_many_k1_and_weights():
    xs = [torch.randn(30, 6) + 3.0, torch.randn(5, 6) - 2.0]
    means, weights, bic = alg.gmm_fit_many(xs, 1, seed=0)
    assert torch.allclose(means[0, 0], xs[0].mean(dim=0), atol=1e-3)
    assert torch.allclose(means[1, 0], xs[1].mean(dim=0), atol=1e-3)
    assert torch.allclose(weights, torch.ones(2, 1), atol=1e-5)
