"""Evolutionary playlist clustering driver.

Reference: /root/reference/tasks/clustering.py (orchestration: batches of
ITERATIONS_PER_BATCH_JOB, elite pool, calibration, stall valve, absorb)
+ clustering_helper.py (one iteration: sample -> scale -> param gen
explore/exploit -> optional PCA -> fit -> playlist trim -> 7-metric
fitness -> naming) + clustering_postprocessing.py (winner cleanup, Top-N
"6+4" diverse selection by centroid max-min distance).

Math runs through cluster/algorithms.py + cluster/fitness.py (torch,
GPU-capable); this module is the search logic. Distributed execution:
run_clustering_task enqueues batch jobs through the task queue, sibling
workers claim them, the parent absorbs results (same shape as the
reference's Postgres-mediated reduction).
"""

from __future__ import annotations

import random
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch

from audiomuse_amd import config as C
from audiomuse_amd.cluster.algorithms import fit_predict
from audiomuse_amd.cluster.fitness import fitness


@dataclass
class TrackRow:
    item_id: str
    title: str = ""
    author: str = ""
    mood_vector: Dict[str, float] = field(default_factory=dict)
    other_features: Dict[str, float] = field(default_factory=dict)


@dataclass
class IterationResult:
    params: Dict
    fitness: Dict[str, float]
    playlists: Dict[str, List[str]]            # name -> item_ids
    centroids: Dict[str, List[float]]

    @property
    def score(self) -> float:
        return self.fitness.get("fitness_score", -1.0)


def _param_space(algorithm: str, n: int, rng: random.Random) -> Dict:
    """Explore ranges come from config (reference PARAMETERS.md:
    NUM_CLUSTERS_MIN/MAX, DBSCAN_*, GMM_*, SPECTRAL_*, PCA_*)."""
    if algorithm == "kmeans":
        hi = max(C.NUM_CLUSTERS_MIN + 1, min(C.NUM_CLUSTERS_MAX, n // 10))
        return {"n_clusters": rng.randint(C.NUM_CLUSTERS_MIN, hi)}
    if algorithm == "dbscan":
        return {"eps": rng.uniform(C.DBSCAN_EPS_MIN, C.DBSCAN_EPS_MAX),
                "min_samples": rng.randint(C.DBSCAN_MIN_SAMPLES_MIN,
                                           C.DBSCAN_MIN_SAMPLES_MAX)}
    if algorithm == "gmm":
        hi = max(C.GMM_N_COMPONENTS_MIN + 1,
                 min(C.GMM_N_COMPONENTS_MAX, n // 10))
        return {"n_components": rng.randint(C.GMM_N_COMPONENTS_MIN, hi),
                "covariance_type": C.GMM_COVARIANCE_TYPE}
    if algorithm == "spectral":
        hi = max(C.SPECTRAL_N_CLUSTERS_MIN + 1,
                 min(C.SPECTRAL_N_CLUSTERS_MAX, n // 20))
        return {"n_clusters": rng.randint(C.SPECTRAL_N_CLUSTERS_MIN, hi),
                "n_neighbors": C.SPECTRAL_N_NEIGHBORS}
    raise ValueError(algorithm)


def _sample_pca(rng: random.Random, dim: int) -> int:
    """Optional PCA stage in the explored space (reference
    PCA_COMPONENTS_MIN/MAX; 0 = off)."""
    lo, hi = C.PCA_COMPONENTS_MIN, min(C.PCA_COMPONENTS_MAX, dim - 1)
    if hi <= 0 or hi < lo:
        return 0
    return rng.randint(lo, hi)


def _mutate(params: Dict, algorithm: str, n: int, rng: random.Random) -> Dict:
    """Elite mutation, deltas from config (reference PARAMETERS.md:
    MUTATION_INT_ABS_DELTA / MUTATION_FLOAT_ABS_DELTA)."""
    p = dict(params)
    di = C.MUTATION_INT_ABS_DELTA
    df = C.MUTATION_FLOAT_ABS_DELTA
    if algorithm in ("kmeans", "spectral"):
        # kmeans additionally jitters proportionally to the current k
        # (reference MUTATION_KMEANS_COORD_FRACTION's role: small
        # relative perturbations of the solution between generations)
        rel = (max(1, int(p["n_clusters"] * C.MUTATION_KMEANS_COORD_FRACTION))
               if algorithm == "kmeans" else di)
        p["n_clusters"] = max(2, p["n_clusters"]
                              + rng.randint(-max(di, rel), max(di, rel)))
    elif algorithm == "gmm":
        p["n_components"] = max(2, p["n_components"] + rng.randint(-di, di))
    else:
        p["eps"] = max(0.05, p["eps"] * rng.uniform(1.0 - df, 1.0 + df))
        p["min_samples"] = max(2, p["min_samples"] + rng.randint(-2, 2))
    return p


def _name_cluster(mood_centroid: Dict[str, float], used: set) -> str:
    """Name from the top moods (reference _name_cluster: predominant
    mood labels joined; de-duplicated with a numeric suffix)."""
    top = sorted(mood_centroid.items(), key=lambda kv: -kv[1])[:2]
    base = " & ".join(k.title() for k, _ in top) if top else "Mix"
    name = f"{base}_automatic"
    i = 2
    while name in used:
        name = f"{base} {i}_automatic"
        i += 1
    used.add(name)
    return name


def _trim_cluster(order: Sequence[int], rows: Sequence[TrackRow],
                  max_per_artist: int, max_songs: int) -> List[int]:
    """Distance-ordered trim with the per-artist cap (reference
    clustering_helper.py:742-780)."""
    count_per_artist: Dict[str, int] = {}
    picked: List[int] = []
    for i in order:
        artist = (rows[i].author or "").strip().lower()
        if max_per_artist > 0 and count_per_artist.get(artist, 0) >= max_per_artist:
            continue
        picked.append(i)
        count_per_artist[artist] = count_per_artist.get(artist, 0) + 1
        if max_songs > 0 and len(picked) >= max_songs:
            break
    return picked


def run_iteration(x: torch.Tensor, rows: Sequence[TrackRow], algorithm: str,
                  params: Dict, *, mood_labels: Sequence[str] = (),
                  other_labels: Sequence[str] = (),
                  pca_components: int = 0, max_songs_per_cluster: int = 0,
                  max_per_artist: Optional[int] = None) -> IterationResult:
    """One clustering iteration (clustering_helper.py:236-338)."""
    from audiomuse_amd.cluster.algorithms import pca_fit_transform

    x = x.float()
    data = x
    # standard-scale (reference StandardScaler)
    mean, std = data.mean(dim=0), data.std(dim=0).clamp(min=1e-9)
    data = (data - mean) / std
    if pca_components and pca_components < data.shape[1]:
        data, _, _ = pca_fit_transform(data, pca_components)
    labels, centers = fit_predict(algorithm, data, params)

    mood_labels = list(mood_labels) or C.MOOD_LABELS
    other_labels = list(other_labels) or C.OTHER_FEATURE_LABELS
    moods = torch.tensor([[r.mood_vector.get(m, 0.0) for m in mood_labels]
                          for r in rows], dtype=torch.float32, device=x.device)
    others = torch.tensor([[r.other_features.get(m, 0.0) for m in other_labels]
                           for r in rows], dtype=torch.float32, device=x.device)
    fit = fitness(data, labels, mood_scores=moods, other_scores=others)

    playlists: Dict[str, List[str]] = {}
    centroids: Dict[str, List[float]] = {}
    used_names: set = set()
    max_pa = C.MAX_SONGS_PER_ARTIST if max_per_artist is None else max_per_artist
    for c in labels[labels >= 0].unique().tolist():
        idx = (labels == c).nonzero(as_tuple=True)[0]
        if idx.numel() == 0:
            continue
        center = data[idx].mean(dim=0)
        dists = (data[idx] - center).norm(dim=1)
        order = idx[dists.argsort()].tolist()
        picked = _trim_cluster(order, rows, max_pa, max_songs_per_cluster)
        if not picked:
            continue
        mood_centroid = {m: float(moods[picked, i].mean())
                         for i, m in enumerate(mood_labels)}
        name = _name_cluster(mood_centroid, used_names)
        playlists[name] = [rows[i].item_id for i in picked]
        centroids[name] = x[picked].mean(dim=0).cpu().tolist()
    return IterationResult(params=params, fitness=fit, playlists=playlists,
                           centroids=centroids)


def _stratified_subset(rows: Sequence[TrackRow], sub: int,
                       rng: random.Random) -> List[int]:
    """Stratified sample by predominant mood (reference
    _get_stratified_song_subset :1211): genres with at least
    MIN_SONGS_PER_GENRE_FOR_STRATIFICATION members are each sampled
    toward the STRATIFIED_SAMPLING_TARGET_PERCENTILE of genre sizes so
    giant genres cannot crowd out the rest of the subset."""
    by_mood: Dict[str, List[int]] = {}
    for i, r in enumerate(rows):
        top = max(r.mood_vector, key=r.mood_vector.get) \
            if r.mood_vector else "unknown"
        by_mood.setdefault(top, []).append(i)
    big = {m: idxs for m, idxs in by_mood.items()
           if len(idxs) >= C.MIN_SONGS_PER_GENRE_FOR_STRATIFICATION}
    if len(big) < 2:
        return rng.sample(range(len(rows)), sub)
    sizes = sorted(len(v) for v in big.values())
    pct = min(max(C.STRATIFIED_SAMPLING_TARGET_PERCENTILE, 0.0), 100.0)
    target = sizes[min(len(sizes) - 1, int(len(sizes) * pct / 100.0))]
    picked: List[int] = []
    for idxs in big.values():
        take = min(len(idxs), target)
        picked.extend(rng.sample(idxs, take))
    small = [i for m, idxs in by_mood.items() if m not in big
             for i in idxs]
    picked.extend(small)
    if len(picked) > sub:
        picked = rng.sample(picked, sub)
    elif len(picked) < sub:
        rest = list(set(range(len(rows))) - set(picked))
        picked.extend(rng.sample(rest, min(sub - len(picked), len(rest))))
    return picked


def evolutionary_search(x: torch.Tensor, rows: Sequence[TrackRow],
                        algorithm: Optional[str] = None, *,
                        runs: Optional[int] = None,
                        elite_size: Optional[int] = None,
                        exploit_prob: Optional[float] = None,
                        stall_limit: int = 30,
                        seed: int = 0, subset: Optional[int] = None,
                        seed_params: Optional[Dict] = None,
                        max_songs_per_cluster: Optional[int] = None,
                        progress_cb=None) -> List[IterationResult]:
    """Explore/exploit search with an elite pool and a stall valve
    (clustering.py:383-1449). Returns elites sorted best-first."""
    algorithm = algorithm or C.CLUSTER_ALGORITHM
    runs = runs or C.CLUSTERING_RUNS
    elite_size = elite_size if elite_size is not None else C.TOP_N_ELITES
    exploit_prob = (exploit_prob if exploit_prob is not None
                    else C.EXPLOITATION_PROBABILITY_CONFIG)
    if max_songs_per_cluster is None:
        max_songs_per_cluster = C.MAX_SONGS_PER_CLUSTER
    rng = random.Random(seed)
    n = x.shape[0]
    sub = min(subset or C.CLUSTERING_SUBSET_SONGS, n)
    elites: List[IterationResult] = []
    stall = 0
    for it in range(runs):
        # subset size wobbles per run (reference
        # SAMPLING_PERCENTAGE_CHANGE_PER_RUN) so elites do not overfit
        # one sample
        wobble = 1.0 + rng.uniform(-C.SAMPLING_PERCENTAGE_CHANGE_PER_RUN,
                                   C.SAMPLING_PERCENTAGE_CHANGE_PER_RUN)
        sub_it = max(16, min(n, int(sub * wobble)))
        if sub_it < n:
            pick = torch.tensor(_stratified_subset(rows, sub_it, rng),
                                device=x.device)
            xs = x[pick]
            rs = [rows[i] for i in pick.tolist()]
        else:
            xs, rs = x, rows
        exploit_allowed = it >= runs * C.EXPLOITATION_START_FRACTION
        if elites and exploit_allowed and rng.random() < exploit_prob:
            params = _mutate(rng.choice(elites).params, algorithm, sub, rng)
        elif seed_params and it == 0:
            params = dict(seed_params)   # calibration probe's winner
        else:
            params = _param_space(algorithm, sub, rng)
        params["seed"] = rng.randint(0, 2**31 - 1)
        pca_k = _sample_pca(rng, int(x.shape[1]))
        try:
            result = run_iteration(xs, rs, algorithm, params,
                                   pca_components=pca_k,
                                   max_songs_per_cluster=max_songs_per_cluster)
        except Exception:
            continue
        best_before = elites[0].score if elites else -1.0
        elites.append(result)
        elites.sort(key=lambda r: -r.score)
        del elites[elite_size:]
        stall = 0 if elites[0].score > best_before + 1e-6 else stall + 1
        if progress_cb is not None:
            progress_cb(it + 1, runs, elites[0].score)
        if stall >= stall_limit:
            break
    return elites


def diverse_top_n(elite: IterationResult, n: Optional[int] = None,
                  min_size: Optional[int] = None) -> Dict[str, List[str]]:
    """Winner post-processing (clustering_postprocessing.py:66): drop tiny
    playlists, then pick Top-N diverse by centroid max-min distance
    (the reference's "6+4": half largest, half most diverse)."""
    n = n or C.TOP_N_PLAYLISTS
    min_size = (min_size if min_size is not None
                else C.MIN_PLAYLIST_SIZE_FOR_TOP_N)
    items = [(name, ids) for name, ids in elite.playlists.items()
             if len(ids) >= min_size]
    if len(items) <= n:
        return dict(items)
    by_size = sorted(items, key=lambda kv: -len(kv[1]))
    keep = by_size[: (n + 1) // 2]
    rest = by_size[(n + 1) // 2:]
    cents = {name: np.asarray(elite.centroids.get(name, []), dtype=np.float32)
             for name, _ in items}
    while len(keep) < n and rest:
        kept_c = [cents[k[0]] for k in keep if cents[k[0]].size]
        best_i, best_d = 0, -1.0
        for i, (name, _ids) in enumerate(rest):
            c = cents[name]
            if not c.size or not kept_c:
                d = 0.0
            else:
                d = min(float(np.linalg.norm(c - kc)) for kc in kept_c)
            if d > best_d:
                best_d, best_i = d, i
        keep.append(rest.pop(best_i))
    return dict(keep)
