"""Clustering fitness metrics on torch.

Reference: /root/reference/tasks/clustering_helper.py:689-1030 — the
7-metric fitness: silhouette / Davies-Bouldin / Calinski-Harabasz plus
mood and other-feature purity & diversity, combined after log1p + z-norm
weighting. Implemented GEMM-shaped for GPU; identical math validated
against sklearn on CPU in tests/test_cluster.py.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from audiomuse_amd.cluster.algorithms import _pairwise_sq


def _valid(labels: torch.Tensor) -> torch.Tensor:
    return labels >= 0


def silhouette_score(x: torch.Tensor, labels: torch.Tensor,
                     max_points: int = 3000, seed: int = 0) -> float:
    """Mean silhouette over (a sample of) labeled points."""
    m = _valid(labels)
    x, labels = x[m].float(), labels[m]
    n = x.shape[0]
    uniq = labels.unique()
    if n < 3 or uniq.numel() < 2:
        return -1.0
    if n > max_points:
        g = torch.Generator().manual_seed(seed)
        idx = torch.randperm(n, generator=g)[:max_points].to(x.device)
        sample_x, sample_l = x[idx], labels[idx]
    else:
        sample_x, sample_l = x, labels
    d = _pairwise_sq(sample_x, x).sqrt()                   # (s, n)
    s_vals = []
    onehot = torch.stack([(labels == c) for c in uniq.tolist()]).float()  # (k, n)
    counts = onehot.sum(dim=1)                             # (k,)
    sums = d @ onehot.T                                    # (s, k)
    for i in range(sample_x.shape[0]):
        ci = int((uniq == sample_l[i]).nonzero()[0])
        same = counts[ci] - 1
        if same <= 0:
            s_vals.append(0.0)
            continue
        a = float(sums[i, ci] / same)
        other = [float(sums[i, j] / counts[j])
                 for j in range(uniq.numel()) if j != ci]
        b = min(other)
        s_vals.append((b - a) / max(a, b, 1e-12))
    return float(torch.tensor(s_vals).mean())


def davies_bouldin(x: torch.Tensor, labels: torch.Tensor) -> float:
    m = _valid(labels)
    x, labels = x[m].float(), labels[m]
    uniq = labels.unique()
    k = uniq.numel()
    if k < 2:
        return float("inf")
    cents = torch.stack([x[labels == c].mean(dim=0) for c in uniq.tolist()])
    scatter = torch.stack([
        (x[labels == c] - cents[i]).norm(dim=1).mean()
        for i, c in enumerate(uniq.tolist())])
    dist = _pairwise_sq(cents, cents).sqrt()
    ratio = (scatter.unsqueeze(0) + scatter.unsqueeze(1)) / (dist + torch.eye(k, device=x.device) * 1e12)
    ratio.fill_diagonal_(0.0)
    return float(ratio.max(dim=1).values.mean())


def calinski_harabasz(x: torch.Tensor, labels: torch.Tensor) -> float:
    m = _valid(labels)
    x, labels = x[m].float(), labels[m]
    n = x.shape[0]
    uniq = labels.unique()
    k = uniq.numel()
    if k < 2 or n <= k:
        return 0.0
    mean = x.mean(dim=0)
    bss = 0.0
    wss = 0.0
    for c in uniq.tolist():
        pts = x[labels == c]
        cent = pts.mean(dim=0)
        bss += pts.shape[0] * float((cent - mean).square().sum())
        wss += float((pts - cent).square().sum())
    if wss <= 0:
        return 0.0
    return (bss / (k - 1)) / (wss / (n - k))


def purity_diversity(score_vectors: torch.Tensor, labels: torch.Tensor,
                     top_k: Optional[int] = None) -> Dict[str, float]:
    """Mood/other-feature purity (mean top-k class share per cluster —
    reference TOP_K_MOODS_FOR_PURITY_CALCULATION) and diversity
    (distinct predominant classes across clusters / k)."""
    if top_k is None:
        from audiomuse_amd import config as C
        top_k = C.TOP_K_MOODS_FOR_PURITY_CALCULATION
    m = _valid(labels)
    sv, labels = score_vectors[m].float(), labels[m]
    uniq = labels.unique()
    if uniq.numel() == 0 or sv.numel() == 0:
        return {"purity": 0.0, "diversity": 0.0}
    predominant = []
    purities = []
    k_eff = max(1, min(int(top_k), sv.shape[1]))
    for c in uniq.tolist():
        mean_scores = sv[labels == c].mean(dim=0)
        total = float(mean_scores.sum())
        topv = mean_scores.topk(k_eff).values
        predominant.append(int(mean_scores.argmax()))
        purities.append(float(topv.sum()) / max(total, 1e-12))
    return {
        "purity": float(torch.tensor(purities).mean()),
        "diversity": len(set(predominant)) / max(len(predominant), 1),
    }


def _default_weights() -> Dict[str, float]:
    """Config-driven fitness weights (reference PARAMETERS.md
    SCORE_WEIGHT_*; clustering_helper.py:689). Read at call time so DB
    overrides apply without a restart."""
    from audiomuse_amd import config as C
    return {
        "silhouette": C.SCORE_WEIGHT_SILHOUETTE,
        "davies_bouldin": C.SCORE_WEIGHT_DAVIES_BOULDIN,
        "calinski_harabasz": C.SCORE_WEIGHT_CALINSKI_HARABASZ,
        "mood_purity": C.SCORE_WEIGHT_PURITY,
        "mood_diversity": C.SCORE_WEIGHT_DIVERSITY,
        "other_purity": C.SCORE_WEIGHT_OTHER_FEATURE_PURITY,
        "other_diversity": C.SCORE_WEIGHT_OTHER_FEATURE_DIVERSITY,
    }


def fitness(x_metrics: torch.Tensor, labels: torch.Tensor,
            mood_scores: Optional[torch.Tensor] = None,
            other_scores: Optional[torch.Tensor] = None,
            weights: Optional[Dict[str, float]] = None) -> Dict[str, float]:
    """Composite fitness (reference combines with log1p + z-ish scaling:
    raw metrics are squashed to comparable ranges before weighting)."""
    w = _default_weights()
    if weights:
        w.update(weights)
    # the configured weights may zero every term that is computable for
    # this call (e.g. geometry-only scoring with the reference defaults,
    # which put all weight on mood purity/diversity) — fall back to
    # equal weights over the available terms so fitness stays ordered
    relevant = ["silhouette", "davies_bouldin", "calinski_harabasz"]
    if mood_scores is not None:
        relevant += ["mood_purity", "mood_diversity"]
    if other_scores is not None:
        relevant += ["other_purity", "other_diversity"]
    if all(w.get(k, 0.0) == 0.0 for k in relevant):
        w = {k: 1.0 for k in w}
    uniq = labels[labels >= 0].unique()
    out: Dict[str, float] = {}
    if uniq.numel() < 2:
        out["fitness_score"] = -1.0
        return out
    sil = silhouette_score(x_metrics, labels)
    db = davies_bouldin(x_metrics, labels)
    ch = calinski_harabasz(x_metrics, labels)
    out.update(silhouette=sil, davies_bouldin=db, calinski_harabasz=ch)
    # squash to [0, 1]-ish: sil already [-1,1]; DB lower-better -> 1/(1+db);
    # CH unbounded -> log1p scaling
    score = (w["silhouette"] * (sil + 1.0) / 2.0
             + w["davies_bouldin"] / (1.0 + max(db, 0.0))
             + w["calinski_harabasz"] * float(torch.log1p(torch.tensor(max(ch, 0.0)))) / 10.0)
    denom = w["silhouette"] + w["davies_bouldin"] + w["calinski_harabasz"]
    if mood_scores is not None:
        pd = purity_diversity(mood_scores, labels)
        out["mood_purity"], out["mood_diversity"] = pd["purity"], pd["diversity"]
        score += w["mood_purity"] * pd["purity"] + w["mood_diversity"] * pd["diversity"]
        denom += w["mood_purity"] + w["mood_diversity"]
    if other_scores is not None:
        pd = purity_diversity(other_scores, labels)
        out["other_purity"], out["other_diversity"] = pd["purity"], pd["diversity"]
        score += w["other_purity"] * pd["purity"] + w["other_diversity"] * pd["diversity"]
        denom += w["other_purity"] + w["other_diversity"]
    out["fitness_score"] = score / max(denom, 1e-12)
    return out
