"""Hyperbolic explorer: Poincare-ball projection + exact distances.

Reference: /root/reference/tasks/hyperbolic_geometry.py:31-99 —
proj(x) = tanh(||x|| / s) * x / ||x||  (scale s calibrated so the
percentile-P norm maps to a target radius), exact Poincare distance
d(u,v) = arccosh(1 + 2||u-v||^2 / ((1-||u||^2)(1-||v||^2))) — and
tasks/hyperbolic_manager.py (tree cache + hyperbolic_similar). Math on
torch, GPU-capable for full-matrix queries.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from audiomuse_amd import config as C

_EPS = 1e-7
_MAX_NORM = 1.0 - 1e-5


def calibrate_scale(norms: torch.Tensor,
                    percentile: Optional[float] = None,
                    target_radius: Optional[float] = None) -> float:
    """Scale s so the percentile-P norm projects to target_radius
    (hyperbolic_geometry.py:101). Defaults: HYPERBOLIC_RADIUS_PERCENTILE
    picks the calibrated percentile and HYPERBOLIC_RADIUS_SCALE scales
    the 0.85 target radius (both reference knobs)."""
    if percentile is None:
        percentile = getattr(C, "HYPERBOLIC_RADIUS_PERCENTILE",
                             C.HYPERBOLIC_SCALE_PERCENTILE)
    if target_radius is None:
        target_radius = min(0.85 * C.HYPERBOLIC_RADIUS_SCALE, 0.999)
    if norms.numel() == 0:
        return 1.0
    p = float(torch.quantile(norms.float(), percentile / 100.0))
    if p <= 0:
        return 1.0
    # tanh(p / s) = target_radius  =>  s = p / atanh(target_radius)
    return p / float(torch.atanh(torch.tensor(target_radius)))


def project(x: torch.Tensor, scale: float) -> torch.Tensor:
    """Poincare-ball projection (hyperbolic_geometry.py:31)."""
    norms = x.norm(dim=-1, keepdim=True).clamp(min=_EPS)
    r = torch.tanh(norms / scale).clamp(max=_MAX_NORM)
    return r * x / norms


def poincare_distance(u: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """Exact distance; u (..., d), v (..., d) broadcastable
    (hyperbolic_geometry.py:55)."""
    diff_sq = (u - v).square().sum(dim=-1)
    un = u.square().sum(dim=-1).clamp(max=_MAX_NORM ** 2)
    vn = v.square().sum(dim=-1).clamp(max=_MAX_NORM ** 2)
    arg = 1.0 + 2.0 * diff_sq / ((1.0 - un) * (1.0 - vn)).clamp(min=_EPS)
    return torch.acosh(arg.clamp(min=1.0))


def distance_matrix(points: torch.Tensor, queries: torch.Tensor
                    ) -> torch.Tensor:
    """(Q, N) Poincare distances, GEMM-shaped expansion."""
    return poincare_distance(queries.unsqueeze(1), points.unsqueeze(0))


class HyperbolicSpace:
    """Projected catalogue + nearest queries (hyperbolic_manager core)."""

    def __init__(self, embeddings: torch.Tensor,
                 percentile: Optional[float] = None):
        norms = embeddings.norm(dim=1)
        self.scale = calibrate_scale(norms, percentile)
        self.points = project(embeddings.float(), self.scale)

    def similar(self, row: int, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
        """(distances, indices) of the k nearest in hyperbolic space."""
        q = self.points[row].unsqueeze(0)
        d = distance_matrix(self.points, q)[0]
        d[row] = float("inf")
        top = torch.topk(d, min(k, d.numel() - 1), largest=False)
        return top.values, top.indices

    def similar_to_vector(self, vec: torch.Tensor, k: int
                          ) -> Tuple[torch.Tensor, torch.Tensor]:
        q = project(vec.float().unsqueeze(0), self.scale)
        d = distance_matrix(self.points, q)[0]
        top = torch.topk(d, min(k, d.numel()), largest=False)
        return top.values, top.indices
