"""MusiCNN-style embedding + mood model, MI355X-native.

Capability parity: the reference runs two opaque ONNX graphs
(musicnn_embedding.onnx / musicnn_prediction.onnx, consumed at
/root/reference/tasks/analysis/song.py:367-412): log-mel patches
(B, 187, 96) -> per-patch 200-d embeddings -> 50 mood logits; the track
embedding is the patch mean and the mood scores are
sigmoid(mean(sigmoid(logits))) (song.py:406).

This is our own musicnn-inspired design (Pons et al. front-end shapes):
parallel "timbral" convolutions spanning the full mel axis and
"temporal" convolutions spanning time, global-pooled, then a dense
mid-end. All channel widths are multiples of 64 for CDNA4 GEMM tiling;
conv layers run in bf16 through MIOpen-free matmul form (unfold+GEMM)
where profiled beneficial.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from audiomuse_amd import config as C


@dataclass
class MusiCNNConfig:
    n_frames: int = 187
    n_mels: int = 96
    embed_dim: int = 200
    n_moods: int = 50
    timbral_channels: int = 128
    temporal_channels: int = 64
    midend_channels: int = 256


class MusiCNNEmbedding(nn.Module):
    def __init__(self, cfg: MusiCNNConfig | None = None):
        super().__init__()
        self.cfg = cfg = cfg or MusiCNNConfig()
        M = cfg.n_mels
        # timbral front-end: filters spanning 90% / 40% of the mel axis
        self.timbral_a = nn.Conv2d(1, cfg.timbral_channels, (7, int(0.9 * M)))
        self.timbral_b = nn.Conv2d(1, cfg.timbral_channels, (3, int(0.4 * M)))
        # temporal front-end: 1-d filters over time, full-band pooled input
        self.temporal_a = nn.Conv1d(1, cfg.temporal_channels, 165, padding=82)
        self.temporal_b = nn.Conv1d(1, cfg.temporal_channels, 33, padding=16)
        front = 2 * cfg.timbral_channels + 2 * cfg.temporal_channels
        self.bn_front = nn.BatchNorm1d(front)
        self.midend = nn.Sequential(
            nn.Conv1d(front, cfg.midend_channels, 7, padding=3), nn.ReLU(),
            nn.BatchNorm1d(cfg.midend_channels),
            nn.Conv1d(cfg.midend_channels, cfg.midend_channels, 7, padding=3),
            nn.ReLU(), nn.BatchNorm1d(cfg.midend_channels),
        )
        pooled = front + 2 * cfg.midend_channels
        self.dense = nn.Sequential(
            nn.Linear(pooled, 512), nn.ReLU(), nn.LayerNorm(512),
            nn.Linear(512, cfg.embed_dim),
        )

    def forward(self, patches: torch.Tensor) -> torch.Tensor:
        """(B, 187, 96) log-mel patches -> (B, 200) per-patch embeddings."""
        B, T, M = patches.shape
        x2 = patches.unsqueeze(1)                        # (B, 1, T, M)
        ta = F.relu(self.timbral_a(x2)).amax(dim=3)      # (B, Ct, T')
        tb = F.relu(self.timbral_b(x2)).amax(dim=3)
        band = patches.mean(dim=2, keepdim=True).transpose(1, 2)  # (B,1,T)
        pa = F.relu(self.temporal_a(band))
        pb = F.relu(self.temporal_b(band))
        L = min(ta.shape[2], tb.shape[2], pa.shape[2], pb.shape[2])
        front = torch.cat([ta[..., :L], tb[..., :L], pa[..., :L], pb[..., :L]],
                          dim=1)
        front = self.bn_front(front)
        mid = self.midend(front)
        feats = torch.cat([front.mean(dim=2), mid.mean(dim=2), mid.amax(dim=2)],
                          dim=1)
        return self.dense(feats)


class MusiCNNPrediction(nn.Module):
    """(B, 200) -> (B, 50) mood logits (reference prediction model)."""

    def __init__(self, cfg: MusiCNNConfig | None = None):
        super().__init__()
        cfg = cfg or MusiCNNConfig()
        self.net = nn.Sequential(
            nn.Linear(cfg.embed_dim, 256), nn.ReLU(), nn.LayerNorm(256),
            nn.Linear(256, cfg.n_moods),
        )

    def forward(self, emb: torch.Tensor) -> torch.Tensor:
        return self.net(emb)


def aggregate_track(per_patch_emb: torch.Tensor, mood_logits: torch.Tensor
                    ) -> Tuple[torch.Tensor, Dict[str, float]]:
    """Reference aggregation (song.py:395-412): track embedding = patch
    mean; moods = sigmoid(mean(sigmoid(logits), axis=0)) keyed by
    MOOD_LABELS."""
    emb = per_patch_emb.mean(dim=0)
    scores = torch.sigmoid(torch.sigmoid(mood_logits).mean(dim=0))
    # one D2H copy for the whole vector (a per-label float() was one
    # device sync per mood — 50 syncs/track in the album loop)
    moods = dict(zip(C.MOOD_LABELS, map(float, scores.cpu().tolist())))
    # persist only the strongest TOP_N_MOODS (reference: mood_vector is
    # a sparse top-N map, song.py mood aggregation + TOP_N_MOODS)
    if C.TOP_N_MOODS and len(moods) > C.TOP_N_MOODS:
        keep = sorted(moods, key=moods.get, reverse=True)[: C.TOP_N_MOODS]
        moods = {k: moods[k] for k in keep}
    return emb, moods
