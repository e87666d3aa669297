"""Voice-activity detection (Silero-class recurrent VAD).

Reference capability (/root/reference/lyrics/silero_onnx.py:95-135):
sequential windowed inference — 512-sample windows @ 16 kHz with 64
samples of left context, recurrent state (2, 1, 128), per-window speech
probability; windows merge into speech segments. This is a first-party
GRU VAD with the same interface and windowing.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

import torch
import torch.nn as nn

WINDOW = 512
CONTEXT = 64
SR = 16000


@dataclass
class VADConfig:
    hidden: int = 128
    features: int = 64


class SileroStyleVAD(nn.Module):
    def __init__(self, cfg: VADConfig | None = None):
        super().__init__()
        self.cfg = cfg = cfg or VADConfig()
        self.frontend = nn.Sequential(
            nn.Conv1d(1, cfg.features, 16, stride=8, padding=4), nn.ReLU(),
            nn.Conv1d(cfg.features, cfg.features, 8, stride=4, padding=2),
            nn.ReLU(),
        )
        self.rnn = nn.GRU(cfg.features, cfg.hidden, num_layers=2,
                          batch_first=True)
        self.head = nn.Linear(cfg.hidden, 1)

    def forward(self, window: torch.Tensor,
                state: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """window (B, WINDOW+CONTEXT); state (2, B, hidden).
        Returns (speech prob (B,), new state)."""
        feats = self.frontend(window.unsqueeze(1)).transpose(1, 2)
        out, new_state = self.rnn(feats, state)
        prob = torch.sigmoid(self.head(out[:, -1]))
        return prob.squeeze(-1), new_state

    def initial_state(self, batch: int = 1, device="cpu",
                      dtype=torch.float32) -> torch.Tensor:
        return torch.zeros(2, batch, self.cfg.hidden, device=device, dtype=dtype)


@torch.inference_mode()
def speech_probabilities(model: SileroStyleVAD,
                         audio: torch.Tensor) -> torch.Tensor:
    """Per-window probabilities over 16 kHz mono audio (sequential,
    stateful — silero_onnx windowed loop)."""
    audio = audio.float().flatten()
    n = audio.shape[0]
    state = model.initial_state(device=audio.device)
    probs = []
    for start in range(0, n - WINDOW + 1, WINDOW):
        lo = max(0, start - CONTEXT)
        chunk = audio[lo : start + WINDOW]
        if chunk.shape[0] < WINDOW + CONTEXT:
            chunk = torch.nn.functional.pad(chunk,
                                            (WINDOW + CONTEXT - chunk.shape[0], 0))
        p, state = model(chunk.unsqueeze(0), state)
        probs.append(float(p))
    return torch.tensor(probs)


def speech_segments(probs: torch.Tensor, threshold: float = 0.5,
                    min_windows: int = 3) -> List[Tuple[float, float]]:
    """[(start_sec, end_sec)] merged speech runs."""
    segs = []
    run_start = None
    for i, p in enumerate(probs.tolist()):
        if p >= threshold:
            if run_start is None:
                run_start = i
        else:
            if run_start is not None and i - run_start >= min_windows:
                segs.append((run_start * WINDOW / SR, i * WINDOW / SR))
            run_start = None
    if run_start is not None and len(probs) - run_start >= min_windows:
        segs.append((run_start * WINDOW / SR, len(probs) * WINDOW / SR))
    return segs


def speech_ratio(probs: torch.Tensor, threshold: float = 0.5) -> float:
    if probs.numel() == 0:
        return 0.0
    return float((probs >= threshold).float().mean())
