"""Smaller feature engines: SemGrove fusion, sonic fingerprint,
playlist ordering.

References:
- SemGrove merge: /root/reference/tasks/sem_grove_manager.py:65-108 —
  per-modality L2 norm + whitening + sqrt(weight)-scaled concat
  (75/25 lyrics/audio), IVF over the merged space.
- Sonic fingerprint: /root/reference/tasks/sonic_fingerprint_manager.py:129
  — top-played tracks, 30-day-half-life exponential recency weights,
  weighted-mean vector, IVF expansion.
- Playlist ordering: /root/reference/tasks/playlist_ordering.py — greedy
  nearest-neighbor walk over (tempo, energy, key) distance with an
  optional energy arc.
"""

from __future__ import annotations

import math
import time
from typing import Dict, List, Optional, Sequence

import numpy as np

from audiomuse_amd import config as C


# -- SemGrove ---------------------------------------------------------------

class SemGroveMerger:
    """Whitening + weighted concat of lyrics (768) and audio (512/200)."""

    def __init__(self, lyrics_weight: Optional[float] = None,
                 audio_weight: Optional[float] = None):
        self.lw = lyrics_weight if lyrics_weight is not None else C.SEM_GROVE_LYRICS_WEIGHT
        self.aw = audio_weight if audio_weight is not None else C.SEM_GROVE_AUDIO_WEIGHT
        self.lyr_mean: Optional[np.ndarray] = None
        self.lyr_std: Optional[np.ndarray] = None
        self.aud_mean: Optional[np.ndarray] = None
        self.aud_std: Optional[np.ndarray] = None

    @staticmethod
    def _l2(x: np.ndarray) -> np.ndarray:
        return x / (np.linalg.norm(x, axis=-1, keepdims=True) + 1e-12)

    def fit(self, lyrics: np.ndarray, audio: np.ndarray) -> None:
        lyr = self._l2(np.asarray(lyrics, dtype=np.float32))
        aud = self._l2(np.asarray(audio, dtype=np.float32))
        self.lyr_mean, self.lyr_std = lyr.mean(0), lyr.std(0) + 1e-6
        self.aud_mean, self.aud_std = aud.mean(0), aud.std(0) + 1e-6

    def merge(self, lyrics: np.ndarray, audio: np.ndarray) -> np.ndarray:
        """(n, 768+audio_d) merged vectors (sem_grove _make_merged_vector)."""
        assert self.lyr_mean is not None, "fit() first"
        lyr = (self._l2(np.asarray(lyrics, dtype=np.float32)) - self.lyr_mean) / self.lyr_std
        aud = (self._l2(np.asarray(audio, dtype=np.float32)) - self.aud_mean) / self.aud_std
        lyr = self._l2(lyr) * math.sqrt(self.lw)
        aud = self._l2(aud) * math.sqrt(self.aw)
        return np.concatenate([lyr, aud], axis=-1).astype(np.float32)


# -- Sonic fingerprint ------------------------------------------------------

def recency_weights(played_at: Sequence[float],
                    now: Optional[float] = None,
                    half_life_days: Optional[float] = None) -> np.ndarray:
    """w = 0.5 ** (age_days / half_life) (sonic_fingerprint_manager)."""
    now = now if now is not None else time.time()
    hl = half_life_days if half_life_days is not None else C.SONIC_FINGERPRINT_HALF_LIFE_DAYS
    ages = np.array([(now - t) / 86400.0 for t in played_at], dtype=np.float64)
    return np.power(0.5, np.clip(ages, 0.0, None) / hl).astype(np.float32)


def sonic_fingerprint(vectors: np.ndarray, played_at: Sequence[float],
                      now: Optional[float] = None) -> Optional[np.ndarray]:
    """Weighted-mean taste vector, L2-normalized
    (generate_sonic_fingerprint :129)."""
    vectors = np.asarray(vectors, dtype=np.float32)
    if vectors.size == 0:
        return None
    w = recency_weights(played_at, now=now)
    if w.sum() <= 0:
        # all plays far older than the half-life window: uniform fallback
        w = np.ones(vectors.shape[0], dtype=np.float32)
    v = (vectors * w[:, None]).sum(axis=0) / w.sum()
    norm = float(np.linalg.norm(v))
    return v / norm if norm > 0 else None


# -- Playlist ordering ------------------------------------------------------

_KEY_POS = {k: i for i, k in enumerate(
    ["C", "C#", "D", "D#", "E", "F", "F#", "G", "G#", "A", "A#", "B"])}


def _track_feature_distance(a: Dict, b: Dict) -> float:
    """tempo/energy/key distance (playlist_ordering.order_playlist)."""
    dt = abs(float(a.get("tempo", 0)) - float(b.get("tempo", 0))) / 60.0
    de = abs(float(a.get("energy", 0)) - float(b.get("energy", 0)))
    ka = _KEY_POS.get(a.get("key", "C"), 0)
    kb = _KEY_POS.get(b.get("key", "C"), 0)
    dk = min(abs(ka - kb), 12 - abs(ka - kb)) / 6.0
    ds = 0.0 if a.get("scale") == b.get("scale") else 0.5
    return dt + de + dk + ds


def order_playlist(tracks: List[Dict], energy_arc: bool = False) -> List[Dict]:
    """Greedy NN walk from the lowest-energy track; optional rising-then-
    falling energy arc reorder."""
    if len(tracks) <= 2:
        return list(tracks)
    remaining = list(tracks)
    remaining.sort(key=lambda t: float(t.get("energy", 0)))
    out = [remaining.pop(0)]
    while remaining:
        last = out[-1]
        best = min(range(len(remaining)),
                   key=lambda i: _track_feature_distance(last, remaining[i]))
        out.append(remaining.pop(best))
    if energy_arc:
        by_energy = sorted(out, key=lambda t: float(t.get("energy", 0)))
        rising = by_energy[0::2]
        falling = by_energy[1::2][::-1]
        out = rising + falling
    return out
