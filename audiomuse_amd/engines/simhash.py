"""Catalogue identity: 200-bit embedding simhash -> canonical ids.

Reference behavior: /root/reference/tasks/simhash.py —
- signature: per-dimension above-row-mean bit, packed big-endian into a
  200-bit integer (signature_batch, simhash.py:127-148)
- canonical id: "fp_" + scheme digit + 50 hex chars of the signature;
  exact-string collisions step to the next free value
  (mint_canonical_id, simhash.py:345)
- candidate lookup: the 200 bits split into max_hamming+1 disjoint
  bands; <= max_hamming flipped bits leave one band intact (pigeonhole),
  so only band-sharing signatures get the XOR+popcount check
  (SignatureIndex, simhash.py:476+)
- confirmation: cosine distance below threshold AND duration agreement
  (confirm gate, simhash.py:238)

The signature math runs in numpy here (identity minting is CPU-side in
the analysis pipeline); the batched GPU embedding path feeds f32 arrays
straight in.
"""

from __future__ import annotations

import hashlib
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

from audiomuse_amd import config as C

# SIMHASH_BITS / SIMHASH_BANDS / CATALOGUE_ID_SCHEME_VERSION come from
# config so the id scheme is a deployment choice, as in the reference
# (config.py CATALOGUE_ID_SCHEME_VERSION; 200-bit scheme 4 default —
# changing them is an id-scheme migration, not a tuning knob)
SIGNATURE_BITS = int(C.SIMHASH_BITS)
SIGNATURE_BYTES = SIGNATURE_BITS // 8          # 25 at the default 200
SIGNATURE_MATCH_MAX_HAMMING = max(int(C.SIMHASH_BANDS) - 1, 1)
_BAND_COUNT = SIGNATURE_MATCH_MAX_HAMMING + 1
_ID_PREFIX = "fp_"
_ID_SCHEME = str(C.CATALOGUE_ID_SCHEME_VERSION)
_HEX_LEN = SIGNATURE_BYTES * 2                 # 50
CANONICAL_ID_LEN = len(_ID_PREFIX) + 1 + _HEX_LEN
_SIGNATURE_MASK = (1 << SIGNATURE_BITS) - 1


def signature_batch(embeddings: Sequence[Optional[np.ndarray]]
                    ) -> List[Optional[int]]:
    """200-bit signatures; None for missing/wrong-dim/non-finite/constant
    rows (simhash.py:127-148)."""
    out: List[Optional[int]] = [None] * len(embeddings)
    rows = []
    pos = []
    for i, e in enumerate(embeddings):
        if e is None:
            continue
        row = np.asarray(e, dtype=np.float32).ravel()
        if row.size != SIGNATURE_BITS or not np.isfinite(row).all():
            continue
        if np.ptp(row) <= 0:
            continue
        rows.append(row)
        pos.append(i)
    if not rows:
        return out
    matrix = np.stack(rows).astype(np.float64)
    matrix -= matrix.mean(axis=1, keepdims=True)
    bits = (matrix > 0).astype(np.uint8)
    packed = np.packbits(bits, axis=1)
    for p, row_bytes in zip(pos, packed):
        out[p] = int.from_bytes(row_bytes.tobytes(), "big")
    return out


def embedding_signature(embedding) -> Optional[int]:
    return signature_batch([embedding])[0]


def canonical_id_str(value: int) -> str:
    return f"{_ID_PREFIX}{_ID_SCHEME}{value & _SIGNATURE_MASK:0{_HEX_LEN}x}"


def mint_canonical_id(signature: int, taken) -> str:
    """Step past exact-string collisions (simhash.py:345)."""
    value = signature & _SIGNATURE_MASK
    item_id = canonical_id_str(value)
    while item_id in taken:
        value = (value + 1) & _SIGNATURE_MASK
        item_id = canonical_id_str(value)
    return item_id


def unsignable_id(server_id: str, provider_track_id: str) -> str:
    """fp_0 fallback id for tracks without a usable embedding."""
    digest = hashlib.sha256(
        f"{server_id}\x00{provider_track_id}".encode()).hexdigest()
    return f"{_ID_PREFIX}0{digest[:_HEX_LEN]}"


def is_signature_id(item_id) -> bool:
    return (isinstance(item_id, str) and len(item_id) == CANONICAL_ID_LEN
            and item_id.startswith(_ID_PREFIX)
            and "1" <= item_id[len(_ID_PREFIX)] <= "9")


def signature_from_id(item_id: str) -> Optional[int]:
    if not is_signature_id(item_id):
        return None
    return int(item_id[len(_ID_PREFIX) + 1:], 16)


def cosine_distance(a, b) -> float:
    """Clipped-to-[0,2] cosine distance (simhash.py confirm gate)."""
    a = np.asarray(a, dtype=np.float64).ravel()
    b = np.asarray(b, dtype=np.float64).ravel()
    if a.size != b.size or a.size == 0:
        return 1.0
    denom = float(np.linalg.norm(a) * np.linalg.norm(b))
    if denom <= 0:
        return 1.0
    return float(np.clip(1.0 - float(np.dot(a, b)) / denom, 0.0, 2.0))


def _band_ranges() -> List[Tuple[int, int]]:
    """_BAND_COUNT disjoint bit ranges covering [0, 200)."""
    base = SIGNATURE_BITS // _BAND_COUNT
    extra = SIGNATURE_BITS % _BAND_COUNT
    ranges = []
    low = 0
    for b in range(_BAND_COUNT):
        width = base + (1 if b < extra else 0)
        ranges.append((low, low + width))
        low += width
    return ranges


_BAND_BITS = _band_ranges()


def _band_key(signature: int, band: int) -> int:
    low, high = _BAND_BITS[band]
    shift = SIGNATURE_BITS - high
    return (signature >> shift) & ((1 << (high - low)) - 1)


def _sig_to_packed(signature: int) -> np.ndarray:
    return np.frombuffer(signature.to_bytes(SIGNATURE_BYTES, "big"),
                         dtype=np.uint8)


class SignatureIndex:
    """Banded Hamming-tolerant lookup (simhash.py SignatureIndex)."""

    def __init__(self, max_hamming: int = SIGNATURE_MATCH_MAX_HAMMING):
        self.max_hamming = min(int(max_hamming), _BAND_COUNT - 1)
        self._bands: List[Dict[int, List[int]]] = [dict() for _ in range(_BAND_COUNT)]
        self._ids: List[str] = []
        self._sigs: List[int] = []
        self._packed: List[np.ndarray] = []
        self._durations: List[float] = []

    def __len__(self) -> int:
        return len(self._ids)

    def add(self, item_id: str, signature: int, duration: float = 0.0) -> None:
        row = len(self._ids)
        self._ids.append(item_id)
        self._sigs.append(signature)
        self._packed.append(_sig_to_packed(signature))
        self._durations.append(float(duration))
        for band in range(_BAND_COUNT):
            self._bands[band].setdefault(_band_key(signature, band), []).append(row)

    def candidates(self, signature: int) -> List[int]:
        rows = set()
        for band in range(_BAND_COUNT):
            rows.update(self._bands[band].get(_band_key(signature, band), ()))
        return sorted(rows)

    def lookup(self, signature: int, duration: Optional[float] = None
               ) -> List[Tuple[str, int]]:
        """[(item_id, hamming)] within max_hamming, optional duration gate."""
        rows = self.candidates(signature)
        if not rows:
            return []
        probe = _sig_to_packed(signature)
        packed = np.stack([self._packed[r] for r in rows])
        ham = np.unpackbits(packed ^ probe, axis=1).sum(axis=1)
        out = []
        for r, h in zip(rows, ham):
            if h > self.max_hamming:
                continue
            if duration is not None and self._durations[r] > 0:
                if abs(self._durations[r] - duration) > C.SIMHASH_CONFIRM_DURATION_SECONDS:
                    continue
            out.append((self._ids[r], int(h)))
        out.sort(key=lambda t: t[1])
        return out


class CatalogResolver:
    """Mint-or-match resolver (simhash.py CatalogResolver:567): an incoming
    embedding either matches an existing recording (banded lookup + cosine
    + duration confirm) or mints a fresh canonical id."""

    def __init__(self, index: Optional[SignatureIndex] = None):
        self.index = index or SignatureIndex()
        self.taken: set = set()
        self.vectors: Dict[str, np.ndarray] = {}

    def register_existing(self, item_id: str, embedding: np.ndarray,
                          duration: float = 0.0) -> None:
        sig = embedding_signature(embedding)
        self.taken.add(item_id)
        if sig is not None:
            self.index.add(item_id, sig, duration)
            self.vectors[item_id] = np.asarray(embedding, dtype=np.float32)

    def resolve(self, embedding: Optional[np.ndarray], duration: float,
                server_id: str, provider_track_id: str,
                confirm_fn=None) -> Tuple[str, bool]:
        """Returns (canonical_id, matched_existing).

        confirm_fn(candidate_id) -> bool is the optional LAST confirm
        gate (reference: the chromaprint bit-match gate,
        CHROMAPRINT_GATE_ENABLED) — a candidate that passes simhash +
        cosine + duration but fails it is treated as a different
        recording."""
        sig = embedding_signature(embedding)
        if sig is None:
            return unsignable_id(server_id, provider_track_id), False
        for cand_id, _ham in self.index.lookup(sig, duration):
            vec = self.vectors.get(cand_id)
            if vec is None:
                continue
            if cosine_distance(embedding, vec) < C.SIMHASH_CONFIRM_COSINE:
                if confirm_fn is not None and not confirm_fn(cand_id):
                    continue       # acoustic fingerprint disagrees
                return cand_id, True
        item_id = mint_canonical_id(sig, self.taken)
        self.register_existing(item_id, embedding, duration)
        return item_id, False
