"""Property-based tests (hypothesis) over the pure transformation
layers: SQL munging, storage codecs, identity normalization, path
rewriting. These are the functions whose edge cases reach production
through USER data, so randomized inputs earn their keep."""

import math

import numpy as np
import pytest
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from audiomuse_amd.analysis.maintenance import normalize_title
from audiomuse_amd.analysis.migration import (detect_path_format,
                                              rewrite_path)
from audiomuse_amd.db.pgwire import qmark_to_dollar, sqlite_dialect_to_pg
from audiomuse_amd.index.ivf import decode_vectors, encode_vectors

# keep CI deterministic-ish and fast
FAST = settings(max_examples=200, deadline=None)


@FAST
@given(st.lists(st.sampled_from(["SELECT a", " WHERE x = ?", ", ?",
                                 " AND s = 'it''s ?'", " -- ?",
                                 " || '?' ||", " ORDER BY 1"]),
                min_size=1, max_size=12))
def test_qmark_translation_counts(parts):
    """$n placeholders appear exactly once per ? OUTSIDE string
    literals, numbered 1..n in order."""
    sql = "".join(parts)
    out = qmark_to_dollar(sql)
    outside = 0
    in_str = False
    i = 0
    while i < len(sql):
        c = sql[i]
        if in_str:
            if c == "'":
                if i + 1 < len(sql) and sql[i + 1] == "'":
                    i += 2
                    continue
                in_str = False
        elif c == "'":
            in_str = True
        elif c == "?":
            outside += 1
        i += 1
    for n in range(1, outside + 1):
        assert f"${n}" in out
    assert "?" not in _strip_literals(out)


def _strip_literals(sql: str) -> str:
    out, in_str = [], False
    i = 0
    while i < len(sql):
        c = sql[i]
        if in_str:
            if c == "'":
                if i + 1 < len(sql) and sql[i + 1] == "'":
                    i += 2
                    continue
                in_str = False
        elif c == "'":
            in_str = True
        else:
            out.append(c)
        i += 1
    return "".join(out)


@FAST
@given(st.text(alphabet=st.characters(blacklist_categories=("Cs",)),
               max_size=120))
def test_dialect_translation_idempotent(sql):
    once = sqlite_dialect_to_pg(sql)
    assert sqlite_dialect_to_pg(once) == once


@FAST
@given(st.integers(0, 2 ** 32 - 1), st.integers(1, 64), st.integers(2, 40))
def test_i8_codec_bounded_error_and_idempotent(seed, d, n):
    g = torch.Generator().manual_seed(seed)
    x = torch.nn.functional.normalize(torch.randn(n, d, generator=g), dim=1)
    enc = encode_vectors(x, "i8")
    dec = decode_vectors(enc, "i8")
    # |x| <= 1 per element after normalize; quantization step 1/127
    assert float((dec - x).abs().max()) <= 0.5 / 127.0 + 1e-6
    # re-encoding decoded values is exact (retrain() relies on this)
    assert torch.equal(encode_vectors(dec, "i8"), enc)


@FAST
@given(st.text(max_size=80))
def test_normalize_title_idempotent_and_casefree(t):
    once = normalize_title(t)
    assert normalize_title(once) == once
    assert normalize_title(t.upper()) == normalize_title(t.lower())


@FAST
@given(st.lists(st.tuples(st.sampled_from(["a", "b", "c", "disc 1"]),
                          st.sampled_from(["x.flac", "y.mp3", "z.wav"])),
                min_size=2, max_size=10),
       st.sampled_from(["/music", "/srv/media/library", "C:\\Music"]))
def test_path_format_detection_recovers_prefix(parts, prefix):
    sep = "\\" if prefix.startswith("C:") else "/"
    paths = [f"{prefix}{sep}{a}{sep}{f}" for a, f in parts]
    fmt = detect_path_format(paths)
    assert fmt["separator"] == sep
    assert fmt["prefix"].startswith(prefix.split(sep)[0] or prefix)


@FAST
@given(st.sampled_from(["/music", "/library/Audio"]),
       st.sampled_from(["/mnt/nas", "/data"]),
       st.lists(st.sampled_from(["Artist", "Album 1", "04 - Song.flac"]),
                min_size=1, max_size=4))
def test_rewrite_path_round_trip(src, dst, tail):
    rule = {"from_prefix": src, "to_prefix": dst,
            "from_separator": "/", "to_separator": "/"}
    back = {"from_prefix": dst, "to_prefix": src,
            "from_separator": "/", "to_separator": "/"}
    p = src + "/" + "/".join(tail)
    assert rewrite_path(rewrite_path(p, rule), back) == p


@FAST
@given(st.integers(0, 2 ** 31 - 1), st.integers(1, 200))
def test_ivf_query_invariants_random_shapes(seed, n):
    """For ANY corpus shape: distances ascend, ids are valid or the -1
    pad, self-query at full probe finds itself first (angular)."""
    from audiomuse_amd.index.ivf import IVFIndex

    g = torch.Generator().manual_seed(seed)
    d = 8 + (seed % 5) * 8
    x = torch.randn(n, d, generator=g)
    idx = IVFIndex.build(x, metric="angular", storage="f32", seed=1)
    k = min(5, n)
    dist, ids = idx.query(x[:1], k=k, nprobe=idx.nlist)
    row_d, row_i = dist.reshape(-1), ids.reshape(-1)
    valid = row_i >= 0
    assert int(row_i[0]) == 0                       # self first
    assert float(row_d[0]) < 1e-4
    vd = row_d[valid]
    assert bool((vd[:-1] <= vd[1:] + 1e-6).all())   # ascending
    assert int(row_i[valid].max()) < n


@FAST
@given(st.integers(0, 2 ** 31 - 1), st.integers(40, 90),
       st.integers(0, 12))
def test_chromaprint_bit_match_shift_invariance(seed, n_frames, shift):
    """A fingerprint matches a time-shifted copy of itself (alignment
    search; overlap stays above CHROMAPRINT_MIN_OVERLAP words) and the
    ratio stays in [0, 1]."""
    import zlib

    from audiomuse_amd.engines.chromaprint import bit_match_ratio

    rng = np.random.default_rng(seed)
    raw = rng.integers(0, 2 ** 32 - 1, size=n_frames, dtype=np.uint32)
    a = zlib.compress(raw.tobytes())          # the compute() wire format
    b = zlib.compress(np.roll(raw, shift).tobytes())
    r = bit_match_ratio(a, b, max_offset=16)
    assert 0.0 <= r <= 1.0
    if n_frames - shift >= 20:
        assert r > 0.95, (shift, r)
    assert bit_match_ratio(a, a) == pytest.approx(1.0)
