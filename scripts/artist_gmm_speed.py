"""Artist-index GMM fitting: batched-across-artists masked EM
(gmm_fit_many) vs per-artist sequential fits on the SAME
device. Synthetic catalogue shaped like a real library: ragged track
counts, 200-d embeddings, BIC over k=2..10."""

import sys
import time

import numpy as np
import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from audiomuse_amd.engines.artist_gmm import ArtistSimilarity  # noqa: E402


def synth_catalogue(n_artists: int, seed: int = 0):
    rng = np.random.default_rng(seed)
    per = {}
    for i in range(n_artists):
        n = int(rng.integers(5, 120))
        k = int(rng.integers(1, 4))
        centers = rng.standard_normal((k, 200)).astype(np.float32) * 2
        per[f"artist{i}"] = (centers[rng.integers(0, k, n)]
                             + rng.standard_normal((n, 200)).astype(
                                 np.float32) * 0.3)
    return per


def main(n_artists=2000):
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    per = synth_catalogue(n_artists)
    items = list(per.items())

    # batched (one masked EM per chunk x k)
    bat = ArtistSimilarity()
    t0 = time.perf_counter()
    bat._fit_catalogue_batched(items, seed=0, device=dev)
    if dev == "cuda":
        torch.cuda.synchronize()
    t_bat = time.perf_counter() - t0

    # sequential per-artist fits on the same device
    from audiomuse_amd.engines.artist_gmm import fit_artist
    seq = ArtistSimilarity()
    sub = items[: max(50, n_artists // 10)]   # sequential is slow: sample
    t0 = time.perf_counter()
    for name, e in sub:
        seq.models[name] = fit_artist(name, e, seed=0)
    if dev == "cuda":
        torch.cuda.synchronize()
    t_seq = (time.perf_counter() - t0) * (len(items) / len(sub))

    print(f"artist GMM fit ({n_artists} artists, 200d, BIC k=2..10) on {dev}:")
    print(f"  batched  : {t_bat:8.2f} s")
    print(f"  per-artist (extrapolated from {len(sub)}): {t_seq:8.2f} s")
    print(f"  speedup  : {t_seq / t_bat:6.1f}x")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 2000)
