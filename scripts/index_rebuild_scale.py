"""Full nine-build index rebuild at library scale: seed a synthetic
catalogue (embeddings + CLAP + lyrics + axis scores + metadata) straight
into the DB, then time `run_all_index_builds` — the operation the
reference schedules nightly and after every analysis run
(index.py:47 _run_all_index_builds). Exercises the batched artist-GMM
path, all IVF builds, the SemGrove merge, both map projections and the
hyperbolic tree persist."""

import sys
import tempfile
import time

import numpy as np
import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from audiomuse_amd.analysis.index import run_all_index_builds  # noqa: E402
from audiomuse_amd.db import connect  # noqa: E402
from audiomuse_amd.db.schema import init_db  # noqa: E402
from audiomuse_amd.db.store import (save_clap_embedding,  # noqa: E402
                                    save_lyrics_embedding,
                                    save_track_analysis_and_embedding)


def seed(conn, n_tracks: int, n_artists: int, seed_: int = 0) -> None:
    rng = np.random.default_rng(seed_)
    artist_centers = rng.standard_normal((n_artists, 200)).astype(
        np.float32) * 2
    t0 = time.perf_counter()
    for i in range(n_tracks):
        a = i % n_artists
        iid = f"fp_4{'%050x' % i}"
        emb = (artist_centers[a]
               + rng.standard_normal(200).astype(np.float32) * 0.4)
        save_track_analysis_and_embedding(
            conn, iid, title=f"T{i}", author=f"Artist {a}",
            album=f"Album {i // 8}", tempo=float(80 + i % 100),
            energy=float((i % 10) / 10), key="C", scale="major",
            duration=200.0,
            mood_vector={"rock": float(i % 3) / 2, "jazz": 0.3},
            other_features={"happy": 0.5}, embedding=emb)
        save_clap_embedding(
            conn, iid, rng.standard_normal(512).astype(np.float32))
        if i % 2 == 0:
            save_lyrics_embedding(
                conn, iid, rng.standard_normal(768).astype(np.float32),
                axis_scores={"love": 0.4, "loss": 0.2})
    conn.commit()
    print(f"seeded {n_tracks} tracks / {n_artists} artists "
          f"in {time.perf_counter() - t0:.1f}s")


def main(n_tracks=100_000, n_artists=4000):
    td = tempfile.mkdtemp()
    conn = connect(f"sqlite:///{td}/scale.db")
    init_db(conn)
    seed(conn, n_tracks, n_artists)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    t0 = time.perf_counter()
    out = run_all_index_builds(conn, device=dev)
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"run_all_index_builds ({n_tracks} tracks, {n_artists} artists, "
          f"{dev}): {dt:.1f} s total")
    for name, status in sorted(out.items()):
        print(f"  {name:24s} {status}")


if __name__ == "__main__":
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 100_000
    a = int(sys.argv[2]) if len(sys.argv) > 2 else 4000
    main(n, a)
