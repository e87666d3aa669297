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


def seed(conn, n_tracks: int, n_artists: int, seed_: int = 0,
         chunk: int = 50_000) -> None:
    """Bulk seeding via executemany chunks (the row-at-a-time store
    helpers are the ingest path's shape; at 1M synthetic rows they are
    pure overhead)."""
    import json as _json

    rng = np.random.default_rng(seed_)
    artist_centers = rng.standard_normal((n_artists, 200)).astype(
        np.float32) * 2
    t0 = time.perf_counter()
    mood = _json.dumps({"rock": 0.5, "jazz": 0.3})
    other = _json.dumps({"happy": 0.5})
    axis = _json.dumps({"love": 0.4, "loss": 0.2})
    for c0 in range(0, n_tracks, chunk):
        c1 = min(c0 + chunk, n_tracks)
        idx = np.arange(c0, c1)
        embs = (artist_centers[idx % n_artists]
                + rng.standard_normal((len(idx), 200)).astype(
                    np.float32) * 0.4)
        claps = rng.standard_normal((len(idx), 512)).astype(np.float32)
        lyr = rng.standard_normal((len(idx), 768)).astype(np.float32)
        ids = [f"fp_4{'%050x' % i}" for i in idx]
        conn.executemany(
            """INSERT INTO score (item_id, title, author, album, tempo,
                   key, scale, mood_vector, other_features, energy,
                   duration) VALUES (?,?,?,?,?,?,?,?,?,?,?)""",
            [(ids[j], f"T{i}", f"Artist {i % n_artists}",
              f"Album {i // 8}", float(80 + i % 100), "C", "major",
              mood, other, 0.5, 200.0) for j, i in enumerate(idx)])
        conn.executemany(
            "INSERT INTO embedding (item_id, embedding) VALUES (?,?)",
            [(ids[j], embs[j].tobytes()) for j in range(len(idx))])
        conn.executemany(
            "INSERT INTO clap_embedding (item_id, embedding) VALUES (?,?)",
            [(ids[j], claps[j].tobytes()) for j in range(len(idx))])
        conn.executemany(
            """INSERT INTO lyrics_embedding (item_id, embedding,
                   axis_scores, lyrics_text, language, instrumental)
               VALUES (?,?,?,?,?,0)""",
            [(ids[j], lyr[j].tobytes(), axis, "la la", "en")
             for j in range(len(idx)) if idx[j] % 2 == 0])
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
