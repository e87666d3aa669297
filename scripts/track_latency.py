"""Interactive single-track analysis latency: one 12 s track through
decode -> resample -> features -> MusiCNN -> CLAP -> identity-ready
embeddings, timed end-to-end (the latency a user sees when analyzing
one new upload; throughput soaks measure the batch regime instead)."""

import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from audiomuse_amd.analysis.pipeline import AnalysisRuntime  # noqa: E402
from audiomuse_amd.mediaserver import make_provider  # noqa: E402


def main(n=30):
    prov = make_provider("synthetic", n_albums=8, tracks_per_album=4,
                         seconds=12.0, sr=44100)
    tracks = prov.get_all_songs()
    blobs = [prov.download_track(t.provider_id) for t in tracks[: n + 2]]
    rt = AnalysisRuntime(device="cuda")
    # warmup (model load, algo search, graph-free steady state)
    rt.analyze_album_batch(blobs[:2])
    torch.cuda.synchronize()
    lats = []
    for b in blobs[2:]:
        t0 = time.perf_counter()
        res = rt.analyze_album_batch([b])[0]
        torch.cuda.synchronize()
        lats.append((time.perf_counter() - t0) * 1000)
        assert res is not None and res.embedding is not None
    lats.sort()
    print(f"single-track analysis latency over {len(lats)} tracks "
          f"(12 s audio, full pipeline):")
    print(f"  p50 {lats[len(lats)//2]:.1f} ms  p90 {lats[int(len(lats)*.9)]:.1f} ms"
          f"  max {lats[-1]:.1f} ms")


if __name__ == "__main__":
    main()
