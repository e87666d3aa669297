"""HTTP serving load: concurrent clients against the real Flask app
(threaded werkzeug, the deployment default) on a seeded catalogue —
/api/similar_tracks + /api/search_tracks + /api/track mixed traffic.
Reports throughput and latency percentiles end to end (HTTP + engine +
GPU scan)."""

import json
import random
import sys
import tempfile
import threading
import time
import urllib.request

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from audiomuse_amd.analysis.index import run_all_index_builds  # noqa: E402
from audiomuse_amd.db import connect  # noqa: E402
from audiomuse_amd.db.schema import init_db  # noqa: E402
from audiomuse_amd.db.store import (save_clap_embedding,  # noqa: E402
                                    save_track_analysis_and_embedding)


def seed(conn, n=20000):
    rng = np.random.default_rng(0)
    ids = []
    for i in range(n):
        iid = f"fp_4{'%050x' % i}"
        ids.append(iid)
        save_track_analysis_and_embedding(
            conn, iid, title=f"Song {i}", author=f"Artist {i % 500}",
            album=f"Album {i % 2000}", tempo=100.0, energy=0.5, key="C",
            scale="major", duration=200.0, mood_vector={"rock": 0.5},
            other_features={},
            embedding=rng.standard_normal(200).astype(np.float32))
        save_clap_embedding(conn, iid,
                            rng.standard_normal(512).astype(np.float32))
    conn.commit()
    return ids


def main(n_clients=16, seconds=20.0, n_tracks=20000, server_procs=1):
    import torch

    from audiomuse_amd.web.app import create_app

    td = tempfile.mkdtemp()
    url = f"sqlite:///{td}/load.db"
    conn = connect(url)
    init_db(conn)
    ids = seed(conn, n_tracks)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    run_all_index_builds(conn, device=dev)
    import logging
    logging.getLogger("werkzeug").setLevel(logging.ERROR)
    import subprocess as _sp

    server_children = []
    if server_procs > 1:
        # N SO_REUSEPORT processes share the port (web/serve.py)
        import socket as s_mod
        probe = s_mod.socket()
        probe.bind(("127.0.0.1", 0))
        port = probe.getsockname()[1]
        probe.close()
        cmd = [sys.executable, "-m", "audiomuse_amd", "web",
               "--host", "127.0.0.1", "--port", str(port), "--reuse-port",
               "--no-auth", "--db", url]
        server_children = [_sp.Popen(cmd) for _ in range(server_procs)]
        srv = None
    else:
        from werkzeug.serving import make_server

        app = create_app(url, auth_disabled=True)
        srv = make_server("127.0.0.1", 0, app, threaded=True)
        port = srv.server_port
        threading.Thread(target=srv.serve_forever, daemon=True).start()

    base = f"http://127.0.0.1:{port}"
    deadline = time.perf_counter() + 120
    while True:
        try:
            urllib.request.urlopen(
                f"{base}/api/similar_tracks?item_id={ids[0]}&n=10",
                timeout=5)
            break
        except Exception:
            if time.perf_counter() > deadline:
                raise
            time.sleep(0.5)

    # clients run as SEPARATE PROCESSES: an in-process thread pool
    # shares the GIL with the (threaded-werkzeug) server and measures
    # its own contention instead of the server
    import subprocess

    client_src = r"""
import json, random, sys, time, urllib.request, urllib.error
base, seed_n, n_ids, seconds = (sys.argv[1], int(sys.argv[2]),
                                int(sys.argv[3]), float(sys.argv[4]))
rng = random.Random(seed_n)
lats, errs = [], {}
stop_at = time.perf_counter() + seconds
while time.perf_counter() < stop_at:
    i = rng.randrange(n_ids)
    iid = "fp_4" + ("%050x" % i)
    r = rng.random()
    if r < 0.6:
        u = base + "/api/similar_tracks?item_id=" + iid + "&n=10"
    elif r < 0.8:
        u = base + "/api/search_tracks?q=song%20" + str(rng.randrange(999))
    else:
        u = base + "/api/track?item_id=" + iid
    t0 = time.perf_counter()
    try:
        with urllib.request.urlopen(u, timeout=20) as resp:
            resp.read()
    except urllib.error.HTTPError as e:
        errs[str(e.code)] = errs.get(str(e.code), 0) + 1
        continue
    except Exception as e:
        errs[type(e).__name__] = errs.get(type(e).__name__, 0) + 1
        continue
    lats.append((time.perf_counter() - t0) * 1000)
print(json.dumps({"lats": lats, "errs": errs}))
"""
    procs = [subprocess.Popen(
        [sys.executable, "-c", client_src, base, str(i), str(len(ids)),
         str(seconds)], stdout=subprocess.PIPE)
        for i in range(n_clients)]
    t0 = time.perf_counter()
    lats, errors = [], {}
    for p in procs:
        out, _ = p.communicate(timeout=seconds + 60)
        d = json.loads(out)
        lats.extend(d["lats"])
        for k, v in d["errs"].items():
            errors[k] = errors.get(k, 0) + v
    wall = time.perf_counter() - t0
    if srv is not None:
        srv.shutdown()
    for c in server_children:
        c.terminate()
    lats.sort()
    q = lambda p: lats[min(int(len(lats) * p), len(lats) - 1)]  # noqa: E731
    print(json.dumps({
        "clients": n_clients, "seconds": round(wall, 1),
        "requests": len(lats), "errors": errors,
        "qps": round(len(lats) / wall, 1),
        "p50_ms": round(q(0.5), 1), "p90_ms": round(q(0.9), 1),
        "p99_ms": round(q(0.99), 1),
        "server_procs": server_procs,
        "catalogue": n_tracks, "mix": "60% similar / 20% search / 20% track"}))


if __name__ == "__main__":
    main(n_clients=int(sys.argv[1]) if len(sys.argv) > 1 else 16,
         server_procs=int(sys.argv[2]) if len(sys.argv) > 2 else 1)
