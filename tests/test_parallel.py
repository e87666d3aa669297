"""Multi-process distributed tests on gloo (world_size 2) — the CPU
stand-in for the 8-GPU RCCL paths (SURVEY.md §4 carry-over: collective
tests at world sizes >1 on one node)."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from audiomuse_amd.engines.projection import umap_project
from audiomuse_amd.parallel.sharded import shard_bounds


def test_shard_bounds_cover_all():
    for n in (10, 11, 17):
        seen = []
        for r in range(4):
            a, b = shard_bounds(n, 4, r)
            seen.extend(range(a, b))
        assert seen == list(range(n))


def test_umap_project_preserves_clusters():
    g = torch.Generator().manual_seed(0)
    centers = torch.randn(3, 16, generator=g) * 6
    assign = torch.arange(120) % 3
    x = centers[assign] + torch.randn(120, 16, generator=g) * 0.3
    emb = umap_project(x, epochs=80, seed=0)
    assert emb.shape == (120, 2)
    # intra-cluster 2-D distances < inter-cluster
    intra, inter = [], []
    for i in range(0, 120, 7):
        for j in range(1, 120, 11):
            d = float((emb[i] - emb[j]).norm())
            (intra if assign[i] == assign[j] else inter).append(d)
    assert np.mean(intra) < 0.5 * np.mean(inter)


def _dist_worker(rank, world, port, fn_name, tmpdir, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
    })
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        result = globals()[fn_name](rank, world, tmpdir)
        q.put((rank, "ok", result))
    except Exception as exc:  # noqa: BLE001
        import traceback
        q.put((rank, "err", traceback.format_exc()))
    finally:
        dist.destroy_process_group()


def _run_dist(fn_name, tmpdir="", world=2):
    import random
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = random.randint(20000, 40000)
    procs = [ctx.Process(target=_dist_worker,
                         args=(r, world, port, fn_name, tmpdir, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, status, payload = q.get(timeout=180)
        assert status == "ok", payload
        results[rank] = payload
    for p in procs:
        p.join(timeout=30)
    return results


# -- distributed bodies (run inside workers) --------------------------------

def _body_sharded_query(rank, world, tmpdir):
    from audiomuse_amd.index.ivf import IVFIndex
    from audiomuse_amd.parallel.sharded import (build_sharded_index,
                                                shard_bounds,
                                                sharded_topk_query)

    torch.manual_seed(0)
    x = torch.randn(800, 32)                      # same full set on all ranks
    a, b = shard_bounds(800, world, rank)
    idx = build_sharded_index(x[a:b], torch.arange(a, b), nlist=8, seed=0)
    q = x[:4] + 0.001
    d, ids = sharded_topk_query(idx, q, k=5, nprobe=8)
    # global top-1 must be the query row itself
    return {"ids0": ids[:, 0].tolist(), "d0": d[:, 0].tolist()}


def _body_dist_kmeans(rank, world, tmpdir):
    from audiomuse_amd.ops.kmeans import minibatch_kmeans

    torch.manual_seed(0)
    full = torch.randn(1000, 8)
    a, b = (0, 500) if rank == 0 else (500, 1000)
    cents = minibatch_kmeans(full[a:b], 10, iters=8, seed=1,
                             group=dist.group.WORLD)
    return {"sum": float(cents.sum())}


def _body_ddp_trainer(rank, world, tmpdir):
    from audiomuse_amd.models.htsat import HTSATConfig
    from audiomuse_amd.parallel.trainer import DistillConfig, DistillTrainer

    tiny = HTSATConfig(n_mels=32, n_frames=64, patch_size=4, embed_dim=32,
                       depths=(1, 1), num_heads=(2, 4), window=4, out_dim=16)
    t = DistillTrainer(DistillConfig(batch=2, student_cfg=tiny,
                                     teacher_cfg=tiny), device="cpu")
    losses = [t.step(i) for i in range(2)]
    # params must be identical across ranks after synced steps
    p = next(iter(t.student.parameters())).detach()
    return {"losses": losses, "psum": float(p.sum())}


# -- tests -------------------------------------------------------------------

@pytest.mark.slow
def test_sharded_query_world2():
    res = _run_dist("_body_sharded_query")
    # both ranks produced the identical global answer
    assert res[0] == res[1]
    assert res[0]["ids0"] == [0, 1, 2, 3]


@pytest.mark.slow
def test_distributed_kmeans_replicated():
    res = _run_dist("_body_dist_kmeans")
    assert abs(res[0]["sum"] - res[1]["sum"]) < 1e-4


@pytest.mark.slow
def test_ddp_distillation_step_syncs():
    res = _run_dist("_body_ddp_trainer")
    assert res[0]["psum"] == pytest.approx(res[1]["psum"], abs=1e-6)
    assert all(np.isfinite(v) for v in res[0]["losses"])
