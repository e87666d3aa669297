"""IVF index engine tests (reference parity: paged_ivf.py / ivf_quant.py)."""

import numpy as np
import pytest
import torch

from audiomuse_amd.index.ivf import (IVFIndex, default_nlist, effective_storage,
                                     encode_vectors)


def _corpus(n=2000, d=64, seed=0):
    g = torch.Generator().manual_seed(seed)
    # clustered data so IVF recall is meaningful
    centers = torch.randn(20, d, generator=g) * 3
    assign = torch.randint(0, 20, (n,), generator=g)
    x = centers[assign] + torch.randn(n, d, generator=g)
    return x


def _brute_force_ids(x, q, k, metric):
    if metric == "angular":
        xn = x / x.norm(dim=1, keepdim=True).clamp(min=1e-12)
        qn = q / q.norm(dim=1, keepdim=True).clamp(min=1e-12)
        d = 1 - qn @ xn.T
    elif metric == "euclidean":
        d = torch.cdist(q, x)
    else:
        d = -(q @ x.T)
    return torch.topk(d, k, dim=1, largest=False).indices


def test_effective_storage_downgrade():
    # ivf_quant.effective_code: i8 is angular-only
    assert effective_storage("i8", "angular") == "i8"
    assert effective_storage("i8", "euclidean") == "f16"
    assert effective_storage("i8", "dot") == "f16"
    assert effective_storage("f32", "euclidean") == "f32"


def test_encode_i8_matches_reference_codec():
    v = torch.tensor([[0.0, 0.5, -0.5, 1.2, -1.2, 0.004]])
    enc = encode_vectors(v, "i8")
    expect = np.clip(np.rint(v.numpy() * 127.0), -127, 127).astype(np.int8)
    np.testing.assert_array_equal(enc.numpy(), expect)


def test_default_nlist_formula():
    assert default_nlist(180_000) == min(int(8 * np.sqrt(180_000)), 8192)
    assert default_nlist(10_000_000) == 8192
    assert default_nlist(1) == 8  # reference formula has no N floor


@pytest.mark.parametrize("metric,storage", [
    ("angular", "i8"), ("angular", "f16"), ("angular", "f32"),
    ("euclidean", "f16"), ("dot", "f32"),
])
def test_build_query_recall(metric, storage):
    x = _corpus()
    idx = IVFIndex.build(x, metric=metric, storage=storage, nlist=32, seed=0)
    q = x[:8] + torch.randn(8, x.shape[1]) * 0.01
    dist, ids = idx.query(q, k=10, nprobe=32)  # probe everything -> exhaustive
    assert dist.shape == (8, 10) and ids.shape == (8, 10)
    # with all cells probed + f32 re-rank, results == brute force
    if metric == "angular":
        xq = x / x.norm(dim=1, keepdim=True).clamp(min=1e-12)
    else:
        xq = x
    expect = _brute_force_ids(xq, q, 10, metric)
    got_sets = [set(r.tolist()) for r in ids]
    exp_sets = [set(r.tolist()) for r in expect]
    overlap = np.mean([len(g & e) / 10 for g, e in zip(got_sets, exp_sets)])
    assert overlap >= 0.9, f"recall {overlap} too low for {metric}/{storage}"


def test_partial_probe_recall_reasonable():
    x = _corpus(4000, 64)
    idx = IVFIndex.build(x, metric="angular", storage="i8", nlist=64, seed=1)
    q = x[100:110]
    _, ids = idx.query(q, k=5, nprobe=16)
    expect = _brute_force_ids(
        x / x.norm(dim=1, keepdim=True).clamp(min=1e-12), q, 5, "angular")
    overlap = np.mean([len(set(a.tolist()) & set(b.tolist())) / 5
                       for a, b in zip(ids, expect)])
    assert overlap >= 0.6


def test_self_query_returns_self_first():
    x = _corpus(500, 32)
    idx = IVFIndex.build(x, metric="angular", nlist=16, seed=2)
    _, ids = idx.query(x[42], k=3, nprobe=16)
    assert int(ids[0]) == 42


def test_custom_ids_and_vector_lookup():
    x = _corpus(300, 16)
    ids = torch.arange(300, dtype=torch.int64) * 7 + 3
    idx = IVFIndex.build(x, ids=ids, metric="angular", nlist=8)
    v = idx.vector_for_id(int(ids[5]))
    assert v is not None
    torch.testing.assert_close(v, x[5], rtol=1e-5, atol=1e-6)
    assert idx.vector_for_id(999999) is None


def test_serialize_roundtrip():
    x = _corpus(400, 24)
    idx = IVFIndex.build(x, metric="angular", storage="i8", nlist=8, seed=3)
    blob = idx.serialize()
    back = IVFIndex.deserialize(blob)
    q = x[:4]
    d1, i1 = idx.query(q, k=5, nprobe=8)
    d2, i2 = back.query(q, k=5, nprobe=8)
    torch.testing.assert_close(d1, d2)
    assert torch.equal(i1, i2)


@pytest.mark.gpu
@pytest.mark.parametrize("metric,storage", [
    ("angular", "i8"), ("angular", "f16"), ("euclidean", "f16"),
    ("dot", "f32"), ("angular", "f32"),
])
def test_native_scan_matches_fallback(metric, storage):
    x = _corpus(3000, 64)
    idx = IVFIndex.build(x, metric=metric, storage=storage, nlist=32,
                         device="cuda", seed=0)
    q = (x[:6] + torch.randn(6, 64) * 0.05).cuda()
    dist_n, row_n = idx.scan(q, nprobe=8)
    # golden: same math on the same packed data via the torch fallback
    q_enc, q_norm = idx._prepare_queries(q)
    probe = idx._rank_cells(q_enc, 8)
    counts = (idx.cell_off[1:] - idx.cell_off[:-1]).long()
    pc = counts[probe.long()]
    cand_off = torch.zeros_like(pc)
    cand_off[:, 1:] = torch.cumsum(pc, dim=1)[:, :-1]
    dist_f = torch.full_like(dist_n, float("inf"))
    row_f = torch.full_like(row_n, -1)
    idx._scan_fallback(q_enc, q_norm, probe, cand_off, dist_f, row_f)
    assert torch.equal(row_n, row_f)
    torch.testing.assert_close(dist_n, dist_f, rtol=1e-4, atol=1e-5)


@pytest.mark.gpu
def test_gpu_query_end_to_end_large():
    torch.manual_seed(0)
    x = torch.randn(200_000, 512)
    idx = IVFIndex.build(x, metric="angular", storage="i8", device="cuda", seed=0)
    q = x[:16].cuda()
    dist, ids = idx.query(q, k=10)
    assert (ids[:, 0].cpu() == torch.arange(16)).all()
    assert torch.isfinite(dist).all()


# -- incremental update (MI355X-native extra: packed-layout splice instead
# -- of the reference's wholesale rebuild) ---------------------------------

def test_incremental_add_found_by_query():
    x = _corpus(1500, 48)
    idx = IVFIndex.build(x, metric="angular", storage="i8", seed=0)
    extra = _corpus(40, 48, seed=9) + 0.05
    new_ids = torch.arange(10_000, 10_040)
    idx.add(extra, new_ids)
    assert idx.n == 1540
    # every added vector finds itself as nearest neighbor
    _, ids = idx.query(extra, k=1, nprobe=idx.nlist)
    assert torch.equal(ids.view(-1), new_ids)
    # packed invariants hold
    assert int(idx.cell_off[-1]) == idx.n
    assert idx.data.shape[0] == idx.ids.shape[0] == idx.vectors_f32.shape[0]


def test_incremental_add_matches_fresh_build_results():
    """After add(), queries return the same neighbors as a scan over the
    union corpus (full-probe, so coarse quantization cannot differ)."""
    x = _corpus(800, 32)
    extra = _corpus(60, 32, seed=5)
    idx = IVFIndex.build(x, metric="angular", storage="f32", seed=0)
    idx.add(extra, torch.arange(800, 860))
    q = _corpus(10, 32, seed=7)
    d, ids = idx.query(q, k=5, nprobe=idx.nlist)
    both = torch.cat([x, extra])
    want = _brute_force_ids(both, q, 5, "angular")
    assert (ids == want).float().mean() > 0.95


def test_incremental_upsert_replaces_vector():
    x = _corpus(500, 32)
    idx = IVFIndex.build(x, metric="angular", storage="f16", seed=0)
    moved = torch.randn(1, 32) * 2
    idx.add(moved, torch.tensor([123]))
    assert idx.n == 500  # replaced, not appended
    v = idx.vector_for_id(123)
    assert torch.allclose(v, moved[0], atol=1e-3)


def test_incremental_remove():
    x = _corpus(600, 32)
    idx = IVFIndex.build(x, metric="euclidean", storage="f32", seed=0)
    gone = torch.arange(0, 50)
    assert idx.remove(gone) == 50
    assert idx.n == 550
    assert idx.remove(gone) == 0  # already gone
    _, ids = idx.query(x[:50], k=3, nprobe=idx.nlist)
    assert not (ids.unsqueeze(-1) == gone.view(1, 1, -1)).any()
    # survivors still found
    _, ids2 = idx.query(x[100:110], k=1, nprobe=idx.nlist)
    assert torch.equal(ids2.view(-1), torch.arange(100, 110))


def test_incremental_add_without_f32_rerank_store():
    x = _corpus(400, 36)
    idx = IVFIndex.build(x, metric="angular", storage="i8", keep_f32=False)
    idx.add(_corpus(10, 36, seed=3), torch.arange(400, 410))
    assert idx.n == 410 and idx.vectors_f32 is None
    d, ids = idx.query(x[:5], k=1, nprobe=idx.nlist, rerank=False)
    assert torch.equal(ids.view(-1), torch.arange(5))
