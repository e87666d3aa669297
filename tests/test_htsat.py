import pytest
import torch

from audiomuse_amd.models.htsat import (HTSATConfig, HTSATEncoder,
                                        clap_track_embedding,
                                        window_partition, window_reverse)


def tiny_cfg():
    return HTSATConfig(n_mels=32, n_frames=64, patch_size=4, embed_dim=32,
                       depths=(1, 1), num_heads=(2, 4), window=4, out_dim=16)


def test_window_partition_roundtrip():
    x = torch.randn(2, 16, 24, 8)
    win = window_partition(x, 4)
    assert win.shape == (2 * 4 * 6, 16, 8)
    back = window_reverse(win, 4, 16, 24)
    assert torch.equal(back, x)


def test_tiny_forward_shapes_and_grad():
    torch.manual_seed(0)
    m = HTSATEncoder(tiny_cfg())
    x = torch.randn(3, 32, 60)  # shorter than n_frames -> padded
    out = m(x)
    assert out.shape == (3, 16)
    out.square().mean().backward()
    grads = [p.grad for p in m.parameters() if p.requires_grad]
    assert all(g is not None for g in grads if g is not None)


def test_forward_deterministic():
    torch.manual_seed(0)
    m = HTSATEncoder(tiny_cfg())
    x = torch.randn(2, 32, 64)
    with torch.no_grad():
        a, b = m(x), m(x)
    assert torch.equal(a, b)


def test_full_config_geometry():
    cfg = HTSATConfig()
    m = HTSATEncoder(cfg)
    # stage windows: 64-token windows (one wavefront) except the 4x32 last stage
    assert m.stage_windows == [8, 8, 8, 4]
    n = sum(p.numel() for p in m.parameters())
    assert 40e6 < n < 60e6  # ~100 MB bf16: same class as the reference student


def test_track_embedding_mean_norm():
    segs = torch.randn(5, 512)
    emb = clap_track_embedding(segs)
    assert emb.shape == (512,)
    assert abs(float(emb.norm()) - 1.0) < 1e-5


@pytest.mark.gpu
def test_htsat_gpu_bf16_forward():
    m = HTSATEncoder(HTSATConfig()).to("cuda", torch.bfloat16).eval()
    x = torch.randn(4, 128, 1001, device="cuda", dtype=torch.bfloat16)
    with torch.inference_mode():
        out = m(x)
    assert out.shape == (4, 512)
    assert torch.isfinite(out.float()).all()
