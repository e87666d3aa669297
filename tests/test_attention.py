"""Fused windowed-attention kernel tests (GPU) + CPU dispatch checks."""

import pytest
import torch

from audiomuse_amd.models.htsat import SwinBlock, _shift_mask


@pytest.mark.gpu
def test_mfma_probe_layout():
    """Pin the gfx950 mfma_f32_16x16x32_bf16 A/B fragment layout against
    torch.matmul with random asymmetric inputs (guide G9)."""
    torch.manual_seed(0)
    import audiomuse_amd._C as C

    A = (torch.randn(16, 32, device="cuda") * 0.5).to(torch.bfloat16)
    B = (torch.randn(32, 16, device="cuda") * 0.5).to(torch.bfloat16)
    D = C.mfma_probe(A.contiguous(), B.contiguous())
    expect = A.float() @ B.float()
    torch.testing.assert_close(D, expect, rtol=2e-2, atol=2e-2)


def _eager_block_forward(blk, x, H, W, mask):
    """Force the eager (non-fused) path for comparison."""
    import torch.nn.functional as F  # noqa: F401
    from audiomuse_amd.models.htsat import window_partition, window_reverse

    B, L, C = x.shape
    shortcut = x
    xn = blk.norm1(x).view(B, H, W, C)
    if blk.shift:
        xn = torch.roll(xn, shifts=(-blk.shift, -blk.shift), dims=(1, 2))
    win = window_partition(xn, blk.window)
    win = blk.attn(win, mask if blk.shift else None)
    xn = window_reverse(win, blk.window, H, W)
    if blk.shift:
        xn = torch.roll(xn, shifts=(blk.shift, blk.shift), dims=(1, 2))
    x = shortcut + xn.reshape(B, L, C)
    return x + blk.mlp(blk.norm2(x))


@pytest.mark.gpu
@pytest.mark.parametrize("shift", [0, 4])
@pytest.mark.parametrize("dim,heads", [(128, 4), (256, 8), (512, 16)])
def test_fused_window_attention_matches_eager(dim, heads, shift):
    torch.manual_seed(dim + shift)
    H, W = 16, 32
    blk = SwinBlock(dim, heads, window=8, shift=shift, mlp_ratio=4.0)
    blk = blk.to("cuda", torch.bfloat16).eval()
    x = torch.randn(2, H * W, dim, device="cuda", dtype=torch.bfloat16)
    mask = _shift_mask(H, W, 8, 4, torch.device("cuda")).to(torch.bfloat16) \
        if shift else None

    with torch.inference_mode():
        assert blk._fused_attn_available(x), "fused path must be active on GPU"
        fused = blk(x, H, W, mask)
    with torch.no_grad(), torch.inference_mode(False):
        # grad-enabled context forces the eager path in _fused_attn_available,
        # but compute under no_grad for numerics
        eager = _eager_block_forward(blk, x.clone(), H, W, mask)

    assert fused.shape == eager.shape
    fused, eager = fused.detach(), eager.detach()  # silence scalar-conv warning
    diff = (fused.float() - eager.float()).abs()
    rel = diff.mean() / eager.float().abs().mean().clamp(min=1e-6)
    assert float(rel) < 3e-2, f"mean rel err {float(rel):.4f}"
    assert float(diff.max()) < 0.5, f"max abs err {float(diff.max()):.4f}"


@pytest.mark.gpu
def test_fused_attention_rows_softmax_sane():
    """Degenerate check: with zero qkv weights + uniform V the output is
    the V bias value (softmax rows sum to 1)."""
    torch.manual_seed(1)
    blk = SwinBlock(128, 4, window=8, shift=0, mlp_ratio=1.0)
    with torch.no_grad():
        blk.attn.qkv.weight.zero_()
        blk.attn.qkv.bias.zero_()
        blk.attn.qkv.bias[2 * 128:].fill_(0.5)   # V bias = 0.5
        blk.attn.rel_bias.zero_()
        blk.attn.proj.weight.copy_(torch.eye(128))
        blk.attn.proj.bias.zero_()
        for m in blk.mlp:
            if hasattr(m, "weight"):
                m.weight.zero_()
                m.bias.zero_()
    blk = blk.to("cuda", torch.bfloat16).eval()
    x = torch.randn(1, 16 * 16, 128, device="cuda", dtype=torch.bfloat16)
    with torch.inference_mode():
        out = blk(x, 16, 16, None)
    expect = x.float() + 0.5
    torch.testing.assert_close(out.float(), expect, rtol=2e-2, atol=2e-2)


def test_cpu_path_unaffected():
    blk = SwinBlock(64, 2, window=4, shift=2, mlp_ratio=2.0)
    x = torch.randn(2, 64, 64)
    mask = _shift_mask(8, 8, 4, 2, torch.device("cpu"))
    out = blk(x, 8, 8, mask)
    assert out.shape == x.shape
