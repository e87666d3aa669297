import pytest
import torch
import torch.nn.functional as F

from audiomuse_amd.ops.norms import FusedLayerNorm


def test_fused_ln_cpu_matches_torch():
    ln = FusedLayerNorm(128)
    torch.nn.init.normal_(ln.weight)
    torch.nn.init.normal_(ln.bias)
    x = torch.randn(4, 10, 128)
    torch.testing.assert_close(ln(x), F.layer_norm(x, (128,), ln.weight, ln.bias, ln.eps))


@pytest.mark.gpu
@pytest.mark.parametrize("dim", [128, 256, 512, 1024])
def test_fused_ln_gpu_matches_reference(dim):
    torch.manual_seed(0)
    ln = FusedLayerNorm(dim).to("cuda", torch.bfloat16)
    with torch.no_grad():
        ln.weight.normal_()
        ln.bias.normal_()
    x = torch.randn(3, 97, dim, device="cuda", dtype=torch.bfloat16)
    with torch.inference_mode():
        got = ln(x)
    expect = F.layer_norm(x.float(), (dim,), ln.weight.float(), ln.bias.float(),
                          ln.eps).to(torch.bfloat16)
    # bf16 output: one-ulp differences expected
    torch.testing.assert_close(got.float(), expect.float(), rtol=2e-2, atol=2e-2)


@pytest.mark.gpu
def test_fused_ln_odd_row_count():
    ln = FusedLayerNorm(256).to("cuda", torch.bfloat16)
    x = torch.randn(1, 5, 256, device="cuda", dtype=torch.bfloat16)
    with torch.inference_mode():
        got = ln(x)
    assert got.shape == x.shape and torch.isfinite(got.float()).all()


@pytest.mark.gpu
def test_mel_fused_quant16_matches_reference():
    from audiomuse_amd.ops import dsp, hip_ops

    cfg = dsp.clap_mel_config()
    audio = (torch.randn(2, 96000, device="cuda") * 0.4).clamp(-1, 1)
    fused = hip_ops.mel_spectrogram(audio, cfg, quantize_int16=True)
    ref = hip_ops.mel_spectrogram(dsp.int16_roundtrip(audio), cfg,
                                  force_reference=True)
    torch.testing.assert_close(fused, ref, rtol=1e-3, atol=2e-3)


@pytest.mark.gpu
@pytest.mark.parametrize("dim", [128, 512])
def test_fused_add_ln_matches_reference(dim):
    import audiomuse_amd._C as C

    torch.manual_seed(0)
    x = torch.randn(5, 33, dim, device="cuda", dtype=torch.bfloat16)
    other = torch.randn_like(x)
    w = torch.randn(dim, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(dim, device="cuda", dtype=torch.bfloat16)
    s, y = C.add_layernorm_bf16(x.contiguous(), other.contiguous(),
                                w.contiguous(), b.contiguous(), 1e-5)
    expect_sum = (x.float() + other.float())
    torch.testing.assert_close(s.float(), expect_sum.to(torch.bfloat16).float(),
                               rtol=2e-2, atol=2e-2)
    expect = F.layer_norm(expect_sum, (dim,), w.float(), b.float(), 1e-5)
    torch.testing.assert_close(y.float(), expect, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_linear_gelu_epilogue_matches_reference():
    import audiomuse_amd._C as C

    torch.manual_seed(0)
    x = torch.randn(7, 33, 256, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(1024, 256, device="cuda", dtype=torch.bfloat16) * 0.05
    b = torch.randn(1024, device="cuda", dtype=torch.bfloat16)
    got = C.linear_gelu(x.contiguous(), w.contiguous(), b.contiguous())
    expect = F.gelu(F.linear(x.float(), w.float(), b.float()),
                    approximate="tanh")
    assert got.shape == expect.shape
    torch.testing.assert_close(got.float(), expect, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_linear_bias_add_matches_reference():
    import audiomuse_amd._C as C

    torch.manual_seed(1)
    x = torch.randn(5, 17, 512, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(128, 512, device="cuda", dtype=torch.bfloat16) * 0.05
    b = torch.randn(128, device="cuda", dtype=torch.bfloat16)
    r = torch.randn(5, 17, 128, device="cuda", dtype=torch.bfloat16)
    got = C.linear_bias_add(x.contiguous(), w.contiguous(), b.contiguous(),
                            r.contiguous())
    expect = F.linear(x.float(), w.float(), b.float()) + r.float()
    torch.testing.assert_close(got.float(), expect, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_linear_bias_matches_reference():
    import audiomuse_amd._C as C

    torch.manual_seed(2)
    x = torch.randn(3, 100, 128, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(384, 128, device="cuda", dtype=torch.bfloat16) * 0.05
    b = torch.randn(384, device="cuda", dtype=torch.bfloat16)
    got = C.linear_bias(x.contiguous(), w.contiguous(), b.contiguous())
    expect = F.linear(x.float(), w.float(), b.float())
    torch.testing.assert_close(got.float(), expect, rtol=3e-2, atol=3e-2)
