"""Opt-in fp8 e4m3 serving path (ops/fp8.py). The headline bench stays
bf16; these tests pin the flag plumbing and the quantization accuracy."""

import pytest
import torch

from audiomuse_amd import config as C
from audiomuse_amd.ops import fp8


def test_off_by_default_and_unavailable_on_cpu():
    assert C.CLAP_FP8_SERVING is False
    assert fp8.serving_enabled() is False
    assert fp8.available(torch.device("cpu")) is False


def test_quantize_weight_accuracy_and_cache():
    w = torch.randn(64, 32) * 0.3
    wq, s = fp8.quantize_weight(w)
    assert wq.dtype == torch.float8_e4m3fn and s.dtype == torch.float32
    deq = wq.float() * s
    rel = (deq - w).abs().mean() / w.abs().mean()
    assert rel < 0.05  # e4m3 has ~2 decimal digits near abs-max scale
    wq2, s2 = fp8.quantize_weight(w)
    assert wq2 is wq and s2 is s  # cached by (data_ptr, version)
    w += 1.0  # version bump invalidates
    wq3, _ = fp8.quantize_weight(w)
    assert wq3 is not wq


@pytest.mark.gpu
def test_scaled_linear_matches_bf16_linear():
    x = torch.randn(512, 256, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(384, 256, device="cuda", dtype=torch.bfloat16) * 0.05
    b = torch.randn(384, device="cuda", dtype=torch.bfloat16)
    out = fp8.scaled_linear(x, w, b)
    ref = torch.nn.functional.linear(x, w, b)
    rel = (out.float() - ref.float()).abs().mean() / ref.float().abs().mean()
    assert out.dtype == torch.bfloat16 and float(rel) < 0.08


@pytest.mark.gpu
def test_fp8_encoder_embeddings_match_bf16(monkeypatch):
    from audiomuse_amd.models.htsat import HTSATConfig, HTSATEncoder

    torch.manual_seed(0)
    model = HTSATEncoder(HTSATConfig()).to("cuda", torch.bfloat16).eval()
    mel = torch.randn(4, 128, 1024, device="cuda", dtype=torch.bfloat16)
    with torch.inference_mode():
        ref = model(mel).float()
        monkeypatch.setattr(C, "CLAP_FP8_SERVING", True)
        out = model(mel).float()
    cos = torch.nn.functional.cosine_similarity(ref, out, dim=1)
    assert float(cos.min()) > 0.98


def test_quantize_weight_under_inference_mode():
    # inference tensors track no version counter (the soak hit this)
    with torch.inference_mode():
        w = torch.randn(16, 8) * 0.2
        wq, s = fp8.quantize_weight(w)
        assert wq.dtype == torch.float8_e4m3fn
        wq2, _ = fp8.quantize_weight(w)
        assert wq2 is wq  # cached under the fallback key


def test_hidden_chain_off_by_default():
    assert fp8.FP8_HIDDEN[0] is False
