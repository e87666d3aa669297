"""Opt-in fp8 (OCP e4m3) serving path for the CLAP encoder's GEMMs.

gfx950's MFMA runs fp8 at 2x the bf16 dense rate (and the fp8 operands
halve the HBM traffic of the weight/activation reads, which is what
actually matters for these memory-bound shapes — see
profiles/r01_kernel_pmc.md). `torch._scaled_mm` maps to hipBLASLt fp8
GEMMs on ROCm and was validated on MI355X (scripts/fp8_probe.py:
637 TF fp8 vs 457 TF bf16 at the encoder's dominant shape).

This is OFF by default: the BASELINE headline metric is bf16 per the
precision contract, and fp8 is an explicitly-labelled serving mode
(`AUDIOMUSE_FP8_SERVING=1`, or `bench.py --fp8` which reports
`"dtype": "fp8_e4m3"` so the number is never mistaken for the bf16
headline). Quantization is dynamic per-tensor e4m3 (weights cached
after first quantization; activations scaled by abs-max per call).

The reference has no quantized serving path (its ONNX students run
fp32 CUDA/DML — /root/reference/tasks/clap_analyzer.py); this is an
MI355X-native extra, not a parity item.

MEASURED STATUS (MI355X, end of round 1): the fused design WINS —
10 811 clips/s vs 9 621 bf16 same box (+11.5%), embedding cosine >
0.98 vs bf16. What made it pay (each step A/B-measured, see
profiles/r01_final_profile.md):
- quantization fused into the producing LayerNorm kernels
  (norms.hip F8 variants): e4m3 emitted via the gfx950 packed-convert
  instruction with a DELAYED per-tensor scale; amax is sampled (1/64
  of blocks) into a 256-slot vector — a naive per-wave atomic on one
  address serialized the whole kernel;
- fp8 GEMMs through the extension's timed algo search
  (gemm_gelu.cpp::linear_fp8): torch._scaled_mm's heuristic pick was
  2.1x SLOWER than bf16 at the stage-1 qkv shape, the searched algo is
  1.4x FASTER;
- only the profitable GEMMs convert (qkv + mlp0 with fused GELU_BIAS
  epilogue); proj and mlp2 stay bf16 (their gain is smaller than any
  quantize cost).
gfx950 note: non-scaled fp8 MFMA runs at the bf16 rate; the entire fp8
win at these memory-bound shapes is halved operand traffic, which is
why the quantize pass must ride a producer kernel, not its own.
"""

from __future__ import annotations

import weakref
from typing import Dict, Optional, Tuple

import torch

from audiomuse_amd import config as C

_E4M3 = torch.float8_e4m3fn
# Keyed by id(owner) with a validating weakref: allocator/GC id reuse
# after a model swap must never return another tensor's quantized data
# (ADVICE r1). The weakref callback evicts entries when the owner dies.
_wcache: Dict[int, tuple] = {}


def _weak_entry(cache: dict, owner, *payload):
    key = id(owner)
    cache[key] = (weakref.ref(owner, lambda _r, k=key, c=cache: c.pop(k, None)),
                  *payload)
    return cache[key]


def _weak_get(cache: dict, owner):
    ent = cache.get(id(owner))
    if ent is None or ent[0]() is not owner:
        return None
    return ent[1:]


def serving_enabled() -> bool:
    return bool(getattr(C, "CLAP_FP8_SERVING", False))


def available(device: Optional[torch.device] = None) -> bool:
    """fp8 scaled-mm usable here (needs a gfx950-class GPU + torch op)."""
    if not hasattr(torch, "_scaled_mm"):
        return False
    if device is not None and device.type != "cuda":
        return False
    return torch.cuda.is_available()


def quantize_weight(w: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """(N, K) bf16/f32 -> (wq (N, K) e4m3 contiguous, scale 0-d f32).

    Cached per weight tensor (weakref-validated): serving weights are
    frozen, so the quantization runs once per weight tensor and the
    entry dies with it.
    """
    try:
        ver = w._version
    except RuntimeError:          # inference tensors track no version
        ver = -1
    hit = _weak_get(_wcache, w)
    if hit is not None and hit[0] == ver:
        return hit[1], hit[2]
    fmax = torch.finfo(_E4M3).max
    s = (w.detach().abs().amax().float() / fmax).clamp(min=1e-12)
    wq = (w.detach().float() / s).clamp(-fmax, fmax).to(_E4M3).contiguous()
    _weak_entry(_wcache, w, ver, wq, s)
    return wq, s


# -- producer-fused path (the profitable one) -------------------------------
# LayerNorm emits e4m3 directly with a DELAYED per-tensor scale: the
# kernel quantizes with the previous step's amax while atomically
# recording the current one (transformer-engine-style). The standalone
# quantize pass — which made unfused fp8 a net loss — disappears.

_E4M3_MAX = 448.0
_ln_state: Dict[int, tuple] = {}


def _state_for(module, device) -> Tuple[torch.Tensor, torch.Tensor]:
    hit = _weak_get(_ln_state, module)
    st = hit[0] if hit is not None else None
    if st is None or st[0].device != device:
        # init scale for |LN out| up to ~8 (gamma ~ 1); self-corrects
        # from the recorded amax after the first step
        scale = torch.full((), 8.0 / _E4M3_MAX, device=device)
        amax = torch.zeros(256, device=device)   # slot-spread (see kernel)
        st = (scale, amax, [False])
        _weak_entry(_ln_state, module, st)
    scale, amax, warm = st
    if warm[0]:
        scale.copy_((amax.max().clamp(min=1e-6) * 1.05) / _E4M3_MAX)
        amax.zero_()
    warm[0] = True
    return scale, amax


_hid_state: Dict[int, tuple] = {}
# fp8 hidden chain (mlp0 emits e4m3 for an fp8 mlp2): measured NEUTRAL
# (10 739 vs 10 811 clips/s) while costing embedding accuracy (the
# e4m3-quantized 4C hidden fails the cosine>0.98 gate), so it is OFF by
# default; AUDIOMUSE_FP8_HIDDEN=1 re-enables for experiments.
FP8_HIDDEN = [bool(getattr(C, "FP8_HIDDEN_ENABLE", False))]
FP8_HIDDEN_ERR = ""

# fp8-ingest attention (QKV GEMM emits e4m3, window kernel reads 8-byte
# fragments): measured NEGATIVE — see FP8_ATTN_ENABLE in config.py for
# the numbers. OFF by default; AUDIOMUSE_FP8_ATTN=1 re-enables for
# experiments. Self-disables if hipBLASLt has no fp8-D algo at the shape.
FP8_ATTN = [bool(getattr(C, "FP8_ATTN_ENABLE", False))]
FP8_ATTN_ERR = ""


def hidden_state(module, device):
    """(scale, inv_scale, amax) for a GEMM's fp8 D output: the GEMM
    records the true amax (AMAX_D pointer), delayed like the LN scales."""
    hit = _weak_get(_hid_state, module)
    st = hit[0] if hit is not None else None
    if st is None or st[0].device != device:
        scale = torch.full((), 4.0 / _E4M3_MAX, device=device)
        inv = torch.full((), _E4M3_MAX / 4.0, device=device)
        amax = torch.zeros((), device=device)
        st = (scale, inv, amax, [False])
        _weak_entry(_hid_state, module, st)
    scale, inv, amax, warm = st
    if warm[0]:
        scale.copy_((amax.clamp(min=1e-6) * 1.05) / _E4M3_MAX)
        inv.copy_(1.0 / scale)
        amax.zero_()
    warm[0] = True
    return scale, inv, amax


def ln_fp8(norm, x: torch.Tensor):
    """FusedLayerNorm -> (y_fp8, scale) via the fused-quantize kernel."""
    from audiomuse_amd.ops import _ext

    ext = _ext.require()
    scale, amax = _state_for(norm, x.device)
    y8 = ext.layernorm_bf16_fp8(
        x.contiguous(), norm.weight.to(torch.bfloat16).contiguous(),
        norm.bias.to(torch.bfloat16).contiguous(), norm.eps, scale, amax)
    return y8, scale


def add_ln_fp8(norm, x: torch.Tensor, other: torch.Tensor):
    """Residual add + LN -> (sum_bf16, y_fp8, scale)."""
    from audiomuse_amd.ops import _ext

    ext = _ext.require()
    scale, amax = _state_for(norm, x.device)
    s, y8 = ext.add_layernorm_bf16_fp8(
        x.contiguous(), other.contiguous(),
        norm.weight.to(torch.bfloat16).contiguous(),
        norm.bias.to(torch.bfloat16).contiguous(), norm.eps, scale, amax)
    return s, y8, scale


def scaled_linear(x: torch.Tensor, weight: torch.Tensor,
                  bias: Optional[torch.Tensor]) -> torch.Tensor:
    """Linear in fp8: x (..., K) bf16 @ weight (N, K) -> (..., N) bf16.

    Dynamic per-tensor activation scale (abs-max), cached per-tensor
    weight scale. `_scaled_mm` needs A row-major and B column-major,
    which `wq.t()` of a contiguous (N, K) quantized weight provides.
    """
    fmax = torch.finfo(_E4M3).max
    lead = x.shape[:-1]
    x2 = x.reshape(-1, x.shape[-1])
    xs = (x2.abs().amax().float() / fmax).clamp(min=1e-12)
    xq = (x2.float() / xs).clamp(-fmax, fmax).to(_E4M3)
    wq, ws = quantize_weight(weight)
    out = torch._scaled_mm(xq, wq.t(), scale_a=xs, scale_b=ws,
                           bias=None if bias is None else bias.to(torch.bfloat16),
                           out_dtype=torch.bfloat16)
    return out.reshape(*lead, -1)
