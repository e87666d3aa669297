"""Windowed attention: torch SDPA reference path.

The HTSAT encoder's hot op — multi-head attention over 8x8 = 64-token
windows with a relative-position bias and (for shifted windows) an
additive group mask. The fused HIP kernel (ops/csrc/attention.hip: one
wavefront per (window, head), mfma_f32_16x16x32_bf16 QK^T/PV, softmax +
bias + inline shift mask in registers) operates on the full
(B, H, W, 3C) image layout and dispatches one level up
(models/htsat.py SwinBlock.forward). This per-window entry point is the
numerics reference (CPU + training path) the kernel is tested against.

Reference behavior being replaced: ONNX Runtime attention inside the
DCLAP student (/root/reference/tasks/clap_analyzer.py:478-500).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F



def _sdpa_reference(q, k, v, bias, mask, scale):
    nW, h, T, d = q.shape
    if mask is None:
        am = bias.to(q.dtype)                              # (h, T, T)
        return F.scaled_dot_product_attention(q, k, v, attn_mask=am, scale=scale)
    nw = mask.shape[0]
    B = nW // nw
    am = (bias.unsqueeze(0) + mask.unsqueeze(1)).to(q.dtype)   # (nw, h, T, T)
    q4 = q.reshape(B, nw * h, T, d)
    k4 = k.reshape(B, nw * h, T, d)
    v4 = v.reshape(B, nw * h, T, d)
    out = F.scaled_dot_product_attention(q4, k4, v4,
                                         attn_mask=am.view(nw * h, T, T),
                                         scale=scale)
    return out.reshape(nW, h, T, d)


def window_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     bias: torch.Tensor, mask: torch.Tensor | None,
                     scale: float) -> torch.Tensor:
    """q, k, v: (nW, heads, T, head_dim); bias: (heads, T, T);
    mask: (windows_per_image, T, T) additive or None. Returns (nW, h, T, d)."""
    # The fused HIP kernel operates on the full (B, H, W, 3C) image layout
    # and is dispatched one level up (models/htsat.py SwinBlock.forward);
    # this per-window entry point always runs the SDPA reference.
    return _sdpa_reference(q, k, v, bias, mask, scale)
