"""Fused normalization ops (HIP on GPU, torch elsewhere)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from audiomuse_amd.ops import _ext


class FusedLayerNorm(nn.LayerNorm):
    """LayerNorm that runs the fused bf16 HIP kernel on the inference
    path (ops/csrc/norms.hip) and eager torch otherwise (training/CPU)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if (x.is_cuda and x.dtype == torch.bfloat16
                and not torch.is_grad_enabled()
                and self.weight is not None and self.bias is not None
                and len(self.normalized_shape) == 1
                and self.normalized_shape[0] <= 4096
                and self.normalized_shape[0] % 4 == 0):
            ext = _ext.native_or_none()
            if ext is not None and hasattr(ext, "layernorm_bf16"):
                return ext.layernorm_bf16(
                    x.contiguous(), self.weight.to(torch.bfloat16).contiguous(),
                    self.bias.to(torch.bfloat16).contiguous(), self.eps)
        return F.layer_norm(x, self.normalized_shape, self.weight, self.bias,
                            self.eps)
