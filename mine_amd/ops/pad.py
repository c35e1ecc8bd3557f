"""Reflection padding on the fused HIP kernels.

Replaces nn.ReflectionPad2d in the decoder's ConvBlock / dispconv heads
(ref network/monodepth2/layers.py:123-138). At the flagship config the
eager PyTorch pad pair (atomic-scatter backward) cost ~24% of the train
step (profiles/r01_flagship_kernel_stats.md); the HIP backward here is a
pure gather (each input pixel sums its <=9 contributing output pixels) —
deterministic and atomic-free.

Layout handling: channels_last tensors map to the kernel's logical
(N,H,W,C) directly (channel-adjacent gathers, coalesced); contiguous
NCHW tensors map as (B*C, H, W, 1). CPU (and exotic dtypes) fall back to
F.pad(mode="reflect").
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from mine_amd.ops.backend import get_extension


def _layout(t: torch.Tensor):
    """-> (flat_contiguous_view, N, H, W, C, channels_last)"""
    B, C, H, W = t.shape
    if t.is_contiguous(memory_format=torch.channels_last):
        return t.permute(0, 2, 3, 1).reshape(-1), B, H, W, C, True
    return t.contiguous().reshape(-1), B * C, H, W, 1, False


class _ReflectPadFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, pad):
        ext = get_extension(required=True)
        flat, N, H, W, C, cl = _layout(x)
        out = ext.reflect_pad_fwd(flat, N, H, W, C, pad)
        ctx.geom = (x.shape, N, H, W, C, cl, pad)
        B = x.shape[0]
        Ho, Wo = H + 2 * pad, W + 2 * pad
        if cl:
            return out.view(B, Ho, Wo, C).permute(0, 3, 1, 2)
        return out.view(B, x.shape[1], Ho, Wo)

    @staticmethod
    def backward(ctx, gout):
        ext = get_extension(required=True)
        shape, N, H, W, C, cl, pad = ctx.geom
        B, Cfull = shape[0], shape[1]
        if cl:
            flat = gout.permute(0, 2, 3, 1).contiguous().reshape(-1)
        else:
            flat = gout.contiguous().reshape(-1)
        gin = ext.reflect_pad_bwd(flat, N, H, W, C, pad)
        if cl:
            gin = gin.view(B, H, W, C).permute(0, 3, 1, 2)
        else:
            gin = gin.view(B, Cfull, H, W)
        return gin, None


def reflection_pad2d(x: torch.Tensor, pad: int = 1) -> torch.Tensor:
    """Reflection-pad a Bx C x H x W tensor by `pad` on every side."""
    if x.is_cuda and x.dtype in (torch.float32, torch.bfloat16):
        return _ReflectPadFn.apply(x, pad)
    return F.pad(x, (pad, pad, pad, pad), mode="reflect")


class ReflectionPad2d(torch.nn.Module):
    """Drop-in nn.ReflectionPad2d running on the HIP gather kernels."""

    def __init__(self, pad: int = 1):
        super().__init__()
        self.pad = int(pad)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return reflection_pad2d(x, self.pad)

    def extra_repr(self) -> str:  # pragma: no cover
        return f"pad={self.pad}"
