"""Fused MPI head: dispconv output -> packed fp32 MPI in one pass.

The decoder head's eager chain (view + sigmoid + abs + cat + permute +
contiguous + float; ref depth_decoder.py:134-146 and the engine's pack)
makes many full passes over the ~200 MB per-scale head tensors. In
channels_last the conv output (B*S, 4, H, W) is laid out exactly as the
packed (B*S, H, W, 4) the renderer consumes, so the fused kernel is one
elementwise pass: rgb=sigmoid, sigma=|x|+1e-4 (or sigmoid alpha), fp32
out.
"""
from __future__ import annotations

import torch

from mine_amd.ops.backend import get_extension


class _MPIHeadFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, z_flat, N, alpha):
        ext = get_extension(required=True)
        out = ext.mpi_head_fwd(z_flat, N, alpha)
        ctx.save_for_backward(z_flat)
        ctx.meta = (N, alpha)
        return out

    @staticmethod
    def backward(ctx, gout):
        ext = get_extension(required=True)
        (z_flat,) = ctx.saved_tensors
        N, alpha = ctx.meta
        gz = ext.mpi_head_bwd(z_flat, gout.contiguous(), N, alpha)
        return gz, None, None


def mpi_head_pack(conv_out: torch.Tensor, B: int, S: int,
                  use_alpha: bool = False) -> torch.Tensor:
    """(B*S, 4, H, W) dispconv output -> (B, S, H, W, 4) packed fp32 MPI.

    GPU channels_last: one fused kernel. Fallback: eager ops of the same
    math (CPU path / exotic layouts).
    """
    BS, C, H, W = conv_out.shape
    assert C == 4 and BS == B * S
    if conv_out.is_cuda and conv_out.dtype in (torch.float32, torch.bfloat16) \
            and conv_out.is_contiguous(memory_format=torch.channels_last):
        z_flat = conv_out.permute(0, 2, 3, 1).reshape(-1)  # zero-copy view
        out = _MPIHeadFn.apply(z_flat, BS * H * W, use_alpha)
        return out.view(B, S, H, W, 4)

    z = conv_out.float().view(B, S, 4, H, W)
    rgb = torch.sigmoid(z[:, :, 0:3])
    sigma = torch.sigmoid(z[:, :, 3:]) if use_alpha \
        else torch.abs(z[:, :, 3:]) + 1e-4
    return torch.cat((rgb, sigma), dim=2).permute(0, 1, 3, 4, 2).contiguous()
