"""General MFMA implicit-GEMM convolution (encoder / neck / base convs).

Carries the ResNet-50 stack (ref network/monodepth2/resnet_encoder.py:
100-108: 7x7 s2 stem, 1x1 / 3x3 bottleneck convs, 1x1 s2 downsamples),
the decoder's receptive-field neck (ref depth_decoder.py:56-61) and the
SplitConvBlock base convs (reflect-padded 3x3, ref monodepth2/
layers.py:123-138) on the hand-written igemm kernels
(ops/csrc/igemm_kernels.hip) — forward, data-grad and weight-grad.
These shapes are tiny (batch 4, L2-resident) and were launch-bound on
the library path; see the kernel header for the design.

Weight packing is ONE kernel launch: a cached device LUT encodes the
fragment order AND any transpose (data-grad) or channel padding (the
stem's C=3), so no permute/pad/cast chain ever materializes.
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from mine_amd.ops.backend import get_extension

_LUT: Dict[Tuple, torch.Tensor] = {}


def _frag_lut(K_log: int, C_log: int, R: int, S: int, C_phys: int,
              K_avail: int, C_avail: int, trans: bool,
              device) -> torch.Tensor:
    """Fragment-order gather LUT (int32; -1 -> packed zero).

    Logical weight L[a, b, tap] with a in [0, K_log) rows and b in
    [0, C_log) contraction channels (C_log % 8 == 0). Physical index
    into the flat (K_phys, C_phys, R, S) tensor:
      trans=False: ((a * C_phys + b) * R*S + tap), valid a<K_avail, b<C_avail
      trans=True:  ((b * C_phys + a) * R*S + tap), valid a<C_avail, b<K_avail
    (the data-grad uses the TRANSPOSED, unflipped weight — the kernel's
    coordinate map walks the taps in reverse).
    """
    key = (K_log, C_log, R, S, C_phys, K_avail, C_avail, trans, str(device))
    lut = _LUT.get(key)
    if lut is not None:
        return lut
    Cv = C_log // 8
    nseg = R * S * Cv
    nchunks = (nseg + 3) // 4
    nK = (K_log + 15) // 16
    nc = torch.arange(nK).view(-1, 1, 1, 1)
    kc = torch.arange(nchunks).view(1, -1, 1, 1)
    lane = torch.arange(64).view(1, 1, -1, 1)
    e = torch.arange(8).view(1, 1, 1, -1)
    seg = kc * 4 + (lane >> 4)
    a = (nc * 16 + (lane & 15)).expand(nK, nchunks, 64, 8)
    tap = seg // Cv
    b = ((seg - tap * Cv) * 8 + e).expand(nK, nchunks, 64, 8)
    tap = tap.expand(nK, nchunks, 64, 8)
    if trans:
        idx = (b * C_phys + a) * (R * S) + tap
        invalid = (seg >= nseg).expand_as(idx) | (a >= C_avail) | \
                  (b >= K_avail)
    else:
        idx = (a * C_phys + b) * (R * S) + tap
        invalid = (seg >= nseg).expand_as(idx) | (a >= K_avail) | \
                  (b >= C_avail)
    idx = torch.where(invalid, torch.tensor(-1), idx)
    lut = idx.reshape(-1).to(device=device, dtype=torch.int32)
    _LUT[key] = lut
    return lut


def pack_weights_general(w: torch.Tensor, trans: bool = False) -> torch.Tensor:
    """(K, C, R, S) contiguous -> fragment-ordered bf16 buffer (one
    launch). trans packs the transposed (C-major) logical layout for the
    data-grad."""
    ext = get_extension(required=True)
    K, C, R, S = w.shape
    Cp = (C + 7) & ~7
    Kp = (K + 7) & ~7
    if trans:
        lut = _frag_lut(Cp, Kp, R, S, C, K, C, True, w.device)
    else:
        lut = _frag_lut(K, Cp, R, S, C, K, C, False, w.device)
    return ext.pack_gather(w.reshape(-1), lut)


class _ConvIgemmFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, stride, pad, reflect):
        ext = get_extension(required=True)
        B, C, Hs, Ws = x.shape
        K, _, R, S = w.shape
        if C % 8:
            x = torch.cat((x, x.new_zeros(B, 8 - C % 8, Hs, Ws)),
                          1).contiguous(memory_format=torch.channels_last)
        Cp = x.shape[1]
        if reflect:
            P, Q = Hs, Ws  # reflect pad keeps size (stride 1, pad (R-1)/2)
        else:
            P = (Hs + 2 * pad - R) // stride + 1
            Q = (Ws + 2 * pad - S) // stride + 1
        wp = pack_weights_general(w)
        M = B * P * Q
        out = ext.conv_igemm_fwd(
            x.permute(0, 2, 3, 1).reshape(-1), wp,
            bias.float() if bias is not None else
            torch.empty(0, device=x.device, dtype=torch.float32),
            M, P, Q, K, Hs, Ws, Cp, R, S,
            stride, 1, -pad, 1, 1 if reflect else 0)
        ctx.save_for_backward(x, w)
        ctx.geom = (B, C, Cp, Hs, Ws, K, R, S, P, Q, stride, pad, reflect)
        ctx.has_bias = bias is not None
        return out.view(B, P, Q, K).permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, gy):
        ext = get_extension(required=True)
        x, w = ctx.saved_tensors  # x already channel-padded
        B, C, Cp, Hs, Ws, K, R, S, P, Q, stride, pad, reflect = ctx.geom
        gy = gy.contiguous(memory_format=torch.channels_last)
        gy_flat = gy.permute(0, 2, 3, 1).reshape(-1)
        x_flat = x.permute(0, 2, 3, 1).reshape(-1)

        gx = gw = gb = None
        Kp = (K + 7) & ~7
        if K % 8:
            # the data-grad reads gy in 8-channel fragments: zero-pad
            # the 4-channel dispconv gradient once here
            gy_p = torch.cat(
                (gy, gy.new_zeros(B, Kp - K, P, Q)),
                1).contiguous(memory_format=torch.channels_last)
            gy_flat_p = gy_p.permute(0, 2, 3, 1).reshape(-1)
        else:
            gy_flat_p = gy_flat
        if ctx.needs_input_grad[1]:
            # wrw maps OUT pixels through the forward tap coordinates
            dw = ext.conv_igemm_wrw(
                x_flat, gy_flat, B * P * Q, P, Q, K, Hs, Ws, Cp, R, S,
                stride, 1, -pad, 1, 1 if reflect else 0)
            gw = dw.view(K, Cp, R, S)[:, :C] if Cp != C \
                else dw.view(K, C, R, S)
            gw = gw.to(w.dtype)
        if ctx.has_bias:
            gb = gy.float().sum((0, 2, 3)).to(w.dtype)
        if ctx.needs_input_grad[0]:
            wtp = pack_weights_general(w, trans=True)
            empty = torch.empty(0, device=x.device, dtype=torch.float32)
            if reflect:
                # grad to the (virtually) padded input, then reflect-fold
                Hp, Wp_ = Hs + 2 * pad, Ws + 2 * pad
                gxp = ext.conv_igemm_fwd(
                    gy_flat_p, wtp, empty, B * Hp * Wp_, Hp, Wp_, Cp,
                    P, Q, Kp, R, S, 1, -1, 0, 1, 0)
                gx = ext.reflect_pad_bwd(gxp, B, Hs, Ws, Cp, pad)
                gx = gx.view(B, Hs, Ws, Cp).permute(0, 3, 1, 2)
            else:
                gxf = ext.conv_igemm_fwd(
                    gy_flat_p, wtp, empty, B * Hs * Ws, Hs, Ws, Cp,
                    P, Q, Kp, R, S, 1, -1, pad, stride, 0)
                gx = gxf.view(B, Hs, Ws, Cp).permute(0, 3, 1, 2)
            if Cp != C:
                gx = gx[:, :C]
        return gx, gw, gb, None, None, None


def conv2d_mfma(x: torch.Tensor, w: torch.Tensor,
                bias: Optional[torch.Tensor] = None, stride: int = 1,
                padding: int = 0, reflect: bool = False) -> torch.Tensor:
    """Conv2d on the hand-written igemm kernels; eager fallback when the
    fast-path conditions don't hold. reflect=True means
    ReflectionPad((R-1)/2) + unpadded conv (stride must be 1)."""
    K, C, R, S = w.shape
    if x.is_cuda and x.dtype == torch.float16 and (R, S) in (
            (1, 1), (3, 3), (7, 7)) and (not reflect or stride == 1):
        # fp16 configs: cast through the bf16 MFMA path (see conv.py)
        return conv2d_mfma(x.to(torch.bfloat16), w, bias, stride, padding,
                           reflect).to(torch.float16)
    usable = (x.is_cuda and x.dtype == torch.bfloat16
              and (R, S) in ((1, 1), (3, 3), (7, 7))
              and w.is_contiguous()
              and x.is_contiguous(memory_format=torch.channels_last)
              and (not reflect or stride == 1))
    if usable:
        return _ConvIgemmFn.apply(x, w, bias,
                                  stride, padding if not reflect
                                  else (R - 1) // 2, reflect)
    if reflect:
        x = F.pad(x, ((S - 1) // 2,) * 2 + ((R - 1) // 2,) * 2,
                  mode="reflect")
        padding = 0
    return F.conv2d(x, w.to(x.dtype),
                    bias.to(x.dtype) if bias is not None else None,
                    stride=stride, padding=padding)


class Conv2dMFMA(nn.Conv2d):
    """nn.Conv2d whose forward runs on the igemm kernels (same
    parameter/state-dict surface; checkpoint contract unchanged)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return conv2d_mfma(x, self.weight, self.bias,
                           stride=self.stride[0], padding=self.padding[0])
