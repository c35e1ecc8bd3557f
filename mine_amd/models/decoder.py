"""MPI decoder — UNet over encoder taps, conditioned on continuous disparity.

Behavioral contract (ref network/monodepth2/depth_decoder.py:35-148):
  * the BxS disparity values are positionally encoded (21 ch for
    multires=10) and concatenated as constant-per-plane channels onto
    EVERY encoder feature map, which is expanded B -> B*S — the
    "continuous depth" mechanism;
  * a receptive-field neck (maxpool + 1x1 2048->512, maxpool + 3x3
    512->256, up + 3x3 256->256, up + 1x1 256->2048, each BN+LeakyReLU(0.1))
    runs at batch B before the expansion;
  * 5 up-stages of [ConvBlock -> nearest x2 -> skip concat -> ConvBlock]
    with channels [256,128,64,32,16], where ConvBlock = reflection-pad
    3x3 conv + BN + ELU (ref network/monodepth2/layers.py:106-138);
  * per-scale `dispconv` 3x3 -> 4 channels split into sigmoid RGB and
    abs(x)+1e-4 sigma (or sigmoid alpha), with optional sigma dropout;
    outputs {("disp", s): BxSx4xH_sxW_s} for s in scales.
"""
from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

import torch
import torch.nn as nn

import torch.nn.functional as F

from mine_amd.utils.embedder import PositionalEncoder


class ConvBlock(nn.Module):
    """ReflectionPad(1) + 3x3 conv + BN + ELU (ref monodepth2/layers.py:106-138).

    The pad runs on the HIP gather kernels (mine_amd/ops/pad.py) — the
    eager pad pair was ~24% of the train step (profiles/r01_*)."""

    def __init__(self, in_ch: int, out_ch: int):
        super().__init__()
        from mine_amd.ops.bn import FusedBNAct
        from mine_amd.ops.pad import ReflectionPad2d
        self.pad = ReflectionPad2d(1)
        self.conv = nn.Conv2d(in_ch, out_ch, 3)
        self.bn = FusedBNAct(out_ch, act="elu")

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from mine_amd.ops.conv import conv3x3_reflect
        # pad folded into the MFMA conv on the GPU fast path; the eager
        # fallback inside conv3x3_reflect pads explicitly
        return self.bn(conv3x3_reflect(x, self.conv.weight, self.conv.bias))


class SplitConvBlock(nn.Module):
    """ConvBlock over cat(x_dec, base broadcast over S, PE broadcast) —
    WITHOUT materializing the B*S expansion.

    The reference expands every skip map B -> B*S, concatenates the
    positional encoding and convolves the (C_dec + C_enc + E)-channel
    stack at batch B*S (ref depth_decoder.py:103-116). Because the
    expanded channels are identical across the S planes, conv linearity
    factors the block EXACTLY (reflection padding preserves constant
    fields, so the PE contribution is a pure per-(b,s,k) bias):

        out = conv_dec(pad(x_dec))                      # batch B*S
            + conv_skip(pad(base))[broadcast over S]    # batch B  (S x fewer FLOPs)
            + (pe @ sum_k3x3 W_pe^T)[:, :, None, None]  # (B*S, C_out) bias

    This removes the expanded-concat materialization (GBs of copies per
    step), runs the wide encoder channels at batch B instead of B*S, and
    eliminates the tile-hostile 1301/661/341/2069-channel conv shapes the
    profile showed MIOpen collapsing on (profiles/r01_progress.md). The
    single `conv` keeps the reference's (C_out, C_dec+C_enc+E, 3, 3)
    weight shape, so checkpoint keys and shapes are unchanged.
    """

    def __init__(self, dec_ch: int, base_ch: int, pe_ch: int, out_ch: int):
        super().__init__()
        from mine_amd.ops.bn import FusedBNAct
        from mine_amd.ops.pad import ReflectionPad2d
        self.dec_ch, self.base_ch, self.pe_ch = dec_ch, base_ch, pe_ch
        self.pad = ReflectionPad2d(1)
        self.conv = nn.Conv2d(dec_ch + base_ch + pe_ch, out_ch, 3)
        self.bn = FusedBNAct(out_ch, act="elu")

    def forward(self, x_dec, base, pe, B: int, S: int) -> torch.Tensor:
        w = self.conv.weight
        d, b = self.dec_ch, self.base_ch

        # Optional side-stream overlap (docs/NEXT.md #4; OFF by default,
        # enabled by SynthesisTask when training.stream_overlap is set):
        # the batch-B base conv + PE bias are independent of the batch-B*S
        # dec conv until the add, so they can run concurrently.
        side = getattr(self, "side_stream", None)
        use_side = side is not None and base.is_cuda
        main = torch.cuda.current_stream() if use_side else None

        def base_part():
            # base part at batch B (bias lives here; it is S-invariant);
            # reflect pad folded into the igemm kernel's coordinate map
            from mine_amd.ops.conv_general import conv2d_mfma
            y_base = conv2d_mfma(base, w[:, d:d + b].contiguous(),
                                 self.conv.bias, reflect=True)
            K = y_base.shape[1]
            # PE part: conv of a spatially-constant field == channel bias
            w_pe = w[:, d + b:].sum((2, 3))  # (K, E)
            bias_pe = torch.matmul(pe.to(w_pe.dtype), w_pe.t())  # (B*S, K)
            # broadcast-add in NHWC so the (B*S,K,H,W) result is
            # channels_last without a transpose
            yb = y_base.permute(0, 2, 3, 1).unsqueeze(1) \
                + bias_pe.view(B, S, 1, 1, K).to(y_base.dtype)
            return yb.view(B * S, y_base.shape[2], y_base.shape[3], K)

        if use_side:
            side.wait_stream(main)
            with torch.cuda.stream(side):
                yb = base_part()
        else:
            yb = base_part()

        if d:
            from mine_amd.ops.conv import conv3x3_reflect
            y_dec = conv3x3_reflect(x_dec, w[:, :d].contiguous(), None)
            if use_side:
                main.wait_stream(side)
                yb.record_stream(main)
            yb = yb + y_dec.permute(0, 2, 3, 1)
        elif use_side:
            main.wait_stream(side)
            yb.record_stream(main)
        return self.bn(yb.permute(0, 3, 1, 2))


def _neck_conv(in_ch: int, out_ch: int, k: int) -> nn.Sequential:
    from mine_amd.ops.bn import FusedBNAct
    from mine_amd.ops.conv_general import Conv2dMFMA
    return nn.Sequential(
        Conv2dMFMA(in_ch, out_ch, k, stride=1, padding=(k - 1) // 2, bias=False),
        FusedBNAct(out_ch, act="lrelu"),
    )


class Conv3x3Refl(nn.Module):
    """ReflectionPad(1) + plain 3x3 conv (the dispconv head)."""

    def __init__(self, in_ch: int, out_ch: int):
        super().__init__()
        from mine_amd.ops.pad import ReflectionPad2d
        self.pad = ReflectionPad2d(1)
        self.conv = nn.Conv2d(in_ch, out_ch, 3)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from mine_amd.ops.conv import conv3x3_reflect
        return conv3x3_reflect(x, self.conv.weight, self.conv.bias)


class MPIDecoder(nn.Module):
    def __init__(self,
                 num_ch_enc: Sequence[int],
                 pos_encoding_multires: int = 10,
                 use_alpha: bool = False,
                 num_output_channels: int = 4,
                 scales: Sequence[int] = (0, 1, 2, 3),
                 use_skips: bool = True,
                 sigma_dropout_rate: float = 0.0):
        super().__init__()
        self.use_alpha = use_alpha
        self.use_skips = use_skips
        self.scales = list(scales)
        self.num_output_channels = num_output_channels
        self.sigma_dropout_rate = sigma_dropout_rate

        self.embedder = PositionalEncoder(pos_encoding_multires, input_dims=1)
        E = self.embedder.out_dim
        self.E = E

        c_last = num_ch_enc[-1]
        self.downsample = nn.MaxPool2d(3, stride=2, padding=1)
        self.upsample = nn.UpsamplingNearest2d(scale_factor=2)
        self.conv_down1 = _neck_conv(c_last, 512, 1)
        self.conv_down2 = _neck_conv(512, 256, 3)
        self.conv_up1 = _neck_conv(256, 256, 3)
        self.conv_up2 = _neck_conv(256, c_last, 1)

        # channels after PE concat (weight shapes keep the reference's
        # concat layout even though the forward factors the concat away)
        ch_dec = [16, 32, 64, 128, 256]
        self.num_ch_enc = list(num_ch_enc)

        self.upconvs0 = nn.ModuleList()
        self.upconvs1 = nn.ModuleList()
        for i in range(4, -1, -1):
            if i == 4:
                self.upconvs0.append(
                    SplitConvBlock(0, num_ch_enc[-1], E, ch_dec[i]))
            else:
                self.upconvs0.append(ConvBlock(ch_dec[i + 1], ch_dec[i]))
            if use_skips and i > 0:
                self.upconvs1.append(
                    SplitConvBlock(ch_dec[i], num_ch_enc[i - 1], E, ch_dec[i]))
            else:
                self.upconvs1.append(ConvBlock(ch_dec[i], ch_dec[i]))
        self.dispconvs = nn.ModuleDict({
            str(s): Conv3x3Refl(ch_dec[s], num_output_channels) for s in self.scales
        })

    def _up(self, x: torch.Tensor) -> torch.Tensor:
        """Nearest x2 upsample on the HIP stream kernel (torch's
        channels_last nearest kernel measured ~15x off roofline),
        autocast-shielded: CUDA autocast promotes upsample_nearest to
        fp32, which poisoned the downstream convs/pads into fp32
        (observed in profiles); outside autocast the op keeps the
        tensor's own dtype."""
        from mine_amd.ops.upsample import upsample_nearest2x
        if x.is_cuda and torch.is_autocast_enabled():
            with torch.autocast("cuda", enabled=False):
                return upsample_nearest2x(x)
        return upsample_nearest2x(x)

    def _expand_with_pe(self, feat: torch.Tensor, pe: torch.Tensor,
                        B: int, S: int) -> torch.Tensor:
        """feat: BxCxHxW -> (B*S)x(C+E)xHxW with the plane's PE broadcast
        over HxW — the reference's materialized expansion (ref
        depth_decoder.py:103-116). Kept as the oracle for
        SplitConvBlock's factored equivalent; not used in forward."""
        _, C, H, W = feat.shape
        f = feat.unsqueeze(1).expand(B, S, C, H, W).reshape(B * S, C, H, W)
        p = pe.to(feat.dtype).unsqueeze(-1).unsqueeze(-1).expand(B * S, self.E, H, W)
        return torch.cat((f, p), dim=1)

    def forward(self, input_features: List[torch.Tensor],
                disparity: torch.Tensor, packed: bool = False
                ) -> Dict[Tuple[str, int], torch.Tensor]:
        """input_features: 5 taps BxCxHxW; disparity: BxS.

        Returns {("disp", s): BxSx4xH_sxW_s} with rgb=sigmoid, sigma=|x|+1e-4.
        With ``packed=True`` each entry is instead the packed fp32
        (B,S,H_s,W_s,4) MPI the fused renderer consumes, produced by the
        one-pass head kernel (mine_amd/ops/head.py).
        """
        B, S = disparity.shape
        pe = self.embedder(disparity.reshape(B * S, 1))  # (B*S, E)

        # receptive-field neck at batch B
        enc_out = input_features[-1]
        x = self.conv_down1(self.downsample(enc_out))
        x = self.conv_down2(self.downsample(x))
        x = self.conv_up1(self._up(x))
        neck = self.conv_up2(self._up(x))  # BxC_lastxH/32xW/32

        outputs: Dict[Tuple[str, int], torch.Tensor] = {}
        for idx, i in enumerate(range(4, -1, -1)):
            if i == 4:
                # B -> B*S happens HERE, factored: neck at batch B + PE bias
                x = self.upconvs0[idx](None, neck, pe, B, S)
            else:
                x = self.upconvs0[idx](x)
            x = self._up(x)
            if self.use_skips and i > 0:
                skip = input_features[i - 1]  # RAW batch-B tap
                if x.shape[-2:] != skip.shape[-2:]:
                    # odd intermediate sizes (e.g. H/32 == 3): align to the
                    # skip tap (the reference decoder requires power-of-two
                    # -divisible sizes and would crash here)
                    x = F.interpolate(x, size=skip.shape[-2:], mode="nearest")
                x = self.upconvs1[idx](x, skip, pe, B, S)
            else:
                x = self.upconvs1[idx](x)
            if i in self.scales:
                out = self.dispconvs[str(i)](x)
                Hs, Ws = out.shape[-2:]
                if packed and not (self.sigma_dropout_rate > 0.0 and self.training):
                    from mine_amd.ops.head import mpi_head_pack
                    outputs[("disp", i)] = mpi_head_pack(out, B, S,
                                                         self.use_alpha)
                    continue
                mpi = out.float().view(B, S, self.num_output_channels, Hs, Ws)
                rgb = torch.sigmoid(mpi[:, :, 0:3])
                if self.use_alpha:
                    sigma = torch.sigmoid(mpi[:, :, 3:])
                else:
                    sigma = torch.abs(mpi[:, :, 3:]) + 1e-4
                if self.sigma_dropout_rate > 0.0 and self.training:
                    sigma = F.dropout2d(
                        sigma.view(B * S, 1, Hs, Ws), p=self.sigma_dropout_rate
                    ).view(B, S, 1, Hs, Ws)
                mpi = torch.cat((rgb, sigma), dim=2)
                if packed:
                    mpi = mpi.permute(0, 1, 3, 4, 2).contiguous()
                outputs[("disp", i)] = mpi
        return outputs
