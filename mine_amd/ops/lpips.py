"""LPIPS perceptual distance (rank-0 eval metric).

The reference depends on the external ``lpips`` package (VGG variant,
ref synthesis_task.py:91-92,342-344). This environment has no network
and no torchvision, so the VGG-16 feature tower and the linear
calibration heads are implemented here directly; calibrated weights can
be loaded from a local state-dict when available
(``eval.lpips_weights`` config key / ``weights_path`` argument).

Without calibration weights the module still computes a deterministic
perceptual distance (unit linear heads over ImageNet-style normalized
features of the randomly-initialized tower, fixed seed) — usable as a
relative eval-tracking metric, clearly NOT the published calibrated
LPIPS. `self.calibrated` records which one you got.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

# VGG-16 feature config (conv layer channel plan, 'M' = maxpool).
_VGG16 = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
          512, 512, 512, "M", 512, 512, 512]
# Feature taps after the ReLU of these conv indices (relu1_2 .. relu5_3).
_TAPS = (1, 3, 6, 9, 12)


class _VGG16Features(nn.Module):
    def __init__(self):
        super().__init__()
        layers: List[nn.Module] = []
        c_in = 3
        for v in _VGG16:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers.append(nn.Conv2d(c_in, v, 3, padding=1))
                layers.append(nn.ReLU(inplace=True))
                c_in = v
        self.features = nn.Sequential(*layers)
        # Conv indices (in self.features) after whose ReLU we tap.
        conv_idx = [i for i, m in enumerate(self.features)
                    if isinstance(m, nn.Conv2d)]
        self._tap_after = {conv_idx[t] + 1 for t in _TAPS}

    def forward(self, x: torch.Tensor) -> List[torch.Tensor]:
        taps = []
        for i, m in enumerate(self.features):
            x = m(x)
            if i in self._tap_after:
                taps.append(x)
        return taps


def _normalize_tensor(t: torch.Tensor, eps: float = 1e-10) -> torch.Tensor:
    norm = torch.sqrt(torch.sum(t * t, dim=1, keepdim=True))
    return t / (norm + eps)


class LPIPS(nn.Module):
    """Learned perceptual image patch similarity, VGG backbone.

    Input images in [0, 1], Bx3xHxW. Returns Bx1x1x1 distances.
    """

    CHANNELS = (64, 128, 256, 512, 512)

    def __init__(self, weights_path: Optional[str] = None):
        super().__init__()
        # input scaling (lpips convention: [-1,1] input, then shift/scale)
        self.register_buffer("shift",
                             torch.tensor([-.030, -.088, -.188]).view(1, 3, 1, 1))
        self.register_buffer("scale",
                             torch.tensor([.458, .448, .450]).view(1, 3, 1, 1))
        with torch.random.fork_rng():
            torch.manual_seed(20210829)  # deterministic uncalibrated tower
            self.net = _VGG16Features()
            self.lins = nn.ModuleList(
                [nn.Conv2d(c, 1, 1, bias=False) for c in self.CHANNELS])
            for lin in self.lins:
                nn.init.constant_(lin.weight, 1.0 / lin.weight.shape[1])
        self.calibrated = False
        if weights_path:
            sd = torch.load(weights_path, map_location="cpu")
            self.load_state_dict(sd, strict=False)
            self.calibrated = True
        for p in self.parameters():
            p.requires_grad_(False)

    def forward(self, img0: torch.Tensor, img1: torch.Tensor) -> torch.Tensor:
        # [0,1] -> [-1,1] -> lpips normalization
        x0 = (2.0 * img0 - 1.0 - self.shift) / self.scale
        x1 = (2.0 * img1 - 1.0 - self.shift) / self.scale
        taps0, taps1 = self.net(x0), self.net(x1)
        val = 0.0
        for t0, t1, lin in zip(taps0, taps1, self.lins):
            d = (_normalize_tensor(t0) - _normalize_tensor(t1)) ** 2
            val = val + lin(d).mean(dim=(2, 3), keepdim=True)
        return val
