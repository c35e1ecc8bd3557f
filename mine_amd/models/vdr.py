"""View-dependent radiance predictor.

Parity component for the reference's
network/monodepth2/view_dependent_radiance_predictor.py (a 4-conv
residual net mapping (rgb, normalized view direction) -> view-dependent
rgb). In the reference it is vestigial: nothing imports it and its
forward returns the input rgb unchanged (ref
view_dependent_radiance_predictor.py:45). Kept here with the same
behavior (and the actual residual path available via
``apply_residual=True``) so the capability surface matches.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class VDRPredictor(nn.Module):
    """(B,S,3,H,W) rgb + (B,3,H,W) unit view dirs -> (B,S,3,H,W) rgb."""

    def __init__(self, hidden: int = 32, apply_residual: bool = False):
        super().__init__()
        self.apply_residual = apply_residual
        self.net = nn.Sequential(
            nn.Conv2d(6, hidden, 3, padding=1),
            nn.ELU(inplace=True),
            nn.Conv2d(hidden, hidden, 3, padding=1),
            nn.ELU(inplace=True),
            nn.Conv2d(hidden, hidden, 3, padding=1),
            nn.ELU(inplace=True),
            nn.Conv2d(hidden, 3, 3, padding=1),
        )

    def forward(self, rgb: torch.Tensor, view_dirs: torch.Tensor) -> torch.Tensor:
        if not self.apply_residual:
            # reference behavior: identity pass-through
            return rgb
        B, S, C, H, W = rgb.shape
        d = view_dirs / (view_dirs.norm(dim=1, keepdim=True) + 1e-8)
        d = d.unsqueeze(1).expand(B, S, 3, H, W)
        x = torch.cat((rgb, d), dim=2).reshape(B * S, 6, H, W)
        res = self.net(x).reshape(B, S, 3, H, W)
        return (rgb + res).clamp(0.0, 1.0)
