"""monodepth2-style geometry/loss helpers (capability parity).

The reference vendors monodepth2's layer zoo
(ref network/monodepth2/layers.py) of which the training path uses only
ConvBlock/Conv3x3/upsample (in mine_amd.models.decoder); the remaining
helpers — depth/disparity conversion, axis-angle pose composition,
backprojection/projection, smoothness, pooled SSIM, depth metrics — are
carried for downstream use. Re-implemented here against the same
contracts (ref network/monodepth2/layers.py:16-271).
"""
from __future__ import annotations

from typing import Dict, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


def disp_to_depth(disp: torch.Tensor, min_depth: float, max_depth: float
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Sigmoid disparity in [0,1] -> (scaled disparity, depth)
    (ref monodepth2/layers.py:16-25)."""
    min_disp = 1.0 / max_depth
    max_disp = 1.0 / min_depth
    scaled = min_disp + (max_disp - min_disp) * disp
    return scaled, 1.0 / scaled


def rot_from_axisangle(vec: torch.Tensor) -> torch.Tensor:
    """Axis-angle (B,1,3) -> 4x4 rotation (Rodrigues;
    ref monodepth2/layers.py:64-103)."""
    angle = torch.norm(vec, 2, 2, True)
    axis = vec / (angle + 1e-7)

    ca, sa = torch.cos(angle), torch.sin(angle)
    C = 1.0 - ca
    x = axis[..., 0].unsqueeze(1)
    y = axis[..., 1].unsqueeze(1)
    z = axis[..., 2].unsqueeze(1)

    rot = torch.zeros((vec.shape[0], 4, 4), device=vec.device, dtype=vec.dtype)
    rot[:, 0, 0] = torch.squeeze(x * x * C + ca)
    rot[:, 0, 1] = torch.squeeze(x * y * C - z * sa)
    rot[:, 0, 2] = torch.squeeze(z * x * C + y * sa)
    rot[:, 1, 0] = torch.squeeze(x * y * C + z * sa)
    rot[:, 1, 1] = torch.squeeze(y * y * C + ca)
    rot[:, 1, 2] = torch.squeeze(y * z * C - x * sa)
    rot[:, 2, 0] = torch.squeeze(z * x * C - y * sa)
    rot[:, 2, 1] = torch.squeeze(y * z * C + x * sa)
    rot[:, 2, 2] = torch.squeeze(z * z * C + ca)
    rot[:, 3, 3] = 1.0
    return rot


def transformation_from_parameters(axisangle: torch.Tensor,
                                   translation: torch.Tensor,
                                   invert: bool = False) -> torch.Tensor:
    """(axisangle (B,1,3), translation (B,1,3)) -> 4x4 transform
    (ref monodepth2/layers.py:28-61)."""
    R = rot_from_axisangle(axisangle)
    t = translation.clone()
    if invert:
        R = R.transpose(1, 2)
        t = t * -1
    T = torch.zeros_like(R)
    T[:, 0, 0] = T[:, 1, 1] = T[:, 2, 2] = T[:, 3, 3] = 1.0
    T[:, :3, 3] = t.squeeze(1)
    # invert: R^T @ T(-t) == (T(t) @ R)^-1
    return torch.matmul(R, T) if invert else torch.matmul(T, R)


class BackprojectDepth(nn.Module):
    """Depth map -> homogeneous point cloud (ref monodepth2/layers.py:141-170)."""

    def __init__(self, batch_size: int, height: int, width: int):
        super().__init__()
        self.batch_size = batch_size
        self.height = height
        self.width = width
        yy, xx = torch.meshgrid(torch.arange(height, dtype=torch.float32),
                                torch.arange(width, dtype=torch.float32),
                                indexing="ij")
        pix = torch.stack((xx.reshape(-1), yy.reshape(-1),
                           torch.ones(height * width)), 0)  # 3xHW
        self.register_buffer("pix_coords",
                             pix.unsqueeze(0).repeat(batch_size, 1, 1),
                             persistent=False)
        self.register_buffer("ones",
                             torch.ones(batch_size, 1, height * width),
                             persistent=False)

    def forward(self, depth: torch.Tensor, inv_K: torch.Tensor) -> torch.Tensor:
        cam = torch.matmul(inv_K[:, :3, :3], self.pix_coords)
        cam = depth.view(self.batch_size, 1, -1) * cam
        return torch.cat((cam, self.ones), 1)  # Bx4xHW


class Project3D(nn.Module):
    """Point cloud -> normalized grid_sample coords
    (ref monodepth2/layers.py:173-195)."""

    def __init__(self, batch_size: int, height: int, width: int,
                 eps: float = 1e-7):
        super().__init__()
        self.batch_size = batch_size
        self.height = height
        self.width = width
        self.eps = eps

    def forward(self, points: torch.Tensor, K: torch.Tensor,
                T: torch.Tensor) -> torch.Tensor:
        P = torch.matmul(K, T)[:, :3, :]
        cam = torch.matmul(P, points)
        pix = cam[:, :2, :] / (cam[:, 2, :].unsqueeze(1) + self.eps)
        pix = pix.view(self.batch_size, 2, self.height, self.width)
        pix = pix.permute(0, 2, 3, 1)
        pix[..., 0] /= self.width - 1
        pix[..., 1] /= self.height - 1
        return (pix - 0.5) * 2


def get_smooth_loss(disp: torch.Tensor, img: torch.Tensor) -> torch.Tensor:
    """Edge-aware first-order smoothness (ref monodepth2/layers.py:204-217).
    Same math as mine_amd.ops.losses.edge_aware_loss_v2 without the
    mean-normalization."""
    gdx = torch.abs(disp[:, :, :, :-1] - disp[:, :, :, 1:])
    gdy = torch.abs(disp[:, :, :-1, :] - disp[:, :, 1:, :])
    gix = torch.mean(torch.abs(img[:, :, :, :-1] - img[:, :, :, 1:]), 1, True)
    giy = torch.mean(torch.abs(img[:, :, :-1, :] - img[:, :, 1:, :]), 1, True)
    return (gdx * torch.exp(-gix)).mean() + (gdy * torch.exp(-giy)).mean()


class PooledSSIM(nn.Module):
    """monodepth2's 3x3 avg-pool SSIM (ref monodepth2/layers.py:220-250);
    distinct from the 11x11 Gaussian SSIM loss (mine_amd.ops.ssim)."""

    def __init__(self):
        super().__init__()
        self.mu_pool = nn.AvgPool2d(3, 1)
        self.refl = nn.ReflectionPad2d(1)
        self.C1 = 0.01 ** 2
        self.C2 = 0.03 ** 2

    def forward(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        x, y = self.refl(x), self.refl(y)
        mu_x, mu_y = self.mu_pool(x), self.mu_pool(y)
        sig_x = self.mu_pool(x ** 2) - mu_x ** 2
        sig_y = self.mu_pool(y ** 2) - mu_y ** 2
        sig_xy = self.mu_pool(x * y) - mu_x * mu_y
        n = (2 * mu_x * mu_y + self.C1) * (2 * sig_xy + self.C2)
        d = (mu_x ** 2 + mu_y ** 2 + self.C1) * (sig_x + sig_y + self.C2)
        return torch.clamp((1 - n / d) / 2, 0, 1)


def compute_depth_errors(gt: torch.Tensor, pred: torch.Tensor
                         ) -> Dict[str, torch.Tensor]:
    """Standard depth metrics (ref monodepth2/layers.py:253-271)."""
    thresh = torch.max(gt / pred, pred / gt)
    out = {
        "a1": (thresh < 1.25).float().mean(),
        "a2": (thresh < 1.25 ** 2).float().mean(),
        "a3": (thresh < 1.25 ** 3).float().mean(),
        "abs_rel": (torch.abs(gt - pred) / gt).mean(),
        "sq_rel": ((gt - pred) ** 2 / gt).mean(),
        "rmse": torch.sqrt(((gt - pred) ** 2).mean()),
        "rmse_log": torch.sqrt(((torch.log(gt) - torch.log(pred)) ** 2).mean()),
    }
    return out
