"""Synthetic scene dataset — geometry-consistent random data.

Produces items with exactly the schema of the LLFF pipeline
(ref input_pipelines/llff/nerf_dataset.py:197-234 + _collate_fn):
    src: {img Bx3xHxW, K, K_inv Bx3x3, xyzs Bx3xN_pt}
    tgt: {img BxLx3xHxW, G_src_tgt BxLx4x4, K/K_inv BxLx3x3, xyzs BxLx3xN_pt}

The images are band-limited random textures, the poses small random
rigid motions, and the sparse "COLMAP" points are sampled on pixel rays
at depths inside the configured disparity range so every loss term
(including the log-disparity terms) is well-defined. Used for
benchmarking (no-network environments) and CPU plumbing tests.
"""
from __future__ import annotations

import math
from typing import Dict, Tuple

import torch
from torch.utils.data import Dataset


def _camera_intrinsics(H: int, W: int, fov_deg: float = 53.13) -> torch.Tensor:
    fx = W * 0.5 / math.tan(math.radians(fov_deg) * 0.5)
    K = torch.tensor([[fx, 0.0, W * 0.5],
                      [0.0, fx, H * 0.5],
                      [0.0, 0.0, 1.0]], dtype=torch.float32)
    return K


def _smooth_noise_image(H: int, W: int, g: torch.Generator) -> torch.Tensor:
    """Band-limited random RGB in [0,1]: upsampled low-res noise + detail."""
    base = torch.rand((3, max(H // 8, 1), max(W // 8, 1)), generator=g)
    img = torch.nn.functional.interpolate(base.unsqueeze(0), size=(H, W),
                                          mode="bilinear", align_corners=False)[0]
    img = 0.8 * img + 0.2 * torch.rand((3, H, W), generator=g)
    return img.clamp(0.0, 1.0)


def _random_rigid(g: torch.Generator, rot_scale: float = 0.05,
                  trans_scale: float = 0.10) -> torch.Tensor:
    """Small random rigid transform (axis-angle via Rodrigues)."""
    axis = torch.randn(3, generator=g)
    axis = axis / (axis.norm() + 1e-8)
    angle = rot_scale * torch.randn(1, generator=g).clamp(-2, 2)
    K = torch.tensor([[0.0, -axis[2], axis[1]],
                      [axis[2], 0.0, -axis[0]],
                      [-axis[1], axis[0], 0.0]])
    R = torch.eye(3) + math.sin(angle) * K + (1 - math.cos(angle)) * (K @ K)
    t = trans_scale * torch.randn(3, generator=g).clamp(-2, 2)
    G = torch.eye(4)
    G[:3, :3] = R
    G[:3, 3] = t
    return G


class SyntheticMPIDataset(Dataset):
    def __init__(self, config, is_validation: bool = False, length: int = None):
        self.H = config["data.img_h"]
        self.W = config["data.img_w"]
        self.n_pt = config["data.visible_point_count"]
        self.L = config["data.num_tgt_views"]
        self.length = length if length is not None else \
            int(config.get("data.synthetic_length", 512))
        if is_validation:
            self.length = min(self.length, 8)
        self.seed_base = 777 if is_validation else 0
        # keep point depths inside the representable disparity range
        d_start = float(config["mpi.disparity_start"])
        d_end = float(config["mpi.disparity_end"])
        self.depth_min = 1.0 / d_start * 1.2
        self.depth_max = min(1.0 / d_end * 0.8, self.depth_min * 50.0)

    def __len__(self) -> int:
        return self.length

    def _sample_points(self, K_inv: torch.Tensor, g: torch.Generator) -> torch.Tensor:
        px = torch.rand(self.n_pt, generator=g) * (self.W - 1)
        py = torch.rand(self.n_pt, generator=g) * (self.H - 1)
        z = self.depth_min * (self.depth_max / self.depth_min) ** \
            torch.rand(self.n_pt, generator=g)
        p = torch.stack((px, py, torch.ones_like(px)), dim=0)  # 3xN
        return (K_inv @ p) * z.unsqueeze(0)

    def __getitem__(self, idx: int) -> Tuple[Dict, list]:
        g = torch.Generator().manual_seed(self.seed_base * 1000003 + idx)
        K = _camera_intrinsics(self.H, self.W)
        K_inv = torch.inverse(K)

        src = {
            "img": _smooth_noise_image(self.H, self.W, g),
            "K": K,
            "K_inv": K_inv,
            "xyzs": self._sample_points(K_inv, g),
        }
        tgts = []
        for _ in range(self.L):
            tgts.append({
                "img": _smooth_noise_image(self.H, self.W, g),
                "K": K.clone(),
                "K_inv": K_inv.clone(),
                "G_src_tgt": _random_rigid(g),
                "xyzs": self._sample_points(K_inv, g),
            })
        return src, tgts


def collate_src_tgt(batch):
    """Stack src dicts and the per-sample tgt LISTS into BxLx... tensors
    (ref input_pipelines/llff/nerf_dataset.py:15-30)."""
    src = {}
    for key in batch[0][0]:
        src[key] = torch.stack([item[0][key] for item in batch], dim=0)
    tgt = {}
    for key in batch[0][1][0]:
        tgt[key] = torch.stack(
            [torch.stack([t[key] for t in item[1]], dim=0) for item in batch], dim=0)
    return src, tgt


SyntheticMPIDataset.collate_fn = staticmethod(collate_src_tgt)
