"""Flowers light-field dataset pipeline.

The reference ships only the ASSETS for its (unreleased) flowers
pipeline: `input_pipelines/flowers/cam_params.txt` — one line per
sub-aperture view `<u>_<v> fx fy cx cy <3x4 pose row-major>` with
NORMALIZED intrinsics — and `dataset_list/{train,test}.list` of eslf
(lenslet) image paths (ref SURVEY §2a "Flowers assets"; the reference's
train.py raises NotImplementedError for the dataset itself). This
module implements the missing pipeline over exactly those formats:

  * `read_cam_params(path)` parses the shipped table;
  * `extract_subaperture(eslf, u, v, grid, offset)` slices view (u, v)
    out of a lenslet image (Lytro eslf layout: pixel (y, x) of view
    (u, v) lives at (y*grid + v + offset, x*grid + u + offset));
  * `FlowersDataset` yields the same (src_item, tgt_items) schema as
    the LLFF/synthetic datasets (src + one random same-grid target with
    relative pose). Flowers is a metric dataset (scale factor 1, ref
    synthesis_task.py:213-214), so the sparse-point channel is a
    fabricated placeholder exactly like the synthetic generator's — the
    disparity-point loss runs at lambda 0 for this dataset.
"""
from __future__ import annotations

import os
import random
from typing import Dict, List, Tuple

import numpy as np
import torch

from mine_amd.data.synthetic import collate_src_tgt  # noqa: F401  (re-export)


def read_cam_params(path: str) -> Dict[Tuple[int, int], dict]:
    """Parse cam_params.txt: {(u, v): {"K_norm": 3x3, "G_cam_world": 4x4}}.

    Intrinsics are normalized by image size (fx etc. in units of W/H);
    the pose rows are a 3x4 world->camera matrix.
    """
    out = {}
    with open(path, "r") as f:
        for line in f:
            parts = line.split()
            if len(parts) != 17:
                continue
            u, v = (int(t) for t in parts[0].split("_"))
            fx, fy, cx, cy = (float(t) for t in parts[1:5])
            P = np.array([float(t) for t in parts[5:17]],
                         dtype=np.float64).reshape(3, 4)
            K = np.array([[fx, 0.0, cx], [0.0, fy, cy], [0.0, 0.0, 1.0]])
            G = np.eye(4)
            G[:3, :] = P
            out[(u, v)] = {"K_norm": K, "G_cam_world": G}
    return out


def extract_subaperture(eslf: np.ndarray, u: int, v: int, grid: int = 14,
                        offset: int = 3) -> np.ndarray:
    """Slice sub-aperture view (u, v) out of an eslf lenslet image
    (H*grid, W*grid, 3) -> (H, W, 3). offset centers the used views in
    the lenslet (the shipped table covers an 8x8 block of a 14x14
    Lytro grid)."""
    return eslf[v + offset::grid, u + offset::grid]


class FlowersDataset(torch.utils.data.Dataset):
    """Light-field training items over the reference's shipped formats.

    root/
      cam_params.txt
      dataset_list/train.list  (or test.list)
      <paths from the list, e.g. imgs/IMG_xxx_eslf.png>
    """

    def __init__(self, config, logger=None, root: str = None,
                 is_validation: bool = False, grid: int = 14,
                 offset: int = 3):
        self.config = config
        self.root = root or config["data.training_set_path"]
        self.is_validation = is_validation
        self.grid = grid
        self.offset = offset
        self.img_w = int(config["data.img_w"])
        self.img_h = int(config["data.img_h"])
        self.n_pts = int(config["data.visible_point_count"])
        self.disp_start = float(config["mpi.disparity_start"])
        self.disp_end = float(config["mpi.disparity_end"])

        self.cams = read_cam_params(os.path.join(self.root, "cam_params.txt"))
        self.views: List[Tuple[int, int]] = sorted(self.cams.keys())
        lst = os.path.join(self.root, "dataset_list",
                           "test.list" if is_validation else "train.list")
        with open(lst, "r") as f:
            self.files = [ln.strip() for ln in f if ln.strip()]
        if logger:
            logger.info("FlowersDataset: %d eslf images, %d views/grid (%s)",
                        len(self.files), len(self.views),
                        "val" if is_validation else "train")
        self.collate_fn = collate_src_tgt

    def __len__(self) -> int:
        return len(self.files)

    def _load_view(self, eslf: np.ndarray, uv: Tuple[int, int]):
        from PIL import Image as PILImage
        sub = extract_subaperture(eslf, uv[0], uv[1], self.grid, self.offset)
        img = PILImage.fromarray(sub).resize((self.img_w, self.img_h),
                                             PILImage.BILINEAR)
        arr = np.asarray(img, dtype=np.float32) / 255.0
        t = torch.from_numpy(arr).permute(2, 0, 1).contiguous()
        cam = self.cams[uv]
        K = cam["K_norm"].copy()
        K[0] *= self.img_w   # de-normalize fx, cx row
        K[1] *= self.img_h
        return t, torch.from_numpy(K).float(), \
            torch.from_numpy(cam["G_cam_world"]).float()

    def _fake_points(self, rng: random.Random) -> torch.Tensor:
        """Placeholder sparse points (disp-point loss is lambda=0 for
        metric datasets; shape parity with LLFF items)."""
        g = torch.Generator().manual_seed(rng.randrange(1 << 31))
        disp = torch.rand(self.n_pts, generator=g) * \
            (self.disp_start - self.disp_end) * 0.9 + self.disp_end * 1.1
        z = torch.reciprocal(disp)
        x = (torch.rand(self.n_pts, generator=g) - 0.5) * z
        y = (torch.rand(self.n_pts, generator=g) - 0.5) * z
        return torch.stack((x, y, z), dim=0)

    def __getitem__(self, idx: int):
        from PIL import Image as PILImage
        rng = random.Random(idx if self.is_validation else None)
        path = os.path.join(self.root, self.files[idx])
        eslf = np.asarray(PILImage.open(path).convert("RGB"))

        src_uv = self.views[len(self.views) // 2] if self.is_validation \
            else rng.choice(self.views)
        others = [uv for uv in self.views if uv != src_uv]
        tgt_uv = others[idx % len(others)] if self.is_validation \
            else rng.choice(others)

        s_img, s_K, s_G = self._load_view(eslf, src_uv)
        t_img, t_K, t_G = self._load_view(eslf, tgt_uv)
        G_src_tgt = s_G @ torch.inverse(t_G)

        src_item = {"img": s_img, "K": s_K, "K_inv": torch.inverse(s_K),
                    "xyzs": self._fake_points(rng)}
        tgt_item = {"img": t_img, "K": t_K, "K_inv": torch.inverse(t_K),
                    "G_src_tgt": G_src_tgt, "xyzs": self._fake_points(rng)}
        return src_item, [tgt_item]
