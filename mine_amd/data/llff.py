"""LLFF / COLMAP-backed dataset.

Re-implements the reference's NeRFDataset behavior
(ref input_pipelines/llff/nerf_dataset.py):
  * each scene dir under `root` holds a COLMAP sparse model
    (``sparse/0``, .bin) and pre-downsampled images in
    ``images_<ratio>`` (val split: ``images_<ratio>_val``,
    ref nerf_dataset.py:47-53);
  * all images are eager-loaded into RAM at init
    (ref nerf_dataset.py:78-81);
  * per image: ``G_cam_world`` from qvec/tvec (ref nerf_dataset.py:143-148),
    K from the COLMAP camera scaled by the actual downsample ratio
    (ref nerf_dataset.py:150-161), and the image's tracked 3D points
    transformed to its camera frame, keeping positive depths
    (ref nerf_dataset.py:163-194);
  * ``__getitem__`` returns (src_item, [tgt_items]): `supervision_count`
    random same-scene target views (validation: the deterministic next
    neighbor), relative pose ``G_src_tgt = G_src_world @ inv(G_tgt_world)``
    and `visible_points_count` randomly sampled sparse points per view
    (ref nerf_dataset.py:197-234,118-126).

Item schema matches mine_amd.data.synthetic exactly, so the two are
interchangeable in train.py and bench.py.
"""
from __future__ import annotations

import os
import random
from typing import List, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset

from mine_amd.data.colmap import read_model
from mine_amd.data.synthetic import collate_src_tgt


def _load_image(path: str, img_wh: Tuple[int, int]) -> torch.Tensor:
    """Load an image file -> float tensor 3xHxW in [0,1], resized to img_wh."""
    from PIL import Image as PILImage
    with PILImage.open(path) as im:
        im = im.convert("RGB")
        if im.size != tuple(img_wh):
            im = im.resize(tuple(img_wh), PILImage.BILINEAR)
        arr = np.asarray(im, dtype=np.float32) / 255.0
    return torch.from_numpy(arr).permute(2, 0, 1).contiguous()


class _ViewRecord:
    __slots__ = ("img", "K", "K_inv", "G_cam_world", "G_world_cam", "xyz_cam")

    def __init__(self, img, K, G_cam_world, xyz_cam):
        self.img = img
        self.K = K
        self.K_inv = torch.inverse(K)
        self.G_cam_world = G_cam_world
        self.G_world_cam = torch.inverse(G_cam_world)
        self.xyz_cam = xyz_cam  # 3xN visible sparse points, camera frame


class NeRFDataset(Dataset):
    """LLFF-style scenes with COLMAP sparse geometry."""

    def __init__(self, config, logger=None, root: str = None,
                 is_validation: bool = False,
                 img_size: Tuple[int, int] = (512, 384),
                 supervision_count: int = 1,
                 visible_points_count: int = 256,
                 img_pre_downsample_ratio: float = 7.875):
        super().__init__()
        self.is_validation = is_validation
        self.img_wh = tuple(img_size)
        self.supervision_count = int(supervision_count)
        self.visible_points_count = int(visible_points_count)
        self.logger = logger

        ratio_str = ("%g" % img_pre_downsample_ratio)
        subdir = f"images_{ratio_str}" + ("_val" if is_validation else "")

        self.scenes: List[List[_ViewRecord]] = []
        self.index: List[Tuple[int, int]] = []  # flat idx -> (scene, view)

        scene_dirs = sorted(
            d for d in os.listdir(root)
            if os.path.isdir(os.path.join(root, d, "sparse", "0")))
        assert scene_dirs, f"no COLMAP scenes under {root!r}"

        for scene in scene_dirs:
            scene_path = os.path.join(root, scene)
            img_dir = os.path.join(scene_path, subdir)
            if not os.path.isdir(img_dir):
                # fall back to the train image folder for val when no
                # dedicated *_val folder was prepared
                img_dir = os.path.join(scene_path, f"images_{ratio_str}")
            if not os.path.isdir(img_dir):
                continue
            views = self._load_scene(scene_path, img_dir)
            if len(views) < 2:
                continue
            si = len(self.scenes)
            self.scenes.append(views)
            self.index.extend((si, vi) for vi in range(len(views)))

        assert self.index, f"no usable scenes/images under {root!r}"
        if logger is not None:
            logger.info(
                "NeRFDataset(%s): %d scenes, %d views",
                "val" if is_validation else "train",
                len(self.scenes), len(self.index))

    # ------------------------------------------------------------------
    def _load_scene(self, scene_path: str, img_dir: str) -> List[_ViewRecord]:
        cameras, images, points3d = read_model(
            os.path.join(scene_path, "sparse", "0"), ".bin")
        views = []
        W_out, H_out = self.img_wh
        for iid in sorted(images):
            im = images[iid]
            img_path = os.path.join(img_dir, im.name)
            if not os.path.exists(img_path):
                stem = os.path.splitext(im.name)[0]
                cands = [p for p in os.listdir(img_dir)
                         if os.path.splitext(p)[0] == stem]
                if not cands:
                    continue
                img_path = os.path.join(img_dir, cands[0])
            img = _load_image(img_path, self.img_wh)

            cam = cameras[im.camera_id]
            K = cam.intrinsic_matrix().copy()
            # scale from COLMAP's full resolution to the output size
            K[0, :] *= W_out / cam.width
            K[1, :] *= H_out / cam.height
            K = torch.from_numpy(K).float()

            R = torch.from_numpy(im.qvec2rotmat()).float()
            t = torch.from_numpy(im.tvec).float()
            G = torch.eye(4)
            G[:3, :3] = R
            G[:3, 3] = t

            pids = [int(p) for p in im.point3D_ids if p >= 0 and int(p) in points3d]
            if pids:
                xyz_w = np.stack([points3d[p].xyz for p in pids], axis=1)  # 3xN
                xyz_c = R.numpy() @ xyz_w + t.numpy()[:, None]
                keep = xyz_c[2] > 1e-4
                xyz_cam = torch.from_numpy(xyz_c[:, keep]).float()
            else:
                xyz_cam = torch.zeros(3, 0)
            if xyz_cam.shape[1] == 0:
                continue
            views.append(_ViewRecord(img, K, G, xyz_cam))
        return views

    # ------------------------------------------------------------------
    def __len__(self) -> int:
        return len(self.index)

    def _sample_points(self, view: _ViewRecord, rng: random.Random) -> torch.Tensor:
        n = view.xyz_cam.shape[1]
        k = self.visible_points_count
        idx = [rng.randrange(n) for _ in range(k)] if n < k else \
            rng.sample(range(n), k)
        return view.xyz_cam[:, idx]

    def __getitem__(self, idx: int):
        si, vi = self.index[idx]
        views = self.scenes[si]
        rng = random.Random(idx if self.is_validation else None)

        src = views[vi]
        src_item = {
            "img": src.img,
            "K": src.K,
            "K_inv": src.K_inv,
            "xyzs": self._sample_points(src, rng),
        }
        tgt_items = []
        others = [i for i in range(len(views)) if i != vi]
        for j in range(self.supervision_count):
            ti = others[(vi + 1 + j - 1) % len(others)] if self.is_validation \
                else rng.choice(others)
            tgt = views[ti]
            G_src_tgt = src.G_cam_world @ tgt.G_world_cam
            tgt_items.append({
                "img": tgt.img,
                "K": tgt.K,
                "K_inv": tgt.K_inv,
                "G_src_tgt": G_src_tgt,
                "xyzs": self._sample_points(tgt, rng),
            })
        return src_item, tgt_items


NeRFDataset.collate_fn = staticmethod(collate_src_tgt)
