#!/usr/bin/env python3
"""Generate the committed tiny LLFF fixture (tests/fixtures/llff_tiny):
one toy COLMAP scene, 64x48, 5 train + 2 val views with smooth textured
images — a few KB, enough for tools/evaluate.py to produce real
PSNR/SSIM numbers in CI. Deterministic; re-run to regenerate."""
import math
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from mine_amd.data import colmap


def smooth_img(rng, H, W):
    """Low-frequency RGB texture (PSNR against noise is meaningless)."""
    y, x = np.mgrid[0:H, 0:W].astype(np.float64)
    img = np.zeros((H, W, 3))
    for c in range(3):
        for _ in range(4):
            fx, fy = rng.uniform(0.5, 3.0, 2)
            ph = rng.uniform(0, 2 * np.pi, 2)
            img[..., c] += np.sin(2 * np.pi * fx * x / W + ph[0]) * \
                np.cos(2 * np.pi * fy * y / H + ph[1])
    img = (img - img.min()) / (img.max() - img.min())
    return (img * 255).astype(np.uint8)


def main():
    from PIL import Image as PILImage
    root = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "tests", "fixtures", "llff_tiny")
    W, H, ratio = 64, 48, "7.875"
    n_train, n_val, n_points = 5, 2, 96
    rng = np.random.default_rng(11)
    scene = os.path.join(root, "scene0")
    sparse = os.path.join(scene, "sparse", "0")
    os.makedirs(sparse, exist_ok=True)
    os.makedirs(os.path.join(scene, f"images_{ratio}"), exist_ok=True)
    os.makedirs(os.path.join(scene, f"images_{ratio}_val"), exist_ok=True)

    f = 0.8 * W
    cameras = {1: colmap.Camera(1, "SIMPLE_RADIAL", W, H,
                                np.array([f, W / 2, H / 2, 0.0]))}
    pts_w = np.stack([rng.uniform(-1.5, 1.5, n_points),
                      rng.uniform(-1.2, 1.2, n_points),
                      rng.uniform(4.0, 10.0, n_points)], axis=0)

    images, points3d = {}, {}
    tracks = {pid: [] for pid in range(1, n_points + 1)}
    n_views = n_train + n_val
    for i in range(1, n_views + 1):
        angle = 0.04 * (i - 1)
        R = np.array([[math.cos(angle), 0, math.sin(angle)],
                      [0, 1, 0],
                      [-math.sin(angle), 0, math.cos(angle)]])
        t = np.array([0.08 * (i - 1), 0.0, 0.0])
        q = colmap.rotmat2qvec(R)
        xyz_c = R @ pts_w + t[:, None]
        uv = xyz_c[:2] / xyz_c[2:]
        px, py = f * uv[0] + W / 2, f * uv[1] + H / 2
        vis = (xyz_c[2] > 0.1) & (px >= 0) & (px < W) & (py >= 0) & (py < H)
        pids = np.where(vis)[0] + 1
        xys = np.stack([px[vis], py[vis]], axis=-1)
        name = f"view_{i:03d}.png"
        images[i] = colmap.Image(i, q, t, 1, name, xys,
                                 pids.astype(np.int64))
        for k, pid in enumerate(pids):
            tracks[int(pid)].append((i, k))
        sub = f"images_{ratio}" if i <= n_train else f"images_{ratio}_val"
        PILImage.fromarray(smooth_img(rng, H, W)).save(
            os.path.join(scene, sub, name))

    for pid in range(1, n_points + 1):
        tr = tracks[pid]
        points3d[pid] = colmap.Point3D(
            pid, pts_w[:, pid - 1],
            np.array([128, 128, 128], dtype=np.uint8), 0.5,
            np.array([a for a, _ in tr], dtype=np.int32),
            np.array([b for _, b in tr], dtype=np.int32))
    colmap.write_model(cameras, images, points3d, sparse)
    print("fixture written to", root)


if __name__ == "__main__":
    main()
