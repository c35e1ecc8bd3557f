#!/usr/bin/env python3
"""Novel-view video synthesis from a single image (inference CLI).

CLI contract follows the reference (ref visualizations/image_to_video.py:259-265):

    python visualizations/image_to_video.py \
        --checkpoint_path /ws/checkpoint.pth [--data_path img_or_scene] \
        --output_dir /out [--gpus 0] [--extra_config '{...}'] \
        [--traj circle|straight-line|double-straight-line] \
        [--benchmark] [--num_frames 90]

Reads ``params.yaml`` sitting next to the checkpoint (the checkpoint-dir
contract, ref image_to_video.py:272-278), computes the MPI ONCE from the
source image, then renders a camera trajectory by per-frame homography
warp + composite of the cached MPI (ref image_to_video.py:90-255).

Output: ``frames/%05d.png`` + ``video.mp4`` when ffmpeg is on PATH
(moviepy/cv2 are not in this environment). ``--benchmark`` times the
per-frame novel-view render (sync-bracketed) and prints one JSON line
with the render FPS — the second half of the BASELINE.json headline
metric.
"""
from __future__ import annotations

import argparse
import json
import math
import os
import shutil
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


# ---------------------------------------------------------------------------
# trajectory synthesis (ref image_to_video.py:22-48,156-190)
# ---------------------------------------------------------------------------


def path_planning(kind: str, dx: float, dy: float, dz: float, frames: int):
    """Per-frame camera offsets for the chosen trajectory."""
    t = np.linspace(0.0, 1.0, frames)
    if kind == "straight-line":
        # cubic ease along (dx,dy,dz) and back
        s = np.concatenate([t[: frames // 2], t[: frames - frames // 2][::-1]])
        s = 3 * s ** 2 - 2 * s ** 3
        xs, ys, zs = dx * s, dy * s, dz * s
    elif kind == "double-straight-line":
        # swing: -d .. +d and back
        s = np.sin(2.0 * math.pi * t)
        xs, ys, zs = dx * s, dy * s, dz * np.abs(s)
    elif kind == "circle":
        xs = dx * np.sin(2.0 * math.pi * t)
        ys = dy * np.cos(2.0 * math.pi * t) - dy
        zs = dz * (1.0 - np.cos(2.0 * math.pi * t)) * 0.5
    else:
        raise ValueError(f"unknown trajectory {kind!r}")
    return np.stack([xs, ys, zs], axis=1)


# per-dataset shift ranges (ref image_to_video.py:156-190)
_TRAJ_PRESETS = {
    "realestate10k": dict(dx=0.06, dy=0.03, dz=0.12),
    "llff": dict(dx=0.15, dy=0.05, dz=0.10),
    "flowers": dict(dx=0.20, dy=0.10, dz=0.30),
    "kitti_raw": dict(dx=0.30, dy=0.05, dz=0.50),
    "dtu": dict(dx=0.15, dy=0.10, dz=0.20),
}


def synthesize_intrinsics(H: int, W: int) -> torch.Tensor:
    """90-degree-FoV pinhole K (ref image_to_video.py:192-202)."""
    f = 0.5 * W
    return torch.tensor([[f, 0.0, W * 0.5],
                         [0.0, f, H * 0.5],
                         [0.0, 0.0, 1.0]], dtype=torch.float32)


def disparity_colormap(disp: torch.Tensor) -> np.ndarray:
    """Min-max normalized disparity -> uint8 turbo-ish colormap HxWx3."""
    d = disp.squeeze().float().cpu()
    d = (d - d.min()) / (d.max() - d.min() + 1e-8)
    d = d.numpy()
    r = np.clip(1.5 - np.abs(2.0 * d - 1.5), 0, 1)
    g = np.clip(1.5 - np.abs(2.0 * d - 1.0), 0, 1)
    b = np.clip(1.5 - np.abs(2.0 * d - 0.5), 0, 1)
    return (np.stack([r, g, b], axis=-1) * 255).astype(np.uint8)


# ---------------------------------------------------------------------------
# generator
# ---------------------------------------------------------------------------


class VideoGenerator:
    """Cache the MPI of one source image, render a trajectory
    (ref image_to_video.py:90-255)."""

    def __init__(self, task, config, device):
        self.task = task
        self.config = config
        self.device = device
        self.H = config["data.img_h"]
        self.W = config["data.img_w"]

    @torch.no_grad()
    def infer_mpi(self, src_img: torch.Tensor, K: torch.Tensor = None):
        """One forward pass: MPI + disparity cached for all later frames
        (ref image_to_video.py:112-153)."""
        from mine_amd.data.synthetic import collate_src_tgt
        from mine_amd.ops import render_src_view
        from mine_amd.utils.geometry import inverse_3x3

        if K is None:
            K = synthesize_intrinsics(self.H, self.W)
        K_inv = torch.inverse(K)
        fake_pts = torch.ones(3, self.config["data.visible_point_count"])
        src_item = {"img": src_img, "K": K, "K_inv": K_inv, "xyzs": fake_pts}
        tgt_item = {"img": src_img.clone(), "K": K.clone(),
                    "K_inv": K_inv.clone(), "G_src_tgt": torch.eye(4),
                    "xyzs": fake_pts.clone()}
        items = collate_src_tgt([(src_item, [tgt_item])])

        self.task.set_data(items)
        endpoints = self.task.network_forward()
        mpi = endpoints["mpi_all_src_list"][0]          # (1,S,H,W,4)
        disparity = endpoints["disparity_all_src"]       # (1,S)

        K_inv_dev = inverse_3x3(self.task.K_src)
        _, _, mpi_blend = render_src_view(
            mpi, disparity, K_inv_dev,
            src_img=self.task.src_imgs,
            bg_depth_inf=self.task.bg_depth_inf,
            use_alpha=self.task.use_alpha)

        self.mpi = mpi_blend
        self.disparity = disparity
        self.K = self.task.K_src
        self.K_inv = K_inv_dev

    @torch.no_grad()
    def render_pose(self, offset) -> dict:
        """Render the cached MPI at one camera offset (ref CS4 subpath;
        scale_factor == 1)."""
        G = torch.eye(4, device=self.device).unsqueeze(0)
        G[0, 0, 3], G[0, 1, 3], G[0, 2, 3] = \
            float(offset[0]), float(offset[1]), float(offset[2])
        return self.task.render_novel_view(
            self.mpi, self.disparity, G, self.K_inv, self.K)

    @torch.no_grad()
    def render_video(self, offsets, out_dir: str, save_depth: bool = False):
        from PIL import Image as PILImage
        frames_dir = os.path.join(out_dir, "frames")
        os.makedirs(frames_dir, exist_ok=True)
        for i, off in enumerate(offsets):
            res = self.render_pose(off)
            rgb = res["tgt_imgs_syn"][0].clamp(0, 1)
            arr = (rgb.permute(1, 2, 0).float().cpu().numpy() * 255).astype(np.uint8)
            PILImage.fromarray(arr).save(
                os.path.join(frames_dir, "%05d.png" % i))
            if save_depth:
                PILImage.fromarray(
                    disparity_colormap(res["tgt_disparity_syn"][0])).save(
                    os.path.join(frames_dir, "disp_%05d.png" % i))
        self._encode(frames_dir, os.path.join(out_dir, "video.mp4"))

    @staticmethod
    def _encode(frames_dir: str, out_path: str, fps: int = 30) -> None:
        ffmpeg = shutil.which("ffmpeg")
        if ffmpeg is None:
            print(f"ffmpeg not found; frames left in {frames_dir}")
            return
        subprocess.run(
            [ffmpeg, "-y", "-loglevel", "error", "-framerate", str(fps),
             "-i", os.path.join(frames_dir, "%05d.png"),
             "-pix_fmt", "yuv420p", out_path], check=True)
        print(f"wrote {out_path}")

    @torch.no_grad()
    def benchmark_fps(self, offsets, warmup: int = 10,
                      use_graph: bool = False) -> float:
        """Sync-bracketed per-frame render timing -> FPS.

        With ``use_graph`` the whole per-frame render (homography build,
        closed-form inverses, fused warp+composite) is captured once into
        a hipGraph and replayed per frame with only the pose buffer
        updated — removing the per-frame launch overhead entirely."""
        is_gpu = self.device.type == "cuda"
        if use_graph and is_gpu:
            pose = torch.eye(4, device=self.device).unsqueeze(0).contiguous()

            def render_static():
                return self.task.render_novel_view(
                    self.mpi, self.disparity, pose, self.K_inv, self.K)

            for off in offsets[:warmup]:
                pose[0, 0:3, 3] = torch.as_tensor(off, device=self.device)
                render_static()
            torch.cuda.synchronize()
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                static_out = render_static()  # noqa: F841 — lives in the pool

            poses = torch.zeros(len(offsets), 3, device=self.device)
            poses.copy_(torch.as_tensor(offsets, dtype=torch.float32))
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for i in range(len(offsets)):
                pose[0, 0:3, 3] = poses[i]
                graph.replay()
            torch.cuda.synchronize()
            return len(offsets) / (time.perf_counter() - t0)

        for off in offsets[:warmup]:
            self.render_pose(off)
        if is_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for off in offsets:
            self.render_pose(off)
        if is_gpu:
            torch.cuda.synchronize()
        return len(offsets) / (time.perf_counter() - t0)


# ---------------------------------------------------------------------------


def load_source_image(data_path, H, W):
    from PIL import Image as PILImage
    if data_path and os.path.isfile(data_path):
        with PILImage.open(data_path) as im:
            im = im.convert("RGB").resize((W, H), PILImage.BILINEAR)
            arr = np.asarray(im, dtype=np.float32) / 255.0
        return torch.from_numpy(arr).permute(2, 0, 1).contiguous()
    # no data on disk: deterministic synthetic texture
    from mine_amd.data.synthetic import _smooth_noise_image
    g = torch.Generator().manual_seed(1234)
    return _smooth_noise_image(H, W, g)


def main() -> int:
    p = argparse.ArgumentParser(description="MPI novel-view video synthesis")
    p.add_argument("--checkpoint_path", type=str, default=None,
                   help="checkpoint.pth; params.yaml read from its dir")
    p.add_argument("--data_path", type=str, default=None,
                   help="source image file (synthetic texture if absent)")
    p.add_argument("--output_dir", type=str, default="video_out")
    p.add_argument("--gpus", type=str, default="0")
    p.add_argument("--extra_config", type=str, default="{}")
    p.add_argument("--traj", type=str, default="circle",
                   choices=["circle", "straight-line", "double-straight-line"])
    p.add_argument("--num_frames", type=int, default=90)
    p.add_argument("--benchmark", action="store_true",
                   help="time per-frame novel-view render, print JSON FPS line")
    p.add_argument("--graph", action="store_true",
                   help="capture the per-frame render in a hipGraph and replay")
    p.add_argument("--save_depth", action="store_true")
    args = p.parse_args()

    os.environ.setdefault("CUDA_VISIBLE_DEVICES", args.gpus.split(",")[0])

    from mine_amd.config import RuntimeState, default_config, load_config
    from mine_amd.engine import SynthesisTask

    if args.checkpoint_path:
        params = os.path.join(os.path.dirname(os.path.abspath(args.checkpoint_path)),
                              "params.yaml")
        cfg = load_config(params, args.extra_config)
        cfg = cfg.replace(**{
            "training.pretrained_checkpoint_path": args.checkpoint_path})
    else:
        cfg = default_config(**json.loads(args.extra_config))
    cfg = cfg.replace(**{"data.per_gpu_batch_size": 1, "eval.lpips": False})

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    task = SynthesisTask(cfg, state=RuntimeState(), is_val=True,
                         device=str(device))

    H, W = cfg["data.img_h"], cfg["data.img_w"]
    src_img = load_source_image(args.data_path, H, W)

    gen = VideoGenerator(task, cfg, device)
    gen.infer_mpi(src_img)

    preset = _TRAJ_PRESETS.get(cfg["data.name"], _TRAJ_PRESETS["realestate10k"])
    offsets = path_planning(args.traj, frames=args.num_frames, **preset)

    if args.benchmark:
        fps = gen.benchmark_fps(offsets, use_graph=args.graph)
        print(json.dumps({
            "metric": "novel-view render FPS",
            "value": round(fps, 2), "unit": "frames/sec",
            "n_gpus": 1, "higher_is_better": True,
            "hipgraph": bool(args.graph),
            "config": {"img_h": H, "img_w": W,
                       "n_planes": int(gen.disparity.shape[1]),
                       "dataset": cfg["data.name"]},
        }))
        return 0

    os.makedirs(args.output_dir, exist_ok=True)
    gen.render_video(offsets, args.output_dir, save_depth=args.save_depth)
    return 0


if __name__ == "__main__":
    sys.exit(main())
