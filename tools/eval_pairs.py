#!/usr/bin/env python3
"""RealEstate10K-protocol pair evaluation.

Consumes the reference's validation-pair JSONL (ref input_pipelines/
realestate10k/test_data_jsons/validation_pairs.json: one object per
line with ``src_img_obj`` and ``tgt_img_obj_<N>_frames`` entries, each
carrying normalized [fx fy cx cy] intrinsics, a 3x4 world-to-camera
pose and a ``frame_ts`` naming the image file). For every pair the MPI
is predicted from the source frame, rendered into the target camera,
and PSNR/SSIM (optionally LPIPS) are averaged per frame-separation.

    python tools/eval_pairs.py --pairs validation_pairs.json \
        --data_root /data/realestate10k/frames \
        --checkpoint_path /ws/v1/checkpoint.pth [--separations 5,10]

Images are looked up as <data_root>/<sequence_id>/<frame_ts>.<ext>.
"""
from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def pose_to_G(vals) -> torch.Tensor:
    """3x4 row-major world-to-camera -> 4x4 G_cam_world."""
    G = torch.eye(4)
    G[:3, :4] = torch.tensor(vals, dtype=torch.float32).view(3, 4)
    return G


def intrinsics_to_K(vals, W: int, H: int) -> torch.Tensor:
    fx, fy, cx, cy = vals
    return torch.tensor([[fx * W, 0.0, cx * W],
                         [0.0, fy * H, cy * H],
                         [0.0, 0.0, 1.0]], dtype=torch.float32)


def find_image(root: str, seq: str, ts: str):
    base = os.path.join(root, seq, str(ts))
    for ext in (".png", ".jpg", ".jpeg"):
        if os.path.exists(base + ext):
            return base + ext
    return None


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--pairs", required=True, help="validation pairs JSONL")
    p.add_argument("--data_root", required=True)
    p.add_argument("--checkpoint_path", type=str, default=None)
    p.add_argument("--extra_config", type=str, default="{}")
    p.add_argument("--separations", type=str, default="5,10")
    p.add_argument("--max_pairs", type=int, default=200)
    p.add_argument("--lpips", action="store_true")
    args = p.parse_args()

    from mine_amd.config import RuntimeState, default_config, load_config
    from mine_amd.data.llff import _load_image
    from mine_amd.engine import SynthesisTask
    from mine_amd.ops import psnr, ssim
    from mine_amd.utils.geometry import inverse_rigid_4x4
    from visualizations.image_to_video import VideoGenerator

    if args.checkpoint_path:
        params = os.path.join(
            os.path.dirname(os.path.abspath(args.checkpoint_path)), "params.yaml")
        cfg = load_config(params, args.extra_config)
        cfg = cfg.replace(**{
            "training.pretrained_checkpoint_path": args.checkpoint_path})
    else:
        cfg = default_config(**json.loads(args.extra_config))
    cfg = cfg.replace(**{"data.per_gpu_batch_size": 1})

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    task = SynthesisTask(cfg, state=RuntimeState(), is_val=True,
                         device=str(device))
    gen = VideoGenerator(task, cfg, device)
    H, W = cfg["data.img_h"], cfg["data.img_w"]
    seps = [int(s) for s in args.separations.split(",")]
    lpips_model = None
    if args.lpips:
        from mine_amd.ops.lpips import LPIPS
        lpips_model = LPIPS().to(device).eval()

    sums = {s: {"psnr": 0.0, "ssim": 0.0, "lpips": 0.0, "n": 0} for s in seps}
    n_lines = 0
    with open(args.pairs) as f, torch.no_grad():
        for line in f:
            if n_lines >= args.max_pairs:
                break
            entry = json.loads(line)
            seq = entry["sequence_id"]
            src_o = entry["src_img_obj"]
            src_path = find_image(args.data_root, seq, src_o["frame_ts"])
            if src_path is None:
                continue
            n_lines += 1
            src_img = _load_image(src_path, (W, H))
            K_src = intrinsics_to_K(src_o["camera_intrinsics"], W, H)
            G_src = pose_to_G(src_o["camera_pose"])
            gen.infer_mpi(src_img, K_src)

            for s in seps:
                tgt_o = entry.get(f"tgt_img_obj_{s}_frames")
                if tgt_o is None:
                    continue
                tgt_path = find_image(args.data_root, seq, tgt_o["frame_ts"])
                if tgt_path is None:
                    continue
                tgt_img = _load_image(tgt_path, (W, H)).unsqueeze(0).to(device)
                G_tgt = pose_to_G(tgt_o["camera_pose"])
                # G_tgt_src = G_tgt_world @ inv(G_src_world)
                G_tgt_src = (G_tgt @ inverse_rigid_4x4(
                    G_src.unsqueeze(0))[0]).unsqueeze(0).to(device)
                res = task.render_novel_view(gen.mpi, gen.disparity,
                                             G_tgt_src, gen.K_inv, gen.K)
                syn = res["tgt_imgs_syn"].clamp(0, 1)
                sums[s]["psnr"] += float(psnr(syn, tgt_img))
                sums[s]["ssim"] += float(ssim(syn, tgt_img))
                if lpips_model is not None:
                    sums[s]["lpips"] += float(lpips_model(syn, tgt_img).mean())
                sums[s]["n"] += 1

    out = {}
    for s in seps:
        n = max(sums[s]["n"], 1)
        out[f"{s}_frames"] = {
            "psnr": round(sums[s]["psnr"] / n, 4),
            "ssim": round(sums[s]["ssim"] / n, 4),
            **({"lpips": round(sums[s]["lpips"] / n, 4)} if lpips_model else {}),
            "n_pairs": sums[s]["n"],
        }
    print(json.dumps(out))
    return 0


if __name__ == "__main__":
    sys.exit(main())
