#!/usr/bin/env python3
"""Standalone checkpoint evaluation: PSNR / SSIM / LPIPS / L1 on a val set.

The reference only evaluates from inside the training loop
(ref synthesis_task.py:476-507) and ships a RealEstate10K pair protocol
(ref input_pipelines/realestate10k/test_data_jsons/validation_pairs.json).
This tool evaluates any checkpoint over the configured dataset's
validation split (LLFF/COLMAP scenes, or the synthetic generator when no
data is on disk) and prints one JSON line of averaged metrics.

    python tools/evaluate.py --checkpoint_path /ws/v1/checkpoint.pth \
        [--extra_config '{...}'] [--max_batches 50] [--lpips]
"""
from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--checkpoint_path", type=str, default=None)
    p.add_argument("--extra_config", type=str, default="{}")
    p.add_argument("--max_batches", type=int, default=50)
    p.add_argument("--lpips", action="store_true",
                   help="also compute (uncalibrated without weights) LPIPS")
    args = p.parse_args()

    import torch
    from torch.utils.data import DataLoader

    from mine_amd.config import RuntimeState, default_config, load_config
    from mine_amd.data import get_dataset
    from mine_amd.engine import SynthesisTask

    if args.checkpoint_path:
        params = os.path.join(
            os.path.dirname(os.path.abspath(args.checkpoint_path)), "params.yaml")
        cfg = load_config(params, args.extra_config)
        cfg = cfg.replace(**{
            "training.pretrained_checkpoint_path": args.checkpoint_path})
    else:
        cfg = default_config(**json.loads(args.extra_config))
    if args.lpips:
        cfg = cfg.replace(**{"eval.lpips": True})

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    task = SynthesisTask(cfg, state=RuntimeState(), is_val=True, device=device)

    ds = get_dataset(cfg, None, is_validation=True)
    dl = DataLoader(ds, batch_size=cfg["data.per_gpu_batch_size"],
                    shuffle=False, collate_fn=ds.collate_fn)

    sums, count = {}, 0
    keys = ("psnr_tgt", "loss_ssim_tgt", "lpips_tgt", "loss_rgb_tgt",
            "loss_rgb_src", "loss_ssim_src")
    with torch.no_grad():
        for i, items in enumerate(dl):
            if i >= args.max_batches:
                break
            task.set_data(items)
            loss_dict, _ = task.loss_fcn(is_val=True)
            B = task.src_imgs.shape[0]
            for k in keys:
                sums[k] = sums.get(k, 0.0) + float(loss_dict[k]) * B
            count += B

    out = {k: round(sums[k] / max(count, 1), 5) for k in keys}
    out["ssim_tgt"] = round(1.0 - out.pop("loss_ssim_tgt"), 5)
    out["ssim_src"] = round(1.0 - out.pop("loss_ssim_src"), 5)
    out["n_images"] = count
    out["dataset"] = cfg["data.name"]
    print(json.dumps(out))
    return 0


if __name__ == "__main__":
    sys.exit(main())
