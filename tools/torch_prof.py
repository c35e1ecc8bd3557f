#!/usr/bin/env python3
"""ATen-op-level profile of the flagship train step (torch.profiler).

Complements rocprofv3 (which aggregates by kernel symbol): this groups
GPU time by the PyTorch op that launched each kernel, so eager
elementwise soup shows up with its op name and shapes.

    python tools/torch_prof.py [--steps 3] [--by-shape]
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--height", type=int, default=256)
    p.add_argument("--width", type=int, default=384)
    p.add_argument("--planes", type=int, default=64)
    p.add_argument("--batch", type=int, default=4)
    p.add_argument("--by-shape", action="store_true")
    p.add_argument("--row-limit", type=int, default=45)
    args = p.parse_args()

    import torch
    from torch.profiler import ProfilerActivity, profile

    from mine_amd.config import default_config
    from mine_amd.data import SyntheticMPIDataset, collate_src_tgt
    from mine_amd.engine import SynthesisTask

    cfg = default_config(**{
        "data.name": "realestate10k",
        "data.img_h": args.height, "data.img_w": args.width,
        "mpi.num_bins_coarse": args.planes,
        "data.per_gpu_batch_size": args.batch,
        "data.visible_point_count": 256, "lr.decay_steps": [4, 8],
    })
    task = SynthesisTask(cfg, device="cuda:0" if torch.cuda.is_available()
                         else "cpu")
    ds = SyntheticMPIDataset(cfg, length=args.batch)
    items = collate_src_tgt([ds[i] for i in range(args.batch)])

    for _ in range(3):
        task.train_step(items)
    torch.cuda.synchronize()

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=args.by_shape) as prof:
        for _ in range(args.steps):
            task.train_step(items)
        torch.cuda.synchronize()

    key = prof.key_averages(group_by_input_shape=args.by_shape)
    print(key.table(sort_by="cuda_time_total", row_limit=args.row_limit,
                    max_name_column_width=55, max_shapes_column_width=60))
    return 0


if __name__ == "__main__":
    sys.exit(main())
