#!/usr/bin/env python3
"""Per-shape conv timing at the flagship config.

Collects every Conv2d call shape in the model (forward hooks), then
times each unique shape's forward / backward-data / backward-weight
(bf16, channels_last) separately. Output: a table sorted by total
ms/step — the worklist for hand-written MFMA conv kernels.

    python tools/conv_shapes.py [--height 256 --width 384 --planes 64 --batch 4]
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--height", type=int, default=256)
    p.add_argument("--width", type=int, default=384)
    p.add_argument("--planes", type=int, default=64)
    p.add_argument("--batch", type=int, default=4)
    p.add_argument("--iters", type=int, default=10)
    args = p.parse_args()

    import torch
    import torch.nn.functional as F

    from mine_amd.config import default_config
    from mine_amd.data import SyntheticMPIDataset, collate_src_tgt
    from mine_amd.engine import SynthesisTask

    assert torch.cuda.is_available()
    cfg = default_config(**{
        "data.name": "realestate10k",
        "data.img_h": args.height, "data.img_w": args.width,
        "mpi.num_bins_coarse": args.planes,
        "data.per_gpu_batch_size": args.batch,
        "data.visible_point_count": 256, "lr.decay_steps": [4, 8],
    })
    task = SynthesisTask(cfg, device="cuda:0")

    shapes = {}  # key -> (count, example)

    def hook(mod, inp, out):
        x = inp[0]
        key = (tuple(x.shape), mod.in_channels, mod.out_channels,
               mod.kernel_size, mod.stride, mod.padding, mod.bias is not None)
        shapes[key] = shapes.get(key, 0) + 1

    handles = []
    for m in list(task.backbone.modules()) + list(task.decoder.modules()):
        if isinstance(m, torch.nn.Conv2d):
            handles.append(m.register_forward_hook(hook))

    ds = SyntheticMPIDataset(cfg, length=args.batch)
    items = collate_src_tgt([ds[i] for i in range(args.batch)])
    task.train_step(items)  # one step records every conv call
    for h in handles:
        h.remove()

    def timeit(fn, iters):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1000.0

    rows = []
    for (xshape, cin, cout, k, stride, pad, has_bias), calls in shapes.items():
        x = torch.randn(*xshape, device="cuda:0", dtype=torch.bfloat16
                        ).contiguous(memory_format=torch.channels_last)
        w = torch.randn(cout, cin, *k, device="cuda:0", dtype=torch.bfloat16
                        ).contiguous(memory_format=torch.channels_last)
        y = F.conv2d(x, w, None, stride, pad)
        gy = torch.randn_like(y)

        t_fwd = timeit(lambda: F.conv2d(x, w, None, stride, pad), args.iters)
        t_bwd_d = timeit(lambda: torch.nn.grad.conv2d_input(
            x.shape, w, gy, stride, pad), args.iters)
        t_bwd_w = timeit(lambda: torch.nn.grad.conv2d_weight(
            x, w.shape, gy, stride, pad), args.iters)
        rows.append((calls * (t_fwd + t_bwd_d + t_bwd_w), calls, xshape,
                     cin, cout, k, stride, t_fwd, t_bwd_d, t_bwd_w))

    rows.sort(reverse=True)
    total = sum(r[0] for r in rows)
    print(f"\n== conv cost at {args.height}x{args.width} N={args.planes} "
          f"B={args.batch}: total {total:.1f} ms/step ==")
    print(f"{'ms/step':>8} {'calls':>5} {'input':>24} {'cin':>5} {'cout':>5} "
          f"{'k':>7} {'s':>5} {'fwd':>7} {'bwd_d':>7} {'bwd_w':>7}")
    for tot, calls, xshape, cin, cout, k, stride, tf, td, tw in rows:
        print(f"{tot:8.2f} {calls:5d} {str(list(xshape)):>24} {cin:5d} "
              f"{cout:5d} {str(k):>7} {str(stride):>5} {tf:7.2f} {td:7.2f} "
              f"{tw:7.2f}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
