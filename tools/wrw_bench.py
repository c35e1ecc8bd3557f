"""Standalone per-shape timing of the conv3x3 wrw kernel vs MIOpen."""
import time
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from mine_amd.ops.backend import get_extension

SHAPES = [  # (N, C, H, W, K) — the flagship decoder's gated convs
    (256, 16, 256, 384, 16),
    (256, 16, 256, 384, 4),
    (256, 32, 128, 192, 32),
    (256, 32, 128, 192, 16),
    (256, 32, 128, 192, 4),
    (256, 64, 64, 96, 64),
    (256, 64, 64, 96, 32),
    (256, 64, 64, 96, 4),
]

def tm(fn, n=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000

ext = get_extension(required=True)
total_hip = total_mi = 0.0
for (N, C, H, W, K) in SHAPES:
    x = torch.randn(N, C, H, W, device="cuda:0", dtype=torch.bfloat16
                    ).contiguous(memory_format=torch.channels_last)
    gy = torch.randn(N, K, H, W, device="cuda:0", dtype=torch.bfloat16
                     ).contiguous(memory_format=torch.channels_last)
    xf = x.permute(0, 2, 3, 1).reshape(-1)
    gf = gy.permute(0, 2, 3, 1).reshape(-1)
    t_hip = tm(lambda: ext.conv3x3_wrw(xf, gf, N, H, W, C, K))
    xp = F.pad(x, (1, 1, 1, 1), mode="reflect")
    w = torch.randn(K, C, 3, 3, device="cuda:0")
    t_mi = tm(lambda: torch.ops.aten.convolution_backward(
        gy, xp, w.to(torch.bfloat16), None, [1, 1], [0, 0], [1, 1],
        False, [0, 0], 1, [False, True, False]))
    gb = (C + K) * 2 * N * H * W / 1e9
    print(f"N{N} C{C} {H}x{W} K{K}: hip {t_hip:7.3f} ms  miopen {t_mi:7.3f} ms"
          f"  min-traffic {gb:.2f} GB -> {gb/t_hip*1000:.0f} GB/s eff")
    total_hip += t_hip
    total_mi += t_mi
print(f"TOTAL hip {total_hip:.2f} ms  miopen {total_mi:.2f} ms")
