"""Standalone per-shape timing of the MFMA conv3x3 fwd + bwd-data."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from mine_amd.ops.conv import conv3x3_reflect, conv3x3_bwd_data

SHAPES = [
    (256, 16, 256, 384, 16),
    (256, 16, 256, 384, 4),
    (256, 32, 128, 192, 32),
    (256, 32, 128, 192, 16),
    (256, 64, 64, 96, 64),
    (256, 64, 64, 96, 32),
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

tot_f = tot_b = tot_mf = tot_mb = 0.0
for (N, C, H, W, K) in SHAPES:
    x = torch.randn(N, C, H, W, device="cuda:0", dtype=torch.bfloat16
                    ).contiguous(memory_format=torch.channels_last)
    w = torch.randn(K, C, 3, 3, device="cuda:0") * 0.2
    gy = torch.randn(N, K, H, W, device="cuda:0", dtype=torch.bfloat16
                     ).contiguous(memory_format=torch.channels_last)
    with torch.no_grad():
        t_f = tm(lambda: conv3x3_reflect(x, w, None))
        t_b = tm(lambda: conv3x3_bwd_data(gy, w))
        xp = F.pad(x, (1, 1, 1, 1), mode="reflect")
        wb = w.to(torch.bfloat16)
        t_mf = tm(lambda: F.conv2d(xp, wb))
        t_mb = tm(lambda: torch.ops.aten.convolution_backward(
            gy, xp, wb, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
            [True, False, False]))
    gb_f = (C + K) * 2 * N * H * W / 1e9
    print(f"N{N} C{C} {H}x{W} K{K}: fwd {t_f:7.3f} (mi {t_mf:7.3f})  "
          f"bwdD {t_b:7.3f} (mi {t_mb:7.3f})  traffic {gb_f:.2f} GB "
          f"-> fwd {gb_f/t_f*1000:.0f} GB/s")
    tot_f += t_f; tot_b += t_b; tot_mf += t_mf; tot_mb += t_mb
print(f"TOTAL fwd {tot_f:.2f} (mi {tot_mf:.2f})  bwdD {tot_b:.2f} (mi {tot_mb:.2f})")
