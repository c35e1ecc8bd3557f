"""Standalone timing of the fused BN kernels at the flagship shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from mine_amd.ops.bn import FusedBNAct

SHAPES = [  # (N, C, H, W) decoder's big BN layers + one encoder-ish
    (256, 16, 256, 384),
    (256, 32, 128, 192),
    (256, 16, 128, 192),
    (256, 64, 64, 96),
    (256, 256, 16, 24),
    (4, 64, 128, 192),
    (4, 256, 64, 96),
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

tot = 0.0
for (N, C, H, W) in SHAPES:
    bn = FusedBNAct(C, act="elu").cuda().train()
    x = torch.randn(N, C, H, W, device="cuda:0", dtype=torch.bfloat16
                    ).contiguous(memory_format=torch.channels_last
                    ).requires_grad_(True)
    gy = torch.randn_like(x)

    def step():
        y = bn(x)
        y.backward(gy)
        x.grad = None

    t = tm(step)
    gb = N * C * H * W * 2 / 1e9
    # stats(1r) + fwd(1r1w) + reduce(2r) + dx(2r1w) = 6r + 2w = 8 passes
    print(f"N{N} C{C} {H}x{W}: fwd+bwd {t:7.3f} ms  tensor {gb*1000:.0f} MB"
          f"  -> {gb*8/t*1000:.0f} GB/s of {6300}")
    tot += t
print(f"TOTAL {tot:.2f} ms")
