"""Minimal dispatch set for PMC counter collection — direct extension
calls, ~40 dispatches total (counter mode costs ~0.1-1 s per dispatch)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

dev = "cuda:0"
torch.manual_seed(0)
from mine_amd.ops.backend import get_extension
ext = get_extension(required=True)

def cl(t):
    return t.contiguous(memory_format=torch.channels_last)

# --- conv3x3 fused reflect fwd + wrw + bwd-data (decoder hot conv) ---
from mine_amd.ops.conv import pack_weights, conv3x3_bwd_data
B, C, H, W, K = 64, 16, 128, 192, 16
x = cl(torch.randn(B, C, H, W, device=dev, dtype=torch.bfloat16))
w = torch.randn(K, C, 3, 3, device=dev) * 0.2
gy = cl(torch.randn(B, K, H, W, device=dev, dtype=torch.bfloat16))
wp = pack_weights(w.to(torch.bfloat16))
out = torch.empty(B * H * W * K, device=dev, dtype=torch.bfloat16)
ext.conv3x3_fwd(x.permute(0, 2, 3, 1).reshape(-1), wp,
                torch.empty(0, device=dev), out, B, H, W, C, K, 0, H, W, 0)
ext.conv3x3_wrw(x.permute(0, 2, 3, 1).reshape(-1),
                gy.permute(0, 2, 3, 1).reshape(-1), B, H, W, C, K)
conv3x3_bwd_data(gy, w)

# --- general igemm fwd/wrw (encoder bottleneck) ---
from mine_amd.ops.conv_general import pack_weights_general
xe = cl(torch.randn(4, 256, 16, 24, device=dev, dtype=torch.bfloat16))
we = torch.randn(256, 256, 3, 3, device=dev) * 0.05
gye = cl(torch.randn(4, 256, 16, 24, device=dev, dtype=torch.bfloat16))
wep = pack_weights_general(we)
M = 4 * 16 * 24
ext.conv_igemm_fwd(xe.permute(0, 2, 3, 1).reshape(-1), wep,
                   torch.empty(0, device=dev), M, 16, 24, 256, 16, 24, 256,
                   3, 3, 1, 1, -1, 1, 0)
ext.conv_igemm_wrw(xe.permute(0, 2, 3, 1).reshape(-1),
                   gye.permute(0, 2, 3, 1).reshape(-1), M, 16, 24, 256,
                   16, 24, 256, 3, 3, 1, 1, -1, 1, 0)

# --- fused BN kernels ---
xb = cl(torch.randn(B, C, H, W, device=dev, dtype=torch.bfloat16))
xb_flat = xb.permute(0, 2, 3, 1).reshape(-1)
Mb = B * H * W
mean, invstd = ext.bn_stats(xb_flat, Mb, C, torch.empty(0, device=dev),
                            torch.empty(0, device=dev), 1e-5, 0.1)
gamma = torch.ones(C, device=dev)
beta = torch.zeros(C, device=dev)
empty_b = torch.empty(0, device=dev, dtype=torch.bfloat16)
y = ext.bn_act_fwd(xb_flat, empty_b, mean, invstd, gamma, beta, Mb, C, 3)
gyb = gy.permute(0, 2, 3, 1).reshape(-1)
red = ext.bn_act_bwd_reduce(xb_flat, empty_b, gyb, mean, invstd, gamma,
                            beta, Mb, C, 3)
ext.bn_act_bwd_dx(xb_flat, empty_b, gyb, mean, invstd, gamma, beta, red,
                  Mb, C, 3, Mb)

# --- renderers (src + tgt fwd, tgt bwd gather mode) ---
from mine_amd.ops import torch_ref as tr
from mine_amd.utils.geometry import inverse_3x3
Bb, S, Hh, Ww = 2, 32, 128, 192
mpi = torch.rand(Bb, S, Hh, Ww, 4, device=dev)
disp, _ = torch.sort(torch.rand(Bb, S, device=dev) * 0.9 + 0.05, dim=1,
                     descending=True)
depths = torch.reciprocal(disp).contiguous()
f = 0.8 * Ww
Km = torch.tensor([[f, 0, Ww / 2], [0, f, Hh / 2], [0, 0, 1.0]],
                  device=dev).unsqueeze(0).repeat(Bb, 1, 1)
K_inv = torch.inverse(Km)
G = torch.eye(4, device=dev).unsqueeze(0).repeat(Bb, 1, 1)
G[:, 0, 3] = 0.1
img = torch.rand(Bb, Hh, Ww, 3, device=dev)
ext.src_composite_fwd(mpi, depths, K_inv.contiguous(), img, False, False)
hinv = tr.homography_tgt_to_src(G, depths, K_inv, Km).contiguous()
m = torch.matmul(G[:, :3, :3], K_inv).contiguous()
tvec = G[:, :3, 3].contiguous()
ext.tgt_composite_fwd(mpi, hinv, m, tvec, depths, False, False)
hfwd = inverse_3x3(hinv.reshape(-1, 3, 3)).reshape(Bb, S, 3, 3).contiguous()
g_rgb = torch.rand(Bb, 3, Hh, Ww, device=dev)
g_depth = torch.rand(Bb, 1, Hh, Ww, device=dev)
ext.tgt_composite_bwd(mpi, hinv, hfwd, m, tvec, depths, False, False,
                      g_rgb, g_depth, 1)

# --- SSIM fwd ---
from mine_amd.ops.ssim import ssim
a = torch.rand(4, 3, 128, 192, device=dev)
b = torch.rand(4, 3, 128, 192, device=dev)
with torch.no_grad():
    ssim(a, b)

# --- upsample ---
xu_flat = cl(torch.randn(64, 16, 64, 96, device=dev, dtype=torch.bfloat16)
             ).permute(0, 2, 3, 1).reshape(-1)
ext.upsample2x_fwd(xu_flat, 64, 64, 96, 16)

torch.cuda.synchronize()
print("pmc targets done")
