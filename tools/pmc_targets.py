"""Minimal dispatch set for PMC counter collection: each hand-written
hot kernel exactly a few times at flagship-like shapes (full-bench PMC
serializes thousands of dispatches and takes forever)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

dev = "cuda:0"
torch.manual_seed(0)

# conv3x3 fused reflect (decoder hot conv) fwd + wrw + bwd-data
from mine_amd.ops.conv import conv3x3_reflect
x = torch.randn(256, 16, 256, 384, device=dev, dtype=torch.bfloat16
                ).contiguous(memory_format=torch.channels_last
                ).requires_grad_(True)
w = (torch.randn(16, 16, 3, 3, device=dev) * 0.2).requires_grad_(True)
for _ in range(2):
    y = conv3x3_reflect(x, w, None)
    y.float().sum().backward()
    x.grad = w.grad = None

# general igemm (encoder bottleneck 3x3 + deep base conv)
from mine_amd.ops.conv_general import conv2d_mfma
xe = torch.randn(4, 256, 16, 24, device=dev, dtype=torch.bfloat16
                 ).contiguous(memory_format=torch.channels_last
                 ).requires_grad_(True)
we = (torch.randn(256, 256, 3, 3, device=dev) * 0.05).requires_grad_(True)
for _ in range(2):
    conv2d_mfma(xe, we, None, padding=1).float().sum().backward()
    xe.grad = we.grad = None

# fused BN+ELU fwd+bwd
from mine_amd.ops.bn import FusedBNAct
bn = FusedBNAct(16, act="elu").to(dev).train()
xb = torch.randn(256, 16, 256, 384, device=dev, dtype=torch.bfloat16
                 ).contiguous(memory_format=torch.channels_last
                 ).requires_grad_(True)
for _ in range(2):
    bn(xb).float().sum().backward()
    xb.grad = None

# fused renderers fwd+bwd (src + tgt incl. gather)
from mine_amd.ops.renderer import pack_mpi, render_src_view, render_tgt_view
B, S, H, W = 4, 64, 256, 384
rgb = torch.rand(B, S, 3, H, W, device=dev)
sig = torch.rand(B, S, 1, H, W, device=dev) * 3 + 1e-4
disp, _ = torch.sort(torch.rand(B, S, device=dev) * 0.9 + 0.05, dim=1,
                     descending=True)
f = 0.8 * W
K = torch.tensor([[f, 0, W / 2], [0, f, H / 2], [0, 0, 1.0]],
                 device=dev).unsqueeze(0).repeat(B, 1, 1)
K_inv = torch.inverse(K)
G = torch.eye(4, device=dev).unsqueeze(0).repeat(B, 1, 1)
G[:, 0, 3] = 0.1
img = torch.rand(B, 3, H, W, device=dev)
mpi = pack_mpi(rgb, sig).requires_grad_(True)
r, d, blend = render_src_view(mpi, disp, K_inv, src_img=img)
tr_, td, tm = render_tgt_view(blend, disp, G, K_inv, K)
(r.sum() + tr_.sum() + td.sum()).backward()

# SSIM fwd+bwd
from mine_amd.ops.ssim import ssim
a = torch.rand(4, 3, 256, 384, device=dev).requires_grad_(True)
b = torch.rand(4, 3, 256, 384, device=dev)
(1 - ssim(a, b)).backward()

# upsample
from mine_amd.ops.upsample import upsample_nearest2x
xu = torch.randn(256, 16, 128, 192, device=dev, dtype=torch.bfloat16
                 ).contiguous(memory_format=torch.channels_last
                 ).requires_grad_(True)
upsample_nearest2x(xu).float().sum().backward()

torch.cuda.synchronize()
print("pmc targets done")
