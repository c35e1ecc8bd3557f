"""Loss functions: edge-aware smoothness (v1/v2) and PSNR.

v1 (ref network/layers.py:54-80) uses kornia-style normalized Sobel
gradients with replicate padding; since kornia is not a dependency the
Sobel operator is implemented directly (3x3 kernels / 8, replicate pad —
matching kornia 0.3.0 `spatial_gradient(..., normalized=True)`), plus a
per-image grad-max edge mask, instance-normalized disparity gradients
and a hinge at `gmin`.

v2 (ref network/layers.py:83-99) is the monodepth2-style mean-normalized
finite-difference smoothness weighted by exp(-|image gradient|).
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

_SOBEL_X = torch.tensor([[-1.0, 0.0, 1.0],
                         [-2.0, 0.0, 2.0],
                         [-1.0, 0.0, 1.0]]) / 8.0
_SOBEL_Y = _SOBEL_X.transpose(0, 1).contiguous()


def spatial_gradient(img: torch.Tensor, normalized: bool = True) -> torch.Tensor:
    """Sobel gradients with replicate padding: BxCxHxW -> BxCx2xHxW
    (dim 2 = [dx, dy]); kornia-0.3.0-compatible."""
    B, C, H, W = img.shape
    kx = _SOBEL_X.to(device=img.device, dtype=img.dtype)
    ky = _SOBEL_Y.to(device=img.device, dtype=img.dtype)
    if not normalized:
        kx = kx * 8.0
        ky = ky * 8.0
    kernel = torch.stack((kx, ky)).unsqueeze(1)  # 2x1x3x3
    x = img.reshape(B * C, 1, H, W)
    x = F.pad(x, (1, 1, 1, 1), mode="replicate")
    out = F.conv2d(x, kernel)  # (B*C)x2xHxW
    return out.reshape(B, C, 2, H, W)


def edge_aware_loss(img: torch.Tensor, disp: torch.Tensor,
                    gmin: float, grad_ratio: float = 0.1) -> torch.Tensor:
    """Edge-aware smoothness v1 (ref network/layers.py:54-80)."""
    grad_img = torch.abs(spatial_gradient(img)).sum(1, keepdim=True).to(torch.float32)
    grad_img_x = grad_img[:, :, 0]
    grad_img_y = grad_img[:, :, 1]
    gmax_x = torch.amax(grad_img_x, dim=(1, 2, 3), keepdim=True)
    gmax_y = torch.amax(grad_img_y, dim=(1, 2, 3), keepdim=True)

    edge_x = torch.clamp(grad_img_x / (gmax_x * grad_ratio), max=1.0)
    edge_y = torch.clamp(grad_img_y / (gmax_y * grad_ratio), max=1.0)

    grad_disp = torch.abs(spatial_gradient(disp, normalized=False))
    gdx = F.instance_norm(grad_disp[:, :, 0]) - gmin
    gdy = F.instance_norm(grad_disp[:, :, 1]) - gmin

    loss_x = torch.clamp(gdx, min=0.0) * (1.0 - edge_x)
    loss_y = torch.clamp(gdy, min=0.0) * (1.0 - edge_y)
    return (loss_x + loss_y).mean()


class _EdgeAwareV2Fn(torch.autograd.Function):
    """Fused edge-aware smoothness v2 (loss_kernels.hip): one reduction
    kernel forward, gather + finish backward; grad flows to disp only
    (images carry no gradient in the training losses)."""

    @staticmethod
    def forward(ctx, img, disp):
        from mine_amd.ops.backend import get_extension
        ext = get_extension(required=True)
        mean_d = disp.detach().mean((1, 2, 3))
        out = ext.eav2_fwd(disp, img, mean_d)
        ctx.save_for_backward(img, disp, mean_d)
        return out[0] + out[1]

    @staticmethod
    def backward(ctx, gl):
        from mine_amd.ops.backend import get_extension
        ext = get_extension(required=True)
        img, disp, mean_d = ctx.saved_tensors
        grad = ext.eav2_bwd(disp, img, mean_d, gl.reshape(1).contiguous())
        return None, grad


def edge_aware_loss_v2(img: torch.Tensor, disp: torch.Tensor) -> torch.Tensor:
    """Mean-normalized edge-aware smoothness (ref network/layers.py:83-99).

    GPU fp32 runs the fused HIP kernels; the eager form is the CPU path
    and the oracle."""
    if (disp.is_cuda and disp.dtype == torch.float32
            and img.dtype == torch.float32 and disp.shape[1] == 1
            and img.shape[1] == 3 and not img.requires_grad
            and disp.is_contiguous()):
        return _EdgeAwareV2Fn.apply(img, disp)

    mean_disp = disp.mean(2, True).mean(3, True)
    d = disp / (mean_disp + 1e-7)

    gdx = torch.abs(d[:, :, :, :-1] - d[:, :, :, 1:])
    gdy = torch.abs(d[:, :, :-1, :] - d[:, :, 1:, :])

    gix = torch.mean(torch.abs(img[:, :, :, :-1] - img[:, :, :, 1:]), 1, keepdim=True)
    giy = torch.mean(torch.abs(img[:, :, :-1, :] - img[:, :, 1:, :]), 1, keepdim=True)

    return (gdx * torch.exp(-gix)).mean() + (gdy * torch.exp(-giy)).mean()


def psnr(img1: torch.Tensor, img2: torch.Tensor) -> torch.Tensor:
    """Peak SNR over [0,1] images (ref network/layers.py:48-51)."""
    mse = ((img1 - img2) ** 2).mean((1, 2, 3))
    return (20.0 * torch.log10(1.0 / torch.sqrt(mse))).mean()
