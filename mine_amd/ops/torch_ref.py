"""Pure-PyTorch reference implementations of the MPI rendering math.

These are the CPU execution path AND the golden oracle every HIP kernel
is tested against. Numerics deliberately reproduce the reference's
cliffs exactly (ref operations/mpi_rendering.py):

  * transparency  t = exp(-sigma * delta), delta padded with 1e3 at the
    far plane
  * accumulated transparency = shifted cumprod(t + 1e-6)
  * depth normalization  / (sum(w) + 1e-5), or +1000*(1-sum(w)) when
    `bg_depth_inf`
  * homography H_tgt_src = K_tgt (R - t n^T / -d) K_src_inv, inverted in
    closed form (the reference used a nan-retrying torch.inverse,
    ref utils.py:96-117 — we use the exact adjugate inverse)
  * grid_sample bilinear, padding_mode="border", align_corners=False
    with the +0.5 pixel-center normalization
    (ref operations/homography_sampler.py:117-141)
"""
from __future__ import annotations

from typing import Tuple

import torch
import torch.nn.functional as F

from mine_amd.utils.geometry import inverse_3x3

# ---------------------------------------------------------------------------
# sampling
# ---------------------------------------------------------------------------


def sample_disparity_linspace(batch_size: int, num_bins: int, start: float, end: float,
                              device=None, stratified: bool = True) -> torch.Tensor:
    """Stratified (jittered) disparity samples in linspace bins, descending
    (near -> far). ref operations/rendering_utils.py:70-88."""
    assert start > end
    edges = torch.linspace(start, end, num_bins + 1, dtype=torch.float32, device=device)
    interval = edges[1] - edges[0]
    starts = edges[:-1].unsqueeze(0).expand(batch_size, num_bins)
    if stratified:
        jitter = torch.rand((batch_size, num_bins), dtype=torch.float32, device=device)
    else:
        jitter = torch.full((batch_size, num_bins), 0.5, dtype=torch.float32, device=device)
    return starts + interval * jitter


def sample_disparity_from_bins(batch_size: int, disparity_edges: torch.Tensor,
                               device=None, stratified: bool = True) -> torch.Tensor:
    """Stratified samples from explicit (descending) bin edges.
    ref operations/rendering_utils.py:47-67."""
    edges = torch.as_tensor(disparity_edges, dtype=torch.float32, device=device)
    assert edges[0] > edges[-1]
    S = edges.numel() - 1
    interval = (edges[1:] - edges[:-1]).unsqueeze(0).expand(batch_size, S)
    starts = edges[:-1].unsqueeze(0).expand(batch_size, S)
    if stratified:
        jitter = torch.rand((batch_size, S), dtype=torch.float32, device=device)
    else:
        jitter = torch.full((batch_size, S), 0.5, dtype=torch.float32, device=device)
    return starts + interval * jitter


def sample_pdf(values: torch.Tensor, weights: torch.Tensor, n_samples: int) -> torch.Tensor:
    """Inverse-CDF sampling (ref operations/rendering_utils.py:91-140).

    values, weights: Bx1xNxS. Returns Bx1xNxn_samples.
    """
    B, _, N, S = weights.shape
    assert values.shape == (B, 1, N, S)

    mid = (values[..., 1:] + values[..., :-1]) * 0.5
    bin_edges = torch.cat((values[..., :1], mid, values[..., -1:]), dim=-1)  # S+1

    pdf = weights / (weights.sum(dim=-1, keepdim=True) + 1e-5)
    cdf = torch.cumsum(pdf, dim=-1)
    cdf = torch.cat((torch.zeros_like(cdf[..., :1]), cdf), dim=-1)  # S+1

    u = torch.rand((B, 1, N, n_samples), dtype=weights.dtype, device=weights.device)

    idx = torch.searchsorted(cdf, u, right=True)
    lo = torch.clamp(idx - 1, min=0)
    hi = torch.clamp(idx, max=S)

    cdf_lo = torch.gather(cdf, -1, lo)
    cdf_hi = torch.gather(cdf, -1, hi)
    bin_lo = torch.gather(bin_edges, -1, lo)
    bin_hi = torch.gather(bin_edges, -1, hi)

    denom = cdf_hi - cdf_lo
    t = (u - cdf_lo) / torch.clamp(denom, min=1e-5)
    t = torch.where(denom <= 1e-4, torch.full_like(t, 0.5), t)
    return bin_lo + t * (bin_hi - bin_lo)


# ---------------------------------------------------------------------------
# geometry
# ---------------------------------------------------------------------------


def make_meshgrid(H: int, W: int, device=None) -> torch.Tensor:
    """Homogeneous pixel grid, 3xHxW fp32: (x, y, 1) at integer pixel coords
    (ref operations/homography_sampler.py:24-33)."""
    y, x = torch.meshgrid(
        torch.arange(H, dtype=torch.float32, device=device),
        torch.arange(W, dtype=torch.float32, device=device),
        indexing="ij",
    )
    return torch.stack((x, y, torch.ones_like(x)), dim=0)


def src_plane_xyz(meshgrid: torch.Tensor, disparity: torch.Tensor,
                  K_inv: torch.Tensor) -> torch.Tensor:
    """Per-plane 3D point maps in the source camera: K^-1 p * depth_s.

    meshgrid: 3xHxW; disparity: BxS; K_inv: Bx3x3 -> BxSx3xHxW
    (ref operations/mpi_rendering.py:140-163).
    """
    B, S = disparity.shape
    H, W = meshgrid.shape[-2:]
    depth = torch.reciprocal(disparity)  # BxS
    rays = torch.matmul(K_inv, meshgrid.reshape(3, -1).to(K_inv.dtype))  # Bx3xHW
    xyz = rays.unsqueeze(1) * depth[:, :, None, None]  # BxSx3xHW
    return xyz.reshape(B, S, 3, H, W)


def tgt_plane_xyz(xyz_src: torch.Tensor, G_tgt_src: torch.Tensor) -> torch.Tensor:
    """Rigid transform of plane point maps: BxSx3xHxW -> BxSx3xHxW
    (ref operations/mpi_rendering.py:166-178)."""
    B, S, _, H, W = xyz_src.shape
    R = G_tgt_src[:, :3, :3].unsqueeze(1)  # Bx1x3x3
    t = G_tgt_src[:, :3, 3].unsqueeze(1)  # Bx1x3
    pts = xyz_src.reshape(B, S, 3, H * W)
    out = torch.matmul(R, pts) + t.unsqueeze(-1)
    return out.reshape(B, S, 3, H, W)


def homography_tgt_to_src(G_tgt_src: torch.Tensor, depths: torch.Tensor,
                          K_src_inv: torch.Tensor, K_tgt: torch.Tensor) -> torch.Tensor:
    """H_src_tgt for every plane: maps a tgt pixel to src pixel coords.

    G_tgt_src: Bx4x4, depths: BxS (plane depths d), K_src_inv/K_tgt: Bx3x3.
    Returns BxSx3x3 = inverse( K_tgt (R - t n^T / -d) K_src_inv ), n=[0,0,1]
    (ref operations/homography_sampler.py:101-114), via exact closed-form
    3x3 inverse.
    """
    B, S = depths.shape
    R = G_tgt_src[:, :3, :3]  # Bx3x3
    t = G_tgt_src[:, :3, 3]  # Bx3
    # t n^T has only the last column nonzero: t
    tn = torch.zeros((B, 3, 3), dtype=R.dtype, device=R.device)
    tn[:, :, 2] = t
    # R - t n^T / -d  =  R + t n^T / d
    Rtnd = R.unsqueeze(1) + tn.unsqueeze(1) / depths[:, :, None, None]  # BxSx3x3
    H_tgt_src = torch.matmul(K_tgt.unsqueeze(1), torch.matmul(Rtnd, K_src_inv.unsqueeze(1)))
    return inverse_3x3(H_tgt_src)


# ---------------------------------------------------------------------------
# compositing
# ---------------------------------------------------------------------------


def alpha_composite(alpha: torch.Tensor, value: torch.Tensor
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Over-composite: BxS x1xHxW alpha, BxSxCxHxW values -> BxCxHxW
    (ref operations/mpi_rendering.py:23-39)."""
    acc = torch.cumprod(1.0 - alpha, dim=1)
    keep = torch.cat((torch.ones_like(acc[:, :1]), acc[:, :-1]), dim=1)
    weights = alpha * keep
    return (value * weights).sum(dim=1), weights


def weighted_sum_mpi(rgb: torch.Tensor, xyz: torch.Tensor, weights: torch.Tensor,
                     bg_depth_inf: bool) -> Tuple[torch.Tensor, torch.Tensor]:
    """Weighted RGB and z-depth sums (ref operations/mpi_rendering.py:70-82)."""
    wsum = weights.sum(dim=1)  # Bx1xHxW
    rgb_out = (weights * rgb).sum(dim=1)
    zsum = (weights * xyz[:, :, 2:]).sum(dim=1)
    if bg_depth_inf:
        depth_out = zsum + (1.0 - wsum) * 1000.0
    else:
        depth_out = zsum / (wsum + 1e-5)
    return rgb_out, depth_out


def volume_composite(rgb: torch.Tensor, sigma: torch.Tensor, xyz: torch.Tensor,
                     bg_depth_inf: bool
                     ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """NeRF-style per-plane volume rendering
    (ref operations/mpi_rendering.py:42-67).

    rgb BxSx3xHxW, sigma BxSx1xHxW, xyz BxSx3xHxW ->
    (rgb_out Bx3xHxW, depth_out Bx1xHxW, transparency_acc BxSx1xHxW,
     weights BxSx1xHxW)
    """
    B, S, _, H, W = sigma.shape
    diff = xyz[:, 1:] - xyz[:, :-1]
    dist = torch.norm(diff, dim=2, keepdim=True)  # Bx(S-1)x1xHxW
    far = torch.full((B, 1, 1, H, W), 1e3, dtype=xyz.dtype, device=xyz.device)
    dist = torch.cat((dist, far), dim=1)

    transparency = torch.exp(-sigma * dist)
    alpha = 1.0 - transparency

    acc = torch.cumprod(transparency + 1e-6, dim=1)
    acc = torch.cat((torch.ones_like(acc[:, :1]), acc[:, :-1]), dim=1)

    weights = acc * alpha
    rgb_out, depth_out = weighted_sum_mpi(rgb, xyz, weights, bg_depth_inf)
    return rgb_out, depth_out, acc, weights


# ---------------------------------------------------------------------------
# warping (reference path: explicit grid + grid_sample)
# ---------------------------------------------------------------------------


def homography_grid_sample(src: torch.Tensor, H_src_tgt: torch.Tensor,
                           H_out: int, W_out: int
                           ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Warp NxCxHxW planes by per-plane 3x3 H_src_tgt (Nx3x3): build the
    target meshgrid, map through H, perspective-divide, bilinear-sample
    with border padding (ref operations/homography_sampler.py:117-141).

    Returns (warped NxCxH_outxW_out, valid mask NxH_outxW_out bool).
    """
    N, C, H_src, W_src = src.shape
    grid = make_meshgrid(H_out, W_out, device=src.device).to(src.dtype)  # 3xHxW
    mapped = torch.matmul(H_src_tgt.to(src.dtype), grid.reshape(3, -1))  # Nx3xHW
    mapped = mapped.reshape(N, 3, H_out, W_out).permute(0, 2, 3, 1)  # NxHxWx3
    uv = mapped[..., :2] / mapped[..., 2:]

    valid = ((uv[..., 0] > -1) & (uv[..., 0] < W_src)
             & (uv[..., 1] > -1) & (uv[..., 1] < H_src))

    gx = (uv[..., 0] + 0.5) / (W_src * 0.5) - 1.0
    gy = (uv[..., 1] + 0.5) / (H_src * 0.5) - 1.0
    sample_grid = torch.stack((gx, gy), dim=-1)
    # grid_sample's backward derives integer scatter indices from the grid
    # values WITHOUT an is-finite check — a NaN coordinate (poisoned batch,
    # degenerate homography) segfaults the CPU kernel. Map non-finite
    # coords far out of bounds: border padding clamps them in forward and
    # the backward treats them as clipped (zero-gradient) taps.
    sample_grid = torch.nan_to_num(sample_grid, nan=-10.0, posinf=-10.0,
                                   neginf=-10.0)
    warped = F.grid_sample(src, sample_grid, mode="bilinear",
                           padding_mode="border", align_corners=False)
    return warped, valid


def render_tgt_reference(mpi_rgb: torch.Tensor, mpi_sigma: torch.Tensor,
                         disparity: torch.Tensor, G_tgt_src: torch.Tensor,
                         K_src_inv: torch.Tensor, K_tgt: torch.Tensor,
                         use_alpha: bool = False, bg_depth_inf: bool = False
                         ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Full target-view render, reference dataflow (materializes the warped
    7-channel plane stack; ref operations/mpi_rendering.py:181-241). Used on
    CPU and as the oracle for the fused HIP kernel.

    Returns (tgt_rgb Bx3xHxW, tgt_depth Bx1xHxW, tgt_mask Bx1xHxW float).
    """
    B, S, _, H, W = mpi_rgb.shape
    depths = torch.reciprocal(disparity)  # BxS

    grid = make_meshgrid(H, W, device=mpi_rgb.device)
    xyz_src = src_plane_xyz(grid, disparity, K_src_inv)
    xyz_tgt = tgt_plane_xyz(xyz_src, G_tgt_src)

    planes = torch.cat((mpi_rgb, mpi_sigma, xyz_tgt), dim=2)  # BxSx7xHxW
    H_src_tgt = homography_tgt_to_src(G_tgt_src, depths, K_src_inv, K_tgt)

    warped, valid = homography_grid_sample(planes.reshape(B * S, 7, H, W),
                                           H_src_tgt.reshape(B * S, 3, 3), H, W)
    warped = warped.reshape(B, S, 7, H, W)
    w_rgb = warped[:, :, 0:3]
    w_sigma = warped[:, :, 3:4]
    w_xyz = warped[:, :, 4:7]

    # cull sigma where the warped plane is behind the target camera
    w_z = w_xyz[:, :, 2:]
    w_sigma = torch.where(w_z >= 0, w_sigma, torch.zeros_like(w_sigma))

    if use_alpha:
        tgt_rgb, _ = alpha_composite(w_sigma, w_rgb)
        tgt_depth, _ = alpha_composite(w_sigma, w_xyz[:, :, 2:])
    else:
        tgt_rgb, tgt_depth, _, _ = volume_composite(w_rgb, w_sigma, w_xyz, bg_depth_inf)

    mask = valid.reshape(B, S, H, W).to(torch.float32).sum(dim=1, keepdim=True)
    return tgt_rgb, tgt_depth, mask


# ---------------------------------------------------------------------------
# sparse gathers
# ---------------------------------------------------------------------------


def gather_pixel_by_pxpy(img: torch.Tensor, pxpy: torch.Tensor) -> torch.Tensor:
    """Round + clamp pixel coords, gather image values: BxCxHxW, Bx2xN ->
    BxCxN (ref operations/rendering_utils.py:27-44)."""
    B, C, H, W = img.shape
    with torch.no_grad():
        pp = torch.round(pxpy).to(torch.int64)
        px = torch.clamp(pp[:, 0:1], 0, W - 1)
        py = torch.clamp(pp[:, 1:2], 0, H - 1)
        idx = px + W * py  # Bx1xN
    return torch.gather(img.reshape(B, C, H * W), 2, idx.expand(B, C, idx.size(2)))


def get_xyz_from_depth(depth: torch.Tensor, K_inv: torch.Tensor) -> torch.Tensor:
    """Backproject a depth map to camera-frame points: Bx1xHxW, Bx3x3 ->
    Bx3xHxW (ref operations/mpi_rendering.py:85-105; unused in the
    reference's train loop, kept for surface parity)."""
    B, _, H, W = depth.shape
    grid = make_meshgrid(H, W, device=depth.device)
    rays = torch.matmul(K_inv, grid.reshape(3, -1).to(K_inv.dtype))  # Bx3xHW
    return (rays * depth.reshape(B, 1, H * W)).reshape(B, 3, H, W)


def disparity_consistency_src_to_tgt(src_disparity: torch.Tensor,
                                     tgt_disparity: torch.Tensor,
                                     G_tgt_src: torch.Tensor,
                                     K_src_inv: torch.Tensor,
                                     K_tgt: torch.Tensor) -> torch.Tensor:
    """Cross-view disparity consistency: backproject the src disparity,
    transform into the tgt camera, project, and compare the induced tgt
    disparity with the rendered one at the projected pixels
    (ref operations/mpi_rendering.py:108-137; unused in the reference's
    train loop)."""
    B, _, H, W = src_disparity.shape
    xyz_src = get_xyz_from_depth(torch.reciprocal(src_disparity), K_src_inv)
    xyz_tgt = tgt_plane_xyz(xyz_src.unsqueeze(1), G_tgt_src).squeeze(1)  # Bx3xHxW
    z = xyz_tgt[:, 2:].clamp_min(1e-6)
    pix = torch.matmul(K_tgt, xyz_tgt.reshape(B, 3, -1))
    pxpy = pix[:, 0:2] / pix[:, 2:].clamp_min(1e-6)
    sampled = gather_pixel_by_pxpy(tgt_disparity, pxpy)  # Bx1xHW
    induced = torch.reciprocal(z.reshape(B, 1, -1))
    return torch.mean(torch.abs(torch.log(sampled.clamp_min(1e-6))
                                - torch.log(induced.clamp_min(1e-6))))
