"""Golden-value tests for the rendering math (the CPU oracle itself).

These independently validate the torch reference implementation with
hand-computed values and geometric invariants; the HIP kernels are then
tested against this oracle in test_gpu_ops.py.
"""
import math

import pytest
import torch

from mine_amd.ops import torch_ref as tr


def _simple_K(B=1, f=100.0, W=64, H=48):
    K = torch.tensor([[f, 0.0, W / 2], [0.0, f, H / 2], [0.0, 0.0, 1.0]])
    return K.unsqueeze(0).repeat(B, 1, 1)


def test_volume_composite_single_pixel_analytic():
    """S=2, one pixel: verify against a hand-derived composite."""
    B, S, H, W = 1, 2, 1, 1
    rgb = torch.tensor([0.25, 0.5]).view(B, S, 1, H, W).expand(B, S, 3, H, W).contiguous()
    sigma = torch.tensor([0.7, 0.3]).view(B, S, 1, H, W)
    # xyz: plane 0 at z=1, plane 1 at z=3 along the optical axis
    xyz = torch.zeros(B, S, 3, H, W)
    xyz[:, 0, 2] = 1.0
    xyz[:, 1, 2] = 3.0

    rgb_out, depth_out, acc, weights = tr.volume_composite(rgb, sigma, xyz, False)

    d0 = 2.0  # |z1 - z0|
    t0 = math.exp(-0.7 * d0)
    t1 = math.exp(-0.3 * 1e3)  # far-plane distance 1e3
    w0 = 1.0 * (1 - t0)
    w1 = (t0 + 1e-6) * (1 - t1)
    expect_rgb = w0 * 0.25 + w1 * 0.5
    expect_depth = (w0 * 1.0 + w1 * 3.0) / (w0 + w1 + 1e-5)

    torch.testing.assert_close(rgb_out.flatten()[0], torch.tensor(expect_rgb),
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(depth_out.flatten()[0], torch.tensor(expect_depth),
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(weights.flatten(),
                               torch.tensor([w0, w1]), rtol=1e-5, atol=1e-7)
    torch.testing.assert_close(acc.flatten(),
                               torch.tensor([1.0, t0 + 1e-6]), rtol=1e-6, atol=1e-7)


def test_volume_composite_bg_depth_inf():
    B, S, H, W = 1, 1, 1, 1
    rgb = torch.full((B, S, 3, H, W), 0.5)
    sigma = torch.full((B, S, 1, H, W), 1e-9)  # nearly transparent
    xyz = torch.zeros(B, S, 3, H, W)
    xyz[:, 0, 2] = 2.0
    _, depth_out, _, _ = tr.volume_composite(rgb, sigma, xyz, True)
    # weights ~0 -> depth ~ 1000 (ref mpi_rendering.py:74-77)
    assert abs(depth_out.item() - 1000.0) < 1.0


def test_opaque_first_plane_dominates():
    B, S, H, W = 1, 4, 2, 2
    rgb = torch.rand(B, S, 3, H, W)
    sigma = torch.full((B, S, 1, H, W), 1e-4)
    sigma[:, 0] = 1e4  # opaque near plane
    disparity = torch.tensor([[1.0, 0.5, 0.25, 0.125]])
    K_inv = torch.inverse(_simple_K(W=W, H=H))
    xyz = tr.src_plane_xyz(tr.make_meshgrid(H, W), disparity, K_inv)
    rgb_out, depth_out, _, _ = tr.volume_composite(rgb, sigma, xyz, False)
    torch.testing.assert_close(rgb_out, rgb[:, 0], rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(depth_out, xyz[:, 0, 2:], rtol=1e-3, atol=1e-3)


def test_src_plane_xyz_geometry():
    """xyz = K^-1 p * depth: principal point maps to (0, 0, depth)."""
    H, W = 48, 64
    K = _simple_K(W=W, H=H)
    K_inv = torch.inverse(K)
    disparity = torch.tensor([[0.5]])  # depth 2
    xyz = tr.src_plane_xyz(tr.make_meshgrid(H, W), disparity, K_inv)
    cx, cy = W // 2, H // 2
    torch.testing.assert_close(xyz[0, 0, :, cy, cx], torch.tensor([0.0, 0.0, 2.0]))
    # a pixel f to the right of center: x = depth * (px - cx)/f = 2 * 10/100
    torch.testing.assert_close(xyz[0, 0, :, cy, cx + 10],
                               torch.tensor([0.2, 0.0, 2.0]), rtol=1e-5, atol=1e-6)


def test_homography_identity_pose():
    """Identity pose => H_src_tgt == identity for every plane."""
    B, S = 2, 4
    G = torch.eye(4).unsqueeze(0).repeat(B, 1, 1)
    depths = torch.tensor([[1.0, 2.0, 4.0, 8.0]]).repeat(B, 1)
    K = _simple_K(B)
    H = tr.homography_tgt_to_src(G, depths, torch.inverse(K), K)
    torch.testing.assert_close(H, torch.eye(3).expand(B, S, 3, 3),
                               rtol=1e-4, atol=1e-5)


def test_homography_translation_shift():
    """Lateral translation tx at plane depth d shifts pixels by f*tx/d."""
    f, W, H = 100.0, 64, 48
    K = _simple_K(f=f, W=W, H=H)
    G = torch.eye(4).unsqueeze(0)
    tx = 0.5
    G[0, 0, 3] = tx  # t_tgt_src: src origin seen from tgt
    d = 2.0
    depths = torch.tensor([[d]])
    Hm = tr.homography_tgt_to_src(G, depths, torch.inverse(K), K)
    # map the tgt center pixel to src: should shift by -f*tx/d in x
    p = torch.tensor([W / 2, H / 2, 1.0])
    q = Hm[0, 0] @ p
    q = q / q[2]
    assert abs((q[0] - W / 2) - (-f * tx / d)) < 1e-3
    assert abs(q[1] - H / 2) < 1e-4


def test_grid_sample_conventions_integer_coords():
    """Warping with identity homography reproduces the image exactly
    (checks the +0.5 / align_corners=False normalization)."""
    torch.manual_seed(0)
    img = torch.rand(3, 2, 9, 13)
    Hm = torch.eye(3).expand(3, 3, 3)
    warped, valid = tr.homography_grid_sample(img, Hm, 9, 13)
    torch.testing.assert_close(warped, img, rtol=1e-5, atol=1e-5)
    assert valid.all()


def test_render_tgt_identity_equals_src_composite():
    torch.manual_seed(3)
    B, S, H, W = 1, 6, 32, 40
    rgb = torch.rand(B, S, 3, H, W)
    sigma = torch.rand(B, S, 1, H, W) * 2
    disparity, _ = torch.sort(torch.rand(B, S) * 0.9 + 0.05, descending=True)
    K = _simple_K(B, W=W, H=H)
    K_inv = torch.inverse(K)
    G = torch.eye(4).unsqueeze(0)

    xyz = tr.src_plane_xyz(tr.make_meshgrid(H, W), disparity, K_inv)
    src_rgb, src_depth, _, _ = tr.volume_composite(rgb, sigma, xyz, False)
    tgt_rgb, tgt_depth, mask = tr.render_tgt_reference(
        rgb, sigma, disparity, G, K_inv, K)

    torch.testing.assert_close(tgt_rgb, src_rgb, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(tgt_depth, src_depth, rtol=1e-4, atol=1e-4)
    assert (mask == S).all()


def test_sigma_zcull():
    """Planes behind the target camera contribute nothing."""
    B, S, H, W = 1, 2, 8, 8
    rgb = torch.rand(B, S, 3, H, W)
    sigma = torch.full((B, S, 1, H, W), 5.0)
    disparity = torch.tensor([[1.0, 0.1]])
    K = _simple_K(B, W=W, H=H)
    # translate the target far along +z so plane 0 (depth 1) is behind it
    G = torch.eye(4).unsqueeze(0)
    G[0, 2, 3] = -5.0  # z_tgt = z_src - 5 -> plane at depth 1 has z=-4 < 0
    tgt_rgb, _, _ = tr.render_tgt_reference(rgb, sigma, disparity, G,
                                            torch.inverse(K), K)
    # only plane 1 (depth 10 -> z=5) can contribute
    assert torch.isfinite(tgt_rgb).all()


def test_gather_pixel_by_pxpy():
    img = torch.arange(12.0).reshape(1, 1, 3, 4)
    pxpy = torch.tensor([[[0.0, 3.4, -2.0], [0.0, 1.6, 5.0]]])  # 1x2x3
    out = tr.gather_pixel_by_pxpy(img, pxpy)
    # (0,0)->0 ; (3,2)->11 (3.4 rounds to 3, 1.6 rounds to 2); (-2,5) clamps to (0,2)->8
    torch.testing.assert_close(out, torch.tensor([[[0.0, 11.0, 8.0]]]))


def test_sample_disparity_stratified_in_bins():
    torch.manual_seed(0)
    d = tr.sample_disparity_linspace(16, 8, 1.0, 0.001)
    assert d.shape == (16, 8)
    edges = torch.linspace(1.0, 0.001, 9)
    for s in range(8):
        assert (d[:, s] <= edges[s] + 1e-6).all()
        assert (d[:, s] >= edges[s + 1] - 1e-6).all()
    # descending
    assert (d[:, :-1] > d[:, 1:]).all()


def test_sample_pdf_concentrates_mass():
    torch.manual_seed(0)
    B, N, S = 1, 1, 8
    values = torch.linspace(1.0, 0.1, S).view(1, 1, 1, S)
    weights = torch.zeros(1, 1, 1, S)
    weights[..., 3] = 1.0  # all mass in bin 3
    samples = tr.sample_pdf(values, weights, 64)
    mid_lo = (values[..., 2] + values[..., 3]) / 2
    mid_hi = (values[..., 3] + values[..., 4]) / 2
    assert ((samples <= mid_lo + 1e-5) & (samples >= mid_hi - 1e-5)).float().mean() > 0.9


def test_alpha_composite_over():
    """alpha=1 at plane 0 -> output = plane 0; uniform alphas follow the
    over operator (ref mpi_rendering.py:23-39)."""
    B, S, H, W = 1, 3, 2, 2
    val = torch.rand(B, S, 3, H, W)
    alpha = torch.zeros(B, S, 1, H, W)
    alpha[:, 0] = 1.0
    out, w = tr.alpha_composite(alpha, val)
    torch.testing.assert_close(out, val[:, 0])
    a = torch.full((B, S, 1, H, W), 0.5)
    out2, w2 = tr.alpha_composite(a, val)
    torch.testing.assert_close(w2[:, 0], torch.full((B, 1, H, W), 0.5))
    torch.testing.assert_close(w2[:, 1], torch.full((B, 1, H, W), 0.25))


def test_get_xyz_from_depth_and_consistency():
    import torch
    from mine_amd.ops import torch_ref as tr

    B, H, W = 1, 8, 10
    f = 6.0
    K = torch.tensor([[f, 0, W / 2], [0, f, H / 2], [0, 0, 1.0]]).unsqueeze(0)
    K_inv = torch.inverse(K)
    depth = torch.full((B, 1, H, W), 4.0)
    xyz = tr.get_xyz_from_depth(depth, K_inv)
    assert xyz.shape == (B, 3, H, W)
    torch.testing.assert_close(xyz[:, 2], depth[:, 0])
    # center pixel backprojects onto the optical axis
    torch.testing.assert_close(xyz[0, :2, H // 2, W // 2],
                               torch.zeros(2), atol=1e-5, rtol=0)

    # identity pose, identical constant disparity -> zero consistency loss
    disp = torch.full((B, 1, H, W), 0.25)
    G_id = torch.eye(4).unsqueeze(0)
    loss = tr.disparity_consistency_src_to_tgt(disp, disp, G_id, K_inv, K)
    assert float(loss) < 1e-6
