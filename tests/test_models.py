import torch

from mine_amd.models import MPIDecoder, ResNetEncoder


def test_encoder_tap_shapes():
    enc = ResNetEncoder()
    assert enc.num_ch_enc == [64, 256, 512, 1024, 2048]
    x = torch.rand(1, 3, 64, 96)
    taps = enc(x)
    assert len(taps) == 5
    expect = [(64, 32, 48), (256, 16, 24), (512, 8, 12), (1024, 4, 6), (2048, 2, 3)]
    for t, (c, h, w) in zip(taps, expect):
        assert t.shape == (1, c, h, w)


def test_encoder_param_count():
    """ResNet-50 sans fc: ~23.5M params (the reference carried +2.05M of
    unused fc; ref SURVEY 2a)."""
    enc = ResNetEncoder()
    n = sum(p.numel() for p in enc.parameters())
    assert 23_000_000 < n < 24_500_000


def test_encoder_no_unused_params():
    enc = ResNetEncoder()
    x = torch.rand(1, 3, 64, 64)
    loss = sum(t.sum() for t in enc(x))
    loss.backward()
    for name, p in enc.named_parameters():
        assert p.grad is not None, name


def test_decoder_output_shapes():
    B, S, H, W = 2, 4, 64, 96
    enc = ResNetEncoder()
    dec = MPIDecoder(enc.num_ch_enc, pos_encoding_multires=10)
    disparity = torch.sort(torch.rand(B, S), descending=True)[0]
    feats = enc(torch.rand(B, 3, H, W))
    out = dec(feats, disparity)
    for s in range(4):
        t = out[("disp", s)]
        assert t.shape == (B, S, 4, H // 2 ** s, W // 2 ** s)
        rgb, sigma = t[:, :, :3], t[:, :, 3:]
        assert (rgb >= 0).all() and (rgb <= 1).all()
        assert (sigma >= 1e-4 - 1e-7).all()


def test_decoder_depends_on_disparity():
    """The continuous-depth conditioning: different disparities must give
    different MPIs from the same image."""
    torch.manual_seed(0)
    enc = ResNetEncoder().eval()
    dec = MPIDecoder(enc.num_ch_enc).eval()
    img = torch.rand(1, 3, 64, 64)
    with torch.no_grad():
        feats = enc(img)
        o1 = dec([f.clone() for f in feats], torch.tensor([[0.9, 0.5]]))
        o2 = dec([f.clone() for f in feats], torch.tensor([[0.8, 0.1]]))
    assert (o1[("disp", 0)] - o2[("disp", 0)]).abs().max() > 1e-6


def test_decoder_packed_output_matches_unpacked():
    import torch
    from mine_amd.models import MPIDecoder, ResNetEncoder

    torch.manual_seed(0)
    enc = ResNetEncoder(50).eval()
    dec = MPIDecoder(num_ch_enc=enc.num_ch_enc).eval()
    img = torch.rand(1, 3, 64, 96)
    disp = torch.linspace(0.9, 0.1, 4).unsqueeze(0)
    with torch.no_grad():
        feats = enc(img)
        out_u = dec(feats, disp)
        out_p = dec(feats, disp, packed=True)
    for s in range(4):
        unpacked = out_u[("disp", s)]            # B,S,4,H,W
        packed = out_p[("disp", s)]              # B,S,H,W,4
        assert packed.shape == (1, 4, 64 >> s, 96 >> s, 4)
        torch.testing.assert_close(
            packed, unpacked.permute(0, 1, 3, 4, 2).contiguous())


def test_split_conv_block_matches_materialized_concat():
    """SplitConvBlock's factored conv == conv over the expanded concat
    (exactness of the B->B*S factorization; reflection pad preserves
    constant PE fields)."""
    import torch
    import torch.nn.functional as F
    from mine_amd.models.decoder import SplitConvBlock

    torch.manual_seed(5)
    B, S, E = 2, 3, 21
    dec_ch, base_ch, out_ch, H, W = 8, 12, 6, 10, 14
    blk = SplitConvBlock(dec_ch, base_ch, E, out_ch).eval()

    x_dec = torch.randn(B * S, dec_ch, H, W)
    base = torch.randn(B, base_ch, H, W)
    pe = torch.randn(B * S, E)

    with torch.no_grad():
        y = blk(x_dec, base, pe, B, S)

        # oracle: materialize expand + concat, one conv, same bn
        base_x = base.unsqueeze(1).expand(B, S, base_ch, H, W
                                          ).reshape(B * S, base_ch, H, W)
        pe_x = pe[:, :, None, None].expand(B * S, E, H, W)
        cat = torch.cat((x_dec, base_x, pe_x), dim=1)
        z = F.conv2d(F.pad(cat, (1, 1, 1, 1), mode="reflect"),
                     blk.conv.weight, blk.conv.bias)
        ref = blk.bn(z)

    torch.testing.assert_close(y, ref, rtol=1e-4, atol=1e-5)

    # no-dec variant (the neck stage)
    blk0 = SplitConvBlock(0, base_ch, E, out_ch).eval()
    with torch.no_grad():
        y0 = blk0(None, base, pe, B, S)
        cat0 = torch.cat((base_x, pe_x), dim=1)
        z0 = F.conv2d(F.pad(cat0, (1, 1, 1, 1), mode="reflect"),
                      blk0.conv.weight, blk0.conv.bias)
        ref0 = blk0.bn(z0)
    torch.testing.assert_close(y0, ref0, rtol=1e-4, atol=1e-5)


def test_monodepth2_layer_zoo():
    import math
    import torch
    from mine_amd.models.layers import (BackprojectDepth, PooledSSIM,
                                        Project3D, compute_depth_errors,
                                        disp_to_depth, get_smooth_loss,
                                        rot_from_axisangle,
                                        transformation_from_parameters)

    # disp_to_depth endpoints
    s, d = disp_to_depth(torch.tensor([0.0, 1.0]), 0.1, 100.0)
    assert torch.allclose(d, torch.tensor([100.0, 0.1]))

    # rotation: 90 deg about z
    aa = torch.tensor([[[0.0, 0.0, math.pi / 2]]])
    R = rot_from_axisangle(aa)[0, :3, :3]
    assert torch.allclose(R @ torch.tensor([1.0, 0.0, 0.0]),
                          torch.tensor([0.0, 1.0, 0.0]), atol=1e-6)

    # transformation invert round-trip
    t = torch.tensor([[[0.1, -0.2, 0.3]]])
    T = transformation_from_parameters(aa, t)
    Ti = transformation_from_parameters(aa, t, invert=True)
    assert torch.allclose(T @ Ti, torch.eye(4).unsqueeze(0), atol=1e-5)

    # backproject/project round-trip at identity pose
    B, H, W = 1, 8, 10
    K = torch.eye(4).unsqueeze(0)
    K[0, 0, 0] = K[0, 1, 1] = 5.0
    K[0, 0, 2], K[0, 1, 2] = W / 2, H / 2
    depth = torch.full((B, 1, H, W), 2.0)
    pts = BackprojectDepth(B, H, W)(depth, torch.inverse(K))
    grid = Project3D(B, H, W)(pts, K, torch.eye(4).unsqueeze(0))
    yy, xx = torch.meshgrid(torch.arange(H, dtype=torch.float32),
                            torch.arange(W, dtype=torch.float32), indexing="ij")
    exp_x = (xx / (W - 1) - 0.5) * 2
    exp_y = (yy / (H - 1) - 0.5) * 2
    assert torch.allclose(grid[0, :, :, 0], exp_x, atol=1e-5)
    assert torch.allclose(grid[0, :, :, 1], exp_y, atol=1e-5)

    # smoothness zero for constant disp; SSIM zero-dissimilarity for x==y
    img = torch.rand(1, 3, 8, 10)
    assert get_smooth_loss(torch.ones(1, 1, 8, 10), img) == 0
    assert PooledSSIM()(img, img).abs().max() < 1e-5

    # depth metrics perfect prediction
    gt = torch.rand(100) + 0.5
    m = compute_depth_errors(gt, gt.clone())
    assert m["a1"] == 1.0 and m["rmse"] < 1e-6
