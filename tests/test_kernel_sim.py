"""CPU simulations of the experimental MFMA kernels' index math.

The HIP sources (ops/csrc/wrw_kernels.hip, the zero-embed mode of
conv_kernels.hip) are round-2 groundwork that cannot be executed here;
these tests re-enact their exact staging + fragment addressing + MFMA
contraction in torch and check the result against autograd, so any edit
that breaks the index math fails HERE before it ever reaches a GPU.
"""
import os
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))


def _reflect1(v, n):
    if v < 0:
        v = -v
    if v >= n:
        v = 2 * (n - 1) - v
    return v


def _img_elem(col, j, NB=64):
    """ops/csrc/wrw_kernels.hip img_elem: the permuted-block
    transpose-read image layout."""
    p = j >> 2
    bi = (p & 1) * (NB // 2) + (p >> 1)
    return bi * 64 + (j & 3) * 16 + col


def test_wrw_v2_tr_image_addressing():
    """The transpose-read image layout is bijective and every
    ds_read_b64_tr_b16 pair (base + (l&15) + jj*16 + (l>>4)*64, and the
    same at +JP*8) reconstructs exactly fragment element
    (col = l&15, j = j0 + (l>>4)*8 + e) — the MFMA A/B element map."""
    JP = 256
    img = {}
    for col in range(16):
        for j in range(JP):
            off = _img_elem(col, j)
            assert off not in img
            img[off] = (col, j)
    assert len(img) == 16 * JP and max(img) == 16 * JP - 1

    for jc in range(JP // 32):
        base = jc * 32 * 8
        for lane in range(64):
            g = lane >> 4
            for jj in range(4):
                off_lo = base + (lane & 15) + jj * 16 + g * 64
                assert img[off_lo] == (lane & 15, jc * 32 + g * 8 + jj)
                off_hi = off_lo + JP * 8
                assert img[off_hi] == (lane & 15, jc * 32 + g * 8 + 4 + jj)


def test_wrw_v2_kernel_index_math():
    """Simulates the v2 conv3x3_wrw_kernel: slab split, x-tiles, three
    dx-shifted A (gy) copies, clamped-reflect B (x) halo rows, 32-wide
    j-chunks, (k, c, dy, dx) accumulator map."""
    torch.manual_seed(4)
    B, C, H, W, K = 2, 16, 5, 40, 24
    P_TILE, JP = 254, 256
    x = torch.randn(B, C, H, W)
    gy = torch.randn(B, K, H, W)

    xn = x.permute(0, 2, 3, 1).double()
    gyn = gy.permute(0, 2, 3, 1).double()
    dw = torch.zeros(K, C, 3, 3, dtype=torch.float64)
    n_slabs = min(4, B * H)

    for kc in range((K + 15) // 16):
        k0 = kc * 16
        for zc in range((C + 31) // 32):
            c0 = zc * 32
            Cg = min(32, C - c0)
            chalves = Cg // 16
            for slab in range(n_slabs):
                r0 = B * H * slab // n_slabs
                r1 = B * H * (slab + 1) // n_slabs
                acc = torch.zeros(3, 3, chalves, 16, 16, dtype=torch.float64)
                for r in range(r0, r1):
                    n, y = divmod(r, H)
                    for x0 in range(0, W, P_TILE):
                        Weff = min(P_TILE, W - x0)
                        A = torch.zeros(3, 16, JP, dtype=torch.float64)
                        for pl in range(P_TILE):
                            px = x0 + pl
                            if px >= W:
                                continue
                            for k in range(16):
                                if k0 + k < K:
                                    for dx in range(3):
                                        A[dx, k, pl + dx] = gyn[n, y, px, k0 + k]
                        Bm = torch.zeros(3, chalves, 16, JP,
                                         dtype=torch.float64)
                        for j in range(JP):
                            u = min(x0 - 1 + j, W)
                            us = _reflect1(u, W)
                            for dy in range(3):
                                yy = _reflect1(y + dy - 1, H)
                                for ch in range(chalves):
                                    for cc in range(16):
                                        Bm[dy, ch, cc, j] = \
                                            xn[n, yy, us, c0 + ch * 16 + cc]
                        jchunks = (Weff + 2 + 31) // 32
                        jmax = jchunks * 32
                        for dx in range(3):
                            for dy in range(3):
                                for ch in range(chalves):
                                    acc[dx, dy, ch] += \
                                        A[dx, :, :jmax] @ Bm[dy, ch, :, :jmax].T
                for dx in range(3):
                    for dy in range(3):
                        for ch in range(chalves):
                            for krow in range(16):
                                for jcol in range(16):
                                    kk, cc = k0 + krow, c0 + ch * 16 + jcol
                                    if kk < K and cc < C:
                                        dw[kk, cc, dy, dx] += \
                                            acc[dx, dy, ch, krow, jcol]

    w_probe = torch.zeros(K, C, 3, 3, requires_grad=True)
    y = F.conv2d(F.pad(x, (1, 1, 1, 1), mode="reflect"), w_probe)
    (y * gy).sum().backward()
    torch.testing.assert_close(dw.float(), w_probe.grad, rtol=1e-4, atol=1e-4)


def test_fwd_kernel_pack_and_zero_embed_math():
    """Simulates conv3x3_fwd_kernel addressing for BOTH pad modes:
    (a) the (cb*9+tap)*8+ci k-ordering of pack_weights against reflect
    conv; (b) the zero-embed logical extent (src_h, src_w, off=1) used by
    the experimental bwd-data path."""
    from mine_amd.ops.conv import _pack_lut

    torch.manual_seed(6)
    C, K, H, W = 8, 16, 4, 9
    x = torch.randn(1, C, H, W)
    w = torch.randn(K, C, 3, 3)

    # (a) reconstruct the GEMM from the pack LUT + per-pixel A rows
    lut = _pack_lut(K, C, False, torch.device("cpu"))
    Cv = C // 8
    nseg = 9 * Cv
    nchunks = (nseg + 3) // 4
    nK = (K + 15) // 16
    flat = torch.cat((w.reshape(-1), torch.zeros(1)))
    wp = flat[lut].view(nK, nchunks, 64, 8)

    xp = F.pad(x, (1, 1, 1, 1), mode="reflect").permute(0, 2, 3, 1)[0]  # H+2,W+2,C
    out = torch.zeros(H, W, K)
    for yy in range(H):
        for xx in range(W):
            for nc in range(nK):
                for kc in range(nchunks):
                    for lane in range(64):
                        j = lane & 15
                        seg = kc * 4 + (lane >> 4)
                        kout = nc * 16 + j
                        if seg >= nseg or kout >= K:
                            continue
                        cb, tap = divmod(seg, 9)
                        dy, dx = divmod(tap, 3)
                        for e in range(8):
                            a = xp[yy + dy, xx + dx, cb * 8 + e]
                            out[yy, xx, kout] += float(a) * float(wp[nc, kc, lane, e])
    ref = F.conv2d(F.pad(x, (1, 1, 1, 1), mode="reflect"), w)[0].permute(1, 2, 0)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)

    # (b) zero-embed: logical (H+2, W+2) image backed by (H, W) with off=1
    def map_zero(v, n_logical, src_n, off):
        v2 = v - off
        return v2 if 0 <= v2 < src_n else None

    Hl, Wl = H + 2, W + 2
    gsrc = x.permute(0, 2, 3, 1)[0]
    outz = torch.zeros(Hl, Wl, K)
    for yy in range(Hl):
        for xx in range(Wl):
            for dy in range(3):
                for dx in range(3):
                    ys = map_zero(yy + dy - 1, Hl, H, 1)
                    xs = map_zero(xx + dx - 1, Wl, W, 1)
                    if ys is None or xs is None:
                        continue
                    for k in range(K):
                        outz[yy, xx, k] += float(
                            (gsrc[ys, xs] * w[k, :, dy, dx]).sum())
    refz = F.conv2d(F.pad(x, (2, 2, 2, 2)), w)[0].permute(1, 2, 0)
    torch.testing.assert_close(outz, refz, rtol=1e-4, atol=1e-4)


def test_warp_backward_gather_decomposition():
    """Round-2 design validation (docs/NEXT.md #2): the tgt-composite
    backward's bilinear scatter equals an INTERIOR per-src-pixel gather
    (inverse-homography window search) plus a rare-path scatter for the
    border-clamped tgt pixels. Proven here in torch before any kernel."""
    from mine_amd.ops import torch_ref as tr

    def make_scene(seed, B=1, S=3, H=14, W=18):
        g = torch.Generator().manual_seed(seed)
        disparity, _ = torch.sort(torch.rand(B, S, generator=g) * 0.9 + 0.05,
                                  dim=1, descending=True)
        f = 0.8 * W
        K = torch.tensor([[f, 0, W / 2], [0, f, H / 2], [0, 0, 1.0]]).unsqueeze(0)
        aa = 0.1 * torch.randn(3, generator=g)
        th = aa.norm()
        k = aa / (th + 1e-9)
        Kx = torch.tensor([[0, -k[2], k[1]], [k[2], 0, -k[0]],
                           [-k[1], k[0], 0.0]])
        R = torch.eye(3) + th.sin() * Kx + (1 - th.cos()) * (Kx @ Kx)
        G = torch.eye(4).unsqueeze(0).clone()
        G[0, :3, :3] = R
        G[0, :3, 3] = 0.2 * torch.randn(3, generator=g)
        payload = torch.randn(B, S, H, W, 4, generator=g)
        return disparity, K, torch.inverse(K), G, payload

    def taps(u, v, H, W):
        u = min(max(u, 0.0), W - 1)
        v = min(max(v, 0.0), H - 1)
        x0, y0 = int(u), int(v)
        x1, y1 = min(x0 + 1, W - 1), min(y0 + 1, H - 1)
        wx, wy = u - x0, v - y0
        return ((x0, y0, (1 - wx) * (1 - wy)), (x1, y0, wx * (1 - wy)),
                (x0, y1, (1 - wx) * wy), (x1, y1, wx * wy))

    for seed in range(3):
        disparity, K, K_inv, G, payload = make_scene(seed)
        B, S, H, W = payload.shape[:4]
        depths = torch.reciprocal(disparity)
        Hinv = tr.homography_tgt_to_src(G, depths, K_inv, K)
        grid = tr.make_meshgrid(H, W)
        uvh = torch.einsum("bsij,jhw->bsihw", Hinv, grid)
        uv = uvh[:, :, :2] / uvh[:, :, 2:]

        scatter = torch.zeros(B, S, H, W, 4)
        gather = torch.zeros(B, S, H, W, 4)
        for b in range(B):
            for s in range(S):
                U, V = uv[b, s, 0], uv[b, s, 1]
                interior = (U >= 0) & (U <= W - 1) & (V >= 0) & (V <= H - 1)
                for y in range(H):
                    for x in range(W):
                        for qx, qy, wt in taps(float(U[y, x]), float(V[y, x]),
                                               H, W):
                            scatter[b, s, qy, qx] += wt * payload[b, s, y, x]
                        if not interior[y, x]:  # rare path in the gather form
                            for qx, qy, wt in taps(float(U[y, x]),
                                                   float(V[y, x]), H, W):
                                gather[b, s, qy, qx] += wt * payload[b, s, y, x]
                Hf = torch.inverse(Hinv[b, s])
                for qy in range(H):
                    for qx in range(W):
                        pc = Hf @ torch.tensor([qx, qy, 1.0])
                        px, py = float(pc[0] / pc[2]), float(pc[1] / pc[2])
                        for ty in range(max(0, int(py) - 4), min(H, int(py) + 6)):
                            for tx in range(max(0, int(px) - 4), min(W, int(px) + 6)):
                                if not interior[ty, tx]:
                                    continue
                                du = float(U[ty, tx]) - qx
                                dv = float(V[ty, tx]) - qy
                                if -1 < du < 1 and -1 < dv < 1:
                                    gather[b, s, qy, qx] += (1 - abs(du)) * \
                                        (1 - abs(dv)) * payload[b, s, ty, tx]
        torch.testing.assert_close(gather, scatter, rtol=1e-4, atol=1e-5)


def test_igemm_kernel_coord_and_pack_math():
    """Simulates conv_igemm_fwd_kernel (ops/csrc/igemm_kernels.hip):
    fragment seg->(tap, c) decomposition over the general pack LUT and
    the (SA, SB, SD, SE) source-coordinate map, for forward (stride
    1/2, zero/reflect pad) AND the data-grad remap, against F.conv2d."""
    from mine_amd.ops.conv_general import _frag_lut

    def reflect(v, n):
        return _reflect1(v, n)

    def src(p, r, SA, SB, SD, SE, n, pad_mode):
        num = p * SA + r * SB + SD
        if SE > 1:
            if num % SE != 0:
                return None
            num //= SE
        if pad_mode == 1:
            return reflect(num, n)
        return num if 0 <= num < n else None

    def run_kernel(x, wfull, P, Q, K, SA, SB, SD, SE, pad_mode, R, S):
        B, C, Hs, Ws = x.shape
        Cv = C // 8
        nseg = R * S * Cv
        nchunks = (nseg + 3) // 4
        nK = (K + 15) // 16
        Cp = (C + 7) & ~7
        lut = _frag_lut(K, Cp, R, S, C, K, C, False,
                        torch.device("cpu")).long()
        flat = wfull.reshape(-1)
        wp = torch.where(lut >= 0, flat[lut.clamp(min=0)],
                         torch.zeros(())).view(nK, nchunks, 64, 8)
        xn = x.permute(0, 2, 3, 1)
        out = torch.zeros(B * P * Q, K)
        for m in range(B * P * Q):
            n, rem = divmod(m, P * Q)
            p, q = divmod(rem, Q)
            for kc in range(nchunks):
                for sub in range(4):
                    seg = kc * 4 + sub
                    if seg >= nseg:
                        continue
                    tap = seg // Cv
                    coct = seg - tap * Cv
                    r, s = divmod(tap, S)
                    ys = src(p, r, SA, SB, SD, SE, Hs, pad_mode)
                    xs = src(q, s, SA, SB, SD, SE, Ws, pad_mode)
                    if ys is None or xs is None:
                        continue
                    a = xn[n, ys, xs, coct * 8:coct * 8 + 8]
                    for nc in range(nK):
                        for j in range(16):
                            kout = nc * 16 + j
                            if kout >= K:
                                continue
                            b = wp[nc, kc, sub * 16 + j]
                            out[m, kout] += (a * b).sum()
        return out.view(B, P, Q, K).permute(0, 3, 1, 2)

    torch.manual_seed(9)
    B, C, H, W = 1, 8, 5, 6
    # fwd: 3x3 stride 1 pad 1 (zero), K=20 (ragged)
    x = torch.randn(B, C, H, W)
    w = torch.randn(20, C, 3, 3)
    got = run_kernel(x, w, H, W, 20, 1, 1, -1, 1, 0, 3, 3)
    torch.testing.assert_close(got, F.conv2d(x, w, padding=1),
                               rtol=1e-4, atol=1e-4)
    # fwd: 3x3 stride 2 pad 1
    got = run_kernel(x, w, (H - 1) // 2 + 1, (W - 1) // 2 + 1, 20,
                     2, 1, -1, 1, 0, 3, 3)
    torch.testing.assert_close(got, F.conv2d(x, w, stride=2, padding=1),
                               rtol=1e-4, atol=1e-4)
    # fwd: reflect pad 1
    got = run_kernel(x, w, H, W, 20, 1, 1, -1, 1, 1, 3, 3)
    torch.testing.assert_close(
        got, F.conv2d(F.pad(x, (1, 1, 1, 1), mode="reflect"), w),
        rtol=1e-4, atol=1e-4)
    # data-grad: stride 2 pad 1 (transposed weights, NOT flipped)
    P, Q = (H - 1) // 2 + 1, (W - 1) // 2 + 1
    gy = torch.randn(B, 20, P, Q)
    w_t = w.permute(1, 0, 2, 3).contiguous()
    # pad K-channel dim (20) for the LUT's 8-divisibility? 20 % 8 != 0:
    # pad gy and w_t to 24 channels like the wrapper does
    gy24 = torch.cat((gy, torch.zeros(B, 4, P, Q)), 1)
    w_t24 = torch.cat((w_t, torch.zeros(C, 4, 3, 3)), 1)
    got = run_kernel(gy24, w_t24, H, W, C, 1, -1, 1, 2, 0, 3, 3)
    xr = x.clone().requires_grad_(True)
    (F.conv2d(xr, w, stride=2, padding=1) * gy).sum().backward()
    torch.testing.assert_close(got, xr.grad, rtol=1e-4, atol=1e-4)


def _img_elem32(col, j):
    """igemm_kernels.hip img_elem32: the 32-pixel (NB=8) transpose-read
    block image used by the general conv weight-grad."""
    pblk = j >> 2
    bi = (pblk & 1) * 4 + (pblk >> 1)
    return bi * 64 + (j & 3) * 16 + col


def test_igemm_wrw_tr_image32_addressing():
    """Bijectivity + fragment reconstruction for the 32-px tr image:
    frag32's two tr reads (lane slice base + (l&15)*4 + (l>>4)*64, and
    +32*8) must deliver element (col=l&15, px=(l>>4)*8+e)."""
    img = {}
    for col in range(16):
        for j in range(32):
            off = _img_elem32(col, j)
            assert off not in img
            img[off] = (col, j)
    assert len(img) == 16 * 32

    for lane in range(64):
        g = lane >> 4
        for jj in range(4):
            # tr semantics (tools/tr_probe.hip): lane l elem jj reads
            # base + (l&15) + jj*16 + g*64
            off_lo = (lane & 15) + jj * 16 + g * 64
            assert img[off_lo] == (lane & 15, g * 8 + jj)
            off_hi = off_lo + 32 * 8
            assert img[off_hi] == (lane & 15, g * 8 + 4 + jj)


def test_igemm_wrw_kernel_math():
    """Simulates conv_igemm_wrw_kernel: (k16, c16, tap) blocks, 32-pixel
    chunks, forward-coordinate tap mapping — against autograd for a
    stride-2 zero-pad 3x3 and the reflect base-conv case."""
    torch.manual_seed(12)

    def src(p, r, SA, SB, SD, SE, n, reflect):
        num = p * SA + r * SB + SD
        if SE > 1:
            if num % SE != 0:
                return None
            num //= SE
        if reflect:
            return _reflect1(num, n)
        return num if 0 <= num < n else None

    def sim_wrw(x, gy, R, stride, pad, reflect):
        B, C, Hs, Ws = x.shape
        K, P, Q = gy.shape[1], gy.shape[2], gy.shape[3]
        xn = x.permute(0, 2, 3, 1).double()
        gyn = gy.permute(0, 2, 3, 1).double().reshape(-1, K)
        M = B * P * Q
        dw = torch.zeros(K, C, R, R, dtype=torch.float64)
        for k0 in range(0, K, 16):
            for c0 in range(0, C, 16):
                for tap in range(R * R):
                    r, s_ = divmod(tap, R)
                    acc = torch.zeros(16, 16, dtype=torch.float64)
                    for ch0 in range(0, M, 32):
                        A = torch.zeros(16, 32, dtype=torch.float64)
                        Bm = torch.zeros(16, 32, dtype=torch.float64)
                        for px in range(32):
                            m = ch0 + px
                            if m >= M:
                                continue
                            A[:, px] = gyn[m, k0:k0 + 16]
                            n, rem = divmod(m, P * Q)
                            p, q = divmod(rem, Q)
                            ys = src(p, r, stride, 1, -pad, 1, Hs, reflect)
                            xs = src(q, s_, stride, 1, -pad, 1, Ws, reflect)
                            if ys is not None and xs is not None:
                                Bm[:, px] = xn[n, ys, xs, c0:c0 + 16]
                        acc += A @ Bm.T
                    dw[k0:k0 + 16, c0:c0 + 16, r, s_] += acc
        return dw.float()

    # stride-2 zero-pad 3x3
    B, C, H, W, K = 2, 16, 6, 7, 16
    x = torch.randn(B, C, H, W)
    w = torch.zeros(K, C, 3, 3, requires_grad=True)
    y = F.conv2d(x, w, stride=2, padding=1)
    gy = torch.randn_like(y)
    (y * gy).sum().backward()
    got = sim_wrw(x, gy, 3, 2, 1, False)
    torch.testing.assert_close(got, w.grad, rtol=1e-4, atol=1e-4)

    # reflect-pad stride-1 (the SplitConvBlock base convs)
    w2 = torch.zeros(K, C, 3, 3, requires_grad=True)
    y2 = F.conv2d(F.pad(x, (1, 1, 1, 1), mode="reflect"), w2)
    gy2 = torch.randn_like(y2)
    (y2 * gy2).sum().backward()
    got2 = sim_wrw(x, gy2, 3, 1, 1, True)
    torch.testing.assert_close(got2, w2.grad, rtol=1e-4, atol=1e-4)
