"""Per-kernel GPU diagnostic: run each fused op at test sizes with
blocking launches, print progress; then time train steps at bench config."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def log(msg):
    print(msg, flush=True)
    torch.cuda.synchronize()

def mk(B=2, S=16, H=48, W=64, seed=0):
    g = torch.Generator().manual_seed(seed)
    rgb = torch.rand(B, S, 3, H, W, generator=g)
    sigma = torch.rand(B, S, 1, H, W, generator=g) * 3 + 1e-4
    disparity, _ = torch.sort(torch.rand(B, S, generator=g) * 0.95 + 0.02, dim=1, descending=True)
    f = 0.8 * W
    K = torch.tensor([[f, 0., W/2], [0., f, H/2], [0., 0., 1.]]).unsqueeze(0).repeat(B, 1, 1)
    K_inv = torch.inverse(K)
    G = torch.eye(4).unsqueeze(0).repeat(B, 1, 1)
    G[:, :3, 3] = 0.15 * torch.randn(B, 3, generator=g)
    img = torch.rand(B, 3, H, W, generator=g)
    dev = lambda t: t.to("cuda:0")
    return tuple(map(dev, (rgb, sigma, disparity, K, K_inv, G, img)))

def main():
    from mine_amd.ops.renderer import pack_mpi, render_src_view, render_tgt_view
    from mine_amd.ops.ssim import ssim

    rgb, sigma, disparity, K, K_inv, G, img = mk()
    mpi = pack_mpi(rgb, sigma)
    log("scene built")

    r, d, bl = render_src_view(mpi, disparity, K_inv)
    log(f"src fwd noblend ok {r.abs().mean().item():.4f}")
    r, d, bl = render_src_view(mpi, disparity, K_inv, src_img=img)
    log(f"src fwd blend ok {r.abs().mean().item():.4f}")

    rr = rgb.clone().requires_grad_(True); ss = sigma.clone().requires_grad_(True)
    m2 = pack_mpi(rr, ss)
    r, d, bl = render_src_view(m2, disparity, K_inv, src_img=img)
    (r.sum() + d.sum() + bl.sum()).backward()
    log(f"src bwd ok {rr.grad.abs().mean().item():.6f} {ss.grad.abs().mean().item():.6f}")

    r, d, mask = render_tgt_view(mpi, disparity, G, K_inv, K)
    log(f"tgt fwd ok {r.abs().mean().item():.4f} mask {mask.mean().item():.2f}")

    rr = rgb.clone().requires_grad_(True); ss = sigma.clone().requires_grad_(True)
    m2 = pack_mpi(rr, ss)
    r, d, mask = render_tgt_view(m2, disparity, G, K_inv, K)
    (r.sum() + d.sum()).backward()
    log(f"tgt bwd ok {rr.grad.abs().mean().item():.6f} {ss.grad.abs().mean().item():.6f}")

    a = torch.rand(2, 3, 64, 96, device="cuda:0"); b = torch.rand(2, 3, 64, 96, device="cuda:0")
    s = ssim(a, b)
    log(f"ssim fwd ok {s.item():.4f} (torch ref {ssim(a, b, force_torch=True).item():.4f})")
    a.requires_grad_(True)
    (1 - ssim(a, b)).backward()
    log(f"ssim bwd ok {a.grad.abs().mean().item():.6f}")

    # bench-config step timing
    from mine_amd.config import default_config
    from mine_amd.data import SyntheticMPIDataset, collate_src_tgt
    from mine_amd.engine import SynthesisTask
    for dtype in ("fp32", "bf16"):
        cfg = default_config(**{
            "data.name": "realestate10k", "data.img_h": 256, "data.img_w": 384,
            "mpi.num_bins_coarse": 64, "data.per_gpu_batch_size": 4,
            "data.visible_point_count": 256, "lr.decay_steps": [4, 8],
            "training.amp_dtype": dtype,
        })
        ds = SyntheticMPIDataset(cfg, length=4)
        items = collate_src_tgt([ds[i] for i in range(4)])
        task = SynthesisTask(cfg, device="cuda:0")
        t0 = time.time()
        task.train_step(items)
        torch.cuda.synchronize()
        log(f"[{dtype}] first step {time.time()-t0:.1f}s")
        t0 = time.time()
        for _ in range(3):
            task.train_step(items)
        torch.cuda.synchronize()
        dt = (time.time()-t0)/3
        log(f"[{dtype}] steady step {dt*1000:.0f} ms -> {4/dt:.1f} imgs/s/gpu")

if __name__ == "__main__":
    main()
