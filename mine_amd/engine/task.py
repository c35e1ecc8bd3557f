"""SynthesisTask — the training/eval engine.

Re-implements the reference's orchestration layer (ref synthesis_task.py)
on top of the fused MI355X rendering ops:

  CS1 loop:  set_data -> loss_fcn (network_forward + 4-scale losses) ->
             backward (bucketed RCCL all-reduce overlapped) -> Adam step
  CS2:       stratified disparity sampling -> encoder taps -> MPI decoder
             (B -> B*S), optional coarse-to-fine PDF resampling
  CS3:       per scale: fused src composite (+RGB blending) ->
             scale factor from sparse COLMAP points -> fused novel-view
             render -> L1/SSIM/edge-aware/log-disparity losses

Differences from the reference, by design:
  * conv stack runs in bf16 autocast + channels_last (fp32 available via
    `training.amp_dtype: fp32`); all rendering/losses in fp32
  * no nn.DataParallel / find_unused_parameters / SyncBN-by-default
  * no per-step `torch.cuda.synchronize()` workarounds (closed-form
    inverses replace the nan-retrying torch.inverse)
"""
from __future__ import annotations

import contextlib
import os
import time
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn.functional as F

from mine_amd.config import Config, RuntimeState
from mine_amd.engine.checkpoint import restore_model, save_checkpoint
from mine_amd.models import MPIDecoder, ResNetEncoder
from mine_amd.ops import (
    edge_aware_loss,
    edge_aware_loss_v2,
    gather_pixel_by_pxpy,
    psnr,
    render_src_view,
    render_tgt_view,
    sample_disparity_from_bins,
    sample_disparity_linspace,
    sample_pdf,
    ssim,
)
from mine_amd.ops import torch_ref
from mine_amd.parallel import GradAllReduceEngine
from mine_amd.utils import AverageMeter
from mine_amd.utils.geometry import inverse_3x3, inverse_rigid_4x4

_LOSS_KEYS = ("loss", "loss_rgb_src", "loss_ssim_src", "loss_disp_pt3dsrc",
              "loss_rgb_tgt", "loss_ssim_tgt", "lpips_tgt", "psnr_tgt",
              "loss_disp_pt3dtgt")


def get_disparity_list(config: Config, B: int, device) -> torch.Tensor:
    """Stratified (or fixed) coarse disparity list, BxS descending
    (ref synthesis_task.py:31-60)."""
    S = config["mpi.num_bins_coarse"]
    start, end = config["mpi.disparity_start"], config["mpi.disparity_end"]
    disp_list = config.get("mpi.disparity_list", None)
    has_list = disp_list is not None and len(disp_list) == S + 1

    if config.get("mpi.fix_disparity", False):
        if has_list:
            d = torch.as_tensor(disp_list[1:], dtype=torch.float32, device=device)
            return d.unsqueeze(0).repeat(B, 1)
        return torch.linspace(start, end, S, dtype=torch.float32,
                              device=device).unsqueeze(0).repeat(B, 1)
    if has_list:
        return sample_disparity_from_bins(B, disp_list, device=device)
    return sample_disparity_linspace(B, S, start, end, device=device)


class SynthesisTask:
    def __init__(self, config: Config, state: Optional[RuntimeState] = None,
                 logger=None, is_val: bool = False, device: Optional[str] = None):
        self.config = config
        self.state = state or RuntimeState()
        self.logger = logger
        self.is_val = is_val

        if device is None:
            device = "cuda:0" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.is_gpu = self.device.type == "cuda"

        if self.is_gpu:
            # The reference sets torch.backends.cudnn.benchmark = True
            # (ref train.py:111-112). On this ROCm 7.2 / MIOpen build the
            # exhaustive-find path it triggers memory-faults the GPU
            # (observed on MI355X during find at the flagship shapes), so
            # it is OFF by default; MIOPEN_FIND_MODE (set to FAST in
            # mine_amd/__init__.py) governs solver selection instead.
            torch.backends.cudnn.benchmark = bool(
                config.get("training.miopen_benchmark", False))

        amp = str(config.get("training.amp_dtype", "bf16")).lower()
        self.amp_dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
                          "fp32": None, "float32": None}[amp]
        # fp16 needs loss scaling (bf16/fp32 do not; driver config 5 runs
        # Flowers in fp16)
        self.grad_scaler = torch.amp.GradScaler("cuda") \
            if (self.is_gpu and self.amp_dtype == torch.float16) else None
        self.channels_last = bool(config.get("training.channels_last", True)) and self.is_gpu

        # ---- models -----------------------------------------------------
        self.backbone = ResNetEncoder(num_layers=50).to(self.device)
        self.decoder = MPIDecoder(
            num_ch_enc=self.backbone.num_ch_enc,
            pos_encoding_multires=config["model.pos_encoding_multires"],
            use_alpha=config.get("mpi.use_alpha", False),
            scales=range(4),
            use_skips=True,
        ).to(self.device)
        if self.channels_last:
            self.backbone = self.backbone.to(memory_format=torch.channels_last)
            self.decoder = self.decoder.to(memory_format=torch.channels_last)

        # ---- optimizer (two LR groups, ref synthesis_task.py:83-87) ------
        self.optimizer = torch.optim.Adam(
            [{"params": self.backbone.parameters(), "lr": config["lr.backbone_lr"]},
             {"params": self.decoder.parameters(), "lr": config["lr.decoder_lr"]}],
            weight_decay=config["lr.weight_decay"])

        # Restore BEFORE the parameter broadcast (ref CS5). Every rank
        # that can see the checkpoint file loads it (workspace is a shared
        # path on one node) — restoring on rank 0 only left the other
        # ranks with cold Adam moments and no resume meta, desyncing
        # epoch counts/LR schedules across ranks (round-1 ADVICE.md).
        # "auto" resumes from the workspace's checkpoint_latest.pth when
        # one exists — an elastic-restart convenience the reference
        # lacked (it always restarted at epoch 1).
        ckpt_path = config.get("training.pretrained_checkpoint_path")
        if ckpt_path == "auto":
            cand = os.path.join(self.state.local_workspace or "",
                                "checkpoint_latest.pth")
            ckpt_path = cand if os.path.exists(cand) else None
        self._restored_meta = {}
        restored_here = False
        if ckpt_path and (self.state.is_rank0 or os.path.exists(ckpt_path)):
            # non-rank0 ranks skip silently if the file is unreachable
            # (rank 0 keeps the reference's hard assert); they then
            # receive the state by broadcast below.
            self._restored_meta = restore_model(
                ckpt_path, self.backbone, self.decoder, self.optimizer,
                logger=logger if self.state.is_rank0 else None) or {}
            restored_here = True
        self._sync_restored_state(restored_here)

        self.grad_engine = None
        if not is_val:
            if bool(config.get("training.sync_batchnorm", False)) and \
                    torch.distributed.is_initialized() and \
                    torch.distributed.get_world_size() > 1:
                # cross-rank BN statistics (the reference's SyncBatchNorm
                # role, ref synthesis_task.py:106-112) via FusedBNAct's own
                # stat all-reduce — torch's convert_sync_batchnorm would
                # drop the fused activation epilogues.
                from mine_amd.ops.bn import FusedBNAct
                for m in list(self.backbone.modules()) + list(self.decoder.modules()):
                    if isinstance(m, FusedBNAct):
                        m.sync = True
            ar_dtype = {"fp32": None, "bf16": torch.bfloat16}[
                str(config.get("training.grad_allreduce_dtype", "fp32"))]
            self.grad_engine = GradAllReduceEngine(
                [self.backbone, self.decoder],
                bucket_mb=float(config.get("training.grad_bucket_mb", 25)),
                allreduce_dtype=ar_dtype,
                timing=bool(config.get("training.comm_timing", False)))
            self.lr_scheduler = torch.optim.lr_scheduler.MultiStepLR(
                self.optimizer, config["lr.decay_steps"],
                gamma=config["lr.decay_gamma"])
            self.backbone.train()
            self.decoder.train()
        else:
            self.backbone.eval()
            self.decoder.eval()

        if self.is_gpu and bool(config.get("training.stream_overlap", False)):
            # EXPERIMENTAL (off by default; docs/NEXT.md #4): run the
            # SplitConvBlocks' batch-B base convs on a side HIP stream,
            # overlapped with the batch-B*S dec convs.
            from mine_amd.models.decoder import SplitConvBlock
            side = torch.cuda.Stream()
            for m in self.decoder.modules():
                if isinstance(m, SplitConvBlock):
                    m.side_stream = side

        self.use_alpha = bool(config.get("mpi.use_alpha", False))
        self.bg_depth_inf = bool(config.get("mpi.is_bg_depth_inf", False))
        self.src_rgb_blending = bool(config.get("training.src_rgb_blending", True))
        self.use_multi_scale = bool(config.get("training.use_multi_scale", True))
        self.scale_factor_is_one = config["data.name"] in ("flowers", "kitti_raw", "dtu")

        self.lpips_model = None
        if bool(config.get("eval.lpips", False)) and self.state.is_rank0:
            from mine_amd.ops.lpips import LPIPS
            self.lpips_model = LPIPS().to(self.device).eval()

        self.train_losses = {k: AverageMeter("train_" + k) for k in _LOSS_KEYS}
        self.val_losses = {k: AverageMeter("val_" + k) for k in _LOSS_KEYS
                           if k != "loss"}

        self.current_epoch = 0
        self.global_step = 0
        # device-side NaN-guard skip counter (read on logging steps only)
        self._nan_skip_count: Optional[torch.Tensor] = None

    # ------------------------------------------------------------------
    def _sync_restored_state(self, restored_here: bool) -> None:
        """Make resume state rank-coherent (round-1 ADVICE.md items 1-2):
        every rank must agree on the resume meta (epoch / global_step /
        LR fast-forward — a mismatch desyncs per-rank collective counts),
        and ranks that could not read the checkpoint file receive the
        Adam state from rank 0 (parameters are broadcast separately by
        GradAllReduceEngine at construction)."""
        if self.is_val or not torch.distributed.is_initialized() or \
                torch.distributed.get_world_size() <= 1:
            return
        dist = torch.distributed
        flags = [None] * dist.get_world_size()
        dist.all_gather_object(flags, bool(restored_here))
        payload = {"meta": self._restored_meta}
        if any(flags) and not all(flags) and self.state.is_rank0:
            # ship Adam state to the ranks that missed the file
            opt_sd = self.optimizer.state_dict()

            def to_cpu(x):
                if torch.is_tensor(x):
                    return x.cpu()
                if isinstance(x, dict):
                    return {k: to_cpu(v) for k, v in x.items()}
                if isinstance(x, list):
                    return [to_cpu(v) for v in x]
                return x
            payload["optimizer"] = to_cpu(opt_sd)
        obj = [payload if self.state.is_rank0 else None]
        dist.broadcast_object_list(obj, src=0)
        if not self.state.is_rank0:
            self._restored_meta = obj[0].get("meta") or {}
            opt_sd = obj[0].get("optimizer")
            if opt_sd is not None and not restored_here:
                self.optimizer.load_state_dict(opt_sd)

    def _autocast(self):
        if self.is_gpu and self.amp_dtype is not None:
            return torch.autocast(device_type="cuda", dtype=self.amp_dtype)
        return contextlib.nullcontext()

    def set_data(self, items, static: bool = False) -> None:
        """Stage a batch on the device (ref synthesis_task.py:184-209).

        static=True copies into persistent device buffers (same shapes
        every step) — the staging the hipGraph-captured step reads."""
        src_items, tgt_items = items
        dev = self.device

        def tod(x):
            return x.to(dev, non_blocking=True)

        L = tgt_items["img"].shape[1]
        assert L == 1, "one target supervision view (ref synthesis_task.py:200-201)"

        if static and getattr(self, "_static_staged", False):
            self.src_imgs.copy_(src_items["img"], non_blocking=True)
            self.K_src.copy_(src_items["K"], non_blocking=True)
            self.K_src_inv.copy_(src_items["K_inv"], non_blocking=True)
            self.pt3d_src.copy_(src_items["xyzs"], non_blocking=True)
            self.tgt_imgs.copy_(tgt_items["img"].squeeze(1), non_blocking=True)
            self.G_src_tgt.copy_(tgt_items["G_src_tgt"].squeeze(1),
                                 non_blocking=True)
            self.K_tgt.copy_(tgt_items["K"].squeeze(1), non_blocking=True)
            self.K_tgt_inv.copy_(tgt_items["K_inv"].squeeze(1),
                                 non_blocking=True)
            self.pt3d_tgt.copy_(tgt_items["xyzs"].squeeze(1),
                                non_blocking=True)
            self.G_tgt_src.copy_(inverse_rigid_4x4(self.G_src_tgt))
            return

        self.src_imgs = tod(src_items["img"]).float()  # Bx3xHxW
        self.K_src = tod(src_items["K"]).float()
        self.K_src_inv = tod(src_items["K_inv"]).float()
        self.pt3d_src = tod(src_items["xyzs"]).float()  # Bx3xN_pt

        self.tgt_imgs = tod(tgt_items["img"]).float().squeeze(1)
        self.G_src_tgt = tod(tgt_items["G_src_tgt"]).float().squeeze(1)
        self.K_tgt = tod(tgt_items["K"]).float().squeeze(1)
        self.K_tgt_inv = tod(tgt_items["K_inv"]).float().squeeze(1)
        self.pt3d_tgt = tod(tgt_items["xyzs"]).float().squeeze(1)

        # closed-form rigid inverse (the reference retried torch.inverse
        # around cuda.synchronize here; ref synthesis_task.py:208-209)
        self.G_tgt_src = inverse_rigid_4x4(self.G_src_tgt)

        if self.channels_last:
            self.src_imgs = self.src_imgs.contiguous(memory_format=torch.channels_last)
            self.tgt_imgs = self.tgt_imgs.contiguous(memory_format=torch.channels_last)
        if static:
            # first static call: the tensors staged above BECOME the
            # persistent buffers
            self._static_staged = True

    # ------------------------------------------------------------------
    def mpi_predictor(self, src_imgs: torch.Tensor, disparity: torch.Tensor
                      ) -> List[torch.Tensor]:
        """Backbone + decoder -> 4 per-scale packed MPIs (B,S,H_s,W_s,4) fp32.

        The decoder packs via the fused head kernel (one pass; the eager
        sigmoid/abs/cat/permute chain was a top profile entry)."""
        with self._autocast():
            feats = self.backbone(src_imgs)
            outputs = self.decoder(feats, disparity, packed=True)
        return [outputs[("disp", s)] for s in range(4)]

    def network_forward(self) -> Dict[str, object]:
        """CS2: disparity sampling -> (coarse-to-fine) MPI prediction."""
        B = self.src_imgs.shape[0]
        S_fine = self.config["mpi.num_bins_fine"]
        disparity_coarse = get_disparity_list(self.config, B, self.device)

        if S_fine > 0:
            with torch.no_grad():
                coarse = self.mpi_predictor(self.src_imgs, disparity_coarse)[0]
                rgb, sigma = coarse[..., 0:3], coarse[..., 3:4]
                rgb = rgb.permute(0, 1, 4, 2, 3)
                sigma = sigma.permute(0, 1, 4, 2, 3)
                grid = torch_ref.make_meshgrid(rgb.shape[-2], rgb.shape[-1],
                                               device=self.device)
                xyz = torch_ref.src_plane_xyz(grid, disparity_coarse, self.K_src_inv)
                _, _, _, weights = torch_ref.volume_composite(
                    rgb, sigma, xyz, self.bg_depth_inf)
                w = weights.mean((2, 3, 4)).unsqueeze(1).unsqueeze(2)  # Bx1x1xS
                vals = disparity_coarse.unsqueeze(1).unsqueeze(2)
                fine = sample_pdf(vals, w, S_fine).squeeze(2).squeeze(1)
                disparity_all = torch.cat((disparity_coarse, fine), dim=1)
                disparity_all, _ = torch.sort(disparity_all, dim=1, descending=True)
        else:
            disparity_all = disparity_coarse

        mpis = self.mpi_predictor(self.src_imgs, disparity_all)
        return {"mpi_all_src_list": mpis, "disparity_all_src": disparity_all}

    # ------------------------------------------------------------------
    def compute_scale_factor(self, disp_syn_pt3d: torch.Tensor,
                             pt3d_disp: torch.Tensor) -> torch.Tensor:
        """exp(mean(log syn - log gt)) per image, or ones for metric
        datasets (ref synthesis_task.py:211-220)."""
        B = pt3d_disp.shape[0]
        if self.scale_factor_is_one:
            return torch.ones(B, dtype=torch.float32, device=self.device)
        return torch.exp(torch.mean(
            torch.log(disp_syn_pt3d) - torch.log(pt3d_disp),
            dim=2, keepdim=False)).squeeze(1)

    def render_novel_view(self, mpi_packed: torch.Tensor, disparity: torch.Tensor,
                          G_tgt_src: torch.Tensor, K_src_inv: torch.Tensor,
                          K_tgt: torch.Tensor, scale_factor=None
                          ) -> Dict[str, torch.Tensor]:
        """Scale-factored novel-view render (ref synthesis_task.py:435-474)."""
        if scale_factor is not None:
            with torch.no_grad():
                G_tgt_src = G_tgt_src.clone()
                sf = scale_factor if torch.is_tensor(scale_factor) else \
                    torch.as_tensor(scale_factor, dtype=torch.float32,
                                    device=G_tgt_src.device)
                G_tgt_src[:, 0:3, 3] = G_tgt_src[:, 0:3, 3] / sf.view(-1, 1)
        tgt_rgb, tgt_depth, tgt_mask = render_tgt_view(
            mpi_packed, disparity, G_tgt_src, K_src_inv, K_tgt,
            bg_depth_inf=self.bg_depth_inf, use_alpha=self.use_alpha)
        return {"tgt_imgs_syn": tgt_rgb,
                "tgt_disparity_syn": torch.reciprocal(tgt_depth),
                "tgt_mask_syn": tgt_mask}

    # ------------------------------------------------------------------
    def loss_fcn_per_scale(self, scale: int, mpi_packed: torch.Tensor,
                           disparity: torch.Tensor, scale_factor=None,
                           is_val: bool = False,
                           monitors: bool = True) -> Tuple[dict, dict, torch.Tensor]:
        cfg = self.config
        # pyramid sized from the decoder's OWN per-scale output: at
        # non-power-of-two resolutions (LLFF 504x378) H//2^s and the
        # up-stage chain disagree by one pixel (the reference crashes
        # there); intrinsics scale per-axis by the actual ratio.
        Hs_m, Ws_m = int(mpi_packed.shape[2]), int(mpi_packed.shape[3])
        if scale == 0 and (Hs_m, Ws_m) == tuple(self.src_imgs.shape[-2:]):
            src_scaled, tgt_scaled = self.src_imgs, self.tgt_imgs
        else:
            src_scaled = F.interpolate(self.src_imgs, size=(Hs_m, Ws_m),
                                       mode="nearest")
            tgt_scaled = F.interpolate(self.tgt_imgs, size=(Hs_m, Ws_m),
                                       mode="nearest")
        B = src_scaled.shape[0]

        sy = Hs_m / self.src_imgs.shape[-2]
        sx = Ws_m / self.src_imgs.shape[-1]
        # python scalars fold into the mul kernels (no host tensor: a
        # pageable H2D would break hipGraph capture)
        K_src_scaled = self.K_src.clone()
        K_src_scaled[:, 0] = K_src_scaled[:, 0] * sx
        K_src_scaled[:, 1] = K_src_scaled[:, 1] * sy
        K_tgt_scaled = self.K_tgt.clone()
        K_tgt_scaled[:, 0] = K_tgt_scaled[:, 0] * sx
        K_tgt_scaled[:, 1] = K_tgt_scaled[:, 1] * sy
        K_src_scaled_inv = inverse_3x3(K_src_scaled)

        # ---- fused src composite + RGB blending -------------------------
        src_imgs_syn, src_depth_syn, mpi_blend = render_src_view(
            mpi_packed, disparity, K_src_scaled_inv,
            src_img=src_scaled if self.src_rgb_blending else None,
            bg_depth_inf=self.bg_depth_inf, use_alpha=self.use_alpha)
        src_disparity_syn = torch.reciprocal(src_depth_syn)

        # ---- scale factor from sparse COLMAP points ---------------------
        src_pt3d_disp = torch.reciprocal(self.pt3d_src[:, 2:, :])
        src_pt3d_pxpy = torch.matmul(K_src_scaled, self.pt3d_src)
        src_pt3d_pxpy = src_pt3d_pxpy[:, 0:2] / src_pt3d_pxpy[:, 2:]
        src_pt3d_disp_syn = gather_pixel_by_pxpy(src_disparity_syn, src_pt3d_pxpy)
        if scale_factor is None:
            scale_factor = self.compute_scale_factor(src_pt3d_disp_syn, src_pt3d_disp)

        # ---- fused novel-view render ------------------------------------
        render_results = self.render_novel_view(mpi_blend, disparity,
                                                self.G_tgt_src, K_src_scaled_inv,
                                                K_tgt_scaled, scale_factor)
        tgt_imgs_syn = render_results["tgt_imgs_syn"]
        tgt_disparity_syn = render_results["tgt_disparity_syn"]
        tgt_mask_syn = render_results["tgt_mask_syn"]

        # ---- losses (ref synthesis_task.py:296-351) ---------------------
        disp_lambda = 0.0 if self.scale_factor_is_one else 1.0
        lam_v1 = cfg.get("loss.smoothness_lambda_v1", 0.5)
        lam_v2 = cfg.get("loss.smoothness_lambda_v2", 1.0)
        gmin = cfg["loss.smoothness_gmin"]
        grad_ratio = cfg.get("loss.smoothness_grad_ratio", 0.1)

        # The src-view L1/SSIM/smooth-v1 terms are MONITORS (no_grad, ref
        # synthesis_task.py:301-306): only computed on steps whose values
        # are read (logging/eval) — the hot loop skips ~3 loss pipelines
        # x 4 scales per step.
        with torch.no_grad():
            if monitors:
                loss_rgb_src = torch.mean(torch.abs(src_imgs_syn - src_scaled))
                loss_ssim_src = 1.0 - ssim(src_imgs_syn, src_scaled)
                loss_smooth_src = edge_aware_loss(
                    src_scaled, src_disparity_syn,
                    gmin=gmin, grad_ratio=grad_ratio)
            else:
                zero = torch.zeros((), device=self.device)
                loss_rgb_src = loss_ssim_src = loss_smooth_src = zero

        src_pt3d_disp_syn_scaled = src_pt3d_disp_syn / scale_factor.view(B, 1, 1)
        loss_disp_pt3dsrc = disp_lambda * torch.mean(torch.abs(
            torch.log(src_pt3d_disp_syn_scaled) - torch.log(src_pt3d_disp)))

        tgt_pt3d_disp = torch.reciprocal(self.pt3d_tgt[:, 2:, :])
        tgt_pt3d_pxpy = torch.matmul(K_tgt_scaled, self.pt3d_tgt)
        tgt_pt3d_pxpy = tgt_pt3d_pxpy[:, 0:2] / tgt_pt3d_pxpy[:, 2:]
        tgt_pt3d_disp_syn = gather_pixel_by_pxpy(tgt_disparity_syn, tgt_pt3d_pxpy)
        tgt_pt3d_disp_syn_scaled = tgt_pt3d_disp_syn / scale_factor.view(B, 1, 1)
        loss_disp_pt3dtgt = disp_lambda * torch.mean(torch.abs(
            torch.log(tgt_pt3d_disp_syn_scaled) - torch.log(tgt_pt3d_disp)))

        valid_mask = torch.ge(tgt_mask_syn,
                              cfg["mpi.valid_mask_threshold"]).to(torch.float32)
        loss_rgb_tgt = (torch.abs(tgt_imgs_syn - tgt_scaled) * valid_mask).mean()

        if lam_v1 != 0.0:
            loss_smooth_tgt = lam_v1 * edge_aware_loss(
                tgt_scaled, tgt_disparity_syn, gmin=gmin,
                grad_ratio=grad_ratio)
        elif monitors:
            with torch.no_grad():
                loss_smooth_tgt = edge_aware_loss(
                    tgt_scaled, tgt_disparity_syn, gmin=gmin,
                    grad_ratio=grad_ratio) * 0.0
        else:
            loss_smooth_tgt = torch.zeros((), device=self.device)
        loss_smooth_tgt_v2 = lam_v2 * edge_aware_loss_v2(tgt_scaled, tgt_disparity_syn)
        loss_smooth_src_v2 = lam_v2 * edge_aware_loss_v2(src_scaled, src_disparity_syn)
        loss_ssim_tgt = 1.0 - ssim(tgt_imgs_syn, tgt_scaled)

        with torch.no_grad():
            if self.lpips_model is not None and is_val and scale == 0:
                lpips_tgt = self.lpips_model(tgt_imgs_syn, tgt_scaled).mean()
            else:
                # zeros((), device) is a device-side fill (graph-capture
                # safe; torch.tensor(0.0, device=...) is a pageable H2D)
                lpips_tgt = torch.zeros((), device=self.device)
            psnr_tgt = psnr(tgt_imgs_syn, tgt_scaled) if monitors else \
                torch.zeros((), device=self.device)

        loss = (loss_disp_pt3dtgt + loss_disp_pt3dsrc
                + loss_rgb_tgt + loss_ssim_tgt
                + loss_smooth_tgt
                + loss_smooth_src_v2 + loss_smooth_tgt_v2)

        loss_dict = {"loss": loss,
                     "loss_rgb_src": loss_rgb_src,
                     "loss_ssim_src": loss_ssim_src,
                     "loss_disp_pt3dsrc": loss_disp_pt3dsrc,
                     "loss_smooth_src": loss_smooth_src,
                     "loss_smooth_tgt": loss_smooth_tgt,
                     "loss_smooth_src_v2": loss_smooth_src_v2,
                     "loss_smooth_tgt_v2": loss_smooth_tgt_v2,
                     "loss_rgb_tgt": loss_rgb_tgt,
                     "loss_ssim_tgt": loss_ssim_tgt,
                     "lpips_tgt": lpips_tgt,
                     "psnr_tgt": psnr_tgt,
                     "loss_disp_pt3dtgt": loss_disp_pt3dtgt}
        vis_dict = {"src_disparity_syn": src_disparity_syn,
                    "tgt_disparity_syn": tgt_disparity_syn,
                    "tgt_imgs_syn": tgt_imgs_syn,
                    "tgt_mask_syn": tgt_mask_syn,
                    "src_imgs_syn": src_imgs_syn}
        return loss_dict, vis_dict, scale_factor

    def loss_fcn(self, is_val: bool, monitors: bool = True) -> Tuple[dict, dict]:
        endpoints = self.network_forward()
        mpis = endpoints["mpi_all_src_list"]
        disparity = endpoints["disparity_all_src"]

        scale_factor = None
        loss_dicts, vis_dicts = [], []
        for scale in range(4):
            ld, vd, scale_factor = self.loss_fcn_per_scale(
                scale, mpis[scale], disparity, scale_factor, is_val=is_val,
                monitors=monitors)
            loss_dicts.append(ld)
            vis_dicts.append(vd)

        loss_dict = loss_dicts[0]
        for s in range(1, 4):
            if self.use_multi_scale:
                loss_dict["loss"] = loss_dict["loss"] + \
                    loss_dicts[s]["loss_rgb_tgt"] + loss_dicts[s]["loss_ssim_tgt"]
            loss_dict["loss"] = loss_dict["loss"] + \
                loss_dicts[s]["loss_disp_pt3dsrc"] + loss_dicts[s]["loss_disp_pt3dtgt"] + \
                loss_dicts[s]["loss_smooth_src_v2"] + loss_dicts[s]["loss_smooth_tgt_v2"]
        return loss_dict, vis_dicts[0]

    # ------------------------------------------------------------------
    def train_step(self, items) -> dict:
        """One optimization step; returns the loss dict."""
        timers = getattr(self, "phase_timers", None)

        def mark(name):
            if timers is not None:
                if self.is_gpu:
                    torch.cuda.synchronize()
                timers.setdefault(name, []).append(time.perf_counter())

        mark("t0")
        self.set_data(items)
        mark("set_data")
        # monitor-only loss terms are read on logging steps only
        log_every = int(self.config.get("training.log_interval", 10))
        want_monitors = (self.global_step % log_every == 0) or \
            bool(self.config.get("training.always_monitors", False))
        loss_dict, _ = self.loss_fcn(is_val=False, monitors=want_monitors)
        mark("forward")
        if self.grad_engine is not None:
            self.grad_engine.zero_grad()
        else:
            self.optimizer.zero_grad(set_to_none=False)
        if self.grad_scaler is not None:
            # fp16: the GradScaler already skips non-finite steps, and it
            # checks the REDUCED gradients (finish_step runs before
            # scaler.step), so the skip decision is rank-coherent.
            self.grad_scaler.scale(loss_dict["loss"]).backward()
            if self.grad_engine is not None:
                self.grad_engine.finish_step()
            mark("backward")
            self.grad_scaler.step(self.optimizer)
            self.grad_scaler.update()
        else:
            loss_dict["loss"].backward()
            if self.grad_engine is not None:
                self.grad_engine.finish_step()
            mark("backward")
            # NaN guard (absent in the reference — SURVEY section 5c),
            # fully DEVICE-SIDE: a non-finite loss zeroes the gradients so
            # nothing poisons the parameters. No `.item()` host sync in
            # the hot loop (the round-1 version stalled the pipeline every
            # step — ADVICE.md item 3); the skip count is a device counter
            # read only on (already-syncing) logging steps. The finite
            # flag is all-reduced with MIN so every rank gates alike and
            # the optimizer states stay bit-identical across ranks.
            # (Adam still applies its momentum-decay update on a gated
            # step — the parameters move slightly along stale momentum but
            # no NaN enters; a pure skip would need a host sync.)
            if bool(self.config.get("training.nan_guard", True)):
                finite = torch.isfinite(
                    loss_dict["loss"].detach()).to(torch.float32)
                if torch.distributed.is_initialized() and \
                        torch.distributed.get_world_size() > 1:
                    torch.distributed.all_reduce(
                        finite, op=torch.distributed.ReduceOp.MIN)
                if self._nan_skip_count is None or \
                        self._nan_skip_count.device != finite.device:
                    self._nan_skip_count = torch.zeros_like(finite)
                self._nan_skip_count += 1.0 - finite
                if self.grad_engine is not None:
                    grads = [b.flat for b in self.grad_engine.buckets]
                else:
                    grads = [p.grad for g in self.optimizer.param_groups
                             for p in g["params"] if p.grad is not None]
                # scale finite grads by the flag, then map NaN/inf -> 0
                # (0 * NaN is NaN, so the multiply alone cannot zero a
                # poisoned buffer); on healthy steps nan_to_num_ is a
                # no-op pass over the ~6 bucket flats (~40 us at HBM bw)
                torch._foreach_mul_(grads, finite)
                for g in grads:
                    torch.nan_to_num_(g, nan=0.0, posinf=0.0, neginf=0.0)
            self.optimizer.step()
        mark("optimizer")
        return loss_dict

    # ------------------------------------------------------------------
    # hipGraph-captured train step (docs/NEXT round-1 item; single GPU)
    # ------------------------------------------------------------------
    def _graph_step_body(self) -> dict:
        """Everything one optimization step does on the device, with no
        host synchronization — the region the hipGraph captures."""
        if self.grad_engine is not None:
            self.grad_engine.zero_grad()
        else:
            self.optimizer.zero_grad(set_to_none=False)
        loss_dict, _ = self.loss_fcn(is_val=False, monitors=False)
        loss_dict["loss"].backward()
        if self.grad_engine is not None:
            self.grad_engine.finish_step()
        if bool(self.config.get("training.nan_guard", True)):
            finite = torch.isfinite(loss_dict["loss"].detach()).to(torch.float32)
            if self._nan_skip_count is None:
                self._nan_skip_count = torch.zeros_like(finite)
            self._nan_skip_count += 1.0 - finite
            grads = [b.flat for b in self.grad_engine.buckets] \
                if self.grad_engine is not None else \
                [p.grad for g in self.optimizer.param_groups
                 for p in g["params"] if p.grad is not None]
            torch._foreach_mul_(grads, finite)
            for g in grads:
                torch.nan_to_num_(g, nan=0.0, posinf=0.0, neginf=0.0)
        self.optimizer.step()
        return loss_dict

    def enable_graph_step(self, items) -> bool:
        """Capture forward+backward+optimizer into ONE hipGraph over
        static input buffers; later steps copy the batch in and replay
        (~600 launches collapse into one). Single-GPU, bf16/fp32, fresh
        Adam state only; returns False (eager path keeps working) when
        any precondition fails."""
        if not (self.is_gpu and self.grad_scaler is None):
            return False
        if torch.distributed.is_initialized() and \
                torch.distributed.get_world_size() > 1:
            return False
        if len(self.optimizer.state) > 0:
            return False  # capturable Adam needs device-side step state
        try:
            # the library conv path allocates workspace per call, which
            # hipGraph capture forbids: force the hand-written kernels
            from mine_amd.ops.conv import set_force_igemm
            set_force_igemm(True)
            # bias correction must be computed on-device per replay
            for g in self.optimizer.param_groups:
                g["capturable"] = True
            self.set_data(items, static=True)
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):  # warmup: allocator + lazy state
                    self._graph_step_body()
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            self._graph = torch.cuda.CUDAGraph()
            # thread_local: autograd's device worker threads must be
            # allowed to record into the capturing stream
            with torch.cuda.graph(self._graph,
                                  capture_error_mode="thread_local"):
                self._graph_loss = self._graph_step_body()
            return True
        except Exception as exc:  # pragma: no cover - depends on runtime
            from mine_amd.ops.conv import set_force_igemm
            set_force_igemm(False)
            self._graph_error = f"{type(exc).__name__}: {exc}"
            if self.logger:
                self.logger.warning("hipGraph capture failed (%s); "
                                    "eager step", self._graph_error)
            self._graph = None
            torch.cuda.synchronize()
            for g in self.optimizer.param_groups:
                g["capturable"] = False
            # warmup may have created capturable (device) step counters;
            # the eager non-capturable Adam wants them on host
            for st in self.optimizer.state.values():
                if "step" in st and torch.is_tensor(st["step"]) and \
                        st["step"].is_cuda:
                    st["step"] = st["step"].cpu()
            return False

    def train_step_graphed(self, items) -> dict:
        """Copy the batch into the static buffers and replay the graph.
        The returned loss dict's tensors are the static graph outputs
        (valid after the replay completes)."""
        self.set_data(items, static=True)
        self._graph.replay()
        return self._graph_loss

    def enable_phase_timers(self) -> None:
        """Sync-bracketed per-phase wall times; read with pop_phase_times()."""
        self.phase_timers = {}

    def pop_phase_times(self) -> dict:
        """Mean seconds per phase since the last call."""
        t = getattr(self, "phase_timers", None)
        if not t or "t0" not in t:
            return {}
        names = ["set_data", "forward", "backward", "optimizer"]
        out = {}
        prev = t["t0"]
        for n in names:
            cur = t.get(n)
            if cur is None or len(cur) != len(prev):
                break
            out[n] = sum(b - a for a, b in zip(prev, cur)) / len(cur)
            prev = cur
        self.phase_timers = {}
        return out

    def train_epoch(self, train_loader, val_loader, epoch: int) -> None:
        cfg = self.config
        if hasattr(train_loader, "sampler") and \
                hasattr(train_loader.sampler, "set_epoch"):
            train_loader.sampler.set_epoch(epoch)
        self.backbone.train()
        self.decoder.train()
        self.current_epoch = epoch
        for m in self.train_losses.values():
            m.reset()

        for step, items in enumerate(train_loader, start=1):
            self.global_step += 1
            loss_dict = self.train_step(items)

            # gate on global_step (the same counter train_step uses to
            # decide whether the monitor losses were computed this step)
            if self.global_step % int(cfg.get("training.log_interval", 10)) \
                    == 0 and self.state.is_rank0:
                self._log_training(epoch, step, len(train_loader), loss_dict)

            ckpt_every = int(cfg.get("training.checkpoint_interval", 5000))
            if step % ckpt_every == 0 and self.state.is_rank0 and \
                    self.state.local_workspace:
                path = os.path.join(self.state.local_workspace, "checkpoint_latest.pth")
                # meta["epoch"] records the last COMPLETED epoch: a
                # mid-epoch save records epoch-1, so resume re-runs the
                # interrupted epoch instead of silently skipping its
                # remainder (round-1 ADVICE.md item 4). global_step keeps
                # its mid-epoch value (monotone; re-run steps re-count).
                save_checkpoint(path, self.backbone, self.decoder, self.optimizer,
                                meta={"epoch": self.current_epoch - 1,
                                      "global_step": self.global_step,
                                      "mid_epoch": True})
                if self.logger:
                    self.logger.info(f"Latest checkpoint saved at {path}")

            eval_every = int(cfg["training.eval_interval"])
            if self.state.is_rank0 and val_loader is not None and \
                    (self.global_step == 2000 or self.global_step % eval_every == 0):
                self.run_eval(val_loader)
                if self.state.local_workspace:
                    path = os.path.join(self.state.local_workspace,
                                        "checkpoint_%012d.pth" % self.global_step)
                    save_checkpoint(path, self.backbone, self.decoder)

    def train(self, train_loader, val_loader=None) -> None:
        start_epoch = 1
        if self._restored_meta and bool(self.config.get("training.fine_tune", False)) is False:
            # resume where the checkpoint left off (improvement over the
            # reference, which always restarted at epoch 1)
            start_epoch = int(self._restored_meta.get("epoch", 0)) + 1
            self.global_step = int(self._restored_meta.get("global_step", 0))
            for _ in range(1, start_epoch):
                self.lr_scheduler.step()
        for epoch in range(start_epoch, self.config["training.epochs"] + 1):
            self.current_epoch = epoch
            self.train_epoch(train_loader, val_loader, epoch)
            self.lr_scheduler.step()
            if self.state.is_rank0 and self.state.local_workspace:
                # end-of-epoch latest save: meta marks the epoch COMPLETE
                path = os.path.join(self.state.local_workspace,
                                    "checkpoint_latest.pth")
                save_checkpoint(path, self.backbone, self.decoder,
                                self.optimizer,
                                meta={"epoch": epoch,
                                      "global_step": self.global_step})
            if self.state.is_rank0 and self.logger:
                self.logger.info("Epoch finished, average losses: ")
                for v in self.train_losses.values():
                    self.logger.info("    {}".format(v))

    # ------------------------------------------------------------------
    def run_eval(self, val_loader) -> None:
        if self.logger:
            self.logger.info("Start running evaluation on validation set:")
        self.backbone.eval()
        self.decoder.eval()
        for m in self.val_losses.values():
            m.reset()
        with torch.no_grad():
            for step, items in enumerate(val_loader):
                self.set_data(items)
                loss_dict, vis_dict = self.loss_fcn(is_val=True)
                B = self.src_imgs.shape[0]
                for key, meter in self.val_losses.items():
                    meter.update(float(loss_dict[key]), n=B)
                if self.state.tb_writer is not None:
                    self._log_val_images(step, vis_dict)
            if self.logger:
                self.logger.info("Evaluation finished, average losses: ")
                for v in self.val_losses.values():
                    self.logger.info("    {}".format(v))
            if self.state.tb_writer is not None:
                for key, meter in self.val_losses.items():
                    self.state.tb_writer.add_scalar(key + "/val", meter.avg,
                                                    self.global_step)
        self.backbone.train()
        self.decoder.train()

    # ------------------------------------------------------------------
    def _log_training(self, epoch, step, n_steps, loss_dict) -> None:
        # logging steps already host-sync on float(loss): piggyback the
        # NaN-guard skip counter read here (never in the hot loop)
        if self._nan_skip_count is not None:
            skips = int(self._nan_skip_count.item())
            if skips and self.logger:
                self.logger.warning(
                    "NaN guard gated %d non-finite step(s) so far "
                    "(gradients zeroed device-side)", skips)
        for key, meter in self.train_losses.items():
            v = float(loss_dict[key])
            meter.update(v)
            if self.state.tb_writer is not None:
                self.state.tb_writer.add_scalar(key + "/train", v, self.global_step)
        if self.logger:
            self.logger.info(
                "epoch [%.3d] step [%d/%d] global_step = %d total_loss = %.4f "
                "encoder_lr = %.7f\n"
                "        src: rgb = %.4f  ssim = %.4f  smooth = %.4f  disp_pt3d = %.4f\n"
                "        tgt: rgb = %.4f  ssim = %.4f  smooth = %.4f  disp_pt3d = %.4f" %
                (epoch, step, n_steps, self.global_step,
                 float(loss_dict["loss"]), self.optimizer.param_groups[0]["lr"],
                 float(loss_dict["loss_rgb_src"]), float(loss_dict["loss_ssim_src"]),
                 float(loss_dict["loss_smooth_src"]), float(loss_dict["loss_disp_pt3dsrc"]),
                 float(loss_dict["loss_rgb_tgt"]), float(loss_dict["loss_ssim_tgt"]),
                 float(loss_dict["loss_smooth_tgt"]), float(loss_dict["loss_disp_pt3dtgt"])))

    def _log_val_images(self, step, vis_dict) -> None:
        from mine_amd.utils import disparity_normalization_vis
        tb = self.state.tb_writer
        gs = self.global_step
        tb.add_images("02_syn_src_images/step_%d" % gs,
                      vis_dict["src_imgs_syn"].clamp(0, 1), step)
        tb.add_images("03_syn_src_disparity_map/step_%d" % gs,
                      disparity_normalization_vis(vis_dict["src_disparity_syn"]), step)
        tb.add_images("04_syn_tgt_images/step_%d" % gs,
                      vis_dict["tgt_imgs_syn"].clamp(0, 1), step)
        tb.add_images("05_syn_tgt_disparity_map/step_%d" % gs,
                      disparity_normalization_vis(vis_dict["tgt_disparity_syn"]), step)
