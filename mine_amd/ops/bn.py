"""Fused BatchNorm + activation on the HIP kernels.

Training-mode BN with fp32 statistics over bf16 (or fp32) channels_last
activations, with the following activation fused into the normalize pass
(and its derivative into the backward): none / relu / lrelu(0.1) / elu /
add_relu (the ResNet residual join y = relu(bn(x) + res)).

Replaces the eager chain the profile showed as ~20% of the step
(cast->MIOpen BN->cast->activation, fwd+bwd; see
profiles/r01_flagship_kernel_stats.md). `FusedBNAct` subclasses
nn.BatchNorm2d so parameter/buffer names (weight, bias, running_mean,
running_var, num_batches_tracked) — and therefore the checkpoint
contract — are unchanged (ref CS5).

Fallback paths (CPU, non-channels_last, exotic dtype, eval-with-grad):
fp32 functional BN + eager activation, numerically equivalent.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from mine_amd.ops.backend import get_extension

_ACT = {"none": 0, "relu": 1, "lrelu": 2, "elu": 3, "add_relu": 4}


class _BNActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x_flat, res_flat, mean, invstd, gamma, beta, M, C, act,
                sync):
        ext = get_extension(required=True)
        y = ext.bn_act_fwd(x_flat, res_flat, mean, invstd,
                           gamma, beta, M, C, act)
        ctx.save_for_backward(x_flat, res_flat, mean, invstd, gamma, beta)
        ctx.geom = (M, C, act, sync)
        return y

    @staticmethod
    def backward(ctx, gy):
        ext = get_extension(required=True)
        x_flat, res_flat, mean, invstd, gamma, beta = ctx.saved_tensors
        M, C, act, sync = ctx.geom
        gy = gy.contiguous()
        red = ext.bn_act_bwd_reduce(x_flat, res_flat, gy, mean, invstd,
                                    gamma, beta, M, C, act)
        # SyncBN backward: dx needs the GLOBAL (dbeta, dgamma) sums and the
        # global batch count; the returned parameter grads stay LOCAL (DDP
        # averages them afterwards, matching torch SyncBatchNorm + DDP).
        dbeta, dgamma = red[:C].clone(), red[C:].clone()
        M_norm = M
        if sync:
            torch.distributed.all_reduce(red)
            M_norm = M * torch.distributed.get_world_size()
        dx, dres = ext.bn_act_bwd_dx(x_flat, res_flat, gy, mean, invstd,
                                     gamma, beta, red, M, C, act, M_norm)
        return (dx, dres if act == _ACT["add_relu"] else None, None, None,
                dgamma, dbeta, None, None, None, None)


def _act_eager(act: str, z: torch.Tensor) -> torch.Tensor:
    if act == "relu":
        return F.relu(z)
    if act == "lrelu":
        return F.leaky_relu(z, 0.1)
    if act == "elu":
        return F.elu(z)
    return z


class FusedBNAct(nn.BatchNorm2d):
    """BatchNorm2d with a fused activation epilogue.

    act: "none" | "relu" | "lrelu" | "elu" | "add_relu".
    For "add_relu", forward takes the residual as the second argument.
    """

    def __init__(self, num_features: int, act: str = "none", **kw):
        super().__init__(num_features, **kw)
        assert act in _ACT, act
        self.act = act
        # Cross-rank statistics sync (the reference's SyncBatchNorm role,
        # ref synthesis_task.py:106-112): when True and a process group is
        # live, the per-channel (sum, sumsq) vectors are all-reduced
        # before normalization — one 2C-float RCCL collective per BN.
        # Enabled by SynthesisTask when training.sync_batchnorm is set;
        # torch's convert_sync_batchnorm would silently DROP the fused
        # activation, so it must not be used on this module.
        self.sync = False

    def _fallback(self, x, res):
        xf = x.float()
        z = F.batch_norm(
            xf, self.running_mean, self.running_var, self.weight, self.bias,
            self.training, self.momentum, self.eps)
        if self.act == "add_relu":
            z = F.relu(z + res.float())
        else:
            z = _act_eager(self.act, z)
        return z.to(x.dtype)

    def forward(self, x: torch.Tensor, res: torch.Tensor = None):
        assert (res is not None) == (self.act == "add_relu")
        use_kernel = (
            x.is_cuda and x.dim() == 4
            and x.dtype in (torch.float32, torch.bfloat16)
            and x.is_contiguous(memory_format=torch.channels_last)
            and (res is None or (res.dtype == x.dtype and res.is_contiguous(
                memory_format=torch.channels_last)))
            and (self.training or not torch.is_grad_enabled()))
        if not use_kernel:
            return self._fallback(x, res)

        ext = get_extension(required=True)
        B, C, H, W = x.shape
        M = B * H * W
        x_flat = x.permute(0, 2, 3, 1).reshape(-1)
        res_flat = res.permute(0, 2, 3, 1).reshape(-1) if res is not None \
            else torch.empty(0, device=x.device, dtype=x.dtype)

        do_sync = self.training and self.sync and \
            torch.distributed.is_initialized() and \
            torch.distributed.get_world_size() > 1
        if self.training:
            if self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
            mom = self.momentum if self.momentum is not None else \
                1.0 / float(self.num_batches_tracked)
            track = self.track_running_stats and self.running_mean is not None
            empty = torch.empty(0, device=x.device, dtype=torch.float32)
            if do_sync:
                sums = ext.bn_sums(x_flat.detach(), M, C)
                torch.distributed.all_reduce(sums)
                Mg = M * torch.distributed.get_world_size()
                mean = sums[:C] / Mg
                var = (sums[C:] / Mg - mean * mean).clamp_min_(0.0)
                invstd = torch.rsqrt(var + self.eps)
                if track:
                    with torch.no_grad():
                        self.running_mean += mom * (mean - self.running_mean)
                        unbiased = var * Mg / max(Mg - 1, 1)
                        self.running_var += mom * (unbiased - self.running_var)
                mean = mean.contiguous()
                invstd = invstd.contiguous()
            else:
                mean, invstd = ext.bn_stats(
                    x_flat.detach(), M, C,
                    self.running_mean if track else empty,
                    self.running_var if track else empty,
                    self.eps, mom)
        else:
            mean = self.running_mean.float()
            invstd = torch.rsqrt(self.running_var.float() + self.eps)

        y = _BNActFn.apply(x_flat, res_flat, mean, invstd,
                           self.weight.float(), self.bias.float(),
                           M, C, _ACT[self.act], do_sync)
        return y.view(B, H, W, C).permute(0, 3, 1, 2)

    def extra_repr(self) -> str:  # pragma: no cover
        return super().extra_repr() + f", act={self.act}"
