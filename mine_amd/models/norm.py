"""BatchNorm that always computes in fp32.

Under bf16 autocast the conv stack feeds bf16 activations; BN statistics
in bf16 are both less accurate and (on ROCm 7.2 + NHWC) route to a MIOpen
batch-norm path that is unstable in low precision. This wrapper runs the
normalization in fp32 and casts back, which is also the numerically
standard choice for mixed-precision training.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class Fp32BatchNorm2d(nn.BatchNorm2d):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dtype == torch.float32:
            return super().forward(x)
        return super().forward(x.float()).to(x.dtype)
