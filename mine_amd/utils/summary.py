"""Scalar/image run logging.

The reference logs scalars and image grids to TensorBoard
(ref train.py:135, synthesis_task.py:502-548). TensorBoard is not
available in this environment, so this module provides a
SummaryWriter-compatible JSONL writer: scalars append to
``scalars.jsonl`` (one {"tag", "value", "step"} object per line — easy
to plot or ingest anywhere) and image batches are written as PNG grids
under ``images/``. train.py prefers torch.utils.tensorboard when
importable and falls back to this.
"""
from __future__ import annotations

import json
import math
import os
import time
from typing import Optional

import torch


def make_image_grid(images: torch.Tensor, ncol: Optional[int] = None
                    ) -> torch.Tensor:
    """Bx3xHxW [0,1] -> 3xH'xW' tiled grid."""
    B, C, H, W = images.shape
    if ncol is None:
        ncol = max(1, int(math.ceil(math.sqrt(B))))
    nrow = (B + ncol - 1) // ncol
    grid = images.new_zeros(C, nrow * H, ncol * W)
    for i in range(B):
        r, c = divmod(i, ncol)
        grid[:, r * H:(r + 1) * H, c * W:(c + 1) * W] = images[i]
    return grid


class JsonlSummaryWriter:
    """Minimal TB-API-compatible writer: JSONL scalars + PNG image grids."""

    def __init__(self, log_dir: str):
        self.log_dir = log_dir
        os.makedirs(log_dir, exist_ok=True)
        self._scalar_path = os.path.join(log_dir, "scalars.jsonl")
        self._img_dir = os.path.join(log_dir, "images")
        self._f = open(self._scalar_path, "a", buffering=1)

    def add_scalar(self, tag: str, value, global_step: int = 0) -> None:
        self._f.write(json.dumps({
            "tag": tag, "value": float(value), "step": int(global_step),
            "ts": time.time()}) + "\n")

    def add_images(self, tag: str, images: torch.Tensor,
                   global_step: int = 0) -> None:
        from PIL import Image as PILImage
        os.makedirs(self._img_dir, exist_ok=True)
        if images.dim() == 3:
            images = images.unsqueeze(0)
        if images.shape[1] == 1:
            images = images.expand(-1, 3, -1, -1)
        grid = make_image_grid(images.detach().float().cpu().clamp(0, 1))
        arr = (grid.permute(1, 2, 0).numpy() * 255).astype("uint8")
        safe_tag = tag.replace("/", "_")
        PILImage.fromarray(arr).save(
            os.path.join(self._img_dir, f"{safe_tag}_{global_step:09d}.png"))

    def flush(self) -> None:
        self._f.flush()

    def close(self) -> None:
        self._f.close()


def create_summary_writer(log_dir: str):
    """torch.utils.tensorboard when importable, else the JSONL writer."""
    try:
        from torch.utils.tensorboard import SummaryWriter
        return SummaryWriter(log_dir=log_dir)
    except Exception:
        return JsonlSummaryWriter(log_dir)
