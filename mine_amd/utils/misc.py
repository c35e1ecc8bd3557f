"""Misc utilities: meters, visualization helpers, logging."""
from __future__ import annotations

import logging
import sys
from typing import Optional

import torch


class AverageMeter:
    """Running average tracker (ref utils.py:120-141)."""

    def __init__(self, name: str, fmt: str = ":f"):
        self.name = name
        self.fmt = fmt
        self.reset()

    def reset(self) -> None:
        self.val = 0.0
        self.avg = 0.0
        self.sum = 0.0
        self.count = 0

    def update(self, val: float, n: int = 1) -> None:
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / self.count

    def __str__(self) -> str:
        fmtstr = "{name} {val" + self.fmt + "} ({avg" + self.fmt + "})"
        return fmtstr.format(**self.__dict__)


def disparity_normalization_vis(disparity: torch.Tensor) -> torch.Tensor:
    """Min-max normalize a Bx1xHxW disparity map to [0,1] (ref utils.py:6-17)."""
    assert disparity.dim() == 4 and disparity.size(1) == 1
    disp_min = torch.amin(disparity, (1, 2, 3), keepdim=True)
    disp_max = torch.amax(disparity, (1, 2, 3), keepdim=True)
    scaled = (disparity - disp_min) / (disp_max - disp_min)
    return torch.clip(scaled, 0.0, 1.0)


def setup_logger(name: str = "mine_amd", log_file: Optional[str] = None,
                 level: int = logging.INFO) -> logging.Logger:
    """Rank-0 logger to stdout (+ optional file), matching the reference's
    format (ref train.py:116-131)."""
    logger = logging.getLogger(name)
    formatter = logging.Formatter("[%(asctime)s %(filename)s] %(message)s")
    handlers = []
    stream_handler = logging.StreamHandler(sys.stdout)
    stream_handler.setFormatter(formatter)
    handlers.append(stream_handler)
    if log_file:
        file_handler = logging.FileHandler(log_file)
        file_handler.setFormatter(formatter)
        handlers.append(file_handler)
    logger.handlers = handlers
    logger.setLevel(level)
    logger.propagate = False
    return logger


def linspace_batch(start: torch.Tensor, end: torch.Tensor,
                   steps: int) -> torch.Tensor:
    """Batched linspace: start/end (B,) -> (B, steps)
    (ref utils.py:70-93, which looped; this is one broadcast)."""
    assert start.shape == end.shape and start.dim() == 1
    t = torch.linspace(0.0, 1.0, steps, dtype=start.dtype,
                       device=start.device)
    return start.unsqueeze(1) + (end - start).unsqueeze(1) * t.unsqueeze(0)
