"""Checkpoint save/restore with the reference's on-disk contract.

Contract (ref synthesis_task.py:629-652, utils.py:40-67):
  * checkpoint.pth is a flat dict {"backbone": sd, "decoder": sd,
    "optimizer": sd} ("optimizer" present in checkpoint_latest.pth,
    absent in step-stamped eval checkpoints);
  * model keys may carry a "module." prefix (saved from DDP-wrapped
    models) — restore strips it;
  * model restore is strict=False with logged key diffs, optimizer
    restore is strict;
  * a params.yaml sits next to checkpoint.pth (the inference entry point
    reads it, ref visualizations/image_to_video.py:272-278).
"""
from __future__ import annotations

import os
from typing import Optional

import torch


def _strip_module_prefix(sd: dict) -> dict:
    return {(k[len("module."):] if k.startswith("module.") else k): v
            for k, v in sd.items()}


def save_checkpoint(path: str, backbone, decoder, optimizer=None,
                    meta: Optional[dict] = None) -> None:
    state = {"backbone": backbone.state_dict(), "decoder": decoder.state_dict()}
    if optimizer is not None:
        state["optimizer"] = optimizer.state_dict()
    if meta:
        # extra key; the reference's loader ignores unknown keys, so the
        # contract stays compatible. Fixes the reference's resume gap
        # (epoch/step not persisted, ref synthesis_task.py:661-663).
        state["meta"] = dict(meta)
    torch.save(state, path)


def restore_model(model_path: Optional[str], backbone, decoder, optimizer=None,
                  logger=None) -> dict:
    """Restore; returns the checkpoint's meta dict ({} if absent)."""
    if not model_path:
        if logger:
            logger.info("Not using pre-trained model...")
        return {}
    assert os.path.exists(model_path), f"Model {model_path} does not exist!"
    state = torch.load(model_path, map_location="cpu", weights_only=False)

    for key, model in (("backbone", backbone), ("decoder", decoder),
                       ("optimizer", optimizer)):
        if model is None or key not in state:
            continue
        sd = _strip_module_prefix(state[key])
        if key != "optimizer":
            model_keys = set(model.state_dict().keys())
            ckpt_keys = set(sd.keys())
            if logger:
                logger.info("[MODEL_RESTORE] missing keys in %s checkpoint: %s"
                            % (key, sorted(model_keys - ckpt_keys)))
                logger.info("[MODEL_RESTORE] missing keys in %s model: %s"
                            % (key, sorted(ckpt_keys - model_keys)))
            model.load_state_dict(sd, strict=False)
        else:
            model.load_state_dict(sd)
    return state.get("meta", {})
