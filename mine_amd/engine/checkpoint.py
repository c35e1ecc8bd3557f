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


def _ref_convs_key(*key_tuple) -> str:
    """The reference decoder's ModuleDict key for a ("upconv", i, j) /
    ("dispconv", s) tuple: `'-'.join(str(key_tuple))` joins the CHARACTERS
    of the tuple's repr (ref network/monodepth2/depth_decoder.py:36-38) —
    reproduced verbatim so released-checkpoint keys resolve."""
    return "-".join(str(tuple(key_tuple)))


def convert_reference_backbone(sd: dict) -> dict:
    """Map a reference ResnetEncoder state dict (torchvision resnet50 under
    an `encoder.` prefix, ref network/monodepth2/resnet_encoder.py:63-86)
    onto mine_amd.models.ResNetEncoder names. The unused classification
    head (`encoder.fc.*`) is dropped — our encoder has none (static
    graph; see models/resnet.py)."""
    out = {}
    for k, v in sd.items():
        if not k.startswith("encoder."):
            out[k] = v
            continue
        k2 = k[len("encoder."):]
        if k2.startswith("fc."):
            continue
        out[k2] = v
    return out


def convert_reference_decoder(sd: dict) -> dict:
    """Map a reference DepthDecoder state dict onto mine_amd MPIDecoder
    names. Weight SHAPES are identical by construction: SplitConvBlock
    keeps the reference's concatenated (C_dec + C_enc + E) input-channel
    layout and order ([x_dec, skip_feat, PE], ref depth_decoder.py:103-137)
    even though its forward factors the concat away.

    Reference keys (ref depth_decoder.py:69-90, layers.py:106-138):
      convs.{key("upconv",i,0)}.conv.conv.{weight,bias} -> upconvs0.{4-i}.conv.*
      convs.{key("upconv",i,0)}.bn.*                    -> upconvs0.{4-i}.bn.*
      convs.{key("upconv",i,1)}.{conv.conv,bn}.*        -> upconvs1.{4-i}.*
      convs.{key("dispconv",s)}.conv.{weight,bias}      -> dispconvs.{s}.conv.*
      conv_down1/2, conv_up1/2 (Sequential conv+BN)      -> unchanged
    """
    mapping = {}
    for i in range(4, -1, -1):
        idx = 4 - i
        for j, ours in ((0, f"upconvs0.{idx}"), (1, f"upconvs1.{idx}")):
            ref = "convs." + _ref_convs_key("upconv", i, j)
            mapping[ref + ".conv.conv"] = ours + ".conv"
            mapping[ref + ".bn"] = ours + ".bn"
    for s in range(4):
        mapping["convs." + _ref_convs_key("dispconv", s) + ".conv"] = \
            f"dispconvs.{s}.conv"
    out = {}
    for k, v in sd.items():
        hit = None
        for ref_prefix, our_prefix in mapping.items():
            if k.startswith(ref_prefix + "."):
                hit = our_prefix + k[len(ref_prefix):]
                break
        out[hit if hit else k] = v
    return out


def maybe_convert_reference_state(state: dict, logger=None) -> dict:
    """Detect a released-MINE-layout checkpoint (ref README.md:43-50,
    utils.py:40-67) and convert its key names in place."""
    bb = state.get("backbone")
    if bb and any(k.startswith(("encoder.", "module.encoder."))
                  for k in bb.keys()):
        if logger:
            logger.info("[MODEL_RESTORE] reference-layout checkpoint "
                        "detected; converting key names")
        state = dict(state)
        state["backbone"] = convert_reference_backbone(
            _strip_module_prefix(bb))
        if "decoder" in state:
            state["decoder"] = convert_reference_decoder(
                _strip_module_prefix(state["decoder"]))
        # the reference optimizer state indexes the reference's parameter
        # order (incl. the dropped `fc`) — not importable
        if state.pop("optimizer", None) is not None and logger:
            logger.info("[MODEL_RESTORE] reference optimizer state "
                        "dropped (parameter sets differ); Adam starts cold")
    return state


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
    state = maybe_convert_reference_state(state, logger=logger)

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
