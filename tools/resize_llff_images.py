#!/usr/bin/env python3
"""Offline LLFF image pre-downsampling (dataset-prep tool).

Equivalent of the reference's
input_pipelines/llff/misc/resize_nerf_llff_images.py: resize every
scene's ``images/`` by `ratio` into ``images_<ratio>/`` (PIL instead of
cv2 — cv2 is not in this environment).

    python tools/resize_llff_images.py --root /data/nerf_llff_data \
        [--ratio 7.875]
"""
from __future__ import annotations

import argparse
import os


def resize_scene(scene_dir: str, ratio: float) -> int:
    from PIL import Image

    src_dir = os.path.join(scene_dir, "images")
    if not os.path.isdir(src_dir):
        return 0
    dst_dir = os.path.join(scene_dir, "images_%g" % ratio)
    os.makedirs(dst_dir, exist_ok=True)
    n = 0
    for name in sorted(os.listdir(src_dir)):
        if os.path.splitext(name)[1].lower() not in (".png", ".jpg", ".jpeg"):
            continue
        with Image.open(os.path.join(src_dir, name)) as im:
            w = int(round(im.width / ratio))
            h = int(round(im.height / ratio))
            im.convert("RGB").resize((w, h), Image.LANCZOS).save(
                os.path.join(dst_dir, name))
        n += 1
    return n


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--root", required=True, help="dir of LLFF scene dirs")
    p.add_argument("--ratio", type=float, default=7.875)
    args = p.parse_args()
    total = 0
    for scene in sorted(os.listdir(args.root)):
        scene_dir = os.path.join(args.root, scene)
        if os.path.isdir(scene_dir):
            n = resize_scene(scene_dir, args.ratio)
            if n:
                print(f"{scene}: {n} images -> images_{args.ratio:g}/")
            total += n
    print(f"done: {total} images")
    return 0


if __name__ == "__main__":
    import sys
    sys.exit(main())
