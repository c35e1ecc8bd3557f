"""Closed-form batched matrix inverses.

The reference fought a CUDA batched-`torch.inverse` nan bug with a
retry-around-`cuda.synchronize` loop (ref utils.py:96-117, referencing
pytorch#47272) plus explicit syncs before every inverse
(ref synthesis_task.py:242-244, homography_sampler.py:111-114).

This framework kills that class of bug: every matrix on the hot path is
3x3 (intrinsics / homographies) or a rigid 4x4 (poses), both of which
have exact closed forms that are branch-free elementwise torch ops (and
trivially fusable inside HIP kernels — the fused warp kernel computes
its own 3x3 inverses in registers).
"""
from __future__ import annotations

import torch


def inverse_3x3(m: torch.Tensor) -> torch.Tensor:
    """Closed-form (adjugate/determinant) inverse of batched 3x3 matrices.

    m: (..., 3, 3). Returns (..., 3, 3). fp32/fp64.
    """
    a = m[..., 0, 0]; b = m[..., 0, 1]; c = m[..., 0, 2]
    d = m[..., 1, 0]; e = m[..., 1, 1]; f = m[..., 1, 2]
    g = m[..., 2, 0]; h = m[..., 2, 1]; i = m[..., 2, 2]

    A = e * i - f * h
    B = -(d * i - f * g)
    C = d * h - e * g
    det = a * A + b * B + c * C

    inv_det = torch.reciprocal(det)
    out = torch.stack(
        (
            A, -(b * i - c * h), (b * f - c * e),
            B, (a * i - c * g), -(a * f - c * d),
            C, -(a * h - b * g), (a * e - b * d),
        ),
        dim=-1,
    ).reshape(m.shape)
    return out * inv_det[..., None, None]


def inverse_rigid_4x4(g: torch.Tensor) -> torch.Tensor:
    """Inverse of batched rigid transforms [[R, t], [0, 1]]: [[R^T, -R^T t], [0, 1]].

    g: (..., 4, 4) with orthonormal rotation block.
    """
    R = g[..., :3, :3]
    t = g[..., :3, 3:4]
    Rt = R.transpose(-1, -2)
    top = torch.cat((Rt, -Rt @ t), dim=-1)  # (...,3,4)
    bottom = torch.zeros_like(top[..., :1, :])
    bottom[..., 0, 3] = 1.0
    return torch.cat((top, bottom), dim=-2)


def inverse_4x4(m: torch.Tensor) -> torch.Tensor:
    """General batched 4x4 inverse via cofactor expansion (exact, no LU).

    Used for pose matrices when they are not guaranteed rigid
    (ref synthesis_task.py:208 inverts G_src_tgt, which IS rigid in all
    shipped datasets — `inverse_rigid_4x4` is the hot-path choice; this
    one is the drop-in for arbitrary inputs).
    """
    # unroll: m[..., i, j]
    m00 = m[..., 0, 0]; m01 = m[..., 0, 1]; m02 = m[..., 0, 2]; m03 = m[..., 0, 3]
    m10 = m[..., 1, 0]; m11 = m[..., 1, 1]; m12 = m[..., 1, 2]; m13 = m[..., 1, 3]
    m20 = m[..., 2, 0]; m21 = m[..., 2, 1]; m22 = m[..., 2, 2]; m23 = m[..., 2, 3]
    m30 = m[..., 3, 0]; m31 = m[..., 3, 1]; m32 = m[..., 3, 2]; m33 = m[..., 3, 3]

    s0 = m00 * m11 - m10 * m01
    s1 = m00 * m12 - m10 * m02
    s2 = m00 * m13 - m10 * m03
    s3 = m01 * m12 - m11 * m02
    s4 = m01 * m13 - m11 * m03
    s5 = m02 * m13 - m12 * m03

    c5 = m22 * m33 - m32 * m23
    c4 = m21 * m33 - m31 * m23
    c3 = m21 * m32 - m31 * m22
    c2 = m20 * m33 - m30 * m23
    c1 = m20 * m32 - m30 * m22
    c0 = m20 * m31 - m30 * m21

    det = s0 * c5 - s1 * c4 + s2 * c3 + s3 * c2 - s4 * c1 + s5 * c0
    inv_det = torch.reciprocal(det)

    r = torch.stack(
        (
            m11 * c5 - m12 * c4 + m13 * c3,
            -m01 * c5 + m02 * c4 - m03 * c3,
            m31 * s5 - m32 * s4 + m33 * s3,
            -m21 * s5 + m22 * s4 - m23 * s3,

            -m10 * c5 + m12 * c2 - m13 * c1,
            m00 * c5 - m02 * c2 + m03 * c1,
            -m30 * s5 + m32 * s2 - m33 * s1,
            m20 * s5 - m22 * s2 + m23 * s1,

            m10 * c4 - m11 * c2 + m13 * c0,
            -m00 * c4 + m01 * c2 - m03 * c0,
            m30 * s4 - m31 * s2 + m33 * s0,
            -m20 * s4 + m21 * s2 - m23 * s0,

            -m10 * c3 + m11 * c1 - m12 * c0,
            m00 * c3 - m01 * c1 + m02 * c0,
            -m30 * s3 + m31 * s1 - m32 * s0,
            m20 * s3 - m21 * s1 + m22 * s0,
        ),
        dim=-1,
    ).reshape(m.shape)
    return r * inv_det[..., None, None]
