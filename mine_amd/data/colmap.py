"""COLMAP sparse-model I/O (cameras / images / points3D, .bin and .txt).

Fresh implementation of the public COLMAP reconstruction file format
(https://colmap.github.io/format.html), providing the reader surface the
reference exposes in input_pipelines/colmap_utils.py (read_model
ref colmap_utils.py:420-439, qvec2rotmat ref colmap_utils.py:454-464)
plus the matching writers.

Binary layout (little-endian):
  cameras.bin : u64 n; per camera: i32 id, i32 model_id, u64 w, u64 h,
                f64 params[model.num_params]
  images.bin  : u64 n; per image: i32 id, f64 qvec[4], f64 tvec[3],
                i32 camera_id, name bytes + NUL, u64 n2d,
                (f64 x, f64 y, i64 point3D_id) * n2d
  points3D.bin: u64 n; per point: i64 id, f64 xyz[3], u8 rgb[3],
                f64 error, u64 track_len, (i32 image_id, i32 p2d_idx)*len
"""
from __future__ import annotations

import os
import struct
from dataclasses import dataclass
from typing import Dict, Tuple

import numpy as np


# ---------------------------------------------------------------------------
# camera models
# ---------------------------------------------------------------------------

# model_id -> (name, num_params)
CAMERA_MODELS: Dict[int, Tuple[str, int]] = {
    0: ("SIMPLE_PINHOLE", 3),
    1: ("PINHOLE", 4),
    2: ("SIMPLE_RADIAL", 4),
    3: ("RADIAL", 5),
    4: ("OPENCV", 8),
    5: ("OPENCV_FISHEYE", 8),
    6: ("FULL_OPENCV", 12),
    7: ("FOV", 5),
    8: ("SIMPLE_RADIAL_FISHEYE", 4),
    9: ("RADIAL_FISHEYE", 5),
    10: ("THIN_PRISM_FISHEYE", 12),
}
CAMERA_MODEL_IDS = {name: mid for mid, (name, _) in CAMERA_MODELS.items()}


@dataclass
class Camera:
    id: int
    model: str
    width: int
    height: int
    params: np.ndarray

    def intrinsic_matrix(self) -> np.ndarray:
        """3x3 K from the pinhole part of the model (distortion ignored)."""
        p = self.params
        if self.model in ("SIMPLE_PINHOLE", "SIMPLE_RADIAL",
                          "SIMPLE_RADIAL_FISHEYE", "RADIAL", "RADIAL_FISHEYE",
                          "FOV"):
            fx = fy = p[0]
            cx, cy = p[1], p[2]
        else:  # PINHOLE / OPENCV families: fx fy cx cy ...
            fx, fy, cx, cy = p[0], p[1], p[2], p[3]
        return np.array([[fx, 0.0, cx], [0.0, fy, cy], [0.0, 0.0, 1.0]],
                        dtype=np.float64)


@dataclass
class Image:
    id: int
    qvec: np.ndarray          # (4,) w x y z
    tvec: np.ndarray          # (3,)
    camera_id: int
    name: str
    xys: np.ndarray           # (N, 2) keypoints
    point3D_ids: np.ndarray   # (N,) int64, -1 = no 3D point

    def qvec2rotmat(self) -> np.ndarray:
        return qvec2rotmat(self.qvec)

    def world_to_cam(self) -> np.ndarray:
        """4x4 G_cam_world."""
        G = np.eye(4)
        G[:3, :3] = qvec2rotmat(self.qvec)
        G[:3, 3] = self.tvec
        return G


@dataclass
class Point3D:
    id: int
    xyz: np.ndarray           # (3,)
    rgb: np.ndarray           # (3,) uint8
    error: float
    image_ids: np.ndarray     # (T,)
    point2D_idxs: np.ndarray  # (T,)


# ---------------------------------------------------------------------------
# quaternion helpers
# ---------------------------------------------------------------------------


def qvec2rotmat(q) -> np.ndarray:
    """Unit quaternion (w,x,y,z) -> 3x3 rotation matrix."""
    w, x, y, z = np.asarray(q, dtype=np.float64)
    return np.array([
        [1 - 2 * (y * y + z * z), 2 * (x * y - w * z), 2 * (x * z + w * y)],
        [2 * (x * y + w * z), 1 - 2 * (x * x + z * z), 2 * (y * z - w * x)],
        [2 * (x * z - w * y), 2 * (y * z + w * x), 1 - 2 * (x * x + y * y)],
    ])


def rotmat2qvec(R) -> np.ndarray:
    """3x3 rotation matrix -> unit quaternion (w,x,y,z), via the
    symmetric-eigenvector method (numerically stable for all traces)."""
    R = np.asarray(R, dtype=np.float64)
    K = np.array([
        [R[0, 0] - R[1, 1] - R[2, 2], R[1, 0] + R[0, 1],
         R[2, 0] + R[0, 2], R[2, 1] - R[1, 2]],
        [R[1, 0] + R[0, 1], R[1, 1] - R[0, 0] - R[2, 2],
         R[2, 1] + R[1, 2], R[0, 2] - R[2, 0]],
        [R[2, 0] + R[0, 2], R[2, 1] + R[1, 2],
         R[2, 2] - R[0, 0] - R[1, 1], R[1, 0] - R[0, 1]],
        [R[2, 1] - R[1, 2], R[0, 2] - R[2, 0],
         R[1, 0] - R[0, 1], R[0, 0] + R[1, 1] + R[2, 2]],
    ]) / 3.0
    vals, vecs = np.linalg.eigh(K)
    q = vecs[[3, 0, 1, 2], np.argmax(vals)]
    return -q if q[0] < 0 else q


# ---------------------------------------------------------------------------
# binary readers
# ---------------------------------------------------------------------------


def _read(f, fmt: str):
    return struct.unpack("<" + fmt, f.read(struct.calcsize("<" + fmt)))


def read_cameras_binary(path: str) -> Dict[int, Camera]:
    cams: Dict[int, Camera] = {}
    with open(path, "rb") as f:
        (n,) = _read(f, "Q")
        for _ in range(n):
            cid, model_id, w, h = _read(f, "iiQQ")
            name, n_params = CAMERA_MODELS[model_id]
            params = np.array(_read(f, "d" * n_params))
            cams[cid] = Camera(cid, name, int(w), int(h), params)
    return cams


def read_images_binary(path: str) -> Dict[int, Image]:
    images: Dict[int, Image] = {}
    with open(path, "rb") as f:
        (n,) = _read(f, "Q")
        for _ in range(n):
            vals = _read(f, "idddddddi")
            iid, camera_id = vals[0], vals[8]
            qvec = np.array(vals[1:5])
            tvec = np.array(vals[5:8])
            name_bytes = bytearray()
            while True:
                c = f.read(1)
                if c == b"\x00" or c == b"":
                    break
                name_bytes += c
            (n2d,) = _read(f, "Q")
            rec = np.frombuffer(f.read(24 * n2d),
                                dtype=[("x", "<f8"), ("y", "<f8"), ("pid", "<i8")])
            images[iid] = Image(
                iid, qvec, tvec, camera_id, name_bytes.decode("utf-8"),
                np.stack([rec["x"], rec["y"]], axis=-1) if n2d else
                np.zeros((0, 2)),
                rec["pid"].copy())
    return images


def read_points3d_binary(path: str) -> Dict[int, Point3D]:
    points: Dict[int, Point3D] = {}
    with open(path, "rb") as f:
        (n,) = _read(f, "Q")
        for _ in range(n):
            vals = _read(f, "qdddBBBd")
            pid = vals[0]
            xyz = np.array(vals[1:4])
            rgb = np.array(vals[4:7], dtype=np.uint8)
            error = vals[7]
            (track_len,) = _read(f, "Q")
            rec = np.frombuffer(f.read(8 * track_len),
                                dtype=[("iid", "<i4"), ("p2d", "<i4")])
            points[pid] = Point3D(pid, xyz, rgb, float(error),
                                  rec["iid"].copy(), rec["p2d"].copy())
    return points


# ---------------------------------------------------------------------------
# text readers
# ---------------------------------------------------------------------------


def _text_lines(path: str):
    with open(path, "r") as f:
        for line in f:
            line = line.strip()
            if line and not line.startswith("#"):
                yield line


def read_cameras_text(path: str) -> Dict[int, Camera]:
    cams: Dict[int, Camera] = {}
    for line in _text_lines(path):
        parts = line.split()
        cid = int(parts[0])
        cams[cid] = Camera(cid, parts[1], int(parts[2]), int(parts[3]),
                           np.array([float(x) for x in parts[4:]]))
    return cams


def read_images_text(path: str) -> Dict[int, Image]:
    images: Dict[int, Image] = {}
    pending = None
    for line in _text_lines(path):
        if pending is None:
            parts = line.split()
            pending = Image(
                int(parts[0]),
                np.array([float(x) for x in parts[1:5]]),
                np.array([float(x) for x in parts[5:8]]),
                int(parts[8]), parts[9],
                np.zeros((0, 2)), np.zeros((0,), dtype=np.int64))
        else:
            vals = np.array(line.split(), dtype=np.float64).reshape(-1, 3)
            pending.xys = vals[:, :2]
            pending.point3D_ids = vals[:, 2].astype(np.int64)
            images[pending.id] = pending
            pending = None
    return images


def read_points3d_text(path: str) -> Dict[int, Point3D]:
    points: Dict[int, Point3D] = {}
    for line in _text_lines(path):
        parts = line.split()
        pid = int(parts[0])
        track = np.array(parts[8:], dtype=np.int64).reshape(-1, 2)
        points[pid] = Point3D(
            pid, np.array([float(x) for x in parts[1:4]]),
            np.array([int(x) for x in parts[4:7]], dtype=np.uint8),
            float(parts[7]), track[:, 0].astype(np.int32),
            track[:, 1].astype(np.int32))
    return points


# ---------------------------------------------------------------------------
# binary writers (round-trip tooling / test fixtures)
# ---------------------------------------------------------------------------


def write_cameras_binary(cams: Dict[int, Camera], path: str) -> None:
    with open(path, "wb") as f:
        f.write(struct.pack("<Q", len(cams)))
        for cam in cams.values():
            mid = CAMERA_MODEL_IDS[cam.model]
            f.write(struct.pack("<iiQQ", cam.id, mid, cam.width, cam.height))
            f.write(struct.pack("<" + "d" * len(cam.params), *cam.params))


def write_images_binary(images: Dict[int, Image], path: str) -> None:
    with open(path, "wb") as f:
        f.write(struct.pack("<Q", len(images)))
        for im in images.values():
            f.write(struct.pack("<idddddddi", im.id, *im.qvec, *im.tvec,
                                im.camera_id))
            f.write(im.name.encode("utf-8") + b"\x00")
            f.write(struct.pack("<Q", len(im.point3D_ids)))
            for (x, y), pid in zip(im.xys, im.point3D_ids):
                f.write(struct.pack("<ddq", x, y, int(pid)))


def write_points3d_binary(points: Dict[int, Point3D], path: str) -> None:
    with open(path, "wb") as f:
        f.write(struct.pack("<Q", len(points)))
        for pt in points.values():
            f.write(struct.pack("<qdddBBBd", pt.id, *pt.xyz,
                                *pt.rgb.astype(np.uint8), pt.error))
            f.write(struct.pack("<Q", len(pt.image_ids)))
            for iid, p2d in zip(pt.image_ids, pt.point2D_idxs):
                f.write(struct.pack("<ii", int(iid), int(p2d)))


def write_model(cameras, images, points3d, path: str) -> None:
    write_cameras_binary(cameras, os.path.join(path, "cameras.bin"))
    write_images_binary(images, os.path.join(path, "images.bin"))
    write_points3d_binary(points3d, os.path.join(path, "points3D.bin"))


# ---------------------------------------------------------------------------
# top-level reader (ref colmap_utils.py:420-439)
# ---------------------------------------------------------------------------


def read_model(path: str, ext: str = ".bin"):
    """Read (cameras, images, points3D) from a COLMAP sparse dir."""
    if ext == ".bin":
        return (read_cameras_binary(os.path.join(path, "cameras.bin")),
                read_images_binary(os.path.join(path, "images.bin")),
                read_points3d_binary(os.path.join(path, "points3D.bin")))
    if ext == ".txt":
        return (read_cameras_text(os.path.join(path, "cameras.txt")),
                read_images_text(os.path.join(path, "images.txt")),
                read_points3d_text(os.path.join(path, "points3D.txt")))
    raise ValueError(f"unknown model extension {ext!r}")
