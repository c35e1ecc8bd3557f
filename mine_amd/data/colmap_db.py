"""COLMAP sqlite database helper (dataset-prep tooling).

Mirrors the tooling surface of the reference's
input_pipelines/database.py (ref database.py:139-227): create a COLMAP
``database.db`` with cameras / images / keypoints / descriptors /
matches tables and insert rows with numpy payloads. Not used by the
training path; kept for dataset preparation parity.

Schema follows the public COLMAP database format
(https://colmap.github.io/database.html).
"""
from __future__ import annotations

import sqlite3
from typing import Optional

import numpy as np

_MAX_IMAGE_ID = 2 ** 31 - 1

_SCHEMA = """
CREATE TABLE IF NOT EXISTS cameras (
    camera_id INTEGER PRIMARY KEY AUTOINCREMENT NOT NULL,
    model INTEGER NOT NULL,
    width INTEGER NOT NULL,
    height INTEGER NOT NULL,
    params BLOB,
    prior_focal_length INTEGER NOT NULL);
CREATE TABLE IF NOT EXISTS images (
    image_id INTEGER PRIMARY KEY AUTOINCREMENT NOT NULL,
    name TEXT NOT NULL UNIQUE,
    camera_id INTEGER NOT NULL,
    prior_qw REAL, prior_qx REAL, prior_qy REAL, prior_qz REAL,
    prior_tx REAL, prior_ty REAL, prior_tz REAL,
    CONSTRAINT image_id_check CHECK(image_id >= 0 and image_id < {maxid}),
    FOREIGN KEY(camera_id) REFERENCES cameras(camera_id));
CREATE TABLE IF NOT EXISTS keypoints (
    image_id INTEGER PRIMARY KEY NOT NULL,
    rows INTEGER NOT NULL, cols INTEGER NOT NULL, data BLOB,
    FOREIGN KEY(image_id) REFERENCES images(image_id) ON DELETE CASCADE);
CREATE TABLE IF NOT EXISTS descriptors (
    image_id INTEGER PRIMARY KEY NOT NULL,
    rows INTEGER NOT NULL, cols INTEGER NOT NULL, data BLOB,
    FOREIGN KEY(image_id) REFERENCES images(image_id) ON DELETE CASCADE);
CREATE TABLE IF NOT EXISTS matches (
    pair_id INTEGER PRIMARY KEY NOT NULL,
    rows INTEGER NOT NULL, cols INTEGER NOT NULL, data BLOB);
CREATE TABLE IF NOT EXISTS two_view_geometries (
    pair_id INTEGER PRIMARY KEY NOT NULL,
    rows INTEGER NOT NULL, cols INTEGER NOT NULL, data BLOB,
    config INTEGER NOT NULL,
    F BLOB, E BLOB, H BLOB);
""".format(maxid=_MAX_IMAGE_ID)


def image_ids_to_pair_id(image_id1: int, image_id2: int) -> int:
    if image_id1 > image_id2:
        image_id1, image_id2 = image_id2, image_id1
    return image_id1 * _MAX_IMAGE_ID + image_id2


def pair_id_to_image_ids(pair_id: int):
    return pair_id // _MAX_IMAGE_ID, pair_id % _MAX_IMAGE_ID


def _blob(arr: np.ndarray) -> bytes:
    return np.ascontiguousarray(arr).tobytes()


class COLMAPDatabase(sqlite3.Connection):
    @staticmethod
    def connect(path: str) -> "COLMAPDatabase":
        return sqlite3.connect(path, factory=COLMAPDatabase)

    def create_tables(self) -> None:
        self.executescript(_SCHEMA)

    def add_camera(self, model: int, width: int, height: int,
                   params: np.ndarray, prior_focal_length: bool = False,
                   camera_id: Optional[int] = None) -> int:
        cur = self.execute(
            "INSERT INTO cameras VALUES (?, ?, ?, ?, ?, ?)",
            (camera_id, model, width, height,
             _blob(np.asarray(params, np.float64)), int(prior_focal_length)))
        return cur.lastrowid

    def add_image(self, name: str, camera_id: int,
                  prior_q=(1.0, 0.0, 0.0, 0.0), prior_t=(0.0, 0.0, 0.0),
                  image_id: Optional[int] = None) -> int:
        cur = self.execute(
            "INSERT INTO images VALUES (?, ?, ?, ?, ?, ?, ?, ?, ?, ?)",
            (image_id, name, camera_id, *prior_q, *prior_t))
        return cur.lastrowid

    def add_keypoints(self, image_id: int, keypoints: np.ndarray) -> None:
        kp = np.asarray(keypoints, np.float32)
        assert kp.ndim == 2 and kp.shape[1] in (2, 4, 6)
        self.execute("INSERT INTO keypoints VALUES (?, ?, ?, ?)",
                     (image_id,) + kp.shape + (_blob(kp),))

    def add_descriptors(self, image_id: int, descriptors: np.ndarray) -> None:
        d = np.asarray(descriptors, np.uint8)
        self.execute("INSERT INTO descriptors VALUES (?, ?, ?, ?)",
                     (image_id,) + d.shape + (_blob(d),))

    def add_matches(self, image_id1: int, image_id2: int,
                    matches: np.ndarray) -> None:
        m = np.asarray(matches, np.uint32)
        assert m.ndim == 2 and m.shape[1] == 2
        if image_id1 > image_id2:
            m = m[:, ::-1]
        pair_id = image_ids_to_pair_id(image_id1, image_id2)
        self.execute("INSERT INTO matches VALUES (?, ?, ?, ?)",
                     (pair_id,) + m.shape + (_blob(m),))

    def add_two_view_geometry(self, image_id1: int, image_id2: int,
                              matches: np.ndarray, F=np.eye(3), E=np.eye(3),
                              H=np.eye(3), config: int = 2) -> None:
        m = np.asarray(matches, np.uint32)
        if image_id1 > image_id2:
            m = m[:, ::-1]
        pair_id = image_ids_to_pair_id(image_id1, image_id2)
        self.execute(
            "INSERT INTO two_view_geometries VALUES (?, ?, ?, ?, ?, ?, ?, ?)",
            (pair_id,) + m.shape + (_blob(m), config,
                                    _blob(np.asarray(F, np.float64)),
                                    _blob(np.asarray(E, np.float64)),
                                    _blob(np.asarray(H, np.float64))))
