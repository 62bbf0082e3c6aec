"""CSV persistence of trajectories and measurements (incl. GNC weights).

Parity: reference src/PGOLogger.cpp (C15): trajectories as
pose_index, qx, qy, qz, qw, tx, ty, tz rows; measurements as
robot_src, pose_src, robot_dst, pose_dst, qx, qy, qz, qw, tx, ty, tz,
kappa, tau, is_known_inlier, weight. The reference is 3D-only
(PGOLogger.cpp:26, 56); we support d=2 by embedding in 3D (theta ->
quaternion about z), and loadTrajectory/loadMeasurements give the
checkpoint/resume path together with PGOAgent.set_x.
"""
from __future__ import annotations

import csv
import os
from typing import List, Optional

import numpy as np

from .liegroups import quat_to_rot, rot_to_quat
from .types import RelativeSEMeasurement


def _embed_rot3(R: np.ndarray) -> np.ndarray:
    if R.shape[0] == 3:
        return R
    R3 = np.eye(3)
    R3[:2, :2] = R
    return R3


def _embed_t3(t: np.ndarray) -> np.ndarray:
    if t.shape[0] == 3:
        return t
    return np.array([t[0], t[1], 0.0])


class PGOLogger:
    def __init__(self, log_directory: str):
        self.log_dir = log_directory
        if log_directory:
            os.makedirs(log_directory, exist_ok=True)

    def _path(self, name: str) -> str:
        return os.path.join(self.log_dir, name)

    def log_trajectory(self, d: int, n: int, T: np.ndarray,
                       filename: str) -> None:
        """T: (d, (d+1) n) trajectory [R1 t1 ...]."""
        if not self.log_dir:
            return
        dh = d + 1
        with open(self._path(filename), "w", newline="") as f:
            w = csv.writer(f)
            w.writerow(["pose_index", "qx", "qy", "qz", "qw",
                        "tx", "ty", "tz"])
            for i in range(n):
                R = _embed_rot3(T[:, i * dh:i * dh + d])
                t = _embed_t3(T[:, i * dh + d])
                q = rot_to_quat(R)
                w.writerow([i, *[f"{v:.17g}" for v in q],
                            *[f"{v:.17g}" for v in t]])

    def load_trajectory(self, filename: str) -> np.ndarray:
        """Returns (3, 4 n) SE(3) trajectory."""
        rows = []
        with open(self._path(filename), newline="") as f:
            rd = csv.reader(f)
            header = next(rd)
            del header
            for row in rd:
                rows.append([float(x) for x in row])
        rows.sort(key=lambda r: r[0])
        n = len(rows)
        T = np.zeros((3, 4 * n))
        for i, row in enumerate(rows):
            _, qx, qy, qz, qw, tx, ty, tz = row
            T[:, i * 4:i * 4 + 3] = quat_to_rot(qx, qy, qz, qw)
            T[:, i * 4 + 3] = [tx, ty, tz]
        return T

    def log_measurements(self, measurements: List[RelativeSEMeasurement],
                         filename: str) -> None:
        if not self.log_dir:
            return
        with open(self._path(filename), "w", newline="") as f:
            w = csv.writer(f)
            w.writerow(["robot_src", "pose_src", "robot_dst", "pose_dst",
                        "qx", "qy", "qz", "qw", "tx", "ty", "tz",
                        "kappa", "tau", "is_known_inlier", "weight"])
            for m in measurements:
                q = rot_to_quat(_embed_rot3(m.R))
                t = _embed_t3(m.t)
                w.writerow([m.r1, m.p1, m.r2, m.p2,
                            *[f"{v:.17g}" for v in q],
                            *[f"{v:.17g}" for v in t],
                            f"{m.kappa:.17g}", f"{m.tau:.17g}",
                            int(m.is_known_inlier), f"{m.weight:.17g}"])

    def load_measurements(self, filename: str,
                          load_weights: bool = False
                          ) -> List[RelativeSEMeasurement]:
        out: List[RelativeSEMeasurement] = []
        with open(self._path(filename), newline="") as f:
            rd = csv.reader(f)
            next(rd)
            for row in rd:
                r1, p1, r2, p2 = (int(x) for x in row[:4])
                qx, qy, qz, qw, tx, ty, tz, kappa, tau = (
                    float(x) for x in row[4:13])
                known = bool(int(row[13]))
                weight = float(row[14]) if load_weights else 1.0
                m = RelativeSEMeasurement(
                    r1, r2, p1, p2, quat_to_rot(qx, qy, qz, qw),
                    np.array([tx, ty, tz]), kappa, tau, weight, known)
                out.append(m)
        return out
