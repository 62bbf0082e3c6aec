"""Structure-of-arrays measurement container for large graphs.

RelativeSEMeasurement objects are convenient at reference-API scale but
a 1M-pose graph has millions of edges — per-edge Python objects cost
minutes and GBs. MeasurementArray holds the same fields as numpy arrays
and feeds the vectorized assembly / partitioning / synthetic paths.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Sequence

import numpy as np

from .types import RelativeSEMeasurement


@dataclass
class MeasurementArray:
    r1: np.ndarray        # (ne,) int64
    r2: np.ndarray
    p1: np.ndarray
    p2: np.ndarray
    R: np.ndarray         # (ne, d, d)
    t: np.ndarray         # (ne, d)
    kappa: np.ndarray     # (ne,)
    tau: np.ndarray
    weight: np.ndarray    # (ne,) fp64
    is_known_inlier: np.ndarray  # (ne,) bool

    def __len__(self) -> int:
        return len(self.p1)

    @property
    def d(self) -> int:
        return self.t.shape[1]

    @staticmethod
    def from_list(meas: Sequence[RelativeSEMeasurement]) -> "MeasurementArray":
        ne = len(meas)
        assert ne > 0
        d = meas[0].d
        return MeasurementArray(
            r1=np.array([m.r1 for m in meas], dtype=np.int64),
            r2=np.array([m.r2 for m in meas], dtype=np.int64),
            p1=np.array([m.p1 for m in meas], dtype=np.int64),
            p2=np.array([m.p2 for m in meas], dtype=np.int64),
            R=np.stack([m.R for m in meas]).astype(np.float64),
            t=np.stack([m.t for m in meas]).astype(np.float64),
            kappa=np.array([m.kappa for m in meas], dtype=np.float64),
            tau=np.array([m.tau for m in meas], dtype=np.float64),
            weight=np.array([m.weight for m in meas], dtype=np.float64),
            is_known_inlier=np.array([m.is_known_inlier for m in meas],
                                     dtype=bool))

    def to_list(self) -> List[RelativeSEMeasurement]:
        return [RelativeSEMeasurement(
            int(self.r1[k]), int(self.r2[k]), int(self.p1[k]),
            int(self.p2[k]), self.R[k].copy(), self.t[k].copy(),
            float(self.kappa[k]), float(self.tau[k]),
            float(self.weight[k]), bool(self.is_known_inlier[k]))
            for k in range(len(self))]

    def select(self, mask: np.ndarray) -> "MeasurementArray":
        return MeasurementArray(
            self.r1[mask], self.r2[mask], self.p1[mask], self.p2[mask],
            self.R[mask], self.t[mask], self.kappa[mask], self.tau[mask],
            self.weight[mask], self.is_known_inlier[mask])

    @staticmethod
    def empty(d: int) -> "MeasurementArray":
        z = np.zeros(0, dtype=np.int64)
        f = np.zeros(0, dtype=np.float64)
        return MeasurementArray(z, z, z.copy(), z.copy(),
                                np.zeros((0, d, d)), np.zeros((0, d)),
                                f, f.copy(), np.ones(0), np.ones(0, bool))


def concat_measurement_arrays(parts: Sequence["MeasurementArray"]
                              ) -> MeasurementArray:
    parts = [p for p in parts if len(p) > 0]
    assert parts
    return MeasurementArray(
        *[np.concatenate([getattr(p, f) for p in parts])
          for f in ("r1", "r2", "p1", "p2", "R", "t", "kappa", "tau",
                    "weight", "is_known_inlier")])


def as_measurement_array(meas) -> MeasurementArray:
    if isinstance(meas, MeasurementArray):
        return meas
    return MeasurementArray.from_list(list(meas))


def partition_measurement_array(ma: MeasurementArray, num_poses: int,
                                part: Sequence[int], num_robots: int):
    """Vectorized split of a global-index MeasurementArray across robots.
    Returns (odometry, private, shared) per robot plus the global<->local
    pose index maps (same conventions as partition.partition_measurements).
    """
    part = np.asarray(part, dtype=np.int64)
    # local index = rank of the pose among its robot's poses, global order
    order = np.argsort(part, kind="stable")
    local_idx = np.zeros(num_poses, dtype=np.int64)
    counts = np.bincount(part, minlength=num_robots)
    starts = np.zeros(num_robots + 1, dtype=np.int64)
    starts[1:] = np.cumsum(counts)
    local_idx[order] = np.arange(num_poses) - starts[part[order]]
    global_of = [order[starts[rb]:starts[rb + 1]] for rb in range(num_robots)]

    src_r = part[ma.p1]
    dst_r = part[ma.p2]
    l1 = local_idx[ma.p1]
    l2 = local_idx[ma.p2]
    odo_mask = (src_r == dst_r) & (ma.p1 + 1 == ma.p2)
    priv_mask = (src_r == dst_r) & ~odo_mask
    shared_mask = src_r != dst_r

    def _mk(mask, rb_mask):
        sel = mask & rb_mask
        out = ma.select(sel)
        out.r1 = src_r[sel].copy()
        out.r2 = dst_r[sel].copy()
        out.p1 = l1[sel].copy()
        out.p2 = l2[sel].copy()
        return out

    odometry, private, shared = [], [], []
    for rb in range(num_robots):
        odometry.append(_mk(odo_mask, src_r == rb))
        private.append(_mk(priv_mask, src_r == rb))
        sh = _mk(shared_mask, (src_r == rb) | (dst_r == rb))
        shared.append(sh)
    return (odometry, private, shared, local_idx, global_of,
            [int(c) for c in counts])


def odometry_initialization_array(d: int, num_poses: int,
                                  odo: MeasurementArray) -> np.ndarray:
    """Vectorized dead-reckoning via prefix composition of SE(d)
    transforms (log-depth doubling scan). Matches reference
    odometryInitialization (DPGO_utils.cpp:411-432)."""
    dh = d + 1
    M = np.tile(np.eye(dh), (num_poses, 1, 1))
    # M[i] = relative transform from pose i-1 to i (identity for i = 0)
    idx = odo.p2
    M[idx, :d, :d] = odo.R
    M[idx, :d, d] = odo.t
    # prefix product P[i] = M[0] @ ... @ M[i]
    P = M.copy()
    shift = 1
    while shift < num_poses:
        # P[i] = P_prev[i - shift] @ P_prev[i] for i >= shift
        head = P[:num_poses - shift]
        tail = P[shift:]
        P = P.copy()
        P[shift:] = np.matmul(head, tail)
        shift *= 2
    T = np.zeros((d, num_poses * dh))
    T_view = T.reshape(d, num_poses, dh).transpose(1, 0, 2)  # (n, d, dh)
    T_view[:] = P[:, :d, :]
    return T
