"""Synthetic pose-graph generators.

Used by bench.py and GPU-box tests (the reference's .g2o datasets are not
shipped with the snapshot copy; there is no network). Shapes mirror the
reference datasets: 3D grid graphs (grid3D/smallGrid3D/tinyGrid3D style:
serpentine odometry chain over an LxLxL lattice + lattice-neighbor loop
closures) and noisy sphere graphs (sphere2500 style), plus a 2D city
grid (city10000 style). Optional outlier loop closures exercise the
GNC_TLS robust pipeline.
"""
from __future__ import annotations

import math
from typing import List, Optional, Tuple

import numpy as np

from .liegroups import project_to_rotation_group, random_rotation, rot2
from .types import RelativeSEMeasurement


def _relative(Ri, ti, Rj, tj, kappa, tau, rng, rot_noise, tran_noise,
              outlier: bool = False, d: int = 3):
    """Noisy relative measurement from pose i to pose j."""
    if outlier:
        R = random_rotation(d, rng)
        t = rng.standard_normal(d) * 5.0
    else:
        R = Ri.T @ Rj
        t = Ri.T @ (tj - ti)
        if rot_noise > 0:
            R = project_to_rotation_group(R @ random_rotation(d, rng, rot_noise))
        if tran_noise > 0:
            t = t + rng.standard_normal(d) * tran_noise
    return R, t


def grid3d(side: int = 5, rot_noise: float = 0.05, tran_noise: float = 0.02,
           kappa: float = 1000.0, tau: float = 100.0,
           loop_closure_prob: float = 1.0, outlier_prob: float = 0.0,
           seed: int = 0) -> Tuple[List[RelativeSEMeasurement], int]:
    """SE(3) grid graph: serpentine odometry over side^3 lattice points +
    loop closures between lattice neighbors."""
    rng = np.random.default_rng(seed)
    n = side ** 3

    # serpentine order -> (x, y, z)
    def coord(idx):
        z = idx // (side * side)
        rem = idx % (side * side)
        y = rem // side
        x = rem % side
        if z % 2 == 1:
            y = side - 1 - y
        if y % 2 == 1:
            x = side - 1 - x
        return np.array([x, y, z], dtype=np.float64)

    # ground-truth poses: position = lattice coord, random small rotations
    Rs = [random_rotation(3, rng, 0.5) for _ in range(n)]
    ts = [coord(i) for i in range(n)]

    pos_of = {}
    for i in range(n):
        pos_of[tuple(int(v) for v in ts[i])] = i

    meas: List[RelativeSEMeasurement] = []
    for i in range(n - 1):
        R, t = _relative(Rs[i], ts[i], Rs[i + 1], ts[i + 1], kappa, tau,
                         rng, rot_noise, tran_noise)
        meas.append(RelativeSEMeasurement(0, 0, i, i + 1, R, t, kappa, tau))
    # lattice-neighbor loop closures
    for i in range(n):
        x, y, z = (int(v) for v in ts[i])
        for dxyz in ((1, 0, 0), (0, 1, 0), (0, 0, 1)):
            nb = (x + dxyz[0], y + dxyz[1], z + dxyz[2])
            j = pos_of.get(nb)
            if j is None or abs(j - i) == 1:
                continue
            if rng.uniform() > loop_closure_prob:
                continue
            outlier = rng.uniform() < outlier_prob
            R, t = _relative(Rs[i], ts[i], Rs[j], ts[j], kappa, tau,
                             rng, rot_noise, tran_noise, outlier)
            a, b = (i, j) if i < j else (j, i)
            if a != i:
                # re-express measurement from a to b
                R = R.T
                t = -R @ t
            meas.append(RelativeSEMeasurement(0, 0, a, b, R, t, kappa, tau))
    return meas, n


def sphere(n: int = 2500, loops_per_pose: float = 1.0,
           rot_noise: float = 0.05, tran_noise: float = 0.05,
           kappa: float = 500.0, tau: float = 100.0,
           outlier_prob: float = 0.0, seed: int = 0
           ) -> Tuple[List[RelativeSEMeasurement], int]:
    """SE(3) sphere graph (sphere2500 style): spiral odometry path on a
    sphere surface + loop closures between nearby rings."""
    rng = np.random.default_rng(seed)
    radius = 10.0
    Rs, ts = [], []
    for i in range(n):
        # spherical spiral
        h = -1.0 + 2.0 * i / (n - 1)
        th = math.acos(max(-1.0, min(1.0, h)))
        ph = math.sqrt(n * math.pi) * th
        p = radius * np.array([math.sin(th) * math.cos(ph),
                               math.sin(th) * math.sin(ph),
                               math.cos(th)])
        ts.append(p)
        Rs.append(random_rotation(3, rng, 0.5))
    meas: List[RelativeSEMeasurement] = []
    for i in range(n - 1):
        R, t = _relative(Rs[i], ts[i], Rs[i + 1], ts[i + 1], kappa, tau,
                         rng, rot_noise, tran_noise)
        meas.append(RelativeSEMeasurement(0, 0, i, i + 1, R, t, kappa, tau))
    # loop closures: connect to spatially-near earlier poses
    n_loops = int(loops_per_pose * n)
    P = np.stack(ts)
    for _ in range(n_loops):
        i = int(rng.integers(0, n))
        dist = np.linalg.norm(P - P[i], axis=1)
        dist[max(0, i - 5):i + 6] = 1e9
        j = int(np.argmin(dist + rng.uniform(0, 0.3, n)))
        if dist[j] > 2.5:
            continue
        a, b = (min(i, j), max(i, j))
        if a == b:
            continue
        outlier = rng.uniform() < outlier_prob
        R, t = _relative(Rs[a], ts[a], Rs[b], ts[b], kappa, tau,
                         rng, rot_noise, tran_noise, outlier)
        meas.append(RelativeSEMeasurement(0, 0, a, b, R, t, kappa, tau))
    return meas, n


def city2d(side: int = 100, rot_noise: float = 0.03, tran_noise: float = 0.05,
           kappa: float = 300.0, tau: float = 150.0,
           outlier_prob: float = 0.0, seed: int = 0
           ) -> Tuple[List[RelativeSEMeasurement], int]:
    """SE(2) city-block graph (city10000 style): serpentine sweep of a
    side x side street grid with lattice loop closures."""
    rng = np.random.default_rng(seed)
    n = side * side

    def coord(idx):
        y = idx // side
        x = idx % side
        if y % 2 == 1:
            x = side - 1 - x
        return np.array([x, y], dtype=np.float64)

    Rs = [rot2(float(rng.uniform(-math.pi, math.pi))) for _ in range(n)]
    ts = [coord(i) for i in range(n)]
    pos_of = {tuple(int(v) for v in ts[i]): i for i in range(n)}

    meas: List[RelativeSEMeasurement] = []
    for i in range(n - 1):
        R, t = _relative(Rs[i], ts[i], Rs[i + 1], ts[i + 1], kappa, tau,
                         rng, rot_noise, tran_noise, d=2)
        meas.append(RelativeSEMeasurement(0, 0, i, i + 1, R, t, kappa, tau))
    for i in range(n):
        x, y = (int(v) for v in ts[i])
        for dxy in ((1, 0), (0, 1)):
            j = pos_of.get((x + dxy[0], y + dxy[1]))
            if j is None or abs(j - i) == 1:
                continue
            outlier = rng.uniform() < outlier_prob
            a, b = (min(i, j), max(i, j))
            R, t = _relative(Rs[a], ts[a], Rs[b], ts[b], kappa, tau,
                             rng, rot_noise, tran_noise, outlier, d=2)
            meas.append(RelativeSEMeasurement(0, 0, a, b, R, t, kappa, tau))
    return meas, n


def triangle_graph() -> Tuple[List[RelativeSEMeasurement], int,
                              np.ndarray]:
    """3-pose exact SE(3) triangle (odometry x2 + 1 loop closure) with
    known ground truth — the consistency fixture of reference
    tests/testTriangleGraph.cpp: exact data => the solver must stay at
    the optimum. Returns (measurements, n, T_truth (3, 12))."""
    rng = np.random.default_rng(42)
    Rs = [np.eye(3)] + [random_rotation(3, rng, 1.0) for _ in range(2)]
    ts = [np.zeros(3), np.array([1.0, 0.2, -0.1]), np.array([1.5, 1.1, 0.4])]
    kappa, tau = 100.0, 100.0
    meas = []
    for (i, j) in ((0, 1), (1, 2), (0, 2)):
        R = Rs[i].T @ Rs[j]
        t = Rs[i].T @ (ts[j] - ts[i])
        meas.append(RelativeSEMeasurement(0, 0, i, j, R, t, kappa, tau))
    T = np.zeros((3, 12))
    for i in range(3):
        T[:, i * 4:i * 4 + 3] = Rs[i]
        T[:, i * 4 + 3] = ts[i]
    return meas, 3, T
