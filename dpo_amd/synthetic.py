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


# ---------------------------------------------------------------------
# Vectorized large-scale generator (SoA): synthetic grid3D at up to
# millions of poses (BASELINE.json config #5: 1M-pose grid, 8 agents,
# robust loop-closure rejection).
# ---------------------------------------------------------------------
def _random_rotations_batch(n: int, rng, scale: float = 0.5) -> np.ndarray:
    """(n, 3, 3) random rotations via batched Rodrigues."""
    w = rng.standard_normal((n, 3))
    nw = np.linalg.norm(w, axis=1, keepdims=True)
    nw[nw == 0] = 1.0
    ang = rng.uniform(0, scale, size=(n, 1))
    w = w / nw * ang
    th = np.linalg.norm(w, axis=1)
    K = np.zeros((n, 3, 3))
    K[:, 0, 1] = -w[:, 2]; K[:, 0, 2] = w[:, 1]
    K[:, 1, 0] = w[:, 2];  K[:, 1, 2] = -w[:, 0]
    K[:, 2, 0] = -w[:, 1]; K[:, 2, 1] = w[:, 0]
    th_safe = np.where(th < 1e-12, 1.0, th)
    a = np.where(th < 1e-12, 1.0, np.sin(th) / th_safe)[:, None, None]
    b = np.where(th < 1e-12, 0.5,
                 (1 - np.cos(th)) / (th_safe ** 2))[:, None, None]
    return np.eye(3)[None] + a * K + b * (K @ K)


def grid3d_soa(side: int, rot_noise: float = 0.05, tran_noise: float = 0.02,
               kappa: float = 1000.0, tau: float = 100.0,
               outlier_prob: float = 0.0, seed: int = 0):
    """Vectorized serpentine-grid SE(3) pose graph as a MeasurementArray.
    Identical structure to grid3d() but scales to millions of poses."""
    from .measurements import MeasurementArray
    rng = np.random.default_rng(seed)
    n = side ** 3
    idx = np.arange(n)
    z = idx // (side * side)
    rem = idx % (side * side)
    y = rem // side
    x = rem % side
    y = np.where(z % 2 == 1, side - 1 - y, y)
    x = np.where(y % 2 == 1, side - 1 - x, x)
    P = np.stack([x, y, z], axis=1).astype(np.float64)
    Rw = _random_rotations_batch(n, rng, 0.5)

    # index lookup grid: pos -> serpentine index
    lookup = np.empty((side, side, side), dtype=np.int64)
    lookup[x, y, z] = idx

    pairs = [np.stack([idx[:-1], idx[1:]], axis=1)]  # odometry chain
    for dxyz in ((1, 0, 0), (0, 1, 0), (0, 0, 1)):
        m = (x + dxyz[0] < side) & (y + dxyz[1] < side) & (z + dxyz[2] < side)
        j = lookup[np.clip(x + dxyz[0], 0, side - 1),
                   np.clip(y + dxyz[1], 0, side - 1),
                   np.clip(z + dxyz[2], 0, side - 1)]
        keep = m & (np.abs(j - idx) != 1)
        a = np.minimum(idx[keep], j[keep])
        b = np.maximum(idx[keep], j[keep])
        pairs.append(np.stack([a, b], axis=1))
    E = np.concatenate(pairs)
    # odometry must come first and stay ordered; loop closures after
    ne = len(E)
    i_, j_ = E[:, 0], E[:, 1]
    # relative measurements with noise
    Rn = _random_rotations_batch(ne, rng, rot_noise) if rot_noise > 0 \
        else np.tile(np.eye(3), (ne, 1, 1))
    Rrel = np.transpose(Rw[i_], (0, 2, 1)) @ Rw[j_] @ Rn
    trel = np.einsum('eij,ej->ei', np.transpose(Rw[i_], (0, 2, 1)),
                     P[j_] - P[i_])
    if tran_noise > 0:
        trel = trel + rng.standard_normal((ne, 3)) * tran_noise
    # outliers among loop closures only
    n_odo = n - 1
    if outlier_prob > 0:
        out_mask = np.zeros(ne, dtype=bool)
        lc = np.arange(n_odo, ne)
        out_mask[lc[rng.uniform(size=len(lc)) < outlier_prob]] = True
        n_out = int(out_mask.sum())
        if n_out:
            Rrel[out_mask] = _random_rotations_batch(n_out, rng, 3.14)
            trel[out_mask] = rng.standard_normal((n_out, 3)) * 5.0
    z64 = np.zeros(ne, dtype=np.int64)
    ma = MeasurementArray(
        r1=z64, r2=z64.copy(), p1=i_.astype(np.int64),
        p2=j_.astype(np.int64), R=Rrel, t=trel,
        kappa=np.full(ne, kappa), tau=np.full(ne, tau),
        weight=np.ones(ne), is_known_inlier=np.ones(ne, dtype=bool))
    ma.outlier_mask = out_mask if outlier_prob > 0 else \
        np.zeros(ne, dtype=bool)
    # stash ground truth (d, n*(d+1)) for warm-start experiments
    T_gt = np.zeros((3, n * 4))
    Tv = T_gt.reshape(3, n, 4).transpose(1, 0, 2)
    Tv[:, :, :3] = Rw
    Tv[:, :, 3] = P
    ma.ground_truth = T_gt
    return ma, n
