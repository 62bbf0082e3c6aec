"""Dataset I/O: .g2o pose-graph files, METIS .graph adjacency files and
partition-assignment files.

Parity: reference src/DPGO_utils.cpp:64-197 (read_g2o_file incl. the
information-divergence-minimizing isotropic precisions), reference
graph/<k>/<preset>/<dataset> partition files and graph/<k>/origin/*.graph
METIS files (examples/MultiRobotExample.cpp:76-110).
"""
from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

import numpy as np

from .liegroups import quat_to_rot, rot2, rot_to_quat
from .types import RelativeSEMeasurement


def read_g2o(path: str) -> Tuple[List[RelativeSEMeasurement], int]:
    """Parse EDGE_SE2 / EDGE_SE3:QUAT measurement lines.

    Returns (measurements, num_poses). Isotropic precisions follow the
    SE-Sync convention used by the reference (DPGO_utils.cpp:125-175):
      SE2: tau = 2 / trace(TranCov^-1) with TranCov from [I11 I12; I12 I22],
           kappa = I33.
      SE3: tau = 3 / trace(TranCov^-1), kappa = 3 / (2 trace(RotCov^-1)).
    VERTEX_* lines are ignored.
    """
    measurements: List[RelativeSEMeasurement] = []
    num_poses = 0
    with open(path, "r") as f:
        for line in f:
            tok = line.split()
            if not tok:
                continue
            tag = tok[0]
            if tag == "EDGE_SE2":
                i, j = int(tok[1]), int(tok[2])
                dx, dy, dth = (float(x) for x in tok[3:6])
                I11, I12, I13, I22, I23, I33 = (float(x) for x in tok[6:12])
                tran_cov = np.array([[I11, I12], [I12, I22]])
                tau = 2.0 / np.trace(np.linalg.inv(tran_cov))
                kappa = I33
                measurements.append(RelativeSEMeasurement(
                    0, 0, i, j, rot2(dth), np.array([dx, dy]), kappa, tau))
            elif tag == "EDGE_SE3:QUAT":
                i, j = int(tok[1]), int(tok[2])
                dx, dy, dz = (float(x) for x in tok[3:6])
                qx, qy, qz, qw = (float(x) for x in tok[6:10])
                I = [float(x) for x in tok[10:31]]
                (I11, I12, I13, I14, I15, I16, I22, I23, I24, I25, I26,
                 I33, I34, I35, I36, I44, I45, I46, I55, I56, I66) = I
                tran_cov = np.array([[I11, I12, I13],
                                     [I12, I22, I23],
                                     [I13, I23, I33]])
                rot_cov = np.array([[I44, I45, I46],
                                    [I45, I55, I56],
                                    [I46, I56, I66]])
                tau = 3.0 / np.trace(np.linalg.inv(tran_cov))
                kappa = 3.0 / (2.0 * np.trace(np.linalg.inv(rot_cov)))
                measurements.append(RelativeSEMeasurement(
                    0, 0, i, j, quat_to_rot(qx, qy, qz, qw),
                    np.array([dx, dy, dz]), kappa, tau))
            elif tag.startswith("VERTEX"):
                continue
            else:
                raise ValueError(f"unrecognized g2o record: {tag}")
            m = measurements[-1]
            num_poses = max(num_poses, m.p1, m.p2)
    return measurements, num_poses + 1


def write_g2o(path: str, measurements: Sequence[RelativeSEMeasurement]) -> None:
    """Write measurements back out (isotropic information matrices)."""
    with open(path, "w") as f:
        for m in measurements:
            if m.d == 2:
                th = float(np.arctan2(m.R[1, 0], m.R[0, 0]))
                # tau = 2/tr(TranCov^-1) inverts to I11=I22=tau, I12=0.
                f.write(f"EDGE_SE2 {m.p1} {m.p2} {m.t[0]:.9g} {m.t[1]:.9g} "
                        f"{th:.9g} {m.tau:.9g} 0 0 {m.tau:.9g} 0 "
                        f"{m.kappa:.9g}\n")
            else:
                q = rot_to_quat(m.R)
                # Inverse of the reader's precision formulas for isotropic
                # info blocks: info_rot = 2*kappa*I, info_tran = tau*I.
                rI = 2.0 * m.kappa
                tI = m.tau
                f.write(
                    f"EDGE_SE3:QUAT {m.p1} {m.p2} "
                    f"{m.t[0]:.9g} {m.t[1]:.9g} {m.t[2]:.9g} "
                    f"{q[0]:.9g} {q[1]:.9g} {q[2]:.9g} {q[3]:.9g} "
                    f"{tI:.9g} 0 0 0 0 0 {tI:.9g} 0 0 0 0 {tI:.9g} 0 0 0 "
                    f"{rI:.9g} 0 0 {rI:.9g} 0 {rI:.9g}\n")


def read_metis_graph(path: str) -> List[List[int]]:
    """Read a METIS-format adjacency file (header 'n m'; then per-vertex
    neighbor lists, 1-based). Returns 0-based adjacency lists."""
    with open(path) as f:
        lines = [ln for ln in (l.strip() for l in f)
                 if ln and not ln.startswith("%")]
    n, _m = (int(x) for x in lines[0].split()[:2])
    adj = [[int(v) - 1 for v in ln.split()] for ln in lines[1:n + 1]]
    return adj


def write_metis_graph(path: str, adj: Sequence[Sequence[int]]) -> None:
    m = sum(len(a) for a in adj) // 2
    with open(path, "w") as f:
        f.write(f"{len(adj)} {m}\n")
        for a in adj:
            f.write(" ".join(str(v + 1) for v in a) + "\n")


def read_partition_file(path: str) -> List[int]:
    """One partition id per pose line (reference
    examples/MultiRobotExample.cpp:78-91 format)."""
    with open(path) as f:
        return [int(ln.strip()) for ln in f if ln.strip()]


def write_partition_file(path: str, part: Sequence[int]) -> None:
    with open(path, "w") as f:
        for p in part:
            f.write(f"{p}\n")


def adjacency_from_measurements(
        measurements: Sequence[RelativeSEMeasurement],
        num_poses: int) -> List[List[int]]:
    """Undirected pose-adjacency (deduplicated) from global-index edges."""
    nbr: List[set] = [set() for _ in range(num_poses)]
    for m in measurements:
        if m.p1 == m.p2:
            continue
        nbr[m.p1].add(m.p2)
        nbr[m.p2].add(m.p1)
    return [sorted(s) for s in nbr]


def load_npz_dataset(path: str) -> Tuple[List[RelativeSEMeasurement], int]:
    """Load a dataset converted to .npz by scripts/convert_datasets
    (same measurement content as the .g2o original)."""
    z = np.load(path)
    d = int(z["d"])
    n = int(z["n"])
    R, t = z["R"], z["t"]
    p1, p2 = z["p1"], z["p2"]
    kappa, tau = z["kappa"], z["tau"]
    meas = [RelativeSEMeasurement(0, 0, int(p1[k]), int(p2[k]),
                                  R[k].copy(), t[k].copy(),
                                  float(kappa[k]), float(tau[k]))
            for k in range(len(p1))]
    return meas, n


def load_dataset(name_or_path: str) -> Tuple[List[RelativeSEMeasurement], int]:
    """Load by dataset name (data/<name>.npz), .npz path or .g2o path."""
    import os
    if name_or_path.endswith(".g2o"):
        return read_g2o(name_or_path)
    if name_or_path.endswith(".npz"):
        return load_npz_dataset(name_or_path)
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    p = os.path.join(here, "data", name_or_path + ".npz")
    if os.path.exists(p):
        return load_npz_dataset(p)
    raise FileNotFoundError(f"dataset {name_or_path} not found")
