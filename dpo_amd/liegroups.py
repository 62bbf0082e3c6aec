"""Small SO(2)/SO(3) helpers used by I/O, initialization and tests."""
from __future__ import annotations

import math

import numpy as np


def rot2(theta: float) -> np.ndarray:
    c, s = math.cos(theta), math.sin(theta)
    return np.array([[c, -s], [s, c]], dtype=np.float64)


def quat_to_rot(qx: float, qy: float, qz: float, qw: float) -> np.ndarray:
    """Unit quaternion (x, y, z, w) -> 3x3 rotation matrix."""
    n = math.sqrt(qx * qx + qy * qy + qz * qz + qw * qw)
    x, y, z, w = qx / n, qy / n, qz / n, qw / n
    return np.array([
        [1 - 2 * (y * y + z * z), 2 * (x * y - z * w), 2 * (x * z + y * w)],
        [2 * (x * y + z * w), 1 - 2 * (x * x + z * z), 2 * (y * z - x * w)],
        [2 * (x * z - y * w), 2 * (y * z + x * w), 1 - 2 * (x * x + y * y)],
    ], dtype=np.float64)


def rot_to_quat(R: np.ndarray) -> np.ndarray:
    """3x3 rotation -> unit quaternion (x, y, z, w), w >= 0."""
    t = np.trace(R)
    if t > 0:
        s = math.sqrt(t + 1.0) * 2.0
        w = 0.25 * s
        x = (R[2, 1] - R[1, 2]) / s
        y = (R[0, 2] - R[2, 0]) / s
        z = (R[1, 0] - R[0, 1]) / s
    else:
        i = int(np.argmax(np.diag(R)))
        j, k = (i + 1) % 3, (i + 2) % 3
        s = math.sqrt(max(R[i, i] - R[j, j] - R[k, k] + 1.0, 0.0)) * 2.0
        q = np.zeros(4)
        q[i] = 0.25 * s
        q[3] = (R[k, j] - R[j, k]) / s
        q[j] = (R[j, i] + R[i, j]) / s
        q[k] = (R[k, i] + R[i, k]) / s
        x, y, z, w = q[0], q[1], q[2], q[3]
    q = np.array([x, y, z, w], dtype=np.float64)
    if q[3] < 0:
        q = -q
    return q


def so3_exp(w: np.ndarray) -> np.ndarray:
    """Rodrigues exponential of a 3-vector."""
    th = float(np.linalg.norm(w))
    K = np.array([[0, -w[2], w[1]], [w[2], 0, -w[0]], [-w[1], w[0], 0]],
                 dtype=np.float64)
    if th < 1e-12:
        return np.eye(3) + K
    return (np.eye(3) + math.sin(th) / th * K
            + (1 - math.cos(th)) / (th * th) * (K @ K))


def random_rotation(d: int, rng: np.random.Generator,
                    scale: float = math.pi) -> np.ndarray:
    """Uniform-ish random rotation; scale < pi gives a small perturbation."""
    if d == 2:
        return rot2(float(rng.uniform(-scale, scale)))
    w = rng.standard_normal(3)
    nw = np.linalg.norm(w)
    if nw > 0:
        w = w / nw * float(rng.uniform(0, scale))
    return so3_exp(w)


def project_to_rotation_group(M: np.ndarray) -> np.ndarray:
    """Nearest SO(d) matrix: SVD with determinant sign fix
    (reference DPGO_utils.cpp:463-477)."""
    U, _, Vt = np.linalg.svd(M)
    if np.linalg.det(U) * np.linalg.det(Vt) < 0:
        U = U.copy()
        U[:, -1] *= -1
    return U @ Vt


def angular_to_chordal_so3(rad: float) -> float:
    """Angular distance (rad) -> chordal (Frobenius) distance on SO(3)
    (reference DPGO_utils.cpp:507-509)."""
    return 2.0 * math.sqrt(2.0) * math.sin(rad / 2.0)


def check_rotation_matrix(R: np.ndarray, tol: float = 1e-8) -> None:
    d = R.shape[0]
    assert abs(np.linalg.det(R) - 1.0) < tol
    assert np.linalg.norm(R.T @ R - np.eye(d)) < tol
