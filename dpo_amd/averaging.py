"""Single rotation/translation/pose averaging and their robust (GNC-TLS)
variants, used for inter-robot frame alignment during distributed
initialization.

Parity: reference DPGO_utils.cpp:518-711. The GNC loop alternates
{weighted closed-form solve, TLS re-weighting, mu <- 1.4 mu} until every
weight saturates at 0 or 1. These run on a handful of candidate
transforms, so they stay host-side (SURVEY.md 2c last row).
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import numpy as np

from .liegroups import check_rotation_matrix, project_to_rotation_group
from .robust import RobustCost
from .types import RobustCostParams, RobustCostType

_W_TOL = 1e-8


def single_translation_averaging(t_vec: Sequence[np.ndarray],
                                 tau: np.ndarray | None = None) -> np.ndarray:
    n = len(t_vec)
    assert n > 0
    w = np.ones(n) if tau is None or len(tau) != n else np.asarray(tau)
    s = sum(wi * ti for wi, ti in zip(w, t_vec))
    return s / w.sum()


def single_rotation_averaging(R_vec: Sequence[np.ndarray],
                              kappa: np.ndarray | None = None) -> np.ndarray:
    n = len(R_vec)
    assert n > 0
    w = np.ones(n) if kappa is None or len(kappa) != n else np.asarray(kappa)
    M = sum(wi * Ri for wi, Ri in zip(w, R_vec))
    return project_to_rotation_group(M)


def single_pose_averaging(R_vec: Sequence[np.ndarray],
                          t_vec: Sequence[np.ndarray],
                          kappa: np.ndarray | None = None,
                          tau: np.ndarray | None = None
                          ) -> Tuple[np.ndarray, np.ndarray]:
    return (single_rotation_averaging(R_vec, kappa),
            single_translation_averaging(t_vec, tau))


def _gnc_loop(solve, residual_sq, n: int, kappa: np.ndarray,
              barc: float, max_iters: int):
    """Shared GNC-TLS alternation. solve(weights) updates the estimate;
    residual_sq() returns per-candidate squared residuals."""
    weights = np.ones(n)
    solve(weights)
    r_sq = residual_sq()
    barc_sq = barc * barc
    mu_init = barc_sq / (2.0 * float(np.max(r_sq)) - barc_sq)
    mu_init = min(mu_init, 1e-5)
    # Negative initial mu means residuals are already small: skip GNC.
    if mu_init > 0:
        params = RobustCostParams(gnc_barc=barc, gnc_max_iters=max_iters,
                                  gnc_init_mu=mu_init)
        cost = RobustCost(RobustCostType.GNC_TLS, params)
        for _ in range(max_iters):
            solve(weights)
            r_sq = residual_sq()
            nc = 0
            for i in range(n):
                wi = cost.weight(float(np.sqrt(r_sq[i])))
                if wi < _W_TOL or wi > 1 - _W_TOL:
                    nc += 1
                weights[i] = wi
            if nc == n:
                break
            cost.update()
    inliers = [i for i in range(n) if weights[i] > 1 - _W_TOL]
    return weights, inliers


def robust_single_rotation_averaging(
        R_vec: Sequence[np.ndarray], kappa: np.ndarray | None,
        error_threshold: float,
        max_iters: int = 1000) -> Tuple[np.ndarray, List[int]]:
    """GNC-TLS rotation averaging (reference DPGO_utils.cpp:567-629).
    Returns (R_opt, inlier_indices)."""
    n = len(R_vec)
    assert n > 0
    k = np.ones(n) if kappa is None or len(kappa) != n else np.asarray(kappa)
    for R in R_vec:
        check_rotation_matrix(R)
    state = {"R": single_rotation_averaging(R_vec, k)}

    def solve(w):
        state["R"] = single_rotation_averaging(R_vec, k * w)

    def res_sq():
        return np.array([k[i] * np.linalg.norm(state["R"] - R_vec[i]) ** 2
                         for i in range(n)])

    _, inliers = _gnc_loop(solve, res_sq, n, k, error_threshold, max_iters)
    return state["R"], inliers


def robust_single_pose_averaging(
        R_vec: Sequence[np.ndarray], t_vec: Sequence[np.ndarray],
        kappa: np.ndarray | None, tau: np.ndarray | None,
        error_threshold: float,
        max_iters: int = 10000
        ) -> Tuple[np.ndarray, np.ndarray, List[int]]:
    """GNC-TLS pose averaging (reference DPGO_utils.cpp:631-711).
    Returns (R_opt, t_opt, inlier_indices). Default weights mirror the
    reference: kappa = 10000, tau = 100 when not provided."""
    n = len(R_vec)
    assert n > 0 and len(t_vec) == n
    k = 10000.0 * np.ones(n) if kappa is None or len(kappa) != n else np.asarray(kappa)
    ta = 100.0 * np.ones(n) if tau is None or len(tau) != n else np.asarray(tau)
    for R in R_vec:
        check_rotation_matrix(R)
    state = {}
    state["R"], state["t"] = single_pose_averaging(R_vec, t_vec, k, ta)

    def solve(w):
        state["R"], state["t"] = single_pose_averaging(
            R_vec, t_vec, k * w, ta * w)

    def res_sq():
        return np.array([
            k[i] * np.linalg.norm(state["R"] - R_vec[i]) ** 2
            + ta[i] * np.linalg.norm(state["t"] - t_vec[i]) ** 2
            for i in range(n)])

    _, inliers = _gnc_loop(solve, res_sq, n, k, error_threshold, max_iters)
    return state["R"], state["t"], inliers


def compute_measurement_error(m, R1: np.ndarray, t1: np.ndarray,
                              R2: np.ndarray, t2: np.ndarray) -> float:
    """kappa ||R1 R - R2||_F^2 + tau ||t2 - t1 - R1 t||^2
    (reference DPGO_utils.cpp:494-500). Accepts lifted blocks (r x d /
    r-vectors) as well as rotations."""
    rot_err = float(np.linalg.norm(R1 @ m.R - R2) ** 2)
    tran_err = float(np.linalg.norm(t2 - t1 - R1 @ m.t) ** 2)
    return m.kappa * rot_err + m.tau * tran_err
