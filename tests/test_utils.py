"""Math-unit tests mirroring reference tests/testUtils.cpp:
lifting-matrix determinism + orthonormality, manifold projection,
chi2inv, robust averaging with planted outliers."""
import math

import numpy as np
import pytest
import torch

from dpo_amd.averaging import (robust_single_pose_averaging,
                               robust_single_rotation_averaging,
                               single_rotation_averaging)
from dpo_amd.liegroups import (angular_to_chordal_so3, project_to_rotation_group,
                               random_rotation, rot_to_quat, quat_to_rot)
from dpo_amd.manifold import LiftedSEManifold, lifting_matrix
from dpo_amd.robust import RobustCost, chi2inv
from dpo_amd.types import RobustCostParams, RobustCostType


def test_lifting_matrix_orthonormal_and_deterministic():
    # reference tests/testUtils.cpp:12-35: all agents must derive the
    # same lifting matrix, and it must be a Stiefel point.
    d, r = 3, 5
    Y1 = lifting_matrix(d, r)
    Y2 = lifting_matrix(d, r)
    assert np.allclose(Y1, Y2)
    assert np.allclose(Y1.T @ Y1, np.eye(d), atol=1e-12)


def test_manifold_project_returns_stiefel_points():
    # reference tests/testUtils.cpp:38-53
    r, d, n = 5, 3, 7
    M = LiftedSEManifold(r, d, n)
    g = torch.Generator().manual_seed(0)
    X = torch.randn((d + 1) * n, r, dtype=torch.float64, generator=g)
    P = M.project(X)
    Pb = P.view(n, d + 1, r)
    for i in range(n):
        Yt = Pb[i, :d, :].numpy()
        assert np.allclose(Yt @ Yt.T, np.eye(d), atol=1e-10)
        # translations untouched
        assert torch.equal(Pb[i, d, :], X.view(n, d + 1, r)[i, d, :])


def test_tangent_projection_idempotent_and_tangent():
    r, d, n = 5, 3, 6
    M = LiftedSEManifold(r, d, n)
    g = torch.Generator().manual_seed(1)
    X = M.project(torch.randn((d + 1) * n, r, dtype=torch.float64, generator=g))
    V = torch.randn((d + 1) * n, r, dtype=torch.float64, generator=g)
    P = M.project_tangent(X, V)
    # tangency: Y^T P_y symmetric-free => Y^T P + P^T Y = 0
    Xb = X.view(n, d + 1, r)
    Pb = P.view(n, d + 1, r)
    for i in range(n):
        Yt, Pt = Xb[i, :d, :].numpy(), Pb[i, :d, :].numpy()
        S = Yt @ Pt.T
        assert np.allclose(S + S.T, 0, atol=1e-10)
    P2 = M.project_tangent(X, P)
    assert torch.allclose(P, P2, atol=1e-10)


def test_chi2inv_against_sampling():
    # reference tests/testUtils.cpp:55-70 (Monte-Carlo quantile check)
    rng = np.random.default_rng(0)
    dof, q = 6, 0.9
    draws = rng.chisquare(dof, size=100_000)
    emp = np.quantile(draws, q)
    assert abs(chi2inv(q, dof) - emp) < 0.05 * emp


def test_project_to_rotation_group():
    rng = np.random.default_rng(3)
    M = rng.standard_normal((3, 3))
    R = project_to_rotation_group(M)
    assert np.allclose(R.T @ R, np.eye(3), atol=1e-12)
    assert abs(np.linalg.det(R) - 1) < 1e-12


def test_quat_roundtrip():
    rng = np.random.default_rng(4)
    for _ in range(20):
        R = random_rotation(3, rng)
        q = rot_to_quat(R)
        assert np.allclose(quat_to_rot(*q), R, atol=1e-12)


def test_robust_rotation_averaging_trivial():
    # single inlier must be exact (reference tests/testUtils.cpp:72-105)
    rng = np.random.default_rng(5)
    R = random_rotation(3, rng)
    ROpt, inliers = robust_single_rotation_averaging(
        [R], None, angular_to_chordal_so3(0.5))
    assert np.allclose(ROpt, R, atol=1e-8)
    assert inliers == [0]


def test_robust_rotation_averaging_with_outliers():
    # 10 inliers + 40 well-separated outliers: recover truth + inlier set
    rng = np.random.default_rng(6)
    R_true = random_rotation(3, rng)
    RVec = []
    for _ in range(10):
        RVec.append(project_to_rotation_group(
            R_true @ random_rotation(3, rng, 0.01)))
    for _ in range(40):
        # outliers at least ~90 degrees away
        RVec.append(project_to_rotation_group(
            R_true @ random_rotation(3, rng) @ random_rotation(3, rng)))
    # reject outliers that landed close to truth
    keep = []
    for i, R in enumerate(RVec):
        if i < 10 or np.linalg.norm(R - R_true) > angular_to_chordal_so3(1.0):
            keep.append(R)
    RVec = keep[:10] + [R for R in keep[10:]]
    ROpt, inliers = robust_single_rotation_averaging(
        RVec, None, angular_to_chordal_so3(0.5))
    assert np.linalg.norm(ROpt - R_true) < 0.05
    assert set(inliers) == set(range(10))


def test_robust_pose_averaging_with_outliers():
    rng = np.random.default_rng(7)
    R_true = random_rotation(3, rng)
    t_true = rng.standard_normal(3)
    RVec, tVec = [], []
    for _ in range(10):
        RVec.append(project_to_rotation_group(
            R_true @ random_rotation(3, rng, 0.005)))
        tVec.append(t_true + 0.001 * rng.standard_normal(3))
    for _ in range(40):
        RVec.append(project_to_rotation_group(
            R_true @ random_rotation(3, rng) @ random_rotation(3, rng)))
        tVec.append(t_true + 5.0 + rng.standard_normal(3))
    cbar = RobustCost.error_threshold_at_quantile(0.9, 3)
    # reference defaults for pose averaging: kappa = 10000, tau = 100
    # (DPGO_utils.cpp:642-643) => outlier residuals >> cbar.
    ROpt, tOpt, inliers = robust_single_pose_averaging(
        RVec, tVec, None, None, cbar)
    assert np.linalg.norm(ROpt - R_true) < 0.05
    assert np.linalg.norm(tOpt - t_true) < 0.05
    assert set(inliers) == set(range(10))


def test_gnc_tls_weight_function():
    p = RobustCostParams(gnc_barc=1.0, gnc_init_mu=1.0)
    c = RobustCost(RobustCostType.GNC_TLS, p)
    # r^2 >= (mu+1)/mu * barc^2 = 2 -> weight 0
    assert c.weight(math.sqrt(2.0) + 1e-9) == 0.0
    # r^2 <= mu/(mu+1) barc^2 = 0.5 -> weight 1
    assert c.weight(math.sqrt(0.5) - 1e-9) == 1.0
    # in between: sqrt(barc^2 mu (mu+1) / r^2) - mu
    r = 1.0
    assert abs(c.weight(r) - (math.sqrt(2.0) - 1.0)) < 1e-12
    c.update()
    assert abs(c.mu - 1.4) < 1e-12


def test_robust_cost_menu():
    c = RobustCost(RobustCostType.L2, RobustCostParams())
    assert c.weight(5.0) == 1.0
    c = RobustCost(RobustCostType.L1, RobustCostParams())
    assert c.weight(4.0) == 0.25
    c = RobustCost(RobustCostType.Huber, RobustCostParams(huber_threshold=2.0))
    assert c.weight(1.0) == 1.0 and c.weight(4.0) == 0.5
    c = RobustCost(RobustCostType.TLS, RobustCostParams(tls_threshold=3.0))
    assert c.weight(2.0) == 1.0 and c.weight(4.0) == 0.0
    c = RobustCost(RobustCostType.GM, RobustCostParams())
    assert abs(c.weight(1.0) - 0.25) < 1e-12


def test_chordal_cgls_matches_direct():
    """The CGLS (torch, GPU-runnable) chordal path must match the direct
    sparse-LU solve: identical rounded rotations and translations to CG
    tolerance on a noisy SE(3) grid."""
    import numpy as np
    from dpo_amd.chordal import chordal_initialization
    from dpo_amd.synthetic import grid3d
    meas, n = grid3d(side=3, seed=11, rot_noise=0.15, tran_noise=0.1)
    T_direct = chordal_initialization(3, n, meas, method="direct")
    T_cgls = chordal_initialization(3, n, meas, method="cgls")
    assert np.abs(T_direct - T_cgls).max() < 1e-5
