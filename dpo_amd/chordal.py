"""Chordal and odometry initialization.

Parity: reference DPGO_utils.cpp:273-461 — B1/B2/B3 matrices of eqs.
(69a-c) of the SE-Sync tech report; rotation chordal relaxation solved as
a sparse least-squares problem with the first pose pinned to identity;
per-pose SO(d) rounding; translations recovered from a second sparse
least-squares solve. The reference uses SuiteSparse SPQR; we solve the
(well-conditioned) normal equations with a sparse Cholesky/LU
factorization on CPU, and a preconditioned CG path for the GPU.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import numpy as np
import scipy.sparse as sp
import scipy.sparse.linalg as spla

from .liegroups import project_to_rotation_group
from .types import RelativeSEMeasurement


def construct_b_matrices(measurements: Sequence[RelativeSEMeasurement],
                         num_poses: int, d: int):
    """Sparse B1 (d m x d n), B2 (d m x d^2 n), B3 (d^2 m x d^2 n)."""
    m = len(measurements)
    d2 = d * d

    rows1: List[int] = []; cols1: List[int] = []; vals1: List[float] = []
    rows2: List[int] = []; cols2: List[int] = []; vals2: List[float] = []
    rows3: List[int] = []; cols3: List[int] = []; vals3: List[float] = []

    for e, ms in enumerate(measurements):
        i, j = ms.p1, ms.p2
        sqrttau = np.sqrt(ms.tau)
        sqrtkappa = np.sqrt(ms.kappa)
        for l in range(d):
            rows1 += [e * d + l, e * d + l]
            cols1 += [i * d + l, j * d + l]
            vals1 += [-sqrttau, sqrttau]
        for k in range(d):
            for rr in range(d):
                rows2.append(d * e + rr)
                cols2.append(d2 * i + d * k + rr)
                vals2.append(-sqrttau * ms.t[k])
        R = ms.R
        for rr in range(d):
            for c in range(d):
                for l in range(d):
                    rows3.append(e * d2 + d * rr + l)
                    cols3.append(i * d2 + d * c + l)
                    vals3.append(-sqrtkappa * R[c, rr])
        for l in range(d2):
            rows3.append(e * d2 + l)
            cols3.append(j * d2 + l)
            vals3.append(sqrtkappa)

    B1 = sp.csc_matrix((vals1, (rows1, cols1)), shape=(d * m, d * num_poses))
    B2 = sp.csc_matrix((vals2, (rows2, cols2)), shape=(d * m, d2 * num_poses))
    B3 = sp.csc_matrix((vals3, (rows3, cols3)), shape=(d2 * m, d2 * num_poses))
    return B1, B2, B3


def _sparse_lsq(A: sp.spmatrix, b: np.ndarray) -> np.ndarray:
    """min ||A x + b||_2 via normal equations with a small Tikhonov-free
    sparse factorization (the systems here are full-rank by construction
    once the first pose is pinned)."""
    import warnings
    AtA = (A.T @ A).tocsc()
    Atb = A.T @ b
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        try:
            x = spla.spsolve(AtA, -Atb)
        except RuntimeError:
            x = None
    if x is None or not np.all(np.isfinite(x)):
        # rank-deficient (e.g. a disconnected local subgraph): min-norm
        # least-squares via LSQR
        x = spla.lsqr(A, -b, atol=1e-12, btol=1e-12, iter_lim=20000)[0]
    return np.asarray(x)


def _sparse_lsq_cgls(A: sp.spmatrix, b: np.ndarray, device: str = "cpu",
                     tol: float = 1e-10, max_iters: int = 20000
                     ) -> np.ndarray:
    """min ||A x + b||_2 via CGLS (CG on the normal equations without
    forming A^T A), with diagonal column scaling. Pure torch ops, so the
    same code runs on CPU or a GPU — the offload path the reference's
    SPQR cannot take. Accuracy matches the direct solve to the CG
    tolerance (see tests/test_utils.py::test_chordal_cgls_matches_direct)."""
    import torch
    A = A.tocsr()
    # column scaling: unit-norm columns precondition the normal equations
    cn = np.sqrt(np.asarray(A.multiply(A).sum(axis=0)).ravel())
    cn[cn == 0.0] = 1.0
    D = sp.diags(1.0 / cn)
    As = (A @ D).tocsr()
    At = As.T.tocsr()

    def to_torch(M):
        return torch.sparse_csr_tensor(
            torch.from_numpy(M.indptr.astype(np.int64)),
            torch.from_numpy(M.indices.astype(np.int64)),
            torch.from_numpy(M.data.astype(np.float64)),
            size=M.shape, device=device)

    tA = to_torch(As)
    tAt = to_torch(At)
    rhs = torch.from_numpy(np.ascontiguousarray(-b)).to(device)
    x = torch.zeros(As.shape[1], dtype=torch.float64, device=device)
    r = rhs.clone()
    s = tAt @ r.unsqueeze(1)
    p = s.clone()
    gamma = float((s * s).sum())
    g0 = gamma
    for _ in range(max_iters):
        if gamma <= tol * tol * max(g0, 1e-300):
            break
        q = tA @ p
        qq = float((q * q).sum())
        if qq == 0.0:
            break
        alpha = gamma / qq
        x += alpha * p.squeeze(1)
        r -= alpha * q.squeeze(1)
        s = tAt @ r.unsqueeze(1)
        g_new = float((s * s).sum())
        p = s + (g_new / gamma) * p
        gamma = g_new
    return (x.cpu().numpy() / cn)


def chordal_initialization(d: int, num_poses: int,
                           measurements: Sequence[RelativeSEMeasurement],
                           method: str = "direct",
                           device: str = "cpu") -> np.ndarray:
    """Returns T (d, (d+1) n): [R1 t1 R2 t2 ...] with pose 0 = identity.
    method="direct": sparse LU of the normal equations (reference-exact
    quality); method="cgls": iterative CGLS in torch, runnable on a GPU."""
    assert measurements, "chordal initialization needs measurements"
    d2 = d * d
    B1, B2, B3 = construct_b_matrices(measurements, num_poses, d)
    lsq = (_sparse_lsq if method == "direct"
           else lambda A, b: _sparse_lsq_cgls(A, b, device=device))

    # Rotations: pin pose 0 to I, solve for the rest.
    B3red = B3[:, d2:]
    Id = np.eye(d)
    cR = B3[:, :d2] @ Id.flatten(order="F")
    rvec = lsq(B3red, cR)
    Rall = np.zeros((d, d * num_poses))
    Rall[:, :d] = Id
    Rall[:, d:] = rvec.reshape(d, d * (num_poses - 1), order="F")
    for i in range(1, num_poses):
        Rall[:, i * d:(i + 1) * d] = project_to_rotation_group(
            Rall[:, i * d:(i + 1) * d])

    # Translations from the rounded rotations.
    t = recover_translations(B1, B2, Rall, method=method, device=device)

    T = np.zeros((d, num_poses * (d + 1)))
    for i in range(num_poses):
        T[:, i * (d + 1):i * (d + 1) + d] = Rall[:, i * d:(i + 1) * d]
        T[:, i * (d + 1) + d] = t[:, i]
    return T


def recover_translations(B1: sp.spmatrix, B2: sp.spmatrix,
                         R: np.ndarray, method: str = "direct",
                         device: str = "cpu") -> np.ndarray:
    """Second least-squares solve for translations given rotations
    (reference DPGO_utils.cpp:434-461)."""
    d = R.shape[0]
    n = R.shape[1] // d
    rvec = R.flatten(order="F")
    B1red = B1[:, d:]
    c = B2 @ rvec
    if method == "direct":
        tred = _sparse_lsq(B1red, c)
    else:
        tred = _sparse_lsq_cgls(B1red, c, device=device)
    t = np.zeros((d, n))
    t[:, 1:] = tred.reshape(d, n - 1, order="F")
    return t


def odometry_initialization(d: int, num_poses: int,
                            odometry: Sequence[RelativeSEMeasurement]
                            ) -> np.ndarray:
    """Dead-reckoning T_{i+1} = T_i * m_i from identity
    (reference DPGO_utils.cpp:411-432). Gap-tolerant: a robot built from
    a non-contiguous (multilevel) partition may be missing odometry
    steps between locally consecutive poses — those gaps propagate with
    an identity step (matching odometry_initialization_array)."""
    dh = d + 1
    step = {}
    for m in odometry:
        assert m.p2 == m.p1 + 1, "odometry edges must link consecutive poses"
        step[m.p1] = m
    T = np.zeros((d, num_poses * dh))
    T[:, 0:d] = np.eye(d)
    for src in range(num_poses - 1):
        dst = src + 1
        Rsrc = T[:, src * dh:src * dh + d]
        tsrc = T[:, src * dh + d]
        m = step.get(src)
        if m is None:
            T[:, dst * dh:dst * dh + d] = Rsrc
            T[:, dst * dh + d] = tsrc
        else:
            T[:, dst * dh:dst * dh + d] = Rsrc @ m.R
            T[:, dst * dh + d] = tsrc + Rsrc @ m.t
    return T


# ---------------------------------------------------------------------
# SoA / GPU-native chordal initialization (no B matrices, no Python
# per-edge loops). Mathematically equivalent to the reference's SPQR
# least-squares (DPGO_utils.cpp:273-409): the rotation stage solves the
# normal equations of min sum_e kappa_e ||R_j - R_i R_e||_F^2 with pose
# 0 pinned — i.e. the rotation connection Laplacian system — by
# preconditioned CG whose SpMV is our block-CSR HIP kernel; rotations
# are rounded to SO(d) by batched SVD; the translation stage solves the
# tau-weighted scalar graph Laplacian the same way. Works on the
# MeasurementArray directly, so a 1M-pose chordal init assembles in
# vectorized numpy and iterates on the GPU.
# ---------------------------------------------------------------------
def _bsr_from_coo(rows, cols, blocks, n, dh, device="cpu"):
    import torch
    from .quadratic import BSRMatrix
    order = np.lexsort((cols, rows))
    rows, cols = rows[order], cols[order]
    blocks = blocks[order]
    key = rows * n + cols
    uniq, inv = np.unique(key, return_inverse=True)
    acc = np.zeros((len(uniq), dh, dh))
    np.add.at(acc, inv, blocks)
    urows = (uniq // n).astype(np.int64)
    ucols = (uniq % n).astype(np.int32)
    row_ptr = np.zeros(n + 1, dtype=np.int32)
    np.add.at(row_ptr, urows + 1, 1)
    row_ptr = np.cumsum(row_ptr, dtype=np.int64).astype(np.int32)
    dev = torch.device(device)
    return BSRMatrix(n, dh,
                     torch.from_numpy(row_ptr).to(dev),
                     torch.from_numpy(ucols).to(dev),
                     torch.from_numpy(acc).to(dev))


def _pcg_block(A, rhs, diag_inv, tol, max_iters):
    """CG on A x = rhs (x: (N, r) torch), Jacobi-scaled."""
    import torch
    x = torch.zeros_like(rhs)
    r = rhs.clone()
    z = r * diag_inv
    p = z.clone()
    rz = float((r * z).sum())
    rhs_n = float((rhs * rhs).sum())
    if rhs_n == 0.0:
        return x, 0
    it = 0
    for it in range(max_iters):
        Ap = A.spmm(p)
        pAp = float((p * Ap).sum())
        if pAp <= 0:
            break
        alpha = rz / pAp
        x += alpha * p
        r -= alpha * Ap
        if float((r * r).sum()) <= tol * tol * rhs_n:
            break
        z = r * diag_inv
        rz_new = float((r * z).sum())
        p = z + (rz_new / rz) * p
        rz = rz_new
    return x, it + 1


def chordal_initialization_soa(ma, num_poses: int, device: str = "cpu",
                               tol: float = 1e-8,
                               max_iters: int = 1000) -> np.ndarray:
    """Chordal initialization from a MeasurementArray (global indices).
    Returns T (d, (d+1) n) with pose 0 = identity, like
    chordal_initialization."""
    import torch
    d = ma.d
    n = num_poses
    p1 = ma.p1.astype(np.int64)
    p2 = ma.p2.astype(np.int64)
    kap = ma.kappa
    Re = ma.R  # (ne, d, d)
    Id = np.eye(d)

    # ---- rotation stage: unknowns X_i = R_i^T, poses 1..n-1 ----------
    # edge (i,j): A_ii += k I, A_jj += k I, A_ij -= k R_e, A_ji -= k R_e^T
    m = n - 1  # unknown poses (shifted by -1)
    i_u = p1 - 1
    j_u = p2 - 1
    rows_l, cols_l, blks_l = [], [], []

    both = (p1 > 0) & (p2 > 0)
    for sel, rr, cc, bb in (
            (p1 > 0, i_u, i_u, kap[:, None, None] * Id[None]),
            (p2 > 0, j_u, j_u, kap[:, None, None] * Id[None]),
            (both, i_u, j_u, -kap[:, None, None] * Re),
            (both, j_u, i_u,
             -kap[:, None, None] * np.transpose(Re, (0, 2, 1)))):
        rows_l.append(rr[sel])
        cols_l.append(cc[sel])
        blks_l.append(np.ascontiguousarray(bb[sel]))
    A = _bsr_from_coo(np.concatenate(rows_l), np.concatenate(cols_l),
                      np.concatenate(blks_l), m, d, device)
    rhs = np.zeros((m, d, d))
    s = (p1 == 0)
    if s.any():  # edge (0, j): RHS_j += k R_e^T
        np.add.at(rhs, j_u[s],
                  kap[s, None, None] * np.transpose(Re[s], (0, 2, 1)))
    s = (p2 == 0)
    if s.any():  # edge (i, 0): RHS_i += k R_e
        np.add.at(rhs, i_u[s], kap[s, None, None] * Re[s])
    dev = torch.device(device)
    rhs_t = torch.from_numpy(rhs.reshape(m * d, d)).to(dev)
    dinv = 1.0 / A.diag_blocks()[:, 0, 0]  # diagonal blocks are c*I
    dinv_t = dinv.repeat_interleave(d).unsqueeze(1)
    X, _ = _pcg_block(A, rhs_t, dinv_t, tol, max_iters)
    # round to SO(d): R_i = proj(X_i^T), batched SVD on the device
    Xt = X.view(m, d, d).transpose(1, 2).contiguous()
    U, _, Vh = torch.linalg.svd(Xt)
    det = torch.linalg.det(U @ Vh)
    D = torch.ones(m, d, dtype=torch.float64, device=dev)
    D[:, -1] = torch.sign(det + (det == 0))
    Rall = (U * D.unsqueeze(1)) @ Vh  # (m, d, d)
    R_full = torch.empty(n, d, d, dtype=torch.float64, device=dev)
    R_full[0] = torch.eye(d, dtype=torch.float64, device=dev)
    R_full[1:] = Rall

    # ---- translation stage: scalar tau-Laplacian, d RHS columns ------
    tau = ma.tau
    c_e = torch.einsum(
        "eij,ej->ei", R_full.index_select(0, torch.from_numpy(p1).to(dev)),
        torch.from_numpy(ma.t).to(dev)).cpu().numpy()  # R_i t_e
    rows_l, cols_l, vals_l = [], [], []
    for sel, rr, cc, vv in (
            (p1 > 0, i_u, i_u, tau),
            (p2 > 0, j_u, j_u, tau),
            (both, i_u, j_u, -tau),
            (both, j_u, i_u, -tau)):
        rows_l.append(rr[sel])
        cols_l.append(cc[sel])
        vals_l.append(vv[sel].reshape(-1, 1, 1))
    At = _bsr_from_coo(np.concatenate(rows_l), np.concatenate(cols_l),
                       np.concatenate(vals_l), m, 1, device)
    rhs_tr = np.zeros((m, d))
    s1 = (p1 > 0)
    np.add.at(rhs_tr, i_u[s1], -(tau[s1, None] * c_e[s1]))
    s2 = (p2 > 0)
    np.add.at(rhs_tr, j_u[s2], tau[s2, None] * c_e[s2])
    rhs_tr_t = torch.from_numpy(rhs_tr).to(dev)
    dinv_tr = (1.0 / At.diag_blocks()[:, 0, 0]).unsqueeze(1)
    tsol, _ = _pcg_block(At, rhs_tr_t, dinv_tr, tol, max_iters)

    dh = d + 1
    T = np.zeros((d, n * dh))
    Tv = T.reshape(d, n, dh).transpose(1, 0, 2)  # (n, d, dh) view
    Tv[:, :, :d] = R_full.cpu().numpy()
    Tv[1:, :, d] = tsol.cpu().numpy()
    return T
