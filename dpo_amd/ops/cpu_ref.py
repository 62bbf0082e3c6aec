"""PyTorch fp64 reference implementations of the dpo_amd hot ops.

Layout convention ("Xt layout", used framework-wide):
  X is a torch tensor of shape (N, r), N = (d+1) * n, fp64, where for pose
  i the rows [i*(d+1), i*(d+1)+d) hold Y_i^T (the transposed Stiefel
  component, d x r) and row i*(d+1)+d holds p_i^T (the translation, r).
  This is the transpose of the reference's r x (d+1)n matrices
  (QuadraticProblem.h:26-30); chosen so the BSR SpMM Q @ X reads/writes
  coalesced rows on the GPU.

The math mirrors:
  * tangent projection P_Y(V) = V - Y sym(Y^T V) per Stiefel block
    (ROPTLIB Stiefel projection, used at QuadraticProblem.cpp:82,95)
  * polar projection to St(d, r) = UV^T from the thin SVD
    (reference DPGO_utils.cpp:479-485)
  * Hessian-vec / gradient SpMM X*Q (QuadraticProblem.cpp:62-73), here
    Q @ Xt in the transposed layout (Q symmetric).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

Tensor = torch.Tensor


def _pose_view(X: Tensor, d: int) -> Tensor:
    """(N, r) -> (n, d+1, r) view."""
    N, r = X.shape
    dh = d + 1
    return X.view(N // dh, dh, r)


def tangent_project(X: Tensor, V: Tensor, d: int) -> Tensor:
    """Project ambient V onto the tangent space of (St(d,r) x R^r)^n at X.

    Stiefel part: P = V - Y sym(Y^T V); Euclidean rows pass through.
    In Xt layout with Yt = X-rows (d x r): S = sym(Yt @ Vt^T) (d x d),
    P_t = Vt - S @ Yt.
    """
    Xb = _pose_view(X, d)
    Vb = _pose_view(V, d).clone()
    Yt = Xb[:, :d, :]                      # (n, d, r)
    Vt = Vb[:, :d, :]
    A = torch.bmm(Yt, Vt.transpose(1, 2))  # (n, d, d) = Y^T V
    S = 0.5 * (A + A.transpose(1, 2))
    Vb[:, :d, :] = Vt - torch.bmm(S, Yt)
    return Vb.view_as(V)


def stiefel_project(M: Tensor, d: int) -> Tensor:
    """Map ambient M to the manifold: per-pose polar (= UV^T) of the Stiefel
    block; Euclidean rows unchanged. Equals the reference's per-pose
    projectToStiefelManifold (DPGO_utils.cpp:479-485,
    LiftedSEManifold.cpp:34-45)."""
    Mb = _pose_view(M, d).clone()
    Yt = Mb[:, :d, :]                      # (n, d, r), wide (d <= r)
    U, _, Vh = torch.linalg.svd(Yt, full_matrices=False)
    Mb[:, :d, :] = torch.bmm(U, Vh)
    return Mb.view_as(M)


def retract(X: Tensor, eta: Tensor, d: int) -> Tensor:
    """Polar retraction R_X(eta) = proj_manifold(X + eta)."""
    return stiefel_project(X + eta, d)


def spmm_q(Q: Tensor, X: Tensor) -> Tensor:
    """Q @ X with Q sparse CSR/COO (N, N) fp64, X (N, r)."""
    return torch.sparse.mm(Q, X)


def precond_factor(Qdiag_blocks: Tensor) -> Tensor:
    """Cholesky factors of the (regularized) diagonal blocks, (n, dh, dh)."""
    return torch.linalg.cholesky(Qdiag_blocks)


def precond_apply(L: Tensor, X: Tensor, V: Tensor, d: int) -> Tensor:
    """Block-Jacobi preconditioner: per-pose solve (LL^T) z = v followed by
    tangent projection at X (the reference applies Cholmod LDL^T of
    Q + 0.1 I then projects, QuadraticProblem.cpp:75-87; block-Jacobi is
    our GPU-friendly substitute — it only needs to precondition)."""
    dh = d + 1
    N, r = V.shape
    Vb = V.view(N // dh, dh, r)
    Z = torch.cholesky_solve(Vb, L)
    return tangent_project(X, Z.reshape(N, r), d)


def sym_block_diag(Q: Tensor, d: int, reg: float = 0.1) -> Tensor:
    """Extract (d+1)x(d+1) diagonal blocks of sparse Q, plus reg * I."""
    dh = d + 1
    N = Q.shape[0]
    n = N // dh
    Qc = Q.coalesce() if Q.layout == torch.sparse_coo else Q.to_sparse_coo().coalesce()
    idx = Qc.indices()
    val = Qc.values()
    rows, cols = idx[0], idx[1]
    bi = rows // dh
    bj = cols // dh
    mask = bi == bj
    br = rows[mask] % dh
    bc = cols[mask] % dh
    blocks = torch.zeros(n, dh, dh, dtype=Q.dtype, device=Q.device)
    blocks.index_put_((bi[mask], br, bc), val[mask], accumulate=True)
    blocks += reg * torch.eye(dh, dtype=Q.dtype, device=Q.device)
    return blocks
