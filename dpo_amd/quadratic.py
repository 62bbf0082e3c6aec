"""Block-sparse SE(d) quadratic cost: Q (connection Laplacian) and G.

The data matrix Q is assembled directly in (d+1)x(d+1) block-CSR (BSR)
form — the layout the CDNA4 kernels consume from HBM — instead of the
reference's scalar triplet assembly via an oriented incidence product
A * Omega * A^T (DPGO_utils.cpp:199-271). The two agree: for an edge
e = (i -> j) with T = [R t; 0 1] and Omega = diag(w*kappa I_d, w*tau),

    Q_ii += T Omega T^T      Q_ij += -T Omega
    Q_jj += Omega            Q_ji += -(T Omega)^T

Per-agent Q additionally gets the shared-edge diagonal corrections
(T Omega T^T at an outgoing public pose, Omega at an incoming one),
mirroring PGOAgent::constructQMatrix (PGOAgent.cpp:720-781).

The linear term G (PGOAgent::constructGMatrix, PGOAgent.cpp:783-859)
couples local public poses to fixed neighbor poses; in the Xt layout the
per-edge updates are Gt[p1] += -(T Omega) @ Xj_t (outgoing) and
Gt[p2] += -(T Omega)^T @ Xi_t (incoming). Every edge contribution is
LINEAR in the GNC weight w, so we store unit-weight blocks once and
rebuild values by scaling — a scatter-add kernel on GPU.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Sequence, Tuple

import numpy as np
import torch

from .types import RelativeSEMeasurement

Tensor = torch.Tensor


def edge_unit_blocks(m: RelativeSEMeasurement) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Unit-weight blocks (B_ii, B_jj, B_ij) of one edge's Q contribution.
    B_ii = T Omega0 T^T, B_jj = Omega0, B_ij = -T Omega0, where Omega0
    uses w = 1."""
    Bii, Bjj, Bij = edge_unit_blocks_batch([m])
    return Bii[0], Bjj[0], Bij[0]


def edge_unit_blocks_batch(meas):
    """Vectorized unit-weight Q blocks for a list of edges (or a
    MeasurementArray): returns (Bii, Bjj, Bij) each (ne, dh, dh)."""
    from .measurements import as_measurement_array
    ma = as_measurement_array(meas)
    ne = len(ma)
    d = ma.d
    dh = d + 1
    R = ma.R
    t = ma.t
    kappa = ma.kappa[:, None, None]
    tau = ma.tau[:, None, None]
    TOm = np.zeros((ne, dh, dh))
    TOm[:, :d, :d] = kappa * R
    TOm[:, :d, d] = tau[:, :, 0] * t
    TOm[:, d, d] = tau[:, 0, 0]
    Bii = np.zeros((ne, dh, dh))
    Bii[:, :d, :d] = kappa * np.eye(d)[None] + tau * np.einsum(
        'ei,ej->eij', t, t)
    Bii[:, :d, d] = tau[:, :, 0] * t
    Bii[:, d, :d] = tau[:, :, 0] * t
    Bii[:, d, d] = tau[:, 0, 0]
    Bjj = np.zeros((ne, dh, dh))
    Bjj[:, :d, :d] = kappa * np.eye(d)[None]
    Bjj[:, d, d] = tau[:, 0, 0]
    return Bii, Bjj, -TOm


@dataclass
class BSRMatrix:
    """Symmetric block-CSR matrix with (d+1)x(d+1) fp64 tiles.

    row_ptr: (n+1,) int32; col_idx: (nnzb,) int32; vals: (nnzb, dh, dh).
    Also keeps a scalar torch CSR mirror for the CPU reference path.
    """

    n: int
    dh: int
    row_ptr: Tensor
    col_idx: Tensor
    vals: Tensor

    _csr_cache: Optional[Tensor] = None

    @property
    def N(self) -> int:
        return self.n * self.dh

    def invalidate(self) -> None:
        self._csr_cache = None

    def to_scalar_csr(self) -> Tensor:
        """Expand to a scalar torch sparse CSR tensor (CPU spmm path)."""
        if self._csr_cache is not None:
            return self._csr_cache
        n, dh = self.n, self.dh
        rp = self.row_ptr.cpu().numpy()
        ci = self.col_idx.cpu().numpy()
        nnzb = ci.shape[0]
        # scalar COO indices for every element of every block
        bro = np.repeat(np.arange(n, dtype=np.int64),
                        np.diff(rp).astype(np.int64))  # block-row per block
        rr = (bro[:, None, None] * dh
              + np.arange(dh)[None, :, None]).repeat(dh, axis=2)
        cc = (ci.astype(np.int64)[:, None, None] * dh
              + np.arange(dh)[None, None, :]).repeat(dh, axis=1)
        v = self.vals.cpu().numpy().reshape(nnzb, dh, dh)
        coo = torch.sparse_coo_tensor(
            np.stack([rr.reshape(-1), cc.reshape(-1)]),
            v.reshape(-1), (self.N, self.N), dtype=torch.float64)
        self._csr_cache = coo.coalesce().to_sparse_csr()
        return self._csr_cache

    def diag_slot_index(self) -> Tensor:
        """(n,) index into vals of each diagonal block (cached)."""
        if getattr(self, "_diag_slots", None) is None:
            rp = self.row_ptr.cpu().numpy().astype(np.int64)
            ci = self.col_idx.cpu().numpy().astype(np.int64)
            bro = np.repeat(np.arange(self.n, dtype=np.int64), np.diff(rp))
            slots = np.nonzero(ci == bro)[0]
            assert len(slots) == self.n, "missing diagonal block"
            self._diag_slots = torch.from_numpy(slots).to(self.vals.device)
        return self._diag_slots

    def diag_blocks(self) -> Tensor:
        """(n, dh, dh) diagonal blocks."""
        return self.vals.index_select(0, self.diag_slot_index())

    def to_scipy(self):
        """Scalar scipy CSR (host) for factorization-based preconditioners."""
        import scipy.sparse as sp
        csr = self.to_scalar_csr()
        return sp.csr_matrix(
            (csr.values().cpu().numpy(),
             csr.col_indices().cpu().numpy(),
             csr.crow_indices().cpu().numpy()),
            shape=(self.N, self.N))

    def to_dense(self) -> Tensor:
        return self.to_scalar_csr().to_dense().to(self.vals.device)

    def spmm(self, X: Tensor) -> Tensor:
        """Q @ X — HIP BSR kernel on GPU, scalar-CSR torch spmm on CPU."""
        if X.is_cuda:
            from .ops import hip_backend
            return hip_backend.bsr_spmm(self.row_ptr, self.col_idx,
                                        self.vals, self.n, self.dh, X)
        return torch.sparse.mm(self.to_scalar_csr(), X)

    def to(self, device) -> "BSRMatrix":
        return BSRMatrix(self.n, self.dh,
                         self.row_ptr.to(device), self.col_idx.to(device),
                         self.vals.to(device))


class QAssembler:
    """Builds and incrementally re-weights a BSR connection Laplacian.

    mode="full": all edges contribute all four blocks (centralized problem
    / private measurements, constructConnectionLaplacianSE).
    Shared edges (flagged) contribute only their local diagonal correction
    (per-agent Q, PGOAgent.cpp:746-776).
    """

    def __init__(self, n: int, d: int,
                 measurements,
                 shared_flags: Optional[Sequence[bool]] = None,
                 local_endpoint: Optional[Sequence[int]] = None):
        from .measurements import as_measurement_array
        self.n, self.d = n, d
        dh = d + 1
        self.dh = dh
        self.meas = as_measurement_array(measurements)
        ne = len(self.meas)
        shared = list(shared_flags) if shared_flags is not None else [False] * ne
        lep = list(local_endpoint) if local_endpoint is not None else [0] * ne

        # ---- block sparsity pattern (vectorized) ---------------------
        p1 = self.meas.p1
        p2 = self.meas.p2
        sh = np.array(shared, dtype=bool)
        lepv = np.array(lep, dtype=np.int64)
        priv = ~sh
        rows = np.concatenate([np.arange(n, dtype=np.int64),
                               p1[priv], p2[priv]])
        cols = np.concatenate([np.arange(n, dtype=np.int64),
                               p2[priv], p1[priv]])
        keys = rows * n + cols
        uk = np.unique(keys)
        u_rows = uk // n
        u_cols = uk % n
        nnzb = len(uk)
        row_ptr = np.zeros(n + 1, dtype=np.int32)
        np.add.at(row_ptr, u_rows + 1, 1)
        row_ptr = np.cumsum(row_ptr).astype(np.int32)
        col_idx = u_cols.astype(np.int32)

        def slot_of(i, j):
            return np.searchsorted(uk, i * n + j)

        # ---- per-edge unit blocks and target slots (vectorized) ------
        ne = len(self.meas)
        Bii, Bjj, Bij = (np.zeros((0, dh, dh)),) * 3
        if ne:
            Bii, Bjj, Bij = edge_unit_blocks_batch(self.meas)
        eidx = np.arange(ne, dtype=np.int64)
        sh_out = sh & (lepv == 0)
        sh_in = sh & (lepv == 1)
        slots = np.concatenate([
            slot_of(p1[priv], p1[priv]),
            slot_of(p2[priv], p2[priv]),
            slot_of(p1[priv], p2[priv]),
            slot_of(p2[priv], p1[priv]),
            slot_of(p1[sh_out], p1[sh_out]),
            slot_of(p2[sh_in], p2[sh_in]),
        ])
        blocks = np.concatenate([
            Bii[priv], Bjj[priv], Bij[priv],
            np.transpose(Bij[priv], (0, 2, 1)),
            Bii[sh_out], Bjj[sh_in],
        ]) if ne else np.zeros((0, dh, dh))
        edge_of = np.concatenate([eidx[priv]] * 4
                                 + [eidx[sh_out], eidx[sh_in]])

        self._slots = torch.from_numpy(np.ascontiguousarray(slots))
        self._blocks = torch.from_numpy(np.ascontiguousarray(blocks))
        self._edge_of = torch.from_numpy(np.ascontiguousarray(edge_of))
        self._nnzb = nnzb
        self.bsr = BSRMatrix(
            n, dh,
            torch.from_numpy(row_ptr), torch.from_numpy(col_idx),
            torch.zeros(nnzb, dh, dh, dtype=torch.float64))

    def assemble(self, weights: Optional[Tensor] = None) -> BSRMatrix:
        """(Re)compute BSR values given per-edge weights (default all 1)."""
        if weights is None:
            weights = torch.from_numpy(self.meas.weight.copy())
        w = weights[self._edge_of]
        vals = torch.zeros(self._nnzb, self.dh, self.dh, dtype=torch.float64)
        vals.index_add_(0, self._slots, self._blocks * w[:, None, None])
        self.bsr.vals = vals
        self.bsr.invalidate()
        return self.bsr


def assemble_connection_laplacian(
        measurements, n: int, d: int,
        weights: Optional[Sequence[float]] = None) -> BSRMatrix:
    """Centralized / private connection Laplacian Q as BSR (parity with
    reference constructConnectionLaplacianSE, DPGO_utils.cpp:265-271;
    weights default to each measurement's stored weight)."""
    qa = QAssembler(n, d, measurements)
    if weights is None:
        return qa.assemble()
    return qa.assemble(torch.tensor(list(weights), dtype=torch.float64))


class GAssembler:
    """Precomputed structure for the per-iteration linear term G.

    For shared edges only. Neighbor poses arrive as a packed tensor
    nbr (n_nbr_poses, dh, r) in a fixed slot order; assemble() performs
    Gt[local_block] += -(w * E0) @ nbr_slot   (outgoing)
    Gt[local_block] += -(w * E0)^T @ nbr_slot (incoming)
    with E0 = T * Omega0. Returns dense Gt (N, r).
    """

    def __init__(self, n: int, d: int,
                 shared_meas,
                 local_endpoint: Sequence[int],
                 nbr_slot: Sequence[int]):
        from .measurements import as_measurement_array
        self.n, self.d = n, d
        dh = d + 1
        ne = len(shared_meas)
        E0 = np.zeros((ne, dh, dh))
        local_pose = np.zeros(ne, dtype=np.int64)
        if ne:
            ma = as_measurement_array(shared_meas)
            kap = ma.kappa[:, None, None]
            tau = ma.tau
            TOm = np.zeros((ne, dh, dh))
            TOm[:, :d, :d] = kap * ma.R
            TOm[:, :d, d] = tau[:, None] * ma.t
            TOm[:, d, d] = tau
            lev = np.array(local_endpoint, dtype=np.int64)
            local_pose = np.where(lev == 0, ma.p1, ma.p2)
            E0 = np.where((lev == 0)[:, None, None], TOm,
                          np.transpose(TOm, (0, 2, 1)))
        self.E0 = torch.from_numpy(E0)
        self.local_pose = torch.from_numpy(local_pose)
        self.nbr_slot = torch.tensor(list(nbr_slot), dtype=torch.int64)

    def assemble(self, nbr_poses: Tensor, weights: Tensor, r: int) -> Tensor:
        """nbr_poses: (n_slots, dh, r) packed neighbor poses (Xt blocks);
        weights: (ne,) per-shared-edge GNC weights. Returns Gt (N, r).
        GPU path: one HIP scatter-add kernel (rebuilt every iteration —
        SURVEY.md 2c row 'G assembly')."""
        dh = self.d + 1
        dev = nbr_poses.device
        if dev.type != "cpu":
            from .ops import hip_backend
            if getattr(self, "_dev_cache", None) is None or \
                    self._dev_cache[0] != dev:
                self._dev_cache = (dev, self.E0.to(dev).contiguous(),
                                   self.local_pose.to(dev),
                                   self.nbr_slot.to(dev))
            _, E0, lp, slots = self._dev_cache
            Gt = torch.empty(self.n * dh, r, dtype=torch.float64, device=dev)
            hip_backend.g_assemble(Gt, E0, lp, slots,
                                   nbr_poses.contiguous(),
                                   weights.to(dev).contiguous(), dh, r)
            return Gt
        E0 = self.E0
        w = weights
        Xn = nbr_poses[self.nbr_slot]                # (ne, dh, r)
        contrib = -torch.bmm(E0 * w[:, None, None], Xn)
        Gt = torch.zeros(self.n * dh, r, dtype=torch.float64)
        Gb = Gt.view(self.n, dh, r)
        Gb.index_add_(0, self.local_pose, contrib)
        return Gt


class QuadraticProblem:
    """Cost oracle f(X) = 0.5 <Q, X^T X> + <X, G> on the lifted manifold.

    In the Xt layout: f = 0.5 * sum((Q @ Xt) * Xt) + sum(Xt * Gt);
    EucGrad_t = Q @ Xt + Gt; Hess-vec_t = Q @ Vt.
    Parity: reference QuadraticProblem.cpp:50-101.
    """

    # Above this many scalar rows the dense-inverse preconditioner is
    # replaced by block-Jacobi (memory: N^2 fp32).
    DENSE_PRECOND_MAX_N = 24576

    def __init__(self, n: int, d: int, r: int, precond: str = "auto"):
        self.n, self.d, self.r = n, d, r
        self.dh = d + 1
        self.N = self.dh * n
        self.Q: Optional[BSRMatrix] = None
        self.Gt: Optional[Tensor] = None
        self.precond_mode = precond
        self._Lpre: Optional[Tensor] = None   # block-Jacobi factors
        self._lu = None                       # CPU exact splu
        self._Minv: Optional[Tensor] = None   # GPU dense fp32 inverse
        from .manifold import LiftedSEManifold
        self.manifold = LiftedSEManifold(r, d, n)

    def set_q(self, Q: BSRMatrix, precond_reg: float = 0.1) -> None:
        """Install Q and rebuild the tCG preconditioner of Q + reg I.

        The reference factors Q + 0.1 I with Cholmod LDL^T once per setQ
        (QuadraticProblem.cpp:31-42). Our modes:
          * "exact" (CPU default): scipy splu of Q + reg I — same quality.
          * "dense" (GPU default for N <= DENSE_PRECOND_MAX_N): explicit
            fp32 inverse, applied as a GEMM (MFMA-friendly; avoids
            serialized sparse triangular solves on the GPU; exactness is
            not required of a CG preconditioner).
          * "jacobi": (d+1)-block diagonal Cholesky (scalable fallback).
        """
        self.Q = Q
        dh = self.dh
        dev = Q.vals.device
        mode = self.precond_mode
        if mode == "auto":
            if dev.type == "cpu":
                mode = "exact"
            else:
                mode = "dense" if self.N <= self.DENSE_PRECOND_MAX_N else "jacobi"
        self._active_precond = mode
        if mode == "exact":
            import scipy.sparse as sp
            import scipy.sparse.linalg as spla
            A = (Q.to_scipy() + precond_reg * sp.eye(self.N)).tocsc()
            self._lu = spla.splu(A)
        elif mode == "dense":
            A = Q.to_dense()
            A += precond_reg * torch.eye(self.N, dtype=A.dtype, device=dev)
            L = torch.linalg.cholesky(A)
            self._Minv = torch.cholesky_inverse(L).to(torch.float32).contiguous()
        else:
            diag = Q.diag_blocks() + precond_reg * torch.eye(
                dh, dtype=torch.float64, device=dev)
            self._Lpre = torch.linalg.cholesky(diag)

    def refresh_preconditioner(self, precond_reg: float = 0.1) -> None:
        """Rebuild the preconditioner after Q values changed in place
        (GNC re-weighting)."""
        self.Q.invalidate()
        mode = self._active_precond
        dh = self.dh
        dev = self.Q.vals.device
        if mode == "exact":
            import scipy.sparse as sp
            import scipy.sparse.linalg as spla
            A = (self.Q.to_scipy() + precond_reg * sp.eye(self.N)).tocsc()
            self._lu = spla.splu(A)
        elif mode == "dense":
            A = self.Q.to_dense()
            A += precond_reg * torch.eye(self.N, dtype=A.dtype, device=dev)
            L = torch.linalg.cholesky(A)
            self._Minv = torch.cholesky_inverse(L).to(
                torch.float32).contiguous()
        else:
            diag = self.Q.diag_blocks() + precond_reg * torch.eye(
                dh, dtype=torch.float64, device=dev)
            self._Lpre = torch.linalg.cholesky(diag).contiguous()

    def set_g(self, Gt: Tensor) -> None:
        self.Gt = Gt

    def _g(self) -> Tensor:
        if self.Gt is None:
            return torch.zeros(self.N, self.r, dtype=torch.float64,
                               device=self.Q.vals.device if self.Q else "cpu")
        return self.Gt

    def f(self, X: Tensor) -> float:
        QX = self.Q.spmm(X)
        return float(0.5 * (QX * X).sum() + (X * self._g()).sum())

    def euc_grad(self, X: Tensor) -> Tensor:
        return self.Q.spmm(X) + self._g()

    def hess_vec(self, V: Tensor) -> Tensor:
        return self.Q.spmm(V)

    def rie_grad(self, X: Tensor) -> Tensor:
        return self.manifold.project_tangent(X, self.euc_grad(X))

    def rie_grad_norm(self, X: Tensor) -> float:
        return float(torch.linalg.norm(self.rie_grad(X)))

    def precondition(self, X: Tensor, V: Tensor) -> Tensor:
        """Apply the preconditioner then tangent-project at X
        (reference QuadraticProblem.cpp:75-87)."""
        mode = self._active_precond
        if mode == "exact":
            import torch as _t
            Z = _t.from_numpy(self._lu.solve(V.cpu().numpy()))
            Z = Z.to(V.device)
        elif mode == "dense":
            Z = (self._Minv @ V.to(torch.float32)).to(torch.float64)
        else:
            Vb = V.view(self.n, self.dh, self.r)
            Z = torch.cholesky_solve(Vb, self._Lpre).reshape(self.N, self.r)
        return self.manifold.project_tangent(X, Z)
