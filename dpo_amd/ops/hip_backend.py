"""ctypes bindings for the dpo_amd HIP extension + GPU op implementations.

Mirrors the cpu_ref op interface for CUDA (ROCm) tensors, and provides
DeviceSolver — the device-resident RBCD trust-region local solve with
on-GPU tCG control (one host sync per solve in the common case).
"""
from __future__ import annotations

import ctypes
import os
from typing import Optional

import torch

Tensor = torch.Tensor

_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                         "hip", "libdpo_hip_ops.so")

if not os.path.exists(_LIB_PATH):
    raise ImportError(f"HIP extension not built: {_LIB_PATH}")

_lib = ctypes.CDLL(_LIB_PATH)

_c = ctypes.c_void_p
_i = ctypes.c_int
_l = ctypes.c_long
_d = ctypes.c_double

_lib.dpo_bsr_spmm.argtypes = [_c, _c, _c, _i, _i, _c, _c, _i, _c, _i, _c]
_lib.dpo_bsr_spmm_mfma.argtypes = [_c, _c, _c, _c, _c, _c, _i, _i, _i, _c]
_lib.dpo_proj_dots.argtypes = [_c, _c, _c, _c, _c, _c, _i, _i, _i, _i, _i,
                               _i, _i, _c]
_lib.dpo_polar_affine.argtypes = [_c, _c, _c, _d, _d, _d, _c, _i, _i, _i,
                                  _c, _i, _c]
_lib.dpo_precond_dense.argtypes = [_c, _c, _c, _i, _i, _c, _i, _c]
_lib.dpo_precond_jacobi.argtypes = [_c, _c, _c, _i, _i, _i, _c, _i, _c]
_lib.dpo_tcg_update.argtypes = [_c, _c, _c, _c, _c, _c, _c, _l, _c]
_lib.dpo_tcg_delta.argtypes = [_c, _c, _c, _l, _c]
_lib.dpo_form_step.argtypes = [_c, _c, _c, _c, _c, _l, _c]
_lib.dpo_axpby.argtypes = [_c, _c, _d, _d, _c, _l, _c]
_lib.dpo_dots.argtypes = [_c, _c, _c, _c, _i, _i, _l, _i, _c]
for name in ("dpo_ctrl_init",):
    _lib.dpo_ctrl_init.argtypes = [_c, _d, _d, _d, _d, _c]
for name in ("dpo_ctrl_z0", "dpo_ctrl_alpha", "dpo_ctrl_rr", "dpo_ctrl_beta",
             "dpo_ctrl_tcg_end", "dpo_ctrl_candidate", "dpo_ctrl_shrink"):
    getattr(_lib, name).argtypes = [_c, _c]
_lib.dpo_ctrl_accept.argtypes = [_c, _d, _c]
_lib.dpo_q_assemble.argtypes = [_c, _c, _c, _c, _c, _i, _i, _l, _c]
_lib.dpo_g_assemble.argtypes = [_c, _c, _c, _c, _c, _c, _i, _i, _i, _l, _c]
_lib.dpo_ctrl_size.restype = _i
_lib.dpo_ctx_create.restype = _c
_lib.dpo_ctx_create.argtypes = [_i, _i, _i, _i]
_lib.dpo_ctx_destroy.argtypes = [_c]
_lib.dpo_ctx_set_problem.argtypes = [_c, _c, _c, _c, _c, _c, _c]
_lib.dpo_rbcd_solve.restype = _i
_lib.dpo_rbcd_solve.argtypes = [_c, _c, _d, _d, _i, _d, _i,
                                ctypes.POINTER(ctypes.c_double), _c]
_lib.dpo_eval_terms.argtypes = [_c, _c, _c, _c]
_lib.dpo_ctx_set_gdata.argtypes = [_c, _c, _c, _c, _c, _i]
_lib.dpo_round_solve.restype = _i
_lib.dpo_round_solve.argtypes = [_c, _c, _c, _d, _d, _i, _d,
                                 ctypes.POINTER(ctypes.c_double), _c]
_lib.dpo_round_eval.argtypes = [_c, _c, _c, _c, _c]
_lib.dpo_round_eval_raw.argtypes = [_c, _c, _c, _c, _c]
_lib.dpo_round_solve_async.argtypes = [_c, _c, _c, _d, _d, _d, _c]
_lib.dpo_round_solve_finish.restype = _i
_lib.dpo_round_solve_finish.argtypes = [
    _c, _i, ctypes.POINTER(ctypes.c_double), _c]
_lib.dpo_round_eval_async.argtypes = [_c, _c, _c, _c, _c]
_lib.dpo_eval_join.argtypes = [_c, _c]
_lib.dpo_gnc_weights.argtypes = [_c, _c, _c, _c, _c, _c, _c, _c, _c, _c,
                                 _c, _c, _c, _i, _i, _i, _d, _d, _c]
_lib.dpo_group_create.restype = _c
_lib.dpo_group_create.argtypes = [
    ctypes.POINTER(ctypes.c_void_p), _i, ctypes.POINTER(ctypes.c_void_p),
    ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_int)]
_lib.dpo_group_destroy.argtypes = [_c]
_lib.dpo_group_solve_start.argtypes = [
    _c, ctypes.POINTER(ctypes.c_int), _i, _d, _d, _d, _c]
_lib.dpo_group_solve_finish.argtypes = [
    _c, ctypes.POINTER(ctypes.c_int), _i, _i, _c]
_lib.dpo_group_eval.argtypes = [_c, _c, _l, _c]

CTRL_SIZE = _lib.dpo_ctrl_size()

# ctrl slots mirrored from dpo_ops.hip
C_STATUS, C_FX, C_GN0SQ = 0, 1, 2
C_RADIUS, C_FPROP, C_DM = 12, 13, 14
C_GN1SQ, C_RHO = 20, 21
C_DOT0, C_DOT1, C_DOT2 = 24, 25, 26
ST_RUN, ST_TCG_STOP, ST_ACCEPTED, ST_GIVE_UP, ST_NO_UPDATE = range(5)
MAX_TCG = 16
GUARD_NONE, GUARD_RUN, GUARD_STOP = -1, ST_RUN, ST_TCG_STOP


def _p(t: Optional[Tensor]):
    return ctypes.c_void_p(0 if t is None else t.data_ptr())


def _stream(t: Tensor):
    return ctypes.c_void_p(torch.cuda.current_stream(t.device).cuda_stream)


def _chk(t: Tensor, dtype=torch.float64):
    assert t.is_cuda and t.dtype == dtype and t.is_contiguous()


# ---------------------------------------------------------------------
# op interface (mirrors cpu_ref)
# ---------------------------------------------------------------------
def tangent_project(X: Tensor, V: Tensor, d: int) -> Tensor:
    _chk(X); _chk(V)
    n = X.shape[0] // (d + 1)
    r = X.shape[1]
    out = torch.empty_like(V)
    _lib.dpo_proj_dots(_p(X), _p(V), None, _p(out), None, None,
                       n, d, r, -1, -1, 0, GUARD_NONE, _stream(X))
    return out


def stiefel_project(M: Tensor, d: int) -> Tensor:
    _chk(M)
    n = M.shape[0] // (d + 1)
    r = M.shape[1]
    out = torch.empty_like(M)
    _lib.dpo_polar_affine(_p(M), None, None, 1.0, 0.0, 0.0, _p(out),
                          n, d, r, None, GUARD_NONE, _stream(M))
    return out


def retract(X: Tensor, eta: Tensor, d: int) -> Tensor:
    _chk(X); _chk(eta)
    n = X.shape[0] // (d + 1)
    r = X.shape[1]
    out = torch.empty_like(X)
    _lib.dpo_polar_affine(_p(X), _p(eta), None, 1.0, 1.0, 0.0, _p(out),
                          n, d, r, None, GUARD_NONE, _stream(X))
    return out


def polar_affine(A: Tensor, B: Optional[Tensor], C: Optional[Tensor],
                 ca: float, cb: float, cc: float, d: int) -> Tensor:
    _chk(A)
    n = A.shape[0] // (d + 1)
    r = A.shape[1]
    out = torch.empty_like(A)
    _lib.dpo_polar_affine(_p(A), _p(B), _p(C), ca, cb, cc, _p(out),
                          n, d, r, None, GUARD_NONE, _stream(A))
    return out


def bsr_spmm(row_ptr: Tensor, col_idx: Tensor, vals: Tensor, n: int,
             dh: int, X: Tensor, out: Optional[Tensor] = None,
             ctrl: Optional[Tensor] = None, guard: int = GUARD_NONE) -> Tensor:
    _chk(vals); _chk(X)
    r = X.shape[1]
    if out is None:
        out = torch.empty_like(X)
    _lib.dpo_bsr_spmm(_p(row_ptr), _p(col_idx), _p(vals), n, dh,
                      _p(X), _p(out), r, _p(ctrl), guard, _stream(X))
    return out


def build_spmm_mfma_groups(row_ptr, col_idx, n: int):
    """Host-side grouped-ELL structure for the fp64-MFMA SpMM A/B
    kernel (d = 3): 4-pose row groups with per-group column unions.
    Returns (grp_ptr, grp_cols, grp_blk) numpy int32 arrays; grp_blk has
    4 entries per (group, union-column) = the Q block index of each pose
    in the group for that column, or -1."""
    import numpy as np
    rp = row_ptr.cpu().numpy() if isinstance(row_ptr, Tensor) else row_ptr
    ci = col_idx.cpu().numpy() if isinstance(col_idx, Tensor) else col_idx
    ngroups = (n + 3) // 4
    grp_ptr = [0]
    grp_cols = []
    grp_blk = []
    for g in range(ngroups):
        rows = range(g * 4, min(g * 4 + 4, n))
        union = {}
        for pi, i in enumerate(rows):
            for p in range(rp[i], rp[i + 1]):
                union.setdefault(int(ci[p]), [-1, -1, -1, -1])[pi] = p
        for j in sorted(union):
            grp_cols.append(j)
            grp_blk.extend(union[j])
        grp_ptr.append(len(grp_cols))
    return (np.asarray(grp_ptr, dtype=np.int32),
            np.asarray(grp_cols, dtype=np.int32),
            np.asarray(grp_blk, dtype=np.int32))


def bsr_spmm_mfma(grp_ptr: Tensor, grp_cols: Tensor, grp_blk: Tensor,
                  vals: Tensor, X: Tensor,
                  out: Optional[Tensor] = None) -> Tensor:
    """fp64-MFMA SpMM (d=3 / dh=4 only); A/B study kernel."""
    _chk(vals); _chk(X)
    r = X.shape[1]
    n = X.shape[0] // 4
    if out is None:
        out = torch.empty_like(X)
    _lib.dpo_bsr_spmm_mfma(_p(grp_ptr), _p(grp_cols), _p(grp_blk),
                           _p(vals), _p(X), _p(out),
                           int(grp_ptr.numel() - 1), n, r, _stream(X))
    return out


def precond_dense(Minv: Tensor, V: Tensor,
                  out: Optional[Tensor] = None,
                  ctrl: Optional[Tensor] = None,
                  guard: int = GUARD_NONE) -> Tensor:
    _chk(Minv, torch.float32); _chk(V)
    N, r = V.shape
    if out is None:
        out = torch.empty_like(V)
    _lib.dpo_precond_dense(_p(Minv), _p(V), _p(out), N, r, _p(ctrl), guard,
                           _stream(V))
    return out


def precond_jacobi(L: Tensor, V: Tensor, dh: int,
                   out: Optional[Tensor] = None,
                   ctrl: Optional[Tensor] = None,
                   guard: int = GUARD_NONE) -> Tensor:
    _chk(L); _chk(V)
    N, r = V.shape
    n = N // dh
    if out is None:
        out = torch.empty_like(V)
    _lib.dpo_precond_jacobi(_p(L), _p(V), _p(out), n, dh, r, _p(ctrl),
                            guard, _stream(V))
    return out


def g_assemble(Gt: Tensor, E0: Tensor, local_pose: Tensor, nbr_slot: Tensor,
               nbr: Tensor, w: Tensor, dh: int, r: int) -> Tensor:
    _chk(Gt)
    ne = E0.shape[0]
    N = Gt.shape[0]
    _lib.dpo_g_assemble(_p(Gt), _p(E0), _p(local_pose), _p(nbr_slot),
                        _p(nbr), _p(w), ne, dh, r, N, _stream(Gt))
    return Gt


def q_assemble(vals: Tensor, blocks: Tensor, slots: Tensor, edge_of: Tensor,
               w: Tensor, dh: int) -> Tensor:
    _chk(vals)
    ncontrib = blocks.shape[0]
    nnzb = vals.shape[0]
    _lib.dpo_q_assemble(_p(vals), _p(blocks), _p(slots), _p(edge_of), _p(w),
                        ncontrib, dh * dh, nnzb, _stream(vals))
    return vals


# ---------------------------------------------------------------------
# Device-resident RBCD local solve
# ---------------------------------------------------------------------
class DeviceSolver:
    """Native (C++) RBCD trust-region local solve with device-resident
    tCG control. Semantics mirror reference QuadraticOptimizer::
    trustRegion with Max_Iteration == 1 (QuadraticOptimizer.cpp:92-110):
    one Steihaug-tCG TR step, radius shrunk /4 until accepted (<= 10
    shrinks). The Krylov path is radius-independent, so rejections replay
    stored snapshots instead of re-running tCG. One host sync per solve
    in the accepted-first-try case; ~1 enqueue call from Python."""

    def __init__(self, n: int, d: int, r: int, device, max_inner: int = 10):
        assert max_inner <= MAX_TCG
        self.n, self.d, self.r = n, d, r
        self.dh = d + 1
        self.N = self.dh * n
        self.max_inner = max_inner
        self.device = torch.device(device)
        with torch.cuda.device(self.device):
            self.handle = _lib.dpo_ctx_create(n, d, r, max_inner)
        self._stats = (ctypes.c_double * 8)()
        self._eval_out = torch.zeros(3, dtype=torch.float64,
                                     device=self.device)
        self._refs = None  # keep problem tensors alive across the call

    def __del__(self):
        try:
            if getattr(self, "handle", None):
                _lib.dpo_ctx_destroy(self.handle)
        except Exception:
            pass

    def _bind(self, problem):
        Q = problem.Q
        G = problem.Gt
        Minv = getattr(problem, "_Minv", None)
        Ljac = getattr(problem, "_Lpre", None)
        if Minv is None and Ljac is not None:
            Ljac = Ljac.contiguous()
        self._refs = (Q.row_ptr, Q.col_idx, Q.vals, G, Minv, Ljac)
        _lib.dpo_ctx_set_problem(self.handle, _p(Q.row_ptr), _p(Q.col_idx),
                                 _p(Q.vals), _p(G), _p(Minv), _p(Ljac))

    def solve(self, problem, X: Tensor, tol: float = 1e-2,
              Delta0: float = 100.0, max_shrink: int = 10,
              accept_rho: float = 0.1, compute_final_gradnorm: bool = True):
        """One RBCD local solve in place on X. Returns a stats dict."""
        self._bind(problem)
        status = _lib.dpo_rbcd_solve(
            self.handle, _p(X), tol, Delta0, max_shrink, accept_rho,
            1 if compute_final_gradnorm else 0, self._stats, _stream(X))
        st = self._stats
        return {
            "status": int(st[0]),
            "f_init": st[1],
            "grad_norm_init": st[2],
            "f_opt": st[3],
            "grad_norm_opt": st[4],
            "rho": st[5],
            "shrinks": int(st[6]),
        }

    def eval_terms(self, problem, X: Tensor) -> Tensor:
        """Device 3-vector [f(X), 0.5<X,G>, ||rgrad||^2]; no host sync."""
        self._bind(problem)
        _lib.dpo_eval_terms(self.handle, _p(X), _p(self._eval_out),
                            _stream(X))
        return self._eval_out

    # --- fused per-round entry points (G assembled in C++) -----------
    def set_gdata(self, E0: Tensor, local_pose: Tensor, nbr_slot: Tensor,
                  w: Tensor) -> None:
        self._grefs = (E0, local_pose, nbr_slot, w)
        _lib.dpo_ctx_set_gdata(self.handle, _p(E0), _p(local_pose),
                               _p(nbr_slot), _p(w), E0.shape[0])

    def bind_problem_static(self, problem) -> None:
        """Bind the Q/preconditioner pointers once (re-call after a Q
        rebuild). The G term is produced in-C++ by the round calls."""
        Q = problem.Q
        Minv = getattr(problem, "_Minv", None)
        Ljac = getattr(problem, "_Lpre", None)
        if Minv is None and Ljac is not None:
            Ljac = Ljac.contiguous()
        self._refs = (Q.row_ptr, Q.col_idx, Q.vals, None, Minv, Ljac)
        _lib.dpo_ctx_set_problem(self.handle, _p(Q.row_ptr), _p(Q.col_idx),
                                 _p(Q.vals), None, _p(Minv), _p(Ljac))

    def round_solve(self, X: Tensor, nbr: Tensor, tol: float = 1e-2,
                    Delta0: float = 100.0) -> int:
        return _lib.dpo_round_solve(self.handle, _p(X), _p(nbr), tol,
                                    Delta0, 10, 0.1, self._stats,
                                    _stream(X))

    def round_eval_raw(self, X: Tensor, nbr: Tensor, out: Tensor) -> None:
        _lib.dpo_round_eval_raw(self.handle, _p(X), _p(nbr), _p(out),
                                _stream(X))

    def round_eval(self, X: Tensor, nbr: Tensor,
                   out: Optional[Tensor] = None) -> Tensor:
        dst = self._eval_out if out is None else out
        _lib.dpo_round_eval(self.handle, _p(X), _p(nbr), _p(dst),
                            _stream(X))
        return dst

    # --- async multi-stream variants (overlap concurrent agents) -----
    def round_solve_async(self, X: Tensor, nbr: Tensor, tol: float = 1e-2,
                          Delta0: float = 100.0) -> None:
        _lib.dpo_round_solve_async(self.handle, _p(X), _p(nbr), tol,
                                   Delta0, 0.1, _stream(X))

    def round_solve_finish(self, X: Tensor) -> int:
        return _lib.dpo_round_solve_finish(self.handle, 10, self._stats,
                                           _stream(X))

    def round_eval_async(self, X: Tensor, nbr: Tensor,
                         out: Optional[Tensor] = None) -> Tensor:
        dst = self._eval_out if out is None else out
        _lib.dpo_round_eval_async(self.handle, _p(X), _p(nbr), _p(dst),
                                  _stream(X))
        return dst

    def eval_join(self, X: Tensor) -> Tensor:
        _lib.dpo_eval_join(self.handle, _stream(X))
        return self._eval_out


class DeviceGroup:
    """Multi-agent round fan-out: one C call per phase instead of a
    Python loop over agents (solve fan-out for the active set, eval
    fan-out + single-gather-kernel for every agent). Pointers (X,
    neighbor buffer, eval scratch) are fixed at construction; the
    per-ctx hipGraph caches stay hot."""

    def __init__(self, solvers, Xs, nbrs, rows):
        n = len(solvers)
        assert n == len(Xs) == len(nbrs) == len(rows)
        self._keep = (solvers, Xs, nbrs)
        handles = (ctypes.c_void_p * n)(
            *[ctypes.c_void_p(s.handle) for s in solvers])
        xp = (ctypes.c_void_p * n)(*[ctypes.c_void_p(x.data_ptr())
                                     for x in Xs])
        np_ = (ctypes.c_void_p * n)(*[ctypes.c_void_p(b.data_ptr())
                                      for b in nbrs])
        rw = (ctypes.c_int * n)(*rows)
        self.n = n
        self.handle = _lib.dpo_group_create(handles, n, xp, np_, rw)
        assert self.handle, "dpo_group_create failed"
        self._js_of = Xs[0]

    def __del__(self):
        try:
            if getattr(self, "handle", None):
                _lib.dpo_group_destroy(self.handle)
        except Exception:
            pass

    def ids(self, idx_list):
        return (ctypes.c_int * len(idx_list))(*idx_list)

    def solve_start(self, ids, tol: float = 1e-2, Delta0: float = 100.0,
                    accept_rho: float = 0.1) -> None:
        _lib.dpo_group_solve_start(self.handle, ids, len(ids), tol,
                                   Delta0, accept_rho,
                                   _stream(self._js_of))

    def solve_finish(self, ids, max_shrink: int = 10) -> None:
        _lib.dpo_group_solve_finish(self.handle, ids, len(ids),
                                    max_shrink, _stream(self._js_of))

    def eval_all(self, out: Tensor) -> None:
        """Evaluate every agent; out is a (num_robots, 3) fp64 device
        matrix — only this group's rows are written."""
        _lib.dpo_group_eval(self.handle, _p(out), out.stride(0),
                            _stream(self._js_of))


def gnc_weights(X: Tensor, nbr: Tensor, g: dict, weights: Tensor,
                d: int, r: int, mu: float, barc_sq: float) -> None:
    """Launch the GNC-TLS weight-update kernel over g['ne'] edges."""
    _lib.dpo_gnc_weights(
        _p(X), _p(nbr), _p(g["e1_idx"]), _p(g["e1_nbr"]), _p(g["e2_idx"]),
        _p(g["e2_nbr"]), _p(g["R"]), _p(g["t"]), _p(g["kappa"]),
        _p(g["tau"]), _p(g["upd"]), _p(g["widx"]), _p(weights),
        g["ne"], d, r, mu, barc_sq, _stream(X))
