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
    """Workspace + orchestration for the trust-region RBCD local solve on
    one agent's problem, entirely on-device.

    Semantics mirror reference QuadraticOptimizer::trustRegion with
    Max_Iteration == 1 (QuadraticOptimizer.cpp:92-110): one Steihaug-tCG
    trust-region step, shrinking the radius /4 until accepted (<= 10
    shrinks). The Krylov path is radius-independent, so the shrink loop
    REPLAYS the stored per-iteration scalars + snapshots instead of
    re-running tCG — each rejection costs one retraction + one f eval.
    """

    def __init__(self, n: int, d: int, r: int, device,
                 max_inner: int = 10):
        assert max_inner <= MAX_TCG
        self.n, self.d, self.r = n, d, r
        self.dh = d + 1
        self.N = self.dh * n
        self.max_inner = max_inner
        dev = torch.device(device)
        total = self.N * r
        f64 = dict(dtype=torch.float64, device=dev)
        self.W = torch.empty(self.N, r, **f64)
        self.grad = torch.empty(self.N, r, **f64)
        self.eta = torch.empty(self.N, r, **f64)
        self.delta = torch.empty(self.N, r, **f64)
        self.rvec = torch.empty(self.N, r, **f64)
        self.z = torch.empty(self.N, r, **f64)
        self.Hd = torch.empty(self.N, r, **f64)
        self.step = torch.empty(self.N, r, **f64)
        self.Xprop = torch.empty(self.N, r, **f64)
        self.eta_snap = torch.empty(max_inner + 1, self.N, r, **f64)
        self.delta_snap = torch.empty(max_inner + 1, self.N, r, **f64)
        self.ctrl = torch.zeros(CTRL_SIZE, **f64)
        self.total = total

    def _precond(self, problem, V, out):
        ctrl, s = self.ctrl, _stream(V)
        if problem._Minv is not None:
            _lib.dpo_precond_dense(_p(problem._Minv), _p(V), _p(out),
                                   self.N, self.r, _p(ctrl), GUARD_RUN, s)
        else:
            _lib.dpo_precond_jacobi(_p(problem._Lpre), _p(V), _p(out),
                                    self.n, self.dh, self.r, _p(ctrl),
                                    GUARD_RUN, s)

    def solve(self, problem, X: Tensor, tol: float = 1e-2,
              Delta0: float = 100.0, max_shrink: int = 10,
              accept_rho: float = 0.1, theta: float = 1.0,
              kappa: float = 0.1, compute_final_gradnorm: bool = True):
        """Run one RBCD local solve in place on X. Returns a stats dict
        (f_init, grad_norm_init, f_opt, grad_norm_opt, status)."""
        Q = problem.Q
        G = problem.Gt
        d, r, n, N = self.d, self.r, self.n, self.N
        ctrl = self.ctrl
        s = _stream(X)
        rp, ci, vals = Q.row_ptr, Q.col_idx, Q.vals

        ctrl.zero_()
        self.eta.zero_()
        self.delta.zero_()
        # --- gradient phase ------------------------------------------
        _lib.dpo_bsr_spmm(_p(rp), _p(ci), _p(vals), n, self.dh, _p(X),
                          _p(self.W), r, None, GUARD_NONE, s)
        # grad = P_X(W + G); C_DOT1 = ||grad||^2 ; C_DOT0 = <W+G, X>
        _lib.dpo_proj_dots(_p(X), _p(self.W), _p(G), _p(self.grad), None,
                           _p(ctrl), n, d, r, C_DOT1, C_DOT0, 0,
                           GUARD_NONE, s)
        if G is not None:
            _lib.dpo_dots(_p(G), _p(X), None, _p(ctrl), C_DOT2, -1,
                          self.total, GUARD_NONE, s)
        # r0 = grad
        _lib.dpo_axpby(_p(self.grad), None, 1.0, 0.0, _p(self.rvec),
                       self.total, s)
        _lib.dpo_ctrl_init(_p(ctrl), tol, Delta0, theta, kappa, s)
        # z0 = P_X(M^-1 r0); C_DOT0 = <z0, r0>
        self._precond(problem, self.rvec, self.z)
        _lib.dpo_proj_dots(_p(X), _p(self.z), None, _p(self.z), _p(self.rvec),
                           _p(ctrl), n, d, r, C_DOT0, -1, 0, GUARD_RUN, s)
        _lib.dpo_ctrl_z0(_p(ctrl), s)
        # delta0 = -z (beta = 0)
        _lib.dpo_tcg_delta(_p(self.delta), _p(self.z), _p(ctrl), self.total, s)

        # --- tCG loop (guarded, unrolled) ----------------------------
        for _ in range(self.max_inner):
            _lib.dpo_bsr_spmm(_p(rp), _p(ci), _p(vals), n, self.dh,
                              _p(self.delta), _p(self.Hd), r, _p(ctrl),
                              GUARD_RUN, s)
            _lib.dpo_proj_dots(_p(X), _p(self.Hd), None, _p(self.Hd),
                               _p(self.delta), _p(ctrl), n, d, r,
                               C_DOT0, -1, 0, GUARD_RUN, s)
            _lib.dpo_ctrl_alpha(_p(ctrl), s)
            _lib.dpo_tcg_update(_p(self.eta), _p(self.rvec), _p(self.delta),
                                _p(self.Hd), _p(self.eta_snap),
                                _p(self.delta_snap), _p(ctrl), self.total, s)
            _lib.dpo_ctrl_rr(_p(ctrl), s)
            self._precond(problem, self.rvec, self.z)
            _lib.dpo_proj_dots(_p(X), _p(self.z), None, _p(self.z),
                               _p(self.rvec), _p(ctrl), n, d, r,
                               C_DOT0, -1, 0, GUARD_RUN, s)
            _lib.dpo_ctrl_beta(_p(ctrl), s)
            _lib.dpo_tcg_delta(_p(self.delta), _p(self.z), _p(ctrl),
                               self.total, s)
        _lib.dpo_ctrl_tcg_end(_p(ctrl), s)

        # --- candidate / shrink loop ---------------------------------
        status = ST_GIVE_UP
        for attempt in range(max_shrink + 1):
            _lib.dpo_ctrl_candidate(_p(ctrl), s)
            _lib.dpo_form_step(_p(self.step), _p(self.eta), _p(self.eta_snap),
                               _p(self.delta_snap), _p(ctrl), self.total, s)
            _lib.dpo_polar_affine(_p(X), _p(self.step), None, 1.0, 1.0, 0.0,
                                  _p(self.Xprop), n, d, r, _p(ctrl),
                                  GUARD_STOP, s)
            _lib.dpo_bsr_spmm(_p(rp), _p(ci), _p(vals), n, self.dh,
                              _p(self.Xprop), _p(self.W), r, _p(ctrl),
                              GUARD_STOP, s)
            _lib.dpo_dots(_p(self.Xprop), _p(self.W), _p(G), _p(ctrl),
                          C_DOT0, C_DOT2, self.total, GUARD_STOP, s)
            _lib.dpo_ctrl_accept(_p(ctrl), accept_rho, s)
            st = int(ctrl[C_STATUS].item())  # host sync
            if st == ST_ACCEPTED:
                X.copy_(self.Xprop)
                status = st
                break
            if st == ST_NO_UPDATE:
                status = st
                break
            _lib.dpo_ctrl_shrink(_p(ctrl), s)
        # --- stats ----------------------------------------------------
        stats_t = self.ctrl.cpu()
        stats = {
            "status": status,
            "f_init": float(stats_t[C_FX]),
            "grad_norm_init": float(stats_t[C_GN0SQ]) ** 0.5,
            "f_opt": float(stats_t[C_FPROP]) if status == ST_ACCEPTED
            else float(stats_t[C_FX]),
            "rho": float(stats_t[C_RHO]),
        }
        if compute_final_gradnorm and status == ST_ACCEPTED:
            ctrl[C_DOT0] = 0.0
            ctrl[C_DOT1] = 0.0
            _lib.dpo_bsr_spmm(_p(rp), _p(ci), _p(vals), n, self.dh, _p(X),
                              _p(self.W), r, None, GUARD_NONE, s)
            _lib.dpo_proj_dots(_p(X), _p(self.W), _p(G), _p(self.grad),
                               None, _p(ctrl), n, d, r, C_DOT1, -1, 0,
                               GUARD_NONE, s)
            stats["grad_norm_opt"] = float(ctrl[C_DOT1].item()) ** 0.5
        elif status in (ST_NO_UPDATE, ST_GIVE_UP):
            stats["grad_norm_opt"] = stats["grad_norm_init"]
        return stats
