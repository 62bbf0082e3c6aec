"""Riemannian trust-region (RTR / truncated CG) and RGD local solvers.

Replaces ROPTLIB's RTRNewton/RSD (the subset the reference exercises —
SURVEY.md 2a) with a host-orchestrated Steihaug-Toint tCG on the lifted
manifold. Semantics mirror reference QuadraticOptimizer.cpp:
  * optimize() records f/gradnorm before+after and asserts monotone
    descent (QuadraticOptimizer.cpp:34-59).
  * trust_region() with max_iterations == 1 implements the RBCD
    per-iteration mode: one TR step with Delta fixed, shrinking the
    radius /4 until a step is accepted, giving up after 10 rejections
    (QuadraticOptimizer.cpp:92-110).
  * Knob sets: RBCD {tol 1e-2, <=10 inner, Delta0 100} (PGOAgent.cpp:
    1134-1137); batch {tol 1e-1, 10 outer, 50 inner, Delta0 10}
    (PGOAgent.cpp:981-984).

tCG inner stop: ||r|| <= ||r0|| * min(||r0||^theta, kappa) with
theta = 1, kappa = 0.1 (standard Steihaug/ROPTLIB defaults); step
acceptance rho > 0.1; radius: shrink x0.25 when rho < 0.25, grow x2 (cap
maximum_Delta) when rho > 0.75 and the boundary was hit.

On CUDA devices the same loop runs with the HIP fused-op backend; the
device-resident tCG (zero host sync) lives in ops/hip and is selected
automatically for CUDA tensors.
"""
from __future__ import annotations

import math
import time
from dataclasses import dataclass
from typing import Optional

import torch

from .quadratic import QuadraticProblem
from .types import OptAlgorithm, OptResult

Tensor = torch.Tensor


@dataclass
class TRParams:
    tolerance: float = 1e-2
    initial_radius: float = 1e1
    max_iterations: int = 1
    max_inner_iterations: int = 50
    min_inner_iterations: int = 0
    theta: float = 1.0
    kappa: float = 0.1
    accept_rho: float = 0.1
    time_bound_s: float = 5.0


def _inner(a: Tensor, b: Tensor) -> float:
    return float((a * b).sum())


def truncated_cg(problem: QuadraticProblem, X: Tensor, grad: Tensor,
                 radius: float, p: TRParams):
    """Preconditioned Steihaug-Toint tCG. Returns (eta, Heta, status)."""
    eta = torch.zeros_like(grad)
    Heta = torch.zeros_like(grad)
    r = grad.clone()
    z = problem.precondition(X, r)
    delta = -z
    e_Pe = 0.0
    e_Pd = 0.0
    r_r = _inner(r, r)
    z_r = _inner(z, r)
    d_Pd = z_r
    norm_r0 = math.sqrt(r_r)
    bound = norm_r0 * min(norm_r0 ** p.theta, p.kappa)
    status = "max_inner"
    for j in range(p.max_inner_iterations):
        Hd = problem.manifold.project_tangent(X, problem.hess_vec(delta))
        d_Hd = _inner(delta, Hd)
        alpha = z_r / d_Hd if d_Hd != 0 else math.inf
        e_Pe_new = e_Pe + 2.0 * alpha * e_Pd + alpha * alpha * d_Pd
        if d_Hd <= 0 or e_Pe_new >= radius * radius:
            # Negative curvature or trust-region boundary: walk to boundary.
            disc = e_Pd * e_Pd + d_Pd * (radius * radius - e_Pe)
            tau = (-e_Pd + math.sqrt(max(disc, 0.0))) / d_Pd if d_Pd > 0 else 0.0
            eta = eta + tau * delta
            Heta = Heta + tau * Hd
            status = "negative_curvature" if d_Hd <= 0 else "exceeded_tr"
            break
        e_Pe = e_Pe_new
        eta = eta + alpha * delta
        Heta = Heta + alpha * Hd
        r = r + alpha * Hd
        r_r = _inner(r, r)
        if j + 1 >= p.min_inner_iterations and math.sqrt(r_r) <= bound:
            status = "converged"
            break
        z = problem.precondition(X, r)
        z_r_new = _inner(z, r)
        beta = z_r_new / z_r
        delta = -z + beta * delta
        e_Pd = beta * (e_Pd + alpha * d_Pd)
        d_Pd = z_r_new + beta * beta * d_Pd
        z_r = z_r_new
    return eta, Heta, status


def _tr_step(problem: QuadraticProblem, X: Tensor, fX: float, grad: Tensor,
             radius: float, p: TRParams):
    """One trust-region step. Returns (X_new, f_new, accepted, hit_boundary,
    tcg_status)."""
    eta, Heta, status = truncated_cg(problem, X, grad, radius, p)
    model_decrease = -_inner(grad, eta) - 0.5 * _inner(eta, Heta)
    X_prop = problem.manifold.retract(X, eta)
    f_prop = problem.f(X_prop)
    rho = (fX - f_prop) / max(model_decrease, 1e-300)
    accepted = (rho > p.accept_rho) and (f_prop <= fX)
    hit_boundary = status in ("negative_curvature", "exceeded_tr")
    if accepted:
        return X_prop, f_prop, True, hit_boundary, status
    return X, fX, False, hit_boundary, status


class QuadraticOptimizer:
    """Local solver driver (parity with reference QuadraticOptimizer)."""

    def __init__(self, problem: QuadraticProblem,
                 algorithm: OptAlgorithm = OptAlgorithm.RTR,
                 params: Optional[TRParams] = None,
                 gd_stepsize: float = 1e-3,
                 verbose: bool = False):
        self.problem = problem
        self.algorithm = algorithm
        self.params = params or TRParams()
        self.gd_stepsize = gd_stepsize
        self.verbose = verbose
        self.result = OptResult()

    def optimize(self, X0: Tensor) -> Tensor:
        t0 = time.perf_counter()
        self.result.f_init = self.problem.f(X0)
        self.result.grad_norm_init = self.problem.rie_grad_norm(X0)
        if self.algorithm == OptAlgorithm.RTR:
            Xopt = self.trust_region(X0)
        else:
            Xopt = self.gradient_descent(X0)
        self.result.elapsed_ms = (time.perf_counter() - t0) * 1e3
        self.result.f_opt = self.problem.f(Xopt)
        self.result.grad_norm_opt = self.problem.rie_grad_norm(Xopt)
        self.result.relative_change = float(
            torch.linalg.norm(Xopt - X0)) / math.sqrt(self.problem.n)
        self.result.success = True
        # Descent-monotonicity invariant (QuadraticOptimizer.cpp:56).
        assert self.result.f_opt <= self.result.f_init + 1e-9 * max(
            1.0, abs(self.result.f_init)), \
            f"non-monotone step: {self.result.f_init} -> {self.result.f_opt}"
        return Xopt

    def trust_region(self, X0: Tensor) -> Tensor:
        p = self.params
        problem = self.problem
        gn0 = problem.rie_grad_norm(X0)
        if gn0 < p.tolerance:
            return X0

        if p.max_iterations == 1:
            # RBCD mode: shrink radius /4 until one step is accepted
            # (QuadraticOptimizer.cpp:92-110).
            radius = p.initial_radius
            fX = problem.f(X0)
            grad = problem.rie_grad(X0)
            total = 0
            while True:
                Xn, fn, accepted, _, status = _tr_step(
                    problem, X0, fX, grad, radius, p)
                if accepted:
                    self.result.tcg_status = status
                    return Xn
                if total > 10:
                    if self.verbose:
                        print("Too many RTR rejections; returning initial guess.")
                    self.result.tcg_status = "rejected"
                    return X0
                radius /= 4.0
                total += 1

        # Batch mode: standard RTR outer loop with radius adaptation.
        X = X0
        fX = problem.f(X)
        radius = p.initial_radius
        max_radius = 5.0 * p.initial_radius
        t_start = time.perf_counter()
        for _ in range(p.max_iterations):
            grad = problem.rie_grad(X)
            if float(torch.linalg.norm(grad)) < p.tolerance:
                break
            eta, Heta, status = truncated_cg(problem, X, grad, radius, p)
            model_decrease = -_inner(grad, eta) - 0.5 * _inner(eta, Heta)
            X_prop = problem.manifold.retract(X, eta)
            f_prop = problem.f(X_prop)
            rho = (fX - f_prop) / max(model_decrease, 1e-300)
            if rho < 0.25:
                radius *= 0.25
            elif rho > 0.75 and status in ("negative_curvature", "exceeded_tr"):
                radius = min(2.0 * radius, max_radius)
            if rho > p.accept_rho and f_prop <= fX:
                X, fX = X_prop, f_prop
            self.result.tcg_status = status
            if time.perf_counter() - t_start > p.time_bound_s:
                break
        return X

    def gradient_descent(self, X0: Tensor) -> Tensor:
        """Single fixed-stepsize preconditioner-free RGD step
        (QuadraticOptimizer.cpp:124-149)."""
        problem = self.problem
        grad = problem.rie_grad(X0)
        return problem.manifold.retract(X0, -self.gd_stepsize * grad)

    def gradient_descent_ls(self, X0: Tensor,
                            max_iterations: int = 10) -> Tensor:
        """Backtracking line-search steepest descent (the reference's
        RSD variant, QuadraticOptimizer.cpp:151-172 — present in its
        API surface though unused by the agents). Armijo condition with
        step halving from a Riemannian-gradient-scaled initial step."""
        problem = self.problem
        X = X0
        fX = problem.f(X)
        for _ in range(max_iterations):
            grad = problem.rie_grad(X)
            gn2 = float((grad * grad).sum())
            if gn2 == 0.0:
                break
            t = 1.0 / (1.0 + gn2 ** 0.5)
            accepted = False
            for _bt in range(20):
                X_prop = problem.manifold.retract(X, -t * grad)
                f_prop = problem.f(X_prop)
                if f_prop <= fX - 1e-4 * t * gn2:
                    X, fX = X_prop, f_prop
                    accepted = True
                    break
                t *= 0.5
            if not accepted:
                break
        return X
