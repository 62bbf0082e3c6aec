"""Robust cost library: weight functions + GNC schedule.

Parity: reference src/DPGO_robust.cpp:23-103 (weight formulas incl. the
GNC-TLS weights of eq. (14) of Yang et al., "Graduated Non-Convexity for
Robust Spatial Perception") and include/DPGO/DPGO_robust.h:107-114
(chi-squared error threshold helper).
"""
from __future__ import annotations

import math

from scipy.stats import chi2

from .types import RobustCostParams, RobustCostType


def chi2inv(quantile: float, dof: int) -> float:
    """Inverse chi-squared CDF (reference DPGO_utils.cpp:502-505)."""
    return float(chi2.ppf(quantile, dof))


class RobustCost:
    """Weight function w(residual) for a menu of robust costs, plus the GNC
    continuation schedule mu <- mu_step * mu (reference DPGO_robust.cpp)."""

    def __init__(self, cost_type: RobustCostType,
                 params: RobustCostParams | None = None):
        self.cost_type = cost_type
        self.params = params or RobustCostParams()
        self.mu = 0.0
        self._gnc_iteration = 0
        self.reset()

    def reset(self) -> None:
        if self.cost_type == RobustCostType.GNC_TLS:
            self.mu = self.params.gnc_init_mu
            self._gnc_iteration = 0

    def weight(self, r: float) -> float:
        p = self.params
        t = self.cost_type
        if t == RobustCostType.L2:
            return 1.0
        if t == RobustCostType.L1:
            return 1.0 / r
        if t == RobustCostType.Huber:
            return 1.0 if r < p.huber_threshold else p.huber_threshold / r
        if t == RobustCostType.TLS:
            return 1.0 if r < p.tls_threshold else 0.0
        if t == RobustCostType.GM:
            a = 1.0 + r * r
            return 1.0 / (a * a)
        if t == RobustCostType.GNC_TLS:
            # GNC paper eq. (14): w = 0 above the upper bound, 1 below the
            # lower bound, sqrt(barc^2 mu (mu+1) / r^2) - mu in between.
            r_sq = r * r
            barc_sq = p.gnc_barc * p.gnc_barc
            upper = (self.mu + 1.0) / self.mu * barc_sq
            lower = self.mu / (self.mu + 1.0) * barc_sq
            if r_sq >= upper:
                return 0.0
            if r_sq <= lower:
                return 1.0
            return math.sqrt(barc_sq * self.mu * (self.mu + 1.0) / r_sq) - self.mu
        raise NotImplementedError(f"weight for {t} not implemented")

    def update(self) -> None:
        """Advance the GNC continuation (no-op for non-GNC costs)."""
        if self.cost_type != RobustCostType.GNC_TLS:
            return
        self._gnc_iteration += 1
        if self._gnc_iteration > self.params.gnc_max_iters:
            return
        self.mu = self.params.gnc_mu_step * self.mu

    @staticmethod
    def error_threshold_at_quantile(quantile: float, dimension: int) -> float:
        """Chi-squared-quantile error threshold for 3D measurements
        (reference DPGO_robust.h:107-114; 6 dof for SE(3))."""
        assert dimension == 3
        assert quantile > 0
        if quantile < 1:
            return math.sqrt(chi2inv(quantile, 6))
        return 1e5
