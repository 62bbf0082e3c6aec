"""Lifted SE(d) product manifold (St(d, r) x R^r)^n.

Replaces the reference's ROPTLIB backend (LiftedSEManifold.cpp:16-45 and
the ROPTLIB Stiefel/ProductManifold subset listed in SURVEY.md 2a) with
batched torch/HIP ops in the Xt layout (see dpo_amd/ops/cpu_ref.py).
"""
from __future__ import annotations

import numpy as np
import torch

from . import ops

Tensor = torch.Tensor


def lifting_matrix(d: int, r: int, seed: int = 1) -> np.ndarray:
    """Deterministic r x d matrix with orthonormal columns, identical for
    every agent/process that calls it with the same seed.

    The reference derives it from srand(1) + ROPTLIB StieVariable::
    RandInManifold (DPGO_utils.cpp:487-492, determinism asserted by
    tests/testUtils.cpp:12-35). Bitwise equality with ROPTLIB is not
    required — the lifted problem is equivariant under any fixed
    orthonormal lift — only cross-agent determinism is. We use a seeded
    Gaussian + QR with sign fix.
    """
    rng = np.random.Generator(np.random.PCG64(seed))
    A = rng.standard_normal((r, d))
    Qm, Rm = np.linalg.qr(A)
    # Fix signs so the factorization (hence the result) is unique.
    s = np.sign(np.diag(Rm))
    s[s == 0] = 1.0
    return Qm * s[None, :]


class LiftedSEManifold:
    """Product manifold ops bound to fixed (r, d, n) and a torch device."""

    def __init__(self, r: int, d: int, n: int):
        assert r >= d
        self.r, self.d, self.n = r, d, n
        self.N = (d + 1) * n

    def project_tangent(self, X: Tensor, V: Tensor) -> Tensor:
        return ops.backend_for(X).tangent_project(X, V, self.d)

    def project(self, M: Tensor) -> Tensor:
        """Metric projection onto the manifold (per-pose polar/SVD;
        reference LiftedSEManifold.cpp:34-45)."""
        return ops.backend_for(M).stiefel_project(M, self.d)

    def retract(self, X: Tensor, eta: Tensor) -> Tensor:
        return ops.backend_for(X).retract(X, eta, self.d)

    def random_point(self, generator: torch.Generator | None = None,
                     device="cpu") -> Tensor:
        M = torch.randn(self.N, self.r, dtype=torch.float64,
                        generator=generator, device=device)
        return self.project(M)
