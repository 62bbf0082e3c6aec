"""dpo_amd — MI355X-native distributed pose-graph optimization (DPGO).

A from-scratch AMD-native framework with the capabilities of the reference
DPGO fork (tjcunhao/dpo): rank-relaxed Riemannian block-coordinate descent
for SE(d) pose-graph synchronization across N agents, with Nesterov
acceleration, graduated non-convexity (GNC) for outlier-robust loop
closures, chordal/odometry initialization, and multi-level graph
partitioning mapping agents to GPUs.

Design (MI355X / gfx950, CDNA4):
  * fp64 core math (matches the reference's Eigen double precision).
  * Hot loop (block-sparse Hessian-vec inside truncated CG, batched Stiefel
    manifold ops) runs as hand-written HIP kernels with device-resident tCG
    control state -> zero host round-trips inside a local RTR solve.
  * One process per GPU; boundary-pose exchange + tiny consensus
    collectives over RCCL (torch.distributed "nccl" backend on ROCm).
  * CPU fallback implemented in PyTorch fp64 for every op: serves as the
    numerics reference for kernel tests and lets the full multi-agent
    pipeline run host-only (gloo backend) for CI without a GPU.
"""

__version__ = "0.1.0"

from .types import (  # noqa: F401
    RelativeSEMeasurement,
    PGOAgentParams,
    PGOAgentState,
    PGOAgentStatus,
    RobustCostType,
    RobustCostParams,
    OptAlgorithm,
)
from .agent import PGOAgent  # noqa: F401
