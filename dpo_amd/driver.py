"""In-process multi-robot RBCD driver.

Parity: reference examples/MultiRobotExample.cpp (C16): partition the
dataset, build agents with a shared lifting matrix, centralized chordal
initialization distributed via set_x, then the greedy-selection RBCD
loop — every iteration all non-selected agents iterate(False), the
selected agent pulls neighbors' public poses + statuses and
iterate(True); the next robot is the argmax of per-robot centralized
Riemannian gradient norms; convergence when the centralized ||grad_R||
drops below 0.1; per-iteration (cost, gradnorm) appended to a trace.

The multi-GPU (one process per GPU over RCCL) variant lives in
dpo_amd/dist_driver.py; this in-process driver doubles as the
single-node / single-GPU path and the algorithmic reference.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from .agent import PGOAgent
from .chordal import chordal_initialization
from .manifold import lifting_matrix
from .partition import (contiguous_partition, multilevel_partition,
                        partition_measurements)
from .io_g2o import adjacency_from_measurements
from .quadratic import QuadraticProblem, assemble_connection_laplacian
from .types import PGOAgentParams, RelativeSEMeasurement, RobustCostType

Tensor = torch.Tensor


@dataclass
class RBCDResult:
    iterations: int = 0
    converged: bool = False
    final_cost: float = 0.0
    final_gradnorm: float = 0.0
    trace: List[Tuple[float, float]] = field(default_factory=list)
    elapsed_s: float = 0.0


class MultiRobotDriver:
    """Simulates num_robots PGOAgents in one process (possibly on one GPU),
    exchanging PoseDicts by direct method call like the reference driver."""

    def __init__(self,
                 measurements: Sequence[RelativeSEMeasurement],
                 num_poses: int,
                 num_robots: int,
                 r: int = 5,
                 partition: str | Sequence[int] = "contiguous",
                 acceleration: bool = False,
                 robust: RobustCostType = RobustCostType.L2,
                 robust_params=None,
                 robust_inner_iters: int = 30,
                 device: str = "cpu",
                 verbose: bool = False,
                 selection: str = "greedy",
                 tr_max_iterations: int = 1):
        self.num_robots = num_robots
        self.verbose = verbose
        self.selection = selection
        d = measurements[0].d
        self.d, self.r, self.n = d, r, num_poses
        self.dh = d + 1
        self.device = device

        # ---- partition -------------------------------------------------
        if isinstance(partition, str):
            if partition == "contiguous":
                part = contiguous_partition(num_poses, num_robots)
            elif partition == "multilevel":
                adj = adjacency_from_measurements(measurements, num_poses)
                part = multilevel_partition(adj, num_robots)
            else:
                raise ValueError(f"unknown partition: {partition}")
        else:
            part = list(partition)
        self.part = part
        (odometry, private_lc, shared_lc, self.pose_map,
         self.pose_to_index, self.pose_counts) = partition_measurements(
            measurements, num_poses, part, num_robots)

        # ---- centralized evaluation problem ----------------------------
        self.central = QuadraticProblem(num_poses, d, r)
        Qc = assemble_connection_laplacian(measurements, num_poses, d)
        if device != "cpu":
            Qc = Qc.to(device)
        self.central.set_q(Qc)

        # ---- centralized chordal init, sliced per agent ----------------
        # (computed before agents so set_pose_graph can take it as TInit
        # and skip redundant per-agent local initializations)
        # The reference example initializes from the centralized chordal
        # relaxation regardless of the robust mode
        # (MultiRobotExample.cpp:185-202).
        T_chordal = chordal_initialization(d, num_poses, measurements)

        def _slice_T(rb):
            if T_chordal is None:
                return None
            Tr = np.zeros((d, self.pose_counts[rb] * self.dh))
            for i in range(self.pose_counts[rb]):
                g = self.pose_to_index[(rb, i)]
                Tr[:, i * self.dh:(i + 1) * self.dh] = \
                    T_chordal[:, g * self.dh:(g + 1) * self.dh]
            return Tr

        # ---- agents ----------------------------------------------------
        self.agents: List[PGOAgent] = []
        for rb in range(num_robots):
            p = PGOAgentParams(d=d, r=r, num_robots=num_robots,
                               acceleration=acceleration,
                               robust_cost_type=robust,
                               robust_opt_inner_iters=robust_inner_iters,
                               verbose=verbose, device=device,
                               tr_max_iterations=tr_max_iterations)
            if robust_params is not None:
                p.robust_cost_params = robust_params
            a = PGOAgent(rb, p)
            if rb > 0:
                a.set_lifting_matrix(self.agents[0].get_lifting_matrix())
            a.set_pose_graph(odometry[rb], private_lc[rb], shared_lc[rb],
                             T_init=_slice_T(rb))
            self.agents.append(a)

        YL = self.agents[0].get_lifting_matrix()
        X_chordal = YL @ T_chordal  # (r, (d+1) n)
        for rb in range(num_robots):
            Xr = np.zeros((r, self.pose_counts[rb] * self.dh))
            for i in range(self.pose_counts[rb]):
                g = self.pose_to_index[(rb, i)]
                Xr[:, i * self.dh:(i + 1) * self.dh] = \
                    X_chordal[:, g * self.dh:(g + 1) * self.dh]
            self.agents[rb].set_x(Xr)

        # Agent-quotient-graph coloring for the "colored" selection rule:
        # agents of one color share no edge, so their simultaneous block
        # updates compose an exact block Gauss-Seidel sweep (the
        # scalable multi-GPU schedule).
        colors = [-1] * num_robots
        for rb in range(num_robots):
            used = {colors[nb] for nb in self.agents[rb].get_neighbors()
                    if colors[nb] >= 0}
            c = 0
            while c in used:
                c += 1
            colors[rb] = c
        self._colors = colors
        self._num_colors = max(colors) + 1 if colors else 1

        self._Xopt = torch.zeros(self.dh * num_poses, r,
                                 dtype=torch.float64,
                                 device=torch.device(device))
        # block scatter index: global block row of each (robot, local) pose
        self._blk_index = [
            torch.tensor([self.pose_to_index[(rb, i)]
                          for i in range(self.pose_counts[rb])],
                         dtype=torch.int64, device=torch.device(device))
            for rb in range(num_robots)]

    # -------------------------------------------------------------------
    def _gather_global_x(self) -> Tensor:
        dh = self.dh
        Xb = self._Xopt.view(self.n, dh, self.r)
        for rb, a in enumerate(self.agents):
            Xa = a.X.view(a.n, dh, self.r)
            Xb.index_copy_(0, self._blk_index[rb], Xa)
        return self._Xopt

    def _exchange_with(self, selected: int, acceleration: bool) -> None:
        sel = self.agents[selected]
        for a in self.agents:
            if a.id == selected:
                continue
            shared = a.get_shared_pose_dict()
            if shared is None:
                continue
            sel.set_neighbor_status(a.get_status())
            sel.update_neighbor_poses(a.id, shared)
        if acceleration:
            for a in self.agents:
                if a.id == selected:
                    continue
                aux = a.get_aux_shared_pose_dict()
                if aux is None:
                    continue
                sel.set_neighbor_status(a.get_status())
                sel.update_aux_neighbor_poses(a.id, aux)

    def _sync_shared_weights(self) -> None:
        """Propagate GNC weights of shared loop closures from the owning
        agent (the lower-id endpoint, which is the one that computes
        them under the owner-computes rule, PGOAgent.cpp:1201-1244) to
        the co-owner, so both agents optimize the same objective.
        The reference does this through its ROS publish step; in-process
        we apply the owner's weights directly after each round."""
        for a in self.agents:
            if not a.publish_weights_requested:
                continue
            for src, dst, w in a.get_shared_measurement_weights():
                other = dst[0] if src[0] == a.id else src[0]
                if other > a.id:  # a owns this edge -> push to co-owner
                    self.agents[other].set_measurement_weight(src, dst, w)
            a.publish_weights_requested = False

    def run(self, max_iters: int = 1000, gradnorm_tol: float = 0.1,
            trace_file: Optional[str] = None) -> RBCDResult:
        res = RBCDResult()
        acceleration = self.agents[0].params.acceleration
        selected = 0
        t0 = time.perf_counter()
        fout = open(trace_file, "w") if trace_file else None
        try:
            for it in range(max_iters):
                if self.selection == "parallel":
                    # dpo_amd extension: all agents exchange + optimize
                    # concurrently each round (Jacobi-style; may stall on
                    # tightly coupled partitions — prefer "colored").
                    for a in self.agents:
                        self._exchange_with(a.id, acceleration)
                    for a in self.agents:
                        a.iterate(True)
                elif self.selection == "colored":
                    # dpo_amd extension: agents of the active color (a
                    # shared-edge independent set) update concurrently =
                    # exact block Gauss-Seidel; scales to one agent/GPU.
                    color = it % self._num_colors
                    active = [a for a in self.agents
                              if self._colors[a.id] == color]
                    for a in self.agents:
                        if self._colors[a.id] != color:
                            a.iterate(False)
                    for a in active:
                        self._exchange_with(a.id, acceleration)
                    for a in active:
                        a.iterate(True)
                else:
                    for a in self.agents:
                        if a.id != selected:
                            a.iterate(False)
                    self._exchange_with(selected, acceleration)
                    self.agents[selected].iterate(True)

                self._sync_shared_weights()
                X = self._gather_global_x()
                rgrad = self.central.rie_grad(X)
                gradnorm = float(torch.linalg.norm(rgrad))
                cost = 2.0 * self.central.f(X)
                res.trace.append((cost, gradnorm))
                if fout:
                    fout.write(f"{cost:.10g},{gradnorm:.10g}\n")
                if self.verbose:
                    print(f"Iter = {it} | robot = {selected} | "
                          f"cost = {cost:.5g} | gradnorm = {gradnorm:.5g}")
                res.iterations = it + 1
                if gradnorm < gradnorm_tol:
                    res.converged = True
                    break

                if self.selection == "greedy":
                    selected = self._select_next(rgrad, selected)
                elif self.selection == "round_robin":
                    selected = (selected + 1) % self.num_robots

                # anchor broadcast for rounding
                M = self.agents[0].get_shared_pose(0)
                if M is not None:
                    for a in self.agents:
                        a.set_global_anchor(M)
            res.final_cost, res.final_gradnorm = res.trace[-1]
        finally:
            if fout:
                fout.close()
        res.elapsed_s = time.perf_counter() - t0
        return res

    def _select_next(self, rgrad: Tensor, current: int) -> int:
        sel = self.agents[current]
        if not sel.get_neighbors():
            return current
        dh = self.dh
        Gb = rgrad.view(self.n, dh, self.r)
        norms = []
        for rb in range(self.num_robots):
            nb = Gb.index_select(0, self._blk_index[rb])
            norms.append(float(nb.pow(2).sum()))
        return int(np.argmax(norms))

    def final_trajectory(self) -> np.ndarray:
        """Global rounded trajectory (d, (d+1) n) using agent 0's anchor."""
        M = self.agents[0].get_shared_pose(0)
        for a in self.agents:
            a.set_global_anchor(M)
        T = np.zeros((self.d, self.dh * self.n))
        for rb, a in enumerate(self.agents):
            Ta = a.get_trajectory_in_global_frame()
            for i in range(a.n):
                g = self.pose_to_index[(rb, i)]
                T[:, g * self.dh:(g + 1) * self.dh] = \
                    Ta[:, i * self.dh:(i + 1) * self.dh]
        return T
