"""PGOAgent — the per-robot (per-GPU) runtime.

Parity target: reference src/PGOAgent.cpp / include/DPGO/PGOAgent.h (C14
in SURVEY.md). State machine WAIT_FOR_DATA -> WAIT_FOR_INITIALIZATION ->
INITIALIZED; synchronous iterate(); Nesterov-accelerated variant
(PGOAgent.cpp:1054-1091) with periodic restart (1033-1052); GNC loop-
closure re-weighting (1181-1245); robust two-stage initialization in the
global frame (250-432); trajectory rounding via the global anchor
(500-519); asynchronous optimization loop (861-916).

Internal state X lives as a torch fp64 tensor in the Xt layout (see
ops/cpu_ref.py) on the configured device; public APIs exchange
numpy r x (d+1) blocks exactly like the reference's PoseDicts.
"""
from __future__ import annotations

import math
import threading
import time
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from .averaging import (compute_measurement_error,
                        robust_single_rotation_averaging,
                        single_translation_averaging)
from .chordal import chordal_initialization, odometry_initialization
from .liegroups import angular_to_chordal_so3, check_rotation_matrix
from .logger import PGOLogger
from .manifold import LiftedSEManifold, lifting_matrix
from .quadratic import GAssembler, QAssembler, QuadraticProblem
from .robust import RobustCost
from .solver import QuadraticOptimizer, TRParams
from .types import (OptAlgorithm, PGOAgentParams, PGOAgentState,
                    PGOAgentStatus, PoseID, RelativeSEMeasurement,
                    RobustCostType)

Tensor = torch.Tensor


def _T_to_Xt(T: np.ndarray, r: int, YLift: np.ndarray) -> np.ndarray:
    """Lift a d x (d+1)n trajectory to the Xt layout (N, r):
    X = YLift @ T  (reference PGOAgent.cpp:183), transposed."""
    X = YLift @ T  # (r, (d+1) n)
    return np.ascontiguousarray(X.T)


class PGOAgent:
    def __init__(self, agent_id: int, params: PGOAgentParams):
        self.id = agent_id
        self.params = params
        self.d = params.d
        self.r = params.r
        self.dh = params.d + 1
        self.n = 1
        self.device = torch.device(params.device)
        self.state = PGOAgentState.WAIT_FOR_DATA
        self.status = PGOAgentStatus(agent_id, self.state, 0, 0, False, 0.0)
        self.robust_cost = RobustCost(params.robust_cost_type,
                                      params.robust_cost_params)
        self.logger = PGOLogger(params.log_directory) if params.log_data else None

        self.instance_number = 0
        self.iteration_number = 0
        self.num_poses_received = 0

        # Measurements
        self.odometry: List[RelativeSEMeasurement] = []
        self.private_lc: List[RelativeSEMeasurement] = []
        self.shared_lc: List[RelativeSEMeasurement] = []

        # Public-pose bookkeeping
        self.local_shared_pose_ids: set[PoseID] = set()
        self.neighbor_shared_pose_ids: set[PoseID] = set()
        self.neighbor_robot_ids: set[int] = set()

        # Neighbor caches (PoseID -> np (r, d+1))
        self.neighbor_pose_dict: Dict[PoseID, np.ndarray] = {}
        self.neighbor_aux_pose_dict: Dict[PoseID, np.ndarray] = {}

        # Optimization state (Xt layout torch tensors)
        self.X: Optional[Tensor] = None
        self.XPrev: Optional[Tensor] = None
        self.XInit: Optional[Tensor] = None
        self.Y: Optional[Tensor] = None     # Nesterov aux
        self.V: Optional[Tensor] = None
        self.gamma = 0.0
        self.alpha = 0.0

        self.T_local_init: Optional[np.ndarray] = None
        self.global_anchor: Optional[np.ndarray] = None  # (r, d+1)

        self.problem: Optional[QuadraticProblem] = None
        self._q_assembler: Optional[QAssembler] = None
        self._g_assembler: Optional[GAssembler] = None
        self._nbr_slot_order: List[PoseID] = []
        self._manifold: Optional[LiftedSEManifold] = None

        self.YLift: Optional[np.ndarray] = None
        if agent_id == 0:
            self.YLift = lifting_matrix(self.d, self.r)

        self.team_status: Dict[int, PGOAgentStatus] = {
            rid: PGOAgentStatus(rid) for rid in range(params.num_robots)}

        # Async loop state
        self._opt_thread: Optional[threading.Thread] = None
        self._end_loop = False
        self._rate = 1.0
        self._lock = threading.RLock()

        self.publish_public_poses_requested = False
        self.publish_weights_requested = False

    # ------------------------------------------------------------------
    # setup
    # ------------------------------------------------------------------
    def set_lifting_matrix(self, M: np.ndarray) -> None:
        assert M.shape == (self.r, self.d)
        self.YLift = M.copy()

    def get_lifting_matrix(self) -> Optional[np.ndarray]:
        return None if self.YLift is None else self.YLift.copy()

    def set_pose_graph(self,
                       odometry: Sequence[RelativeSEMeasurement],
                       private_loop_closures: Sequence[RelativeSEMeasurement],
                       shared_loop_closures: Sequence[RelativeSEMeasurement],
                       T_init: Optional[np.ndarray] = None) -> None:
        assert self.state == PGOAgentState.WAIT_FOR_DATA
        if (len(odometry) == 0 and len(private_loop_closures) == 0
                and len(shared_loop_closures) == 0):
            return
        for m in odometry:
            assert m.r1 == self.id and m.r2 == self.id and m.p1 + 1 == m.p2
            self.n = max(self.n, m.p2 + 1)
            self.odometry.append(m.copy())
        for m in private_loop_closures:
            assert m.r1 == self.id and m.r2 == self.id
            self.n = max(self.n, m.p1 + 1, m.p2 + 1)
            mm = m.copy()
            mm.is_known_inlier = False
            self.private_lc.append(mm)
        for m in shared_loop_closures:
            mm = m.copy()
            mm.is_known_inlier = False
            if m.r1 == self.id:
                assert m.r2 != self.id
                self.n = max(self.n, m.p1 + 1)
                self.local_shared_pose_ids.add((self.id, m.p1))
                self.neighbor_shared_pose_ids.add((m.r2, m.p2))
                self.neighbor_robot_ids.add(m.r2)
            else:
                assert m.r2 == self.id
                self.n = max(self.n, m.p2 + 1)
                self.local_shared_pose_ids.add((self.id, m.p2))
                self.neighbor_shared_pose_ids.add((m.r1, m.p1))
                self.neighbor_robot_ids.add(m.r1)
            self.shared_lc.append(mm)

        # Odometry edges are known inliers with fixed weight 1.
        for m in self.odometry:
            m.is_known_inlier = True

        self._manifold = LiftedSEManifold(self.r, self.d, self.n)
        self.problem = QuadraticProblem(self.n, self.d, self.r)
        self._build_assemblers()
        self._construct_q()

        if (T_init is not None and T_init.shape ==
                (self.d, self.dh * self.n)):
            self.T_local_init = T_init.copy()
        else:
            self._local_initialization()

        self.state = PGOAgentState.WAIT_FOR_INITIALIZATION

        if self.id == 0 or not self.params.multirobot_initialization:
            assert self.YLift is not None
            Xt = _T_to_Xt(self.T_local_init, self.r, self.YLift)
            self.X = torch.from_numpy(Xt).to(self.device)
            self.XInit = self.X.clone()
            self.state = PGOAgentState.INITIALIZED
            if self.params.acceleration:
                self._initialize_acceleration()
            if self.logger:
                self.logger.log_trajectory(self.d, self.n, self.T_local_init,
                                           "trajectory_initial.csv")

    def _build_assemblers(self) -> None:
        all_meas = self.odometry + self.private_lc + self.shared_lc
        npriv = len(self.odometry) + len(self.private_lc)
        shared_flags = [False] * npriv + [True] * len(self.shared_lc)
        local_ep = [0] * npriv + [
            0 if m.r1 == self.id else 1 for m in self.shared_lc]
        self._all_meas = all_meas
        self._q_assembler = QAssembler(self.n, self.d, all_meas,
                                       shared_flags, local_ep)
        # Neighbor slot order: deterministic sorted PoseIDs.
        self._nbr_slot_order = sorted(self.neighbor_shared_pose_ids)
        slot_of = {pid: k for k, pid in enumerate(self._nbr_slot_order)}
        nbr_slots = []
        for m in self.shared_lc:
            pid = (m.r2, m.p2) if m.r1 == self.id else (m.r1, m.p1)
            nbr_slots.append(slot_of[pid])
        self._g_assembler = GAssembler(
            self.n, self.d, self.shared_lc,
            [0 if m.r1 == self.id else 1 for m in self.shared_lc],
            nbr_slots)

    def _construct_q(self) -> None:
        Q = self._q_assembler.assemble(self._weights_tensor())
        if self.device.type != "cpu":
            Q = Q.to(self.device)
        self.problem.set_q(Q)

    def _construct_g(self, pose_dict: Dict[PoseID, np.ndarray]) -> bool:
        """Build the linear term from cached neighbor poses. Returns False
        (skip update) when any required pose is missing
        (PGOAgent.cpp:806-814)."""
        n_slots = len(self._nbr_slot_order)
        buf = np.zeros((n_slots, self.dh, self.r))
        for k, pid in enumerate(self._nbr_slot_order):
            v = pose_dict.get(pid)
            if v is None:
                if self.params.verbose:
                    print(f"agent {self.id}: missing neighbor pose {pid}")
                return False
            buf[k] = v.T  # (r, dh) -> (dh, r)
        if getattr(self, "_soa", False):
            w = torch.from_numpy(self._shared_ma.weight.copy())
        else:
            w = torch.tensor([m.weight for m in self.shared_lc],
                             dtype=torch.float64)
        nbr = torch.from_numpy(buf).to(self.device)
        Gt = self._g_assembler.assemble(nbr, w, self.r)
        self.problem.set_g(Gt)
        return True

    def _local_initialization(self) -> None:
        meas = self.odometry + self.private_lc
        if not meas:
            # isolated / boundary-only agent: identity local init
            dh = self.dh
            T = np.zeros((self.d, self.n * dh))
            for i in range(self.n):
                T[:, i * dh:i * dh + self.d] = np.eye(self.d)
            self.T_local_init = T
        elif self.params.robust_cost_type == RobustCostType.L2:
            self.T_local_init = chordal_initialization(self.d, self.n, meas)
        else:
            # Robust mode: don't trust loop closures; odometry init
            # (PGOAgent.cpp:952-957).
            self.T_local_init = odometry_initialization(
                self.d, self.n, self.odometry)

    # ------------------------------------------------------------------
    # pose access (PoseDict exchange API)
    # ------------------------------------------------------------------
    def set_x(self, Xin: np.ndarray) -> None:
        """Warm start / external estimate, r x (d+1)n (reference setX,
        PGOAgent.cpp:55-68). Also the checkpoint-resume entry point."""
        with self._lock:
            assert self.state != PGOAgentState.WAIT_FOR_DATA
            assert Xin.shape == (self.r, self.dh * self.n)
            self.state = PGOAgentState.INITIALIZED
            self.X = torch.from_numpy(
                np.ascontiguousarray(Xin.T)).to(self.device)
            if self.params.acceleration:
                self._initialize_acceleration()

    def get_x(self) -> np.ndarray:
        with self._lock:
            return self.X.cpu().numpy().T.copy()

    def _block(self, M: Tensor, idx: int) -> np.ndarray:
        b = M[idx * self.dh:(idx + 1) * self.dh, :]
        return b.cpu().numpy().T.copy()  # (r, d+1)

    def get_shared_pose(self, index: int) -> Optional[np.ndarray]:
        if self.state != PGOAgentState.INITIALIZED or index >= self.n:
            return None
        with self._lock:
            return self._block(self.X, index)

    def get_shared_pose_dict(self) -> Optional[Dict[PoseID, np.ndarray]]:
        if self.state != PGOAgentState.INITIALIZED:
            return None
        with self._lock:
            return {pid: self._block(self.X, pid[1])
                    for pid in self.local_shared_pose_ids}

    def get_aux_shared_pose_dict(self) -> Optional[Dict[PoseID, np.ndarray]]:
        assert self.params.acceleration
        if self.state != PGOAgentState.INITIALIZED:
            return None
        with self._lock:
            return {pid: self._block(self.Y, pid[1])
                    for pid in self.local_shared_pose_ids}

    def update_neighbor_poses(self, neighbor_id: int,
                              pose_dict: Dict[PoseID, np.ndarray]) -> None:
        assert neighbor_id != self.id
        nbr_state = self.team_status[neighbor_id].state
        if (self.state == PGOAgentState.WAIT_FOR_INITIALIZATION
                and nbr_state == PGOAgentState.INITIALIZED):
            self.initialize_in_global_frame(neighbor_id, pose_dict)
        for pid, var in pose_dict.items():
            assert pid[0] == neighbor_id
            self.num_poses_received += 1
            if pid not in self.neighbor_shared_pose_ids:
                continue
            if (self.state == PGOAgentState.INITIALIZED
                    and nbr_state == PGOAgentState.INITIALIZED):
                with self._lock:
                    self.neighbor_pose_dict[pid] = np.asarray(var).copy()

    def update_aux_neighbor_poses(self, neighbor_id: int,
                                  pose_dict: Dict[PoseID, np.ndarray]) -> None:
        assert self.params.acceleration and neighbor_id != self.id
        for pid, var in pose_dict.items():
            assert pid[0] == neighbor_id
            self.num_poses_received += 1
            if pid not in self.neighbor_shared_pose_ids:
                continue
            if (self.state == PGOAgentState.INITIALIZED and
                    self.team_status[neighbor_id].state
                    == PGOAgentState.INITIALIZED):
                with self._lock:
                    self.neighbor_aux_pose_dict[pid] = np.asarray(var).copy()

    def set_neighbor_status(self, status: PGOAgentStatus) -> None:
        self.team_status[status.agent_id] = status

    def get_status(self) -> PGOAgentStatus:
        # Sync the live state into the shared status (reference
        # PGOAgent.h:282-288).
        self.status.agent_id = self.id
        self.status.state = self.state
        self.status.instance_number = self.instance_number
        self.status.iteration_number = self.iteration_number
        return self.status

    def get_neighbors(self) -> List[int]:
        return sorted(self.neighbor_robot_ids)

    def has_shared_lc(self) -> bool:
        """True when this agent has inter-robot loop closures (works for
        both object-mode and SoA agents)."""
        if getattr(self, "_soa", False):
            return len(self._shared_ma) > 0
        return bool(self.shared_lc)

    def get_neighbor_public_poses(self, neighbor_id: int) -> List[int]:
        assert neighbor_id in self.neighbor_robot_ids
        return [p for (rid, p) in self.neighbor_shared_pose_ids
                if rid == neighbor_id]

    def set_global_anchor(self, M: np.ndarray) -> None:
        assert M.shape == (self.r, self.dh)
        self.global_anchor = M.copy()

    # ------------------------------------------------------------------
    # distributed initialization (robust frame alignment)
    # ------------------------------------------------------------------
    def _find_shared_lc_with_neighbor(self, nid: PoseID) -> RelativeSEMeasurement:
        for m in self.shared_lc:
            if ((m.r1 == nid[0] and m.p1 == nid[1])
                    or (m.r2 == nid[0] and m.p2 == nid[1])):
                return m
        raise RuntimeError("Cannot find shared loop closure with neighbor.")

    def _compute_neighbor_transform(self, nid: PoseID,
                                    var: np.ndarray) -> np.ndarray:
        """Candidate alignment T_world2_world1 from one shared edge
        (reference PGOAgent.cpp:250-288)."""
        assert self.YLift is not None
        d, dh = self.d, self.dh
        m = self._find_shared_lc_with_neighbor(nid)
        dT = np.eye(dh)
        dT[:d, :d] = m.R
        dT[:d, d] = m.t
        T_w2_f2 = np.eye(dh)
        # Round the received lifted pose back to SE(d).
        T_w2_f2[:d, :] = self.YLift.T @ var
        T = self.T_local_init
        T_w1_f1 = np.eye(dh)
        if m.r1 == nid[0]:
            # Incoming edge: neighbor owns the tail.
            T_f1_f2 = np.linalg.inv(dT)
            T_w1_f1[:d, :] = T[:, m.p2 * dh:(m.p2 + 1) * dh]
        else:
            T_f1_f2 = dT
            T_w1_f1[:d, :] = T[:, m.p1 * dh:(m.p1 + 1) * dh]
        T_w2_f1 = T_w2_f2 @ np.linalg.inv(T_f1_f2)
        T_w2_w1 = T_w2_f1 @ np.linalg.inv(T_w1_f1)
        check_rotation_matrix(T_w2_w1[:d, :d], tol=1e-6)
        return T_w2_w1

    def _compute_robust_neighbor_transform_two_stage(
            self, neighbor_id: int,
            pose_dict: Dict[PoseID, np.ndarray]) -> np.ndarray:
        """GNC rotation averaging + inlier translation averaging
        (reference PGOAgent.cpp:290-331)."""
        R_vec, t_vec = [], []
        for nid, var in pose_dict.items():
            if nid in self.neighbor_shared_pose_ids:
                T = self._compute_neighbor_transform(nid, np.asarray(var))
                R_vec.append(T[:self.d, :self.d])
                t_vec.append(T[:self.d, self.d])
        if not R_vec:
            raise RuntimeError("no shared edges with neighbor")
        max_rot_err = angular_to_chordal_so3(0.5)  # ~30 deg
        R_opt, inliers = robust_single_rotation_averaging(
            R_vec, None, max_rot_err)
        if len(inliers) == 0:
            raise RuntimeError("empty inlier set in robust initialization")
        t_opt = single_translation_averaging([t_vec[i] for i in inliers])
        T_opt = np.eye(self.dh)
        T_opt[:self.d, :self.d] = R_opt
        T_opt[:self.d, self.d] = t_opt
        return T_opt

    def _compute_robust_neighbor_transform(
            self, neighbor_id: int,
            pose_dict: Dict[PoseID, np.ndarray]) -> np.ndarray:
        """Single-stage GNC pose averaging over candidate alignments
        (reference PGOAgent.cpp:333-367; the two-stage variant below is
        what initializeInGlobalFrame uses by default)."""
        from .averaging import robust_single_pose_averaging
        from .robust import RobustCost
        R_vec, t_vec = [], []
        for nid, var in pose_dict.items():
            if nid in self.neighbor_shared_pose_ids:
                T = self._compute_neighbor_transform(nid, np.asarray(var))
                R_vec.append(T[:self.d, :self.d])
                t_vec.append(T[:self.d, self.d])
        if not R_vec:
            raise RuntimeError("no shared edges with neighbor")
        kappa = 1.82 * np.ones(len(R_vec))   # rot stddev ~30 deg
        tau = 0.01 * np.ones(len(R_vec))     # trans stddev ~10 m
        cbar = RobustCost.error_threshold_at_quantile(0.9, 3)
        R_opt, t_opt, inliers = robust_single_pose_averaging(
            R_vec, t_vec, kappa, tau, cbar)
        if len(inliers) == 0:
            raise RuntimeError("empty inlier set in robust initialization")
        T_opt = np.eye(self.dh)
        T_opt[:self.d, :self.d] = R_opt
        T_opt[:self.d, self.d] = t_opt
        return T_opt

    def get_aux_shared_pose(self, index: int) -> Optional[np.ndarray]:
        """Nesterov auxiliary pose block (reference PGOAgent.cpp:85-93)."""
        assert self.params.acceleration
        if self.state != PGOAgentState.INITIALIZED or index >= self.n:
            return None
        with self._lock:
            return self._block(self.Y, index)

    @staticmethod
    def is_duplicate_measurement(m: RelativeSEMeasurement,
                                 measurements) -> bool:
        """Reference PGOAgent.cpp:1291-1299."""
        return any(m.r1 == m2.r1 and m.r2 == m2.r2
                   and m.p1 == m2.p1 and m.p2 == m2.p2
                   for m2 in measurements)

    def initialize_in_global_frame(self, neighbor_id: int,
                                   pose_dict: Dict[PoseID, np.ndarray]) -> None:
        assert self.YLift is not None
        with self._lock:
            self.neighbor_pose_dict.clear()
            self.neighbor_aux_pose_dict.clear()
            try:
                T_w2_w1 = self._compute_robust_neighbor_transform_two_stage(
                    neighbor_id, pose_dict)
            except RuntimeError:
                if self.params.verbose:
                    print("Robust initialization unsuccessful; will retry.")
                return
            d, dh = self.d, self.dh
            T = self.T_local_init.copy()
            for i in range(self.n):
                Ti = np.eye(dh)
                Ti[:d, :] = T[:, i * dh:(i + 1) * dh]
                Tn = T_w2_w1 @ Ti
                T[:, i * dh:(i + 1) * dh] = Tn[:d, :]
            Xt = _T_to_Xt(T, self.r, self.YLift)
            self.X = torch.from_numpy(Xt).to(self.device)
            self.XInit = self.X.clone()
            self.state = PGOAgentState.INITIALIZED
            if self.params.acceleration:
                self._initialize_acceleration()
            if self.logger:
                self.logger.log_trajectory(d, self.n, T,
                                           "trajectory_initial.csv")

    # ------------------------------------------------------------------
    # iteration
    # ------------------------------------------------------------------
    def iterate(self, do_optimization: bool) -> None:
        self.iteration_number += 1

        if self.iteration_number == 50 and self.logger:
            T = self.get_trajectory_in_global_frame()
            if T is not None:
                self.logger.log_trajectory(self.d, self.n, T,
                                           "trajectory_early_stop.csv")

        if self._should_update_loop_closure_weights():
            self.update_loop_closures_weights()
            self.robust_cost.update()
            if not self.params.robust_opt_warm_start:
                assert self.XInit is not None
                self.X = self.XInit.clone()
            if self.params.acceleration:
                self._initialize_acceleration()

        if self.state != PGOAgentState.INITIALIZED:
            return

        with self._lock:
            self.XPrev = self.X.clone()
            if self.params.acceleration:
                self._update_gamma()
                self._update_alpha()
                self._update_y()
                success = self._update_x(do_optimization, acceleration=True)
                self._update_v()
                if self._should_restart():
                    self._restart_nesterov(do_optimization)
                self.publish_public_poses_requested = True
            else:
                success = self._update_x(do_optimization, acceleration=False)
                if do_optimization:
                    self.publish_public_poses_requested = True

            if do_optimization:
                self.status.state = self.state
                self.status.instance_number = self.instance_number
                self.status.iteration_number = self.iteration_number
                self.status.relative_change = float(
                    torch.linalg.norm(self.X - self.XPrev)) / math.sqrt(self.n)
                ready = success
                if self.status.relative_change > self.params.rel_change_tol:
                    ready = False
                if (self._converged_loop_closure_ratio()
                        < self.params.robust_opt_min_convergence_ratio):
                    ready = False
                self.status.ready_to_terminate = ready

    def _update_x(self, do_optimization: bool, acceleration: bool) -> bool:
        if not do_optimization:
            if acceleration:
                self.X = self.Y.clone()
            return True
        assert self.state == PGOAgentState.INITIALIZED

        if self.params.robust_cost_type != RobustCostType.L2:
            self._construct_q()

        pose_dict = (self.neighbor_aux_pose_dict if acceleration
                     else self.neighbor_pose_dict)
        if self.has_shared_lc() and not self._construct_g(pose_dict):
            return False
        if not self.has_shared_lc():
            self.problem.set_g(torch.zeros(
                self.dh * self.n, self.r, dtype=torch.float64,
                device=self.device))

        # RBCD knob set (PGOAgent.cpp:1134-1137).
        X_start = self.Y if acceleration else self.X
        if (self.device.type != "cpu"
                and self.params.algorithm == OptAlgorithm.RTR):
            # Device-resident solve: tCG control state lives on the GPU,
            # one host sync per local solve.
            if getattr(self, "_dev_solver", None) is None:
                from .ops.hip_backend import DeviceSolver
                self._dev_solver = DeviceSolver(self.n, self.d, self.r,
                                                self.device, max_inner=10)
            X_work = X_start.clone() if acceleration else self.X
            for _ in range(self.params.tr_max_iterations):
                stats = self._dev_solver.solve(self.problem, X_work,
                                               tol=self.params.inner_tol,
                                               Delta0=100.0)
            self.X = X_work
            from .types import OptResult
            res = OptResult(success=True,
                            f_init=stats["f_init"], f_opt=stats["f_opt"],
                            grad_norm_init=stats["grad_norm_init"],
                            grad_norm_opt=stats.get("grad_norm_opt", 0.0))
            self.last_opt_result = res
            return True
        tr = TRParams(tolerance=self.params.inner_tol, initial_radius=100.0,
                      max_iterations=self.params.tr_max_iterations,
                      max_inner_iterations=10)
        opt = QuadraticOptimizer(self.problem, self.params.algorithm, tr,
                                 verbose=self.params.verbose)
        self.X = opt.optimize(X_start)
        self.last_opt_result = opt.result
        return True

    # --- Nesterov acceleration (PGOAgent.cpp:1054-1091) ---------------
    def _initialize_acceleration(self) -> None:
        assert self.params.acceleration
        if self.state == PGOAgentState.INITIALIZED:
            self.XPrev = self.X.clone()
            self.gamma = 0.0
            self.alpha = 0.0
            self.V = self.X.clone()
            self.Y = self.X.clone()

    def _update_gamma(self) -> None:
        K = self.params.num_robots
        self.gamma = (1 + math.sqrt(1 + 4 * K * K * self.gamma * self.gamma)) \
            / (2 * K)

    def _update_alpha(self) -> None:
        self.alpha = 1.0 / (self.gamma * self.params.num_robots)

    def _update_y(self) -> None:
        M = (1 - self.alpha) * self.X + self.alpha * self.V
        self.Y = self._manifold.project(M)

    def _update_v(self) -> None:
        M = self.V + self.gamma * (self.X - self.Y)
        self.V = self._manifold.project(M)

    def _should_restart(self) -> bool:
        if self.params.acceleration:
            return (self.iteration_number + 1) % self.params.restart_interval == 0
        return False

    def _restart_nesterov(self, do_optimization: bool) -> None:
        if self.params.acceleration and self.state == PGOAgentState.INITIALIZED:
            self.X = self.XPrev.clone()
            self._update_x(do_optimization, acceleration=False)
            self.V = self.X.clone()
            self.Y = self.X.clone()
            self.gamma = 0.0
            self.alpha = 0.0

    # --- GNC weight updates (PGOAgent.cpp:1174-1245) ------------------
    def _should_update_loop_closure_weights(self) -> bool:
        if self.params.robust_cost_type == RobustCostType.L2:
            return False
        return (self.iteration_number + 1) % self.params.robust_opt_inner_iters == 0

    def _lifted_blocks(self, idx: int) -> Tuple[np.ndarray, np.ndarray]:
        Xi = self.X[idx * self.dh:(idx + 1) * self.dh, :].cpu().numpy().T
        return Xi[:, :self.d], Xi[:, self.d]

    def update_loop_closures_weights(self) -> None:
        assert self.state == PGOAgentState.INITIALIZED
        if getattr(self, "_soa", False):
            raise RuntimeError(
                "SoA agents update GNC weights via the packed GPU path "
                "(_packed_update_weights); robust SoA needs a cuda device")
        for m in self.private_lc:
            if m.is_known_inlier:
                continue
            Y1, p1 = self._lifted_blocks(m.p1)
            Y2, p2 = self._lifted_blocks(m.p2)
            residual = math.sqrt(compute_measurement_error(m, Y1, p1, Y2, p2))
            m.weight = self.robust_cost.weight(residual)
        # Owner-computes rule: agent i updates weights with j > i.
        for m in self.shared_lc:
            if m.is_known_inlier:
                continue
            if m.r1 == self.id:
                if m.r2 < self.id:
                    continue
                Y1, p1 = self._lifted_blocks(m.p1)
                nbr = self.neighbor_pose_dict.get((m.r2, m.p2))
                if nbr is None:
                    continue
                Y2, p2 = nbr[:, :self.d], nbr[:, self.d]
            else:
                if m.r1 < self.id:
                    continue
                Y2, p2 = self._lifted_blocks(m.p2)
                nbr = self.neighbor_pose_dict.get((m.r1, m.p1))
                if nbr is None:
                    continue
                Y1, p1 = nbr[:, :self.d], nbr[:, self.d]
            residual = math.sqrt(compute_measurement_error(m, Y1, p1, Y2, p2))
            m.weight = self.robust_cost.weight(residual)
        self.publish_weights_requested = True

    def set_measurement_weight(self, src: PoseID, dst: PoseID,
                               weight: float) -> bool:
        """Apply a weight computed by the owning agent (weight sync,
        PGOAgent.cpp owner-computes rule)."""
        for m in self.shared_lc:
            if (m.r1, m.p1) == src and (m.r2, m.p2) == dst:
                m.weight = weight
                return True
        return False

    def get_shared_measurement_weights(self) -> List[Tuple[PoseID, PoseID, float]]:
        return [((m.r1, m.p1), (m.r2, m.p2), m.weight) for m in self.shared_lc]

    def _converged_loop_closure_ratio(self) -> float:
        if self.params.robust_cost_type != RobustCostType.GNC_TLS:
            return 1.0
        total = converged = 0
        for m in self.private_lc + self.shared_lc:
            if m.is_known_inlier:
                continue
            total += 1
            if m.weight in (0.0, 1.0):
                converged += 1
        return converged / total if total else 1.0

    # ------------------------------------------------------------------
    # termination / rounding / reset
    # ------------------------------------------------------------------
    def should_terminate(self) -> bool:
        if self.iteration_number > self.params.max_num_iters:
            return True
        for rid in range(self.params.num_robots):
            st = self.team_status[rid]
            if st.state != PGOAgentState.INITIALIZED:
                return False
        for rid in range(self.params.num_robots):
            if not self.team_status[rid].ready_to_terminate:
                return False
        return True

    def get_trajectory_in_local_frame(self) -> Optional[np.ndarray]:
        """Round to SE(d) in the frame of the agent's first pose
        (PGOAgent.cpp:481-498)."""
        if self.state != PGOAgentState.INITIALIZED:
            return None
        with self._lock:
            X = self.X.cpu().numpy().T  # (r, N)
            return self._round_trajectory(X, X[:, :self.d],
                                          X[:, self.d])

    def get_trajectory_in_global_frame(self) -> Optional[np.ndarray]:
        if self.global_anchor is None:
            return None
        if self.state != PGOAgentState.INITIALIZED:
            return None
        with self._lock:
            X = self.X.cpu().numpy().T
            anchor = self.global_anchor
            return self._round_trajectory(X, anchor[:, :self.d],
                                          anchor[:, self.d])

    def _round_trajectory(self, X: np.ndarray, Ya: np.ndarray,
                          pa: np.ndarray) -> np.ndarray:
        from .liegroups import project_to_rotation_group
        d, dh = self.d, self.dh
        T = Ya.T @ X  # (d, N)
        t0 = Ya.T @ pa
        out = T.copy()
        for i in range(self.n):
            out[:, i * dh:i * dh + d] = project_to_rotation_group(
                T[:, i * dh:i * dh + d])
            out[:, i * dh + d] -= t0
        return out

    def get_pose_in_global_frame(self, pose_id: int) -> Optional[np.ndarray]:
        if self.global_anchor is None or self.state != PGOAgentState.INITIALIZED:
            return None
        if pose_id >= self.n:
            return None
        with self._lock:
            Xi = self.X[pose_id * self.dh:(pose_id + 1) * self.dh, :]
            Ya = self.global_anchor[:, :self.d]
            pa = self.global_anchor[:, self.d]
            Ti = Ya.T @ Xi.cpu().numpy().T
            Ti[:, self.d] -= Ya.T @ pa
            return Ti

    def get_neighbor_pose_in_global_frame(self, neighbor_id: int,
                                          pose_id: int
                                          ) -> Optional[np.ndarray]:
        """Round a cached neighbor pose to SE(d) in the global frame
        (reference PGOAgent.cpp:540-562)."""
        if self.global_anchor is None or self.state != PGOAgentState.INITIALIZED:
            return None
        with self._lock:
            v = self.neighbor_pose_dict.get((neighbor_id, pose_id))
            if v is None:
                return None
            Ya = self.global_anchor[:, :self.d]
            pa = self.global_anchor[:, self.d]
            Ti = Ya.T @ v
            Ti[:, self.d] -= Ya.T @ pa
            return Ti

    def local_pose_graph_optimization(self) -> np.ndarray:
        """Single-robot full-batch RTR at r = d (reference
        PGOAgent.cpp:964-990; batch knob set at 981-984)."""
        if self.T_local_init is None:
            self._local_initialization()
        from .quadratic import assemble_connection_laplacian
        meas = self.odometry + self.private_lc
        Q = assemble_connection_laplacian(meas, self.n, self.d)
        problem = QuadraticProblem(self.n, self.d, self.d)
        if self.device.type != "cpu":
            Q = Q.to(self.device)
        problem.set_q(Q)
        tr = TRParams(tolerance=1e-1, initial_radius=10.0,
                      max_iterations=10, max_inner_iterations=50)
        opt = QuadraticOptimizer(problem, OptAlgorithm.RTR, tr,
                                 verbose=self.params.verbose)
        X0 = torch.from_numpy(
            np.ascontiguousarray(self.T_local_init.T)).to(self.device)
        Xopt = opt.optimize(X0)
        self.last_opt_result = opt.result
        return Xopt.cpu().numpy().T.copy()

    def reset(self) -> None:
        self.end_optimization_loop()
        if self.logger:
            meas = self.odometry + self.private_lc + self.shared_lc
            if meas:
                self.logger.log_measurements(meas, "measurements.csv")
            T = self.get_trajectory_in_global_frame()
            if T is not None:
                self.logger.log_trajectory(self.d, self.n, T,
                                           "trajectory_optimized.csv")
            # pre-rounding lifted estimate (checkpoint; reference writes
            # X.txt at PGOAgent.cpp:602) - resume via numpy load + set_x
            if self.X is not None and self.logger.log_dir:
                import os as _os
                np.save(_os.path.join(self.logger.log_dir, "X.npy"),
                        self.X.cpu().numpy().T)
        self.instance_number += 1
        self.iteration_number = 0
        self.num_poses_received = 0
        self.state = PGOAgentState.WAIT_FOR_DATA
        self.status = PGOAgentStatus(self.id, self.state,
                                     self.instance_number, 0, False, 0.0)
        self.odometry.clear()
        self.private_lc.clear()
        self.shared_lc.clear()
        self.neighbor_pose_dict.clear()
        self.neighbor_aux_pose_dict.clear()
        self.local_shared_pose_ids.clear()
        self.neighbor_shared_pose_ids.clear()
        self.neighbor_robot_ids.clear()
        self.team_status = {rid: PGOAgentStatus(rid)
                            for rid in range(self.params.num_robots)}
        self.problem = None
        self.robust_cost.reset()
        self.global_anchor = None
        self.T_local_init = None
        self.XInit = None
        self.n = 1
        self.X = None

    # ------------------------------------------------------------------
    # SoA setup for large graphs (MeasurementArray end-to-end; no
    # per-edge Python objects). Feeds the packed GPU path incl. robust.
    # ------------------------------------------------------------------
    def set_pose_graph_arrays(self, odometry_ma, private_ma, shared_ma,
                              T_init: Optional[np.ndarray] = None) -> None:
        """Array-based set_pose_graph for large graphs. odometry/private/
        shared are MeasurementArrays in LOCAL indices (r1/r2 = robot ids).
        Robust (non-L2) agents use vectorized odometry initialization."""
        from .measurements import (MeasurementArray,
                                   concat_measurement_arrays,
                                   odometry_initialization_array)
        assert self.state == PGOAgentState.WAIT_FOR_DATA
        self._soa = True
        d = self.d
        odometry_ma.is_known_inlier[:] = True
        private_ma.is_known_inlier[:] = False
        shared_ma.is_known_inlier[:] = False
        self._odo_ma, self._priv_ma, self._shared_ma = \
            odometry_ma, private_ma, shared_ma
        n = 1
        for ma in (odometry_ma, private_ma):
            if len(ma):
                n = max(n, int(ma.p1.max()) + 1, int(ma.p2.max()) + 1)
        sh = shared_ma
        if len(sh):
            mine1 = sh.r1 == self.id
            local_p = np.where(mine1, sh.p1, sh.p2)
            nbr_r = np.where(mine1, sh.r2, sh.r1)
            nbr_p = np.where(mine1, sh.p2, sh.p1)
            n = max(n, int(local_p.max()) + 1)
            self.local_shared_pose_ids = {
                (self.id, int(p)) for p in np.unique(local_p)}
            nbr_ids = np.unique(np.stack([nbr_r, nbr_p], 1), axis=0)
            self.neighbor_shared_pose_ids = {
                (int(a), int(b)) for a, b in nbr_ids}
            self.neighbor_robot_ids = {int(x) for x in np.unique(nbr_r)}
        self.n = n
        self._manifold = LiftedSEManifold(self.r, d, n)
        self.problem = QuadraticProblem(n, d, self.r)

        npriv = len(odometry_ma) + len(private_ma)
        nsh = len(shared_ma)
        all_ma = concat_measurement_arrays(
            [odometry_ma, private_ma, shared_ma])
        shared_flags = [False] * npriv + [True] * nsh
        lep = [0] * npriv + [0 if r1 == self.id else 1 for r1 in sh.r1]
        self._all_ma = all_ma
        self._q_assembler = QAssembler(n, d, all_ma, shared_flags, lep)
        self._nbr_slot_order = sorted(self.neighbor_shared_pose_ids)
        slot_of = {pid: k for k, pid in enumerate(self._nbr_slot_order)}
        if nsh:
            nbr_slots = [slot_of[(int(a), int(b))]
                         for a, b in zip(nbr_r, nbr_p)]
        else:
            nbr_slots = []
        self._g_assembler = GAssembler(
            n, d, shared_ma,
            [0 if r1 == self.id else 1 for r1 in sh.r1], nbr_slots)
        self._construct_q()
        if T_init is not None:
            self.T_local_init = T_init
        elif self.params.robust_cost_type != RobustCostType.L2:
            self.T_local_init = odometry_initialization_array(
                d, n, odometry_ma)
        else:
            self.T_local_init = chordal_initialization(
                d, n, concat_measurement_arrays(
                    [odometry_ma, private_ma]).to_list())
        assert self.YLift is not None
        Xt = _T_to_Xt(self.T_local_init, self.r, self.YLift)
        self.X = torch.from_numpy(Xt).to(self.device)
        self.XInit = self.X.clone()
        self.state = PGOAgentState.INITIALIZED

    def _weights_tensor(self) -> Tensor:
        if getattr(self, "_soa", False):
            return torch.from_numpy(self._all_ma.weight.copy())
        return torch.tensor([m.weight for m in self._all_meas],
                            dtype=torch.float64)

    # ------------------------------------------------------------------
    # packed fast path (GPU): device-tensor neighbor buffers + native
    # C++ solve/eval, driven by DistributedRBCDDriver._run_packed.
    # ------------------------------------------------------------------
    def _ensure_packed(self, dev) -> None:
        if getattr(self, "_packed_ready", False):
            return
        n_slots = len(self._nbr_slot_order)
        self._nbr_buffer = torch.zeros(max(n_slots, 1), self.dh, self.r,
                                       dtype=torch.float64, device=dev)
        self._nbr_buffer_aux = torch.zeros_like(self._nbr_buffer)
        # full per-measurement weight vector on device (odometry |
        # private | shared); the shared tail doubles as the G-assembly
        # weight view and the exchange payload.
        self._all_weights_dev = self._weights_tensor().to(dev)
        if getattr(self, "_soa", False):
            nsh = len(self._shared_ma)
        else:
            nsh = len(self.shared_lc)
        self._shared_w_off = self._all_weights_dev.numel() - nsh
        self._w_shared_dev = self._all_weights_dev[self._shared_w_off:]
        if getattr(self, "_dev_solver", None) is None:
            from .ops.hip_backend import DeviceSolver
            self._dev_solver = DeviceSolver(self.n, self.d, self.r, dev,
                                            max_inner=10)
        ga = self._g_assembler
        if getattr(ga, "_dev_cache", None) is None or ga._dev_cache[0] != dev:
            ga._dev_cache = (dev, ga.E0.to(dev).contiguous(),
                             ga.local_pose.to(dev), ga.nbr_slot.to(dev))
        _, E0, lp, slots = ga._dev_cache
        self._dev_solver.set_gdata(E0, lp, slots, self._w_shared_dev)
        self._dev_solver.bind_problem_static(self.problem)
        if self.params.acceleration:
            self.Y = self.X.clone()
            self.V = self.X.clone()
            self.gamma = 0.0
            self.alpha = 0.0
        self._packed_ready = True

    def _packed_solve(self, accel: bool) -> None:
        if accel:
            self.X.copy_(self.Y)
        nbr = self._nbr_buffer_aux if accel else self._nbr_buffer
        for _ in range(self.params.tr_max_iterations):
            self._dev_solver.round_solve(self.X, nbr,
                                         tol=self.params.inner_tol,
                                         Delta0=100.0)

    def _packed_eval(self, out=None):
        """Device 3-vector [f, 0.5<X,G>, ||rgrad||^2] with fresh G.
        With an explicit out row the RAW enqueue is used so the whole
        phase can be captured into one driver-level graph."""
        if out is not None:
            self._dev_solver.round_eval_raw(self.X, self._nbr_buffer, out)
            return out
        return self._dev_solver.round_eval(self.X, self._nbr_buffer)

    def _packed_solve_async(self, accel: bool, first: bool = True) -> None:
        if accel and first:
            self.X.copy_(self.Y)
        nbr = self._nbr_buffer_aux if accel else self._nbr_buffer
        self._dev_solver.round_solve_async(self.X, nbr,
                                           tol=self.params.inner_tol)

    def _packed_solve_finish(self) -> None:
        self._dev_solver.round_solve_finish(self.X)

    def _packed_eval_async(self, out=None):
        return self._dev_solver.round_eval_async(self.X, self._nbr_buffer,
                                                 out)

    def _packed_eval_join(self):
        return self._dev_solver.eval_join(self.X)

    # --- packed robust (GNC) support ----------------------------------
    def _packed_gnc_setup(self, dev) -> None:
        """Device tensors for the GNC weight-update kernel. SoA mode."""
        if getattr(self, "_gnc_ready", False):
            return
        from .measurements import concat_measurement_arrays
        assert getattr(self, "_soa", False), \
            "packed robust mode requires set_pose_graph_arrays"
        priv, sh = self._priv_ma, self._shared_ma
        n_odo = len(self._odo_ma)
        n_priv = len(priv)
        slot_of = {pid: k for k, pid in enumerate(self._nbr_slot_order)}
        import numpy as np
        ne = n_priv + len(sh)
        e1_idx = np.zeros(ne, dtype=np.int64)
        e2_idx = np.zeros(ne, dtype=np.int64)
        e1_nbr = np.zeros(ne, dtype=np.uint8)
        e2_nbr = np.zeros(ne, dtype=np.uint8)
        upd = np.zeros(ne, dtype=np.uint8)
        widx = np.zeros(ne, dtype=np.int64)
        R = np.zeros((ne, self.d, self.d))
        t = np.zeros((ne, self.d))
        kap = np.zeros(ne)
        tau = np.zeros(ne)
        if n_priv:
            e1_idx[:n_priv] = priv.p1
            e2_idx[:n_priv] = priv.p2
            upd[:n_priv] = (~priv.is_known_inlier).astype(np.uint8)
            widx[:n_priv] = n_odo + np.arange(n_priv)
            R[:n_priv] = priv.R
            t[:n_priv] = priv.t
            kap[:n_priv] = priv.kappa
            tau[:n_priv] = priv.tau
        if len(sh):
            mine1 = sh.r1 == self.id
            other = np.where(mine1, sh.r2, sh.r1)
            for k in range(len(sh)):
                e = n_priv + k
                if mine1[k]:
                    e1_idx[e] = sh.p1[k]
                    e2_idx[e] = slot_of[(int(sh.r2[k]), int(sh.p2[k]))]
                    e2_nbr[e] = 1
                else:
                    e1_idx[e] = slot_of[(int(sh.r1[k]), int(sh.p1[k]))]
                    e1_nbr[e] = 1
                    e2_idx[e] = sh.p2[k]
            # owner-computes: this agent updates edges whose OTHER robot
            # has a larger id (PGOAgent.cpp:1201-1235)
            upd[n_priv:] = ((other > self.id)
                            & ~sh.is_known_inlier).astype(np.uint8)
            widx[n_priv:] = n_odo + n_priv + np.arange(len(sh))
            R[n_priv:] = sh.R
            t[n_priv:] = sh.t
            kap[n_priv:] = sh.kappa
            tau[n_priv:] = sh.tau
        self._gnc = {
            "e1_idx": torch.from_numpy(e1_idx).to(dev),
            "e2_idx": torch.from_numpy(e2_idx).to(dev),
            "e1_nbr": torch.from_numpy(e1_nbr).to(dev),
            "e2_nbr": torch.from_numpy(e2_nbr).to(dev),
            "upd": torch.from_numpy(upd).to(dev),
            "widx": torch.from_numpy(widx).to(dev),
            "R": torch.from_numpy(R).to(dev).contiguous(),
            "t": torch.from_numpy(t).to(dev).contiguous(),
            "kappa": torch.from_numpy(kap).to(dev),
            "tau": torch.from_numpy(tau).to(dev),
            "ne": ne,
        }
        # device-side Q assembly structure (values rebuilt in place)
        qa = self._q_assembler
        self._q_dev = {
            "slots": qa._slots.to(dev),
            "blocks": qa._blocks.to(dev).contiguous(),
            "edge_of": qa._edge_of.to(dev),
        }
        self._gnc_ready = True

    def _packed_update_weights(self) -> None:
        """Launch the GNC weight kernel over private + shared LCs."""
        from .ops import hip_backend as hb
        g = self._gnc
        if g["ne"] == 0:
            return
        p = self.params.robust_cost_params
        hb.gnc_weights(self.X, self._nbr_buffer, g, self._all_weights_dev,
                       self.d, self.r, self.robust_cost.mu,
                       p.gnc_barc ** 2)

    def _packed_rebuild_q(self) -> None:
        """Re-assemble Q values from current weights (in place) and
        refresh the preconditioner factors."""
        from .ops import hip_backend as hb
        Q = self.problem.Q
        hb.q_assemble(Q.vals, self._q_dev["blocks"], self._q_dev["slots"],
                      self._q_dev["edge_of"], self._all_weights_dev, self.dh)
        self.problem.refresh_preconditioner()
        self._dev_solver.bind_problem_static(self.problem)
        self._packed_generation = getattr(self, "_packed_generation", 0) + 1

    def _packed_shared_weights(self) -> Tensor:
        return self._w_shared_dev

    def _packed_nesterov_pre(self) -> None:
        from .ops import hip_backend as hb
        self._update_gamma()
        self._update_alpha()
        self._XPrev_packed = self.X.clone()
        self.Y = hb.polar_affine(self.X, self.V, None,
                                 1.0 - self.alpha, self.alpha, 0.0, self.d)

    def _packed_nesterov_post(self, it: int) -> None:
        from .ops import hip_backend as hb
        self.V = hb.polar_affine(self.V, self.X, self.Y,
                                 1.0, self.gamma, -self.gamma, self.d)
        if (self.iteration_number + it + 1) % self.params.restart_interval == 0:
            # periodic restart (PGOAgent.cpp:1040-1052)
            self.X.copy_(self._XPrev_packed)
            self._dev_solver.round_solve(self.X, self._nbr_buffer,
                                         tol=self.params.inner_tol,
                                         Delta0=100.0)
            self.V.copy_(self.X)
            self.Y.copy_(self.X)
            self.gamma = 0.0
            self.alpha = 0.0

    # ------------------------------------------------------------------
    # asynchronous optimization loop (PGOAgent.cpp:861-916)
    # ------------------------------------------------------------------
    def start_optimization_loop(self, rate_hz: float) -> None:
        assert not self.params.acceleration, \
            "asynchronous updates require non-accelerated mode"
        if self.is_optimization_running():
            return
        self._rate = rate_hz
        self._end_loop = False
        self._opt_thread = threading.Thread(
            target=self._run_optimization_loop, daemon=True)
        self._opt_thread.start()

    def _run_optimization_loop(self) -> None:
        rng = np.random.default_rng()
        while not self._end_loop:
            # Poisson clock: exponentially distributed sleep.
            time.sleep(float(rng.exponential(1.0 / self._rate)))
            if self._end_loop:
                break
            self.iterate(True)

    def end_optimization_loop(self) -> None:
        if not self.is_optimization_running():
            return
        self._end_loop = True
        self._opt_thread.join()
        self._opt_thread = None
        self._end_loop = False

    def is_optimization_running(self) -> bool:
        return self._opt_thread is not None
