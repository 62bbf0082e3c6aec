"""Distributed multi-robot RBCD driver: one process per GPU over RCCL.

Maps agents to ranks (agent a -> rank a % world_size); each round:
  1. the active agent(s) run the device-resident local solve,
  2. one packed all-gather moves every agent's public poses + status
     (payload is tens of KB — latency-bound over xGMI, so one fused
     collective beats per-neighbor p2p),
  3. each agent rebuilds G from fresh neighbor poses and evaluates its
     LOCAL Riemannian gradient norm — which equals the block of the
     CENTRALIZED gradient (the local Q carries the shared-edge diagonal
     corrections and G carries the cross terms, so local grad == central
     block grad when neighbor data is fresh), and the centralized cost is
     sum_agents(f_local - 0.5 <X, G>). One small all-reduce therefore
     reproduces the reference driver's centralized evaluation
     (MultiRobotExample.cpp:279-325) without gathering trajectories.
  4. greedy argmax selection / colored schedule, termination check.

Works identically under gloo (CPU, CI) and RCCL (MI355X); also usable
with world_size == 1 (all agents on one device).
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from .agent import PGOAgent
from .chordal import chordal_initialization
from .comm import Comm
from .driver import RBCDResult
from .io_g2o import adjacency_from_measurements
from .partition import (contiguous_partition, multilevel_partition,
                        partition_measurements)
from .types import PGOAgentParams, PGOAgentState, PGOAgentStatus, \
    RelativeSEMeasurement, RobustCostType

Tensor = torch.Tensor
STATUS_LEN = 6


def _adjacency_from_ma(ma, num_poses):
    nbr = [set() for _ in range(num_poses)]
    for a, b in zip(ma.p1, ma.p2):
        if a != b:
            nbr[a].add(int(b))
            nbr[b].add(int(a))
    return [sorted(x) for x in nbr]


class DistributedRBCDDriver:
    def __init__(self,
                 measurements: Sequence[RelativeSEMeasurement],
                 num_poses: int,
                 num_robots: int,
                 comm: Comm,
                 r: int = 5,
                 partition: str | Sequence[int] = "multilevel",
                 acceleration: bool = False,
                 robust: RobustCostType = RobustCostType.L2,
                 device: str = "cpu",
                 verbose: bool = False,
                 selection: str = "greedy",
                 inner_tol: float = 1e-2,
                 tr_max_iterations: int = 1):
        self.comm = comm
        self.num_robots = num_robots
        self.verbose = verbose
        self.selection = selection
        self.device = device
        from .measurements import MeasurementArray
        self._soa = isinstance(measurements, MeasurementArray)
        if (self._soa and robust != RobustCostType.L2
                and not str(device).startswith("cuda")):
            raise ValueError(
                "robust (non-L2) SoA driver needs a cuda device: GNC "
                "weight updates run in the packed GPU path "
                "(_packed_update_weights); on CPU pass object-mode "
                "measurements (list of RelativeSEMeasurement)")
        d = measurements.d if self._soa else measurements[0].d
        self.d, self.r, self.n_global = d, r, num_poses
        self.dh = d + 1
        self.acceleration = acceleration
        self.robust = robust

        # ---- identical partition on every rank (deterministic) ---------
        if isinstance(partition, str):
            if partition == "contiguous":
                part = contiguous_partition(num_poses, num_robots)
            else:
                if self._soa:
                    adj = _adjacency_from_ma(measurements, num_poses)
                else:
                    adj = adjacency_from_measurements(measurements,
                                                      num_poses)
                part = multilevel_partition(adj, num_robots)
        else:
            part = list(partition)
        if self._soa:
            from .measurements import partition_measurement_array
            (odometry, private_lc, shared_lc, local_idx, global_of,
             self.pose_counts) = partition_measurement_array(
                measurements, num_poses, part, num_robots)
            self.pose_to_index = {}
            for rb in range(num_robots):
                for i, g in enumerate(global_of[rb]):
                    self.pose_to_index[(rb, int(i))] = int(g)
            # global cross-edge exchange maps (GNC weight sync)
            partv = np.asarray(part, dtype=np.int64)
            src_r = partv[measurements.p1]
            dst_r = partv[measurements.p2]
            cross = np.nonzero(src_r != dst_r)[0]
            self._n_cross = len(cross)
            cross_rank = {int(g): i for i, g in enumerate(cross)}
            self._cross_map = []     # per agent: global positions
            self._cross_owned = []   # per agent: owned mask
            for rb in range(num_robots):
                sel = np.nonzero((src_r != dst_r)
                                 & ((src_r == rb) | (dst_r == rb)))[0]
                self._cross_map.append(
                    np.array([cross_rank[int(g)] for g in sel],
                             dtype=np.int64))
                other = np.where(src_r[sel] == rb, dst_r[sel], src_r[sel])
                self._cross_owned.append(other > rb)
        else:
            (odometry, private_lc, shared_lc, self.pose_map,
             self.pose_to_index, self.pose_counts) = partition_measurements(
                measurements, num_poses, part, num_robots)
            # global cross-edge enumeration (GNC weight sync on the dict
            # path; owner-computes rule — PGOAgent.cpp:1201-1244): each
            # inter-robot loop closure gets one global slot, keyed by its
            # (global src, global dst) pose pair in input order.
            partv = np.asarray(part, dtype=np.int64)
            cross_fifo: Dict[Tuple[int, int], List[int]] = {}
            n_cross = 0
            for m in measurements:
                if partv[m.p1] != partv[m.p2]:
                    cross_fifo.setdefault((m.p1, m.p2), []).append(n_cross)
                    n_cross += 1
            self._n_cross = n_cross
            # per-robot: shared_lc edge k -> global cross slot, owned mask
            self._cross_map = []
            self._cross_owned = []
            taken: Dict[Tuple[int, int], int] = {}
            for rb in range(num_robots):
                idxs = []
                owned = []
                seen: Dict[Tuple[int, int], int] = {}
                for m in shared_lc[rb]:
                    g1 = self.pose_to_index[(m.r1, m.p1)]
                    g2 = self.pose_to_index[(m.r2, m.p2)]
                    k = seen.get((g1, g2), 0)
                    seen[(g1, g2)] = k + 1
                    idxs.append(cross_fifo[(g1, g2)][k])
                    other = m.r2 if m.r1 == rb else m.r1
                    owned.append(other > rb)
                self._cross_map.append(np.asarray(idxs, dtype=np.int64))
                self._cross_owned.append(np.asarray(owned, dtype=bool))

        # ---- agents owned by this rank ---------------------------------
        self.owner = [a % comm.world_size for a in range(num_robots)]
        self.local_agents: Dict[int, PGOAgent] = {}
        from .manifold import lifting_matrix
        YL = lifting_matrix(d, r)
        if self._soa:
            # global-frame initialization: an externally provided warm
            # start (ma.warm_start) wins; else L2 mode runs the SoA/GPU
            # chordal init (vectorized assembly + BSR-kernel PCG —
            # reference MultiRobotExample.cpp:185-202 semantics at SoA
            # scale); robust mode keeps the odometry dead-reckoning
            # prefix scan (reference PGOAgent.cpp:952-957 does not trust
            # loop closures before GNC).
            if getattr(measurements, "warm_start", None) is not None:
                T_chordal_pre = measurements.warm_start
            elif robust == RobustCostType.L2:
                from .chordal import chordal_initialization_soa
                T_chordal_pre = chordal_initialization_soa(
                    measurements, num_poses, device=device,
                    tol=1e-6, max_iters=500)
            else:
                from .measurements import odometry_initialization_array
                odo_mask = (measurements.p1 + 1 == measurements.p2)
                T_chordal_pre = odometry_initialization_array(
                    d, num_poses, measurements.select(odo_mask))
        else:
            T_chordal_pre = chordal_initialization(
                d, num_poses, measurements) \
                if robust == RobustCostType.L2 else None
        Tg3 = None
        if T_chordal_pre is not None:
            Tg3 = np.ascontiguousarray(
                T_chordal_pre.reshape(d, num_poses, self.dh).transpose(
                    1, 0, 2))  # (n, d, dh) for fast slicing
        for rb in range(num_robots):
            if self.owner[rb] != comm.rank:
                continue
            p = PGOAgentParams(d=d, r=r, num_robots=num_robots,
                               acceleration=acceleration,
                               robust_cost_type=robust,
                               verbose=verbose, device=device,
                               inner_tol=inner_tol,
                               tr_max_iterations=tr_max_iterations)
            a = PGOAgent(rb, p)
            a.set_lifting_matrix(YL)
            T_init = None
            if Tg3 is not None:
                gidx = np.array([self.pose_to_index[(rb, i)]
                                 for i in range(self.pose_counts[rb])],
                                dtype=np.int64)
                T_init = np.ascontiguousarray(
                    Tg3[gidx].transpose(1, 0, 2).reshape(
                        d, self.pose_counts[rb] * self.dh))
            if self._soa:
                a.set_pose_graph_arrays(odometry[rb], private_lc[rb],
                                        shared_lc[rb], T_init=T_init)
            else:
                a.set_pose_graph(odometry[rb], private_lc[rb],
                                 shared_lc[rb], T_init=T_init)
            self.local_agents[rb] = a

        # ---- public-pose packing layout (global, same on all ranks) ----
        # For each agent: sorted local public pose indices. An agent's
        # packed payload = [status(6) | anchor(dh*r if agent 0) |
        #                   pub poses (dh*r each) | aux poses if accel].
        self.pub_idx: List[List[int]] = []
        pub_sets = self._public_pose_sets(shared_lc, num_robots)
        for rb in range(num_robots):
            self.pub_idx.append(sorted(pub_sets[rb]))
        blk = self.dh * self.r
        self.payload_sizes = []
        for rb in range(num_robots):
            sz = STATUS_LEN + len(self.pub_idx[rb]) * blk
            if rb == 0:
                sz += blk  # anchor = agent 0 pose 0
            if acceleration:
                sz += len(self.pub_idx[rb]) * blk
            self.payload_sizes.append(sz)
        self.rank_agents = [
            [rb for rb in range(num_robots) if self.owner[rb] == rk]
            for rk in range(comm.world_size)]
        self.rank_sizes = [
            sum(self.payload_sizes[rb] for rb in agents)
            for agents in self.rank_agents]

        # ---- centralized chordal init (identical on all ranks) ---------
        T_chordal = T_chordal_pre
        if T_chordal is not None:
            X_chordal = YL @ T_chordal
            for rb, a in self.local_agents.items():
                Xr = np.zeros((r, self.pose_counts[rb] * self.dh))
                for i in range(self.pose_counts[rb]):
                    g = self.pose_to_index[(rb, i)]
                    Xr[:, i * self.dh:(i + 1) * self.dh] = \
                        X_chordal[:, g * self.dh:(g + 1) * self.dh]
                a.set_x(Xr)

        # coloring for the colored schedule
        from .measurements import MeasurementArray as _MA
        nbrs = [set() for _ in range(num_robots)]
        for rb in range(num_robots):
            sl = shared_lc[rb]
            if isinstance(sl, _MA):
                if len(sl):
                    o = np.where(sl.r1 == rb, sl.r2, sl.r1)
                    nbrs[rb] = {int(x) for x in np.unique(o)}
            else:
                for m in sl:
                    o = m.r2 if m.r1 == rb else m.r1
                    nbrs[rb].add(o)
        colors = [-1] * num_robots
        for rb in range(num_robots):
            used = {colors[o] for o in nbrs[rb] if colors[o] >= 0}
            c = 0
            while c in used:
                c += 1
            colors[rb] = c
        self._colors = colors
        self._num_colors = max(colors) + 1 if colors else 1

    def _robust_is_l2(self) -> bool:
        return all(a.params.robust_cost_type == RobustCostType.L2
                   for a in self.local_agents.values())

    @staticmethod
    def _public_pose_sets(shared_lc, num_robots):
        from .measurements import MeasurementArray
        pub = [set() for _ in range(num_robots)]
        for rb in range(num_robots):
            sl = shared_lc[rb]
            if isinstance(sl, MeasurementArray):
                if len(sl):
                    local = np.where(sl.r1 == rb, sl.p1, sl.p2)
                    pub[rb] = {int(x) for x in np.unique(local)}
            else:
                for m in sl:
                    if m.r1 == rb:
                        pub[rb].add(m.p1)
                    else:
                        pub[rb].add(m.p2)
        return pub

    # -------------------------------------------------------------------
    def _pack_rank_payload(self) -> Tensor:
        blk = self.dh * self.r
        parts = []
        for rb in self.rank_agents[self.comm.rank]:
            a = self.local_agents[rb]
            st = a.get_status()
            parts.append(torch.from_numpy(st.as_vector()))
            if a.state == PGOAgentState.INITIALIZED:
                Xb = a.X.view(a.n, self.dh, self.r)
                idx = torch.tensor(self.pub_idx[rb], dtype=torch.int64,
                                   device=a.X.device)
                pubs = Xb.index_select(0, idx).reshape(-1).cpu()
                anchor = Xb[0].reshape(-1).cpu() if rb == 0 else None
                aux = None
                if self.acceleration and a.Y is not None:
                    Yb = a.Y.view(a.n, self.dh, self.r)
                    aux = Yb.index_select(0, idx).reshape(-1).cpu()
            else:
                pubs = torch.zeros(len(self.pub_idx[rb]) * blk,
                                   dtype=torch.float64)
                anchor = torch.zeros(blk, dtype=torch.float64) \
                    if rb == 0 else None
                aux = torch.zeros_like(pubs) if self.acceleration else None
            if anchor is not None:
                parts.append(anchor)
            parts.append(pubs)
            if self.acceleration:
                parts.append(aux)
        return torch.cat(parts) if parts else torch.zeros(0, dtype=torch.float64)

    def _eval_scalars(self, a: PGOAgent) -> np.ndarray:
        if a.state != PGOAgentState.INITIALIZED or a.problem is None:
            return np.zeros(3)
        ok = True
        if a.has_shared_lc():
            ok = a._construct_g(a.neighbor_pose_dict)
        if not ok:
            return np.zeros(3)
        f = a.problem.f(a.X)
        half_xg = 0.5 * float((a.X * a.problem._g()).sum())
        gn2 = float(torch.linalg.norm(a.problem.rie_grad(a.X))) ** 2
        return np.array([f, half_xg, gn2])

    def _unpack_and_update(self, flats: List[Tensor]):
        """Distribute gathered public poses/statuses to local agents."""
        blk = self.dh * self.r
        statuses: Dict[int, PGOAgentStatus] = {}
        anchor = None
        pose_payloads: Dict[int, Tuple[np.ndarray, Optional[np.ndarray]]] = {}
        for rk, flat in enumerate(flats):
            f = flat.cpu().numpy()
            off = 0
            for rb in self.rank_agents[rk]:
                st = PGOAgentStatus.from_vector(f[off:off + STATUS_LEN])
                off += STATUS_LEN
                statuses[rb] = st
                if rb == 0:
                    anchor = f[off:off + blk].reshape(self.dh, self.r)
                    off += blk
                npub = len(self.pub_idx[rb])
                pubs = f[off:off + npub * blk].reshape(npub, self.dh, self.r)
                off += npub * blk
                aux = None
                if self.acceleration:
                    aux = f[off:off + npub * blk].reshape(npub, self.dh,
                                                          self.r)
                    off += npub * blk
                pose_payloads[rb] = (pubs, aux)
        # update local agents' neighbor caches
        for rb, a in self.local_agents.items():
            for nb in a.get_neighbors():
                st = statuses[nb]
                a.set_neighbor_status(st)
                pubs, aux = pose_payloads[nb]
                pd = {(nb, p): pubs[k].T.copy()
                      for k, p in enumerate(self.pub_idx[nb])}
                a.update_neighbor_poses(nb, pd)
                if self.acceleration and aux is not None \
                        and st.state == PGOAgentState.INITIALIZED:
                    ad = {(nb, p): aux[k].T.copy()
                          for k, p in enumerate(self.pub_idx[nb])}
                    a.update_aux_neighbor_poses(nb, ad)
            if anchor is not None and np.any(anchor):
                a.set_global_anchor(np.ascontiguousarray(anchor.T))
        return statuses

    def _dict_sync_weights(self) -> None:
        """All ranks: collect owner-computed GNC weights of inter-robot
        loop closures and apply them on both co-owners (one all-reduce of
        n_cross doubles; mirrors the packed path's _w_exch)."""
        w = torch.zeros(self._n_cross, dtype=torch.float64)
        for rb, a in self.local_agents.items():
            cm, own = self._cross_map[rb], self._cross_owned[rb]
            if not len(cm):
                continue
            ws = np.array([m.weight for m in a.shared_lc])
            sel = np.nonzero(own)[0]
            if len(sel):
                w[torch.from_numpy(cm[sel])] = torch.from_numpy(ws[sel])
        self.comm.all_reduce_sum_(w)
        wn = w.numpy()
        for rb, a in self.local_agents.items():
            cm = self._cross_map[rb]
            for k, m in enumerate(a.shared_lc):
                m.weight = float(wn[cm[k]])
            a.publish_weights_requested = False

    def _evaluate(self) -> Tuple[float, np.ndarray]:
        """Centralized cost + per-agent centralized block grad-norm^2,
        from fresh neighbor data: one all-reduce of num_robots+1 fp64."""
        vec = torch.zeros(self.num_robots + 1, dtype=torch.float64)
        for rb, a in self.local_agents.items():
            ev = self._eval_scalars(a)
            vec[0] += ev[0] - ev[1]
            vec[1 + rb] = ev[2]
        self.comm.all_reduce_sum_(vec)
        v = vec.numpy()
        return float(v[0]), v[1:]

    def run(self, max_iters: int = 1000, gradnorm_tol: float = 0.1,
            trace_file: Optional[str] = None,
            time_limit_s: Optional[float] = None) -> RBCDResult:
        import torch
        if (str(self.device).startswith("cuda")
                and (self._robust_is_l2() or self._soa)
                and all(a.state == PGOAgentState.INITIALIZED
                        for a in self.local_agents.values())):
            return self._run_packed(max_iters, gradnorm_tol, trace_file,
                                    time_limit_s)
        res = RBCDResult()
        selected = 0
        t0 = time.perf_counter()
        # round 0 pre-exchange so G terms exist before the first solve
        flats = self.comm.all_gather_flat(self._pack_rank_payload(),
                                          self.rank_sizes)
        self._unpack_and_update(flats)
        del flats
        fout = open(trace_file, "w") if (trace_file and
                                         self.comm.rank == 0) else None
        # the reference checks convergence BEFORE the first iteration
        # (MultiRobotExample.cpp:302-305): an already-converged start
        # reports 0 iterations (e.g. kitti_08)
        if gradnorm_tol > 0.0:
            cost0, gn20 = self._evaluate()
            gradnorm0 = float(np.sqrt(gn20.sum()))
            if gradnorm0 < gradnorm_tol:
                res.converged = True
                res.final_cost = 2.0 * cost0
                res.final_gradnorm = gradnorm0
                if fout:
                    fout.close()
                res.elapsed_s = time.perf_counter() - t0
                return res
        for it in range(max_iters):
            if self.selection == "colored":
                color = it % self._num_colors
                active = [rb for rb in range(self.num_robots)
                          if self._colors[rb] == color]
            elif self.selection == "colored_greedy":
                # activate the whole independent set (color) that
                # contains the max-gradient agent: greedy's targeting
                # with colored's concurrent block updates
                color = self._colors[selected]
                active = [rb for rb in range(self.num_robots)
                          if self._colors[rb] == color]
            elif self.selection == "round_robin":
                active = [it % self.num_robots]
            else:
                active = [selected]
            for rb, a in self.local_agents.items():
                a.iterate(rb in active)
            if self.robust != RobustCostType.L2 and not self._soa \
                    and self._n_cross:
                self._dict_sync_weights()
            flats = self.comm.all_gather_flat(self._pack_rank_payload(),
                                              self.rank_sizes)
            self._unpack_and_update(flats)
            cost, gn2 = self._evaluate()
            gradnorm = float(np.sqrt(gn2.sum()))
            res.trace.append((2.0 * cost, gradnorm))
            if fout:
                fout.write(f"{2.0 * cost:.10g},{gradnorm:.10g}\n")
            if self.verbose and self.comm.rank == 0:
                print(f"iter {it} active {active} cost {2 * cost:.5g} "
                      f"gn {gradnorm:.5g}")
            res.iterations = it + 1
            if gradnorm < gradnorm_tol:
                res.converged = True
                break
            if time_limit_s and time.perf_counter() - t0 > time_limit_s:
                break
            if self.selection != "round_robin":
                selected = int(np.argmax(gn2))
        if res.trace:
            res.final_cost, res.final_gradnorm = res.trace[-1]
        if fout:
            fout.close()
        res.elapsed_s = time.perf_counter() - t0
        return res

    # ===================================================================
    # Packed fast path (GPU, L2 cost, all agents initialized):
    # device-tensor payloads end-to-end, native C++ solve + eval, one
    # host sync per round. Used automatically by run() when applicable.
    # ===================================================================
    def _packed_setup(self):
        if getattr(self, "_pk", None) is not None:
            return
        import torch
        dev = torch.device(self.device)
        blk = self.dh * self.r
        pk = {}
        pk["rank_payload_len"] = [
            sum(len(self.pub_idx[rb]) * blk for rb in agents)
            for agents in self.rank_agents]
        # global offsets: agent rb's block inside its owner's payload
        glob_off = {}
        for rk, agents in enumerate(self.rank_agents):
            o = 0
            for rb in agents:
                glob_off[rb] = o
                o += len(self.pub_idx[rb]) * blk
        pk["glob_off"] = glob_off

        # ---- consolidated per-rank buffers --------------------------
        # One X buffer and one neighbor buffer for all local agents:
        # pack/scatter become single index ops, and each agent's tensors
        # are stable contiguous views (the hipGraph keys stay valid).
        my = self.rank_agents[self.comm.rank]
        for rb in my:
            self.local_agents[rb]._ensure_packed(dev)
        pose_off = {}
        off = 0
        for rb in my:
            pose_off[rb] = off
            off += self.local_agents[rb].n
        rank_X = torch.zeros(off * self.dh, self.r, dtype=torch.float64,
                             device=dev)
        for rb in my:
            a = self.local_agents[rb]
            o = pose_off[rb] * self.dh
            rank_X[o:o + a.n * self.dh].copy_(a.X)
            a.X = rank_X[o:o + a.n * self.dh]
            if a.params.acceleration:
                a.Y = a.X.clone()
                a.V = a.X.clone()
        pk["rank_X"] = rank_X
        slot_off = {}
        soff = 0
        for rb in my:
            slot_off[rb] = soff
            soff += max(len(self.local_agents[rb]._nbr_slot_order), 1)
        rank_nbr = torch.zeros(max(soff, 1), self.dh, self.r,
                               dtype=torch.float64, device=dev)
        rank_nbr_aux = torch.zeros_like(rank_nbr)
        for rb in my:
            a = self.local_agents[rb]
            ns = max(len(a._nbr_slot_order), 1)
            a._nbr_buffer = rank_nbr[slot_off[rb]:slot_off[rb] + ns]
            a._nbr_buffer_aux = rank_nbr_aux[slot_off[rb]:slot_off[rb] + ns]
        pk["rank_nbr"] = rank_nbr
        pk["rank_nbr_aux"] = rank_nbr_aux

        # single pack index: global pose-block rows of every public pose
        pack_rows = []
        for rb in my:
            for p in self.pub_idx[rb]:
                pack_rows.append(pose_off[rb] + p)
        pk["pack_rows"] = torch.tensor(pack_rows, dtype=torch.int64,
                                       device=dev)
        # scatter: one (src positions -> nbr slots) pair per SOURCE rank
        pub_payload_pos = {}  # (nb agent, pose) -> index in owner payload
        for rb in range(self.num_robots):
            base = glob_off[rb] // blk
            for k, p in enumerate(self.pub_idx[rb]):
                pub_payload_pos[(rb, p)] = base + k
        per_src = {}
        for rb in my:
            a = self.local_agents[rb]
            for slot, (nb, p) in enumerate(a._nbr_slot_order):
                rk = self.owner[nb]
                per_src.setdefault(rk, ([], []))
                per_src[rk][0].append(pub_payload_pos[(nb, p)])
                per_src[rk][1].append(slot_off[rb] + slot)
        pk["scatter_rank"] = {
            rk: (torch.tensor(src, dtype=torch.int64, device=dev),
                 torch.tensor(dst, dtype=torch.int64, device=dev))
            for rk, (src, dst) in per_src.items()}
        pk["dev"] = dev
        self._pk = pk

    def _packed_pack(self, use_aux: bool = False):
        import torch
        pk = self._pk
        if not self.rank_agents[self.comm.rank]:
            return torch.zeros(0, dtype=torch.float64, device=pk["dev"])
        if use_aux:
            parts = []
            for rb in self.rank_agents[self.comm.rank]:
                a = self.local_agents[rb]
                src = a.Y if a.Y is not None else a.X
                n_pub = len(self.pub_idx[rb])
                Xb = src.view(a.n, self.dh, self.r)
                idx = torch.tensor(self.pub_idx[rb], dtype=torch.int64,
                                   device=pk["dev"])
                parts.append(Xb.index_select(0, idx).reshape(-1))
            return torch.cat(parts) if parts else torch.zeros(
                0, dtype=torch.float64, device=pk["dev"])
        Xb = pk["rank_X"].view(-1, self.dh, self.r)
        return Xb.index_select(0, pk["pack_rows"]).reshape(-1)

    def _packed_scatter(self, flats, aux: bool = False):
        pk = self._pk
        blk = self.dh * self.r
        target = pk["rank_nbr_aux"] if aux else pk["rank_nbr"]
        for rk, (src_idx, dst_slots) in pk["scatter_rank"].items():
            blkv = flats[rk].view(-1, self.dh, self.r)
            target.index_copy_(0, dst_slots, blkv.index_select(0, src_idx))

    def _packed_exchange_xy(self, sizes):
        """Accelerated mode: ONE all-gather carries both the X public
        poses and the Nesterov aux poses Y ([X | Y] halves per rank),
        instead of two collectives per round (round-1 VERDICT item 9).
        Callers must have run _packed_nesterov_pre so Y is current."""
        import torch
        payload = torch.cat([self._packed_pack(),
                             self._packed_pack(use_aux=True)])
        sizes2 = [2 * s for s in sizes]
        flats = self.comm.all_gather_flat(payload, sizes2)
        self._packed_scatter([f[:s] for f, s in zip(flats, sizes)])
        self._packed_scatter([f[s:] for f, s in zip(flats, sizes)],
                             aux=True)

    def _packed_eval_phase(self, evalmat):
        """Per-round evaluation of every local agent. The whole phase
        (zeroing + every agent's G assembly + cost/gradient kernels,
        writing directly into evalmat rows) is captured once into a
        single hipGraph via torch.cuda.graph and replayed per round;
        re-captured when an agent's problem pointers change (GNC Q
        rebuild)."""
        import torch
        modes = getattr(self, "_eval_env_modes", None)
        if modes is None:
            # env decisions are fixed per driver (read once, not per round)
            import os as _os
            _sync_eval_default = ("0" if self.selection in (
                "colored", "colored_greedy") else "1")
            modes = self._eval_env_modes = (
                _os.environ.get("DPO_DRIVER_EVAL_GRAPH", "0") == "1",
                _os.environ.get("DPO_SYNC_EVAL", _sync_eval_default) == "1")
        driver_graph, sync_eval = modes
        gen = sum(getattr(a, "_packed_generation", 0)
                  for a in self.local_agents.values())
        cache = getattr(self, "_eval_graph", None)
        if not driver_graph:
            # Default: per-agent hip-captured eval graphs (or the colored
            # schedule's per-stream fan-out). A single driver-level
            # torch-captured graph over all agents measured ~20% faster
            # but perturbs the evaluation run-to-run at the 1e-15 level,
            # which the greedy selection chaotically amplifies into
            # iteration-count scatter — parity with the reference's
            # deterministic trajectories wins. Opt in with
            # DPO_DRIVER_EVAL_GRAPH=1.
            if sync_eval:
                evalmat.zero_()
                for rb, a in self.local_agents.items():
                    a._packed_eval(out=None)
                    evalmat[rb] = a._dev_solver._eval_out
            elif getattr(self, "_group", None) is not None:
                # native fan-out: one C call + one gather kernel
                if not self._group_all:
                    evalmat.zero_()
                self._group.eval_all(evalmat)
            else:
                evalmat.zero_()
                for rb, a in self.local_agents.items():
                    a._packed_eval_async()
                for rb, a in self.local_agents.items():
                    evalmat[rb] = a._packed_eval_join()
            return
        # ROCm 7.2: a long-lived graph exec intermittently degrades after
        # a few hundred replays (stale node state; fresh capture of the
        # identical sequence is correct). Re-capture periodically —
        # amortized cost ~1%.
        if cache is not None and cache[2] >= 128:
            cache = None
        if cache is None or cache[1] != gen:
            g = torch.cuda.CUDAGraph()
            # warmup on a side stream (required before capture)
            side = torch.cuda.Stream()
            with torch.cuda.stream(side):
                evalmat.zero_()
                for rb, a in self.local_agents.items():
                    a._packed_eval(out=evalmat[rb])
            torch.cuda.current_stream().wait_stream(side)
            with torch.cuda.graph(g):
                evalmat.zero_()
                for rb, a in self.local_agents.items():
                    a._packed_eval(out=evalmat[rb])
            self._eval_graph = [g, gen, 0]
            cache = self._eval_graph
        cache[0].replay()
        cache[2] += 1

    def _run_packed(self, max_iters, gradnorm_tol, trace_file, time_limit_s):
        import torch
        res = RBCDResult()
        self._packed_setup()
        pk = self._pk
        dev = pk["dev"]
        selected = 0
        t0 = time.perf_counter()
        # persistent eval buffers: stable pointers keep the native
        # gather/eval graph caches hot across repeated run() calls
        # (bench episodes)
        if getattr(self, "_evalmat", None) is None \
                or self._evalmat.device != dev:
            self._evalmat = torch.zeros(self.num_robots, 3,
                                        dtype=torch.float64, device=dev)
        evalmat = self._evalmat
        evalmat.zero_()
        sizes = pk["rank_payload_len"]
        accel = self.acceleration
        # Nesterov host-side scalars per agent
        if accel:
            for a in self.local_agents.values():
                a.gamma = 0.0
                a.alpha = 0.0
            # round-0 pre-exchange carries X and the first Y in one
            # collective; per-round exchanges then stay single too
            for a in self.local_agents.values():
                a._packed_nesterov_pre()
            self._packed_exchange_xy(sizes)
        else:
            flats = self.comm.all_gather_flat(self._packed_pack(), sizes)
            self._packed_scatter(flats)
        fout = open(trace_file, "w") if (trace_file and
                                         self.comm.rank == 0) else None
        robust_mode = not self._robust_is_l2()
        if robust_mode:
            for a in self.local_agents.values():
                a._packed_gnc_setup(dev)
            self._w_exch = torch.zeros(self._n_cross, dtype=torch.float64,
                                       device=dev)
            self._cross_map_t = {
                rb: torch.from_numpy(self._cross_map[rb]).to(dev)
                for rb in self.local_agents}
            self._cross_owned_t = {
                rb: torch.from_numpy(
                    np.nonzero(self._cross_owned[rb])[0]).to(dev)
                for rb in self.local_agents}
        if not hasattr(self, "_round_counter"):
            self._round_counter = 0
        inner = next(iter(self.local_agents.values())).params \
            .robust_opt_inner_iters if self.local_agents else 30
        # reference semantics: convergence is checked BEFORE the first
        # iteration (an already-converged start reports 0 iterations)
        if gradnorm_tol > 0.0:
            self._packed_eval_phase(evalmat)
            self.comm.all_reduce_sum_(evalmat)
            ev0 = evalmat.cpu().numpy()
            gradnorm0 = float(np.sqrt(ev0[:, 2].sum()))
            if gradnorm0 < gradnorm_tol:
                res.converged = True
                res.final_cost = 2.0 * float((ev0[:, 0] - ev0[:, 1]).sum())
                res.final_gradnorm = gradnorm0
                if fout:
                    fout.close()
                res.elapsed_s = time.perf_counter() - t0
                self._sync_anchor()
                return res
        # Pipelined evaluation readback: when no per-round host decision
        # is needed (no greedy selection, no gradient-norm stop test),
        # the per-round eval scalars land in a device-side ring and are
        # read back every `chunk` rounds — the eval kernels of round t
        # then overlap the solves of round t+1 instead of draining the
        # GPU at a .cpu() sync every round.
        import os as _os_
        _sync_solve = _os_.environ.get("DPO_SYNC_SOLVE", "0") == "1"
        _timing = _os_.environ.get("DPO_TIME_PHASES", "0") == "1"
        ph = [0.0] * 5  # solve-enqueue, solve-finish, pack+scatter,
        #                 eval-enqueue, ring-flush
        # native multi-agent fan-out (one C call per phase, fixed
        # pointers keep the per-agent hipGraph caches hot)
        if (str(dev).startswith("cuda") and self.local_agents
                and getattr(self, "_group", None) is None
                and _os_.environ.get("DPO_NO_GROUP", "0") != "1"):
            from .ops.hip_backend import DeviceGroup
            rbs = sorted(self.local_agents)
            self._group = DeviceGroup(
                [self.local_agents[rb]._dev_solver for rb in rbs],
                [self.local_agents[rb].X for rb in rbs],
                [self.local_agents[rb]._nbr_buffer for rb in rbs],
                rbs)
            self._group_lidx = {rb: i for i, rb in enumerate(rbs)}
            self._group_all = len(rbs) == self.num_robots
        group = getattr(self, "_group", None)
        use_group_solve = (group is not None and not accel
                           and not _sync_solve)
        inner_tol = next(iter(self.local_agents.values())) \
            .params.inner_tol if self.local_agents else 1e-2
        tr_steps = next(iter(self.local_agents.values())) \
            .params.tr_max_iterations if self.local_agents else 1
        if self.selection in ("colored", "colored_greedy"):
            color_active = [
                [rb for rb in range(self.num_robots)
                 if self._colors[rb] == c]
                for c in range(self._num_colors)]
            if group is not None:
                color_ids = [
                    group.ids([self._group_lidx[rb] for rb in ca
                               if rb in self.local_agents])
                    for ca in color_active]
        # Pipelined eval readback width. tol<=0: nothing to decide per
        # round -> deep ring. tol>0 with a selection that needs no
        # per-round argmax (colored): still pipeline, checking the
        # convergence test on ring flushes — the crossing ITERATION is
        # read exactly from the per-round ring entries, only the break
        # happens up to chunk-1 rounds later (extra monotone rounds).
        if _os_.environ.get("DPO_DRIVER_EVAL_GRAPH", "0") == "1":
            chunk = 1
        elif gradnorm_tol <= 0.0 and self.selection not in (
                "greedy", "colored_greedy"):
            chunk = 16
        elif self.selection == "colored":
            chunk = 8
        else:
            chunk = 1
        if chunk > 1:
            er = getattr(self, "_evalring", None)
            if er is None or er.shape[0] != chunk:
                self._evalring = torch.zeros(chunk, self.num_robots, 3,
                                             dtype=torch.float64,
                                             device=dev)
            evalring = self._evalring
            evalring.zero_()
        else:
            evalring = None
        pending = 0   # rounds evaluated but not yet read back

        def flush_ring():
            nonlocal pending
            if pending == 0:
                return
            slab = evalring[:pending]
            self.comm.all_reduce_sum_(slab)
            evh = slab.cpu().numpy()
            for k in range(pending):
                c_ = float((evh[k, :, 0] - evh[k, :, 1]).sum())
                g_ = float(np.sqrt(evh[k, :, 2].sum()))
                res.trace.append((2.0 * c_, g_))
                if fout:
                    fout.write(f"{2.0 * c_:.10g},{g_:.10g}\n")
            slab.zero_()
            pending = 0

        for it in range(max_iters):
            self._round_counter += 1
            if chunk > 1 and robust_mode \
                    and self._round_counter % inner == 0:
                flush_ring()
            if robust_mode and self._round_counter % inner == 0:
                # GNC re-weighting round (reference iterate():
                # shouldUpdateLoopClosureWeights -> update -> mu step)
                for rb, a in self.local_agents.items():
                    a._packed_update_weights()
                self._w_exch.zero_()
                for rb, a in self.local_agents.items():
                    own = self._cross_owned_t[rb]
                    if own.numel():
                        pos = self._cross_map_t[rb].index_select(0, own)
                        self._w_exch.index_copy_(
                            0, pos, a._w_shared_dev.index_select(0, own))
                self.comm.all_reduce_sum_(self._w_exch)
                for rb, a in self.local_agents.items():
                    if self._cross_map_t[rb].numel():
                        a._w_shared_dev.copy_(
                            self._w_exch.index_select(
                                0, self._cross_map_t[rb]))
                for rb, a in self.local_agents.items():
                    a._packed_rebuild_q()
                    a.robust_cost.update()
            if self.selection == "colored":
                active = color_active[it % self._num_colors]
            elif self.selection == "colored_greedy":
                active = color_active[self._colors[selected]]
            elif self.selection == "round_robin":
                selected = it % self.num_robots
                active = [selected]
            else:
                active = [selected]
            # (accelerated mode: Y for this round was computed and
            # exchanged together with X at the end of the previous round
            # — the reference pulls aux dicts right before the selected
            # iterate, MultiRobotExample.cpp:259-273, and nothing updates
            # X/V between our round-end pre() and this solve, so the
            # values are identical with half the collectives.)
            # Concurrent active agents' solves overlap on per-agent HIP
            # streams (data-flow fences in dpo_ops.hip order the cached
            # solve/eval graphs against this stream's pack/scatter; the
            # async path is bitwise-deterministic).
            if _timing:
                tA = time.perf_counter()
            if use_group_solve:
                if self.selection == "colored":
                    gids = color_ids[it % self._num_colors]
                elif self.selection == "colored_greedy":
                    gids = color_ids[self._colors[selected]]
                elif selected in self.local_agents:
                    gids = group.ids([self._group_lidx[selected]])
                else:
                    gids = group.ids([])
                if _timing:
                    tB = time.perf_counter()
                if len(gids):
                    for _k in range(tr_steps):
                        group.solve_start(gids, tol=inner_tol)
                        group.solve_finish(gids)
            else:
                for rb, a in self.local_agents.items():
                    if accel and rb not in active:
                        a.X.copy_(a.Y)
                if _timing:
                    tB = time.perf_counter()
                if _sync_solve:
                    for rb, a in self.local_agents.items():
                        if rb in active:
                            a._packed_solve(accel)  # loops tr steps
                else:
                    for _k in range(tr_steps):
                        for rb in active:
                            if rb in self.local_agents:
                                self.local_agents[rb]._packed_solve_async(
                                    accel, first=(_k == 0))
                        for rb in active:
                            if rb in self.local_agents:
                                self.local_agents[rb]._packed_solve_finish()
            if _timing:
                tC = time.perf_counter()
            if accel:
                for a in self.local_agents.values():
                    a._packed_nesterov_post(it)
                for a in self.local_agents.values():
                    a._packed_nesterov_pre()
                self._packed_exchange_xy(sizes)
            else:
                flats = self.comm.all_gather_flat(self._packed_pack(),
                                                  sizes)
                self._packed_scatter(flats)
            if _timing:
                tD = time.perf_counter()
                ph[0] += tB - tA
                ph[1] += tC - tB
                ph[2] += tD - tC
            # evaluation (fresh neighbor data); agents fan out on their
            # own streams and join back before the packed reduce
            if chunk > 1:
                self._packed_eval_phase(evalring[pending])
                pending += 1
                res.iterations = it + 1
                if _timing:
                    ph[3] += time.perf_counter() - tD
                if pending == chunk or it + 1 == max_iters:
                    tE = time.perf_counter()
                    scan_from = len(res.trace)
                    flush_ring()
                    if _timing:
                        ph[4] += time.perf_counter() - tE
                    if gradnorm_tol > 0.0:
                        for k in range(scan_from, len(res.trace)):
                            if res.trace[k][1] < gradnorm_tol:
                                res.converged = True
                                res.iterations = k + 1
                                res.trace = res.trace[:k + 1]
                                break
                        if res.converged:
                            break
                    if time_limit_s and \
                            time.perf_counter() - t0 > time_limit_s:
                        break
                continue
            self._packed_eval_phase(evalmat)
            self.comm.all_reduce_sum_(evalmat)
            ev = evalmat.cpu().numpy()          # the round's one host sync
            cost = float((ev[:, 0] - ev[:, 1]).sum())
            gn2 = ev[:, 2]
            gradnorm = float(np.sqrt(gn2.sum()))
            res.trace.append((2.0 * cost, gradnorm))
            if fout:
                fout.write(f"{2.0 * cost:.10g},{gradnorm:.10g}\n")
            res.iterations = it + 1
            if gradnorm < gradnorm_tol:
                res.converged = True
                break
            if time_limit_s and time.perf_counter() - t0 > time_limit_s:
                break
            if self.selection != "round_robin":
                selected = int(np.argmax(gn2))
        if chunk > 1:
            flush_ring()   # safety net: all exits above should have flushed
        if _timing and self.comm.rank == 0 and res.iterations:
            k = res.iterations
            print("phase ms/round: solve_enq %.3f finish %.3f pack %.3f "
                  "eval %.3f flush %.3f" % tuple(p / k * 1e3 for p in ph))
        if res.trace:
            res.final_cost, res.final_gradnorm = res.trace[-1]
        if fout:
            fout.close()
        res.elapsed_s = time.perf_counter() - t0
        # sync dict-path state (global anchor for rounding)
        self._sync_anchor()
        return res

    def snapshot_initial_state(self) -> None:
        """Record every local agent's current iterate so repeated
        benchmark episodes can re-run the identical solve-to-target
        problem (bench.py). Call once after construction."""
        self._X0 = {rb: a.X.detach().clone()
                    for rb, a in self.local_agents.items()}

    def restore_initial_state(self) -> None:
        """Reset iterates (and Nesterov state) to the snapshot; the next
        run() starts from the same cold optimization state. Packed-path
        buffers are updated in place, so cached hipGraphs stay valid."""
        for rb, a in self.local_agents.items():
            a.X.copy_(self._X0[rb])
            if a.params.acceleration and a.Y is not None:
                a.Y.copy_(a.X)
                a.V.copy_(a.X)
                a.gamma = 0.0
                a.alpha = 0.0
            a.iteration_number = 0
        self._round_counter = 0

    def gather_final_trajectory(self):
        """Rounded global trajectory (d, (d+1) n) on rank 0 (reference
        PartitionInitial.cpp:329-335 output). Returns None on other
        ranks."""
        import torch
        self._sync_anchor()
        dh = self.dh
        buf = torch.zeros(self.n_global, self.d, dh, dtype=torch.float64)
        for rb, a in self.local_agents.items():
            T = a.get_trajectory_in_global_frame()
            if T is None:
                continue
            Tb = T.reshape(self.d, a.n, dh).transpose(1, 0, 2)
            for i in range(a.n):
                buf[self.pose_to_index[(rb, i)]] = torch.from_numpy(
                    np.ascontiguousarray(Tb[i]))
        self.comm.all_reduce_sum_(buf.view(-1))
        if self.comm.rank != 0:
            return None
        return np.ascontiguousarray(
            buf.numpy().transpose(1, 0, 2).reshape(self.d,
                                                   self.n_global * dh))

    def _sync_anchor(self):
        import torch
        blk = self.dh * self.r
        dev = self._pk["dev"] if getattr(self, "_pk", None) else "cpu"
        buf = torch.zeros(blk, dtype=torch.float64, device=dev)
        if 0 in self.local_agents:
            a0 = self.local_agents[0]
            buf.copy_(a0.X.view(a0.n, self.dh, self.r)[0].reshape(-1))
        self.comm.all_reduce_sum_(buf)
        M = buf.cpu().numpy().reshape(self.dh, self.r).T
        for a in self.local_agents.values():
            a.set_global_anchor(np.ascontiguousarray(M))
