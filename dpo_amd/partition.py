"""Pose-graph partitioning: contiguous chunks, partition files, and a
built-in multi-level partitioner (so the framework does not depend on an
offline KaHIP run, unlike the reference whose graph/<k>/<preset> files
were produced out-of-repo — SURVEY.md C18).

Multilevel scheme (KaHIP/METIS-style): heavy-edge-matching coarsening ->
greedy graph-growing initial partition -> boundary FM refinement at every
uncoarsening level. Quality target: the cut-edge reductions of
BASELINE.md (e.g. city10000 naive 33448 -> ~260 at k=5).
"""
from __future__ import annotations

import heapq
from typing import Dict, List, Sequence, Tuple

import numpy as np

from .types import RelativeSEMeasurement


def _native_multilevel(adj_lists, k, imbalance, seed, n_restarts):
    try:
        import ctypes
        import os
        lib_path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "ops", "hip", "libdpo_hip_ops.so")
        lib = ctypes.CDLL(lib_path)
        fn = lib.dpo_partition_multilevel
        fn.restype = ctypes.c_double
        fn.argtypes = [ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p,
                       ctypes.c_void_p, ctypes.c_int, ctypes.c_double,
                       ctypes.c_int, ctypes.c_uint, ctypes.c_void_p]
    except OSError:
        return None
    n = len(adj_lists)
    xadj = np.zeros(n + 1, dtype=np.int32)
    for u, a in enumerate(adj_lists):
        xadj[u + 1] = xadj[u] + len(a)
    adjncy = np.zeros(max(int(xadj[n]), 1), dtype=np.int32)
    for u, a in enumerate(adj_lists):
        adjncy[xadj[u]:xadj[u + 1]] = a
    out = np.zeros(n, dtype=np.int32)
    fn(n, xadj.ctypes.data, adjncy.ctypes.data, None, k,
       float(imbalance), int(n_restarts), int(seed) & 0xFFFFFFFF,
       out.ctypes.data)
    return [int(x) for x in out]


def contiguous_partition(num_poses: int, k: int) -> List[int]:
    """Naive sequential chunking (reference MultiRobotExample.cpp:93-110)."""
    per = num_poses // k
    part = []
    for i in range(num_poses):
        p = min(i // per, k - 1) if per > 0 else 0
        part.append(p)
    return part


def cut_edges(adj: Sequence[Sequence[int]], part: Sequence[int]) -> int:
    """Number of cut edge endpoints / 2 (undirected cut size)."""
    c = 0
    for u, nbrs in enumerate(adj):
        for v in nbrs:
            if part[u] != part[v]:
                c += 1
    return c // 2


class _Graph:
    """Weighted graph in adjacency form for the multilevel partitioner."""

    def __init__(self, adj: List[Dict[int, float]], vwgt: np.ndarray):
        self.adj = adj
        self.vwgt = vwgt
        self.n = len(adj)


def _build_graph(adj_lists: Sequence[Sequence[int]]) -> _Graph:
    adj = [dict((v, 1.0) for v in nbrs if v != u)
           for u, nbrs in enumerate(adj_lists)]
    return _Graph(adj, np.ones(len(adj_lists)))


def _coarsen(g: _Graph) -> Tuple[_Graph, np.ndarray]:
    """One heavy-edge-matching coarsening pass. Returns (coarse, map)."""
    n = g.n
    match = -np.ones(n, dtype=np.int64)
    # visit in random-ish but deterministic order: by degree ascending
    order = sorted(range(n), key=lambda u: len(g.adj[u]))
    for u in order:
        if match[u] >= 0:
            continue
        best, bw = -1, -1.0
        for v, w in g.adj[u].items():
            if match[v] < 0 and w > bw:
                best, bw = v, w
        if best >= 0:
            match[u] = best
            match[best] = u
        else:
            match[u] = u
    cmap = -np.ones(n, dtype=np.int64)
    nc = 0
    for u in range(n):
        if cmap[u] < 0:
            v = match[u]
            cmap[u] = nc
            cmap[v] = nc
            nc += 1
    cadj: List[Dict[int, float]] = [dict() for _ in range(nc)]
    cvwgt = np.zeros(nc)
    for u in range(n):
        cu = cmap[u]
        cvwgt[cu] += g.vwgt[u] if match[u] != u or True else g.vwgt[u]
    # fix double count: each coarse vertex counted once per fine vertex
    cvwgt = np.zeros(nc)
    for u in range(n):
        cvwgt[cmap[u]] += g.vwgt[u]
    for u in range(n):
        cu = cmap[u]
        for v, w in g.adj[u].items():
            cv = cmap[v]
            if cu == cv:
                continue
            cadj[cu][cv] = cadj[cu].get(cv, 0.0) + w
    return _Graph(cadj, cvwgt), cmap


def _initial_partition(g: _Graph, k: int, max_wgt: float,
                       seed: int = 1) -> np.ndarray:
    """Greedy graph growing: BFS regions from spread-out seeds."""
    n = g.n
    part = -np.ones(n, dtype=np.int64)
    rng = np.random.default_rng(seed)
    wgts = np.zeros(k)
    # pick k seeds: farthest-point style via BFS from random start
    seeds = [int(rng.integers(n))]
    for _ in range(1, k):
        dist = -np.ones(n, dtype=np.int64)
        q = list(seeds)
        for s in seeds:
            dist[s] = 0
        head = 0
        while head < len(q):
            u = q[head]; head += 1
            for v in g.adj[u]:
                if dist[v] < 0:
                    dist[v] = dist[u] + 1
                    q.append(v)
        dist[dist < 0] = 10 ** 9
        seeds.append(int(np.argmax(dist)))
    # grow regions with a priority on connection strength
    heaps: List[list] = [[] for _ in range(k)]
    for p, s in enumerate(seeds):
        part[s] = p
        wgts[p] += g.vwgt[s]
        for v, w in g.adj[s].items():
            heapq.heappush(heaps[p], (-w, v))
    active = True
    while active:
        active = False
        for p in np.argsort(wgts):  # grow lightest region first
            h = heaps[p]
            while h:
                negw, v = heapq.heappop(h)
                if part[v] >= 0:
                    continue
                if wgts[p] + g.vwgt[v] > max_wgt:
                    break
                part[v] = p
                wgts[p] += g.vwgt[v]
                for u, w in g.adj[v].items():
                    if part[u] < 0:
                        heapq.heappush(h, (-w, u))
                active = True
                break
    # stragglers (disconnected): assign to lightest part
    for u in range(n):
        if part[u] < 0:
            p = int(np.argmin(wgts))
            part[u] = p
            wgts[p] += g.vwgt[u]
    return part


def _fm_refine(g: _Graph, part: np.ndarray, k: int, max_wgt: float,
               passes: int = 8) -> None:
    """Boundary FM refinement: greedy positive-gain moves with balance."""
    wgts = np.zeros(k)
    for u in range(g.n):
        wgts[part[u]] += g.vwgt[u]
    for _ in range(passes):
        moved = 0
        # build gain heap over boundary vertices
        heap = []
        for u in range(g.n):
            pu = part[u]
            conn = np.zeros(k)
            for v, w in g.adj[u].items():
                conn[part[v]] += w
            ext = conn.copy()
            ext[pu] = -1.0
            best_p = int(np.argmax(ext))
            gain = conn[best_p] - conn[pu]
            if gain > 0 or (gain == 0 and wgts[pu] > wgts[best_p] + g.vwgt[u]):
                heap.append((-gain, u, best_p))
        heapq.heapify(heap)
        while heap:
            ngain, u, tp = heapq.heappop(heap)
            pu = part[u]
            if pu == tp:
                continue
            if wgts[tp] + g.vwgt[u] > max_wgt:
                continue
            # recompute gain (lazy heap)
            conn = np.zeros(k)
            for v, w in g.adj[u].items():
                conn[part[v]] += w
            gain = conn[tp] - conn[pu]
            if gain < -ngain - 1e-12:  # stale entry
                if gain > 0:
                    heapq.heappush(heap, (-gain, u, tp))
                continue
            if gain < 0:
                continue
            if gain == 0 and wgts[pu] <= wgts[tp] + g.vwgt[u]:
                continue
            part[u] = tp
            wgts[pu] -= g.vwgt[u]
            wgts[tp] += g.vwgt[u]
            moved += 1
            for v in g.adj[u]:
                pv = part[v]
                conn_v = np.zeros(k)
                for x, w in g.adj[v].items():
                    conn_v[part[x]] += w
                ext = conn_v.copy()
                ext[pv] = -1.0
                bp = int(np.argmax(ext))
                gv = conn_v[bp] - conn_v[pv]
                if gv > 0:
                    heapq.heappush(heap, (-gv, v, bp))
        if moved == 0:
            break


def multilevel_partition(adj_lists: Sequence[Sequence[int]], k: int,
                         imbalance: float = 0.05,
                         coarsen_to: int = 0,
                         seed: int = 1,
                         n_restarts: int = 0) -> List[int]:
    """Multi-level k-way partition of an undirected graph.

    Uses the native C++ partitioner (heavy-edge matching + graph
    growing + hill-climbing FM with rollback + connectivity fixup +
    iterated V-cycles and boundary-blob kicks, dpo_partition.cpp) when
    the extension is built; falls back to the pure-Python
    implementation below. n_restarts=0 picks a size-adaptive default:
    32 restarts for graphs up to 50k vertices (sub-second, best
    quality), 8 beyond (the V-cycle/kick machinery still runs)."""
    if k <= 1:
        return [0] * len(adj_lists)
    if n_restarts <= 0:
        n_restarts = 32 if len(adj_lists) <= 50000 else 8
    native = _native_multilevel(adj_lists, k, imbalance, seed, n_restarts)
    if native is not None:
        return native
    g0 = _build_graph(adj_lists)
    total = float(g0.vwgt.sum())
    max_wgt = (1.0 + imbalance) * total / k
    target = coarsen_to or max(20 * k, 80)

    levels: List[Tuple[_Graph, np.ndarray]] = []
    g = g0
    while g.n > target:
        gc, cmap = _coarsen(g)
        if gc.n >= g.n * 0.95:  # matching stalled
            break
        levels.append((g, cmap))
        g = gc

    part = _initial_partition(g, k, max_wgt, seed)
    _fm_refine(g, part, k, max_wgt)

    while levels:
        gf, cmap = levels.pop()
        part = part[cmap]
        _fm_refine(gf, part, k, max_wgt)
    return [int(p) for p in part]


def partition_measurements(
        measurements: Sequence[RelativeSEMeasurement],
        num_poses: int, part: Sequence[int], num_robots: int
        ) -> Tuple[List[List[RelativeSEMeasurement]],
                   List[List[RelativeSEMeasurement]],
                   List[List[RelativeSEMeasurement]],
                   Dict[int, Tuple[int, int]],
                   Dict[Tuple[int, int], int],
                   List[int]]:
    """Split a global-index dataset across robots given a pose->robot map.

    Local indices preserve global order within each robot (reference
    MultiRobotExample.cpp:76-151). Returns (odometry, private_lc,
    shared_lc, pose_map global->(robot, local), pose_to_index
    (robot, local)->global, pose_counts)."""
    pose_map: Dict[int, Tuple[int, int]] = {}
    pose_to_index: Dict[Tuple[int, int], int] = {}
    counts = [0] * num_robots
    for g in range(num_poses):
        rb = part[g]
        pid = (rb, counts[rb])
        pose_map[g] = pid
        pose_to_index[pid] = g
        counts[rb] += 1

    odometry: List[List[RelativeSEMeasurement]] = [[] for _ in range(num_robots)]
    private_lc: List[List[RelativeSEMeasurement]] = [[] for _ in range(num_robots)]
    shared_lc: List[List[RelativeSEMeasurement]] = [[] for _ in range(num_robots)]
    for mIn in measurements:
        src_r, src_i = pose_map[mIn.p1]
        dst_r, dst_i = pose_map[mIn.p2]
        m = RelativeSEMeasurement(src_r, dst_r, src_i, dst_i,
                                  mIn.R.copy(), mIn.t.copy(),
                                  mIn.kappa, mIn.tau, mIn.weight,
                                  mIn.is_known_inlier)
        if src_r == dst_r:
            if mIn.p1 + 1 == mIn.p2:
                odometry[src_r].append(m)
            else:
                private_lc[src_r].append(m)
        else:
            shared_lc[src_r].append(m)
            shared_lc[dst_r].append(m.copy())
    return odometry, private_lc, shared_lc, pose_map, pose_to_index, counts


def fixup_odometry_chains(
        odometry: List[List[RelativeSEMeasurement]],
        private_lc: List[List[RelativeSEMeasurement]],
        counts: Sequence[int]) -> None:
    """After an arbitrary partition, a robot's "odometry" edges (global
    i -> i+1 within one robot) are generally NOT a full local chain.
    PGOAgent's odometry invariants require p1 + 1 == p2 locally; edges
    violating that are reclassified as private loop closures, and agents
    may end up without a spanning chain. The reference sidesteps this by
    only feeding it contiguous or KaHIP partitions whose local order
    keeps chains intact; we reclassify for full generality."""
    for rb in range(len(odometry)):
        keep = []
        for m in odometry[rb]:
            if m.p1 + 1 == m.p2:
                keep.append(m)
            else:
                private_lc[rb].append(m)
        odometry[rb] = keep
