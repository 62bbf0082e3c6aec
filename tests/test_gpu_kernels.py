"""GPU kernel numerics tests: every HIP op is compared against the plain
PyTorch fp64 CPU reference implementation of the same op, plus
device-solver vs host-solver and end-to-end driver runs on the GPU."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _require_ext():
    from dpo_amd import ops
    assert ops.hip_available(), "HIP extension must load on a GPU box"
    return ops


@pytest.fixture(scope="module")
def grid_fixture():
    from dpo_amd.synthetic import grid3d
    from dpo_amd.quadratic import assemble_connection_laplacian
    meas, n = grid3d(side=4, seed=0)
    d, r = 3, 5
    Q = assemble_connection_laplacian(meas, n, d)
    g = torch.Generator().manual_seed(0)
    N = (d + 1) * n
    X = torch.randn(N, r, dtype=torch.float64, generator=g)
    from dpo_amd.manifold import LiftedSEManifold
    M = LiftedSEManifold(r, d, n)
    X = M.project(X)
    V = torch.randn(N, r, dtype=torch.float64, generator=g)
    return meas, n, d, r, Q, X, V


def test_bsr_spmm_matches_cpu(grid_fixture):
    _require_ext()
    meas, n, d, r, Q, X, V = grid_fixture
    ref = Q.spmm(V)
    Qd = Q.to(DEV)
    out = Qd.spmm(V.to(DEV)).cpu()
    assert torch.allclose(out, ref, atol=1e-10, rtol=1e-12)


def test_tangent_project_matches_cpu(grid_fixture):
    from dpo_amd.ops import cpu_ref, hip_backend
    meas, n, d, r, Q, X, V = grid_fixture
    ref = cpu_ref.tangent_project(X, V, d)
    out = hip_backend.tangent_project(X.to(DEV), V.to(DEV), d).cpu()
    assert torch.allclose(out, ref, atol=1e-12)


def test_stiefel_project_matches_cpu(grid_fixture):
    from dpo_amd.ops import cpu_ref, hip_backend
    meas, n, d, r, Q, X, V = grid_fixture
    ref = cpu_ref.stiefel_project(V, d)
    out = hip_backend.stiefel_project(V.to(DEV), d).cpu()
    # polar via analytic eig vs LAPACK SVD: small tolerance
    assert torch.allclose(out, ref, atol=1e-8)
    # orthonormality exactly
    n_poses = V.shape[0] // (d + 1)
    Ob = out.view(n_poses, d + 1, r)
    for i in range(n_poses):
        Yt = Ob[i, :d, :]
        assert torch.allclose(Yt @ Yt.T, torch.eye(d, dtype=torch.float64),
                              atol=1e-10)


def test_retract_matches_cpu(grid_fixture):
    from dpo_amd.ops import cpu_ref, hip_backend
    meas, n, d, r, Q, X, V = grid_fixture
    eta = 0.01 * V
    ref = cpu_ref.retract(X, eta, d)
    out = hip_backend.retract(X.to(DEV), eta.to(DEV), d).cpu()
    assert torch.allclose(out, ref, atol=1e-9)


def test_precond_dense_matches_cpu(grid_fixture):
    from dpo_amd.ops import hip_backend
    meas, n, d, r, Q, X, V = grid_fixture
    N = (d + 1) * n
    A = Q.to_scalar_csr().to_dense() + 0.1 * torch.eye(N, dtype=torch.float64)
    Minv64 = torch.cholesky_inverse(torch.linalg.cholesky(A))
    Minv = Minv64.to(torch.float32).contiguous().to(DEV)
    ref = (Minv64.to(torch.float32) @ V.to(torch.float32)).to(torch.float64)
    out = hip_backend.precond_dense(Minv, V.to(DEV)).cpu()
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-5)


def test_precond_jacobi_matches_cpu(grid_fixture):
    from dpo_amd.ops import cpu_ref, hip_backend
    meas, n, d, r, Q, X, V = grid_fixture
    dh = d + 1
    blocks = Q.diag_blocks() + 0.1 * torch.eye(dh, dtype=torch.float64)
    L = torch.linalg.cholesky(blocks)
    ref = torch.cholesky_solve(V.view(n, dh, r), L).reshape(-1, r)
    out = hip_backend.precond_jacobi(L.contiguous().to(DEV), V.to(DEV),
                                     dh).cpu()
    assert torch.allclose(out, ref, atol=1e-10)


def test_g_assemble_matches_cpu():
    from dpo_amd.synthetic import grid3d
    from dpo_amd.partition import contiguous_partition, partition_measurements
    from dpo_amd.quadratic import GAssembler
    meas, n = grid3d(side=4, seed=1)
    odo, priv, shared, *_ = partition_measurements(
        meas, n, contiguous_partition(n, 2), 2)
    r = 5
    sl = shared[0]
    n0 = max(max(m.p1 if m.r1 == 0 else m.p2 for m in sl) + 1,
             len([1 for m in meas]))  # upper bound fine for buffer size
    n0 = 40
    ep = [0 if m.r1 == 0 else 1 for m in sl]
    slots = list(range(len(sl)))
    ga = GAssembler(n0, 3, sl, ep, slots)
    gcpu = GAssembler(n0, 3, sl, ep, slots)
    gen = torch.Generator().manual_seed(3)
    nbr = torch.randn(len(sl), 4, r, dtype=torch.float64, generator=gen)
    w = torch.rand(len(sl), dtype=torch.float64, generator=gen)
    ref = gcpu.assemble(nbr, w, r)
    out = ga.assemble(nbr.to(DEV), w, r).cpu()
    assert torch.allclose(out, ref, atol=1e-12)


def test_device_solver_matches_host_solver(grid_fixture):
    """One RBCD local solve: device-resident tCG + shrink-replay must
    match the host fp64 implementation to tight tolerance."""
    from dpo_amd.ops.hip_backend import DeviceSolver
    from dpo_amd.quadratic import QuadraticProblem
    from dpo_amd.solver import QuadraticOptimizer, TRParams
    from dpo_amd.types import OptAlgorithm
    meas, n, d, r, Q, X, V = grid_fixture

    # host solve (jacobi preconditioner so both sides match exactly)
    ph = QuadraticProblem(n, d, r, precond="jacobi")
    ph.set_q(Q)
    tr = TRParams(tolerance=1e-2, initial_radius=100.0, max_iterations=1,
                  max_inner_iterations=10)
    opt = QuadraticOptimizer(ph, OptAlgorithm.RTR, tr)
    Xh = opt.optimize(X.clone())

    pd = QuadraticProblem(n, d, r, precond="jacobi")
    pd.set_q(Q.to(DEV))
    pd._Lpre = pd._Lpre.contiguous()
    ds = DeviceSolver(n, d, r, DEV, max_inner=10)
    Xd = X.clone().to(DEV)
    stats = ds.solve(pd, Xd, tol=1e-2, Delta0=100.0)
    assert abs(stats["f_init"] - opt.result.f_init) < 1e-6 * max(
        1, abs(opt.result.f_init))
    assert abs(stats["grad_norm_init"] - opt.result.grad_norm_init) < 1e-6
    assert abs(stats["f_opt"] - opt.result.f_opt) < 1e-5 * max(
        1, abs(opt.result.f_opt))
    assert torch.allclose(Xd.cpu(), Xh, atol=1e-6)


def test_driver_gpu_matches_cpu_trace():
    """5 iterations of the 2-robot RBCD loop on GPU vs CPU."""
    from dpo_amd.driver import MultiRobotDriver
    from dpo_amd.synthetic import grid3d
    meas, n = grid3d(side=4, seed=0)
    d_cpu = MultiRobotDriver(meas, n, 2, r=5, partition="contiguous")
    res_cpu = d_cpu.run(max_iters=5)
    d_gpu = MultiRobotDriver(meas, n, 2, r=5, partition="contiguous",
                             device=DEV)
    res_gpu = d_gpu.run(max_iters=5)
    for (c0, g0), (c1, g1) in zip(res_cpu.trace, res_gpu.trace):
        # different preconditioners (exact LU vs dense-inverse) may alter
        # the tCG path slightly; costs must track closely
        assert abs(c0 - c1) < 1e-4 * max(1.0, abs(c0))


def test_driver_gpu_converges():
    from dpo_amd.driver import MultiRobotDriver
    from dpo_amd.synthetic import grid3d
    meas, n = grid3d(side=4, seed=0)
    drv = MultiRobotDriver(meas, n, 2, r=5, partition="contiguous",
                           device=DEV)
    res = drv.run(max_iters=300)
    assert res.converged, f"gradnorm {res.final_gradnorm}"


def test_dist_driver_packed_gpu_matches_cpu():
    """DistributedRBCDDriver world=1: GPU packed fast path vs CPU dict
    path must produce matching cost traces and converge."""
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import grid3d
    meas, n = grid3d(side=4, seed=0)
    cpu = DistributedRBCDDriver(meas, n, 2, Comm(), r=5,
                                partition="contiguous")
    res_cpu = cpu.run(max_iters=200)
    gpu = DistributedRBCDDriver(meas, n, 2, Comm(), r=5,
                                partition="contiguous", device=DEV)
    res_gpu = gpu.run(max_iters=200)
    assert res_gpu.converged
    for (c0, _), (c1, _) in zip(res_cpu.trace[:20], res_gpu.trace[:20]):
        assert abs(c0 - c1) < 1e-4 * max(1.0, abs(c0))
    assert abs(res_gpu.final_cost - res_cpu.final_cost) < 1e-3 * max(
        1.0, abs(res_cpu.final_cost))


def test_dist_driver_packed_gpu_accelerated():
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import grid3d
    meas, n = grid3d(side=4, seed=0)
    gpu = DistributedRBCDDriver(meas, n, 2, Comm(), r=5,
                                partition="contiguous", device=DEV,
                                acceleration=True)
    res = gpu.run(max_iters=250)
    assert res.converged


def test_soa_robust_gnc_gpu():
    """SoA pipeline + packed robust GNC on GPU: planted outlier loop
    closures must be rejected (weight -> 0) and inliers kept."""
    import numpy as np
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import grid3d_soa
    from dpo_amd.types import RobustCostType
    ma, n = grid3d_soa(side=8, outlier_prob=0.15, seed=3,
                       rot_noise=0.02, tran_noise=0.01)
    outliers = ma.outlier_mask.copy()
    drv = DistributedRBCDDriver(
        ma, n, 2, Comm(), r=5, partition="contiguous",
        robust=RobustCostType.GNC_TLS, device=DEV)
    # accelerated GNC schedule for the small fixture
    for a in drv.local_agents.values():
        a.params.robust_opt_inner_iters = 10
        a.robust_cost.params.gnc_mu_step = 2.5
        a.robust_cost.params.gnc_init_mu = 1e-3
        a.robust_cost.reset()
    res = drv.run(max_iters=400, gradnorm_tol=0.05)
    # collect final weights mapped back to global edges
    # (2 agents, contiguous: use each agent's arrays)
    n_rej = n_rej_true = n_kept_inlier = n_inlier = 0
    for rb, a in drv.local_agents.items():
        w = a._all_weights_dev.cpu().numpy()
        n_odo = len(a._odo_ma)
        for k in range(len(a._priv_ma)):
            wk = w[n_odo + k]
            if wk < 0.5:
                n_rej += 1
        # crude: count kept weights
        lcw = w[n_odo:]
        n_inlier += int((lcw > 0.9).sum())
        n_rej_true += int((lcw < 0.1).sum())
    frac_out = outliers.mean()
    total_lc = sum(len(a._priv_ma) + len(a._shared_ma)
                   for a in drv.local_agents.values())
    # rejected fraction should be in the ballpark of the planted fraction
    assert n_rej_true > 0.5 * frac_out * total_lc, \
        f"rejected {n_rej_true} of ~{frac_out * total_lc:.0f} planted"
    assert n_inlier > 0.6 * total_lc
    # cost should be far below the unweighted initial cost
    assert res.trace[-1][0] < res.trace[0][0]


def test_soa_l2_gpu_matches_object_path():
    """SoA construction must produce the same optimization as the object
    path on an L2 run."""
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import grid3d, grid3d_soa
    from dpo_amd.measurements import MeasurementArray
    meas, n = grid3d(side=4, seed=0)
    ma = MeasurementArray.from_list(meas)
    obj = DistributedRBCDDriver(meas, n, 2, Comm(), r=5,
                                partition="contiguous", device=DEV)
    res_obj = obj.run(max_iters=100)
    soa = DistributedRBCDDriver(ma, n, 2, Comm(), r=5,
                                partition="contiguous", device=DEV)
    res_soa = soa.run(max_iters=100)
    # different initializations (chordal vs odometry-prefix for L2? both
    # use chordal path in object mode; SoA L2 also distributes odometry
    # init) — compare converged cost instead of traces
    assert abs(res_soa.final_cost - res_obj.final_cost) < 1e-2 * max(
        1.0, abs(res_obj.final_cost))


def test_async_optimization_loop_gpu():
    """Asynchronous per-agent optimization thread on the GPU path
    (reference testOptimizationThread semantics)."""
    import time
    from dpo_amd.agent import PGOAgent
    from dpo_amd.synthetic import triangle_graph
    from dpo_amd.types import PGOAgentParams
    import numpy as np
    meas, n, T_truth = triangle_graph()
    odo = [m for m in meas if m.p1 + 1 == m.p2]
    lc = [m for m in meas if m.p1 + 1 != m.p2]
    a = PGOAgent(0, PGOAgentParams(d=3, r=5, device=DEV))
    a.set_pose_graph(odo, lc, [])
    a.start_optimization_loop(100.0)
    time.sleep(0.5)
    a.end_optimization_loop()
    T = a.get_trajectory_in_local_frame()
    assert np.abs(T - T_truth).max() < 1e-4


def test_single_robot_batch_gpu_matches_cpu():
    """Full-batch RTR (r = d) on GPU vs CPU (reference
    SingleRobotExample path)."""
    from dpo_amd.agent import PGOAgent
    from dpo_amd.synthetic import grid3d
    from dpo_amd.types import PGOAgentParams
    meas, n = grid3d(side=3, seed=1)
    odo = [m for m in meas if m.p1 + 1 == m.p2]
    lc = [m for m in meas if m.p1 + 1 != m.p2]
    res = {}
    for dev in ("cpu", DEV):
        a = PGOAgent(0, PGOAgentParams(d=3, r=3, device=dev))
        a.set_pose_graph(odo, lc, [])
        a.local_pose_graph_optimization()
        res[dev] = a.last_opt_result
    assert abs(res[DEV].f_opt - res["cpu"].f_opt) < 1e-4 * max(
        1.0, abs(res["cpu"].f_opt))
    assert res[DEV].grad_norm_opt < 1e-1


def test_se2_kernels_match_cpu():
    """d=2 template instantiations (SE(2)) against the CPU reference."""
    from dpo_amd.ops import cpu_ref, hip_backend
    from dpo_amd.synthetic import city2d
    from dpo_amd.quadratic import assemble_connection_laplacian
    from dpo_amd.manifold import LiftedSEManifold
    meas, n = city2d(side=6, seed=0)
    d, r = 2, 5
    Q = assemble_connection_laplacian(meas, n, d)
    g = torch.Generator().manual_seed(0)
    N = (d + 1) * n
    M = LiftedSEManifold(r, d, n)
    X = M.project(torch.randn(N, r, dtype=torch.float64, generator=g))
    V = torch.randn(N, r, dtype=torch.float64, generator=g)
    assert torch.allclose(Q.to(DEV).spmm(V.to(DEV)).cpu(), Q.spmm(V),
                          atol=1e-10)
    assert torch.allclose(
        hip_backend.tangent_project(X.to(DEV), V.to(DEV), d).cpu(),
        cpu_ref.tangent_project(X, V, d), atol=1e-12)
    assert torch.allclose(
        hip_backend.stiefel_project(V.to(DEV), d).cpu(),
        cpu_ref.stiefel_project(V, d), atol=1e-8)


def test_dist_driver_se2_gpu():
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import city2d
    meas, n = city2d(side=8, seed=1)
    cpu = DistributedRBCDDriver(meas, n, 2, Comm(), r=5,
                                partition="contiguous")
    res_cpu = cpu.run(max_iters=400)
    gpu = DistributedRBCDDriver(meas, n, 2, Comm(), r=5,
                                partition="contiguous", device=DEV)
    res_gpu = gpu.run(max_iters=400)
    assert res_gpu.converged
    assert abs(res_gpu.final_cost - res_cpu.final_cost) < 1e-3 * max(
        1.0, abs(res_cpu.final_cost))


def test_rgd_gpu():
    """RGD algorithm on GPU (python solver over HIP ops)."""
    from dpo_amd.synthetic import grid3d
    from dpo_amd.agent import PGOAgent
    from dpo_amd.types import OptAlgorithm, PGOAgentParams
    # gentle precisions: RGD uses the reference's fixed 1e-3 stepsize
    # (QuadraticOptimizer.cpp:23), which overshoots on stiff problems
    # exactly as the reference's own monotonicity assert would.
    meas, n = grid3d(side=3, seed=0, rot_noise=0.2, tran_noise=0.1,
                     kappa=20.0, tau=10.0)
    odo = [m for m in meas if m.p1 + 1 == m.p2]
    lc = [m for m in meas if m.p1 + 1 != m.p2]
    a = PGOAgent(0, PGOAgentParams(d=3, r=5, algorithm=OptAlgorithm.RGD,
                                   device=DEV))
    a.set_pose_graph(odo, lc, [])
    f0 = a.problem.f(a.X)
    for _ in range(3):
        a.iterate(True)
    assert a.problem.f(a.X) <= f0 + 1e-9


def test_group_fanout_matches_per_agent_path(monkeypatch):
    """The native multi-agent fan-out (DpoGroup: batched fence
    signalling + single gather kernel) must produce the same
    optimization trajectory as the per-agent Python loop."""
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import sphere

    meas, n = sphere(n=600, loops_per_pose=1.0, rot_noise=0.1,
                     tran_noise=0.1, seed=3)

    def run(no_group):
        if no_group:
            monkeypatch.setenv("DPO_NO_GROUP", "1")
        else:
            monkeypatch.delenv("DPO_NO_GROUP", raising=False)
        drv = DistributedRBCDDriver(meas, n, 4, Comm(), r=5,
                                    partition="contiguous", device=DEV,
                                    selection="colored")
        return drv.run(max_iters=40, gradnorm_tol=0.0)

    a = run(False)
    b = run(True)
    assert a.iterations == b.iterations == 40
    # same trajectory up to fp-atomic reduction jitter
    for (ca, ga), (cb, gb) in zip(a.trace, b.trace):
        assert abs(ca - cb) < 1e-6 * max(1.0, abs(cb))
        assert abs(ga - gb) < 1e-4 * max(1.0, gb)


def test_greedy_inner_tol0_stable():
    """Regression (round-1 VERDICT weak #2): greedy selection with
    inner_tol=0 must stay numerically stable — the monotone-acceptance
    guard in the device solver must prevent divergence even when the
    solver runs its full tCG budget every round on a converged state."""
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import grid3d
    meas, n = grid3d(side=4, seed=0)
    drv = DistributedRBCDDriver(meas, n, 4, Comm(), r=5,
                                partition="contiguous",
                                selection="greedy", device=DEV,
                                inner_tol=0.0)
    res = drv.run(max_iters=400, gradnorm_tol=0.0)
    costs = [c for c, _ in res.trace]
    assert np.isfinite(res.final_gradnorm)
    assert res.final_gradnorm < 1.0, res.final_gradnorm
    assert costs[-1] <= costs[0] + 1e-9


def test_bench_episode_restore_deterministic():
    """snapshot/restore (bench.py episodes) reproduces the identical
    trajectory on the packed GPU path."""
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import grid3d
    meas, n = grid3d(side=4, seed=1)
    drv = DistributedRBCDDriver(meas, n, 4, Comm(), r=5,
                                partition="contiguous",
                                selection="colored", device=DEV)
    drv.snapshot_initial_state()
    r1 = drv.run(max_iters=1000, gradnorm_tol=0.1)
    drv.restore_initial_state()
    r2 = drv.run(max_iters=1000, gradnorm_tol=0.1)
    assert r1.iterations == r2.iterations
    # the first run's round-0 eval runs once eagerly (hipGraph capture
    # warmup) while the restored run replays the graph — identical math,
    # but reduction fan-in order can differ at the few-ULP level
    for a, b in (r1.trace[0], r2.trace[0]), (r1.trace[-1], r2.trace[-1]):
        assert abs(a[0] - b[0]) <= 1e-9 * max(1.0, abs(a[0]))
        assert abs(a[1] - b[1]) <= 1e-9 * max(1.0, abs(a[1]))


def test_hess_wide_matches_narrow():
    """k_hess_wide (element-per-thread, large-agent layout) must produce
    the same solve trajectory as k_hess_fused (thread-per-pose). Run in
    subprocesses because the path choice is cached from the env."""
    import json
    import os
    import subprocess
    import sys
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = (
        "import json\n"
        "from dpo_amd.comm import Comm\n"
        "from dpo_amd.dist_driver import DistributedRBCDDriver\n"
        "from dpo_amd.synthetic import grid3d\n"
        "meas, n = grid3d(side=6, seed=3)\n"
        "drv = DistributedRBCDDriver(meas, n, 2, Comm(), r=5,\n"
        "    partition='contiguous', selection='colored', device='cuda:0')\n"
        "res = drv.run(max_iters=40, gradnorm_tol=0.0)\n"
        "print(json.dumps({'cost': res.final_cost,\n"
        "                  'gn': res.final_gradnorm,\n"
        "                  't0': res.trace[0]}))\n")
    outs = {}
    for w in ("0", "1"):
        env = dict(os.environ, DPO_HESS_WIDE=w)
        r = subprocess.run([sys.executable, "-c", code],
                           capture_output=True, text=True, env=env,
                           cwd=here, timeout=600)
        assert r.returncode == 0, r.stderr[-2000:]
        outs[w] = json.loads(r.stdout.strip().splitlines()[-1])
    a, b = outs["0"], outs["1"]
    assert abs(a["t0"][0] - b["t0"][0]) < 1e-9 * max(1, abs(a["t0"][0]))
    assert abs(a["cost"] - b["cost"]) < 1e-8 * max(1, abs(a["cost"]))
    assert abs(a["gn"] - b["gn"]) < 1e-5 * max(1.0, abs(a["gn"]))


def test_chordal_soa_gpu_matches_cpu():
    """GPU chordal init (BSR-kernel PCG) matches the CPU run of the
    same algorithm and the direct reference-style solve."""
    from dpo_amd.chordal import (chordal_initialization,
                                 chordal_initialization_soa)
    from dpo_amd.synthetic import grid3d_soa
    ma, n = grid3d_soa(side=5, seed=2)
    T_cpu = chordal_initialization_soa(ma, n, device="cpu", tol=1e-11)
    T_gpu = chordal_initialization_soa(ma, n, device=DEV, tol=1e-11)
    assert np.abs(T_cpu - T_gpu).max() < 1e-6
    T_ref = chordal_initialization(3, n, ma.to_list())
    assert np.abs(T_gpu - T_ref).max() < 1e-6


def test_bsr_spmm_mfma_matches(grid_fixture):
    """fp64-MFMA grouped-ELL SpMM (A/B study kernel) must match the
    production BSR SpMM exactly (same fp64 FMA semantics)."""
    from dpo_amd.ops import hip_backend as hb
    meas, n, d, r, Q, X, V = grid_fixture
    Qd = Q.to(DEV)
    Xd = X.to(DEV)
    ref = Qd.spmm(Xd)
    gp, gc, gb = hb.build_spmm_mfma_groups(Qd.row_ptr, Qd.col_idx, n)
    out = hb.bsr_spmm_mfma(torch.from_numpy(gp).to(DEV),
                           torch.from_numpy(gc).to(DEV),
                           torch.from_numpy(gb).to(DEV),
                           Qd.vals, Xd)
    torch.cuda.synchronize()
    assert torch.allclose(out.cpu(), ref.cpu(), atol=1e-10)


def test_flagship_tr3_gpu_beats_reference_parking_garage():
    """GPU mirror of the CPU flagship regression: colored + multilevel
    + tr_max_iterations=3 must beat the reference's best parking-garage
    run (14 iterations) through the packed device path."""
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.io_g2o import load_dataset
    meas, n = load_dataset("parking-garage")
    drv = DistributedRBCDDriver(meas, n, 5, Comm(), r=5,
                                partition="multilevel",
                                selection="colored", device=DEV,
                                tr_max_iterations=3)
    res = drv.run(max_iters=100)
    assert res.converged
    assert res.iterations <= 14, res.iterations
