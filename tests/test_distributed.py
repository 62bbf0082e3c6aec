"""Multi-process distributed driver tests (gloo backend, CPU, world=2).
Validates that the RCCL-path code (one process per GPU in production)
produces the same trace as the single-process driver."""
import json
import os
import tempfile

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world, result_dir, mode):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29781"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from dpo_amd.comm import TorchDistComm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import grid3d
    from dpo_amd.types import RobustCostType

    meas, n = grid3d(side=4, seed=0)
    comm = TorchDistComm("cpu")
    sel = "colored" if mode == "accel_colored" else \
        ("greedy" if mode == "accel" else mode)
    drv = DistributedRBCDDriver(
        meas, n, 4, comm, r=5, partition="contiguous",
        selection=sel,
        acceleration=mode.startswith("accel"))
    res = drv.run(max_iters=250)
    if rank == 0:
        with open(os.path.join(result_dir, "res.json"), "w") as f:
            json.dump({"iters": res.iterations, "conv": res.converged,
                       "cost": res.final_cost,
                       "trace0": res.trace[0], "trace5": res.trace[5]}, f)
    dist.destroy_process_group()


def _run_world2(mode):
    return _run_world(2, mode)


def _run_world(world, mode):
    with tempfile.TemporaryDirectory() as td:
        mp.spawn(_worker, args=(world, td, mode), nprocs=world, join=True)
        with open(os.path.join(td, "res.json")) as f:
            return json.load(f)


def _run_single(mode):
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import grid3d
    meas, n = grid3d(side=4, seed=0)
    sel = "colored" if mode == "accel_colored" else \
        ("greedy" if mode == "accel" else mode)
    drv = DistributedRBCDDriver(
        meas, n, 4, Comm(), r=5, partition="contiguous",
        selection=sel,
        acceleration=mode.startswith("accel"))
    return drv.run(max_iters=250)


@pytest.mark.parametrize("mode", ["greedy", "colored"])
def test_world2_matches_single_process(mode):
    ref = _run_single(mode)
    out = _run_world2(mode)
    assert out["conv"] == ref.converged
    assert out["iters"] == ref.iterations
    assert abs(out["cost"] - ref.final_cost) < 1e-6 * max(1, abs(ref.final_cost))
    # early-trace agreement (bitwise-deterministic math on CPU)
    assert abs(out["trace5"][0] - ref.trace[5][0]) < 1e-8


def test_world2_accelerated():
    out = _run_world2("accel")
    assert out["conv"]


def test_world4_accel_colored_combined_payload():
    """World=4 (one agent per rank), accelerated + colored: exercises the
    folded [X|Y] single-collective exchange layout on the dict path and
    multi-rank agent spreading (8-GPU readiness)."""
    out = _run_world(4, "accel_colored")
    ref = _run_single("accel_colored")
    assert out["conv"] == ref.converged
    assert out["iters"] == ref.iterations
    assert abs(out["cost"] - ref.final_cost) \
        < 1e-6 * max(1, abs(ref.final_cost))


def test_bench_torchrun_world2_cpu():
    """bench.py under torch.distributed.run with 2 ranks (gloo, CPU):
    the exact launch path the driver's multi-GPU SCALE run uses. The
    episode must converge with the same iteration count as world=1."""
    import subprocess
    import sys
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29787", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--dataset", "smallGrid3D",
         "--agents", "4"],
        capture_output=True, text=True, timeout=600, cwd=here)
    assert out.returncode == 0, out.stderr[-2000:]
    o = json.loads(out.stdout.strip().splitlines()[-1])
    assert o["n_gpus"] == 2
    assert o["config"]["converged"] is True
    assert o["config"]["iterations"] == 56  # world=1 count (smallGrid3D)
