#!/usr/bin/env python3
"""dpo_amd flagship benchmark — distributed RBCD pose-graph optimization.

Contract (driver-run):
  python bench.py --gpus N --steps K --warmup W
  For N > 1 the driver launches this under torch.distributed.run with one
  rank per GPU (RCCL over xGMI); rank/world read from the environment.

Workload: the reference's headline configuration (BASELINE.json metric
"wall-clock to target cost + iters-to-converge, sphere2500 ... at
1/2/4/8 GPUs") on a SYNTHETIC sphere2500-shaped SE(3) pose graph
(2500 poses, ~2x loop-closure density of sphere2500; there is no network
for datasets, so the graph is generated with the same shape/noise class
and random-init ground truth). 8 PGOAgents partitioned with the built-in
multilevel partitioner, r = 5, fp64 RBCD with the colored
(graph-colored block Gauss-Seidel) schedule — the framework's scalable
production schedule (non-adjacent agents solve concurrently; the
reference's greedy single-agent selection is available via
--selection greedy). One step = one synchronized RBCD round (the active
color's trust-region solves + boundary-pose all-gather +
centralized-gradient evaluation). The round schedule is identical for
every N (agents spread over ranks) => strong scaling; value = whole-job
rounds/s.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=60)
    ap.add_argument("--warmup", type=int, default=15)
    ap.add_argument("--poses", type=int, default=2500)
    ap.add_argument("--agents", type=int, default=8)
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--selection", type=str, default="colored")
    args = ap.parse_args()

    import torch

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_gpu = torch.cuda.is_available()
    if args.device:
        device = args.device
    elif use_gpu:
        device = f"cuda:{local_rank}"
    else:
        device = "cpu"
    if device.startswith("cuda"):
        torch.cuda.set_device(device)

    from dpo_amd.comm import init_from_env
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import sphere

    comm = init_from_env(device)

    # Synthetic sphere2500-shaped SE(3) graph, identical on all ranks.
    meas, n = sphere(n=args.poses, loops_per_pose=1.5, rot_noise=0.2,
                     tran_noise=0.3, seed=12345)

    # inner_tol=0 forces the local trust-region solver to run its full
    # tCG + acceptance sequence on every agent in every round, even once
    # the instance converges: the per-step work is state-independent, so
    # no work is ever skipped inside the timed region regardless of the
    # chosen step count.
    drv = DistributedRBCDDriver(
        meas, n, args.agents, comm, r=5, partition="multilevel",
        device=device, selection=args.selection, inner_tol=0.0)

    def sync():
        comm.barrier()
        if device.startswith("cuda"):
            torch.cuda.synchronize()

    # Settle device clocks/caches with ~1 s of throwaway rounds on a
    # SEPARATE driver instance (observed ~1-in-6 runs starting at half
    # throughput for tens of ms after init — a power/clock ramp).
    # The measured driver below still starts from the cold optimization
    # state, so the timed region's work is unchanged. Fixed iteration
    # count: all ranks must agree (collectives inside).
    if device.startswith("cuda"):
        settle = DistributedRBCDDriver(
            meas, n, args.agents, comm, r=5, partition="multilevel",
            device=device, selection=args.selection, inner_tol=0.0)
        settle.run(max_iters=1200, gradnorm_tol=0.0)
        del settle
        torch.cuda.synchronize()

    # warmup (untimed)
    drv.run(max_iters=args.warmup, gradnorm_tol=0.0)

    sync()
    t0 = time.perf_counter()
    res = drv.run(max_iters=args.steps, gradnorm_tol=0.0)
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if comm.world_size > 1:
        import torch.distributed as dist
        tt = t.to(device if device.startswith("cuda") else "cpu")
        dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        elapsed = float(tt.item())

    steps_done = res.iterations
    value = steps_done / elapsed
    if rank == 0:
        out = {
            "metric": "rbcd_rounds_per_s",
            "value": value,
            "unit": "rounds/s",
            "n_gpus": world if use_gpu else args.gpus,
            "steps": steps_done,
            "warmup": args.warmup,
            "ms_per_step": elapsed / steps_done * 1e3,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": "synthetic sphere2500-shaped SE(3) graph (no network; "
                    "random-init ground truth, same shape/noise class)",
            "config": {
                "model": "DPGO RBCD r=5 (colored block-Gauss-Seidel schedule)",
                "poses": n,
                "edges": len(meas),
                "agents": args.agents,
                "parallelism": f"rbcd-dist{world}",
                "final_cost": res.final_cost,
                "final_gradnorm": res.final_gradnorm,
                "selection": args.selection,
                "device": device,
            },
        }
        print(json.dumps(out))
    return 0


if __name__ == "__main__":
    sys.exit(main())
