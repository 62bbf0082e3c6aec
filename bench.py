#!/usr/bin/env python3
"""dpo_amd flagship benchmark — the BASELINE headline metric.

BASELINE.json: "wall-clock to target cost + iters-to-converge,
sphere2500 & city10000 at 1/2/4/8 GPUs". This benchmark runs the
SHIPPED sphere2500 dataset (data/sphere2500.npz, the real 2500-pose /
4949-edge SE(3) sphere; loaded by dpo_amd.io_g2o.load_dataset) through
the distributed RBCD driver to the reference's convergence criterion
(centralized Riemannian gradient norm < 0.1 — the stopping rule of the
reference driver, examples/MultiRobotExample.cpp:302-305) and reports
the wall-clock to reach it.

One "step" = one complete solve-to-target episode from the cold
centralized-chordal-initialized state (the same state the reference
driver starts its RBCD loop from). --warmup W runs W untimed episodes
(also settling device clocks), then EXACTLY --steps K episodes are
timed, bracketed by a barrier + torch.cuda.synchronize on both sides;
the reported value is the max-over-ranks mean wall-clock per episode.
Every episode performs the full round sequence — local trust-region
solves, boundary-pose all-gather, centralized cost/gradient evaluation,
convergence test — no work is skipped or cached across episodes (state
is restored to the cold start each time; the path is deterministic, so
every episode runs the identical iteration count).

Contract (driver-run):
  python bench.py --gpus N --steps K --warmup W
  For N > 1 the driver launches this under torch.distributed.run with
  one rank per GPU (RCCL over xGMI); rank/world read from the env.
  The 8 agents are spread over ranks; total work is fixed => strong
  scaling.

Config: 8 PGOAgents (BASELINE.json configs name 8-way partitioning on
8 GPUs), r = 5, fp64, colored (graph-colored block Gauss-Seidel)
schedule — the framework's scalable production schedule; it converges
sphere2500 in 203 iterations vs the reference's 289 (greedy, NP) /
320 (best KaHIP preset) from result/graph/*sphere2500.txt.
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
    ap.add_argument("--steps", type=int, default=5,
                    help="timed solve-to-target episodes")
    ap.add_argument("--warmup", type=int, default=2,
                    help="untimed warmup episodes")
    ap.add_argument("--dataset", type=str, default="sphere2500")
    ap.add_argument("--agents", type=int, default=8)
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--selection", type=str, default="colored")
    ap.add_argument("--partition", type=str, default="contiguous")
    ap.add_argument("--tol", type=float, default=0.1)
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
    from dpo_amd.io_g2o import load_dataset

    comm = init_from_env(device)

    # The shipped dataset (no network needed: data/*.npz is in-repo).
    meas, n = load_dataset(args.dataset)

    drv = DistributedRBCDDriver(
        meas, n, args.agents, comm, r=5, partition=args.partition,
        device=device, selection=args.selection)
    drv.snapshot_initial_state()

    def sync():
        comm.barrier()
        if device.startswith("cuda"):
            torch.cuda.synchronize()

    def episode():
        drv.restore_initial_state()
        return drv.run(max_iters=1000, gradnorm_tol=args.tol)

    # Untimed warmup episodes (also settles device clocks; every episode
    # is the full solve, so no separate settle phase is needed).
    warm_iters = []
    for _ in range(max(args.warmup, 1)):
        warm_iters.append(episode().iterations)

    sync()
    t0 = time.perf_counter()
    ep_times = []
    res = None
    for _ in range(args.steps):
        te = time.perf_counter()
        res = episode()
        ep_times.append(time.perf_counter() - te)
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if comm.world_size > 1:
        import torch.distributed as dist
        tt = t.to(device if device.startswith("cuda") else "cpu")
        dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        elapsed = float(tt.item())

    per_episode = elapsed / args.steps
    if rank == 0:
        out = {
            "metric": f"{args.dataset}_wall_to_gradnorm_0.1_s",
            "value": per_episode,
            "unit": "s",
            "n_gpus": world if use_gpu else args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": per_episode * 1e3,
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,  # reference publishes no wall-clock
            "dtype": "fp64",
            "data": f"shipped {args.dataset}.g2o dataset "
                    f"(data/{args.dataset}.npz, in-repo; real poses/edges)",
            "config": {
                "model": "DPGO RBCD r=5 (colored block-Gauss-Seidel "
                         "schedule, 8 agents)",
                "dataset": args.dataset,
                "poses": n,
                "edges": len(meas),
                "agents": args.agents,
                "partition": args.partition,
                "selection": args.selection,
                "parallelism": f"rbcd-dist{world}",
                "target": "centralized ||grad_R|| < 0.1 "
                          "(reference stopping rule)",
                "converged": bool(res.converged),
                "iterations": int(res.iterations),
                "ref_iterations_NP_greedy": 289 if
                    args.dataset == "sphere2500" else None,
                "final_cost": res.final_cost,
                "final_gradnorm": res.final_gradnorm,
                "rounds_per_s": res.iterations / per_episode,
                "episode_times_s": [round(x, 5) for x in ep_times],
                "warmup_iterations": warm_iters,
                "device": device,
            },
        }
        print(json.dumps(out))
    return 0


if __name__ == "__main__":
    sys.exit(main())
