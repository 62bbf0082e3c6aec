#!/usr/bin/env python3
"""Headline parity runs: iterations-to-convergence (centralized
||grad_R|| < 0.1) and wall-clock on the shipped datasets, matching the
reference driver configuration (5 robots, r=5, greedy RBCD).

Usage: python scripts/headline.py --dataset sphere2500 --robots 5 \
          [--device cuda:0] [--partition contiguous|multilevel] \
          [--selection greedy|colored] [--accel] [--max-iters 1000]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset", required=True)
    ap.add_argument("--robots", type=int, default=5)
    ap.add_argument("--r", type=int, default=5)
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--partition", default="contiguous")
    ap.add_argument("--selection", default="greedy")
    ap.add_argument("--accel", action="store_true")
    ap.add_argument("--tr-iters", type=int, default=1,
                    help="trust-region steps per RBCD round (reference: 1)")
    ap.add_argument("--max-iters", type=int, default=1000)
    ap.add_argument("--tol", type=float, default=0.1)
    ap.add_argument("--trace", default=None)
    ap.add_argument("--dump-trajectory", default=None,
                    help="write the final rounded global trajectory as "
                         "CSV (reference PartitionInitial.cpp:329-335)")
    ap.add_argument("--driver", default="dist", choices=["dist", "local"])
    args = ap.parse_args()

    from dpo_amd.io_g2o import load_dataset
    from dpo_amd.comm import init_from_env

    meas, n = load_dataset(args.dataset)
    t_setup = time.perf_counter()
    if args.driver == "dist":
        from dpo_amd.dist_driver import DistributedRBCDDriver
        comm = init_from_env(args.device)
        drv = DistributedRBCDDriver(
            meas, n, args.robots, comm, r=args.r,
            partition=args.partition, acceleration=args.accel,
            device=args.device, selection=args.selection,
            tr_max_iterations=args.tr_iters)
    else:
        from dpo_amd.driver import MultiRobotDriver
        drv = MultiRobotDriver(
            meas, n, args.robots, r=args.r, partition=args.partition,
            acceleration=args.accel, device=args.device,
            selection=args.selection,
            tr_max_iterations=args.tr_iters)
    setup_s = time.perf_counter() - t_setup
    res = drv.run(max_iters=args.max_iters, gradnorm_tol=args.tol,
                  trace_file=args.trace)
    out = {
        "dataset": args.dataset, "robots": args.robots, "r": args.r,
        "poses": n, "edges": len(meas),
        "partition": args.partition, "selection": args.selection,
        "accel": args.accel, "device": args.device,
        "iterations": res.iterations, "converged": res.converged,
        "final_cost": res.final_cost, "final_gradnorm": res.final_gradnorm,
        "wall_s": res.elapsed_s, "setup_s": setup_s,
        "ms_per_iter": res.elapsed_s / max(res.iterations, 1) * 1e3,
    }
    rank = int(os.environ.get("RANK", "0"))
    if args.dump_trajectory:
        from dpo_amd.logger import PGOLogger
        T = (drv.final_trajectory() if args.driver == "local"
             else drv.gather_final_trajectory())
        if T is not None and rank == 0:
            lg = PGOLogger(os.path.dirname(args.dump_trajectory) or ".")
            lg.log_trajectory(meas[0].d, n, T,
                              os.path.basename(args.dump_trajectory))
    if rank == 0:
        print(json.dumps(out))


if __name__ == "__main__":
    main()
