#!/usr/bin/env python3
"""Full-dataset parity sweep: iterations-to-convergence + final cost for
every shipped dataset, reference driver configuration (5 robots, r=5,
greedy RBCD, NP + multilevel partitions)."""
import glob
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cuda:0")
    ap.add_argument("--partition", default="contiguous")
    ap.add_argument("--robots", type=int, default=5)
    ap.add_argument("--selection", default="greedy")
    ap.add_argument("--accel", action="store_true")
    ap.add_argument("--tr-iters", type=int, default=1,
                    help="trust-region steps per RBCD round (reference: 1)")
    ap.add_argument("--max-iters", type=int, default=1000)
    ap.add_argument("--datasets", default="",
                    help="comma-separated subset (default: all)")
    args = ap.parse_args()
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.io_g2o import load_dataset
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    names = sorted(os.path.splitext(os.path.basename(p))[0]
                   for p in glob.glob(os.path.join(here, "data", "*.npz")))
    if args.datasets:
        keep = {s.strip() for s in args.datasets.split(",")}
        names = [n for n in names if n in keep]
    for name in names:
        try:
            meas, n = load_dataset(name)
            if n <= args.robots:
                continue
            drv = DistributedRBCDDriver(
                meas, n, args.robots, Comm(), r=5,
                partition=args.partition, device=args.device,
                selection=args.selection, acceleration=args.accel,
                tr_max_iterations=args.tr_iters)
            res = drv.run(max_iters=args.max_iters)
            print(json.dumps({
                "dataset": name, "poses": n, "edges": len(meas),
                "partition": args.partition, "selection": args.selection,
                "accel": args.accel, "robots": args.robots,
                "iterations": res.iterations, "converged": res.converged,
                "final_cost": res.final_cost,
                "final_gradnorm": res.final_gradnorm,
                "wall_s": res.elapsed_s}), flush=True)
        except Exception as e:  # noqa: BLE001
            print(json.dumps({"dataset": name, "error": str(e)}),
                  flush=True)


if __name__ == "__main__":
    main()
