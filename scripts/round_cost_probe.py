#!/usr/bin/env python3
"""Probe per-round wall cost of the bench configuration across the
convergence trajectory: times 25-round segments of a single episode so
late-round slowdowns (e.g. trust-region shrink-replay storms near the
optimum) show up directly. Also reports a tol=0 free-run segment rate
for comparison."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import argparse
    import torch
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset", default="sphere2500")
    ap.add_argument("--agents", type=int, default=8)
    ap.add_argument("--device", default="cuda:0")
    ap.add_argument("--selection", default="colored")
    ap.add_argument("--segments", type=int, default=10)
    ap.add_argument("--seglen", type=int, default=25)
    ap.add_argument("--inner-tol", type=float, default=1e-2)
    ap.add_argument("--partition", default="contiguous")
    args = ap.parse_args()
    from dpo_amd.comm import init_from_env
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.io_g2o import load_dataset
    meas, n = load_dataset(args.dataset)
    comm = init_from_env(args.device)
    drv = DistributedRBCDDriver(meas, n, args.agents, comm, r=5,
                                partition=args.partition,
                                device=args.device,
                                selection=args.selection,
                                inner_tol=args.inner_tol)
    drv.snapshot_initial_state()
    # warmup episode (graph capture etc.)
    drv.run(max_iters=40, gradnorm_tol=0.0)
    drv.restore_initial_state()
    torch.cuda.synchronize()
    out = []
    for seg in range(args.segments):
        t0 = time.perf_counter()
        res = drv.run(max_iters=args.seglen, gradnorm_tol=0.0)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        out.append({"seg": seg, "rounds": res.iterations,
                    "ms_per_round": round(dt / res.iterations * 1e3, 3),
                    "gradnorm": res.final_gradnorm})
        print(json.dumps(out[-1]), flush=True)


if __name__ == "__main__":
    main()
