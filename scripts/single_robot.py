#!/usr/bin/env python3
"""Single-robot full-batch pose-graph optimization (the reference's
examples/SingleRobotExample.cpp): chordal init + RTR at r = d (no rank
relaxation) with the batch knob set {tol 1e-1, 10 outer, 50 inner,
Delta0 10}.

python scripts/single_robot.py --dataset sphere2500 [--device cuda:0]
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
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--dump-trajectory", default=None)
    args = ap.parse_args()

    from dpo_amd.agent import PGOAgent
    from dpo_amd.io_g2o import load_dataset
    from dpo_amd.types import PGOAgentParams

    meas, n = load_dataset(args.dataset)
    d = meas[0].d
    for m in meas:
        m.r1 = m.r2 = 0
    odometry = [m for m in meas if m.p1 + 1 == m.p2]
    loop_closures = [m for m in meas if m.p1 + 1 != m.p2]

    a = PGOAgent(0, PGOAgentParams(d=d, r=d, device=args.device))
    a.set_pose_graph(odometry, loop_closures, [])
    t0 = time.perf_counter()
    Topt = a.local_pose_graph_optimization()
    wall = time.perf_counter() - t0
    res = a.last_opt_result
    out = {
        "dataset": args.dataset, "poses": n, "edges": len(meas),
        "device": args.device,
        "f_init": res.f_init, "f_opt": res.f_opt,
        "grad_norm_init": res.grad_norm_init,
        "grad_norm_opt": res.grad_norm_opt,
        "wall_s": wall,
    }
    print(json.dumps(out))
    if args.dump_trajectory:
        from dpo_amd.logger import PGOLogger
        lg = PGOLogger(os.path.dirname(args.dump_trajectory) or ".")
        lg.log_trajectory(d, n, Topt,
                          os.path.basename(args.dump_trajectory))


if __name__ == "__main__":
    main()
