#!/usr/bin/env python3
"""Chordal-initialization evaluation over datasets (the reference's
examples/ChordalInitializationExample.cpp): cost and Riemannian gradient
norm of the chordal relaxation solution on each dataset.

python scripts/chordal_eval.py [--datasets sphere2500 CSAIL ...]
"""
import argparse
import glob
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--datasets", nargs="*", default=None)
    args = ap.parse_args()

    import numpy as np
    import torch
    from dpo_amd.chordal import chordal_initialization
    from dpo_amd.io_g2o import load_dataset
    from dpo_amd.quadratic import (QuadraticProblem,
                                   assemble_connection_laplacian)

    names = args.datasets
    if not names:
        here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        names = sorted(os.path.splitext(os.path.basename(p))[0]
                       for p in glob.glob(os.path.join(here, "data", "*.npz")))
    for name in names:
        meas, n = load_dataset(name)
        d = meas[0].d
        t0 = time.perf_counter()
        T = chordal_initialization(d, n, meas)
        wall = time.perf_counter() - t0
        Q = assemble_connection_laplacian(meas, n, d)
        prob = QuadraticProblem(n, d, d, precond="jacobi")
        prob.set_q(Q)
        X = torch.from_numpy(np.ascontiguousarray(T.T))
        print(json.dumps({
            "dataset": name, "poses": n, "edges": len(meas),
            "chordal_cost": 2.0 * prob.f(X),
            "chordal_gradnorm": prob.rie_grad_norm(X),
            "wall_s": wall,
        }))


if __name__ == "__main__":
    main()
