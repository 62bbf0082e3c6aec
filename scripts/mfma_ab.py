#!/usr/bin/env python3
"""A/B study: fp64-MFMA BSR SpMM vs the scalar-FMA SpMM kernels
(round-1 VERDICT item 4; BASELINE north star asks for the MFMA decision
to be settled with measured data).

Contenders on the same block-sparse Q (d=3, 4x4 blocks, r=5):
  * k_bsr_spmm        — element-per-thread scalar-FMA (production)
  * k_bsr_spmm_mfma_d3 — v_mfma_f64_16x16x4_f64 over 4-pose row groups
                         with column-union grouped-ELL padding

Measured at agent scale (~2.7k poses) and 1M-agent scale (~125k poses).
Prints one JSON line per (scale, kernel); writes profiles/mfma_ab.json.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench_case(side, device, reps=200):
    import numpy as np
    import torch
    from dpo_amd.ops import hip_backend as hb
    from dpo_amd.quadratic import assemble_connection_laplacian
    from dpo_amd.synthetic import grid3d_soa

    ma, n = grid3d_soa(side=side, seed=1)
    Q = assemble_connection_laplacian(ma, n, 3).to(device)
    rp, ci, vals = Q.row_ptr, Q.col_idx, Q.vals
    g = torch.Generator().manual_seed(0)
    X = torch.randn(4 * n, 5, dtype=torch.float64, generator=g).to(device)

    # reference (torch sparse on device)
    ref = Q.spmm(X)

    gp, gc, gb = hb.build_spmm_mfma_groups(rp, ci, n)
    gp_t = torch.from_numpy(gp).to(device)
    gc_t = torch.from_numpy(gc).to(device)
    gb_t = torch.from_numpy(gb).to(device)
    nnz = int(vals.numel() // 16)
    padded = int(gb.size)  # 4 slots per union column
    fill = nnz / padded

    out = torch.empty_like(X)
    results = []

    def timeit(fn, name, extra=None):
        fn()  # warmup + numerics
        torch.cuda.synchronize()
        err = float((out - ref).abs().max())
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / reps
        flops = 2.0 * nnz * 16 * 5
        rec = {"side": side, "poses": n, "nnz_blocks": nnz,
               "kernel": name, "us": dt * 1e6,
               "gflops_useful": flops / dt / 1e9,
               "max_abs_err": err}
        if extra:
            rec.update(extra)
        results.append(rec)
        print(json.dumps(rec), flush=True)

    timeit(lambda: hb.bsr_spmm(rp, ci, vals, n, 4, X, out=out),
           "k_bsr_spmm")
    timeit(lambda: hb.bsr_spmm_mfma(gp_t, gc_t, gb_t, vals, X, out=out),
           "k_bsr_spmm_mfma_d3",
           {"union_fill": fill,
            "useful_frac_of_mfma_flops": fill * 5.0 / 16.0})
    return results


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cuda:0")
    ap.add_argument("--out", default="profiles/mfma_ab.json")
    ap.add_argument("--sides", nargs="*", type=int, default=[14, 50])
    args = ap.parse_args()
    allr = []
    for s in args.sides:
        allr += bench_case(s, args.device)
    if args.out:
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(allr, f, indent=1)


if __name__ == "__main__":
    main()
