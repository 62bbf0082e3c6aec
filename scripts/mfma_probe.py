#!/usr/bin/env python3
"""Empirical probe of the v_mfma_f64_16x16x4_f64 fragment layout via
k_bsr_spmm_mfma_d3 on a trivial graph: 4 poses, Q = block-diagonal with
distinct asymmetric 4x4 blocks, X distinct. out_pose must equal
B_pose @ X_pose; prints the achieved vs expected tiles so a mapping
error shows its permutation structure."""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dpo_amd.ops import hip_backend as hb  # noqa: E402

dev = "cuda:0"
n, r = 4, 5
rp = torch.tensor([0, 1, 2, 3, 4], dtype=torch.int32, device=dev)
ci = torch.tensor([0, 1, 2, 3], dtype=torch.int32, device=dev)
vals = torch.arange(64, dtype=torch.float64, device=dev).reshape(4, 4, 4)
vals = vals + torch.rand(4, 4, 4, dtype=torch.float64, device=dev)
X = torch.arange(n * 4 * r, dtype=torch.float64,
                 device=dev).reshape(n * 4, r) / 7.0

gp, gc, gb = hb.build_spmm_mfma_groups(rp, ci, n)
out = hb.bsr_spmm_mfma(torch.from_numpy(gp).to(dev),
                       torch.from_numpy(gc).to(dev),
                       torch.from_numpy(gb).to(dev), vals, X)
torch.cuda.synchronize()
ref = torch.einsum("pij,pjk->pik", vals, X.view(n, 4, r))
got = out.view(n, 4, r)
print("max err:", float((got - ref).abs().max()))
for p in range(2):
    print(f"--- pose {p} expected:\n", ref[p].cpu().numpy())
    print(f"--- pose {p} got:\n", got[p].cpu().numpy())
