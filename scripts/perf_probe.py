"""Probe run-to-run throughput variance: three timed 60-round segments
per process; a 'slow run' that is slow in ALL segments points at
process-persistent state (clock/placement), slow in one at transients."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dpo_amd.comm import Comm
from dpo_amd.dist_driver import DistributedRBCDDriver
from dpo_amd.synthetic import sphere

meas, n = sphere(n=2500, loops_per_pose=1.5, rot_noise=0.2,
                 tran_noise=0.3, seed=12345)
drv = DistributedRBCDDriver(meas, n, 8, Comm(), r=5,
                            partition="multilevel", device="cuda:0",
                            selection="colored", inner_tol=0.0)
drv.run(max_iters=15, gradnorm_tol=0.0)
rates = []
for seg in range(3):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    drv.run(max_iters=60, gradnorm_tol=0.0)
    torch.cuda.synchronize()
    rates.append(60 / (time.perf_counter() - t0))
print("segments r/s: " + " ".join("%.0f" % r for r in rates))
