import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from dpo_amd.comm import Comm
from dpo_amd.dist_driver import DistributedRBCDDriver
from dpo_amd.synthetic import sphere

meas, n = sphere(n=2500, loops_per_pose=1.5, rot_noise=0.2,
                 tran_noise=0.3, seed=12345)
for trial in range(3):
    drv = DistributedRBCDDriver(meas, n, 8, Comm(), r=5,
                                partition="multilevel", device="cuda:0",
                                selection="colored", inner_tol=0.0)
    drv.run(max_iters=15, gradnorm_tol=0.0)   # warmup, like bench
    res = drv.run(max_iters=75, gradnorm_tol=0.0)
    bad = [(i, c, g) for i, (c, g) in enumerate(res.trace)
           if not (g == g) or g > 1e6]
    print("trial", trial, "final gn %.4g cost %.2f" %
          (res.final_gradnorm, res.final_cost), "bad rounds:", bad[:8])
    for i in range(55, 75):
        c, g = res.trace[i]
        print("  r%02d cost %.2f gn %.4g" % (i, c, g))
