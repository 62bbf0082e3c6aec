import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from dpo_amd.comm import Comm
from dpo_amd.dist_driver import DistributedRBCDDriver
from dpo_amd.io_g2o import load_dataset
meas, n = load_dataset("sphere2500")
for k in range(3):
    drv = DistributedRBCDDriver(meas, n, 5, Comm(), r=5,
                                partition="contiguous", device="cuda:0")
    res = drv.run(max_iters=1000)
    print(f"run {k}: iters={res.iterations} conv={res.converged} gn={res.final_gradnorm:.4f}")
