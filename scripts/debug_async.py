import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from dpo_amd.comm import Comm
from dpo_amd.dist_driver import DistributedRBCDDriver
from dpo_amd.io_g2o import load_dataset

meas, n = load_dataset("city10000")
os.environ["DPO_SYNC_SOLVE"] = "0"
os.environ["DPO_SYNC_EVAL"] = "1"
drv = DistributedRBCDDriver(meas, n, 5, Comm(), r=5,
                            partition="multilevel", device="cuda:0")

import dpo_amd.agent as am
orig_finish = am.PGOAgent._packed_solve_finish
state = {"bad": None}
def patched(self):
    orig_finish(self)
    st = self._dev_solver._stats
    if abs(st[1]) > 1e15 and state["bad"] is None:
        state["bad"] = self
        raise SystemExit(0)
am.PGOAgent._packed_solve_finish = patched

try:
    res = drv.run(max_iters=120)
except SystemExit:
    pass
a = state["bad"]
print("bad agent:", a.id if a else None)
if a:
    # sanity of inputs
    print("X finite:", bool(torch.isfinite(a.X).all().item()),
          "max", float(a.X.abs().max()))
    print("nbr finite:", bool(torch.isfinite(a._nbr_buffer).all().item()),
          "max", float(a._nbr_buffer.abs().max()))
    print("Q finite:", bool(torch.isfinite(a.problem.Q.vals).all().item()))
    print("Minv finite:", bool(torch.isfinite(a.problem._Minv).all().item()))
    # 1. replay the cached graph again (same pointers)
    ds = a._dev_solver
    st1 = ds.round_solve(a.X.clone() if False else a.X, a._nbr_buffer)
    print("re-run cached graph: status", st1, "f_init", ds._stats[1])
    # 2. fresh DeviceSolver, eager first call
    from dpo_amd.ops.hip_backend import DeviceSolver
    ds2 = DeviceSolver(a.n, a.d, a.r, a.device, max_inner=10)
    ds2.set_gdata(*(ds._grefs))
    ds2.bind_problem_static(a.problem)
    st2 = ds2.round_solve(a.X, a._nbr_buffer)
    print("fresh solver: status", st2, "f_init", ds2._stats[1],
          "f_opt", ds2._stats[3])
