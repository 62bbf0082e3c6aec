"""Debug the GPU driver stall: run until frozen, then dissect one solve."""
import torch

from dpo_amd.driver import MultiRobotDriver
from dpo_amd.synthetic import grid3d
from dpo_amd.ops import hip_backend as hb

DEV = "cuda:0"

meas, n = grid3d(side=4, seed=0)
drv = MultiRobotDriver(meas, n, 2, r=5, partition="contiguous", device=DEV)

res = drv.run(max_iters=300)
print("converged:", res.converged, "final gn:", res.final_gradnorm)
tail = [g for (_, g) in res.trace[-10:]]
print("gradnorm tail:", tail)

# Which agent is selected & what does its solver do?
import numpy as np
X = drv._gather_global_x()
rgrad = drv.central.rie_grad(X)
Gb = rgrad.view(drv.n, drv.dh, drv.r)
norms = []
for rb in range(drv.num_robots):
    nb = Gb.index_select(0, drv._blk_index[rb])
    norms.append(float(nb.pow(2).sum()))
print("per-agent central gn2:", norms)
sel = int(np.argmax(norms))
a = drv.agents[sel]
# refresh neighbor data as the driver would
drv._exchange_with(sel, False)
ok = a._construct_g(a.neighbor_pose_dict)
print("G ok:", ok)
p = a.problem
print("local f", p.f(a.X), "local gn", p.rie_grad_norm(a.X))

ds = a._dev_solver
Xw = a.X.clone()
stats = ds.solve(p, Xw, tol=1e-2, Delta0=100.0)
print("device solve stats:", stats)
ctrl = ds.ctrl.cpu().numpy()
print("ctrl: status", ctrl[0], "radius", ctrl[12], "dm", ctrl[14],
      "rho", ctrl[21], "fprop", ctrl[13], "fX", ctrl[1],
      "hlen", ctrl[22], "J", ctrl[11], "use_cur", ctrl[17])
print("hist z_r", ctrl[32:32+6])
print("hist dHd", ctrl[48:48+6])

# host solver on the same problem (same jacobi? p uses dense on GPU)
from dpo_amd.solver import QuadraticOptimizer, TRParams
from dpo_amd.types import OptAlgorithm
tr = TRParams(tolerance=1e-2, initial_radius=100.0, max_iterations=1,
              max_inner_iterations=10)
opt = QuadraticOptimizer(p, OptAlgorithm.RTR, tr)
Xh = opt.optimize(a.X.clone())
print("host result: f", opt.result.f_init, "->", opt.result.f_opt,
      "gn", opt.result.grad_norm_init, "->", opt.result.grad_norm_opt,
      "tcg", opt.result.tcg_status)
