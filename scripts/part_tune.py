import os, sys, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from dpo_amd.comm import Comm
from dpo_amd.dist_driver import DistributedRBCDDriver
from dpo_amd.io_g2o import load_dataset, adjacency_from_measurements
from dpo_amd.partition import _native_multilevel, cut_edges

ds = sys.argv[1] if len(sys.argv) > 1 else "city10000"
meas, n = load_dataset(ds)
adj = adjacency_from_measurements(meas, n)
for (imb, nr, seed) in [(0.05, 8, 1), (0.03, 16, 1), (0.05, 16, 7),
                        (0.10, 16, 1), (0.03, 16, 21), (0.05, 32, 3)]:
    part = _native_multilevel(adj, 5, imb, seed, nr)
    cut = cut_edges(adj, part)
    bpose = len({u for u in range(n) for v in adj[u] if part[u] != part[v]})
    drv = DistributedRBCDDriver(meas, n, 5, Comm(), r=5, partition=part,
                                device="cuda:0")
    res = drv.run(max_iters=1000)
    print(json.dumps({"imb": imb, "restarts": nr, "seed": seed, "cut": cut,
                      "boundary_poses": bpose, "iters": res.iterations,
                      "conv": res.converged}), flush=True)
