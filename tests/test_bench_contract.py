"""bench.py is the driver-facing benchmark contract: one JSON line with
the documented schema, runnable on CPU with small sizes. The headline
measures the BASELINE metric (wall-clock to the reference convergence
criterion on the shipped datasets); here we exercise the identical code
path on the small shipped smallGrid3D dataset so the CPU test stays
fast."""
import json
import os
import subprocess
import sys

import pytest


def test_bench_json_contract():
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(here, "bench.py"),
         "--steps", "3", "--warmup", "1", "--dataset", "smallGrid3D",
         "--agents", "4"],
        capture_output=True, text=True, timeout=600, cwd=here)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    o = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in o, f"missing {key}"
    assert o["metric"] == "smallGrid3D_wall_to_gradnorm_0.1_s"
    assert o["steps"] == 3
    assert o["higher_is_better"] is False
    assert o["scaling"] == "strong"
    assert o["dtype"] == "fp64"
    assert o["value"] > 0
    assert o["config"]["converged"] is True
    assert o["config"]["iterations"] > 0
    # deterministic episodes: warmup episode matches the timed ones
    assert all(w == o["config"]["iterations"]
               for w in o["config"]["warmup_iterations"])
