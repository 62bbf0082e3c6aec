"""bench.py is the driver-facing benchmark contract: one JSON line with
the documented schema, runnable on CPU with small sizes."""
import json
import os
import subprocess
import sys

import pytest


def test_bench_json_contract():
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(here, "bench.py"),
         "--steps", "4", "--warmup", "1", "--poses", "200"],
        capture_output=True, text=True, timeout=600, cwd=here)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    o = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in o, f"missing {key}"
    assert o["metric"] == "rbcd_rounds_per_s"
    assert o["steps"] == 4
    assert o["higher_is_better"] is True
    assert o["scaling"] == "strong"
    assert o["dtype"] == "fp64"
    assert o["value"] > 0
    assert o["config"]["agents"] == 8
