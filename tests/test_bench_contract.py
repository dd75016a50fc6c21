"""bench.py driver contract: both modes run on CPU and print one valid
result JSON line with the required fields (the driver parses this)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _run(argv):
    out = subprocess.run([sys.executable, "bench.py"] + argv,
                         capture_output=True, text=True, cwd=REPO,
                         timeout=600,
                         env={**os.environ, "PYTHONPATH": REPO})
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, out.stdout[-2000:]
    return json.loads(lines[0])


def test_bench_serving_mode_contract():
    res = _run(["--steps", "1", "--warmup", "1"])
    assert REQUIRED <= set(res)
    assert res["value"] > 0
    assert res["config"]["mode"] == "serving"
    assert res["config"]["parallelism"] == "dp1"
    assert res["higher_is_better"] is True


def test_bench_engine_mode_contract():
    res = _run(["--mode", "engine", "--steps", "2", "--warmup", "1"])
    assert REQUIRED <= set(res)
    assert res["value"] > 0
    assert res["config"]["mode"] == "engine"
