"""bench.py driver-contract test: one JSON line with the required fields."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--device", "cpu",
         "--n", "1024", "--nb", "256", "--steps", "1", "--warmup", "0"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    j = json.loads(lines[0])
    assert j["metric"] == "cholesky_fp64_gflops"
    assert j["unit"] == "GFlop/s"
    assert j["higher_is_better"] is True
    assert j["scaling"] == "strong"
    assert j["dtype"] == "fp64"
    assert j["n_gpus"] == 1 and j["steps"] == 1 and j["warmup"] == 0
    assert j["value"] > 0 and j["ms_per_step"] > 0
    for k in ("model", "n", "nb", "parallelism"):
        assert k in j["config"]
    assert "synthetic" in j["data"]
