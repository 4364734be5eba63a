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


def test_bench_dist_cpu_2rank():
    """2-rank gloo run of bench.py (exercises the RCCL-order dry-run guard
    and the distributed POTRF path through the driver entry point)."""
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29571",
                "DLAF_BENCH_N": "1024", "DLAF_BENCH_NB": "256"})
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29572", os.path.join(ROOT, "bench.py"),
         "--device", "cpu", "--steps", "1", "--warmup", "0", "--gpus", "2"],
        capture_output=True, text=True, timeout=600, env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    j = json.loads(lines[0])
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"] == "grid1x2"
