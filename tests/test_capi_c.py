"""Build & run the C-ABI smoke test against libdlaf_c.so.

Verifies the reference-parity C API surface (SURVEY.md section 2.8,
include/dlaf_c/*): a plain C program links libdlaf_c.so, factorizes,
inverts and diagonalizes through the embedded dlaf_amd runtime.
"""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_c_abi_smoke(tmp_path):
    lib = os.path.join(ROOT, "libdlaf_c.so")
    if not os.path.exists(lib):
        r = subprocess.run(["bash", os.path.join(ROOT, "tools", "build_capi.sh")],
                           capture_output=True, text=True)
        assert r.returncode == 0, r.stderr
    exe = str(tmp_path / "test_dlaf_c")
    r = subprocess.run(
        ["gcc", "-O2", os.path.join(ROOT, "tests", "c", "test_dlaf_c.c"),
         "-I", os.path.join(ROOT, "include"),
         "-L", ROOT, "-ldlaf_c", "-lm", f"-Wl,-rpath,{ROOT}", "-o", exe],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    r = subprocess.run([exe], capture_output=True, text=True, timeout=280,
                       env={**os.environ, "DLAF_AMD_PYROOT": ROOT})
    assert r.returncode == 0, f"rc={r.returncode}\n{r.stdout}\n{r.stderr}"
    assert "OK" in r.stdout, r.stdout


@pytest.mark.timeout(300)
def test_c_abi_distributed_2rank(tmp_path):
    """1x2-grid POTRF through the C ABI: two C processes rendezvous via the
    torchrun-style env (gloo backend inside the embedded runtime)."""
    lib = os.path.join(ROOT, "libdlaf_c.so")
    if not os.path.exists(lib):
        r = subprocess.run(["bash", os.path.join(ROOT, "tools", "build_capi.sh")],
                           capture_output=True, text=True)
        assert r.returncode == 0, r.stderr
    exe = str(tmp_path / "test_dlaf_c_dist")
    r = subprocess.run(
        ["gcc", "-O2", os.path.join(ROOT, "tests", "c", "test_dlaf_c_dist.c"),
         "-I", os.path.join(ROOT, "include"),
         "-L", ROOT, "-ldlaf_c", "-lm", f"-Wl,-rpath,{ROOT}", "-o", exe],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    base = {**os.environ, "DLAF_AMD_PYROOT": ROOT, "WORLD_SIZE": "2",
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "CUDA_VISIBLE_DEVICES": ""}
    procs = [subprocess.Popen([exe], stdout=subprocess.PIPE,
                              stderr=subprocess.PIPE, text=True,
                              env={**base, "RANK": str(rk),
                                   "LOCAL_RANK": str(rk)})
             for rk in range(2)]
    outs = []
    for p in procs:
        try:
            out, errs = p.communicate(timeout=280)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            raise
        outs.append((p.returncode, out, errs))
    for rk, (rc, out, errs) in enumerate(outs):
        assert rc == 0, f"rank {rk} rc={rc}\n{out}\n{errs}"
        assert f"OK rank {rk}" in out, out
