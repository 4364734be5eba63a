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
