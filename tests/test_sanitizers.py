"""SURVEY §5 race-detection CI: the native control plane is built and
stress-run under ASan and TSan every round (tools/sanitize).  The
driver compiles control_plane.cc standalone (no interpreter) and
hammers register/heartbeat/put/get/incr/barrier from 8 client threads;
any sanitizer report fails the run."""

import os
import shutil
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CLANG = "/opt/rocm/lib/llvm/bin/clang++"


@pytest.mark.skipif(not os.path.exists(CLANG) or shutil.which("python3-config") is None,
                    reason="sanitizer toolchain unavailable")
def test_control_plane_asan_tsan(tmp_path):
    r = subprocess.run(
        ["bash", "tools/sanitize/run.sh", str(tmp_path)],
        cwd=ROOT, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "sanitize OK" in r.stdout
    assert "WARNING: ThreadSanitizer" not in r.stderr
    assert "AddressSanitizer" not in r.stderr
