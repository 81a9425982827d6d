"""ASan/UBSan gate over the standalone C ABI (the
reference ships no sanitizer coverage; here the dlopen-able C library —
the surface foreign engines embed — runs its compiled consumer under
AddressSanitizer+UBSan with leak detection on)."""

import os
import shutil
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_capi_under_asan():
    if shutil.which("g++") is None or shutil.which("gcc") is None:
        pytest.skip("no host toolchain")
    p = subprocess.run([os.path.join(REPO, "scripts", "sanitize_capi.sh")],
                       capture_output=True, text=True, timeout=600)
    assert p.returncode == 0, p.stdout[-2000:] + p.stderr[-2000:]
    assert "sanitize_capi: PASS" in p.stdout
