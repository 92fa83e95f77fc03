"""ASan+UBSan harness over the native host-code codec paths.

SURVEY.md §5.2: the reference has no native sanitizer coverage; the HIP
engine's host code gets real coverage here (tests/asan_main.cpp drives
the H.264 encoder, JPEG encoder, downscaler and NAL assembly under
-fsanitize=address,undefined)."""

import shutil
import subprocess

import pytest


def test_native_host_code_under_asan_ubsan():
    if shutil.which("g++") is None:
        pytest.skip("g++ not available")
    r = subprocess.run(["make", "-C", "native", "asan"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert "sanitizer harness ok" in r.stdout
