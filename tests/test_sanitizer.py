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


def test_packaging_recipes_reference_real_entry_points():
    """The rpm/apk/arch/AppImage recipes must name only entry points that
    exist in pyproject.toml and addon .so files that the build produces."""
    import os
    import re
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    eps = set(re.findall(r"^(selkies[\w-]*)\s*=",
                         open(os.path.join(root, "pyproject.toml")).read(),
                         re.M))
    assert eps, "no console scripts found in pyproject"
    for rel in ("infra/packaging/selkies-amd.spec",
                "infra/packaging/APKBUILD",
                "infra/packaging/PKGBUILD",
                "infra/appimage/build_appimage.sh",
                "infra/packaging/build_deb.sh"):
        path = os.path.join(root, rel)
        text = open(path).read()
        for name in re.findall(r"\bselkies[\w-]*\b", text):
            if name.startswith("selkies-amd") or name == "selkies_amd":
                continue
            if name in ("selkies",) or name in eps:
                continue
            assert name in eps, f"{rel} references unknown script {name}"
        assert "js-interposer" in text and "fake-udev" in text
