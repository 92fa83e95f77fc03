"""Test-suite registry and tier runner (reference tests/suites.py:21-103
architecture: every suite is declared in one place with its selection,
tier, and hardware requirement; CI and humans invoke suites by name).

Usage:
    python -m tests.suites list
    python -m tests.suites run <suite> [extra pytest args...]

Tiers:
    0  pure-python / native-extension unit + protocol tests (always run)
    1  integration: live servers, sockets, multi-process gloo ranks
    2  gpu: requires an MI355X (pytest -m gpu)
    3  perf: bench contracts (timing-sensitive; run on quiet machines)
"""

from __future__ import annotations

import subprocess
import sys
from dataclasses import dataclass, field
from pathlib import Path

TESTS_DIR = Path(__file__).resolve().parent


@dataclass(frozen=True)
class Suite:
    name: str
    description: str
    tier: int
    paths: tuple[str, ...] = ()          # test files (relative to tests/)
    marker: str = "not gpu"              # pytest -m expression
    extra_args: tuple[str, ...] = ()


SUITES: dict[str, Suite] = {}


def _register(s: Suite):
    SUITES[s.name] = s
    return s


_register(Suite(
    "codecs", "Entropy/transform conformance vs the from-spec decoders "
    "(H.264, HEVC, JPEG, Opus)", 0,
    ("test_h264.py", "test_hevc.py", "test_jpeg.py", "test_opus.py",
     "test_opus_js.py", "test_rate_control.py")))
_register(Suite(
    "engine", "Capture engine, damage, scaling, audio engine, recording",
    0, ("test_engine.py", "test_engine_h264.py", "test_audio.py",
        "test_recording.py", "test_cursor.py", "test_watermark.py")))
_register(Suite(
    "protocol", "Wire protocol, relay backpressure, settings, fuzzing",
    0, ("test_protocol.py", "test_relay.py", "test_settings.py",
        "test_fuzz.py", "test_pacer.py", "test_transfers.py")))
_register(Suite(
    "input", "Input dispatch, clipboard, gamepad, interposer, wayland",
    0, ("test_clipboard_input.py", "test_gamepad.py",
        "test_touch_gamepad.py", "test_interposer.py", "test_fake_udev.py",
        "test_wayland.py")))
_register(Suite(
    "server", "Live WS server, roles, transfers, multi-display, webrtc "
    "loopback, dashboard", 1,
    ("test_server.py", "test_roles.py", "test_player_seats.py",
     "test_multidisplay.py", "test_webrtc_stack.py", "test_rtc_config.py",
     "test_dashboard_api.py", "test_computer_use.py",
     "test_display_utils.py", "test_advice_fixes.py")))
_register(Suite(
    "distributed", "Multi-process rank tests (gloo here, RCCL on GPU)",
    1, ("test_tile_comm.py",)))
_register(Suite(
    "sanitize", "ASan/UBSan native harness + packaging lint", 1,
    ("test_sanitizer.py",)))
_register(Suite(
    "gpu", "Byte-identity + numerics on a real MI355X", 2,
    ("test_gpu_h264.py", "test_gpu_hevc.py", "test_gpu_jpeg.py",
     "test_gpu_tile_comm.py"), marker="gpu"))
_register(Suite(
    "perf", "Bench contract guards (driver JSON line shape)", 3,
    ("test_bench_contract.py",)))
_register(Suite(
    "all-cpu", "Everything that runs without a GPU", 1, (),
    marker="not gpu"))


def pytest_args(suite: Suite) -> list[str]:
    args = ["-q", "-m", suite.marker]
    args += [str(TESTS_DIR / p) for p in suite.paths] or [str(TESTS_DIR)]
    args += list(suite.extra_args)
    return args


def main(argv: list[str]) -> int:
    if not argv or argv[0] == "list":
        for s in sorted(SUITES.values(), key=lambda s: (s.tier, s.name)):
            print(f"tier {s.tier}  {s.name:12s} {s.description}")
        return 0
    if argv[0] == "run" and len(argv) >= 2:
        suite = SUITES.get(argv[1])
        if suite is None:
            print(f"unknown suite {argv[1]!r}; try `list`",
                  file=sys.stderr)
            return 2
        cmd = [sys.executable, "-m", "pytest"] + pytest_args(suite) \
            + argv[2:]
        return subprocess.call(cmd)
    print(__doc__, file=sys.stderr)
    return 2


if __name__ == "__main__":
    raise SystemExit(main(sys.argv[1:]))
