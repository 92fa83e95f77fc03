"""Guards the driver-facing contracts: bench.py's JSON line (the CI
driver parses these exact fields), the static web client's JS syntax,
and the pipeline-depth interface on the CPU tier."""

import json
import shutil
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parents[1]


def test_bench_json_contract():
    """`python bench.py --cpu --steps 2` emits one JSON line with every
    field the driver consumes, with sane values."""
    p = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--cpu", "--steps", "2",
         "--warmup", "1", "--width", "320", "--height", "192"],
        capture_output=True, text=True, timeout=300)
    assert p.returncode == 0, p.stderr[-800:]
    line = [l for l in p.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in d, f"missing driver field {key}"
    assert d["metric"] == "encoded_fps_1080p60_h264"
    assert d["n_gpus"] == 1 and d["steps"] == 2
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["data"] == "synthetic" and d["value"] > 0
    cfg = d["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism",
                "resolution", "latency_p50_ms", "latency_p95_ms"):
        assert key in cfg, f"missing config field {key}"
    assert cfg["resolution"] == "320x192"


node = shutil.which("node")


@pytest.mark.skipif(node is None, reason="node not available")
def test_web_client_js_syntax():
    """Every shipped static JS file parses (node --check)."""
    web = REPO / "selkies_amd" / "web"
    files = sorted(web.glob("*.js"))
    assert files
    for f in files:
        p = subprocess.run([node, "--check", str(f)],
                           capture_output=True, text=True, timeout=60)
        assert p.returncode == 0, f"{f.name}: {p.stderr}"


def test_cpu_pipeline_ignores_depth():
    """pipeline_depth is a HIP-pipeline feature; the CPU pipelines take
    the same interface but stay synchronous — same outputs, flush is a
    no-op."""
    hipflux = pytest.importorskip("hipflux")
    if not hipflux.native_available():
        pytest.skip("native module not built")
    import numpy as np
    from hipflux import _native
    rng = np.random.default_rng(3)
    frames = [np.ascontiguousarray(
        rng.integers(0, 256, (128, 192, 4), dtype=np.uint8))
        for _ in range(4)]
    a = _native._pipeline_encode("cpu", frames, 192, 128, 26, 64, 1)
    b = _native._pipeline_encode("cpu", frames, 192, 128, 26, 64, 1,
                                 pipeline_depth=2)
    assert len(a) == len(b) == 4
    for fa, fb in zip(a, b):
        assert [(bytes(d), y) for d, y, *_ in fa] == \
               [(bytes(d), y) for d, y, *_ in fb]
