"""Web-client Opus decoder (selkies_amd/web/opus-decoder.js) anchored
against the C++ encoder: node decodes real packets and the output must
match tests/opus_ref_decoder.py sample-for-sample (float32-basis
tolerance).  Mirrors the dashboard node-harness pattern
(tests/test_dashboard_api.py)."""

import json
import pathlib
import shutil
import subprocess

import numpy as np
import pytest

hipflux = pytest.importorskip("hipflux")
if not hipflux.native_available():
    pytest.skip("hipflux native module not built", allow_module_level=True)
if shutil.which("node") is None:
    pytest.skip("node not available", allow_module_level=True)

from hipflux import _native
from opus_ref_decoder import OpusDecoder

WEB = pathlib.Path(__file__).resolve().parents[1] / "selkies_amd" / "web"

HARNESS = """
const {OpusDecoder} = require(process.argv[2]);
const pkts = JSON.parse(require("fs").readFileSync(process.argv[3], "utf8"));
const dec = new OpusDecoder();
const out = [];
for (const hex of pkts) {
  const pkt = Uint8Array.from(Buffer.from(hex, "hex"));
  out.push(Array.from(dec.decodePacket(pkt)));
}
process.stdout.write(JSON.stringify(out));
"""


def _encode_tone(n_frames=6, bitrate=96000, freq=440.0):
    enc = _native.OpusEncoder(bitrate)
    t = np.arange(960 * n_frames) / 48000.0
    pcm = (6000 * np.sin(2 * np.pi * freq * t)).astype(np.int16)
    return [bytes(enc.encode(pcm[i * 960:(i + 1) * 960], 1))
            for i in range(n_frames)]


def _js_decode(tmp_path, pkts):
    pkts_file = tmp_path / "pkts.json"
    pkts_file.write_text(json.dumps([p.hex() for p in pkts]))
    harness = tmp_path / "run.js"
    harness.write_text(HARNESS)
    r = subprocess.run(
        ["node", str(harness), str(WEB / "opus-decoder.js"),
         str(pkts_file)],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    return [np.asarray(f) for f in json.loads(r.stdout)]


def test_js_decoder_matches_reference(tmp_path):
    pkts = _encode_tone()
    js = _js_decode(tmp_path, pkts)
    ref = OpusDecoder()
    for i, p in enumerate(pkts):
        want = ref.decode_packet(p)
        got = js[i]
        assert got.shape == want.shape
        peak = max(np.abs(want).max(), 1e-9)
        # JS basis is float32; reference is float64
        assert np.abs(got - want).max() / peak < 2e-4, i


def test_js_decoder_tone_snr(tmp_path):
    pkts = _encode_tone(n_frames=8, bitrate=128000)
    js = _js_decode(tmp_path, pkts)
    out = np.concatenate(js[2:])
    t = np.arange(out.size) / 48000.0
    ref = np.vstack([np.sin(2 * np.pi * 440 * t),
                     np.cos(2 * np.pi * 440 * t)]).T
    coef, *_ = np.linalg.lstsq(ref, out, rcond=None)
    resid = out - ref @ coef
    snr = 10 * np.log10((out ** 2).sum() / max((resid ** 2).sum(), 1e-12))
    assert snr > 18.0, snr


def test_js_decoder_rejects_garbage(tmp_path):
    harness = tmp_path / "rej.js"
    harness.write_text("""
const {OpusDecoder} = require(process.argv[2]);
const dec = new OpusDecoder();
let threw = false;
try { dec.decodePacket(Uint8Array.from([0x00, 1, 2, 3])); }
catch (e) { threw = true; }
process.stdout.write(threw ? "ok" : "bad");
""")
    r = subprocess.run(["node", str(harness), str(WEB / "opus-decoder.js")],
                       capture_output=True, text=True, timeout=60)
    assert r.returncode == 0 and r.stdout == "ok", (r.stdout, r.stderr)
