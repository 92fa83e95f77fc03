"""Dashboard postMessage API contract (node audit of
selkies_amd/web/postmessage-bridge.js against the reference
window-messaging contract, addons/selkies-web-core/README.md:44-170):
same-origin enforcement, settings mapping, pipeline control, clipboard,
stats round-trip and client->dashboard notifications."""

import json
import os
import shutil
import subprocess

import pytest

if shutil.which("node") is None:
    pytest.skip("node not available", allow_module_level=True)

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

HARNESS = r"""
const { installPostMessageBridge } = require(process.argv[1]);

// stub window: same-origin parent capture + message dispatch
const posted = [];
const listeners = [];
const win = {
  location: { origin: "https://host.example" },
  parent: { postMessage: (m, o) => posted.push({ m, o }) },
  addEventListener: (t, fn) => { if (t === "message") listeners.push(fn); },
};
win.parent.self = win.parent;   // parent !== win

const sent = [];
const calls = [];
const core = {
  send: (m) => sent.push(m),
  getStats: () => ({ clientFps: 42, encoderName: "h264enc-striped" }),
  setVolume: (v) => calls.push(["volume", v]),
  setMute: (v) => calls.push(["mute", v]),
  setGamepadEnabled: (v) => calls.push(["gamepad", v]),
  resizeToWindow: () => calls.push(["resizeToWindow"]),
  showVirtualKeyboard: () => calls.push(["vkb"]),
  setRenderFlag: (k, v) => calls.push(["render", k, v]),
};

const bridge = installPostMessageBridge(core, win);
const dispatch = (data, origin) =>
    listeners.forEach((fn) => fn({ data, origin:
        origin || "https://host.example" }));

// 1. same-origin enforcement
dispatch({ type: "command", value: "evil" }, "https://attacker.example");
// 2. settings mapping (video_bitrate -> video_bitrate_kbps)
dispatch({ type: "settings", settings: {
    framerate: 120, video_bitrate: 8000, encoder: "jpeg",
    scaling_dpi: 144 } });
// 3. resolution + pipeline + volume + clipboard + command + stats
dispatch({ type: "setManualResolution", width: 2560, height: 1440 });
dispatch({ type: "resetResolutionToWindow" });
dispatch({ type: "pipelineControl", pipeline: "video", enabled: false });
dispatch({ type: "setVolume", value: 0.5 });
dispatch({ type: "setMute", value: true });
dispatch({ type: "gamepadControl", enabled: false });
dispatch({ type: "clipboardUpdateFromUI", text: "héllo" });
dispatch({ type: "command", value: "xdotool key a" });
dispatch({ type: "sidebarVisibilityChanged", isOpen: true });
dispatch({ type: "getStats" });

// 4. client -> dashboard notifications
bridge.onServerSettings({ encoder: { value: "h264enc-striped" } });
bridge.onClipboard("from server");
bridge.onRole("controller");
bridge.onStats({ cpu: 10 });     // sidebar open -> forwarded

console.log(JSON.stringify({ sent, calls, posted }));
"""


def run_harness():
    bridge = os.path.join(ROOT, "selkies_amd", "web",
                          "postmessage-bridge.js")
    r = subprocess.run(["node", "-e", HARNESS, bridge],
                       capture_output=True, text=True, timeout=60)
    assert r.returncode == 0, r.stderr
    return json.loads(r.stdout)


def test_postmessage_contract():
    out = run_harness()
    sent, calls = out["sent"], out["calls"]
    # cross-origin message dropped: no "cmd,evil"
    assert "cmd,evil" not in sent
    # settings mapped onto the SETTINGS wire verb with server knob names
    st = next(s for s in sent if s.startswith("SETTINGS,"))
    payload = json.loads(st.split(",", 1)[1])
    assert payload["framerate"] == 120
    assert payload["video_bitrate_kbps"] == 8000
    assert payload["encoder"] == "jpeg"
    assert "s,144" in sent
    assert "r,2560x1440" in sent
    assert "STOP_VIDEO" in sent
    assert "cmd,xdotool key a" in sent
    # clipboard base64 (utf-8)
    import base64
    cw = next(s for s in sent if s.startswith("cw,"))
    assert base64.b64decode(cw[3:]).decode() == "héllo"
    assert ["volume", 0.5] in calls
    assert ["mute", True] in calls
    assert ["gamepad", False] in calls
    assert ["resizeToWindow"] in calls


def test_postmessage_telemetry():
    out = run_harness()
    posted = [p["m"] for p in out["posted"]]
    types = [p["type"] for p in posted]
    assert "stats" in types
    stats = next(p for p in posted if p["type"] == "stats")
    assert stats["data"]["clientFps"] == 42
    assert "pipelineStatusUpdate" in types
    assert "sidebarButtonStatusUpdate" in types
    assert "serverSettings" in types
    clip = next(p for p in posted if p["type"] == "clipboardContentUpdate")
    assert clip["text"] == "from server"
    role = next(p for p in posted if p["type"] == "clientRoleUpdate")
    assert role["role"] == "controller"
    # onStats forwarded because the sidebar reported itself open
    assert sum(1 for t in types if t == "stats") >= 2
    # every post targets the same origin
    assert all(p["o"] == "https://host.example" for p in out["posted"])
