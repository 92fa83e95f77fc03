"""Touch/physical gamepad JS core, exercised headlessly under node.

The pure-logic core of selkies_amd/web/gamepad.js (wire encoder, layout
geometry, touch state machine) is a CommonJS module, so these tests run
it with the system node, feed synthetic touch sequences, and check the
emitted `js,*` wire verbs — then replay those exact verbs through the
real GamepadHub/SocketGamepad to prove the full browser->server->
interposer-socket path end to end.

Reference parity: addons/universal-touch-gamepad/universalTouchGamepad.js
(sticks with tap->L3/R3, 8-way d-pad, button overlay) and
addons/selkies-web-core/lib/gamepad.js (getGamepads polling).
"""

import asyncio
import json
import shutil
import struct
import subprocess
from pathlib import Path

import pytest

import selkies_amd.gamepad as G

JS = Path(__file__).resolve().parents[1] / "selkies_amd" / "web" / "gamepad.js"

node = shutil.which("node")
pytestmark = pytest.mark.skipif(node is None, reason="node not available")


def run_js(body: str) -> list[str]:
    """Run a JS snippet with the module loaded; return emitted wire verbs."""
    script = f"""
const G = require({json.dumps(str(JS))});
const out = [];
const send = (m) => out.push(m);
{body}
console.log(JSON.stringify(out));
"""
    p = subprocess.run([node, "-e", script], capture_output=True, text=True,
                       timeout=30)
    assert p.returncode == 0, p.stderr
    return json.loads(p.stdout.strip().splitlines()[-1])


def test_encoder_dedup_and_format():
    out = run_js("""
const e = new G.GamepadEncoder(send);
e.connect(0, "Pad", 18, 4);
e.connect(0, "Pad", 18, 4);        // duplicate: no resend
e.button(0, 0, true);
e.button(0, 0, true);              // unchanged: suppressed
e.button(0, 0, false);
e.axis(0, 1, 0.5004);              // quantized to 3 decimals
e.axis(0, 1, 0.5006);              // within epsilon: suppressed
e.axis(0, 1, 0.02);                // inside deadzone -> 0
e.button(1, 0, true);              // unconnected pad: dropped
e.disconnect(0);
e.disconnect(0);
""")
    assert out == [
        "js,c,0,UGFk,18,4",
        "js,b,0,0,1",
        "js,b,0,0,0",
        "js,a,0,1,0.5",
        "js,a,0,1,0",
        "js,d,0",
    ]


def test_physical_poll_maps_standard_gamepad():
    out = run_js("""
const e = new G.GamepadEncoder(send);
const pad = {connected: true, id: "X",
             buttons: [{pressed: true}, {pressed: false}],
             axes: [0.25, -1.5]};
G.pollPhysical(e, [pad, null]);
pad.buttons[0].pressed = false;
G.pollPhysical(e, [pad, null]);
G.pollPhysical(e, [null, null]);   // unplugged -> js,d
""")
    assert out[0] == "js,c,0,WA==,2,2"
    assert "js,b,0,0,1" in out and "js,a,0,0,0.25" in out
    assert "js,a,0,1,-1" in out          # clamped to [-1, 1]
    assert "js,b,0,0,0" in out[4:]
    assert out[-1] == "js,d,0"


def test_touch_stick_drag_and_release():
    out = run_js("""
let t = 0;
const e = new G.GamepadEncoder(send);
const c = new G.TouchGamepadCore(e, 3, "modern",
    {now: () => t, setTimeout: (f) => f(), aspect: 16/9});
// left stick is at (0.17, 0.74) r=0.12 — drag right half-deflection
c.touchStart(7, 0.17, 0.74);
t += 100;
c.touchMove(7, 0.17 + 0.06 * 9/16, 0.74);
t += 400;                           // slow: not a tap
c.touchEnd(7);
""")
    assert out[0].startswith("js,c,3,")
    assert "js,a,3,0,0.5" in out            # half deflection on axis 0
    assert "js,a,3,0,0" in out[2:]          # release recenters
    assert not any(",b,3,10," in m for m in out)   # no L3 tap


def test_touch_stick_tap_clicks_l3():
    out = run_js("""
let t = 0;
const e = new G.GamepadEncoder(send);
const c = new G.TouchGamepadCore(e, 3, "modern",
    {now: () => t, setTimeout: (f) => f(), aspect: 16/9});
c.touchStart(1, 0.17, 0.74);
t += 80;                            // quick, no travel -> stick click
c.touchEnd(1);
c.touchStart(2, 0.83, 0.74);        // right stick tap -> R3
t += 80;
c.touchEnd(2);
""")
    assert "js,b,3,10,1" in out and "js,b,3,10,0" in out
    assert "js,b,3,11,1" in out and "js,b,3,11,0" in out


def test_touch_dpad_8way_and_face_buttons():
    out = run_js("""
const e = new G.GamepadEncoder(send);
const c = new G.TouchGamepadCore(e, 3, "classic", {aspect: 1});
// classic d-pad center (0.16, 0.62) r=0.16: press right edge
c.touchStart(1, 0.16 + 0.12, 0.62);
c.touchMove(1, 0.16 + 0.09, 0.62 - 0.09);   // diagonal up-right
c.touchEnd(1);
// face button A (button 0) at (0.84, 0.72)
c.touchStart(2, 0.84, 0.72);
c.touchEnd(2);
c.detach();
""")
    assert "js,b,3,15,1" in out                  # right pressed
    i_up = out.index("js,b,3,12,1")              # diagonal adds up
    assert i_up > out.index("js,b,3,15,1")
    assert "js,b,3,15,0" in out[i_up:]           # release clears both
    assert "js,b,3,12,0" in out[i_up:]
    assert "js,b,3,0,1" in out and "js,b,3,0,0" in out
    assert out[-1] == "js,d,3"


def test_touch_outside_controls_ignored():
    out = run_js("""
const e = new G.GamepadEncoder(send);
const c = new G.TouchGamepadCore(e, 3, "modern", {aspect: 16/9});
const hit = c.touchStart(1, 0.5, 0.5);    // dead center: no control
send("hit=" + hit);
""")
    assert out[-1] == "hit=false"
    assert len(out) == 2                  # connect + the probe line


def test_layouts_fit_viewport():
    out = run_js("""
for (const prof of ["modern", "classic"]) {
  for (const c of G.layoutControls(prof)) {
    if (c.cx < 0 || c.cx > 1 || c.cy < 0 || c.cy > 1)
      send("offscreen:" + prof + ":" + c.id);
    if (c.kind === "button" && (c.button < 0 || c.button >= 18))
      send("badbutton:" + prof + ":" + c.id);
  }
}
send("ok");
""")
    assert out == ["ok"]


def test_touch_verbs_drive_socket_gamepad(tmp_path):
    """End to end: JS touch session -> wire verbs -> GamepadHub ->
    interposer socket events."""
    verbs = run_js("""
let t = 0;
const e = new G.GamepadEncoder(send);
const c = new G.TouchGamepadCore(e, 0, "modern",
    {now: () => t, setTimeout: (f) => f(), aspect: 16/9});
c.touchStart(1, 0.87, 0.485);       // face A (button 0)
c.touchEnd(1);
c.touchStart(2, 0.17, 0.74);        // left stick full right
c.touchMove(2, 0.17 + 0.2, 0.74);
t += 500;
c.touchEnd(2);
c.detach();
""")
    assert verbs[0].startswith("js,c,0,")

    async def main():
        hub = G.GamepadHub(socket_dir=str(tmp_path), prefer_uinput=False)
        await hub.handle(verbs[0])
        pad = hub.pads[0]
        reader, writer = await asyncio.open_unix_connection(pad.path)
        await reader.readexactly(G.JS_CONFIG.size)
        for v in verbs[1:]:
            await hub.handle(v)
        # button 0 press + release, axis 0 deflect + recenter all arrive
        # (events with the 0x80 INIT flag are the joydev initial-state
        # burst; skip them)
        seen = []
        while len(seen) < 6:
            ev = await asyncio.wait_for(reader.readexactly(8), timeout=5)
            _, value, etype, num = struct.unpack("<IhBB", ev)
            if etype & 0x80:
                continue
            seen.append((etype, num, value))
        writer.close()
        await hub.close()
        assert (1, 0, 1) in seen and (1, 0, 0) in seen      # button A
        axis_vals = [v for (t_, n, v) in seen if t_ == 2 and n == 0]
        assert max(axis_vals) > 30000 and 0 in axis_vals    # full deflect
    asyncio.new_event_loop().run_until_complete(main())
