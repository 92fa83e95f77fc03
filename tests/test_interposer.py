"""LD_PRELOAD joystick interposer <-> GamepadHub socket server, end to end:
a subprocess with the shim preloaded opens /dev/input/js0, queries joydev
ioctls, and reads live js_event records."""

import asyncio
import base64
import json
import os
import subprocess
import sys

import pytest

from selkies_amd import gamepad as G

SHIM_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "addons", "js-interposer")
SHIM = os.path.join(SHIM_DIR, "selkies_js_interposer.so")

CHILD = r"""
import array, fcntl, json, os, struct, sys
fd = os.open("/dev/input/js0", os.O_RDONLY)
out = {}
buf = array.array("B", [0])
fcntl.ioctl(fd, 0x80016A11, buf)            # JSIOCGAXES
out["axes"] = buf[0]
buf = array.array("B", [0])
fcntl.ioctl(fd, 0x80016A12, buf)            # JSIOCGBUTTONS
out["buttons"] = buf[0]
name = array.array("B", [0] * 64)
fcntl.ioctl(fd, 0x80406A13, name)           # JSIOCGNAME(64)
out["name"] = name.tobytes().split(b"\0")[0].decode()
ver = array.array("I", [0])
fcntl.ioctl(fd, 0x80046A01, ver)            # JSIOCGVERSION
out["version"] = ver[0]
events = []
# skip INIT events, then read two live ones
while len(events) < 2:
    t, v, ty, num = struct.unpack("<IhBB", os.read(fd, 8))
    if ty & 0x80:
        continue
    events.append([v, ty, num])
out["events"] = events
os.close(fd)
print(json.dumps(out))
"""


@pytest.fixture(scope="module")
def shim():
    subprocess.run(["make", "-C", SHIM_DIR], check=True,
                   capture_output=True)
    assert os.path.exists(SHIM)
    return SHIM


def test_interposer_end_to_end(shim, tmp_path):
    async def main():
        hub = G.GamepadHub(socket_dir=str(tmp_path), prefer_uinput=False)
        name_b64 = base64.b64encode(b"Selkies X360").decode()
        await hub.handle(f"js,c,0,{name_b64},11,4")

        env = dict(os.environ, LD_PRELOAD=shim,
                   SELKIES_JS_SOCKET_PATH=str(tmp_path))
        proc = await asyncio.create_subprocess_exec(
            sys.executable, "-c", CHILD, env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE)

        # feed events once the client has had time to connect
        await asyncio.sleep(0.5)
        await hub.handle("js,b,0,0,1")
        await hub.handle("js,a,0,1,0.5")
        try:
            out, err = await asyncio.wait_for(proc.communicate(), 10)
        finally:
            if proc.returncode is None:
                proc.kill()
        assert proc.returncode == 0, err.decode()
        data = json.loads(out.decode())
        assert data["name"] == "Selkies X360"
        assert data["buttons"] == 11
        assert data["axes"] == 8
        assert data["version"] >= 0x020100
        ev = data["events"]
        assert ev[0] == [1, G.JS_EVENT_BUTTON, 0]
        assert ev[1][1] == G.JS_EVENT_AXIS and ev[1][2] == 1
        assert abs(ev[1][0] - 16383) < 10
        await hub.close()

    asyncio.new_event_loop().run_until_complete(main())


EVDEV_CHILD = r"""
import array, fcntl, json, os, stat, struct, sys
st = os.stat("/dev/input/event1000")
out = {"is_chr": stat.S_ISCHR(st.st_mode)}
fd = os.open("/dev/input/event1000", os.O_RDONLY)
ver = array.array("i", [0])
fcntl.ioctl(fd, 0x80044501, ver)            # EVIOCGVERSION
out["version"] = ver[0]
iid = array.array("H", [0] * 4)
fcntl.ioctl(fd, 0x80084502, iid)            # EVIOCGID
out["id"] = iid.tolist()
name = array.array("B", [0] * 64)
fcntl.ioctl(fd, 0x80404506, name)           # EVIOCGNAME(64)
out["name"] = name.tobytes().split(b"\0")[0].decode()
bits = array.array("B", [0] * 8)
fcntl.ioctl(fd, 0x80084520, bits)           # EVIOCGBIT(0, 8)
out["evtypes"] = bits[0]
keybits = array.array("B", [0] * 96)
fcntl.ioctl(fd, 0x80604521, keybits)        # EVIOCGBIT(EV_KEY, 96)
btn_a = 0x130
out["has_btn_a"] = bool(keybits[btn_a // 8] & (1 << (btn_a % 8)))
absinfo = array.array("i", [0] * 6)
fcntl.ioctl(fd, 0x80184540, absinfo)        # EVIOCGABS(ABS_X)
out["abs_x"] = absinfo.tolist()
events = []
while len(events) < 4:
    raw = os.read(fd, 24)
    sec, usec, etype, code, value = struct.unpack("<QQHHi", raw)
    events.append([etype, code, value])
out["events"] = events
os.close(fd)
print(json.dumps(out))
"""


def test_interposer_evdev_surface(shim, tmp_path):
    """SDL/evdev-style consumption: /dev/input/event1000 stats as a char
    device, answers the EVIOC* family, and read() delivers translated
    input_event records with EV_SYN framing."""
    async def main():
        hub = G.GamepadHub(socket_dir=str(tmp_path), prefer_uinput=False)
        name_b64 = base64.b64encode(b"Selkies X360").decode()
        await hub.handle(f"js,c,0,{name_b64},11,4")

        env = dict(os.environ, LD_PRELOAD=shim,
                   SELKIES_JS_SOCKET_PATH=str(tmp_path))
        proc = await asyncio.create_subprocess_exec(
            sys.executable, "-c", EVDEV_CHILD, env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE)
        await asyncio.sleep(0.5)
        await hub.handle("js,b,0,0,1")     # button 0 (BTN_A) down
        await hub.handle("js,a,0,0,-0.5")  # ABS_X left
        try:
            out, err = await asyncio.wait_for(proc.communicate(), 10)
        finally:
            if proc.returncode is None:
                proc.kill()
        assert proc.returncode == 0, err.decode()
        data = json.loads(out.decode())
        assert data["is_chr"]
        assert data["version"] == 0x010001
        assert data["id"][0] == 3          # BUS_USB
        assert data["name"] == "Selkies X360"
        assert data["evtypes"] & 0b1011 == 0b1011   # SYN|KEY|ABS
        assert data["has_btn_a"]
        assert data["abs_x"][1:3] == [-32767, 32767]
        ev = data["events"]
        # button down + SYN, axis + SYN
        assert ev[0][:2] == [1, 0x130] and ev[0][2] == 1
        assert ev[1] == [0, 0, 0]
        assert ev[2][0] == 3 and abs(ev[2][2] + 16383) < 10
        assert ev[3] == [0, 0, 0]
        await hub.close()

    asyncio.new_event_loop().run_until_complete(main())
