"""Gamepad subsystem: config struct, joydev event ABI, socket server
round-trip, wire-verb handling (reference architecture SURVEY.md §2.4)."""

import asyncio
import base64
import struct

import pytest

from selkies_amd import gamepad as G


def test_js_event_abi():
    # struct js_event is 8 bytes: u32 time, s16 value, u8 type, u8 number
    assert G.JS_EVENT.size == 8
    b = G.JS_EVENT.pack(0x11223344, -32767, G.JS_EVENT_AXIS, 3)
    t, v, ty, num = G.JS_EVENT.unpack(b)
    assert (t, v, ty, num) == (0x11223344, -32767, 2, 3)


def test_input_event_abi():
    # 64-bit input_event is 24 bytes
    assert G.INPUT_EVENT.size == 24


def test_uinput_ioctl_numbers():
    # known kernel values: UI_DEV_CREATE = 0x5501, UI_DEV_DESTROY = 0x5502,
    # UI_SET_EVBIT = _IOW('U', 100, int) = 0x40045564
    assert G.UI_DEV_CREATE == 0x5501
    assert G.UI_DEV_DESTROY == 0x5502
    assert G.UI_SET_EVBIT == 0x40045564
    assert G.UI_SET_KEYBIT == 0x40045565
    assert G.UI_SET_ABSBIT == 0x40045567


def test_uinput_user_dev_layout():
    # name[80] + 4*u16 + u32 + 4*64*s32 = 80+8+4+1024 = 1116
    assert G.UINPUT_USER_DEV.size == 1116


def test_config_roundtrip():
    cfg = G.make_js_config("Test Pad", 11, 4)
    parsed = G.parse_js_config(cfg)
    assert parsed["name"] == "Test Pad"
    assert parsed["num_btns"] == 11
    assert parsed["num_axes"] == 8          # 4 sticks + triggers + hat
    assert parsed["vendor"] == 0x045E
    assert parsed["btn_map"][0] == G.BTN_A


def test_socket_gamepad_roundtrip(tmp_path):
    async def main():
        hub = G.GamepadHub(socket_dir=str(tmp_path), prefer_uinput=False)
        name = base64.b64encode(b"PadX").decode()
        await hub.handle(f"js,c,0,{name},11,4")
        assert 0 in hub.pads
        pad = hub.pads[0]

        reader, writer = await asyncio.open_unix_connection(pad.path)
        cfg = G.parse_js_config(await reader.readexactly(G.JS_CONFIG.size))
        assert cfg["name"] == "PadX"
        # initial INIT events: num_btns + num_axes
        n_init = cfg["num_btns"] + cfg["num_axes"]
        for _ in range(n_init):
            data = await reader.readexactly(8)
            _, _, ty, _ = G.JS_EVENT.unpack(data)
            assert ty & G.JS_EVENT_INIT
        # live events
        await asyncio.sleep(0.05)
        await hub.handle("js,b,0,0,1")
        await hub.handle("js,a,0,1,-0.5")
        t1 = G.JS_EVENT.unpack(await reader.readexactly(8))
        t2 = G.JS_EVENT.unpack(await reader.readexactly(8))
        assert t1[2] == G.JS_EVENT_BUTTON and t1[3] == 0 and t1[1] == 1
        assert t2[2] == G.JS_EVENT_AXIS and t2[3] == 1
        assert abs(t2[1] + 16383) < 10
        writer.close()
        await hub.close()

    asyncio.new_event_loop().run_until_complete(main())


def test_disconnect_removes_socket(tmp_path):
    async def main():
        hub = G.GamepadHub(socket_dir=str(tmp_path), prefer_uinput=False)
        await hub.handle("js,c,1")
        path = hub.pads[1].path
        import os
        assert os.path.exists(path)
        await hub.handle("js,d,1")
        assert not os.path.exists(path)
        await hub.close()

    asyncio.new_event_loop().run_until_complete(main())
