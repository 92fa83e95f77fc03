"""Gamepad passthrough: interposer socket servers + uinput kernel backend.

Re-implements the reference gamepad architecture (SURVEY.md §2.4:
js-interposer LD_PRELOAD + SelkiesGamepad socket servers, UInputGamepad via
raw ioctls) with our own wire/struct contract:

* Browser events arrive as text verbs:
    js,c,<idx>,<b64 name>,<num_btns>,<num_axes>   connect pad
    js,b,<idx>,<button>,<0|1>                     button
    js,a,<idx>,<axis>,<float -1..1>               axis
    js,d,<idx>                                    disconnect
* Each pad runs a unix-socket server (SELKIES_JS_SOCKET_PATH/selkies_js<idx>.sock).
  A connecting client (the LD_PRELOAD interposer inside the app container)
  first receives a JsConfig struct, then a stream of Linux joydev
  `struct js_event` records (public kernel ABI).
* When /dev/uinput is writable, UInputGamepad creates a real kernel device
  instead (Xbox-360-compatible mapping), using pure-Python struct/ioctl
  math (same approach the reference validates against a C ground truth).
"""

from __future__ import annotations

import asyncio
import base64
import binascii
import logging
import os
import struct
import time
from typing import Optional

logger = logging.getLogger("selkies.gamepad")

# ---- public kernel ABI -----------------------------------------------------
# struct js_event { __u32 time; __s16 value; __u8 type; __u8 number; }
JS_EVENT = struct.Struct("<IhBB")
JS_EVENT_BUTTON = 0x01
JS_EVENT_AXIS = 0x02
JS_EVENT_INIT = 0x80

# struct input_event (64-bit): timeval(2x long) + u16 type u16 code s32 value
INPUT_EVENT = struct.Struct("<qqHHi")
EV_SYN, EV_KEY, EV_ABS = 0x00, 0x01, 0x03

# Xbox-360-style mapping from the browser Standard Gamepad
BTN_A, BTN_B, BTN_X, BTN_Y = 0x130, 0x131, 0x133, 0x134
BTN_TL, BTN_TR = 0x136, 0x137
BTN_SELECT, BTN_START, BTN_MODE = 0x13A, 0x13B, 0x13C
BTN_THUMBL, BTN_THUMBR = 0x13D, 0x13E
ABS_X, ABS_Y, ABS_Z = 0x00, 0x01, 0x02
ABS_RX, ABS_RY, ABS_RZ = 0x03, 0x04, 0x05
ABS_HAT0X, ABS_HAT0Y = 0x10, 0x11

# standard gamepad button index -> evdev button code (None => hat)
STANDARD_BTNS = [BTN_A, BTN_B, BTN_X, BTN_Y, BTN_TL, BTN_TR, None, None,
                 BTN_SELECT, BTN_START, BTN_THUMBL, BTN_THUMBR,
                 None, None, None, None, BTN_MODE]
# indices 6/7 are analog triggers (map to ABS_Z / ABS_RZ),
# 12..15 are the dpad (ABS_HAT0Y-, HAT0Y+, HAT0X-, HAT0X+)
STANDARD_AXES = [ABS_X, ABS_Y, ABS_RX, ABS_RY]

# ---- our interposer config struct ------------------------------------------
# magic "SJSG", u16 version, u16 vendor, u16 product, u16 num_btns,
# u16 num_axes, name[128], btn_map[64] u16 (evdev codes),
# axes_map[16] u8 (evdev ABS codes)
JS_CONFIG = struct.Struct("<4sHHHHH128s64H16B")
JS_CONFIG_MAGIC = b"SJSG"
JS_CONFIG_VERSION = 1


def make_js_config(name: str, num_btns: int, num_axes: int,
                   vendor: int = 0x045E, product: int = 0x028E) -> bytes:
    btn_map = [0] * 64
    for i in range(min(num_btns, len(STANDARD_BTNS))):
        btn_map[i] = STANDARD_BTNS[i] or 0
    axes_map = [0] * 16
    for i, code in enumerate([ABS_X, ABS_Y, ABS_Z, ABS_RX, ABS_RY, ABS_RZ,
                              ABS_HAT0X, ABS_HAT0Y][:min(16, num_axes + 4)]):
        axes_map[i] = code
    return JS_CONFIG.pack(JS_CONFIG_MAGIC, JS_CONFIG_VERSION, vendor,
                          product, num_btns, num_axes + 4,
                          name.encode()[:127], *btn_map, *axes_map)


def parse_js_config(data: bytes) -> dict:
    vals = JS_CONFIG.unpack(data)
    assert vals[0] == JS_CONFIG_MAGIC
    return {
        "version": vals[1], "vendor": vals[2], "product": vals[3],
        "num_btns": vals[4], "num_axes": vals[5],
        "name": vals[6].split(b"\0")[0].decode(),
        "btn_map": list(vals[7:7 + 64]),
        "axes_map": list(vals[7 + 64:7 + 64 + 16]),
    }


def now_ms32() -> int:
    return int(time.monotonic() * 1000) & 0xFFFFFFFF


class SocketGamepad:
    """Unix-socket joydev-protocol server for one pad."""

    def __init__(self, index: int, socket_dir: str, name: str = "Selkies "
                 "Virtual Gamepad", num_btns: int = 11, num_axes: int = 4):
        self.index = index
        self.path = os.path.join(socket_dir, f"selkies_js{index}.sock")
        self.config = make_js_config(name, num_btns, num_axes)
        self._server: Optional[asyncio.AbstractServer] = None
        self._writers: list[asyncio.StreamWriter] = []

    async def start(self):
        os.makedirs(os.path.dirname(self.path), exist_ok=True)
        try:
            os.unlink(self.path)
        except OSError:
            pass
        self._server = await asyncio.start_unix_server(self._on_client,
                                                       self.path)

    async def _on_client(self, reader, writer):
        writer.write(self.config)
        # initial neutral state (JS_EVENT_INIT records, joydev semantics)
        cfg = parse_js_config(self.config)
        for b in range(cfg["num_btns"]):
            writer.write(JS_EVENT.pack(now_ms32(), 0,
                                       JS_EVENT_BUTTON | JS_EVENT_INIT, b))
        for a in range(cfg["num_axes"]):
            writer.write(JS_EVENT.pack(now_ms32(), 0,
                                       JS_EVENT_AXIS | JS_EVENT_INIT, a))
        try:
            await writer.drain()
        except ConnectionError:
            return
        self._writers.append(writer)
        try:
            while await reader.read(4096):
                pass  # interposer never sends meaningful data back
        except ConnectionError:
            pass
        finally:
            if writer in self._writers:
                self._writers.remove(writer)
            writer.close()

    def _send(self, payload: bytes):
        for w in list(self._writers):
            try:
                w.write(payload)
            except Exception:
                if w in self._writers:
                    self._writers.remove(w)

    def button(self, number: int, pressed: bool):
        self._send(JS_EVENT.pack(now_ms32(), 1 if pressed else 0,
                                 JS_EVENT_BUTTON, number))

    def axis(self, number: int, value: float):
        v = max(-32767, min(32767, int(value * 32767)))
        self._send(JS_EVENT.pack(now_ms32(), v, JS_EVENT_AXIS, number))

    async def stop(self):
        if self._server:
            self._server.close()
            await self._server.wait_closed()
        for w in self._writers:
            w.close()
        self._writers.clear()
        try:
            os.unlink(self.path)
        except OSError:
            pass


# ---- uinput backend ---------------------------------------------------------
# ioctl request computation (public kernel _IOW/_IO macros)
def _IOW(typ, nr, size):
    return (1 << 30) | (size << 16) | (ord(typ) << 8) | nr


def _IO(typ, nr):
    return (ord(typ) << 8) | nr


UI_SET_EVBIT = _IOW("U", 100, 4)
UI_SET_KEYBIT = _IOW("U", 101, 4)
UI_SET_ABSBIT = _IOW("U", 103, 4)
UI_DEV_CREATE = _IO("U", 1)
UI_DEV_DESTROY = _IO("U", 2)

# struct uinput_user_dev: name[80], input_id{bus,vendor,product,version u16},
# ff_effects_max u32, absmax[64] s32, absmin[64] s32, absfuzz[64] s32,
# absflat[64] s32
UINPUT_USER_DEV = struct.Struct("<80sHHHHI" + "64i" * 4)
BUS_USB = 0x03


def uinput_writable(path: str = "/dev/uinput") -> bool:
    return os.access(path, os.W_OK)


class UInputGamepad:
    """Creates a real kernel gamepad device through /dev/uinput."""

    def __init__(self, name="Selkies Virtual Gamepad", vendor=0x045E,
                 product=0x028E):
        import fcntl
        self._fd = os.open("/dev/uinput", os.O_WRONLY | os.O_NONBLOCK)
        f = self._fd
        fcntl.ioctl(f, UI_SET_EVBIT, struct.pack("<i", EV_KEY))
        fcntl.ioctl(f, UI_SET_EVBIT, struct.pack("<i", EV_ABS))
        fcntl.ioctl(f, UI_SET_EVBIT, struct.pack("<i", EV_SYN))
        for code in [b for b in STANDARD_BTNS if b]:
            fcntl.ioctl(f, UI_SET_KEYBIT, struct.pack("<i", code))
        for code in (ABS_X, ABS_Y, ABS_Z, ABS_RX, ABS_RY, ABS_RZ,
                     ABS_HAT0X, ABS_HAT0Y):
            fcntl.ioctl(f, UI_SET_ABSBIT, struct.pack("<i", code))
        absmax = [0] * 64
        absmin = [0] * 64
        for code in (ABS_X, ABS_Y, ABS_RX, ABS_RY):
            absmax[code], absmin[code] = 32767, -32768
        for code in (ABS_Z, ABS_RZ):
            absmax[code], absmin[code] = 255, 0
        for code in (ABS_HAT0X, ABS_HAT0Y):
            absmax[code], absmin[code] = 1, -1
        dev = UINPUT_USER_DEV.pack(name.encode()[:79], BUS_USB, vendor,
                                   product, 0x110, 0,
                                   *absmax, *absmin, *([0] * 64),
                                   *([0] * 64))
        os.write(f, dev)
        import fcntl as _f
        _f.ioctl(f, UI_DEV_CREATE)

    def _emit(self, etype: int, code: int, value: int):
        os.write(self._fd, INPUT_EVENT.pack(0, 0, etype, code, value))

    def button(self, number: int, pressed: bool):
        code = STANDARD_BTNS[number] if number < len(STANDARD_BTNS) else None
        if code:
            self._emit(EV_KEY, code, 1 if pressed else 0)
        elif number in (6, 7):           # analog triggers
            self._emit(EV_ABS, ABS_Z if number == 6 else ABS_RZ,
                       255 if pressed else 0)
        elif number in (12, 13):         # dpad up/down
            self._emit(EV_ABS, ABS_HAT0Y,
                       (-1 if number == 12 else 1) if pressed else 0)
        elif number in (14, 15):         # dpad left/right
            self._emit(EV_ABS, ABS_HAT0X,
                       (-1 if number == 14 else 1) if pressed else 0)
        self._emit(EV_SYN, 0, 0)

    def axis(self, number: int, value: float):
        if number < len(STANDARD_AXES):
            self._emit(EV_ABS, STANDARD_AXES[number],
                       max(-32768, min(32767, int(value * 32767))))
            self._emit(EV_SYN, 0, 0)

    def close(self):
        import fcntl
        try:
            fcntl.ioctl(self._fd, UI_DEV_DESTROY)
        finally:
            os.close(self._fd)


class GamepadHub:
    """Owns all pads; consumes the js,* wire verbs."""

    MAX_PADS = 4

    def __init__(self, socket_dir: Optional[str] = None,
                 prefer_uinput: bool = True):
        self.socket_dir = socket_dir or os.environ.get(
            "SELKIES_JS_SOCKET_PATH", "/tmp/selkies_js")
        self.prefer_uinput = prefer_uinput
        self.pads: dict[int, object] = {}

    async def handle(self, msg: str):
        try:
            await self._handle(msg)
        except (ValueError, OverflowError, IndexError, binascii.Error):
            logger.debug("malformed js verb dropped: %r", msg[:80])

    async def _handle(self, msg: str):
        parts = msg.split(",")
        if len(parts) < 3 or parts[0] != "js":
            return
        op, idx = parts[1], int(parts[2])
        if idx < 0 or idx >= self.MAX_PADS:
            return
        if op == "c":
            name = base64.b64decode(parts[3]).decode() if len(parts) > 3 \
                else "Selkies Virtual Gamepad"
            nb = int(parts[4]) if len(parts) > 4 else 17
            na = int(parts[5]) if len(parts) > 5 else 4
            await self._connect(idx, name, nb, na)
        elif op == "d":
            await self._disconnect(idx)
        elif op == "b" and idx in self.pads:
            self.pads[idx].button(int(parts[3]), parts[4] == "1")
        elif op == "a" and idx in self.pads:
            self.pads[idx].axis(int(parts[3]), float(parts[4]))

    async def _connect(self, idx, name, nb, na):
        if idx in self.pads:
            return
        if self.prefer_uinput and uinput_writable():
            try:
                self.pads[idx] = UInputGamepad(name)
                logger.info("gamepad %d via uinput", idx)
                return
            except Exception as exc:
                logger.info("uinput unavailable (%r); using socket", exc)
        pad = SocketGamepad(idx, self.socket_dir, name, nb, na)
        await pad.start()
        self.pads[idx] = pad
        logger.info("gamepad %d via interposer socket %s", idx, pad.path)

    async def _disconnect(self, idx):
        pad = self.pads.pop(idx, None)
        if pad is None:
            return
        if isinstance(pad, SocketGamepad):
            await pad.stop()
        else:
            pad.close()

    async def close(self):
        for idx in list(self.pads):
            await self._disconnect(idx)
