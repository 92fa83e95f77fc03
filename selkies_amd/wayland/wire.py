"""Wayland wire protocol: framing, argument marshalling, interface
tables (transcribed from wayland.xml / xdg-shell.xml message orders).

One shared table keeps compositor and client in sync; the framing
follows the protocol spec: each message is a 32-bit object id, then a
32-bit word with size in the upper 16 bits (bytes, including the 8-byte
header) and opcode in the lower 16. Args are 32-bit LE words; strings
are u32 length (incl NUL) + padded bytes; fds travel as SCM_RIGHTS
ancillary data in order of appearance.
"""

from __future__ import annotations

import array
import socket
import struct

# ---- interface message signatures -----------------------------------------
# sig chars: i=int32 u=uint32 f=fixed24.8 s=string n=new_id o=object
#            a=array h=fd  N=new_id with interface+version (registry.bind)

INTERFACES = {
    "wl_display": {
        "requests": [("sync", "n"), ("get_registry", "n")],
        "events": [("error", "ous"), ("delete_id", "u")],
    },
    "wl_registry": {
        "requests": [("bind", "uN")],
        "events": [("global", "usu"), ("global_remove", "u")],
    },
    "wl_callback": {
        "requests": [],
        "events": [("done", "u")],
    },
    "wl_compositor": {
        "requests": [("create_surface", "n"), ("create_region", "n")],
        "events": [],
    },
    "wl_surface": {
        "requests": [("destroy", ""), ("attach", "oii"), ("damage", "iiii"),
                     ("frame", "n"), ("set_opaque_region", "o"),
                     ("set_input_region", "o"), ("commit", ""),
                     ("set_buffer_transform", "i"),
                     ("set_buffer_scale", "i"),
                     ("damage_buffer", "iiii")],
        "events": [("enter", "o"), ("leave", "o")],
    },
    "wl_region": {
        "requests": [("destroy", ""), ("add", "iiii"),
                     ("subtract", "iiii")],
        "events": [],
    },
    "wl_shm": {
        "requests": [("create_pool", "nhi")],
        "events": [("format", "u")],
    },
    "wl_shm_pool": {
        "requests": [("create_buffer", "niiiiu"), ("destroy", ""),
                     ("resize", "i")],
        "events": [],
    },
    "wl_buffer": {
        "requests": [("destroy", "")],
        "events": [("release", "")],
    },
    "wl_seat": {
        "requests": [("get_pointer", "n"), ("get_keyboard", "n"),
                     ("get_touch", "n"), ("release", "")],
        "events": [("capabilities", "u"), ("name", "s")],
    },
    "wl_pointer": {
        "requests": [("set_cursor", "uoii"), ("release", "")],
        "events": [("enter", "uoff"), ("leave", "uo"), ("motion", "uff"),
                   ("button", "uuuu"), ("axis", "uuf"), ("frame", ""),
                   ("axis_source", "u"), ("axis_stop", "uu"),
                   ("axis_discrete", "ui")],
    },
    "wl_keyboard": {
        "requests": [("release", "")],
        "events": [("keymap", "uhu"), ("enter", "uoa"), ("leave", "uo"),
                   ("key", "uuuu"), ("modifiers", "uuuuu"),
                   ("repeat_info", "ii")],
    },
    "wl_output": {
        "requests": [("release", "")],
        "events": [("geometry", "iiiiissi"), ("mode", "uiii"),
                   ("done", ""), ("scale", "i")],
    },
    "xdg_wm_base": {
        "requests": [("destroy", ""), ("create_positioner", "n"),
                     ("get_xdg_surface", "no"), ("pong", "u")],
        "events": [("ping", "u")],
    },
    "xdg_surface": {
        "requests": [("destroy", ""), ("get_toplevel", "n"),
                     ("get_popup", "noo"), ("set_window_geometry", "iiii"),
                     ("ack_configure", "u")],
        "events": [("configure", "u")],
    },
    "xdg_toplevel": {
        "requests": [("destroy", ""), ("set_parent", "o"),
                     ("set_title", "s"), ("set_app_id", "s"),
                     ("show_window_menu", "ouii"), ("move", "ou"),
                     ("resize", "ouu"), ("set_max_size", "ii"),
                     ("set_min_size", "ii"), ("set_maximized", ""),
                     ("unset_maximized", ""), ("set_fullscreen", "o"),
                     ("unset_fullscreen", ""), ("set_minimized", "")],
        "events": [("configure", "iia"), ("close", "")],
    },
    "wl_data_device_manager": {
        "requests": [("create_data_source", "n"), ("get_data_device", "no")],
        "events": [],
    },
    "wl_data_source": {
        "requests": [("offer", "s"), ("destroy", ""), ("set_actions", "u")],
        "events": [("target", "s"), ("send", "sh"), ("cancelled", ""),
                   ("dnd_drop_performed", ""), ("dnd_finished", ""),
                   ("action", "u")],
    },
    "wl_data_device": {
        "requests": [("start_drag", "oou"), ("set_selection", "ou"),
                     ("release", "")],
        "events": [("data_offer", "n"), ("enter", "uoffo"), ("leave", ""),
                   ("motion", "uff"), ("drop", ""), ("selection", "o")],
    },
    "wl_data_offer": {
        "requests": [("accept", "us"), ("receive", "sh"), ("destroy", ""),
                     ("finish", ""), ("set_actions", "uu")],
        "events": [("offer", "s"), ("source_actions", "u"), ("action", "u")],
    },
}


def fixed(v: float) -> int:
    return int(round(v * 256.0)) & 0xFFFFFFFF


def unfixed(v: int) -> float:
    if v >= 1 << 31:
        v -= 1 << 32
    return v / 256.0


def marshal(obj_id: int, opcode: int, sig: str, args) -> tuple[bytes, list]:
    """Returns (payload bytes, fds)."""
    body = b""
    fds = []
    it = iter(args)
    for c in sig:
        if c in "iu":
            body += struct.pack("<i" if c == "i" else "<I",
                                next(it) & 0xFFFFFFFF
                                if c == "u" else next(it))
        elif c == "f":
            body += struct.pack("<I", fixed(next(it)))
        elif c in "no":
            body += struct.pack("<I", next(it) or 0)
        elif c == "s":
            s = next(it).encode() + b"\x00"
            body += struct.pack("<I", len(s)) + s + b"\x00" * (-len(s) % 4)
        elif c == "a":
            a = bytes(next(it))
            body += struct.pack("<I", len(a)) + a + b"\x00" * (-len(a) % 4)
        elif c == "h":
            fds.append(next(it))
        elif c == "N":
            # registry bind: interface string, version, new_id
            iface, version, new_id = next(it), next(it), next(it)
            s = iface.encode() + b"\x00"
            body += struct.pack("<I", len(s)) + s + b"\x00" * (-len(s) % 4)
            body += struct.pack("<II", version, new_id)
    hdr = struct.pack("<II", obj_id, ((len(body) + 8) << 16) | opcode)
    return hdr + body, fds


def unmarshal(sig: str, data: bytes, fds: list):
    """Decode args per sig; consumes fds from the front of `fds`."""
    out = []
    off = 0
    for c in sig:
        if c == "i":
            (v,) = struct.unpack_from("<i", data, off)
            out.append(v)
            off += 4
        elif c in "uno":
            (v,) = struct.unpack_from("<I", data, off)
            out.append(v)
            off += 4
        elif c == "f":
            (v,) = struct.unpack_from("<I", data, off)
            out.append(unfixed(v))
            off += 4
        elif c == "s":
            (n,) = struct.unpack_from("<I", data, off)
            off += 4
            out.append(data[off:off + n - 1].decode() if n else "")
            off += (n + 3) & ~3
        elif c == "a":
            (n,) = struct.unpack_from("<I", data, off)
            off += 4
            out.append(data[off:off + n])
            off += (n + 3) & ~3
        elif c == "h":
            out.append(fds.pop(0))
        elif c == "N":
            (n,) = struct.unpack_from("<I", data, off)
            off += 4
            iface = data[off:off + n - 1].decode()
            off += (n + 3) & ~3
            ver, nid = struct.unpack_from("<II", data, off)
            off += 8
            out.extend([iface, ver, nid])
    return out


def send_msg(sock: socket.socket, payload: bytes, fds: list) -> None:
    if fds:
        sock.sendmsg([payload], [(socket.SOL_SOCKET, socket.SCM_RIGHTS,
                                  array.array("i", fds).tobytes())])
    else:
        sock.sendall(payload)


class MsgReader:
    """Buffers a unix stream + ancillary fds and yields whole messages."""

    def __init__(self, sock: socket.socket):
        self.sock = sock
        self.buf = b""
        self.fds: list[int] = []

    def pump(self) -> bool:
        """Read once; False on EOF."""
        try:
            data, anc, _, _ = self.sock.recvmsg(65536, 4096)
        except (BlockingIOError, InterruptedError):
            return True
        if not data:
            return False
        for level, typ, raw in anc:
            if level == socket.SOL_SOCKET and typ == socket.SCM_RIGHTS:
                a = array.array("i")
                a.frombytes(raw[:len(raw) // 4 * 4])
                self.fds.extend(a.tolist())
        self.buf += data
        return True

    def messages(self):
        """Yield (obj_id, opcode, body) for complete buffered messages."""
        while len(self.buf) >= 8:
            obj_id, sz_op = struct.unpack_from("<II", self.buf)
            size = sz_op >> 16
            if size < 8 or len(self.buf) < size:
                return
            body = self.buf[8:size]
            self.buf = self.buf[size:]
            yield obj_id, sz_op & 0xFFFF, body
