"""Minimal Wayland client used by the integration tests (and as the
reference consumer of the compositor): binds globals, maps an xdg
toplevel backed by a wl_shm buffer, and records seat/data-device
events."""

from __future__ import annotations

import mmap
import os
import socket
import time

from .wire import INTERFACES, MsgReader, marshal, send_msg, unmarshal


class WaylandClient:
    def __init__(self, display_name: str):
        runtime = os.environ.get("XDG_RUNTIME_DIR") or "/tmp"
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.connect(os.path.join(runtime, display_name))
        self.sock.setblocking(False)
        self.reader = MsgReader(self.sock)
        self.next_id = 2
        self.objects = {1: "wl_display"}
        self.globals = {}           # interface -> (name, version)
        self.outputs = {}           # oid -> dict
        self.events = []            # (iface, event, args)
        self.keymap = None
        self.selection_mimes = []
        self._offer = None
        self.registry = self._new("wl_registry")
        self._req(1, "wl_display", "get_registry", self.registry)
        self.roundtrip()

    # ---- plumbing ----------------------------------------------------------
    def _new(self, iface):
        oid = self.next_id
        self.next_id += 1
        self.objects[oid] = iface
        return oid

    def _req(self, oid, iface, name, *args):
        # fds ride inline at their signature position
        reqs = INTERFACES[iface]["requests"]
        for op, (rname, sig) in enumerate(reqs):
            if rname == name:
                payload, efds = marshal(oid, op, sig, list(args))
                send_msg(self.sock, payload, efds)
                return
        raise KeyError(f"{iface}.{name}")

    def pump(self, timeout=0.5):
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            import select
            r, _, _ = select.select([self.sock], [], [],
                                    max(0, deadline - time.monotonic()))
            if not r:
                break
            if not self.reader.pump():
                break
            for oid, opcode, body in self.reader.messages():
                self._event(oid, opcode, body)
        return self

    def roundtrip(self):
        cb = self._new("wl_callback")
        self._req(1, "wl_display", "sync", cb)
        self._done = False
        deadline = time.monotonic() + 5
        while not self._done and time.monotonic() < deadline:
            self.pump(0.1)
        return self

    def _event(self, oid, opcode, body):
        iface = self.objects.get(oid)
        if iface is None:
            return
        events = INTERFACES[iface]["events"]
        if opcode >= len(events):
            return
        name, sig = events[opcode]
        args = unmarshal(sig, body, self.reader.fds)
        self.events.append((iface, name, args))
        if iface == "wl_callback" and name == "done":
            self._done = True
        elif iface == "wl_registry" and name == "global":
            gname, giface, ver = args
            if giface == "wl_output":
                oid2 = self._new("wl_output")
                self._req(self.registry, "wl_registry", "bind", gname,
                          "wl_output", 3, oid2)
                self.outputs[oid2] = {"name": gname}
            else:
                self.globals[giface] = (gname, ver)
        elif iface == "wl_output":
            self.outputs.setdefault(oid, {})[name] = args
        elif iface == "wl_keyboard" and name == "keymap":
            fmt, fd, size = args
            mm = mmap.mmap(fd, size, prot=mmap.PROT_READ)
            self.keymap = mm[:size].decode(errors="replace")
            mm.close()
            os.close(fd)
        elif iface == "xdg_wm_base" and name == "ping":
            self._req(oid, "xdg_wm_base", "pong", args[0])
        elif iface == "xdg_surface" and name == "configure":
            self._req(oid, "xdg_surface", "ack_configure", args[0])
        elif iface == "wl_data_device" and name == "data_offer":
            self._offer = args[0]
            self.objects[args[0]] = "wl_data_offer"
            self._offer_mimes = []
        elif iface == "wl_data_offer" and name == "offer":
            self._offer_mimes.append(args[0])
        elif iface == "wl_data_device" and name == "selection":
            self.selection_mimes = list(getattr(self, "_offer_mimes", []))
            self.selection_offer = args[0] or None
        elif iface == "wl_data_source" and name == "send":
            mime, fd = args
            try:
                os.write(fd, getattr(self, "source_data", b""))
            finally:
                os.close(fd)

    # ---- conveniences ------------------------------------------------------
    def bind(self, giface):
        gname, ver = self.globals[giface]
        oid = self._new(giface)
        self._req(self.registry, "wl_registry", "bind", gname, giface,
                  ver, oid)
        return oid

    def setup_seat(self):
        self.seat = self.bind("wl_seat")
        self.keyboard = self._new("wl_keyboard")
        self._req(self.seat, "wl_seat", "get_keyboard", self.keyboard)
        self.pointer = self._new("wl_pointer")
        self._req(self.seat, "wl_seat", "get_pointer", self.pointer)
        return self.roundtrip()

    def map_window(self, w=320, h=200, title="win"):
        comp = self.bind("wl_compositor")
        shm = self.bind("wl_shm")
        wm = self.bind("xdg_wm_base")
        self.surface = self._new("wl_surface")
        self._req(comp, "wl_compositor", "create_surface", self.surface)
        xs = self._new("xdg_surface")
        self._req(wm, "xdg_wm_base", "get_xdg_surface", xs, self.surface)
        tl = self._new("xdg_toplevel")
        self._req(xs, "xdg_surface", "get_toplevel", tl)
        self._req(tl, "xdg_toplevel", "set_title", title)
        self.roundtrip()
        # shm buffer
        size = w * h * 4
        fd = os.memfd_create("selkies-client-buf")
        os.truncate(fd, size)
        self.shm_map = mmap.mmap(fd, size)
        self.shm_map[:] = b"\x42" * size
        pool = self._new("wl_shm_pool")
        self._req(shm, "wl_shm", "create_pool", pool, fd, size)
        os.close(fd)
        buf = self._new("wl_buffer")
        self._req(pool, "wl_shm_pool", "create_buffer", buf, 0, w, h,
                  w * 4, 1)
        self._req(self.surface, "wl_surface", "attach", buf, 0, 0)
        self._req(self.surface, "wl_surface", "commit")
        return self.roundtrip()

    def setup_data_device(self):
        mgr = self.bind("wl_data_device_manager")
        self.ddm = mgr
        self.data_device = self._new("wl_data_device")
        self._req(mgr, "wl_data_device_manager", "get_data_device",
                  self.data_device, self.seat)
        return self.roundtrip()

    def set_selection(self, text: bytes,
                      mime="text/plain;charset=utf-8"):
        self.source_data = text
        src = self._new("wl_data_source")
        self._req(self.ddm, "wl_data_device_manager", "create_data_source",
                  src)
        self._req(src, "wl_data_source", "offer", mime)
        self._req(self.data_device, "wl_data_device", "set_selection",
                  src, 1)
        return self.roundtrip()

    def read_selection(self, mime="text/plain;charset=utf-8", timeout=2.0):
        assert self.selection_offer
        r, w = os.pipe()
        self._req(self.selection_offer, "wl_data_offer", "receive", mime, w)
        os.close(w)
        os.set_blocking(r, False)
        out = b""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            self.pump(0.05)
            try:
                chunk = os.read(r, 65536)
                if not chunk:
                    break
                out += chunk
            except BlockingIOError:
                continue
        os.close(r)
        return out

    def key_events(self):
        return [a for i, n, a in self.events
                if i == "wl_keyboard" and n == "key"]

    def pointer_events(self, kind=None):
        return [(n, a) for i, n, a in self.events if i == "wl_pointer"
                and (kind is None or n == kind)]

    def close(self):
        self.sock.close()
