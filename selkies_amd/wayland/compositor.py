"""Headless Wayland compositor (phase 1 of the Wayland backend).

The image ships no wayland-server library or wlroots, so the compositor
speaks the wire protocol directly (wire.py) on a unix socket in
XDG_RUNTIME_DIR, on its own thread with a selectors loop. It provides
the surface the pixelflux contract needs:

* globals: wl_compositor, wl_shm, wl_seat (keyboard+pointer),
  xdg_wm_base, wl_data_device_manager, one wl_output per configured
  output;
* output management: create/reposition/destroy outputs (globals come and
  go), list_outputs;
* window management: xdg toplevels tracked with title/app_id/geometry,
  list_windows, move_window_to_output, focus;
* input injection: inject_key / inject_mouse_* deliver real
  wl_keyboard/wl_pointer events to the focused client (keymap delivered
  as an xkb_v1 memfd per the protocol);
* clipboard data-control: set_clipboard (compositor-owned selection),
  clipboard_types_app / clipboard_read_app (reads a client's selection
  through wl_data_offer.receive), clipboard_write_app, clipboard_clear.

Committed shm buffers stay mapped, so `composite_frame()` renders the
current desktop into a BGRX array — the capture seam for the encode
engine's Wayland backend.
"""

from __future__ import annotations

import errno
import mmap
import os
import selectors
import socket
import struct
import threading
import time
from typing import Optional

from .wire import INTERFACES, MsgReader, marshal, send_msg, unmarshal

# minimal us(qwerty) xkb_v1 keymap: keycodes are evdev code + 8 (the
# X11-compatible offset every toolkit expects)
XKB_KEYMAP = """xkb_keymap {
xkb_keycodes "evdev" {
    minimum = 8;
    maximum = 255;
    <ESC> = 9;   <AE01> = 10; <AE02> = 11; <AE03> = 12; <AE04> = 13;
    <AE05> = 14; <AE06> = 15; <AE07> = 16; <AE08> = 17; <AE09> = 18;
    <AE10> = 19; <AD01> = 24; <AD02> = 25; <AD03> = 26; <AD04> = 27;
    <AD05> = 28; <AD06> = 29; <AD07> = 30; <AD08> = 31; <AD09> = 32;
    <AD10> = 33; <AC01> = 38; <AC02> = 39; <AC03> = 40; <AC04> = 41;
    <AC05> = 42; <AC06> = 43; <AC07> = 44; <AC08> = 45; <AC09> = 46;
    <AC10> = 47; <AB01> = 52; <AB02> = 53; <AB03> = 54; <AB04> = 55;
    <AB05> = 56; <AB06> = 57; <AB07> = 58; <SPCE> = 65; <RTRN> = 36;
    <LFSH> = 50; <LCTL> = 37; <LALT> = 64; <BKSP> = 22; <TAB> = 23;
};
xkb_types "basic" { type "ONE_LEVEL" { modifiers = none; level_name[1] = "Any"; }; };
xkb_compatibility "basic" { };
xkb_symbols "us" {
    key <ESC> { [ Escape ] }; key <SPCE> { [ space ] };
    key <RTRN> { [ Return ] }; key <BKSP> { [ BackSpace ] };
    key <TAB> { [ Tab ] };
    key <AE01> { [ 1 ] }; key <AE02> { [ 2 ] }; key <AE03> { [ 3 ] };
    key <AE04> { [ 4 ] }; key <AE05> { [ 5 ] }; key <AE06> { [ 6 ] };
    key <AE07> { [ 7 ] }; key <AE08> { [ 8 ] }; key <AE09> { [ 9 ] };
    key <AE10> { [ 0 ] };
    key <AD01> { [ q ] }; key <AD02> { [ w ] }; key <AD03> { [ e ] };
    key <AD04> { [ r ] }; key <AD05> { [ t ] }; key <AD06> { [ y ] };
    key <AD07> { [ u ] }; key <AD08> { [ i ] }; key <AD09> { [ o ] };
    key <AD10> { [ p ] };
    key <AC01> { [ a ] }; key <AC02> { [ s ] }; key <AC03> { [ d ] };
    key <AC04> { [ f ] }; key <AC05> { [ g ] }; key <AC06> { [ h ] };
    key <AC07> { [ j ] }; key <AC08> { [ k ] }; key <AC09> { [ l ] };
    key <AC10> { [ semicolon ] };
    key <AB01> { [ z ] }; key <AB02> { [ x ] }; key <AB03> { [ c ] };
    key <AB04> { [ v ] }; key <AB05> { [ b ] }; key <AB06> { [ n ] };
    key <AB07> { [ m ] };
    key <LFSH> { [ Shift_L ] }; key <LCTL> { [ Control_L ] };
    key <LALT> { [ Alt_L ] };
};
};
"""



# keysym -> X keycode for the XKB_KEYMAP above (evdev code = keycode - 8);
# used by the Wayland input backend to map wire-protocol keysyms
KEYSYM_TO_XKEYCODE = {}
def _build_keysym_map():
    rows = {
        9: 0xFF1B, 36: 0xFF0D, 65: 0x20, 22: 0xFF08, 23: 0xFF09,
        50: 0xFFE1, 37: 0xFFE3, 64: 0xFFE9,
    }
    for i, ch in enumerate("1234567890"):
        rows[10 + i] = ord(ch)
    for i, ch in enumerate("qwertyuiop"):
        rows[24 + i] = ord(ch)
    for i, ch in enumerate("asdfghjkl;"):
        rows[38 + i] = ord(ch)
    for i, ch in enumerate("zxcvbnm"):
        rows[52 + i] = ord(ch)
    for code, ks in rows.items():
        KEYSYM_TO_XKEYCODE[ks] = code
        # uppercase letters share the keycode
        if ord('a') <= ks <= ord('z'):
            KEYSYM_TO_XKEYCODE[ks - 32] = code
_build_keysym_map()


class Output:
    def __init__(self, name_id: int, w: int, h: int, x: int = 0, y: int = 0,
                 label: str = "HEADLESS-1"):
        self.name_id = name_id          # registry global name
        self.w, self.h, self.x, self.y = w, h, x, y
        self.label = label


class Surface:
    def __init__(self, client, oid):
        self.client = client
        self.oid = oid
        self.pending_buffer = None      # (pool, offset, w, h, stride, fmt)
        self.buffer = None
        self.xdg_surface = None
        self.toplevel = None
        self.title = ""
        self.app_id = ""
        self.w = 0
        self.h = 0
        self.output: Optional[Output] = None
        self.mapped = False


class Client:
    _next = [1]

    def __init__(self, comp, sock):
        self.comp = comp
        self.sock = sock
        self.reader = MsgReader(sock)
        self.objects: dict[int, tuple[str, object]] = {
            1: ("wl_display", None)}
        self.cid = Client._next[0]
        Client._next[0] += 1
        self.keyboards: list[int] = []
        self.pointers: list[int] = []
        self.data_devices: list[int] = []
        self.serial = 0
        self.pools: dict[int, tuple[mmap.mmap, int]] = {}
        self.selection_source: Optional[int] = None  # wl_data_source oid
        self.selection_mimes: list[str] = []
        self.dead = False

    def next_serial(self):
        self.serial += 1
        return self.serial

    def send(self, oid: int, iface: str, event: str, *args):
        # fds ride INLINE at their signature position ('h' args)
        events = INTERFACES[iface]["events"]
        for op, (name, sig) in enumerate(events):
            if name == event:
                payload, efds = marshal(oid, op, sig, list(args))
                try:
                    send_msg(self.sock, payload, efds)
                except OSError:
                    self.dead = True
                return
        raise KeyError(f"{iface}.{event}")


class Compositor:
    """One instance per WAYLAND_DISPLAY socket."""

    def __init__(self, display_name: str = "selkies-wl-0",
                 width: int = 1920, height: int = 1080):
        self.display_name = display_name
        runtime = os.environ.get("XDG_RUNTIME_DIR") or "/tmp"
        self.path = os.path.join(runtime, display_name)
        self.outputs: list[Output] = [Output(100, width, height)]
        self._next_output_name = 101
        self.clients: list[Client] = []
        self.surfaces: list[Surface] = []
        self.focus: Optional[Surface] = None
        self.pointer_xy = (0.0, 0.0)
        self.pointer_focus: Optional[Surface] = None
        # compositor-owned clipboard ('' = none); app selection wins when set
        self.clipboard_text: Optional[str] = None
        self.clipboard_mime = "text/plain;charset=utf-8"
        self._lock = threading.RLock()
        self._sel = selectors.DefaultSelector()
        self._wake_r, self._wake_w = os.pipe()
        self._pending: list = []
        self._thread: Optional[threading.Thread] = None
        self._stop = False

        try:
            os.unlink(self.path)
        except FileNotFoundError:
            pass
        self._listen = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self._listen.bind(self.path)
        self._listen.listen(8)
        self._listen.setblocking(False)

    # ---- lifecycle ---------------------------------------------------------
    def start(self):
        self._sel.register(self._listen, selectors.EVENT_READ, "listen")
        self._sel.register(self._wake_r, selectors.EVENT_READ, "wake")
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="selkies-wayland")
        self._thread.start()
        return self

    def stop(self):
        self._stop = True
        os.write(self._wake_w, b"x")
        if self._thread:
            self._thread.join(timeout=5)
        try:
            os.unlink(self.path)
        except OSError:
            pass

    def _post(self, fn):
        """Run fn on the compositor thread (thread-safe injection)."""
        with self._lock:
            self._pending.append(fn)
        os.write(self._wake_w, b"x")

    def _run(self):
        while not self._stop:
            for key, _ in self._sel.select(timeout=0.5):
                if key.data == "wake":
                    os.read(self._wake_r, 4096)
                elif key.data == "listen":
                    try:
                        s, _ = self._listen.accept()
                    except OSError:
                        continue
                    s.setblocking(False)
                    c = Client(self, s)
                    self.clients.append(c)
                    self._sel.register(s, selectors.EVENT_READ, c)
                else:
                    self._service(key.data)
            with self._lock:
                pending, self._pending = self._pending, []
            for fn in pending:
                try:
                    fn()
                except Exception:
                    pass
            self._reap()

    def _reap(self):
        for c in [c for c in self.clients if c.dead]:
            try:
                self._sel.unregister(c.sock)
            except Exception:
                pass
            try:
                c.sock.close()
            except OSError:
                pass
            self.clients.remove(c)
            self.surfaces = [s for s in self.surfaces if s.client is not c]
            if self.focus and self.focus.client is c:
                self.focus = next((s for s in self.surfaces if s.mapped),
                                  None)

    def _service(self, c: Client):
        if not c.reader.pump():
            c.dead = True
            return
        for oid, opcode, body in c.reader.messages():
            entry = c.objects.get(oid)
            if entry is None:
                continue
            iface, state = entry
            reqs = INTERFACES[iface]["requests"]
            if opcode >= len(reqs):
                continue
            name, sig = reqs[opcode]
            try:
                args = unmarshal(sig, body, c.reader.fds)
            except Exception:
                c.dead = True
                return
            try:
                self._dispatch(c, oid, iface, state, name, args)
            except Exception:
                import traceback
                traceback.print_exc()
                c.dead = True
                return

    # ---- request handling --------------------------------------------------
    GLOBALS = [("wl_compositor", 4), ("wl_shm", 1), ("wl_seat", 7),
               ("xdg_wm_base", 2), ("wl_data_device_manager", 3)]

    def _dispatch(self, c: Client, oid, iface, state, name, args):
        if iface == "wl_display":
            if name == "sync":
                cb = args[0]
                c.objects[cb] = ("wl_callback", None)
                c.send(cb, "wl_callback", "done", c.next_serial())
                c.send(1, "wl_display", "delete_id", cb)
            elif name == "get_registry":
                reg = args[0]
                c.objects[reg] = ("wl_registry", None)
                for i, (gname, ver) in enumerate(self.GLOBALS):
                    c.send(reg, "wl_registry", "global", i + 1, gname, ver)
                for out in self.outputs:
                    c.send(reg, "wl_registry", "global", out.name_id,
                           "wl_output", 3)
        elif iface == "wl_registry" and name == "bind":
            gname, giface, ver, nid = args
            c.objects[nid] = (giface, None)
            if giface == "wl_shm":
                c.send(nid, "wl_shm", "format", 0)      # argb8888
                c.send(nid, "wl_shm", "format", 1)      # xrgb8888
            elif giface == "wl_seat":
                c.send(nid, "wl_seat", "capabilities", 3)  # ptr|kbd
                c.send(nid, "wl_seat", "name", "seat0")
            elif giface == "wl_output":
                out = next((o for o in self.outputs
                            if o.name_id == gname), self.outputs[0])
                c.objects[nid] = ("wl_output", out)
                self._send_output_info(c, nid, out)
        elif iface == "wl_compositor" and name == "create_surface":
            s = Surface(c, args[0])
            c.objects[args[0]] = ("wl_surface", s)
            self.surfaces.append(s)
        elif iface == "wl_surface":
            self._surface_req(c, state, name, args)
        elif iface == "wl_shm" and name == "create_pool":
            nid, fd, size = args
            mm = mmap.mmap(fd, size)
            os.close(fd)
            c.objects[nid] = ("wl_shm_pool", nid)
            c.pools[nid] = (mm, size)
        elif iface == "wl_shm_pool":
            if name == "create_buffer":
                nid, off, w, h, stride, fmt = args
                mm, _ = c.pools[state]
                c.objects[nid] = ("wl_buffer", (mm, off, w, h, stride, fmt))
            elif name == "destroy":
                c.objects.pop(oid, None)
        elif iface == "wl_buffer" and name == "destroy":
            c.objects.pop(oid, None)
            c.send(1, "wl_display", "delete_id", oid)
        elif iface == "wl_seat":
            if name == "get_keyboard":
                kid = args[0]
                c.objects[kid] = ("wl_keyboard", None)
                c.keyboards.append(kid)
                self._send_keymap(c, kid)
                if self.focus and self.focus.client is c:
                    c.send(kid, "wl_keyboard", "enter", c.next_serial(),
                           self.focus.oid, b"")
            elif name == "get_pointer":
                pid = args[0]
                c.objects[pid] = ("wl_pointer", None)
                c.pointers.append(pid)
        elif iface == "xdg_wm_base":
            if name == "get_xdg_surface":
                nid, surf_oid = args
                s = c.objects[surf_oid][1]
                s.xdg_surface = nid
                c.objects[nid] = ("xdg_surface", s)
            elif name == "pong":
                pass
        elif iface == "xdg_surface":
            s = state
            if name == "get_toplevel":
                s.toplevel = args[0]
                c.objects[args[0]] = ("xdg_toplevel", s)
                s.output = self.outputs[0]
                # initial configure: let the client pick its size
                c.send(args[0], "xdg_toplevel", "configure", 0, 0, b"")
                c.send(s.xdg_surface, "xdg_surface", "configure",
                       c.next_serial())
            elif name == "ack_configure":
                pass
        elif iface == "xdg_toplevel":
            s = state
            if name == "set_title":
                s.title = args[0]
            elif name == "set_app_id":
                s.app_id = args[0]
            elif name == "destroy":
                s.mapped = False
                c.objects.pop(oid, None)
        elif iface == "wl_data_device_manager":
            if name == "create_data_source":
                c.objects[args[0]] = ("wl_data_source", [])
            elif name == "get_data_device":
                c.objects[args[0]] = ("wl_data_device", None)
                c.data_devices.append(args[0])
        elif iface == "wl_data_source":
            if name == "offer":
                state.append(args[0])
            elif name == "destroy":
                if c.selection_source == oid:
                    c.selection_source = None
                c.objects.pop(oid, None)
        elif iface == "wl_data_device":
            if name == "set_selection":
                src = args[0]
                c.selection_source = src or None
                c.selection_mimes = (list(c.objects[src][1])
                                     if src else [])
                if src:
                    self.clipboard_text = None  # app selection wins
                self._broadcast_selection(owner=c)
        elif iface == "wl_data_offer":
            if name == "receive":
                mime, fd = args
                self._offer_receive(c, state, mime, fd)
            elif name == "destroy":
                c.objects.pop(oid, None)

    def _surface_req(self, c, s: Surface, name, args):
        if name == "attach":
            buf_oid = args[0]
            s.pending_buffer = (c.objects[buf_oid][1]
                                if buf_oid else None)
        elif name == "commit":
            if s.pending_buffer is not None:
                s.buffer = s.pending_buffer
                _, _, w, h, _, _ = s.buffer
                s.w, s.h = w, h
                if not s.mapped and s.toplevel is not None:
                    s.mapped = True
                    if self.focus is None:
                        self._set_focus(s)
        elif name == "destroy":
            s.mapped = False
            if s in self.surfaces:
                self.surfaces.remove(s)
            if self.focus is s:
                self.focus = None

    # ---- seat / focus ------------------------------------------------------
    def _send_keymap(self, c: Client, kid: int):
        data = XKB_KEYMAP.encode()
        fd = os.memfd_create("selkies-xkb")
        os.write(fd, data)
        os.lseek(fd, 0, os.SEEK_SET)
        c.send(kid, "wl_keyboard", "keymap", 1, fd, len(data))
        os.close(fd)
        c.send(kid, "wl_keyboard", "repeat_info", 25, 400)

    def _set_focus(self, s: Optional[Surface]):
        old = self.focus
        if old is s:
            return
        if old is not None:
            for kid in old.client.keyboards:
                old.client.send(kid, "wl_keyboard", "leave",
                                old.client.next_serial(), old.oid)
        self.focus = s
        if s is not None:
            for kid in s.client.keyboards:
                s.client.send(kid, "wl_keyboard", "enter",
                              s.client.next_serial(), s.oid, b"")
            self._send_selection_to(s.client)

    def _ms(self):
        return int(time.monotonic() * 1000) & 0x7FFFFFFF

    # ---- public control API (thread-safe; mirrors the pixelflux wayland
    # contract) ---------------------------------------------
    def list_outputs(self):
        return [{"name": o.name_id, "label": o.label, "x": o.x, "y": o.y,
                 "width": o.w, "height": o.h} for o in self.outputs]

    def create_output(self, width, height, x=0, y=0, label=None):
        out = Output(self._next_output_name, width, height, x, y,
                     label or f"HEADLESS-{self._next_output_name}")
        self._next_output_name += 1
        self.outputs.append(out)

        def announce():
            for c in self.clients:
                for oid, (iface, st) in list(c.objects.items()):
                    if iface == "wl_registry":
                        c.send(oid, "wl_registry", "global", out.name_id,
                               "wl_output", 3)
        self._post(announce)
        return out.name_id

    def reposition_output(self, name_id, x, y, width=None, height=None):
        for o in self.outputs:
            if o.name_id == name_id:
                o.x, o.y = x, y
                if width:
                    o.w = width
                if height:
                    o.h = height
                def update():
                    for c in self.clients:
                        for oid, (iface, st) in list(c.objects.items()):
                            if iface == "wl_output" and st is o:
                                self._send_output_info(c, oid, o)
                self._post(update)
                return True
        return False

    def destroy_output(self, name_id):
        for o in list(self.outputs):
            if o.name_id == name_id and len(self.outputs) > 1:
                self.outputs.remove(o)
                def remove():
                    for c in self.clients:
                        for oid, (iface, st) in list(c.objects.items()):
                            if iface == "wl_registry":
                                c.send(oid, "wl_registry", "global_remove",
                                       o.name_id)
                self._post(remove)
                return True
        return False

    def _send_output_info(self, c, oid, out):
        c.send(oid, "wl_output", "geometry", out.x, out.y, 520, 320, 0,
               "selkies", out.label, 0)
        c.send(oid, "wl_output", "mode", 3, out.w, out.h, 60000)
        c.send(oid, "wl_output", "scale", 1)
        c.send(oid, "wl_output", "done")

    def list_windows(self):
        return [{"id": s.oid * 1000 + s.client.cid, "title": s.title,
                 "app_id": s.app_id, "width": s.w, "height": s.h,
                 "output": s.output.name_id if s.output else None,
                 "focused": s is self.focus}
                for s in self.surfaces if s.mapped]

    def move_window_to_output(self, win_id, output_name):
        for s in self.surfaces:
            if s.oid * 1000 + s.client.cid == win_id:
                for o in self.outputs:
                    if o.name_id == output_name:
                        s.output = o
                        return True
        return False

    def get_realized_geometry(self):
        xs = [o.x for o in self.outputs]
        ys = [o.y for o in self.outputs]
        x2 = [o.x + o.w for o in self.outputs]
        y2 = [o.y + o.h for o in self.outputs]
        return {"x": min(xs), "y": min(ys), "width": max(x2) - min(xs),
                "height": max(y2) - min(ys)}

    # ---- input injection ---------------------------------------------------
    def inject_key(self, evdev_code: int, pressed: bool):
        def do():
            s = self.focus
            if not s:
                return
            c = s.client
            for kid in c.keyboards:
                c.send(kid, "wl_keyboard", "key", c.next_serial(),
                       self._ms(), evdev_code, 1 if pressed else 0)
        self._post(do)

    def inject_mouse_move(self, x: float, y: float):
        def do():
            self.pointer_xy = (x, y)
            s = self.focus
            if not s:
                return
            c = s.client
            for pid in c.pointers:
                if self.pointer_focus is not s:
                    c.send(pid, "wl_pointer", "enter", c.next_serial(),
                           s.oid, x, y)
                c.send(pid, "wl_pointer", "motion", self._ms(), x, y)
                c.send(pid, "wl_pointer", "frame")
            self.pointer_focus = s
        self._post(do)

    def inject_mouse_button(self, button: int, pressed: bool):
        def do():
            s = self.pointer_focus or self.focus
            if not s:
                return
            c = s.client
            for pid in c.pointers:
                c.send(pid, "wl_pointer", "button", c.next_serial(),
                       self._ms(), button, 1 if pressed else 0)
                c.send(pid, "wl_pointer", "frame")
        self._post(do)

    def inject_mouse_scroll(self, dx: float, dy: float):
        def do():
            s = self.pointer_focus or self.focus
            if not s:
                return
            c = s.client
            for pid in c.pointers:
                if dy:
                    c.send(pid, "wl_pointer", "axis", self._ms(), 0, dy)
                if dx:
                    c.send(pid, "wl_pointer", "axis", self._ms(), 1, dx)
                c.send(pid, "wl_pointer", "frame")
        self._post(do)

    def get_xkb_keymap_string(self) -> str:
        return XKB_KEYMAP

    # ---- clipboard (data-control) -----------------------------------------
    def set_clipboard(self, text: str,
                      mime: str = "text/plain;charset=utf-8"):
        """Compositor-owned selection offered to every data device."""
        def do():
            self.clipboard_text = text
            self.clipboard_mime = mime
            for c in self.clients:
                c.selection_source = None
            self._broadcast_selection(owner=None)
        self._post(do)

    def clipboard_clear(self):
        self.set_clipboard("")

    def _broadcast_selection(self, owner: Optional[Client]):
        for c in self.clients:
            self._send_selection_to(c, owner)

    def _send_selection_to(self, c: Client, owner: Optional[Client] = None):
        if owner is None:
            owner = next((cl for cl in self.clients
                          if cl.selection_source is not None), None)
        mimes = (owner.selection_mimes if owner and owner.selection_source
                 else ([self.clipboard_mime]
                       if self.clipboard_text is not None else []))
        for dd in c.data_devices:
            if not mimes:
                c.send(dd, "wl_data_device", "selection", 0)
                continue
            offer_id = 0xFF000000 + c.next_serial()
            c.objects[offer_id] = ("wl_data_offer",
                                   owner.cid if owner and
                                   owner.selection_source else -1)
            c.send(dd, "wl_data_device", "data_offer", offer_id)
            for m in mimes:
                c.send(offer_id, "wl_data_offer", "offer", m)
            c.send(dd, "wl_data_device", "selection", offer_id)

    def _offer_receive(self, c: Client, src_tag, mime: str, fd: int):
        """Client asked for the selection contents on fd."""
        if src_tag == -1 or src_tag is None:
            data = (self.clipboard_text or "").encode()
            try:
                os.write(fd, data)
            finally:
                os.close(fd)
            return
        # app-owned: forward a send event (with the fd) to the owner
        owner = next((cl for cl in self.clients if cl.cid == src_tag), None)
        if owner and owner.selection_source:
            owner.send(owner.selection_source, "wl_data_source", "send",
                       mime, fd)
        os.close(fd)

    def clipboard_types_app(self):
        owner = next((c for c in self.clients
                      if c.selection_source is not None), None)
        if owner:
            return list(owner.selection_mimes)
        return [self.clipboard_mime] if self.clipboard_text is not None \
            else []

    def clipboard_read_app(self, mime="text/plain;charset=utf-8",
                           timeout=2.0) -> bytes:
        """Read the current selection (app- or compositor-owned)."""
        owner = next((c for c in self.clients
                      if c.selection_source is not None), None)
        if owner is None:
            return (self.clipboard_text or "").encode()
        r, w = os.pipe()
        def do():
            owner.send(owner.selection_source, "wl_data_source", "send",
                       mime, w)
            os.close(w)
        self._post(do)
        os.set_blocking(r, False)
        out = b""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            try:
                chunk = os.read(r, 65536)
                if not chunk:
                    break
                out += chunk
            except BlockingIOError:
                time.sleep(0.01)
        os.close(r)
        return out

    # ---- engine capture bridge --------------------------------------------
    def start_shm_publisher(self, path: str, fps: float = 60.0):
        """Publish composited frames into a seqlock shm file the native
        engine maps via its "shm:<path>" capture backend (header: magic
        'HFSH', seq, w, h; even seq = stable — native/capture_shm.cpp)."""
        import mmap
        import struct as _st
        out = self.outputs[0]
        size = 16 + out.w * out.h * 4
        with open(path, "wb") as f:
            f.write(_st.pack("<IIII", 0x48534648, 0, out.w, out.h))
            f.truncate(size)
        f = open(path, "r+b")
        mm = mmap.mmap(f.fileno(), size)
        self._shm_stop = False

        def loop():
            seq = 0
            period = 1.0 / max(1.0, fps)
            while not self._shm_stop:
                t0 = time.monotonic()
                try:
                    _, _, fb = self.composite_frame()
                except Exception:
                    break
                seq += 1                       # odd: writing
                mm[4:8] = _st.pack("<I", seq)
                mm[16:16 + len(fb)] = fb
                seq += 1                       # even: stable
                mm[4:8] = _st.pack("<I", seq)
                dt = period - (time.monotonic() - t0)
                if dt > 0:
                    time.sleep(dt)
            mm.close()
            f.close()

        self._shm_thread = threading.Thread(target=loop, daemon=True,
                                            name="wl-shm-pub")
        self._shm_thread.start()
        return path

    def stop_shm_publisher(self):
        self._shm_stop = True
        t = getattr(self, "_shm_thread", None)
        if t is not None:
            t.join(timeout=2)
            self._shm_thread = None

    # ---- capture seam ------------------------------------------------------
    def composite_frame(self, output_name=None):
        """Render mapped surfaces' committed shm buffers into a BGRX
        bytearray (w, h, bytes) for the primary (or given) output."""
        out = next((o for o in self.outputs if o.name_id == output_name),
                   self.outputs[0])
        fb = bytearray(out.w * out.h * 4)
        for s in self.surfaces:
            if not s.mapped or s.buffer is None or s.output is not out:
                continue
            mm, off, w, h, stride, fmt = s.buffer
            for row in range(min(h, out.h)):
                src0 = off + row * stride
                dst0 = (row * out.w) * 4
                n = min(w, out.w) * 4
                fb[dst0:dst0 + n] = mm[src0:src0 + n]
        return out.w, out.h, bytes(fb)


# ---- module-level contract (pixelflux API names) --------------------------

_compositor: Optional[Compositor] = None
_comp_lock = threading.Lock()


def ensure_wayland_display(name: str = "selkies-wl-0", width: int = 1920,
                           height: int = 1080) -> str:
    """Idempotent compositor bring-up; returns the WAYLAND_DISPLAY name
    (reference contract: pixelflux.ensure_wayland_display,
    stream_server.py:866)."""
    global _compositor
    with _comp_lock:
        if _compositor is None:
            _compositor = Compositor(name, width, height).start()
        return _compositor.display_name


def get_wayland_display_name() -> Optional[str]:
    return _compositor.display_name if _compositor else None


def get_compositor() -> Optional[Compositor]:
    return _compositor


def shutdown_wayland_display():
    global _compositor
    with _comp_lock:
        if _compositor is not None:
            _compositor.stop()
            _compositor = None
