"""Clipboard wire verbs + input dispatcher coverage."""

import base64

from selkies_amd.clipboard import MemoryClipboard
from selkies_amd.input_handler import InputDispatcher, RecordingBackend


def make_dispatcher():
    cb = MemoryClipboard()
    backend = RecordingBackend()
    d = InputDispatcher(backend, on_clipboard=cb.write,
                        clipboard_read=cb.read)
    return d, backend, cb


def test_clipboard_write_and_read():
    d, _, cb = make_dispatcher()
    payload = base64.b64encode("héllo 世界".encode()).decode()
    d.on_message(f"cw,{payload}")
    assert cb.read() == "héllo 世界"
    reply = d.on_message("cr,")
    verb, rest = reply.split(",", 1)
    assert verb == "clipboard"
    assert base64.b64decode(rest).decode() == "héllo 世界"


def test_clipboard_disabled():
    cb = MemoryClipboard()
    d = InputDispatcher(RecordingBackend(), on_clipboard=cb.write,
                        clipboard_read=cb.read, enable_clipboard=False)
    payload = base64.b64encode(b"secret").decode()
    d.on_message(f"cw,{payload}")
    assert cb.read() == ""
    assert d.on_message("cr,") is None


def test_button_mask_diffing():
    d, backend, _ = make_dispatcher()
    d.on_message("m,5,5,1")       # left down
    d.on_message("m,6,6,3")       # +middle
    d.on_message("m,7,7,0")       # all up
    btns = [e for e in backend.events if e[0] == "btn"]
    assert ("btn", 1, True) in btns and ("btn", 2, True) in btns
    assert ("btn", 1, False) in btns and ("btn", 2, False) in btns


def test_scroll_and_relative():
    d, backend, _ = make_dispatcher()
    d.on_message("sw,u,2")
    d.on_message("m2,-3,4,0")
    assert backend.events.count(("btn", 4, True)) == 2
    assert ("rel", -3, 4) in backend.events


def test_atomic_char_typing():
    d, backend, _ = make_dispatcher()
    text = base64.b64encode("aÉ".encode()).decode()
    d.on_message(f"co,{text}")
    keys = [e for e in backend.events if e[0] == "key"]
    assert ("key", ord("a"), True) in keys
    assert ("key", 0xC9, True) in keys          # É latin-1 keysym


def test_key_release_all_on_reset():
    d, backend, _ = make_dispatcher()
    d.on_message("kd,65")
    d.on_message("kd,0xff08" if False else "kd,66")
    d.on_message("kr,")
    ups = [e for e in backend.events if e[0] == "key" and not e[2]]
    assert ("key", 65, False) in ups and ("key", 66, False) in ups


def test_spare_keycode_pool_rotation():
    """Unmapped keysyms claim spare keycodes; LRU rotation when the pool
    is exhausted (reference keysym->spare-keycode overlay)."""
    from selkies_amd.input_handler import SpareKeycodePool

    remaps = []
    pool = SpareKeycodePool([250, 251], lambda kc, ks: remaps.append(
        (kc, ks)))
    assert pool.keycode_for(0x1001000) == 250
    assert pool.keycode_for(0x1001001) == 251
    # cached: no new remap
    assert pool.keycode_for(0x1001000) == 250
    assert remaps == [(250, 0x1001000), (251, 0x1001001)]
    # pool full: evict the LRU entry (0x1001001 after the re-touch above)
    assert pool.keycode_for(0x1001002) == 251
    assert remaps[-1] == (251, 0x1001002)
    # the evicted keysym re-claims via another eviction
    assert pool.keycode_for(0x1001001) == 250


def test_spare_keycode_pool_empty():
    from selkies_amd.input_handler import SpareKeycodePool
    pool = SpareKeycodePool([], lambda kc, ks: None)
    assert pool.keycode_for(0x100) is None


def _mk_dispatcher(**kw):
    from selkies_amd.input_handler import InputDispatcher, RecordingBackend
    writes = []
    binaries = []
    d = InputDispatcher(RecordingBackend(),
                        on_clipboard=writes.append,
                        on_clipboard_binary=lambda m, b: binaries.append(
                            (m, b)),
                        **kw)
    return d, writes, binaries


def test_multipart_text_clipboard():
    """cws/cwd/cwe chunked text transfer (reference multipart clipboard
    protocol): chunks reassemble, delivered once at cwe."""
    import base64
    d, writes, _ = _mk_dispatcher()
    payload = "hello " * 1000
    raw = payload.encode()
    d.on_message(f"cws,t1,{len(raw)}")
    for i in range(0, len(raw), 1024):
        chunk = base64.b64encode(raw[i:i + 1024]).decode()
        d.on_message(f"cwd,t1,{chunk}")
        assert not writes                  # nothing delivered mid-transfer
    d.on_message("cwe,t1")
    assert writes == [payload]


def test_multipart_binary_clipboard_gated_and_delivered():
    import base64
    # gate off: start rejected, chunks ignored
    d, writes, binaries = _mk_dispatcher(enable_binary_clipboard=False)
    d.on_message("cbs,t2,image/png,8")
    d.on_message("cbd,t2," + base64.b64encode(b"PNGDATA!").decode())
    d.on_message("cbe,t2")
    assert not binaries
    # gate on: delivered with mime
    d, writes, binaries = _mk_dispatcher(enable_binary_clipboard=True)
    d.on_message("cbs,t3,image/png,8")
    d.on_message("cbd,t3," + base64.b64encode(b"PNGDATA!").decode())
    d.on_message("cbe,t3")
    assert binaries == [("image/png", b"PNGDATA!")]


def test_multipart_clipboard_bounds_and_mismatch():
    import base64
    d, writes, _ = _mk_dispatcher()
    # oversize declared: rejected at start
    d.on_message(f"cws,big,{64 * 1024 * 1024}")
    d.on_message("cwd,big," + base64.b64encode(b"x").decode())
    d.on_message("cwe,big")
    assert not writes
    # id mismatch mid-transfer aborts
    d.on_message("cws,a,4")
    d.on_message("cwd,b," + base64.b64encode(b"zz").decode())
    d.on_message("cwe,a")
    assert not writes
    # overflow beyond declared size aborts
    d.on_message("cws,c,2")
    d.on_message("cwd,c," + base64.b64encode(b"zzzz").decode())
    d.on_message("cwe,c")
    assert not writes


def test_single_binary_cb_verb():
    import base64
    d, _, binaries = _mk_dispatcher(enable_binary_clipboard=True)
    d.on_message("cb,application/x-test," +
                 base64.b64encode(b"\x00\x01\x02").decode())
    assert binaries == [("application/x-test", b"\x00\x01\x02")]
