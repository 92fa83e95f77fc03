"""fake-udev shim driven through the real libudev ABI via ctypes:
enumeration reflects socket presence; monitor surfaces hotplug."""

import ctypes
import os
import select
import subprocess
import time

import pytest

SHIM_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "addons", "fake-udev")
LIB = os.path.join(SHIM_DIR, "libudev.so.1")


@pytest.fixture(scope="module")
def libudev():
    subprocess.run(["make", "-C", SHIM_DIR], check=True, capture_output=True)
    lib = ctypes.CDLL(LIB)
    for fn in ("udev_new", "udev_enumerate_new",
               "udev_enumerate_get_list_entry", "udev_list_entry_get_next",
               "udev_list_entry_get_name", "udev_device_new_from_syspath",
               "udev_device_get_devnode", "udev_device_get_property_value",
               "udev_device_get_parent", "udev_device_get_sysattr_value",
               "udev_monitor_new_from_netlink",
               "udev_monitor_receive_device", "udev_device_get_action"):
        getattr(lib, fn).restype = ctypes.c_void_p
    lib.udev_list_entry_get_name.restype = ctypes.c_char_p
    lib.udev_device_get_devnode.restype = ctypes.c_char_p
    lib.udev_device_get_property_value.restype = ctypes.c_char_p
    lib.udev_device_get_sysattr_value.restype = ctypes.c_char_p
    lib.udev_device_get_action.restype = ctypes.c_char_p
    lib.udev_monitor_get_fd.restype = ctypes.c_int
    return lib


def enum_names(lib, udev):
    e = lib.udev_enumerate_new(ctypes.c_void_p(udev))
    lib.udev_enumerate_add_match_subsystem(ctypes.c_void_p(e), b"input")
    lib.udev_enumerate_scan_devices(ctypes.c_void_p(e))
    names = []
    ent = lib.udev_enumerate_get_list_entry(ctypes.c_void_p(e))
    while ent:
        names.append(lib.udev_list_entry_get_name(ctypes.c_void_p(ent)))
        ent = lib.udev_list_entry_get_next(ctypes.c_void_p(ent))
    lib.udev_enumerate_unref(ctypes.c_void_p(e))
    return [n.decode() for n in names]


def test_enumerate_and_device(libudev, tmp_path, monkeypatch):
    monkeypatch.setenv("SELKIES_JS_SOCKET_PATH", str(tmp_path))
    (tmp_path / "selkies_js0.sock").touch()
    (tmp_path / "selkies_js2.sock").touch()
    u = libudev.udev_new()
    names = enum_names(libudev, u)
    # each present pad enumerates a joydev node AND its evdev sibling
    assert len(names) == 4
    assert names[0].endswith("js0") and names[1].endswith("event1000")
    assert names[2].endswith("js2") and names[3].endswith("event1002")

    d = libudev.udev_device_new_from_syspath(
        ctypes.c_void_p(u), names[0].encode())
    assert libudev.udev_device_get_devnode(
        ctypes.c_void_p(d)) == b"/dev/input/js0"
    assert libudev.udev_device_get_property_value(
        ctypes.c_void_p(d), b"ID_INPUT_JOYSTICK") == b"1"
    parent = libudev.udev_device_get_parent(ctypes.c_void_p(d))
    assert parent
    assert libudev.udev_device_get_sysattr_value(
        ctypes.c_void_p(parent), b"name") == b"Selkies Virtual Gamepad"


def test_monitor_hotplug(libudev, tmp_path, monkeypatch):
    monkeypatch.setenv("SELKIES_JS_SOCKET_PATH", str(tmp_path))
    u = libudev.udev_new()
    m = libudev.udev_monitor_new_from_netlink(ctypes.c_void_p(u), b"udev")
    libudev.udev_monitor_enable_receiving(ctypes.c_void_p(m))
    fd = libudev.udev_monitor_get_fd(ctypes.c_void_p(m))
    assert fd >= 0

    (tmp_path / "selkies_js1.sock").touch()
    r, _, _ = select.select([fd], [], [], 3)
    assert r, "inotify did not fire on socket creation"
    d = libudev.udev_monitor_receive_device(ctypes.c_void_p(m))
    assert d
    assert libudev.udev_device_get_action(ctypes.c_void_p(d)) == b"add"
    assert libudev.udev_device_get_devnode(
        ctypes.c_void_p(d)) == b"/dev/input/js1"

    os.unlink(tmp_path / "selkies_js1.sock")
    r, _, _ = select.select([fd], [], [], 3)
    assert r
    d = libudev.udev_monitor_receive_device(ctypes.c_void_p(m))
    assert libudev.udev_device_get_action(ctypes.c_void_p(d)) == b"remove"


def test_event_device_from_syspath(libudev):
    lib = libudev
    lib.udev_device_get_sysname.restype = ctypes.c_char_p
    u = lib.udev_new()
    d = lib.udev_device_new_from_syspath(
        ctypes.c_void_p(u),
        b"/sys/devices/virtual/input/selkies-input1/event1001")
    assert d
    assert lib.udev_device_get_devnode(
        ctypes.c_void_p(d)) == b"/dev/input/event1001"
    assert lib.udev_device_get_sysname(
        ctypes.c_void_p(d)) == b"event1001"
