"""Clipboard sync backends (reference ladder surveyed at SURVEY.md §2.1
input_handler.py:20-47: XFixes monitor / compositor data-control / external
tool fallback).

This environment ships no clipboard CLI tools and headless operation is the
default, so the ladder here is: external tool (xclip/xsel/wl-paste) when one
exists, else an in-memory clipboard (which still gives full client<->server
clipboard semantics for single-server use)."""

from __future__ import annotations

import logging
import shutil
import subprocess
from typing import Optional

logger = logging.getLogger("selkies.clipboard")


class MemoryClipboard:
    """In-memory clipboard store."""

    def __init__(self):
        self._text = ""

    def read(self) -> str:
        return self._text

    def write(self, text: str) -> None:
        self._text = text


class ToolClipboard(MemoryClipboard):
    """Drives a real session clipboard through xclip/xsel/wl-clipboard,
    keeping the memory copy as cache + fallback."""

    def __init__(self, display: Optional[str] = None):
        super().__init__()
        self.display = display
        self._read_cmd = None
        self._write_cmd = None
        if shutil.which("xclip"):
            self._read_cmd = ["xclip", "-selection", "clipboard", "-o"]
            self._write_cmd = ["xclip", "-selection", "clipboard", "-i"]
        elif shutil.which("xsel"):
            self._read_cmd = ["xsel", "-b", "-o"]
            self._write_cmd = ["xsel", "-b", "-i"]
        elif shutil.which("wl-paste"):
            self._read_cmd = ["wl-paste", "-n"]
            self._write_cmd = ["wl-copy"]
        if self._read_cmd is None:
            raise RuntimeError("no clipboard tool available")

    def _env(self):
        import os
        env = dict(os.environ)
        if self.display:
            env["DISPLAY"] = self.display
        return env

    def read(self) -> str:
        try:
            out = subprocess.run(self._read_cmd, capture_output=True,
                                 text=True, env=self._env(), timeout=5)
            if out.returncode == 0:
                self._text = out.stdout
        except (OSError, subprocess.TimeoutExpired) as exc:
            logger.debug("clipboard read failed: %r", exc)
        return self._text

    def write(self, text: str) -> None:
        super().write(text)
        try:
            subprocess.run(self._write_cmd, input=text, text=True,
                           env=self._env(), timeout=5)
        except (OSError, subprocess.TimeoutExpired) as exc:
            logger.debug("clipboard write failed: %r", exc)


def make_clipboard(display: Optional[str] = None) -> MemoryClipboard:
    try:
        cb = ToolClipboard(display)
        logger.info("clipboard via %s", cb._read_cmd[0])
        return cb
    except Exception:
        return MemoryClipboard()
