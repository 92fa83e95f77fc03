"""`selkies-gpu-probe`: recommend a capture/encode configuration for
container entrypoints (reference gpu_probe.py:115 — prints a backend
recommendation string)."""

from __future__ import annotations

import json
import os
import sys


def probe() -> dict:
    info = {"hip_devices": 0, "gpu_encode": False, "display": None,
            "recommendation": "cpu"}
    try:
        import hipflux
        info["hip_devices"] = hipflux.hip_device_count()
        info["native"] = hipflux.native_available()
    except Exception as exc:
        info["native"] = False
        info["error"] = repr(exc)
    info["display"] = os.environ.get("DISPLAY") or None
    if info["hip_devices"] > 0:
        info["gpu_encode"] = True
        info["recommendation"] = "x11-gpu" if info["display"] else \
            "synthetic-gpu"
    else:
        info["recommendation"] = "x11-cpu" if info["display"] else \
            "synthetic-cpu"
    return info


def main(argv=None) -> int:
    info = probe()
    if argv and "--json" in argv or "--json" in sys.argv[1:]:
        print(json.dumps(info))
    else:
        print(info["recommendation"])
    return 0


if __name__ == "__main__":
    sys.exit(main())
