#!/bin/bash
# Build a .deb of selkies-amd (reference parity: infra/packaging deb
# scripts). Layout: the wheel's contents installed into
# /usr/lib/python3/dist-packages + entry-point shims in /usr/bin.
# Requires: a built native module (make -C native) and dpkg-deb.
set -euo pipefail
cd "$(dirname "$0")/../.."

VERSION=$(python3 -c "import re;print(re.search(r'version = \"([^\"]+)\"', open('pyproject.toml').read()).group(1))")
ARCH=$(dpkg --print-architecture 2>/dev/null || echo amd64)
ROOT=$(mktemp -d)
PKG="$ROOT/selkies-amd_${VERSION}_${ARCH}"
SITE="$PKG/usr/lib/python3/dist-packages"

mkdir -p "$SITE" "$PKG/usr/bin" "$PKG/DEBIAN" "$PKG/usr/lib/selkies-amd"

# wheel -> site-packages
rm -rf build_deb_wheel && mkdir build_deb_wheel
pip3 wheel . --no-build-isolation --no-deps -w build_deb_wheel >/dev/null
python3 -c "
import glob, zipfile
w = glob.glob('build_deb_wheel/*.whl')[0]
zipfile.ZipFile(w).extractall('$SITE')
"
rm -rf build_deb_wheel

# LD_PRELOAD addons (joystick interposer + fake-udev), if built
for so in addons/js-interposer/*.so addons/fake-udev/*.so; do
  [ -f "$so" ] && cp "$so" "$PKG/usr/lib/selkies-amd/"
done

# entry points
for name in selkies selkies-resize selkies-gpu-probe selkies-mux; do
  mod=$(python3 - "$name" <<'PY'
import sys
eps = {
    "selkies": "selkies_amd.__main__:main",
    "selkies-resize": "selkies_amd.display_utils:resize_entrypoint",
    "selkies-gpu-probe": "selkies_amd.gpu_probe:main",
    "selkies-mux": "selkies_amd.mp4:mux_entrypoint",
}
print(eps[sys.argv[1]])
PY
)
  module=${mod%%:*}; func=${mod##*:}
  cat > "$PKG/usr/bin/$name" <<SH
#!/usr/bin/python3
import sys
from $module import $func
sys.exit($func())
SH
  chmod 755 "$PKG/usr/bin/$name"
done

cat > "$PKG/DEBIAN/control" <<CTL
Package: selkies-amd
Version: $VERSION
Architecture: $ARCH
Maintainer: selkies-amd
Depends: python3 (>= 3.10), python3-aiohttp, python3-numpy, python3-psutil, python3-prometheus-client
Section: net
Priority: optional
Description: MI355X-native low-latency HTML5 remote desktop streaming
 GPU H.264/JPEG stripe encoder (HIP/gfx950), WebSocket + WebRTC
 transports, input/clipboard/gamepad injection.
CTL

mkdir -p dist
dpkg-deb --build --root-owner-group "$PKG" dist/ >/dev/null
rm -rf "$ROOT"
ls -l dist/selkies-amd_${VERSION}_${ARCH}.deb
