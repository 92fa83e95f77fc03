#!/bin/bash
# AppImage recipe (reference parity: infra/appimage). Builds the AppDir
# and, when appimagetool is available, the .AppImage; otherwise leaves
# the AppDir (and a tarball of it) so the layout is verifiable offline.
set -euo pipefail
cd "$(dirname "$0")/../.."

OUT=${1:-build/appimage}
APP="$OUT/selkies-amd.AppDir"
rm -rf "$APP"
mkdir -p "$APP/usr/bin" "$APP/usr/lib/selkies-amd" "$APP/usr/share/applications"

# wheel into a private site dir
pip3 wheel . --no-build-isolation --no-deps -w "$OUT/wheel" >/dev/null
SITE="$APP/usr/lib/selkies-amd/site-packages"
mkdir -p "$SITE"
python3 -c "import glob,zipfile;zipfile.ZipFile(glob.glob('$OUT/wheel/*.whl')[0]).extractall('$SITE')"

for so in addons/js-interposer/*.so addons/fake-udev/*.so; do
  [ -f "$so" ] && cp "$so" "$APP/usr/lib/selkies-amd/"
done

cat > "$APP/AppRun" <<'RUN'
#!/bin/bash
HERE=$(dirname "$(readlink -f "$0")")
export PYTHONPATH="$HERE/usr/lib/selkies-amd/site-packages:$PYTHONPATH"
exec python3 -m selkies_amd "$@"
RUN
chmod +x "$APP/AppRun"

cat > "$APP/selkies-amd.desktop" <<'DESK'
[Desktop Entry]
Name=selkies-amd
Exec=AppRun
Icon=selkies-amd
Type=Application
Categories=Network;
DESK
: > "$APP/selkies-amd.png"

if command -v appimagetool >/dev/null 2>&1; then
  appimagetool "$APP" "$OUT/selkies-amd.AppImage"
else
  tar -C "$OUT" -czf "$OUT/selkies-amd.AppDir.tar.gz" "$(basename "$APP")"
  echo "appimagetool not present: AppDir tarball at $OUT/selkies-amd.AppDir.tar.gz"
fi
