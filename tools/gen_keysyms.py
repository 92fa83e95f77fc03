#!/usr/bin/env python3
"""Generate selkies_amd/keysyms.py from the system X11 keysymdef.h
(authoritative public ABI — no table transcription risk)."""

import re
import sys

HEADER = "/usr/include/X11/keysymdef.h"
OUT = "selkies_amd/keysyms.py"


def main():
    pat = re.compile(
        r"^#define XK_(\w+)\s+0x([0-9a-fA-F]+)\s*(?:/\*.*?U\+([0-9a-fA-F]{4,6}).*?\*/)?")
    name_to_keysym = {}
    keysym_to_unicode = {}
    for line in open(HEADER):
        m = pat.match(line)
        if not m:
            continue
        name, ks = m.group(1), int(m.group(2), 16)
        if name not in name_to_keysym:
            name_to_keysym[name] = ks
        if m.group(3):
            keysym_to_unicode.setdefault(ks, int(m.group(3), 16))
    with open(OUT, "w") as f:
        f.write('"""X11 keysym tables — GENERATED from %s by '
                'tools/gen_keysyms.py.\nEquivalent role to the reference '
                'server_keysym_map.py (SURVEY.md §2.1)."""\n\n' % HEADER)
        f.write("NAME_TO_KEYSYM = {\n")
        for k, v in sorted(name_to_keysym.items()):
            f.write(f"    {k!r}: {v:#x},\n")
        f.write("}\n\nKEYSYM_TO_UNICODE = {\n")
        for k, v in sorted(keysym_to_unicode.items()):
            f.write(f"    {k:#x}: {v:#x},\n")
        f.write("}\n\nKEYSYM_TO_NAME = {v: k for k, v in "
                "NAME_TO_KEYSYM.items()}\n\n")
        f.write('''
def unicode_to_keysym(cp: int) -> int:
    """Unicode codepoint -> keysym (latin-1 direct, else 0x01000000+cp)."""
    if cp < 0x100:
        return cp
    return 0x01000000 + cp


def keysym_to_unicode(ks: int) -> int:
    if ks < 0x100:
        return ks
    if ks >= 0x01000000:
        return ks - 0x01000000
    return KEYSYM_TO_UNICODE.get(ks, 0)
''')
    print(f"wrote {OUT}: {len(name_to_keysym)} names, "
          f"{len(keysym_to_unicode)} unicode mappings")


if __name__ == "__main__":
    sys.exit(main())
