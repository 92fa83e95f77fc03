"""Summarize per-kernel GPU time from a rocprofv3 sqlite results db.

Usage: python tools/kstats.py <results.db> [top_n]
Introspects the rocpd schema (table/column names vary between ROCm
releases) and prints total-ms / calls / avg-us per kernel, hottest first.
"""
import sqlite3
import sys


def main(path: str, top: int = 15) -> int:
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = {r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")}
    kd = next((t for t in tables if "kernel_dispatch" in t), None)
    if kd is None:
        print("no kernel_dispatch table; tables:", sorted(tables))
        return 1
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({kd})")]
    start = next(c for c in cols if c in ("start", "start_timestamp",
                                          "begin", "start_ns"))
    end = next(c for c in cols if c in ("end", "end_timestamp", "end_ns"))
    # kernel name: either a direct column or via a string/info table join
    namecol = next((c for c in cols if "name" in c), None)
    rows = None
    if namecol:
        rows = cur.execute(
            f"SELECT {namecol}, COUNT(*), SUM({end}-{start})"
            f" FROM {kd} GROUP BY 1").fetchall()
    else:
        kid = next((c for c in cols if "kernel_id" in c or c == "kernel"),
                   None)
        ki = next((t for t in tables if "kernel" in t and "info" in t),
                  None)
        st = next((t for t in tables if t.endswith("_string")), None)
        if kid and ki:
            kcols = [r[1] for r in cur.execute(f"PRAGMA table_info({ki})")]
            kname = next((c for c in kcols if "name" in c), None)
            kpk = next((c for c in kcols if c in ("id", "kernel_id")), None)
            q = (f"SELECT i.{kname}, COUNT(*), SUM(d.{end}-d.{start}) "
                 f"FROM {kd} d JOIN {ki} i ON d.{kid}=i.{kpk} GROUP BY 1")
            rows = cur.execute(q).fetchall()
            if st and rows and isinstance(rows[0][0], int):
                scols = [r[1] for r in cur.execute(
                    f"PRAGMA table_info({st})")]
                spk = scols[0]
                sval = next(c for c in scols if c in ("string", "value",
                                                      "str"))
                names = dict(cur.execute(
                    f"SELECT {spk}, {sval} FROM {st}"))
                rows = [(names.get(r[0], r[0]), r[1], r[2]) for r in rows]
    if not rows:
        print("could not resolve kernel names; columns:", cols)
        return 1
    rows.sort(key=lambda r: -(r[2] or 0))
    print(f"{'total ms':>10} {'calls':>7} {'avg us':>9}  kernel")
    for name, calls, total in rows[:top]:
        if total is None:
            continue
        nm = str(name).split("(")[0][:70]
        print(f"{total / 1e6:10.2f} {calls:7d} {total / calls / 1e3:9.1f}"
              f"  {nm}")
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2
                  else 15))
