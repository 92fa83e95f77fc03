"""Summarize per-kernel GPU time from a rocprofv3 sqlite results db.

Usage: python tools/kstats.py <results.db> [top_n]
Uses the rocpd views (ROCm 7.x): kernel dispatches joined to
rocpd_info_kernel_symbol, plus memory-copy totals.
"""
import sqlite3
import sys


def main(path: str, top: int = 15) -> int:
    db = sqlite3.connect(path)
    cur = db.cursor()
    rows = cur.execute(
        "SELECT s.display_name, COUNT(*), SUM(d.end - d.start),"
        " AVG(d.workgroup_size_x)"
        " FROM rocpd_kernel_dispatch d"
        " JOIN rocpd_info_kernel_symbol s ON d.kernel_id = s.id"
        " GROUP BY 1 ORDER BY 3 DESC").fetchall()
    print(f"{'total ms':>10} {'calls':>7} {'avg us':>9} {'wgx':>5}  kernel")
    for name, calls, total, wgx in rows[:top]:
        nm = str(name).split("(")[0][:66]
        print(f"{total / 1e6:10.2f} {calls:7d} {total / calls / 1e3:9.1f}"
              f" {int(wgx):5d}  {nm}")
    copies = cur.execute(
        "SELECT st.string, COUNT(*), SUM(c.end - c.start), SUM(c.size)"
        " FROM rocpd_memory_copy c"
        " JOIN rocpd_string st ON c.name_id = st.id"
        " GROUP BY 1 ORDER BY 3 DESC").fetchall()
    for name, calls, total, size in copies:
        print(f"{total / 1e6:10.2f} {calls:7d} {total / calls / 1e3:9.1f}"
              f" {'':>5}  {name} ({size / 1e6:.1f} MB)")
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2
                  else 15))
