"""Summarize rocprofv3 rocpd sqlite output into a kernel-stats table."""
import glob
import sqlite3
import sys

out = []
for db in sorted(glob.glob(sys.argv[1])):
    c = sqlite3.connect(db)
    t = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")]
    if not t:
        continue
    sfx = t[0].replace('rocpd_kernel_dispatch', '')
    q = (f"SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6, "
         f"AVG(k.end-k.start)/1e3 FROM rocpd_kernel_dispatch{sfx} k "
         f"JOIN rocpd_info_kernel_symbol{sfx} ks ON k.kernel_id = ks.id "
         f"GROUP BY 1 ORDER BY 3 DESC LIMIT 10")
    out.append(f"== {db}")
    for n, cnt, tot, avg in c.execute(q):
        out.append(f"| {(n or '?').split('(')[0][:58]:58s} | n={cnt:5d} "
                   f"| total {tot:8.2f} ms | avg {avg:8.1f} us |")
print("\n".join(out))
