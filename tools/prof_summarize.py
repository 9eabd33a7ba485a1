"""Summarize a rocprofv3 SQLite results DB into a small kernel-stats CSV."""
import csv
import glob
import sqlite3
import sys

db = glob.glob(sys.argv[1])[0]
out = sys.argv[2]
con = sqlite3.connect(db)
cur = con.cursor()
tabs = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")]
sfx = tabs[0].replace("rocpd_kernel_dispatch_", "")
q = f"""SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6, AVG(d.end-d.start)/1e6
FROM rocpd_kernel_dispatch_{sfx} d
JOIN rocpd_info_kernel_symbol_{sfx} s ON d.kernel_id=s.id
GROUP BY s.display_name ORDER BY SUM(d.end-d.start) DESC"""
rows = list(cur.execute(q))
tot = sum(r[2] for r in rows)
with open(out, "w", newline="") as f:
    w = csv.writer(f)
    w.writerow(["kernel", "calls", "total_ms", "avg_ms", "pct"])
    for name, n, tms, avg in rows:
        w.writerow([name, n, round(tms, 3), round(avg, 4), round(100 * tms / tot, 2)])
print(f"total kernel ms: {tot:.1f}; rows: {len(rows)} -> {out}")
for name, n, tms, avg in rows[:14]:
    print(f"{tms:9.2f} ms {100*tms/tot:5.1f}% n={n:6d} {name[:70]}")
