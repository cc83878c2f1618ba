"""Summarize a rocprofv3 rocpd .db: per-kernel total time, calls, %."""
import sqlite3, sys, re

db = sys.argv[1]
con = sqlite3.connect(db)
cur = con.cursor()
cur.execute("SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")
t = cur.fetchone()[0]
u = t.replace("rocpd_kernel_dispatch_", "")
q = f"""
SELECT ks.display_name, COUNT(*), SUM(kd.end - kd.start)/1e6, AVG(kd.end-kd.start)/1e3
FROM rocpd_kernel_dispatch_{u} kd
JOIN rocpd_info_kernel_symbol_{u} ks ON kd.kernel_id = ks.id
GROUP BY ks.display_name ORDER BY SUM(kd.end-kd.start) DESC LIMIT 40
"""
rows = cur.execute(q).fetchall()
total = sum(r[2] for r in rows)
print(f"{'kernel':<72} {'calls':>7} {'total_ms':>10} {'avg_us':>9} {'%':>6}")
for name, calls, ms, avg in rows:
    short = re.sub(r"<[^>]*>", "", name)[:70]
    print(f"{short:<72} {calls:>7} {ms:>10.2f} {avg:>9.1f} {100*ms/total:>5.1f}%")
print(f"TOTAL GPU time: {total:.1f} ms")
