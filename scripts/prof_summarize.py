"""Summarize a rocprofv3 rocpd .db into a small CSV (top kernels by time)."""
import sqlite3, sys, csv, glob
db = glob.glob(sys.argv[1])[0]
out = sys.argv[2]
c = sqlite3.connect(db)
suf = [r[0] for r in c.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")
    if r[0].startswith('rocpd_metadata')][0].replace('rocpd_metadata', '')
rows = list(c.execute(f"""
  SELECT substr(ks.display_name,1,90) n, COUNT(*) calls,
         SUM(kd.end-kd.start)/1e6 ms
  FROM rocpd_kernel_dispatch{suf} kd
  JOIN rocpd_info_kernel_symbol{suf} ks ON kd.kernel_id = ks.id
  GROUP BY 1 ORDER BY ms DESC"""))
tot = sum(r[2] for r in rows)
with open(out, 'w', newline='') as f:
    w = csv.writer(f)
    w.writerow(['kernel', 'calls', 'total_ms', 'pct'])
    for n, calls, ms in rows:
        w.writerow([n, calls, f"{ms:.3f}", f"{100*ms/tot:.2f}"])
print(f"total {tot:.1f} ms over {len(rows)} kernels -> {out}")
for n, calls, ms in rows[:12]:
    print(f"{ms:8.1f}ms {100*ms/tot:5.1f}% x{calls:6d}  {n[:70]}")
