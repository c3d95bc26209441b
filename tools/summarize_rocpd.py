"""Summarize a rocprofv3 rocpd SQLite DB (ROCm 7.2 schema) into a
top-kernels text table: python tools/summarize_rocpd.py in.db out.txt"""
import sqlite3
import sys

db = sqlite3.connect(sys.argv[1])
q = """
SELECT ks.display_name, COUNT(*) c, SUM(kd.end - kd.start)/1000.0 tot,
       AVG(kd.end - kd.start)/1000.0 avg,
       MAX(ks.arch_vgpr_count), MAX(ks.sgpr_count),
       MAX(kd.grid_size_x * kd.grid_size_y * kd.grid_size_z)
FROM rocpd_kernel_dispatch kd
JOIN rocpd_info_kernel_symbol ks ON kd.kernel_id = ks.id
GROUP BY ks.display_name ORDER BY tot DESC LIMIT 40
"""
rows = list(db.execute(q))
tot = sum(r[2] for r in rows)
span, busy = next(iter(db.execute(
    "SELECT (MAX(end)-MIN(start))/1e9, SUM(end-start)/1e9 "
    "FROM rocpd_kernel_dispatch"
)))
with open(sys.argv[2], "w") as out:
    out.write(f"span_s {span:.3f}  busy_s {busy:.3f}  "
              f"busy_frac {busy/span:.3f}\n")
    out.write(f"{'pct':>6} {'calls':>7} {'avg_us':>9} {'tot_us':>11} "
              f"{'vgpr':>5} {'sgpr':>5} {'grid':>9}  name\n")
    for name, c, t, a, v, sg, g in rows:
        out.write(f"{100*t/tot:6.2f} {c:7d} {a:9.2f} {t:11.1f} "
                  f"{v:5d} {sg:5d} {g:9d}  {name[:110]}\n")
print("ok")
