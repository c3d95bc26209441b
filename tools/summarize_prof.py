import sqlite3, sys
db = sqlite3.connect(sys.argv[1])
out = open(sys.argv[2], 'w')
out.write(f"{'pct':>6} {'calls':>7} {'avg_us':>10} {'total_us':>12}  name\n")
for name, calls, tot, avg, pct in db.execute("SELECT * FROM top_kernels LIMIT 40"):
    out.write(f"{pct:6.2f} {calls:7d} {avg:10.2f} {tot:12.1f}  {name[:150]}\n")
out.close()
