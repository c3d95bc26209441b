import sqlite3, sys
db = sqlite3.connect(sys.argv[1])
out = open(sys.argv[2], 'w')
cols = [d[1] for d in db.execute("PRAGMA table_info(counters_collection)")]
out.write("columns: " + ",".join(cols) + "\n")
# aggregate counter values per kernel name
q = """SELECT kernel_name, counter_name, SUM(value), COUNT(*)
       FROM counters_collection GROUP BY kernel_name, counter_name"""
try:
    rows = list(db.execute(q))
except Exception as e:
    out.write(f"agg query failed: {e}\nfalling back to raw dump\n")
    rows = []
    for r in db.execute("SELECT * FROM counters_collection LIMIT 200"):
        out.write(repr(r)[:300] + "\n")
agg = {}
for kn, cn, v, c in rows:
    agg.setdefault(kn, {})[cn] = (v, c)
for kn, d in sorted(agg.items(),
                    key=lambda t: -t[1].get("SQ_VALU_MFMA_BUSY_CYCLES", (0,0))[0]):
    out.write(f"\n{kn[:120]}\n")
    for cn, (v, c) in sorted(d.items()):
        out.write(f"   {cn}: total {v:.3e} over {c} dispatches\n")
out.close()
