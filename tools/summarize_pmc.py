"""Summarize a rocprofv3 --pmc rocpd SQLite DB per kernel.

Counter rows are per-SE instances linked to dispatches via
rocpd_pmc_event.event_id = rocpd_kernel_dispatch.event_id (both
reference rocpd_event ids); rocpd_info_pmc.name is plain TEXT.
Usage: python tools/summarize_pmc.py <results.db> [top_n]
"""

import sqlite3
import sys


def main():
    db = sqlite3.connect(sys.argv[1])
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 24
    rows = list(db.execute(
        """
        SELECT ks.display_name, p.name,
               SUM(pe.value), COUNT(DISTINCT kd.id),
               SUM(kd.end - kd.start)
        FROM rocpd_pmc_event pe
        JOIN rocpd_info_pmc p ON pe.pmc_id = p.id
        JOIN rocpd_kernel_dispatch kd ON pe.event_id = kd.event_id
        JOIN rocpd_info_kernel_symbol ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name, p.name
        """
    ))
    # collect per-kernel: {counter: total}, dispatch count, total ns
    per = {}
    for nm, ctr, val, nd, ns in rows:
        d = per.setdefault(nm, {"n": nd, "ns": ns, "c": {}})
        d["c"][ctr] = val
        d["n"] = max(d["n"], nd)
        d["ns"] = max(d["ns"], ns)
    order = sorted(per.items(), key=lambda kv: -kv[1]["ns"])[:top]
    ctrs = sorted({c for d in per.values() for c in d["c"]})
    print(f"{'kernel':58s} {'disp':>6s} {'us/disp':>8s} "
          + " ".join(f"{c.replace('SQ_',''):>22s}/disp" for c in ctrs))
    for nm, d in order:
        n = max(1, d["n"])
        cols = " ".join(
            f"{d['c'].get(c, 0.0) / n:27.3e}" for c in ctrs
        )
        short = nm.split("(")[0][:58]
        print(f"{short:58s} {d['n']:6d} {d['ns'] / 1e3 / n:8.2f} {cols}")


if __name__ == "__main__":
    main()
