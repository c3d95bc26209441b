"""Tests for the rocprofv3 rocpd summarizers (tools/).

These scripts are the committed-evidence pipeline (profiles/*.txt); pin
them against tiny synthetic DBs in the ROCm 7.2 rocpd schema so schema
assumptions (PMC rows link to dispatches via event_id; counter names
are plain TEXT) stay checked on CPU.
"""

import sqlite3
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parents[1]


def _mk_db(path, with_pmc=False):
    db = sqlite3.connect(path)
    db.executescript(
        """
        CREATE TABLE rocpd_info_kernel_symbol (
            id INTEGER, display_name TEXT,
            arch_vgpr_count INTEGER, sgpr_count INTEGER);
        CREATE TABLE rocpd_kernel_dispatch (
            id INTEGER, kernel_id INTEGER, event_id INTEGER,
            start BIGINT, end BIGINT,
            grid_size_x INTEGER, grid_size_y INTEGER,
            grid_size_z INTEGER);
        """
    )
    db.execute("INSERT INTO rocpd_info_kernel_symbol VALUES "
               "(1, 'void k_fast(double*)', 32, 16)")
    db.execute("INSERT INTO rocpd_info_kernel_symbol VALUES "
               "(2, 'void k_slow(double*)', 64, 32)")
    # k_slow: one 10 us dispatch; k_fast: two 1 us dispatches
    rows = [
        (1, 2, 1, 1000, 11000, 256, 1, 1),
        (2, 1, 2, 20000, 21000, 64, 1, 1),
        (3, 1, 3, 30000, 31000, 64, 1, 1),
    ]
    db.executemany(
        "INSERT INTO rocpd_kernel_dispatch VALUES (?,?,?,?,?,?,?,?)",
        rows,
    )
    if with_pmc:
        db.executescript(
            """
            CREATE TABLE rocpd_info_pmc (id INTEGER, name TEXT);
            CREATE TABLE rocpd_pmc_event (
                id INTEGER, event_id INTEGER, pmc_id INTEGER,
                value REAL);
            """
        )
        db.execute("INSERT INTO rocpd_info_pmc VALUES (7, 'SQ_WAVES')")
        # two per-SE instance rows per dispatch, summed by the tool
        pmc = [
            (1, 1, 7, 100.0), (2, 1, 7, 100.0),
            (3, 2, 7, 10.0), (4, 2, 7, 10.0),
            (5, 3, 7, 10.0), (6, 3, 7, 10.0),
        ]
        db.executemany(
            "INSERT INTO rocpd_pmc_event VALUES (?,?,?,?)", pmc
        )
    db.commit()
    db.close()


def test_summarize_rocpd(tmp_path):
    dbp = tmp_path / "r.db"
    outp = tmp_path / "out.txt"
    _mk_db(dbp)
    subprocess.run(
        [sys.executable, str(ROOT / "tools" / "summarize_rocpd.py"),
         str(dbp), str(outp)],
        check=True, cwd=ROOT,
    )
    lines = outp.read_text().splitlines()
    assert lines[0].startswith("span_s")
    # sorted by total time: k_slow (10 us) above k_fast (2 us total)
    assert "k_slow" in lines[2] and "k_fast" in lines[3]
    assert " 10.00 " in lines[2]  # avg_us of the slow kernel


def test_summarize_pmc(tmp_path):
    dbp = tmp_path / "p.db"
    _mk_db(dbp, with_pmc=True)
    res = subprocess.run(
        [sys.executable, str(ROOT / "tools" / "summarize_pmc.py"),
         str(dbp)],
        check=True, cwd=ROOT, capture_output=True, text=True,
    )
    lines = res.stdout.splitlines()
    assert "WAVES" in lines[0]
    # per-dispatch sums: k_slow 200 waves / 1 dispatch, k_fast
    # (2 dispatches x 20 summed) / 2 = 20
    slow = next(ln for ln in lines if "k_slow" in ln)
    fast = next(ln for ln in lines if "k_fast" in ln)
    assert "2.000e+02" in slow
    assert "2.000e+01" in fast
    # ordered by total time
    assert lines.index(slow) < lines.index(fast)
