#!/usr/bin/env python3
"""Top-kernel summary of a rocprofv3 rocpd SQLite DB (kernel-trace runs).

Usage: python scripts/prof_summary.py '<glob of .db files>' [steps]
Prints total kernel time and the top kernels by total duration; if steps
is given, also per-step columns.
"""

import glob
import sqlite3
import sys


def main():
    pat = sys.argv[1]
    steps = float(sys.argv[2]) if len(sys.argv) > 2 else None
    paths = sorted(glob.glob(pat))
    assert paths, f"no DB matches {pat}"
    db = paths[-1]
    con = sqlite3.connect(db)
    tabs = [r[0] for r in
            con.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tabs if t.startswith("rocpd_kernel_dispatch_")]
    assert kd, f"no kernel dispatch table in {db}"
    sfx = kd[0][len("rocpd_kernel_dispatch_"):]
    tot = con.execute(
        f"SELECT SUM(end-start)/1e6 FROM rocpd_kernel_dispatch_{sfx}"
    ).fetchone()[0]
    n_disp = con.execute(
        f"SELECT COUNT(*) FROM rocpd_kernel_dispatch_{sfx}").fetchone()[0]
    per = f" ({tot/steps:.2f} ms/step, {n_disp/steps:.0f} disp/step)" \
        if steps else ""
    print(f"# {db}\n# total kernel {tot:.1f} ms, {n_disp} dispatches{per}")
    q = (f"SELECT s.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 ms, "
         f"AVG(k.end-k.start)/1e3 avg "
         f"FROM rocpd_kernel_dispatch_{sfx} k "
         f"JOIN rocpd_info_kernel_symbol_{sfx} s ON k.kernel_id = s.id "
         f"GROUP BY s.display_name ORDER BY ms DESC LIMIT 30")
    for name, n, ms, avg in con.execute(q):
        stp = f" {ms/steps:6.3f}/st" if steps else ""
        print(f"{ms:9.2f} ms  n={n:5d}  avg={avg:8.1f} us{stp}  {name[:70]}")


if __name__ == "__main__":
    main()
