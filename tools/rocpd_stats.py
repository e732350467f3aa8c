"""Summarize a rocprofv3 rocpd sqlite database into a per-kernel
stats table (and per-kernel PMC counter sums when present).

Usage: python tools/rocpd_stats.py <results.db-or-dir> [--min-pct 0.1]
"""

import argparse
import glob
import os
import re
import sqlite3


def short_name(n):
    n = re.sub(r"\(.*", "", n)
    n = n.split("void ")[-1].strip()
    return n[:68]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("path")
    ap.add_argument("--min-pct", type=float, default=0.05)
    p = ap.parse_args()

    path = p.path
    if os.path.isdir(path):
        cands = glob.glob(os.path.join(path, "**", "*results.db"),
                          recursive=True)
        if not cands:
            raise SystemExit(f"no results.db under {path}")
        path = cands[0]
    con = sqlite3.connect(path)
    tabs = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def tab(prefix):
        for t in tabs:
            if t.startswith(prefix):
                return t
        return None

    kd, ks = tab("rocpd_kernel_dispatch"), tab("rocpd_info_kernel_symbol")
    rows = con.execute(
        f"SELECT s.display_name, count(*), sum(d.end - d.start), "
        f"s.arch_vgpr_count, s.sgpr_count, max(d.group_segment_size) "
        f"FROM {kd} d JOIN {ks} s ON d.kernel_id = s.id "
        f"GROUP BY s.display_name ORDER BY sum(d.end - d.start) DESC"
    ).fetchall()
    total_ns = sum(r[2] for r in rows) or 1
    ndisp = sum(r[1] for r in rows)
    print(f"# rocpd kernel stats: {path}")
    print(f"# total GPU kernel time: {total_ns/1e6:.2f} ms over "
          f"{ndisp} dispatches")
    print(f"{'kernel':<68} {'calls':>6} {'total_ms':>9} {'avg_us':>9} "
          f"{'%':>6} {'vgpr':>5} {'lds':>7}")
    for name, calls, ns, vgpr, _sgpr, lds in rows:
        pct = 100 * ns / total_ns
        if pct < p.min_pct:
            continue
        print(f"{short_name(name):<68} {calls:>6} {ns/1e6:>9.3f} "
              f"{ns/1e3/calls:>9.1f} {pct:>6.2f} {vgpr:>5} {lds:>7}")

    pmc_ev, pmc_info = tab("rocpd_pmc_event"), tab("rocpd_info_pmc")
    npmc = con.execute(f"SELECT count(*) FROM {pmc_ev}").fetchone()[0] \
        if pmc_ev else 0
    if npmc:
        print("\n# per-kernel PMC sums")
        rows = con.execute(
            f"SELECT s.display_name, i.name, count(*), sum(e.value) "
            f"FROM {pmc_ev} e "
            f"JOIN {kd} d ON e.event_id = d.event_id "
            f"JOIN {ks} s ON d.kernel_id = s.id "
            f"JOIN {pmc_info} i ON e.pmc_id = i.id "
            f"GROUP BY s.display_name, i.name "
            f"ORDER BY sum(e.value) DESC").fetchall()
        print(f"{'kernel':<60} {'counter':<16} {'calls':>6} "
              f"{'sum':>16} {'per-call':>14}")
        for name, cname, calls, val in rows[:40]:
            print(f"{short_name(name):<60} {cname:<16} {calls:>6} "
                  f"{val:>16.0f} {val/calls:>14.1f}")


if __name__ == "__main__":
    main()
