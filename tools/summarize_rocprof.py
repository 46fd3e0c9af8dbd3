#!/usr/bin/env python3
"""Aggregate a rocprofv3 rocpd SQLite DB into a compact per-kernel summary
(kernel-trace timings and/or PMC counter sums).  Run ON the GPU box so only
the small text summary travels back, not the multi-MB DB."""
import argparse
import glob
import sqlite3
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db_glob")
    ap.add_argument("--out", default="-")
    ap.add_argument("--top", type=int, default=25)
    args = ap.parse_args()
    import os
    pattern = args.db_glob
    if os.path.isdir(pattern):  # accept a rocprofv3 -d output directory
        paths = sorted(glob.glob(os.path.join(pattern, "**", "*.db"),
                                 recursive=True))
    else:
        paths = sorted(glob.glob(pattern))
    if not paths:
        print(f"no db matches {args.db_glob}", file=sys.stderr)
        sys.exit(1)
    lines = []
    for path in paths:
        db = sqlite3.connect(path)
        cur = db.cursor()
        tables = [r[0] for r in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'")]
        u = None
        for t in tables:
            if t.startswith("rocpd_kernel_dispatch_"):
                u = t.replace("rocpd_kernel_dispatch_", "")
        if u is None:
            continue
        lines.append(f"== {path} ==")
        # timings
        rows = cur.execute(f"""
            SELECT k.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
                   AVG(d.end-d.start)/1e3
            FROM rocpd_kernel_dispatch_{u} d
            JOIN rocpd_info_kernel_symbol_{u} k ON d.kernel_id=k.id
            GROUP BY k.display_name ORDER BY 3 DESC LIMIT ?""",
            (args.top,)).fetchall()
        if rows and rows[0][2] is not None:
            lines.append(f"{'kernel':<60} {'calls':>6} {'tot_ms':>9} {'avg_us':>8}")
            for name, n, ms, avg in rows:
                lines.append(f"{str(name)[:58]:<60} {n:>6} {ms:>9.2f} {avg:>8.1f}")
        # PMC counters, if any
        if f"rocpd_pmc_event_{u}" in tables:
            npmc = cur.execute(
                f"SELECT COUNT(*) FROM rocpd_pmc_event_{u}").fetchone()[0]
            if npmc:
                try:
                    pmc = cur.execute(f"""
                        SELECT k.display_name, pd.name, SUM(p.value), COUNT(*)
                        FROM rocpd_pmc_event_{u} p
                        JOIN rocpd_kernel_dispatch_{u} d ON p.event_id = d.event_id
                        JOIN rocpd_info_kernel_symbol_{u} k ON d.kernel_id = k.id
                        JOIN rocpd_info_pmc_{u} pd ON p.pmc_id = pd.id
                        GROUP BY k.display_name, pd.name
                        ORDER BY k.display_name, pd.name""").fetchall()
                    lines.append(
                        f"{'kernel':<52} {'counter':<26} {'sum':>16} {'n':>6}")
                    for name, ctr, val, n in pmc:
                        lines.append(
                            f"{str(name)[:50]:<52} {str(ctr):<26} {val:>16.0f} {n:>6}")
                except sqlite3.Error as e:
                    lines.append(f"[pmc join failed: {e}; schemas follow]")
                    for t in (f"rocpd_pmc_event_{u}", f"rocpd_info_pmc_{u}",
                              f"rocpd_kernel_dispatch_{u}"):
                        cols = [r[1] for r in cur.execute(f"PRAGMA table_info({t})")]
                        lines.append(f"  {t}: {cols}")
        db.close()
    text = "\n".join(lines) + "\n"
    if args.out == "-":
        print(text)
    else:
        open(args.out, "w").write(text)


if __name__ == "__main__":
    main()
