"""Summarize a rocprofv3 rocpd sqlite database into a per-kernel table.

Usage: python tools/rocpd_kernel_stats.py <db path> [top_n]

Schema-introspective: finds the kernel-symbol table (kernel_name + register
counts) and the dispatch table (start/end timestamps keyed by symbol id)
whatever their exact rocpd_* names are in this ROCm build.
"""

import sqlite3
import sys


def main():
    db, top_n = sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 20
    con = sqlite3.connect(db)
    tables = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    cols = {t: [c[1] for c in con.execute(f"PRAGMA table_info('{t}')")]
            for t in tables}

    sym = next(t for t in tables if "kernel_name" in cols[t])
    disp = next(t for t in tables
                if "start" in cols[t] and "end" in cols[t]
                and any("kernel" in c and "id" in c for c in cols[t]))
    key = next(c for c in cols[disp] if "kernel" in c and "id" in c)
    sid = "id" if "id" in cols[sym] else cols[sym][0]

    def pick(*names):
        return next((c for c in cols[sym] if c in names), None)

    vgpr = pick("arch_vgpr_count", "vgpr_count", "vgpr")
    agpr = pick("accum_vgpr_count", "agpr_count", "agpr")
    lds = pick("group_segment_size", "lds_size", "static_lds_size")
    extra = ", ".join(f"s.{c}" for c in (vgpr, agpr, lds) if c)
    extra = (", " + extra) if extra else ""

    rows = con.execute(
        f"SELECT s.kernel_name, COUNT(*), SUM(d.end - d.start)/1e6, "
        f"AVG(d.end - d.start)/1e3{extra} "
        f"FROM '{disp}' d JOIN '{sym}' s ON d.{key} = s.{sid} "
        f"GROUP BY s.kernel_name ORDER BY SUM(d.end - d.start) DESC "
        f"LIMIT {top_n}").fetchall()
    total = con.execute(
        f"SELECT SUM(end - start)/1e6, COUNT(*) FROM '{disp}'").fetchone()

    print(f"Total GPU kernel time: {total[0]:.2f} ms, "
          f"{total[1]} dispatches")
    print("| total ms | n | avg µs | VGPR | AGPR | LDS B | kernel |")
    print("|---|---|---|---|---|---|---|")
    for r in rows:
        name, n, tot, avg = r[0], r[1], r[2], r[3]
        regs = list(r[4:]) + [""] * (3 - len(r[4:]))
        print(f"| {tot:8.2f} | {n} | {avg:8.1f} | {regs[0]} | {regs[1]} | "
              f"{regs[2]} | `{name[:64]}` |")


if __name__ == "__main__":
    main()
