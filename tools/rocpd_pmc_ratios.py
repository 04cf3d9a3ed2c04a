"""Per-kernel PMC ratio table from a rocprofv3 --pmc rocpd database.

Uses the flat `counters_collection` view (kernel_name, counter_name, value).
Ratios are against SQ_WAVE_CYCLES (PMC serialization inflates absolute
times; ratios stay valid).
"""

import collections
import sqlite3
import sys


def main():
    con = sqlite3.connect(sys.argv[1])
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    for kn, cn, val in con.execute(
            "SELECT kernel_name, counter_name, SUM(value) "
            "FROM counters_collection GROUP BY kernel_name, counter_name"):
        agg[kn][cn] = val or 0.0
    print("| kernel | wave-cycles | wait-any | wait-inst | active-inst "
          "| bank-conf | mfma-busy |")
    print("|---|---|---|---|---|---|---|")
    rows = sorted(agg.items(), key=lambda x: -x[1]["SQ_WAVE_CYCLES"])[:6]
    for kn, d in rows:
        wc = d["SQ_WAVE_CYCLES"] or 1.0

        def pct(n):
            return f"{100.0 * d[n] / wc:.1f}%"

        print(f"| `{kn.split('(')[0][:48]}` | {wc:.2e} | {pct('SQ_WAIT_ANY')} "
              f"| {pct('SQ_WAIT_INST_ANY')} | {pct('SQ_ACTIVE_INST_ANY')} "
              f"| {pct('SQ_LDS_BANK_CONFLICT')} "
              f"| {pct('SQ_VALU_MFMA_BUSY_CYCLES')} |")


if __name__ == "__main__":
    main()
