"""Summarize a rocprofv3 results.db kernel trace into a text table
(per-kernel total/avg time, call count, share). Usage:
    python scripts/prof_summary.py <results.db> [steps] > profiles/xxx.txt
"""

import collections
import sqlite3
import sys


def main():
    path = sys.argv[1]
    steps = float(sys.argv[2]) if len(sys.argv) > 2 else None
    db = sqlite3.connect(path)
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE "
        "'rocpd_kernel_dispatch%'")][0]
    sfx = t.replace("rocpd_kernel_dispatch_", "")
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
           AVG(kd.end-kd.start)/1e3, MAX(ks.arch_vgpr_count),
           MAX(ks.group_segment_size)
    FROM rocpd_kernel_dispatch_{sfx} kd
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC
    """
    rows = list(cur.execute(q))
    total = sum(r[2] for r in rows)
    print(f"{'total_ms':>9} {'calls':>6} {'avg_us':>9} {'%':>5} "
          f"{'vgpr':>5} {'lds':>6}  kernel")
    for name, n, ms, us, vgpr, lds in rows:
        # strip return type / namespace noise so every row is attributable
        # (a bare split("(") turned "(anonymous namespace)::k<...>(args)"
        # into "void " — the round-1 unnamed rows)
        short = name.replace("(anonymous namespace)::", "")
        short = short.replace("tdsa::", "")
        if short.startswith("void "):
            short = short[5:]
        depth = 0
        for i, c in enumerate(short):  # split at the ARG paren, not template ones
            if c == "<":
                depth += 1
            elif c == ">":
                depth -= 1
            elif c == "(" and depth == 0:
                short = short[:i]
                break
        if len(short) > 80:
            short = short[:80]
        print(f"{ms:9.2f} {n:6d} {us:9.1f} {100*ms/total:5.1f} "
              f"{vgpr:5d} {lds:6d}  {short}")
    print(f"\nTOTAL kernel time: {total:.1f} ms"
          + (f"  ({total/steps:.1f} ms/step over {steps:g} steps)"
             if steps else ""))


if __name__ == "__main__":
    main()
