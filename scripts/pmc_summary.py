"""Summarize a rocprofv3 --pmc results.db: per-kernel counter totals and
derived rates (MFMA busy fraction, VALU-instrs per wave-cycle, LDS instrs).
Usage: python scripts/pmc_summary.py <results.db>
Collect counters in their OWN run (no --sys-trace/--runtime-trace — the
combination is refused by gpurun)."""

import collections
import sqlite3
import sys


def main():
    db = sqlite3.connect(sys.argv[1])
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE "
        "'rocpd_pmc_event%'")][0]
    sfx = t.replace("rocpd_pmc_event_", "")
    # discover join columns defensively (schema varies across rocprofv3)
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({t})")]
    kd_cols = [r[1] for r in cur.execute(
        f"PRAGMA table_info(rocpd_kernel_dispatch_{sfx})")]
    pmc_id = next((c for c in cols if "pmc" in c.lower()), None)
    disp = next((c for c in cols if "dispatch" in c.lower()
                 or "event" in c.lower() or "corr" in c.lower()), None)
    kd_key = "id"
    if disp and "corr" in disp.lower():
        kd_key = next((c for c in kd_cols if "corr" in c.lower()), "id")
    if pmc_id is None or disp is None:
        print("pmc_event columns:", cols)
        print("kernel_dispatch columns:", kd_cols)
        raise SystemExit("cannot infer join columns")
    rows = list(cur.execute(f"""
        SELECT ks.display_name, pi.name, SUM(pe.value)
        FROM {t} pe
        JOIN rocpd_info_pmc_{sfx} pi ON pe.{pmc_id} = pi.id
        JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.{disp} = kd.{kd_key}
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name, pi.name"""))
    per_kernel = collections.defaultdict(dict)
    for kname, cname, val in rows:
        short = kname.replace("(anonymous namespace)::", "")
        short = short.replace("tdsa::", "")
        if short.startswith("void "):
            short = short[5:]
        depth = 0
        for i, c in enumerate(short):
            if c == "<":
                depth += 1
            elif c == ">":
                depth -= 1
            elif c == "(" and depth == 0:
                short = short[:i]
                break
        per_kernel[short[:60]][cname] = val
    for k, cs in sorted(per_kernel.items()):
        print(k)
        for cname, val in sorted(cs.items()):
            print(f"    {cname:<36} {val:>18,.0f}")
        wc = cs.get("SQ_WAVE_CYCLES") or cs.get("SQ_BUSY_CYCLES")
        mfma = cs.get("SQ_VALU_MFMA_BUSY_CYCLES")
        valu = cs.get("SQ_INSTS_VALU")
        lds = cs.get("SQ_INSTS_LDS")
        if wc:
            if mfma:
                print(f"    -> MFMA busy / wave-cycle: {mfma / wc:.3f}")
            if valu:
                print(f"    -> VALU insts / wave-cycle: {valu / wc:.3f}")
            if lds:
                print(f"    -> LDS insts / wave-cycle: {lds / wc:.3f}")


if __name__ == "__main__":
    main()
