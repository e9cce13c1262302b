"""Digest multi-GPU scaling results (round-3 prep: the driver emits
SCALE_rNN.json with per-N bench values; bench.py prints a [comm-stats]
stderr line on multi-rank runs).

Usage:
    python scripts/scale_analysis.py SCALE_r02.json
    python scripts/scale_analysis.py bench_n1.json bench_n2.json ...

Prints per-N throughput, weak-scaling efficiency vs N=1, per-GPU
tokens/s, and — when comm-stats lines are pasted on stdin — the implied
xGMI bytes/step against the 153 GB/s per-link ring bound.
"""

import json
import sys


def load_points(paths):
    pts = []
    for path in paths:
        with open(path) as f:
            data = json.load(f)
        if isinstance(data, dict) and "points" in data:  # driver SCALE file
            for p in data["points"]:
                pts.append(p)
        elif isinstance(data, list):
            pts.extend(data)
        else:
            pts.append(data)
    out = []
    for p in pts:
        n = p.get("n_gpus") or p.get("n") or 1
        v = p.get("value")
        ms = p.get("ms_per_step")
        if v is not None:
            out.append((int(n), float(v), float(ms) if ms else None))
    return sorted(out)


def main():
    pts = load_points(sys.argv[1:])
    if not pts:
        print("no bench points found")
        return
    base_n, base_v, _ = pts[0]
    per_gpu_base = base_v / base_n
    print(f"{'N':>3} {'tokens/s':>12} {'per-GPU':>12} {'weak-eff':>9} "
          f"{'ms/step':>8}")
    for n, v, ms in pts:
        eff = (v / n) / per_gpu_base
        print(f"{n:>3} {v:>12,.0f} {v / n:>12,.0f} {eff:>8.1%} "
              f"{ms if ms is not None else float('nan'):>8.2f}")
    print()
    print("interpretation aids:")
    print(" - weak-eff drop at N=2 but flat 2->8: fixed per-step comm cost")
    print("   (exposed step-end broadcast; try --parallel zero2flat)")
    print(" - eff decaying with N: per-link-bound ring collectives; check")
    print("   the [comm-stats] stderr lines: gb_per_step x (N-1)/N per rank")
    print("   over 153 GB/s/link ~= lower-bound comm ms hidden or exposed")
    print(" - compare zero2 vs zero2flat at the same N before kernel work")


if __name__ == "__main__":
    main()
