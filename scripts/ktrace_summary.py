"""Summarize a rocprofv3 kernel-trace csv: steady-state per-kernel totals.

Usage: python scripts/ktrace_summary.py TRACE.csv [steady_frac] > summary.txt
Steady window = last `steady_frac` of the wall-clock span (default 0.35),
excluding capture/MIOpen-find warmup at the start.
"""
import csv
import re
import sys
from collections import defaultdict


def shorten(n):
    n = re.sub(r"<.*", "", n)
    n = re.sub(r"\(.*", "", n)
    return n[:80]


def main():
    path = sys.argv[1]
    frac = float(sys.argv[2]) if len(sys.argv) > 2 else 0.35
    rows = []
    with open(path) as f:
        for d in csv.DictReader(f):
            rows.append((d["Kernel_Name"], int(d["Start_Timestamp"]),
                         int(d["End_Timestamp"]), d["Stream_Id"]))
    t0 = min(r[1] for r in rows)
    t1 = max(r[2] for r in rows)
    print(f"total window {(t1 - t0) / 1e9:.2f}s, {len(rows)} dispatches")
    # anchor the steady window to the sgd_step dispatches (one per train
    # batch, absent from capture warmup / MIOpen find): last `frac` of them
    sgd = sorted(r[1] for r in rows if "sgd_step" in r[0] or "multi_tensor_apply" in r[0])
    if sgd:
        n = max(1, int(len(sgd) * frac))
        lo = sgd[-n]
        print(f"anchor: last {n} of {len(sgd)} sgd steps")
    else:
        lo = t0 + int((t1 - t0) * (1 - frac))
    ss = [r for r in rows if r[1] >= lo]
    sswall = (t1 - lo) / 1e9
    agg = defaultdict(lambda: [0, 0])
    for name, s, e, _ in ss:
        k = shorten(name)
        agg[k][0] += 1
        agg[k][1] += e - s
    tot = sum(v[1] for v in agg.values())
    print(f"steady window {sswall:.2f}s, busy {tot/1e9:.2f}s "
          f"({100 * tot / 1e9 / sswall:.0f}%), {len(ss)} dispatches")
    for k, (c, ns) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:30]:
        print(f"{ns/1e6:9.2f} ms {c:6d} x {ns/c/1000:8.2f} us  {k}")


if __name__ == "__main__":
    main()
