#!/usr/bin/env python3
"""Aggregate a rocprofv3 --pmc counter_collection.csv into per-kernel
means (markdown). Usage: pmc_agg.py <dir-with-csv> [out.md]

Robust to the two rocprofv3 CSV layouts seen on ROCm 7.x: one row per
(dispatch, counter) with Counter-Name/Counter-Value columns, or one row
per dispatch with a column per counter.
"""
import csv
import glob
import sys
from collections import defaultdict


def short(name: str, n: int = 60) -> str:
    name = name.split("(")[0].strip()
    return name if len(name) <= n else name[: n - 1] + "…"


def main() -> int:
    d = sys.argv[1]
    files = sorted(glob.glob(f"{d}/**/*counter_collection.csv", recursive=True))
    if not files:
        print(f"no counter_collection.csv under {d}", file=sys.stderr)
        return 1
    # agg[kernel][counter] = [sum, n_dispatches]
    agg = defaultdict(lambda: defaultdict(lambda: [0.0, 0]))
    counters = []
    for f in files:
        rows = list(csv.DictReader(open(f)))
        if not rows:
            continue
        cols = rows[0].keys()
        kcol = next(c for c in cols if "Kernel" in c and "Name" in c)
        ncol = next((c for c in cols if "Counter" in c and "Name" in c), None)
        if ncol:  # long layout
            vcol = next(c for c in cols if "Counter" in c and "Value" in c)
            per_dispatch = defaultdict(dict)
            dcol = next((c for c in cols if "Dispatch" in c), None)
            for r in rows:
                key = (r[kcol], r.get(dcol, id(r)))
                per_dispatch[key][r[ncol]] = float(r[vcol])
            for (kern, _), cv in per_dispatch.items():
                for cname, v in cv.items():
                    a = agg[kern][cname]
                    a[0] += v
                    a[1] += 1
                    if cname not in counters:
                        counters.append(cname)
        else:  # wide layout
            skip = {kcol} | {c for c in cols if any(
                s in c for s in ("Id", "Name", "Size", "gpu", "queue", "pid",
                                 "tid", "Begin", "End", "Agent", "Grid",
                                 "Workgroup", "LDS", "VGPR", "SGPR", "sec"))}
            for r in rows:
                for c in cols:
                    if c in skip or not r[c]:
                        continue
                    try:
                        v = float(r[c])
                    except ValueError:
                        continue
                    a = agg[r[kcol]][c]
                    a[0] += v
                    a[1] += 1
                    if c not in counters:
                        counters.append(c)
    lines = ["| kernel | n | " + " | ".join(counters) + " |",
             "|---|---|" + "---|" * len(counters)]
    for kern, cv in sorted(agg.items(),
                           key=lambda kv: -max(v[0] for v in kv[1].values())):
        n = max((v[1] for v in cv.values()), default=0)
        vals = []
        for c in counters:
            s, cnt = cv.get(c, [0.0, 0])
            vals.append(f"{s / cnt:,.0f}" if cnt else "-")
        lines.append(f"| `{short(kern)}` | {n} | " + " | ".join(vals) + " |")
    out = "\n".join(lines) + "\n"
    if len(sys.argv) > 2:
        open(sys.argv[2], "w").write(out)
    print(out)
    return 0


if __name__ == "__main__":
    sys.exit(main())
