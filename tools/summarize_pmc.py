#!/usr/bin/env python3
"""Summarize rocprofv3 --pmc counter CSVs per kernel.

Usage: python tools/summarize_pmc.py gpurun_out/pmc/*/ [out.md]

Reads every *counter_collection.csv under the given dirs (long format: one
row per counter per dispatch), sums counter values per kernel name, and
prints a markdown table. Derived columns when the inputs allow:
  mfma_busy_pct = SQ_VALU_MFMA_BUSY_CYCLES / (SQ_WAVE_CYCLES / waves...) —
                  reported raw; interpret with the microarch guide.
  hbm_rd/wr_GB  = TCC_EA0_*REQ_sum * 64 B (NOTE: on gfx950 FETCH under-counts
                  wide coalesced reads 2x — calibrate before absolutes).
"""
import csv
import glob
import os
import sys
from collections import defaultdict


def load(paths):
    # {kernel: {counter: sum}}, {kernel: dispatches}
    agg = defaultdict(lambda: defaultdict(float))
    disp = defaultdict(set)
    for d in paths:
        for f in glob.glob(os.path.join(d, "**", "*counter_collection.csv"),
                           recursive=True):
            with open(f) as fh:
                for row in csv.DictReader(fh):
                    name = row.get("Kernel_Name") or row.get("KernelName") or "?"
                    cname = row.get("Counter_Name") or row.get("CounterName")
                    cval = row.get("Counter_Value") or row.get("CounterValue")
                    if not cname or cval in (None, ""):
                        continue
                    agg[name][cname] += float(cval)
                    key = row.get("Dispatch_Id") or row.get("DispatchId")
                    disp[name].add((f, key))
    return agg, disp


def short(name, n=70):
    name = name.split("(")[0]
    return (name if len(name) <= n else name[: n - 3] + "...").replace("|", "\\|")


def main():
    args = [a for a in sys.argv[1:]]
    out = None
    if args and args[-1].endswith(".md"):
        out = args.pop()
    agg, disp = load(args or ["gpurun_out/pmc"])
    counters = sorted({c for v in agg.values() for c in v})
    lines = ["| kernel | dispatches | " + " | ".join(counters) + " |",
             "|---" * (2 + len(counters)) + "|"]
    order = sorted(agg, key=lambda k: -max(agg[k].values(), default=0))
    for k in order[:40]:
        vals = " | ".join(f"{agg[k].get(c, 0):.3e}" for c in counters)
        lines.append(f"| `{short(k)}` | {len(disp[k])} | {vals} |")
    text = "\n".join(lines) + "\n"
    if out:
        open(out, "w").write(text)
        print(f"wrote {out}")
    else:
        print(text)


if __name__ == "__main__":
    main()
