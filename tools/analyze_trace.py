#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel_trace.csv into a steady-state per-step
kernel table.

MIOpen's find phase runs every candidate algorithm once (including
multi-ms naive kernels) and dominates a whole-run `--stats` summary; this
tool drops everything up to the last find-phase kernel and normalizes by
step count (inferred from a reference kernel that runs a known number of
times per step).

Usage:
    rocprofv3 --kernel-trace --output-format csv -d OUT -- python bench.py ...
    python tools/analyze_trace.py OUT/**/*kernel_trace.csv [--per-step N:NAME]

--per-step defaults to 32:fattn_score_fwd (the decoder unroll runs the
fused attention kernel seq_len=32 times per training step).
"""

import argparse
import csv
from collections import defaultdict


def main():
    p = argparse.ArgumentParser()
    p.add_argument("trace")
    p.add_argument("--per-step", default="32:fattn_score_fwd",
                   help="N:SUBSTR — kernel SUBSTR runs N times per step")
    p.add_argument("--top", type=int, default=25)
    args = p.parse_args()
    n_per, ref_name = args.per_step.split(":", 1)
    n_per = int(n_per)

    rows = list(csv.DictReader(open(args.trace)))
    find_end = max((int(r["End_Timestamp"]) for r in rows
                    if "naive_conv" in r["Kernel_Name"]), default=0)
    agg = defaultdict(lambda: [0, 0])
    tot = 0
    for r in rows:
        s = int(r["Start_Timestamp"])
        if s <= find_end:
            continue
        d = int(r["End_Timestamp"]) - s
        k = r["Kernel_Name"][:70]
        agg[k][0] += d
        agg[k][1] += 1
        tot += d
    refs = [k for k in agg if ref_name in k]
    if not refs:
        raise SystemExit(f"reference kernel {ref_name!r} not in trace")
    nst = agg[refs[0]][1] / n_per
    ndisp = sum(c for _, (d, c) in agg.items())
    print(f"steady-state: {tot / 1e6:.1f} ms kernel over ~{nst:.1f} steps "
          f"-> {tot / 1e6 / nst:.2f} ms/step, {ndisp / nst:.0f} disp/step")
    for k, (d, c) in sorted(agg.items(), key=lambda x: -x[1][0])[:args.top]:
        print(f"{d / 1e6 / nst:7.2f} ms/step {100 * d / tot:5.1f}% "
              f"{c / nst:7.1f}/step  {k}")


if __name__ == "__main__":
    main()
