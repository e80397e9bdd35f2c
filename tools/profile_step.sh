#!/bin/bash
# rocprofv3 recipes for the WAP train step on an MI355X box.
# Run from the repo root on the GPU box:  bash tools/profile_step.sh
# (cd to /tmp first for rocprof scratch, per pool guidance)
set -ex
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
OUT=${1:-gpurun_out/prof}

# per-kernel time breakdown of the graphed train step
rocprofv3 --kernel-trace --stats --output-format csv -d "$OUT" -o step \
  -- python bench.py --steps 4 --warmup 2 --no-decode

# PMC counters (own run; never combine --pmc with sys/runtime traces)
rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,SQ_WAVE_CYCLES,SQ_LDS_BANK_CONFLICT,SQ_WAIT_ANY \
  --kernel-trace --output-format csv -d "$OUT" -o conv_pmc \
  -- python tools/bench_conv.py
