#!/usr/bin/env bash
# PMC-counter capture recipe for the gfx950 kernels (next measurement round).
#
# gpurun REFUSES rocprofv3 commands that combine --pmc (or -i counter files)
# with -s/--sys-trace, -r/--runtime-trace or the hip/hsa/memory-copy/
# scratch-memory/marker trace domains (suspected node-crasher) — so counters
# are collected in their OWN run, kernel-trace/stats in another.
#
# Usage on a GPU box:  bash tools/rocprof_pmc.sh "python bench.py --steps 3 --warmup 2"
set -euo pipefail
CMD=${1:-"python bench.py --steps 3 --warmup 2"}
cd /tmp && export TMPDIR=/tmp
OUT=${GRAFT_REPO_ROOT:-$PWD}/gpurun_out/pmc
mkdir -p "$OUT"
# MFMA utilization + LDS conflicts + HBM traffic, one metric set per run
rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_INSTS_MFMA -d "$OUT/mfma" -- bash -c "$CMD"
rocprofv3 --pmc SQ_INSTS_LDS SQ_WAIT_ANY -d "$OUT/lds" -- bash -c "$CMD"
rocprofv3 --pmc TCC_EA0_RDREQ_sum TCC_EA0_WRREQ_sum -d "$OUT/hbm" -- bash -c "$CMD"
echo "counter DBs under $OUT — summarize with tools/summarize_prof.py"
