#!/bin/bash
# Provenance: exact commands that produced the committed summaries.
# Run on an MI355X box (via gpurun); outputs land in gpurun_out/, then
# summarize.py distills them into profiles/.
set -e
cd "$(dirname "$0")/.."
export TMPDIR=/tmp
OUT=gpurun_out/profile_$(date +%s 2>/dev/null || echo run)
mkdir -p "$OUT"
# kernel timing (never combine --pmc with trace domains)
rocprofv3 --kernel-trace --stats -d "$OUT" -o trace -- \
    python bench.py --steps 6 --warmup 2 --no-cpu-baseline
# HBM traffic, separate single-counter passes (gfx950: FETCH_SIZE reads
# HALF the bytes of a wide coalesced stream — double before comparing)
rocprofv3 --pmc FETCH_SIZE -d "$OUT" -o fetch -- \
    python bench.py --steps 3 --warmup 1 --no-cpu-baseline
rocprofv3 --pmc WRITE_SIZE -d "$OUT" -o write -- \
    python bench.py --steps 3 --warmup 1 --no-cpu-baseline
rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY \
    -d "$OUT" -o sq -- python bench.py --steps 2 --warmup 1 --no-cpu-baseline --batch 512
echo "now: python profiles/summarize.py $OUT/<x>_results.db > profiles/<label>.txt"
