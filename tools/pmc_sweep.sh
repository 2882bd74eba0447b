#!/bin/bash
# Round-2 PMC collection plan (docs/ROADMAP.md §Perf 3/4): per-kernel
# hardware counters for the BN/LN/optimizer kernels that round 1 only
# profiled with --stats. Run ON A GPU BOX via gpurun, e.g.
#
#   /usr/local/graft/bin/gpurun --timeout 900 -- bash tools/pmc_sweep.sh
#
# gpurun REFUSES rocprofv3 runs that mix --pmc with sys/runtime/hip/hsa
# trace domains — each line below uses --pmc with --kernel-trace only,
# per the pool rules. Outputs land in gpurun_out/pmc/ (merged back).
set -e
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export TMPDIR=/tmp
OUT=gpurun_out/pmc
mkdir -p "$OUT"

# memory-traffic counters: are the BN reduce kernels fetch-bound?
rocprofv3 --kernel-trace --pmc FETCH_SIZE WRITE_SIZE \
    --output-format csv -d "$OUT/bn_fetch" -- \
    python tools/bench_kernels.py --reps 20 > "$OUT/bn_fetch.log" 2>&1

# wave occupancy + LDS conflicts on the same suite
rocprofv3 --kernel-trace --pmc SQ_WAVES SQ_INSTS_LDS SQC_DCACHE_REQ \
    --output-format csv -d "$OUT/bn_occ" -- \
    python tools/bench_kernels.py --reps 20 > "$OUT/bn_occ.log" 2>&1

# end-to-end ResNet-50 step: MFMA mix of MIOpen conv vs our kernels
rocprofv3 --kernel-trace --pmc SQ_INSTS_MFMA SQ_WAVE_CYCLES \
    --output-format csv -d "$OUT/r50_mfma" -- \
    python bench.py --steps 3 --warmup 2 > "$OUT/r50_mfma.log" 2>&1

echo "pmc sweep done:"; du -sh "$OUT"
