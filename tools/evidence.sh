#!/bin/bash
# Round evidence collection on the GPU box (run under gpurun).
# Produces: gpurun_out/bench_*.json, gpurun_out/prof_*/ (rocprofv3 dbs),
# gpurun_out/*_summary.txt (prof_summary tables), gpurun_out/pytest_gpu.log
set -x
export TMPDIR=/tmp
python -m pytest tests -m gpu -q 2>&1 | tail -2 | tee gpurun_out/pytest_gpu.log

# bench lines (roofline + cpu_baseline in each)
python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 > gpurun_out/bench_bm25_10M.json
for wl in bm25 range agg; do
  python bench.py --workload $wl --docs 100000000 --steps 20 --warmup 3 \
    2>/dev/null | tail -1 > gpurun_out/bench_${wl}_100M.json
done

# kernel-trace profiles of the dominant kernels (no PMC mixed in)
for wl in bm25 agg range; do
  rocprofv3 --kernel-trace --stats -d gpurun_out/prof_$wl -o $wl -- \
    python bench.py --workload $wl --docs 100000000 --steps 5 --warmup 2 \
    --cpu-baseline-steps 0 > /dev/null 2>&1
  db=$(ls gpurun_out/prof_$wl/*.db 2>/dev/null | head -1)
  [ -n "$db" ] && python tools/prof_summary.py "$db" gpurun_out/prof_${wl}_summary.txt
done

# PMC passes (separate from any tracing): HBM traffic of the agg kernel,
# issue/stall mix of the decode kernel
rocprofv3 --pmc FETCH_SIZE -d gpurun_out/pmc_fetch -o fetch -- \
  python bench.py --workload agg --docs 100000000 --steps 2 --warmup 1 \
  --cpu-baseline-steps 0 > /dev/null 2>&1
rocprofv3 --pmc WRITE_SIZE -d gpurun_out/pmc_write -o write -- \
  python bench.py --workload agg --docs 100000000 --steps 2 --warmup 1 \
  --cpu-baseline-steps 0 > /dev/null 2>&1
rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_ACTIVE_INST_ANY SQ_WAIT_INST_ANY \
  -d gpurun_out/pmc_sq -o sq -- \
  python bench.py --workload bm25 --docs 100000000 --steps 2 --warmup 1 \
  --cpu-baseline-steps 0 > /dev/null 2>&1
for p in pmc_fetch pmc_write pmc_sq; do
  db=$(ls gpurun_out/$p/*.db 2>/dev/null | head -1)
  [ -n "$db" ] && python tools/prof_summary.py "$db" gpurun_out/${p}_summary.txt
done
rm -rf gpurun_out/prof_*/ gpurun_out/pmc_fetch gpurun_out/pmc_write gpurun_out/pmc_sq
ls -la gpurun_out/
