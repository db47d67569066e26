#!/bin/bash
# Round-2 evidence collection on the GPU box (run under gpurun).
# Produces gpurun_out/r02final_*: bench JSON lines (roofline+cpu_baseline),
# rocprofv3 kernel-trace summaries for the dominant kernels, and PMC passes
# (FETCH/WRITE for bm25+agg, SQ wait/issue mix for bm25) — collected in
# separate runs per the rocprofv3 constraints.
set -x
export TMPDIR=/tmp
R=${GRAFT_REPO_ROOT:-/root/repo}
cd "$R"

python -m pytest tests -m gpu -q 2>&1 | tail -2 | tee gpurun_out/r02final_pytest_gpu.log

# bench lines at the headline + secondary configs
python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 > gpurun_out/r02final_bench_bm25_100M.json
python bench.py --workload range --steps 20 --warmup 5 2>/dev/null | tail -1 > gpurun_out/r02final_bench_range_100M.json
python bench.py --workload agg --steps 20 --warmup 5 2>/dev/null | tail -1 > gpurun_out/r02final_bench_agg_100M.json
python bench.py --workload config5 --docs 125000000 --splits 8 --steps 10 --warmup 3 2>/dev/null | tail -1 > gpurun_out/r02final_bench_config5.json
python bench.py --docs 10000000 --steps 30 --warmup 5 2>/dev/null | tail -1 > gpurun_out/r02final_bench_bm25_10M.json

# kernel-trace profiles (no PMC mixed in); summaries cross-check the bench
# ms_per_launch numbers
cd /tmp
for wl in bm25 agg range; do
  rocprofv3 --kernel-trace --stats -d "$R/gpurun_out/prof_$wl" -o $wl -- \
    python "$R/bench.py" --workload $wl --docs 100000000 --steps 5 --warmup 2 \
    --cpu-baseline-steps 0 > /dev/null 2>&1
  db=$(ls "$R"/gpurun_out/prof_$wl/*.db 2>/dev/null | head -1)
  [ -n "$db" ] && python "$R/tools/prof_summary.py" "$db" "$R/gpurun_out/r02final_prof_${wl}_summary.txt"
done

# PMC passes (separate from tracing)
rocprofv3 --pmc FETCH_SIZE -d "$R/gpurun_out/pmc_f_bm25" -o f -- \
  python "$R/bench.py" --steps 2 --warmup 1 --cpu-baseline-steps 0 > /dev/null 2>&1
rocprofv3 --pmc WRITE_SIZE -d "$R/gpurun_out/pmc_w_bm25" -o w -- \
  python "$R/bench.py" --steps 2 --warmup 1 --cpu-baseline-steps 0 > /dev/null 2>&1
rocprofv3 --pmc FETCH_SIZE -d "$R/gpurun_out/pmc_f_agg" -o f -- \
  python "$R/bench.py" --workload agg --steps 2 --warmup 1 --cpu-baseline-steps 0 > /dev/null 2>&1
rocprofv3 --pmc WRITE_SIZE -d "$R/gpurun_out/pmc_w_agg" -o w -- \
  python "$R/bench.py" --workload agg --steps 2 --warmup 1 --cpu-baseline-steps 0 > /dev/null 2>&1
rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_ACTIVE_INST_ANY SQ_WAIT_INST_ANY \
  -d "$R/gpurun_out/pmc_sq_bm25" -o sq -- \
  python "$R/bench.py" --steps 2 --warmup 1 --cpu-baseline-steps 0 > /dev/null 2>&1
cd "$R"
for p in pmc_f_bm25 pmc_w_bm25 pmc_f_agg pmc_w_agg pmc_sq_bm25; do
  db=$(ls gpurun_out/$p/*.db 2>/dev/null | head -1)
  [ -n "$db" ] && python tools/prof_summary.py "$db" gpurun_out/r02final_${p}_summary.txt
done
rm -rf gpurun_out/prof_bm25 gpurun_out/prof_agg gpurun_out/prof_range \
       gpurun_out/pmc_f_bm25 gpurun_out/pmc_w_bm25 gpurun_out/pmc_f_agg \
       gpurun_out/pmc_w_agg gpurun_out/pmc_sq_bm25
ls -la gpurun_out/ | grep r02final
