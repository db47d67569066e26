"""Pin bench.py's driver-contract JSON line (single process, oracle
engine, tiny synthetic split): exactly one JSON line on stdout with every
key the driver parses, correct types, and the BASELINE.json metric name.
The world-size-2 torchrun path is covered by test_bench_distributed.py;
the measured numbers always come from the HIP product on the GPU box."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(extra):
    env = dict(os.environ, QW_BENCH_ENGINE="oracle")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--docs", "50000",
         "--steps", "3", "--warmup", "1", "--cpu-baseline-steps", "0"] + extra,
        env=env, capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.strip().splitlines()
             if ln.startswith("{")]
    assert len(lines) == 1, out.stdout[-1000:]
    return json.loads(lines[0])


def test_default_line_shape():
    j = run_bench([])
    # BASELINE.json's "metric" is prose ("leaf_search docs/sec + p50
    # latency, 3-term BM25 over 100M docs, ..."); the line carries its slug
    assert j["metric"] == "leaf_search_docs_per_sec"
    assert j["unit"] == "docs/s"
    assert "p50_ms" in j  # the baseline metric's latency half
    assert j["n_gpus"] == 1 and j["steps"] == 3 and j["warmup"] == 1
    assert isinstance(j["value"], (int, float)) and j["value"] > 0
    assert isinstance(j["ms_per_step"], (int, float)) and j["ms_per_step"] > 0
    assert j["higher_is_better"] is True
    assert j["scaling"] == "weak"
    assert j["vs_baseline"] is None  # no published number for this metric
    assert j["data"] == "synthetic"
    assert isinstance(j["dtype"], str)
    cfg = j["config"]
    assert cfg["workload"].startswith("3term_bm25_or")
    assert cfg["docs_per_gpu"] == 50000
    assert cfg["parallelism"] == "split-dp1"
    assert "model" not in cfg  # non-neural workload: no model keys
    # roofline/cpu_baseline are null on the oracle dry run (no HIP events),
    # but the keys must exist for the driver
    assert "roofline" in j and "cpu_baseline" in j


def test_workload_lines_exist():
    names = {"range": "must_info_tenant_range_50000",
             "agg": "datehisto_terms_agg_50000"}
    for wl, expect in names.items():
        j = run_bench(["--workload", wl])
        assert j["value"] > 0, wl
        assert j["config"]["workload"] == expect
        assert j["config"]["docs_per_gpu"] == 50000, wl
