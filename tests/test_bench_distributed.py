"""End-to-end dry run of bench.py's N>1 contract path on CPU (gloo,
world_size=2): the exact code the driver executes for the scaling curve —
torchrun env, per-rank split generation, the packed-tensor
distributed_merge, max-over-ranks timing and the rank-0 JSON line — with
the oracle standing in for the GPU engine (QW_BENCH_ENGINE=oracle; the
measured runs always use the HIP product)."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench_world2(extra_args):
    env = dict(os.environ, QW_BENCH_ENGINE="oracle",
               MASTER_ADDR="127.0.0.1")
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29531", os.path.join(REPO, "bench.py"),
           "--gpus", "2", "--docs", "60000", "--steps", "3", "--warmup", "1",
           "--cpu-baseline-steps", "0"] + extra_args
    out = subprocess.run(cmd, env=env, capture_output=True, text=True,
                         timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.strip().splitlines()
             if ln.startswith("{")]
    assert len(lines) == 1, out.stdout[-1500:]  # exactly one rank-0 JSON line
    return json.loads(lines[0])


def test_bench_world2_bm25():
    j = run_bench_world2([])
    assert j["n_gpus"] == 2
    assert j["scaling"] == "weak"
    assert j["config"]["parallelism"] == "split-dp2"
    assert j["config"]["docs_per_gpu"] == 60000
    # whole-job value counts BOTH ranks' docs
    assert j["value"] > 0
    assert j["metric"] == "leaf_search_docs_per_sec"
    # merged hits exist and num_hits sums both ranks' splits
    assert j["config"]["num_hits_per_split"] > 0


def test_bench_world2_agg():
    j = run_bench_world2(["--workload", "agg"])
    assert j["n_gpus"] == 2
    # match_all counts every doc across both ranks
    assert j["config"]["num_hits_per_split"] == 60000
