"""Bisect the composite-agg GPU segfault: run each source combination in a
subprocess so a crash doesn't end the sweep. Usage (on the GPU box):
  python tools/dbg_comp.py            # driver: sweep configs
  python tools/dbg_comp.py run <idx>  # child: one config
"""
import json
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

CONFIGS = [
    {"sources": [{"u": {"terms": {"field": "user"}}}]},
    {"sources": [{"h": {"terms": {"field": "host", "missing_bucket": True}}}]},
    {"sources": [{"r": {"histogram": {"field": "resp", "interval": 50}}}]},
    {"sources": [{"u": {"terms": {"field": "user"}}},
                 {"r": {"histogram": {"field": "resp", "interval": 50}}}]},
    {"sources": [{"h": {"terms": {"field": "host", "missing_bucket": True}}},
                 {"u": {"terms": {"field": "user"}}}]},
    {"sources": [{"h": {"terms": {"field": "host", "missing_bucket": True}}},
                 {"u": {"terms": {"field": "user"}}},
                 {"r": {"histogram": {"field": "resp", "interval": 50}}}]},
]


def corpus():
    import random

    from quickwit_amd import splitgen
    rng = random.Random(3)
    hosts = [None, "10.0.0.1", "10.0.0.2", "10.0.0.3"]
    names = ["ann", "bob", "cat", "dan", "eve"]
    schema = {"timestamp_field": "timestamp", "fields":
              splitgen.HDFS_SCHEMA["fields"] +
              [{"name": "host", "type": "str", "fast": True},
               {"name": "user", "type": "str", "fast": True},
               {"name": "resp", "type": "i64", "fast": True}]}
    out = []
    for s in range(2):
        docs = []
        for i in range(400):
            d = {"timestamp": 1700000000 + i, "severity_text": "INFO",
                 "body": "x", "tenant_id": i % 3,
                 "user": rng.choice(names), "resp": rng.choice([0, 30, 100, 120])}
            h = rng.choice(hosts)
            if h is not None:
                d["host"] = h
            docs.append(d)
        w = splitgen.SplitWriter(schema, f"comp-{s}")
        w.add_documents(docs)
        out.append((f"comp-{s}", len(docs), w.finalize()))
    return schema, out


def child(idx):
    from quickwit_amd.api import GpuSearcher, OracleSearcher, make_leaf_request
    schema, splits = corpus()
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    for sid, n, data in splits:
        gpu.add_split(sid, data)
        cpu.add_split(sid, data)
    aggs = {"c": {"composite": dict(CONFIGS[idx], size=100)}}
    req = make_leaf_request({"type": "match_all"}, schema,
                            [(s, n) for s, n, _ in splits], max_hits=0,
                            aggregation=aggs)
    g = gpu.leaf_search(req)
    e = cpu.leaf_search(req)
    gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
    ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
    print("MATCH" if gj == ej else "DIFF")
    if gj != ej:
        print("GPU:", json.dumps(gj)[:400])
        print("ORA:", json.dumps(ej)[:400])


if __name__ == "__main__":
    if len(sys.argv) > 2 and sys.argv[1] == "run":
        child(int(sys.argv[2]))
        sys.exit(0)
    for i in range(len(CONFIGS)):
        r = subprocess.run(
            [sys.executable, os.path.abspath(__file__), "run", str(i)],
            capture_output=True, text=True, timeout=300)
        tail = (r.stdout + r.stderr).strip().splitlines()[-3:]
        print(f"config {i}: rc={r.returncode} {' | '.join(tail)}", flush=True)
