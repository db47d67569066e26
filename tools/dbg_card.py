"""Debug str-column cardinality mismatch: single split, one agg at a time,
print both engines' finalized JSON. Run on the GPU box."""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import random

from quickwit_amd import splitgen
from quickwit_amd.api import GpuSearcher, OracleSearcher, make_leaf_request

rng = random.Random(13)
docs = []
for i in range(500):
    d = {"timestamp": 1700000000 + (i % 97), "severity_text":
         rng.choice(["INFO", "WARN", "ERROR"]), "body": "x",
         "tenant_id": i % 41}
    if rng.random() < 0.7:
        d["opt_u"] = rng.randrange(29)
    docs.append(d)
schema = {"timestamp_field": "timestamp", "fields":
          splitgen.HDFS_SCHEMA["fields"] +
          [{"name": "opt_u", "type": "u64", "fast": True}]}
w = splitgen.SplitWriter(schema, "card-0")
w.add_documents(docs)
data = w.finalize()
gpu, cpu = GpuSearcher(device=0), OracleSearcher()
gpu.add_split("card-0", data)
cpu.add_split("card-0", data)

CASES = [
    {"sev": {"cardinality": {"field": "severity_text"}}},
    {"sev_t": {"terms": {"field": "severity_text", "size": 10}}},
    {"ten": {"cardinality": {"field": "tenant_id"}}},
    {"sev": {"cardinality": {"field": "severity_text"}},
     "ten": {"cardinality": {"field": "tenant_id"}}},
]
for aggs in CASES:
    req = make_leaf_request({"type": "match_all"}, schema,
                            [("card-0", len(docs))], max_hits=0,
                            aggregation=aggs)
    try:
        g = gpu.leaf_search(req)
        gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
    except Exception as ex:
        gj = f"ERR {ex}"
    e = cpu.leaf_search(req)
    ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
    print(list(aggs), "GPU:", json.dumps(gj)[:200], "ORA:",
          json.dumps(ej)[:200], flush=True)
