"""Kernel timing experiment: where does k_leaf_tile time go?

Times the dominant kernel via qw_kernel_stats (HIP events) across request
variants that toggle kernel phases on/off:
  count_only   — decode+score+count, NO candidate collection (collect=0)
  collect      — + per-doc candidate append (the suspected atomic hot spot)
  agg_only     — decode+count+aggregations, no collection
  matchall_agg — column-scan aggregations only (no posting decode)
Usage: python tools/exp_kernel.py [--docs N]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from quickwit_amd import proto, splitgen
from quickwit_amd.api import GpuSearcher, make_leaf_request


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=10_000_000)
    ap.add_argument("--reps", type=int, default=5)
    args = ap.parse_args()

    t0 = time.time()
    data = splitgen.generate_split(0, args.docs, seed=42)
    sid = "synthetic-42-0000"
    print(f"gen {time.time()-t0:.1f}s", flush=True)
    s = GpuSearcher(device=0)
    s.add_split(sid, data)

    terms = ["w%05d" % i for i in (9, 10, 11)]
    q3 = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": t} for t in terms]}
    aggs = {"per_day": {"date_histogram": {"field": "timestamp",
                                           "fixed_interval": "3600000ms"}},
            "per_tenant": {"terms": {"field": "tenant_name", "size": 10}}}
    score = [{"field_name": "_score", "sort_order": 1}]
    variants = [
        ("count_only", dict(query_ast=q3, max_hits=0)),
        ("collect_top10", dict(query_ast=q3, max_hits=10, sort_fields=score)),
        ("collect_top1000", dict(query_ast=q3, max_hits=1000, sort_fields=score)),
        ("agg_only", dict(query_ast=q3, max_hits=0, aggregation=aggs)),
        ("matchall_agg", dict(query_ast={"type": "match_all"}, max_hits=0,
                              aggregation=aggs)),
        ("matchall_range", dict(query_ast={
            "type": "range", "field": "tenant_id",
            "lower_bound": {"included": 100},
            "upper_bound": {"excluded": 200}}, max_hits=100)),
    ]
    for name, kw in variants:
        qa = kw.pop("query_ast")
        req = make_leaf_request(qa, splitgen.HDFS_SCHEMA, [(sid, args.docs)], **kw)
        req_pb = proto.encode("LeafSearchRequest", req)
        s.leaf_search_raw(req_pb)
        s.leaf_search_raw(req_pb)
        s.kernel_stats_reset()
        tw0 = time.perf_counter()
        for _ in range(args.reps):
            s.leaf_search_raw(req_pb)
        wall = (time.perf_counter() - tw0) / args.reps * 1e3
        stats = {}
        for k in ("union_bm25", "range_filter", "column_agg", "topk_select"):
            ms, n = s.kernel_stats(k)
            if n:
                stats[k] = f"{ms/n:.3f}ms x{n}"
        print(f"{name:18s} wall={wall:8.3f}ms  {stats}", flush=True)


if __name__ == "__main__":
    main()
