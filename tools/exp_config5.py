"""Where does config5's per-split kernel time go? Ablates the query parts
over one 15.6M-doc split and prints the dominant-kernel ms per launch."""
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from quickwit_amd import proto, splitgen  # noqa: E402
from quickwit_amd.api import GpuSearcher, make_leaf_request  # noqa: E402

DOCS = 15_625_000
TERMS = ["w%05d" % i for i in (9, 10, 11)]
SHOULD = [{"type": "term", "field": "body", "value": t} for t in TERMS]
FILT = [{"type": "range", "field": "tenant_id",
         "lower_bound": {"included": 100}, "upper_bound": {"excluded": 900}}]
AGG = {"per_tenant": {"terms": {"field": "tenant_name", "size": 10}}}
SCORE = [{"field_name": "_score", "sort_order": 1}]


def main():
    t0 = time.time()
    data = splitgen.generate_split(0, DOCS, seed=42)
    print(f"gen {time.time()-t0:.1f}s", flush=True)
    s = GpuSearcher(device=0)
    s.add_split("s", data)
    variants = [
        ("full_config5", {"type": "bool", "should": SHOULD, "filter": FILT},
         1000, AGG, SCORE),
        ("no_agg", {"type": "bool", "should": SHOULD, "filter": FILT},
         1000, None, SCORE),
        ("no_filter", {"type": "bool", "should": SHOULD}, 1000, AGG, SCORE),
        ("no_collect", {"type": "bool", "should": SHOULD, "filter": FILT},
         0, AGG, None),
        ("agg_only_matchall", {"type": "match_all"}, 0, AGG, None),
        ("bm25_top10", {"type": "bool", "should": SHOULD}, 10, None, SCORE),
    ]
    for name, q, mh, agg, sort in variants:
        req = make_leaf_request(q, splitgen.HDFS_SCHEMA, [("s", DOCS)],
                                max_hits=mh, sort_fields=sort,
                                aggregation=agg)
        pb = proto.encode("LeafSearchRequest", req)
        for _ in range(3):
            s.leaf_search_raw(pb)
        s.kernel_stats_reset()
        t1 = time.perf_counter()
        for _ in range(10):
            s.leaf_search_raw(pb)
        wall = (time.perf_counter() - t1) / 10 * 1e3
        parts = []
        for k in ("union_bm25", "column_agg", "range_filter", "topk_select"):
            try:
                ms, n = s.kernel_stats(k)
                if n:
                    parts.append(f"{k}={ms/n:.3f}ms x{n}")
            except KeyError:
                pass
        print(f"{name:18s} wall {wall:6.3f} ms | " + "  ".join(parts),
              flush=True)


if __name__ == "__main__":
    main()
