"""Multi-segment splits (QWA2) on the CPU oracle — the reference's splits
carry multiple tantivy segments and collection is per segment
(collector.rs:475-594); PartialHit.segment_ord + (split, segment, doc)
tie-breaks (sorting.md:14-17) must hold across the container boundary.

Equivalence oracle: a QWA2 split [segA, segB] under split id "s" must
answer exactly like the same two images searched as separate splits
"s-a" < "s-b" (same lexicographic order as segment ords), with hits
relabeled (s-a, 0, d) -> (s, 0, d) and (s-b, 0, d) -> (s, 1, d).
"""
import math

import pytest

from quickwit_amd import splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

NA, NB = 30_000, 20_000
SCHEMA = splitgen.HDFS_SCHEMA


@pytest.fixture(scope="module", autouse=True)
def build_all():
    import __graft_entry__
    __graft_entry__.build()


@pytest.fixture(scope="module")
def searchers():
    a = splitgen.generate_split(0, NA, seed=9)
    b = splitgen.generate_split(1, NB, seed=9)
    multi = OracleSearcher()
    multi.add_split("s", splitgen.concat_segments([a, b], "s"))
    flat = OracleSearcher()
    flat.add_split("s-a", a)
    flat.add_split("s-b", b)
    return multi, flat


def relabel(h):
    seg = {"s-a": 0, "s-b": 1}[h["split_id"]]
    out = dict(h)
    out["split_id"] = "s"
    if seg:
        out["segment_ord"] = seg
    else:
        out.pop("segment_ord", None)
    return out


def run_pair(searchers, query, **kw):
    multi, flat = searchers
    rm = multi.leaf_search(make_leaf_request(
        query, SCHEMA, [("s", NA + NB)], **kw))
    rf = flat.leaf_search(make_leaf_request(
        query, SCHEMA, [("s-a", NA), ("s-b", NB)], **kw))
    return rm, rf


def assert_equiv(rm, rf):
    assert rm.get("num_hits", 0) == rf.get("num_hits", 0)
    mh = rm.get("partial_hits", [])
    fh = [relabel(h) for h in rf.get("partial_hits", [])]
    assert mh == fh


def test_bm25_across_segments(searchers):
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i}
        for i in (9, 10, 11)]}
    rm, rf = run_pair(searchers, q, max_hits=20,
                      sort_fields=[{"field_name": "_score", "sort_order": 1}])
    assert_equiv(rm, rf)
    assert {h.get("segment_ord", 0) for h in rm["partial_hits"]} == {0, 1}


def test_doc_order_hits_across_segments(searchers):
    rm, rf = run_pair(searchers, {"type": "match_all"}, max_hits=15)
    assert_equiv(rm, rf)


def test_field_sort_two_keys_across_segments(searchers):
    q = {"type": "term", "field": "severity_text", "value": "INFO"}
    rm, rf = run_pair(searchers, q, max_hits=25, sort_fields=[
        {"field_name": "timestamp", "sort_order": 1},
        {"field_name": "tenant_id", "sort_order": 0}])
    assert_equiv(rm, rf)


def test_aggregations_merge_across_segments(searchers):
    multi, flat = searchers
    aggs = {"per_hour": {"date_histogram": {"field": "timestamp",
                                            "fixed_interval": "3600000ms"}},
            "per_tenant": {"terms": {"field": "tenant_name", "size": 10}},
            "stats": {"stats": {"field": "tenant_id"}}}
    q = {"type": "term", "field": "severity_text", "value": "INFO"}
    rm, rf = run_pair(searchers, q, max_hits=0, aggregation=aggs)
    assert rm["num_hits"] == rf["num_hits"]
    gm = multi.finalize_agg_json(rm["intermediate_aggregation_result"], aggs)
    gf = flat.finalize_agg_json(rf["intermediate_aggregation_result"], aggs)
    assert gm == gf


def test_search_after_pages_across_segments(searchers):
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i}
        for i in (9, 10, 11)]}
    multi, flat = searchers
    sort = [{"field_name": "_score", "sort_order": 1}]
    seen = []
    cursor = None
    for _page in range(4):
        req = make_leaf_request(q, SCHEMA, [("s", NA + NB)], max_hits=7,
                                sort_fields=sort)
        if cursor:
            req["search_request"]["search_after"] = cursor
        r = multi.leaf_search(req)
        hits = r.get("partial_hits", [])
        if not hits:
            break
        seen.extend(hits)
        cursor = hits[-1]
    # pages concatenate to the unpaged top-28 exactly
    r_all = multi.leaf_search(make_leaf_request(
        q, SCHEMA, [("s", NA + NB)], max_hits=len(seen), sort_fields=sort))
    assert seen == r_all["partial_hits"]
    assert {h.get("segment_ord", 0) for h in seen} == {0, 1}


def test_count_and_stats_per_split(searchers):
    rm, rf = run_pair(searchers, {"type": "term", "field": "severity_text",
                                  "value": "ERROR"}, max_hits=0)
    assert rm["num_hits"] == rf["num_hits"]
    # one split attempted/succeeded for the QWA2 container, two for flat
    assert rm.get("num_attempted_splits", 0) == 1
    assert rm.get("num_successful_splits", 0) == 1
    assert rf.get("num_attempted_splits", 0) == 2
    s = rm["resource_stats"]["split_resources_sum"]
    assert s["split_num_docs"] == NA + NB
