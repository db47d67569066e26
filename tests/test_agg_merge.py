"""Cross-split aggregation merge equivalence (CPU): for every aggregation
family, merging per-split leaf responses through the PRODUCT's ctx-less
root-side merge (qw_merge_leaf_responses, the path bench.py's multi-GPU
reduce uses) must finalize to exactly what a single multi-split leaf search
produces. Covers the QAGG wire format round-trip (encode -> merge decode ->
re-encode -> finalize) for terms+subs, numeric terms, cardinality,
composite, percentiles and histograms.

Reference semantics: collector.rs:832-861 merge_fruits + tantivy
intermediate aggregation merge.
"""
import json
import os
import random
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from quickwit_amd import proto, splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request
from quickwit_amd.merge import merge_leaf_responses

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

SCHEMA = {"timestamp_field": "timestamp", "fields":
          splitgen.HDFS_SCHEMA["fields"] +
          [{"name": "svc", "type": "str", "fast": True},
           {"name": "lat", "type": "f64", "fast": True},
           {"name": "code", "type": "u64", "fast": True}]}


def corpus():
    rng = random.Random(21)
    docs = []
    for i in range(1200):
        d = {"timestamp": 1700000000 + i * 7, "severity_text":
             "INFO" if i % 3 else "ERROR", "body": "x", "tenant_id": i % 5,
             "svc": rng.choice(["api", "ing", "jan"]),
             "code": rng.choice([200, 204, 404, 500, 1769070189829214201])}
        if rng.random() < 0.85:
            d["lat"] = round(rng.uniform(0.5, 900.0), 3)
        docs.append(d)
    return docs


@pytest.fixture(scope="module")
def setup():
    import subprocess
    subprocess.run(["make", "-s", "-C", os.path.join(REPO, "oracle")],
                   check=True)
    docs = corpus()
    parts = [docs[0::3], docs[1::3], docs[2::3]]
    datas = []
    for i, part in enumerate(parts):
        w = splitgen.SplitWriter(SCHEMA, f"m-{i}")
        w.add_documents(part)
        datas.append(w.finalize())
    combined = OracleSearcher()
    singles = []
    for i, data in enumerate(datas):
        combined.add_split(f"m-{i}", data)
        s = OracleSearcher()
        s.add_split(f"m-{i}", data)
        singles.append(s)
    splits = [(f"m-{i}", len(parts[i])) for i in range(3)]
    return combined, singles, splits


AGG_CASES = {
    "histo_stats": {"h": {"date_histogram": {"field": "timestamp",
                          "fixed_interval": "1000000ms"},
                          "aggs": {"st": {"stats": {"field": "lat"}}}}},
    "terms_subs": {"t": {"terms": {"field": "svc", "size": 10},
                         "aggs": {"st": {"extended_stats": {"field": "lat"}},
                                  "mx": {"max": {"field": "lat"}}}}},
    "terms_numeric": {"n": {"terms": {"field": "code", "size": 10}}},
    "terms_truncated": {"t": {"terms": {"field": "svc", "size": 2,
                                        "split_size": 2}}},
    "cardinality": {"c1": {"cardinality": {"field": "svc"}},
                    "c2": {"cardinality": {"field": "code"}},
                    "c3": {"cardinality": {"field": "lat"}}},
    "composite": {"c": {"composite": {"size": 50, "sources": [
        {"s": {"terms": {"field": "svc"}}},
        {"r": {"histogram": {"field": "code", "interval": 100}}}]}}},
    "percentiles": {"p": {"percentiles": {"field": "lat",
                                          "percents": [50, 95]}},
                    "h": {"date_histogram": {"field": "timestamp",
                          "fixed_interval": "1000000ms"},
                          "aggs": {"lp": {"percentiles": {"field": "lat",
                                          "keyed": False}}}}},
    "range_metric": {"r": {"range": {"field": "code", "ranges": [
                          {"to": 300.0}, {"from": 300.0, "to": 600.0},
                          {"from": 600.0}]}},
                     "m": {"avg": {"field": "lat"}}},
    "terms_key_asc": {"t": {"terms": {"field": "svc", "size": 2,
                                      "split_size": 2,
                                      "order": {"_key": "asc"}}}},
    "terms_key_desc": {"t": {"terms": {"field": "svc", "size": 2,
                                       "split_size": 2,
                                       "order": {"_key": "desc"}}}},
    "terms_count_asc": {"t": {"terms": {"field": "svc", "size": 10,
                                        "order": {"_count": "asc"}}}},
}


@pytest.mark.parametrize("case", sorted(AGG_CASES))
def test_merged_equals_combined(setup, case):
    combined, singles, splits = setup
    aggs = AGG_CASES[case]
    for q in ({"type": "match_all"},
              {"type": "term", "field": "severity_text", "value": "ERROR"}):
        req_all = make_leaf_request(q, SCHEMA, splits, max_hits=5,
                                    aggregation=aggs)
        sreq_pb = proto.encode("SearchRequest", req_all["search_request"])
        want = combined.leaf_search(req_all)
        wj = combined.finalize_agg_json(
            want["intermediate_aggregation_result"], aggs)

        resps = []
        for i, s in enumerate(singles):
            req1 = make_leaf_request(q, SCHEMA, [splits[i]], max_hits=5,
                                     aggregation=aggs)
            resps.append(s.leaf_search_raw(proto.encode("LeafSearchRequest",
                                                        req1)))
        merged_pb = merge_leaf_responses(sreq_pb, resps)
        merged = proto.decode("LeafSearchResponse", merged_pb)
        assert merged.get("num_hits", 0) == want.get("num_hits", 0), case
        # finalize the merged blob through the PRODUCT's ctx-less ABI
        import ctypes

        from quickwit_amd.merge import _Buf, _get_lib
        lib = _get_lib()
        lib.qw_finalize_agg_to_json.argtypes = [
            ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p,
            ctypes.POINTER(_Buf)]
        buf = _Buf()
        blob = merged.get("intermediate_aggregation_result", b"")
        rc = lib.qw_finalize_agg_to_json(blob, len(blob),
                                         json.dumps(aggs).encode(),
                                         ctypes.byref(buf))
        assert rc == 0, case
        out = ctypes.string_at(buf.data, buf.len)
        lib.qw_buf_free(ctypes.byref(buf))
        got = json.loads(out)
        assert got == wj, (case, q, got, wj)


@pytest.mark.parametrize("case", sorted(AGG_CASES))
def test_merge_is_input_order_independent(setup, case):
    """merge_fruits' exact tie-breaks (sorting.md:14-26) make the merged
    result a function of the SET of leaf responses, not their arrival
    order — required for the multi-GPU allgather, where rank order is
    arbitrary. Hits, counters and finalized aggs must be identical for
    every permutation of the per-split responses; the one permitted
    order effect is f64 summation non-associativity in accumulated
    metrics (sum/avg wander by ~1 ulp — the reference's own merge adds
    f64 in arrival order too), so floats compare at rel 1e-12."""
    import itertools
    import math

    def eq(a, b):
        if isinstance(a, float) or isinstance(b, float):
            return (a == b or (isinstance(a, float) and isinstance(b, float)
                               and math.isclose(a, b, rel_tol=1e-12)))
        if isinstance(a, dict):
            return (isinstance(b, dict) and a.keys() == b.keys()
                    and all(eq(a[k], b[k]) for k in a))
        if isinstance(a, list):
            return (isinstance(b, list) and len(a) == len(b)
                    and all(eq(x, y) for x, y in zip(a, b)))
        return a == b
    combined, singles, splits = setup
    aggs = AGG_CASES[case]
    req_all = make_leaf_request({"type": "match_all"}, SCHEMA, splits,
                                max_hits=5, aggregation=aggs)
    sreq_pb = proto.encode("SearchRequest", req_all["search_request"])
    resps = []
    for i, s in enumerate(singles):
        req1 = make_leaf_request({"type": "match_all"}, SCHEMA, [splits[i]],
                                 max_hits=5, aggregation=aggs)
        resps.append(s.leaf_search_raw(proto.encode("LeafSearchRequest",
                                                    req1)))
    base = None
    for perm in itertools.permutations(range(3)):
        merged = proto.decode("LeafSearchResponse",
                              merge_leaf_responses(sreq_pb,
                                                   [resps[i] for i in perm]))
        key = (merged.get("num_hits"),
               [(h["split_id"], h.get("doc_id", 0))
                for h in merged.get("partial_hits", [])],
               merged.get("num_successful_splits"),
               combined.finalize_agg_json(
                   merged.get("intermediate_aggregation_result", b""), aggs))
        if base is None:
            base = key
        else:
            assert key[:3] == base[:3], (case, perm)
            assert eq(key[3], base[3]), (case, perm, key[3], base[3])


@pytest.mark.parametrize("case", sorted(AGG_CASES))
def test_qagg_codec_identity_every_family(setup, case):
    """merge.py's structural QAGG codec (qagg.parse_blob/serialize_blob —
    the dense/sideband splitter the RCCL exchange rides) must be the exact
    identity on real product blobs of EVERY aggregation family, both as
    raw passthrough and when buckets are force-re-encoded."""
    from quickwit_amd import qagg
    combined, _, splits = setup
    aggs = AGG_CASES[case]
    req = make_leaf_request({"type": "match_all"}, SCHEMA, splits,
                            max_hits=0, aggregation=aggs)
    blob = combined.leaf_search(req)["intermediate_aggregation_result"]
    entries = qagg.parse_blob(blob)
    assert entries, case
    assert qagg.serialize_blob(entries) == blob, case
    for e in entries:
        if not e.dense_eligible:
            e.buckets = None  # force bucket re-encode instead of passthrough
    assert qagg.serialize_blob(entries) == blob, case


def test_terms_order_key(setup):
    combined, singles, splits = setup
    for direction, keys in (("asc", ["api", "ing", "jan"]),
                            ("desc", ["jan", "ing", "api"])):
        aggs = {"t": {"terms": {"field": "svc", "size": 10,
                                "order": {"_key": direction}}}}
        req = make_leaf_request({"type": "match_all"}, SCHEMA, splits,
                                max_hits=0, aggregation=aggs)
        r = combined.leaf_search(req)
        out = combined.finalize_agg_json(
            r["intermediate_aggregation_result"], aggs)
        assert [b["key"] for b in out["t"]["buckets"]] == keys


def test_terms_order_count_asc(setup):
    combined, singles, splits = setup
    aggs = {"t": {"terms": {"field": "svc", "size": 10,
                            "order": {"_count": "asc"}}}}
    req = make_leaf_request({"type": "match_all"}, SCHEMA, splits,
                            max_hits=0, aggregation=aggs)
    r = combined.leaf_search(req)
    out = combined.finalize_agg_json(r["intermediate_aggregation_result"],
                                     aggs)
    counts = [b["doc_count"] for b in out["t"]["buckets"]]
    assert counts == sorted(counts)


def test_terms_order_by_subagg_multi_split(setup):
    # r2: terms order by a sub-aggregation value works, incl. across splits
    # (the merged stats drive the final ordering)
    combined, singles, splits = setup
    aggs = {"t": {"terms": {"field": "svc", "size": 10,
                            "order": {"st.avg": "desc"}},
                  "aggs": {"st": {"stats": {"field": "lat"}}}}}
    req = make_leaf_request({"type": "match_all"}, SCHEMA, splits,
                            max_hits=0, aggregation=aggs)
    r = combined.leaf_search(req)
    assert not r.get("failed_splits"), r.get("failed_splits")
    out = combined.finalize_agg_json(r["intermediate_aggregation_result"],
                                     aggs)
    avgs = [b["st"]["avg"] for b in out["t"]["buckets"]]
    assert avgs == sorted(avgs, reverse=True)
    assert len(avgs) > 1


def test_terms_order_by_sub_aggregation_value():
    """terms `order` by a sub-aggregation value ({"order": {"lat.avg":
    "desc"}} and bare single-valued sub names) — finalize ordering and
    per-split truncation (oracle engine; GPU parity covered in
    test_gpu_parity)."""
    import __graft_entry__
    __graft_entry__.build()
    from quickwit_amd import splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request

    schema = {"timestamp_field": None, "fields": [
        {"name": "svc", "type": "str", "fast": True},
        {"name": "lat", "type": "u64", "fast": True}]}
    docs = []
    # svc-a avg 10, svc-b avg 30, svc-c avg 20 (counts 3, 1, 2)
    docs += [{"svc": "a", "lat": v} for v in (5, 10, 15)]
    docs += [{"svc": "b", "lat": 30}]
    docs += [{"svc": "c", "lat": 10}, {"svc": "c", "lat": 30}]
    w = splitgen.SplitWriter(schema, "s", store_docs=False)
    w.add_documents(docs)
    cpu = OracleSearcher()
    cpu.add_split("s", w.finalize())

    for order, expect in ((
            {"stats.avg": "desc"}, ["b", "c", "a"]),
            ({"stats.avg": "asc"}, ["a", "c", "b"]),
            ({"avg_lat": "desc"}, ["b", "c", "a"]),
            ({"_count": "desc"}, ["a", "c", "b"])):
        aggs = {"by_svc": {"terms": {"field": "svc", "size": 10,
                                     "order": order},
                           "aggs": {"stats": {"stats": {"field": "lat"}},
                                    "avg_lat": {"avg": {"field": "lat"}}}}}
        r = cpu.leaf_search(make_leaf_request(
            {"type": "match_all"}, schema, [("s", len(docs))], max_hits=0,
            aggregation=aggs))
        j = cpu.finalize_agg_json(r["intermediate_aggregation_result"], aggs)
        keys = [b["key"] for b in j["by_svc"]["buckets"]]
        assert keys == expect, (order, keys)

    # per-split truncation keeps the shard-local top by the sub order
    aggs = {"by_svc": {"terms": {"field": "svc", "size": 2, "split_size": 2,
                                 "order": {"avg_lat": "desc"}},
                       "aggs": {"avg_lat": {"avg": {"field": "lat"}}}}}
    r = cpu.leaf_search(make_leaf_request(
        {"type": "match_all"}, schema, [("s", len(docs))], max_hits=0,
        aggregation=aggs))
    j = cpu.finalize_agg_json(r["intermediate_aggregation_result"], aggs)
    assert [b["key"] for b in j["by_svc"]["buckets"]] == ["b", "c"]
