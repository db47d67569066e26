"""GPU vs oracle parity (the judged gate — DESIGN.md §4).

Every test runs the same LeafSearchRequest through the product HIP path
(libquickwit_amd.so on cuda:0) and the CPU oracle, and compares at the
response level: num_hits / doc ids / bucket counts bit-exact, BM25 scores
within 1e-5 relative (the tolerance BASELINE.json's north star states —
f32 accumulation order differs between LDS atomics and the scalar oracle).
Hit ORDER must agree exactly wherever scores are not within-tolerance ties;
within a score tie group the ids must be the same set (SURVEY §7 hard-part a).
"""
import json
import math
import os

import pytest

from quickwit_amd import splitgen
from quickwit_amd.api import GpuSearcher, OracleSearcher, make_leaf_request

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
NDOCS = 100_000
SID = "synthetic-42-0000"
SCHEMA = splitgen.HDFS_SCHEMA
REL = 1e-5


@pytest.fixture(scope="module", autouse=True)
def build_all():
    import __graft_entry__
    __graft_entry__.build()


@pytest.fixture(scope="module")
def searchers():
    data = splitgen.generate_split(0, NDOCS, seed=42)
    gpu = GpuSearcher(device=0)
    cpu = OracleSearcher()
    gpu.add_split(SID, data)
    cpu.add_split(SID, data)
    return gpu, cpu


def run_both(searchers, query, **kw):
    gpu, cpu = searchers
    req = make_leaf_request(query, SCHEMA, kw.pop("splits", [(SID, NDOCS)]), **kw)
    return gpu.leaf_search(req), cpu.leaf_search(req)


def hid(h):
    # proto3 omits zero-valued scalars on the wire (doc_id 0, num_hits 0)
    return (h.get("split_id", ""), h.get("doc_id", 0))


def hscore(h):
    return h.get("sort_value", {}).get("f64", 0.0)


def assert_hits_equal(got, exp, scored):
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    g, e = got.get("partial_hits", []), exp.get("partial_hits", [])
    assert len(g) == len(e)
    if not scored:
        assert [hid(h) for h in g] == [hid(h) for h in e]
        return
    for gh, eh in zip(g, e):
        gs, es = hscore(gh), hscore(eh)
        assert math.isclose(gs, es, rel_tol=REL, abs_tol=1e-9), (gs, es)
    # id-exact within score groups: group expected ids by rounded score
    def groups(hits):
        out = {}
        for h in hits:
            out.setdefault(round(hscore(h), 4), set()).add(hid(h))
        return out
    # boundary group of the truncated top-K may legitimately differ -> compare
    # the interior strictly and the boundary as subset-compatible
    ge, ee = groups(g), groups(e)
    for key in set(ge) & set(ee):
        if ge[key] != ee[key]:
            boundary = min(ge)  # worst (lowest) score group can be clipped
            assert key == boundary, (key, ge[key], ee[key])


# ------------------------------------------------------------ term / boolean
def test_term_query_raw_tokenizer(searchers):
    got, exp = run_both(searchers,
                        {"type": "term", "field": "severity_text", "value": "ERROR"},
                        max_hits=50)
    assert_hits_equal(got, exp, scored=False)


def test_three_term_or_bm25(searchers):
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i} for i in range(3)]}
    got, exp = run_both(searchers, q, max_hits=100,
                        sort_fields=[{"field_name": "_score", "sort_order": 1}])
    assert_hits_equal(got, exp, scored=True)


def test_three_term_or_count_only(searchers):
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i} for i in range(3)]}
    got, exp = run_both(searchers, q, max_hits=0)
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)


def test_full_text_and_scored(searchers):
    q = {"type": "full_text", "field": "body", "text": "w00000 w00001",
         "params": {"mode": {"type": "bool", "operator": "and"}}}
    got, exp = run_both(searchers, q, max_hits=50,
                        sort_fields=[{"field_name": "_score", "sort_order": 1}])
    assert_hits_equal(got, exp, scored=True)


def test_bool_must_term_plus_range_filter(searchers):
    q = {"type": "bool",
         "must": [{"type": "term", "field": "severity_text", "value": "INFO"}],
         "filter": [{"type": "range", "field": "tenant_id",
                     "lower_bound": {"included": 10}, "upper_bound": {"included": 50}}]}
    got, exp = run_both(searchers, q, max_hits=200)
    assert_hits_equal(got, exp, scored=False)


def test_bool_must_not(searchers):
    q = {"type": "bool",
         "should": [{"type": "term", "field": "severity_text", "value": "WARN"},
                    {"type": "term", "field": "severity_text", "value": "ERROR"}],
         "must_not": [{"type": "term", "field": "body", "value": "w00000"}]}
    got, exp = run_both(searchers, q, max_hits=100)
    assert_hits_equal(got, exp, scored=False)


def test_minimum_should_match_2(searchers):
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i} for i in range(4)],
        "minimum_should_match": 2}
    got, exp = run_both(searchers, q, max_hits=100)
    assert_hits_equal(got, exp, scored=False)


def test_absent_term(searchers):
    got, exp = run_both(searchers,
                        {"type": "term", "field": "body", "value": "zzznope"},
                        max_hits=10)
    assert got.get("num_hits", 0) == 0 and exp.get("num_hits", 0) == 0
    assert got.get("partial_hits", []) == []


# ------------------------------------------------------------ ranges / preds
def test_range_only_u64(searchers):
    q = {"type": "range", "field": "tenant_id",
         "lower_bound": {"included": 100}, "upper_bound": {"excluded": 200}}
    got, exp = run_both(searchers, q, max_hits=150)
    assert_hits_equal(got, exp, scored=False)


def test_range_datetime_rfc3339(searchers):
    q = {"type": "range", "field": "timestamp",
         "lower_bound": {"included": "2023-11-15T00:00:00Z"},
         "upper_bound": {"excluded": "2023-11-16T00:00:00Z"}}
    got, exp = run_both(searchers, q, max_hits=100)
    assert_hits_equal(got, exp, scored=False)


def test_timestamp_pruning_window(searchers):
    q = {"type": "term", "field": "severity_text", "value": "INFO"}
    got, exp = run_both(searchers, q, max_hits=100,
                        start_timestamp=splitgen.T0_EPOCH_S + 5 * 86400,
                        end_timestamp=splitgen.T0_EPOCH_S + 6 * 86400)
    assert_hits_equal(got, exp, scored=False)


def test_match_all(searchers):
    got, exp = run_both(searchers, {"type": "match_all"}, max_hits=25)
    assert_hits_equal(got, exp, scored=False)
    assert got.get("num_hits") == NDOCS


def test_start_offset(searchers):
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i} for i in range(3)]}
    got, exp = run_both(searchers, q, max_hits=20, start_offset=30,
                        sort_fields=[{"field_name": "_score", "sort_order": 1}])
    assert_hits_equal(got, exp, scored=True)


def test_asc_score_sort(searchers):
    q = {"type": "term", "field": "body", "value": "w00005"}
    got, exp = run_both(searchers, q, max_hits=40,
                        sort_fields=[{"field_name": "_score", "sort_order": 0}])
    assert_hits_equal(got, exp, scored=True)


# ------------------------------------------------------------ aggregations
AGGS = {
    "per_day": {"date_histogram": {"field": "timestamp",
                                   "fixed_interval": "86400000ms"},
                "aggs": {"tenant_stats": {"stats": {"field": "tenant_id"}}}},
    "per_tenant": {"terms": {"field": "tenant_name", "size": 10}},
}


def agg_json(searcher, resp):
    return searcher.finalize_agg_json(resp["intermediate_aggregation_result"], AGGS)


def approx_json(got, exp, path=""):
    if isinstance(exp, dict):
        assert set(got) == set(exp), (path, got, exp)
        for k in exp:
            approx_json(got[k], exp[k], f"{path}.{k}")
    elif isinstance(exp, list):
        assert len(got) == len(exp), (path, got, exp)
        for i, (g, e) in enumerate(zip(got, exp)):
            approx_json(g, e, f"{path}[{i}]")
    elif isinstance(exp, (int, float)) and not isinstance(exp, bool):
        assert math.isclose(float(got), float(exp), rel_tol=1e-9, abs_tol=1e-9), \
            (path, got, exp)
    else:
        assert got == exp, (path, got, exp)


def test_aggregations_under_match_all(searchers):
    gpu, cpu = searchers
    got, exp = run_both(searchers, {"type": "match_all"}, max_hits=0,
                        aggregation=AGGS)
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    approx_json(agg_json(gpu, got), agg_json(cpu, exp))


def test_aggregations_under_term_query(searchers):
    gpu, cpu = searchers
    q = {"type": "term", "field": "severity_text", "value": "ERROR"}
    got, exp = run_both(searchers, q, max_hits=0, aggregation=AGGS)
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    approx_json(agg_json(gpu, got), agg_json(cpu, exp))


# ------------------------------------------------------------ golden vectors
def test_bm25_golden_scores_on_gpu():
    """The reference's exact BM25 scores (tests.rs:600-691) through the HIP
    kernels — same golden suite the oracle is pinned by."""
    with open(os.path.join(REPO, "tests", "golden", "bm25_sort.json")) as f:
        g = json.load(f)
    schema = {"timestamp_field": None, "fields": [
        {"name": f["name"], "type": "text", "tokenizer": "default",
         "record": f["record"], "fieldnorms": f["fieldnorms"]}
        for f in g["schema"]]}
    w = splitgen.SplitWriter(schema, "split-bm25")
    w.add_documents(g["docs"])
    data = w.finalize()
    s = GpuSearcher(device=0)
    s.add_split("split-bm25", data)
    for case in g["cases"]:
        req = make_leaf_request(
            case["query_ast"], schema, [("split-bm25", len(g["docs"]))],
            max_hits=1000, sort_fields=[{"field_name": "_score", "sort_order": 1}])
        resp = s.leaf_search(req)
        got = [(hscore(h), h.get("doc_id", 0)) for h in resp["partial_hits"]]
        exp = case["expected"]
        assert len(got) == len(exp), (case["name"], got, exp)
        for (gs, gd), (es, ed) in zip(got, exp):
            assert gd == ed, (case["name"], got, exp)
            assert math.isclose(gs, es, rel_tol=1e-5), (case["name"], gs, es)


# ------------------------------------------------------------ multi-split
def test_two_splits_one_call():
    datas = [splitgen.generate_split(i, 30_000, seed=9) for i in range(2)]
    sids = [f"synthetic-9-{i:04d}" for i in range(2)]
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    for i in range(2):
        gpu.add_split(sids[i], datas[i])
        cpu.add_split(sids[i], datas[i])
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i} for i in range(3)]}
    req = make_leaf_request(q, SCHEMA, [(s, 30_000) for s in sids], max_hits=30,
                            sort_fields=[{"field_name": "_score", "sort_order": 1}])
    got, exp = gpu.leaf_search(req), cpu.leaf_search(req)
    assert_hits_equal(got, exp, scored=True)
    assert got.get("num_successful_splits") == 2


# ------------------------------------------------------------ instrumentation
def test_kernel_stats_populated(searchers):
    gpu, _ = searchers
    gpu.kernel_stats_reset()
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i} for i in range(3)]}
    req = make_leaf_request(q, SCHEMA, [(SID, NDOCS)], max_hits=10,
                            sort_fields=[{"field_name": "_score", "sort_order": 1}])
    gpu.leaf_search(req)
    ms, n = gpu.kernel_stats("union_bm25")
    assert n == 1 and ms > 0


# ------------------------------------------------- golden agg scenarios (GPU)
def _es_query_to_ast(q):
    if "match_all" in q:
        return {"type": "match_all"}
    if "bool" in q:
        out = {"type": "bool"}
        for clause in ("must", "must_not", "should", "filter"):
            if clause in q["bool"]:
                items = q["bool"][clause]
                if isinstance(items, dict):
                    items = [items]
                out[clause] = [_es_query_to_ast(i) for i in items]
        return out
    if "exists" in q:
        return {"type": "field_presence", "field": q["exists"]["field"]}
    raise ValueError(f"unsupported es query: {q}")


@pytest.mark.parametrize("case_name", [
    "date_histogram_basic",
    "date_histogram_extended_bounds",
    "date_histogram_stats_subagg",
    "date_histogram_stats_subagg_exists_filter",
    "terms_full",
    "histogram_interval50",
])
def test_agg_golden_scenarios_on_gpu(case_name):
    """The reference's rest-api-tests aggregation scenarios (exact expected
    JSON) through the HIP kernels — same suite that pins the oracle."""
    with open(os.path.join(REPO, "tests", "golden", "aggregations.json")) as f:
        g = json.load(f)
    type_map = {"str_fast": "str"}
    schema = {"timestamp_field": None, "fields": [
        dict(f, type=type_map.get(f["type"], f["type"]), fast=True)
        for f in g["schema"]]}
    s = GpuSearcher(device=0)
    splits = []
    for i, docs in enumerate(g["splits"]):
        w = splitgen.SplitWriter(schema, f"agg-split-{i}")
        w.add_documents(docs)
        s.add_split(f"agg-split-{i}", w.finalize())
        splits.append((f"agg-split-{i}", len(docs)))
    case = g["cases"][case_name]
    aggs = case["request"]["aggs"]
    ast = _es_query_to_ast(case["request"]["query"])
    req = make_leaf_request(ast, schema, splits, max_hits=0, aggregation=aggs)
    resp = s.leaf_search(req)
    got = s.finalize_agg_json(resp["intermediate_aggregation_result"], aggs)
    approx_json(got, case["expected"], case_name)


# ------------------------------------------------- sort by fast field (wide
# candidate records; collector.rs:403-414 sort-key extraction + sorting.md
# tie-breaks). Integer keys are deterministic => order must match EXACTLY.
def sv_of(h):
    v = h.get("sort_value", {})
    return (v.get("u64"), v.get("i64"), v.get("f64"), v.get("boolean"))


def assert_exact(got, exp):
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    g, e = got.get("partial_hits", []), exp.get("partial_hits", [])
    assert [(hid(h), sv_of(h)) for h in g] == [(hid(h), sv_of(h)) for h in e]


TERM_Q = {"type": "term", "field": "severity_text", "value": "WARN"}


@pytest.mark.parametrize("order", [0, 1])
def test_sort_by_timestamp(searchers, order):
    got, exp = run_both(searchers, TERM_Q, max_hits=25, sort_fields=[
        {"field_name": "timestamp", "sort_order": order}])
    assert_exact(got, exp)


@pytest.mark.parametrize("order", [0, 1])
def test_sort_by_timestamp_match_all(searchers, order):
    got, exp = run_both(searchers, {"type": "match_all"}, max_hits=15,
                        sort_fields=[{"field_name": "timestamp",
                                      "sort_order": order}])
    assert_exact(got, exp)


def test_sort_by_u64_fast_field(searchers):
    q = {"type": "bool",
         "must": [{"type": "term", "field": "severity_text", "value": "ERROR"}]}
    got, exp = run_both(searchers, q, max_hits=30, sort_fields=[
        {"field_name": "tenant_id", "sort_order": 1}])
    assert_exact(got, exp)


def test_sort_by_str_ord(searchers):
    got, exp = run_both(searchers, TERM_Q, max_hits=20, sort_fields=[
        {"field_name": "tenant_name", "sort_order": 0}])
    assert_exact(got, exp)


def test_sort_by_unknown_field_is_none(searchers):
    # unknown sort field: every key None -> GlobalDocId tie-break only
    got, exp = run_both(searchers, TERM_Q, max_hits=10, sort_fields=[
        {"field_name": "no_such_field", "sort_order": 1}])
    assert_exact(got, exp)


def test_sort_two_fields_ts_then_tenant(searchers):
    # timestamps collide at second granularity -> sort_value2 + doc tie-breaks
    got, exp = run_both(searchers, {"type": "match_all"}, max_hits=40,
                        sort_fields=[
                            {"field_name": "timestamp", "sort_order": 1},
                            {"field_name": "tenant_id", "sort_order": 0}])
    assert_exact(got, exp)


def test_sort_field_then_score(searchers):
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i}
        for i in range(3)]}
    got, exp = run_both(searchers, q, max_hits=20, sort_fields=[
        {"field_name": "tenant_name", "sort_order": 1},
        {"field_name": "_score", "sort_order": 1}])
    # primary keys integer-exact; secondary f32 scores may tie within REL
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    g, e = got.get("partial_hits", []), exp.get("partial_hits", [])
    assert [hid(h) for h in g] == [hid(h) for h in e]
    assert [sv_of(h) for h in g] == [sv_of(h) for h in e]


def test_sort_score_then_timestamp(searchers):
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i}
        for i in range(3)]}
    got, exp = run_both(searchers, q, max_hits=20, sort_fields=[
        {"field_name": "_score", "sort_order": 1},
        {"field_name": "timestamp", "sort_order": 1}])
    g, e = got.get("partial_hits", []), exp.get("partial_hits", [])
    assert len(g) == len(e)
    for gh, eh in zip(g, e):
        gs, es = hscore(gh), hscore(eh)
        assert math.isclose(gs, es, rel_tol=REL, abs_tol=1e-9), (gs, es)


# ------------------------------------------------- multi-split + pruning
# (CanSplitDoBetter, leaf.rs:1337-1553). Pruning must be invisible in the
# response: the oracle searches every split un-pruned; identical results.
@pytest.fixture(scope="module")
def multi():
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    metas = []
    for i in range(3):
        nd = 4000 + 700 * i
        data = splitgen.generate_split(i, nd, seed=7)
        sid = f"synthetic-7-{i:04d}"
        gpu.add_split(sid, data)
        cpu.add_split(sid, data)
        # split time ranges from the generated data (seconds, inclusive)
        from quickwit_amd import splitread
        sp = splitread.Split(data)
        ts, _ = sp.fast_column("timestamp")
        metas.append({"split_id": sid, "num_docs": nd,
                      "timestamp_start": int(ts.min() // 1000),
                      "timestamp_end": int(ts.max() // 1000)})

    def run(query, **kw):
        req = make_leaf_request(query, SCHEMA, metas, **kw)
        return gpu.leaf_search(req), cpu.leaf_search(req)
    return run


def test_multi_split_default_order(multi):
    # no sort fields -> SplitIdHigher: later (lower-id) splits demoted to
    # count-only once the top-K is full; hits + num_hits must be unchanged
    got, exp = multi({"type": "term", "field": "severity_text",
                      "value": "INFO"}, max_hits=7)
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    assert [hid(h) for h in got.get("partial_hits", [])] == \
           [hid(h) for h in exp.get("partial_hits", [])]


def test_multi_split_match_all_upfront_demotion(multi):
    # simple-all query -> optimize() demotes all but min_required_splits
    got, exp = multi({"type": "match_all"}, max_hits=5)
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    assert [hid(h) for h in got.get("partial_hits", [])] == \
           [hid(h) for h in exp.get("partial_hits", [])]


@pytest.mark.parametrize("order", [0, 1])
def test_multi_split_sort_timestamp_pruning(multi, order):
    # SplitTimestampHigher/Lower with second-granularity worst-hit feedback
    got, exp = multi({"type": "term", "field": "severity_text",
                      "value": "ERROR"}, max_hits=9, sort_fields=[
        {"field_name": "timestamp", "sort_order": order}])
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    g = [(hid(h), sv_of(h)) for h in got.get("partial_hits", [])]
    e = [(hid(h), sv_of(h)) for h in exp.get("partial_hits", [])]
    assert g == e


def test_multi_split_ts_range_rewrite(multi):
    # request-level ts bounds covered by a split's range are dropped per
    # split (remove_redundant_timestamp_range); results must be identical
    got, exp = multi({"type": "term", "field": "severity_text",
                      "value": "WARN"}, max_hits=10,
                     start_timestamp=1, end_timestamp=2**31)
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    assert [hid(h) for h in got.get("partial_hits", [])] == \
           [hid(h) for h in exp.get("partial_hits", [])]


# ------------------------------------------------- leaf search result cache
# (LeafSearchCache, leaf_cache.rs: per-(split, canonical request) memoization)
def test_leaf_cache_hit_identical_and_stats():
    data = splitgen.generate_split(0, 20_000, seed=42)
    gpu = GpuSearcher(device=0,
                      config={"partial_request_cache_capacity": 1 << 20})
    gpu.add_split(SID, data)
    meta = {"split_id": SID, "num_docs": 20_000,
            "timestamp_start": 1, "timestamp_end": 2**31}
    req = make_leaf_request(TERM_Q, SCHEMA, [meta], max_hits=10)
    r1 = gpu.leaf_search(req)
    r2 = gpu.leaf_search(req)
    assert r1.get("num_hits") == r2.get("num_hits")
    assert [hid(h) for h in r1.get("partial_hits", [])] == \
           [hid(h) for h in r2.get("partial_hits", [])]
    st = r2["resource_stats"]
    assert st.get("partial_result_cache_num_splits", 0) == 1
    assert st.get("partial_result_cache_num_docs", 0) == 20_000
    # request-level ts bounds are canonicalized into the merged time range:
    # bounds wider than the split's range hit the same entry
    r3 = gpu.leaf_search(make_leaf_request(
        TERM_Q, SCHEMA, [meta],
        max_hits=10, start_timestamp=0, end_timestamp=2**32))
    assert r3["resource_stats"].get("partial_result_cache_num_splits", 0) == 1
    # a different query misses
    r4 = gpu.leaf_search(make_leaf_request(
        {"type": "term", "field": "severity_text", "value": "ERROR"},
        SCHEMA, [(SID, 20_000)], max_hits=10))
    assert r4["resource_stats"].get("partial_result_cache_num_splits", 0) == 0


def test_leaf_cache_disabled_by_default():
    data = splitgen.generate_split(0, 20_000, seed=42)
    gpu = GpuSearcher(device=0)
    gpu.add_split(SID, data)
    req = make_leaf_request(TERM_Q, SCHEMA, [(SID, 20_000)], max_hits=10)
    gpu.leaf_search(req)
    r2 = gpu.leaf_search(req)
    assert r2["resource_stats"].get("partial_result_cache_num_splits", 0) == 0


# ------------------------------------------------- term_set / wildcard / cache
# (query_ast/term_set_query.rs, wildcard_query.rs, cache_node.rs)
def test_term_set_union(searchers):
    q = {"type": "term_set", "terms_per_field": {
        "severity_text": ["WARN", "FATAL"],
        "body": ["w00003"]}}
    got, exp = run_both(searchers, q, max_hits=20)
    assert_hits_equal(got, exp, scored=False)


def test_term_set_inside_filter(searchers):
    q = {"type": "bool",
         "must": [{"type": "term", "field": "severity_text", "value": "INFO"}],
         "filter": [{"type": "term_set",
                     "terms_per_field": {"body": ["w00001", "w00002"]}}]}
    got, exp = run_both(searchers, q, max_hits=20)
    assert_hits_equal(got, exp, scored=False)


def test_cache_node_is_transparent(searchers):
    inner = {"type": "term", "field": "severity_text", "value": "ERROR"}
    got, exp = run_both(searchers, {"type": "cache", "inner": inner},
                        max_hits=15)
    got2, _ = run_both(searchers, inner, max_hits=15)
    assert_hits_equal(got, exp, scored=False)
    assert [hid(h) for h in got.get("partial_hits", [])] == \
           [hid(h) for h in got2.get("partial_hits", [])]


def test_wildcard_should(searchers):
    q = {"type": "wildcard", "field": "body", "value": "w0000*"}
    got, exp = run_both(searchers, q, max_hits=25)
    assert_hits_equal(got, exp, scored=False)


def test_wildcard_question_mark(searchers):
    q = {"type": "wildcard", "field": "body", "value": "w?0001"}
    got, exp = run_both(searchers, q, max_hits=25)
    assert_hits_equal(got, exp, scored=False)


def test_wildcard_under_must_with_range(searchers):
    q = {"type": "bool",
         "must": [{"type": "wildcard", "field": "severity_text",
                   "value": "ERR*"}],
         "filter": [{"type": "range", "field": "tenant_id",
                     "lower_bound": {"included": 0},
                     "upper_bound": {"excluded": 500}}]}
    got, exp = run_both(searchers, q, max_hits=20)
    assert_hits_equal(got, exp, scored=False)


def test_wildcard_must_not(searchers):
    q = {"type": "bool",
         "must": [{"type": "term", "field": "severity_text", "value": "WARN"}],
         "must_not": [{"type": "wildcard", "field": "body", "value": "w0001*"}]}
    got, exp = run_both(searchers, q, max_hits=20)
    assert_hits_equal(got, exp, scored=False)


def test_or_of_terms_under_must(searchers):
    # OR-of-terms as a must clause -> one must-GROUP on the GPU
    q = {"type": "bool",
         "must": [{"type": "bool", "should": [
             {"type": "term", "field": "severity_text", "value": "WARN"},
             {"type": "term", "field": "severity_text", "value": "ERROR"}]},
             {"type": "bool", "should": [
                 {"type": "term", "field": "body", "value": "w00000"},
                 {"type": "term", "field": "body", "value": "w00009"}]}]}
    got, exp = run_both(searchers, q, max_hits=30)
    assert_hits_equal(got, exp, scored=False)


def test_wildcard_scored_rejected(searchers):
    gpu, cpu = searchers
    req = make_leaf_request({"type": "wildcard", "field": "body",
                             "value": "w*"}, SCHEMA, [(SID, NDOCS)],
                            max_hits=5,
                            sort_fields=[{"field_name": "_score",
                                          "sort_order": 1}])
    for s in (gpu, cpu):
        resp = s.leaf_search(req)
        fs = resp.get("failed_splits", [])
        assert fs and "const-score" in fs[0]["error"]


# ------------------------------------------------- leaf_list_terms
# (list_terms.rs:211-322: per-split dict range scan, k-merge + dedup).
# Checked against an independent Python walk of the same dictionaries.
def test_leaf_list_terms(searchers):
    gpu, _ = searchers
    from quickwit_amd import splitread
    data = splitgen.generate_split(0, NDOCS, seed=42)
    sp = splitread.Split(data)
    vocab = sp.terms("body")

    def expect(start=None, end=None, limit=None):
        terms = [t for t in vocab
                 if (start is None or t >= start) and (end is None or t < end)]
        return terms[:limit] if limit is not None else terms

    def run(**kw):
        ltr = {"index_id_patterns": ["bench-index"], "field": "body"}
        for k in ("start_key", "end_key"):
            if k in kw and kw[k] is not None:
                ltr[k] = kw[k].encode()
        if kw.get("max_hits") is not None:
            ltr["max_hits"] = kw["max_hits"]
        resp = gpu.leaf_list_terms({
            "list_terms_request": ltr,
            "split_offsets": [{"split_id": SID, "num_docs": NDOCS}]})
        return [t.decode() for t in resp.get("terms", [])]

    assert run(max_hits=50) == [t for t in expect(limit=50)]
    assert run(start_key="w00100", end_key="w00200", max_hits=1000) == \
        expect(start="w00100", end="w00200", limit=1000)
    got_all = run()
    assert got_all == expect()


def test_leaf_list_terms_missing_field(searchers):
    gpu, _ = searchers
    resp = gpu.leaf_list_terms({
        "list_terms_request": {"index_id_patterns": ["x"], "field": "nope"},
        "split_offsets": [{"split_id": SID, "num_docs": NDOCS}]})
    fs = resp.get("failed_splits", [])
    assert fs and "couldn't get field" in fs[0]["error"]
    assert resp.get("num_attempted_splits", 0) == 0


# ------------------------------------------------- fetch_docs phase 2
# (fetch_docs.rs; root.rs:903): search -> fetch the top-K docs' stored JSON
def test_fetch_docs_roundtrip():
    import random
    rng = random.Random(3)
    docs = []
    for i in range(3000):
        docs.append({
            "timestamp": 1700000000 + rng.randrange(100000),
            "tenant_id": rng.randrange(50),
            "severity_text": rng.choice(["INFO", "ERROR", "WARN"]),
            "body": " ".join("tok%03d" % rng.randrange(200)
                             for _ in range(rng.randrange(3, 20)))})
    w = splitgen.SplitWriter(SCHEMA, "fd-split")
    w.add_documents(docs)
    gpu = GpuSearcher(device=0)
    gpu.add_split("fd-split", w.finalize())
    req = make_leaf_request({"type": "term", "field": "severity_text",
                             "value": "ERROR"}, SCHEMA,
                            [("fd-split", len(docs))], max_hits=25)
    resp = gpu.leaf_search(req)
    hits = resp["partial_hits"]
    assert hits
    fresp = gpu.fetch_docs({
        "partial_hits": hits,
        "split_offsets": [{"split_id": "fd-split", "num_docs": len(docs)}],
        "doc_mapper": json.dumps(SCHEMA)})
    got = fresp["hits"]
    assert len(got) == len(hits)
    for lh in got:
        doc_id = lh["partial_hit"].get("doc_id", 0)
        stored = json.loads(lh["leaf_json"])
        assert stored == docs[doc_id]
        assert stored["severity_text"] == "ERROR"


def test_fetch_docs_multi_split_and_errors():
    w1 = splitgen.SplitWriter(SCHEMA, "fd-a")
    w1.add_documents([{"timestamp": 1, "tenant_id": 1,
                       "severity_text": "INFO", "body": "aa"}] * 10)
    w2 = splitgen.SplitWriter(SCHEMA, "fd-b")
    w2.add_documents([{"timestamp": 2, "tenant_id": 2,
                       "severity_text": "WARN", "body": "bb"}] * 10)
    gpu = GpuSearcher(device=0)
    gpu.add_split("fd-a", w1.finalize())
    gpu.add_split("fd-b", w2.finalize())
    fresp = gpu.fetch_docs({
        "partial_hits": [{"split_id": "fd-b", "doc_id": 3},
                         {"split_id": "fd-a", "doc_id": 7},
                         {"split_id": "fd-a"}],  # doc_id 0 (proto3 default)
        "split_offsets": [{"split_id": "fd-a", "num_docs": 10},
                          {"split_id": "fd-b", "num_docs": 10}],
        "doc_mapper": json.dumps(SCHEMA)})
    hits = fresp["hits"]
    assert [(h["partial_hit"].get("split_id"), h["partial_hit"].get("doc_id", 0))
            for h in hits] == [("fd-a", 0), ("fd-a", 7), ("fd-b", 3)]
    assert json.loads(hits[2]["leaf_json"])["severity_text"] == "WARN"
    with pytest.raises(RuntimeError, match="unknown split"):
        gpu.fetch_docs({"partial_hits": [{"split_id": "nope", "doc_id": 0}],
                        "split_offsets": [], "doc_mapper": "{}"})


# ------------------------------------------------- search_after pagination
# (top_k_collector.rs:663-700 generic filter + SearchAfterSegment :821-872)
def paginate(searchers, query, sort_fields, page, npages, **kw):
    gpu, cpu = searchers
    g_pages, e_pages = [], []
    g_after = e_after = None
    for _ in range(npages):
        greq = make_leaf_request(query, SCHEMA, [(SID, NDOCS)], max_hits=page,
                                 sort_fields=sort_fields, **kw)
        ereq = json.loads(json.dumps(greq))
        if g_after is not None:
            greq["search_request"]["search_after"] = g_after
            ereq["search_request"]["search_after"] = e_after
        got = gpu.leaf_search(greq)
        exp = cpu.leaf_search(ereq)
        gh = got.get("partial_hits", [])
        eh = exp.get("partial_hits", [])
        g_pages.append(gh)
        e_pages.append(eh)
        if not gh or not eh:
            break
        g_after, e_after = gh[-1], eh[-1]
    return g_pages, e_pages


def test_search_after_timestamp_pages(searchers):
    g, e = paginate(searchers, TERM_Q,
                    [{"field_name": "timestamp", "sort_order": 1}], 20, 4)
    for gp, ep in zip(g, e):
        assert [(hid(h), sv_of(h)) for h in gp] == \
               [(hid(h), sv_of(h)) for h in ep]
    # pages must not overlap
    seen = [hid(h) for p in g for h in p]
    assert len(seen) == len(set(seen))


def test_search_after_score_pages(searchers):
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i}
        for i in range(3)]}
    g, e = paginate(searchers, q,
                    [{"field_name": "_score", "sort_order": 1}], 15, 3)
    for gp, ep in zip(g, e):
        assert [hid(h) for h in gp] == [hid(h) for h in ep]
        for gh, eh in zip(gp, ep):
            assert math.isclose(hscore(gh), hscore(eh), rel_tol=REL,
                                abs_tol=1e-9)
    seen = [hid(h) for p in g for h in p]
    assert len(seen) == len(set(seen))


def test_search_after_doc_order_pages(searchers):
    g, e = paginate(searchers, TERM_Q, None, 25, 3)
    for gp, ep in zip(g, e):
        assert [hid(h) for h in gp] == [hid(h) for h in ep]
    seen = [hid(h) for p in g for p_h in [p] for h in p_h]
    assert len(seen) == len(set(seen))


def test_search_after_value_only_cursor(searchers):
    # cursor without a doc address: equal sort values are EXCLUDED
    gpu, cpu = searchers
    req = make_leaf_request(TERM_Q, SCHEMA, [(SID, NDOCS)], max_hits=10,
                            sort_fields=[{"field_name": "tenant_id",
                                          "sort_order": 1}])
    first = gpu.leaf_search(req)["partial_hits"]
    cursor = {"sort_value": first[-1]["sort_value"]}  # no split/doc
    req2 = make_leaf_request(TERM_Q, SCHEMA, [(SID, NDOCS)], max_hits=10,
                             sort_fields=[{"field_name": "tenant_id",
                                           "sort_order": 1}])
    req2["search_request"]["search_after"] = cursor
    got = gpu.leaf_search(req2)
    exp = cpu.leaf_search(req2)
    assert [(hid(h), sv_of(h)) for h in got.get("partial_hits", [])] == \
           [(hid(h), sv_of(h)) for h in exp.get("partial_hits", [])]


# ------------------------------------------------- predicate cache (§8f.2)
# cache nodes in filter position become device-resident HitSet bitmaps
def test_cache_node_filter_bitmap_parity(searchers):
    gpu, cpu = searchers
    q = {"type": "bool",
         "must": [{"type": "term", "field": "severity_text", "value": "INFO"}],
         "filter": [{"type": "cache", "inner":
                     {"type": "range", "field": "tenant_id",
                      "lower_bound": {"included": 50},
                      "upper_bound": {"excluded": 400}}}]}
    got, exp = run_both(searchers, q, max_hits=20)
    assert_hits_equal(got, exp, scored=False)
    # second call reuses the bitmap: only the main kernel launches
    gpu.kernel_stats_reset()
    req = make_leaf_request(q, SCHEMA, [(SID, NDOCS)], max_hits=20)
    gpu.leaf_search(req)
    ms, n = gpu.kernel_stats("union_bm25")
    assert n == 1  # one main launch; no bitmap rebuild


def test_cache_node_root_and_nested(searchers):
    q = {"type": "cache", "inner":
         {"type": "bool",
          "must": [{"type": "term", "field": "severity_text",
                    "value": "WARN"}],
          "filter": [{"type": "cache", "inner":
                      {"type": "range", "field": "tenant_id",
                       "lower_bound": {"included": 0},
                       "upper_bound": {"excluded": 100}}}]}}
    got, exp = run_both(searchers, q, max_hits=15)
    assert_hits_equal(got, exp, scored=False)


def test_cache_node_scored_transparent(searchers):
    # scoring context: cache wrappers are transparent, BM25 unchanged
    q = {"type": "cache", "inner": {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i}
        for i in range(3)]}}
    got, exp = run_both(searchers, q, max_hits=10, sort_fields=[
        {"field_name": "_score", "sort_order": 1}])
    assert_hits_equal(got, exp, scored=True)


def test_must_not_only_bool(searchers):
    # bool with only must_not: implicit match_all base
    # (tantivy_query_ast.rs:310-322)
    q = {"type": "bool", "must_not": [
        {"type": "term", "field": "severity_text", "value": "INFO"}]}
    got, exp = run_both(searchers, q, max_hits=15)
    assert_hits_equal(got, exp, scored=False)


def test_f64_column_sort_range_agg():
    # dynamic-style f64 fast column: range pred, sort, histogram + stats
    docs = [{"timestamp": 1700000000 + i, "severity_text": "INFO",
             "body": "x", "tenant_id": i % 5} for i in range(500)]
    import random
    rng = random.Random(11)
    for d in docs:
        if rng.random() < 0.9:
            d["score_f"] = round(rng.uniform(-50, 50), 3)
    schema = {"timestamp_field": "timestamp", "fields":
              splitgen.HDFS_SCHEMA["fields"] +
              [{"name": "score_f", "type": "f64", "fast": True}]}
    w = splitgen.SplitWriter(schema, "f64s")
    w.add_documents(docs)
    data = w.finalize()
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    gpu.add_split("f64s", data)
    cpu.add_split("f64s", data)

    def both(q, **kw):
        req = make_leaf_request(q, schema, [("f64s", len(docs))], **kw)
        return gpu.leaf_search(req), cpu.leaf_search(req)

    rq = {"type": "range", "field": "score_f",
          "lower_bound": {"included": -10.5}, "upper_bound": {"excluded": 20.25}}
    got, exp = both(rq, max_hits=30)
    assert got.get("num_hits", 0) == exp.get("num_hits", 0)
    assert [hid(h) for h in got.get("partial_hits", [])] == \
           [hid(h) for h in exp.get("partial_hits", [])]

    for order in (0, 1):
        got, exp = both({"type": "match_all"}, max_hits=20, sort_fields=[
            {"field_name": "score_f", "sort_order": order}])
        assert [(hid(h), sv_of(h)) for h in got.get("partial_hits", [])] == \
               [(hid(h), sv_of(h)) for h in exp.get("partial_hits", [])]

    aggs = {"h": {"histogram": {"field": "score_f", "interval": 10.0},
                  "aggs": {"st": {"stats": {"field": "score_f"}}}}}
    req = make_leaf_request({"type": "match_all"}, schema,
                            [("f64s", len(docs))], max_hits=0, aggregation=aggs)
    g = gpu.leaf_search(req)
    e = cpu.leaf_search(req)
    gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
    ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
    for gb, eb in zip(gj["h"]["buckets"], ej["h"]["buckets"]):
        assert gb["doc_count"] == eb["doc_count"] and gb["key"] == eb["key"]
        for k in ("count", "min", "max"):
            assert gb["st"][k] == eb["st"][k], (k, gb, eb)
        assert math.isclose(gb["st"]["sum"], eb["st"]["sum"], rel_tol=1e-12)


# ------------------------------------------------- metric aggregations
def test_top_level_metric_aggs(searchers):
    gpu, cpu = searchers
    aggs = {"ts_stats": {"stats": {"field": "timestamp"}},
            "t_ext": {"extended_stats": {"field": "tenant_id"}},
            "t_avg": {"avg": {"field": "tenant_id"}},
            "t_cnt": {"value_count": {"field": "tenant_id"}}}
    req = make_leaf_request(TERM_Q, SCHEMA, [(SID, NDOCS)], max_hits=0,
                            aggregation=aggs)
    g = gpu.leaf_search(req)
    e = cpu.leaf_search(req)
    gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
    ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
    for name in aggs:
        for k, ev in ej[name].items():
            gv = gj[name][k]
            if isinstance(ev, dict):
                for k2 in ev:
                    assert math.isclose(gv[k2], ev[k2], rel_tol=1e-9), (name, k, k2)
            elif isinstance(ev, float):
                assert math.isclose(gv, ev, rel_tol=1e-9), (name, k, gv, ev)
            else:
                assert gv == ev, (name, k, gv, ev)


# ------------------------------------------------- nested boolean shapes
# (the recursive device-bitmap evaluator fallback — product.cpp bitmap_eval)
NESTED_SHAPES = [
    # must_not over a conjunction
    {"type": "bool", "must_not": [
        {"type": "bool", "must": [
            {"type": "term", "field": "severity_text", "value": "INFO"},
            {"type": "term", "field": "body", "value": "w00001"}]}]},
    # OR of ANDs
    {"type": "bool", "should": [
        {"type": "bool", "must": [
            {"type": "term", "field": "severity_text", "value": "ERROR"},
            {"type": "term", "field": "body", "value": "w00002"}]},
        {"type": "bool", "must": [
            {"type": "term", "field": "severity_text", "value": "WARN"},
            {"type": "term", "field": "body", "value": "w00003"}]}]},
    # presence and range in should position
    {"type": "bool", "should": [
        {"type": "field_presence", "field": "tenant_id"},
        {"type": "range", "field": "tenant_id",
         "lower_bound": {"included": 990}, "upper_bound": "Unbounded"}]},
    # deep nesting with must_not inside should inside must
    {"type": "bool", "must": [
        {"type": "bool", "should": [
            {"type": "bool", "must_not": [
                {"type": "term", "field": "severity_text", "value": "DEBUG"}]},
            {"type": "term", "field": "body", "value": "w00004"}]},
        {"type": "range", "field": "tenant_id",
         "lower_bound": {"included": 0}, "upper_bound": {"excluded": 800}}]},
    # nested bool under filter with its own must_not
    {"type": "bool",
     "must": [{"type": "term", "field": "severity_text", "value": "INFO"}],
     "filter": [{"type": "bool",
                 "should": [{"type": "term", "field": "body", "value": "w00005"},
                            {"type": "term", "field": "body", "value": "w00006"}],
                 "must_not": [{"type": "term", "field": "body",
                               "value": "w00007"}]}]},
]


@pytest.mark.parametrize("qi", range(len(NESTED_SHAPES)))
def test_nested_boolean_bitmap_fallback(searchers, qi):
    got, exp = run_both(searchers, NESTED_SHAPES[qi], max_hits=25)
    assert_hits_equal(got, exp, scored=False)


def test_nested_boolean_with_sort_and_agg(searchers):
    q = NESTED_SHAPES[1]
    got, exp = run_both(searchers, q, max_hits=20, sort_fields=[
        {"field_name": "timestamp", "sort_order": 1}])
    assert [(hid(h), sv_of(h)) for h in got.get("partial_hits", [])] == \
           [(hid(h), sv_of(h)) for h in exp.get("partial_hits", [])]
    gpu, cpu = searchers
    aggs = {"pt": {"terms": {"field": "tenant_name", "size": 5}}}
    req = make_leaf_request(q, SCHEMA, [(SID, NDOCS)], max_hits=0,
                            aggregation=aggs)
    g = gpu.leaf_search(req)
    e = cpu.leaf_search(req)
    gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
    ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
    assert gj == ej


def test_bitmap_fallback_memoized(searchers):
    # second identical nested query reuses the memoized subtree bitmaps:
    # only the main search kernel launches
    gpu, _ = searchers
    q = NESTED_SHAPES[0]
    req = make_leaf_request(q, SCHEMA, [(SID, NDOCS)], max_hits=5)
    gpu.leaf_search(req)
    gpu.kernel_stats_reset()
    gpu.leaf_search(req)
    ms, n = gpu.kernel_stats("union_bm25")
    ms2, n2 = gpu.kernel_stats("range_filter")
    assert n + n2 == 1, (n, n2)


# ------------------------------------------------- numeric terms aggregation
def test_numeric_terms_aggs():
    # terms keyed by numeric fast-column values (device hash table vs the
    # oracle's map; keys must keep full 64-bit precision — the
    # high_prec_test golden pins u64 values beyond 2^53)
    import random
    rng = random.Random(7)
    docs = []
    for i in range(3000):
        d = {"timestamp": 1700000000 + i, "severity_text":
             "INFO" if i % 3 else "ERROR", "body": "x",
             "tenant_id": i % 5}
        if rng.random() < 0.8:
            d["big_u"] = 1769070189829214201 + rng.randrange(6)
        d["neg_i"] = rng.choice([-40, -3, 0, 12])
        d["val_f"] = rng.choice([-2.5, 0.0, 0.125, 3e9])
        docs.append(d)
    docs[17]["big_u"] = 2**64 - 1  # the hash-table sentinel value itself
    schema = {"timestamp_field": "timestamp", "fields":
              splitgen.HDFS_SCHEMA["fields"] +
              [{"name": "big_u", "type": "u64", "fast": True},
               {"name": "neg_i", "type": "i64", "fast": True},
               {"name": "val_f", "type": "f64", "fast": True}]}
    w = splitgen.SplitWriter(schema, "numt")
    w.add_documents(docs)
    data = w.finalize()
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    gpu.add_split("numt", data)
    cpu.add_split("numt", data)

    aggs = {"by_u": {"terms": {"field": "big_u", "size": 10}},
            "by_i": {"terms": {"field": "neg_i", "size": 10}},
            "by_f": {"terms": {"field": "val_f", "size": 10}}}
    for q in ({"type": "match_all"},
              {"type": "term", "field": "severity_text", "value": "ERROR"}):
        req = make_leaf_request(q, schema, [("numt", len(docs))], max_hits=0,
                                aggregation=aggs)
        g = gpu.leaf_search(req)
        e = cpu.leaf_search(req)
        gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
        ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
        assert gj == ej, (q, gj, ej)
        keys = [b["key"] for b in gj["by_u"]["buckets"]]
        assert all(isinstance(k, int) for k in keys)  # full u64 precision
        if q["type"] == "match_all":
            assert 2**64 - 1 in keys
            assert sum(b["doc_count"] for b in gj["by_i"]["buckets"]) == \
                len(docs)


def test_numeric_terms_split_size_truncation():
    # split_size truncation + doc_count_error_upper_bound over numeric keys
    docs = [{"timestamp": 1700000000 + i, "severity_text": "INFO",
             "body": "x", "tenant_id": i % 7} for i in range(700)]
    schema = splitgen.HDFS_SCHEMA
    w = splitgen.SplitWriter(schema, "numtr")
    w.add_documents(docs)
    data = w.finalize()
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    gpu.add_split("numtr", data)
    cpu.add_split("numtr", data)
    aggs = {"t": {"terms": {"field": "tenant_id", "size": 3,
                            "split_size": 4}}}
    req = make_leaf_request({"type": "match_all"}, schema,
                            [("numtr", len(docs))], max_hits=0,
                            aggregation=aggs)
    g = gpu.leaf_search(req)
    e = cpu.leaf_search(req)
    gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
    ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
    assert gj == ej
    assert len(gj["t"]["buckets"]) == 3
    assert gj["t"]["doc_count_error_upper_bound"] > 0


# ------------------------------------------------- composite aggregation
def test_composite_agg_parity():
    # terms + histogram sources with missing_bucket and after-key pagination
    # (the aggregations scenario's host/name/response composite), two splits
    # so the canonical-key merge crosses split-local ord spaces
    import random
    rng = random.Random(3)
    hosts = [None, "10.0.0.1", "10.0.0.2", "10.0.0.3"]
    names = ["ann", "bob", "cat", "dan", "eve"]
    schema = {"timestamp_field": "timestamp", "fields":
              splitgen.HDFS_SCHEMA["fields"] +
              [{"name": "host", "type": "str", "fast": True},
               {"name": "user", "type": "str", "fast": True},
               {"name": "resp", "type": "i64", "fast": True}]}
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    splits = []
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
        data = w.finalize()
        gpu.add_split(f"comp-{s}", data)
        cpu.add_split(f"comp-{s}", data)
        splits.append((f"comp-{s}", len(docs)))

    def both(aggs):
        req = make_leaf_request({"type": "match_all"}, schema, splits,
                                max_hits=0, aggregation=aggs)
        g = gpu.leaf_search(req)
        e = cpu.leaf_search(req)
        gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
        ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
        assert gj == ej, (gj, ej)
        return gj

    base = {"sources": [
        {"h": {"terms": {"field": "host", "missing_bucket": True}}},
        {"u": {"terms": {"field": "user"}}},
        {"r": {"histogram": {"field": "resp", "interval": 50}}}]}
    aggs = {"c": {"composite": dict(base, size=7)}}
    page1 = both(aggs)["c"]
    assert len(page1["buckets"]) == 7
    assert page1["buckets"][0]["key"]["h"] is None  # missing bucket first
    keys = [tuple((b["key"][k] is None, b["key"][k] or 0 if k == "r" else
                   str(b["key"][k])) for k in ("h", "u", "r"))
            for b in page1["buckets"]]
    assert keys == sorted(keys)
    # paginate with after (typed key serialization like quickwit's)
    ak = page1["after_key"]
    assert ak == page1["buckets"][-1]["key"]
    after = {"h": "str:" + ak["h"] if ak["h"] is not None else None,
             "u": "str:" + ak["u"], "r": f'f64:{ak["r"]}'}
    aggs2 = {"c": {"composite": dict(base, size=500, after=after)}}
    page2 = both(aggs2)["c"]
    # pages partition the full bucket space
    aggs_all = {"c": {"composite": dict(base, size=1000)}}
    full = both(aggs_all)["c"]
    got = [json.dumps(b) for b in page1["buckets"] + page2["buckets"]]
    assert got == [json.dumps(b) for b in full["buckets"]]
    total = sum(b["doc_count"] for b in full["buckets"])
    assert total == 800  # every doc lands somewhere (h missing-bucketed)


def test_cardinality_aggs():
    # exact distinct counts over str / numeric / nullable columns (declared
    # deviation from the reference's HLL++ sketch: exact values, which equal
    # the golden's outputs at scenario cardinalities)
    import random
    rng = random.Random(13)
    docs = []
    for i in range(2500):
        d = {"timestamp": 1700000000 + (i % 97), "severity_text":
             rng.choice(["INFO", "WARN", "ERROR"]), "body": "x",
             "tenant_id": i % 41,
             "svc": rng.choice(["api", "ingest", "janitor"])}
        if rng.random() < 0.7:
            d["opt_u"] = rng.randrange(29)
        docs.append(d)
    schema = {"timestamp_field": "timestamp", "fields":
              splitgen.HDFS_SCHEMA["fields"] +
              [{"name": "opt_u", "type": "u64", "fast": True},
               {"name": "svc", "type": "str", "fast": True}]}
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    splits = []
    for s in range(2):
        w = splitgen.SplitWriter(schema, f"card-{s}")
        w.add_documents(docs[s::2])
        data = w.finalize()
        gpu.add_split(f"card-{s}", data)
        cpu.add_split(f"card-{s}", data)
        splits.append((f"card-{s}", len(docs[s::2])))
    aggs = {"sev": {"cardinality": {"field": "svc"}},
            "ten": {"cardinality": {"field": "tenant_id"}},
            "opt": {"cardinality": {"field": "opt_u"}},
            "ts": {"cardinality": {"field": "timestamp"}}}
    for q in ({"type": "match_all"},
              {"type": "term", "field": "severity_text", "value": "WARN"}):
        req = make_leaf_request(q, schema, splits, max_hits=0,
                                aggregation=aggs)
        g = gpu.leaf_search(req)
        e = cpu.leaf_search(req)
        gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
        ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
        assert gj == ej, (q, gj, ej)
        if q["type"] == "match_all":
            assert gj == {"sev": {"value": 3.0}, "ten": {"value": 41.0},
                          "opt": {"value": 29.0}, "ts": {"value": 97.0}}


# ------------------------------------------------- percentiles aggregation
def test_percentiles_aggs():
    # DDSketch restatement: sub-agg under date_histogram (the golden's shape)
    # and top-level, vs the oracle; bucketing must be bit-identical (shared
    # boundary tables), so exact JSON equality is required
    import random
    rng = random.Random(5)
    docs = []
    for i in range(4000):
        d = {"timestamp": (1700000000 + i % 7200) * 1000,
             "severity_text": "INFO" if i % 4 else "ERROR", "body": "x",
             "tenant_id": i % 3}
        if rng.random() < 0.9:
            d["lat"] = round(rng.uniform(0.0, 5000.0), 3)
        docs.append(d)
    schema = {"timestamp_field": None, "fields":
              [{"name": "timestamp", "type": "i64", "fast": True},
               {"name": "severity_text", "type": "text", "tokenizer": "raw",
                "fast": True},
               {"name": "body", "type": "text"},
               {"name": "tenant_id", "type": "u64", "fast": True},
               {"name": "lat", "type": "f64", "fast": True}]}
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    splits = []
    for s in range(2):
        w = splitgen.SplitWriter(schema, f"perc-{s}")
        w.add_documents(docs[s::2])
        data = w.finalize()
        gpu.add_split(f"perc-{s}", data)
        cpu.add_split(f"perc-{s}", data)
        splits.append((f"perc-{s}", len(docs[s::2])))

    aggs = {"per_hour": {"date_histogram": {"field": "timestamp",
                         "fixed_interval": "3600000ms"},
                         "aggs": {"lat_p": {"percentiles": {"field": "lat",
                                  "percents": [50, 85, 99],
                                  "keyed": False}}}},
            "lat_top": {"percentiles": {"field": "lat"}}}
    for q in ({"type": "match_all"},
              {"type": "term", "field": "severity_text", "value": "ERROR"}):
        req = make_leaf_request(q, schema, splits, max_hits=0,
                                aggregation=aggs)
        g = gpu.leaf_search(req)
        e = cpu.leaf_search(req)
        gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
        ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
        assert gj == ej, (q, gj, ej)
    # sanity: p50 < p85 < p99 and all within the value range
    b0 = gj["per_hour"]["buckets"][0]["lat_p"]["values"]
    vals = [v["value"] for v in b0]
    assert vals == sorted(vals) and 0 <= vals[0] <= 5100
    top = gj["lat_top"]["values"]
    assert set(top) == {"1.0", "5.0", "25.0", "50.0", "75.0", "95.0", "99.0"}


def test_terms_sub_aggregations():
    # stats-family sub-aggs under terms buckets (per-ord 40B slots on device
    # vs the oracle), incl. nullable sub columns and f64 sums
    import random
    rng = random.Random(9)
    docs = []
    for i in range(3000):
        d = {"timestamp": 1700000000 + i, "severity_text":
             "INFO" if i % 5 else "WARN", "body": "x", "tenant_id": i % 4,
             "svc": rng.choice(["api", "ing", "jan", "ui"])}
        if rng.random() < 0.8:
            d["lat"] = round(rng.uniform(1, 100), 2)
        docs.append(d)
    schema = {"timestamp_field": "timestamp", "fields":
              splitgen.HDFS_SCHEMA["fields"] +
              [{"name": "svc", "type": "str", "fast": True},
               {"name": "lat", "type": "f64", "fast": True}]}
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    splits = []
    for s in range(2):
        w = splitgen.SplitWriter(schema, f"tsub-{s}")
        w.add_documents(docs[s::2])
        data = w.finalize()
        gpu.add_split(f"tsub-{s}", data)
        cpu.add_split(f"tsub-{s}", data)
        splits.append((f"tsub-{s}", len(docs[s::2])))
    aggs = {"by_svc": {"terms": {"field": "svc", "size": 3},
                       "aggs": {"lat_stats": {"stats": {"field": "lat"}},
                                "lat_ext": {"extended_stats":
                                            {"field": "lat"}},
                                "lat_cnt": {"value_count":
                                            {"field": "lat"}}}}}
    for q in ({"type": "match_all"},
              {"type": "term", "field": "severity_text", "value": "WARN"}):
        req = make_leaf_request(q, schema, splits, max_hits=0,
                                aggregation=aggs)
        g = gpu.leaf_search(req)
        e = cpu.leaf_search(req)
        gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
        ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
        for gb, eb in zip(gj["by_svc"]["buckets"], ej["by_svc"]["buckets"]):
            assert gb["key"] == eb["key"]
            assert gb["doc_count"] == eb["doc_count"]
            for nm in ("lat_stats", "lat_ext"):
                for k2, ev in eb[nm].items():
                    gv = gb[nm][k2]
                    if isinstance(ev, dict):
                        for k3 in ev:
                            assert math.isclose(gv[k3], ev[k3],
                                                rel_tol=1e-9), (nm, k2, k3)
                    elif isinstance(ev, float):
                        assert math.isclose(gv, ev, rel_tol=1e-9), (nm, k2)
                    else:
                        assert gv == ev, (nm, k2)
            assert gb["lat_cnt"] == eb["lat_cnt"]
        assert gj["by_svc"]["sum_other_doc_count"] == \
            ej["by_svc"]["sum_other_doc_count"]


def test_terms_order_key_on_gpu(searchers):
    # order=_key flows through the product's per-split truncation (assembly)
    gpu, cpu = searchers
    for direction in ("asc", "desc"):
        aggs = {"t": {"terms": {"field": "tenant_name", "size": 5,
                                "split_size": 5,
                                "order": {"_key": direction}}}}
        req = make_leaf_request({"type": "match_all"}, SCHEMA, [(SID, NDOCS)],
                                max_hits=0, aggregation=aggs)
        g = gpu.leaf_search(req)
        e = cpu.leaf_search(req)
        gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
        ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
        assert gj == ej, (direction, gj, ej)
        keys = [b["key"] for b in gj["t"]["buckets"]]
        assert keys == sorted(keys, reverse=direction == "desc")


# ---------------------------------------------------------------- agg_fast
# Regression: the agg_fast EPILOGUE branch (any query that is not pure
# match_all-no-collect: filters, or match_all + hit collection) must apply
# the same lds_rep bucket-index mapping as the pure-agg tile loop; when
# n_buckets*2 <= AGG_LDS_BUCKETS the LDS histogram is 2-way replicated and
# the end-of-kernel flush sums interleaved pairs. Bare aggs (no sub-aggs)
# keep agg_fast=1; the per-hour histogram (~720 buckets) gets lds_rep=2.
BARE_AGGS = {
    "per_hour": {"date_histogram": {"field": "timestamp",
                                    "fixed_interval": "3600000ms"}},
    "per_tenant": {"terms": {"field": "tenant_name", "size": 10}},
}


def _bare_agg_both(searchers, query, max_hits=0):
    gpu, cpu = searchers
    req = make_leaf_request(query, SCHEMA, [(SID, NDOCS)],
                            max_hits=max_hits, aggregation=BARE_AGGS,
                            sort_fields=None)
    g = gpu.leaf_search(req)
    e = cpu.leaf_search(req)
    gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], BARE_AGGS)
    ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], BARE_AGGS)
    assert g.get("num_hits", 0) == e.get("num_hits", 0)
    return gj, ej


def test_agg_fast_bare_histogram_under_term_filter(searchers):
    q = {"type": "term", "field": "severity_text", "value": "INFO"}
    gj, ej = _bare_agg_both(searchers, q)
    assert gj == ej


def test_agg_fast_bare_histogram_under_range_filter(searchers):
    q = {"type": "range", "field": "tenant_id",
         "lower_bound": {"included": 100}, "upper_bound": {"excluded": 600}}
    gj, ej = _bare_agg_both(searchers, q)
    assert gj == ej


def test_agg_fast_bare_histogram_match_all_with_hits(searchers):
    # match_all + max_hits>0 routes through the epilogue (collect), not the
    # pure-agg tile loop
    gj, ej = _bare_agg_both(searchers, {"type": "match_all"}, max_hits=10)
    assert gj == ej


def test_agg_fast_bare_histogram_only_under_filter(searchers):
    # histogram WITHOUT the terms agg (af_terms=false leg)
    gpu, cpu = searchers
    aggs = {"per_hour": BARE_AGGS["per_hour"]}
    q = {"type": "term", "field": "severity_text", "value": "ERROR"}
    req = make_leaf_request(q, SCHEMA, [(SID, NDOCS)], max_hits=0,
                            aggregation=aggs)
    g = gpu.leaf_search(req)
    e = cpu.leaf_search(req)
    gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
    ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
    assert gj == ej


# ------------------------------------------------------- absence cache
def test_absence_cache_hit_miss_and_parity():
    """Negative term cache (leaf.rs:761-827): a required term proven absent
    populates the cache; the next search for it short-circuits the split
    (hit) with a response identical to the oracle's. Should-position absent
    terms must NOT populate it."""
    data = splitgen.generate_split(0, 30_000, seed=11)
    gpu = GpuSearcher(device=0)
    cpu = OracleSearcher()
    gpu.add_split("abs-split", data)
    cpu.add_split("abs-split", data)
    assert gpu.absence_cache_stats() == (0, 0, 0)

    q_absent = {"type": "bool", "must": [
        {"type": "term", "field": "body", "value": "zz_not_a_term"},
        {"type": "term", "field": "severity_text", "value": "INFO"}]}
    req = make_leaf_request(q_absent, SCHEMA, [("abs-split", 30_000)],
                            max_hits=5, aggregation={
                                "h": {"date_histogram": {
                                    "field": "timestamp",
                                    "fixed_interval": "86400000ms"}}})
    g1, e1 = gpu.leaf_search(req), cpu.leaf_search(req)
    assert g1.get("num_hits", 0) == 0 == e1.get("num_hits", 0)
    h, m, e = gpu.absence_cache_stats()
    assert (h, m) == (0, 1) and e == 1  # miss recorded, absence learned

    # second identical search: cache HIT short-circuits; response identical
    # up to timing (resource_stats carries wall-clock microseconds)
    def strip_stats(r):
        return {k: v for k, v in r.items() if k != "resource_stats"}
    g2 = gpu.leaf_search(req)
    h2, m2, _ = gpu.absence_cache_stats()
    assert (h2, m2) == (1, 1)
    assert strip_stats(g2) == strip_stats(g1) == strip_stats(e1)

    # a different query sharing the absent required term also hits
    q2 = {"type": "bool", "filter": [
        {"type": "term", "field": "body", "value": "zz_not_a_term"}]}
    req2 = make_leaf_request(q2, SCHEMA, [("abs-split", 30_000)], max_hits=3)
    g3 = gpu.leaf_search(req2)
    assert gpu.absence_cache_stats()[0] == 2
    assert g3.get("num_hits", 0) == 0 == \
        cpu.leaf_search(req2).get("num_hits", 0)

    # should-position absence is NOT required -> no new entries, no hit;
    # present terms still match
    q3 = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "zz_other_missing"},
        {"type": "term", "field": "body", "value": "w00000"}]}
    req3 = make_leaf_request(q3, SCHEMA, [("abs-split", 30_000)], max_hits=5,
                             sort_fields=[{"field_name": "_score",
                                           "sort_order": 1}])
    g4, e4 = gpu.leaf_search(req3), cpu.leaf_search(req3)
    assert g4["num_hits"] == e4["num_hits"] > 0
    assert gpu.absence_cache_stats()[2] == 1  # unchanged

    # removing the split drops its keys
    gpu._lib.qw_ctx_remove_split.argtypes = [
        __import__("ctypes").c_void_p, __import__("ctypes").c_char_p]
    assert gpu._lib.qw_ctx_remove_split(gpu._ctx, b"abs-split") == 0
    assert gpu.absence_cache_stats()[2] == 0


# ------------------------------------------------- memory budget (permits)
def test_memory_budget_accounting_and_refusal():
    """HBM accounting (search_permit_provider.rs:43-110 analog): add_split
    and search scratch are accounted against the ctx budget; an over-budget
    add_split is refused with QW_ERR_OVER_MEMORY_BUDGET."""
    data = splitgen.generate_split(0, 20_000, seed=5)
    g = GpuSearcher(device=0)
    g.add_split("m1", data)
    u1, b1, s1 = g.memory_stats()
    assert s1 == len(data) + 64
    assert u1 >= s1
    assert b1 > s1  # default budget resolved from device free memory
    req = make_leaf_request({"type": "match_all"}, SCHEMA, [("m1", 20_000)],
                            max_hits=5)
    g.leaf_search(req)
    u2, _, _ = g.memory_stats()
    assert u2 >= u1  # search scratch accounted
    # removing the split releases its accounted bytes
    import ctypes
    g._lib.qw_ctx_remove_split.argtypes = [ctypes.c_void_p, ctypes.c_char_p]
    assert g._lib.qw_ctx_remove_split(g._ctx, b"m1") == 0
    u3, _, s3 = g.memory_stats()
    assert s3 == 0 and u3 == u2 - s1

    # a ctx with a tiny configured budget refuses the split up front
    g2 = GpuSearcher(device=0, config={"hbm_memory_budget": 4096})
    with pytest.raises(RuntimeError, match="budget"):
        g2.add_split("m1", data)
    assert g2.memory_stats()[1] == 4096


# ---------------------------------------------------- multi-segment (QWA2)
def test_multi_segment_split_parity():
    """QWA2 multi-segment split through the HIP path vs the oracle:
    per-segment collection (collector.rs:475-594), (split, segment_ord,
    doc) tie-breaks, agg merge across segments, absence-cache semantics
    (a term must be absent from EVERY segment to be cached)."""
    a = splitgen.generate_split(0, 30_000, seed=9)
    b = splitgen.generate_split(1, 20_000, seed=9)
    multi = splitgen.concat_segments([a, b], "s")
    gpu = GpuSearcher(device=0)
    cpu = OracleSearcher()
    gpu.add_split("s", multi)
    cpu.add_split("s", multi)
    splits = [("s", 50_000)]

    # scored top-K across segments
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i}
        for i in (9, 10, 11)]}
    req = make_leaf_request(q, SCHEMA, splits, max_hits=20,
                            sort_fields=[{"field_name": "_score",
                                          "sort_order": 1}])
    g, e = gpu.leaf_search(req), cpu.leaf_search(req)
    assert_hits_equal(g, e, scored=True)
    assert {h.get("segment_ord", 0) for h in g["partial_hits"]} == {0, 1}
    for gh, eh in zip(g["partial_hits"], e["partial_hits"]):
        assert gh.get("segment_ord", 0) == eh.get("segment_ord", 0)

    # two-key field sort
    req = make_leaf_request(
        {"type": "term", "field": "severity_text", "value": "INFO"},
        SCHEMA, splits, max_hits=25,
        sort_fields=[{"field_name": "timestamp", "sort_order": 1},
                     {"field_name": "tenant_id", "sort_order": 0}])
    g, e = gpu.leaf_search(req), cpu.leaf_search(req)
    assert g["num_hits"] == e["num_hits"]
    assert [(h.get("segment_ord", 0), h.get("doc_id", 0))
            for h in g["partial_hits"]] == \
        [(h.get("segment_ord", 0), h.get("doc_id", 0))
         for h in e["partial_hits"]]

    # aggregations merge across segments
    aggs = {"per_hour": {"date_histogram": {"field": "timestamp",
                                            "fixed_interval": "3600000ms"}},
            "per_tenant": {"terms": {"field": "tenant_name", "size": 10}}}
    req = make_leaf_request({"type": "match_all"}, SCHEMA, splits,
                            max_hits=0, aggregation=aggs)
    g, e = gpu.leaf_search(req), cpu.leaf_search(req)
    assert g["num_hits"] == e["num_hits"] == 50_000
    gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
    ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
    assert gj == ej

    # search_after pagination crossing a segment boundary
    q = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i}
        for i in (9, 10, 11)]}
    sort = [{"field_name": "_score", "sort_order": 1}]
    seen = []
    cursor = None
    for _page in range(3):
        req = make_leaf_request(q, SCHEMA, splits, max_hits=9,
                                sort_fields=sort)
        if cursor:
            req["search_request"]["search_after"] = cursor
        r = gpu.leaf_search(req)
        hits = r.get("partial_hits", [])
        if not hits:
            break
        seen.extend(hits)
        cursor = hits[-1]
    r_all = cpu.leaf_search(make_leaf_request(
        q, SCHEMA, splits, max_hits=len(seen), sort_fields=sort))
    assert [(h.get("segment_ord", 0), h.get("doc_id", 0)) for h in seen] == \
        [(h.get("segment_ord", 0), h.get("doc_id", 0))
         for h in r_all["partial_hits"]]


def test_multi_segment_absence_cache_all_segments_rule():
    """A required term present in one segment must NOT be cached as absent;
    a term absent from every segment must."""
    # craft two doc-writer segments: 'onlyseg0' appears only in segment 0
    w0 = splitgen.SplitWriter(
        {"timestamp_field": None,
         "fields": [{"name": "body", "type": "text", "tokenizer": "default",
                     "record": "freq", "fieldnorms": True}]}, "seg0",
        store_docs=False)
    w0.add_documents([{"body": "onlyseg0 shared"}, {"body": "shared"}])
    w1 = splitgen.SplitWriter(
        {"timestamp_field": None,
         "fields": [{"name": "body", "type": "text", "tokenizer": "default",
                     "record": "freq", "fieldnorms": True}]}, "seg1",
        store_docs=False)
    w1.add_documents([{"body": "shared word"}, {"body": "word"}])
    multi = splitgen.concat_segments([w0.finalize(), w1.finalize()], "ms")
    schema = {"timestamp_field": None, "fields": [
        {"name": "body", "type": "text", "tokenizer": "default",
         "record": "freq", "fieldnorms": True}]}
    gpu = GpuSearcher(device=0)
    gpu.add_split("ms", multi)

    q1 = {"type": "bool", "must": [
        {"type": "term", "field": "body", "value": "onlyseg0"}]}
    r = gpu.leaf_search(make_leaf_request(q1, schema, [("ms", 4)], max_hits=5))
    assert r["num_hits"] == 1
    assert gpu.absence_cache_stats()[2] == 0  # present in seg0: not cached

    q2 = {"type": "bool", "must": [
        {"type": "term", "field": "body", "value": "nowhere"}]}
    r = gpu.leaf_search(make_leaf_request(q2, schema, [("ms", 4)], max_hits=5))
    assert r.get("num_hits", 0) == 0
    assert gpu.absence_cache_stats()[2] == 1  # absent everywhere: cached
    r2 = gpu.leaf_search(make_leaf_request(q2, schema, [("ms", 4)], max_hits=5))
    assert r2.get("num_hits", 0) == 0
    assert gpu.absence_cache_stats()[0] >= 1  # second probe hits


# ---------------------------------------------------- mixed-type columns
def test_mixed_type_column_sort_parity():
    """Mixed-type dynamic column (u64-beyond-i64 + i64 + f64 + bool):
    numeric sort order across types, typed echo in sort_value, search_after
    band through the f64-key domain — GPU vs oracle."""
    schema = {"timestamp_field": None, "fields": [
        {"name": "mx", "type": "mixed", "fast": True},
        {"name": "body", "type": "text", "tokenizer": "default",
         "record": "freq", "fieldnorms": True}]}
    w = splitgen.SplitWriter(schema, "mx-split", store_docs=False)
    w.add_documents([
        {"mx": 18000000000000000000, "body": "a"},
        {"mx": 0, "body": "a"},
        {"mx": True, "body": "a"},
        {"mx": 10.5, "body": "a"},
        {"mx": -10, "body": "a"},
        {"body": "a"},  # missing -> None, sorts last
    ])
    data = w.finalize()
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    gpu.add_split("mx-split", data)
    cpu.add_split("mx-split", data)
    for order in (0, 1):
        req = make_leaf_request({"type": "match_all"}, schema,
                                [("mx-split", 6)], max_hits=10,
                                sort_fields=[{"field_name": "mx",
                                              "sort_order": order}])
        g, e = gpu.leaf_search(req), cpu.leaf_search(req)
        assert g["num_hits"] == e["num_hits"] == 6
        assert g["partial_hits"] == e["partial_hits"], (order, g, e)
        vals = [h.get("sort_value") for h in g["partial_hits"]]
        if order == 0:  # asc: -10, 0, true, 10.5, 18e18, None
            assert vals == [{"i64": -10}, {"u64": 0}, {"boolean": True},
                            {"f64": 10.5}, {"u64": 18000000000000000000},
                            None]
    # search_after from a negative i64 cursor, ascending
    req = make_leaf_request({"type": "match_all"}, schema, [("mx-split", 6)],
                            max_hits=10,
                            sort_fields=[{"field_name": "mx",
                                          "sort_order": 0}])
    req["search_request"]["search_after"] = {"sort_value": {"i64": -10}}
    g, e = gpu.leaf_search(req), cpu.leaf_search(req)
    assert g["partial_hits"] == e["partial_hits"]
    assert [h.get("sort_value") for h in g["partial_hits"]][:4] == [
        {"u64": 0}, {"boolean": True}, {"f64": 10.5},
        {"u64": 18000000000000000000}]


# --------------------------------------------------------------- phrases
def test_phrase_parity_gpu():
    """Multi-token phrases through k_phrase_bitmap (device HitSet path) vs
    the oracle: bare phrase, under bool filter/must/must_not, repeated
    tokens, absent token, multi-block postings."""
    import random as _r
    schema = {"timestamp_field": None, "fields": [
        {"name": "body", "type": "text", "tokenizer": "default",
         "record": "position", "fieldnorms": True},
        {"name": "sev", "type": "u64", "fast": True}]}
    rng = _r.Random(77)
    vocab = ["alpha", "beta", "gamma", "delta", "eps", "zeta"]
    docs = [{"body": " ".join(rng.choice(vocab)
                              for _ in range(rng.randint(3, 12))),
             "sev": rng.randint(0, 9)} for _ in range(3000)]
    w = splitgen.SplitWriter(schema, "ph", store_docs=False)
    w.add_documents(docs)
    data = w.finalize()
    gpu, cpu = GpuSearcher(device=0), OracleSearcher()
    gpu.add_split("ph", data)
    cpu.add_split("ph", data)

    def phrase(toks):
        return {"type": "full_text", "field": "body", "text": " ".join(toks),
                "params": {"mode": {"type": "phrase"}}}

    queries = [
        phrase(["alpha", "beta"]),
        phrase(["beta", "beta"]),
        phrase(["gamma", "delta", "eps"]),
        phrase(["eps", "eps", "eps"]),
        phrase(["alpha", "missingtok"]),
        {"type": "bool", "filter": [phrase(["alpha", "beta"])],
         "must": [{"type": "term", "field": "body", "value": "gamma"}]},
        {"type": "bool", "filter": [phrase(["alpha", "beta"])],
         "must": [{"type": "range", "field": "sev",
                   "lower_bound": {"included": 3},
                   "upper_bound": {"excluded": 8}}]},
        {"type": "bool",
         "must": [{"type": "term", "field": "body", "value": "gamma"}],
         "must_not": [phrase(["alpha", "beta"])]},
    ]
    for q in queries:
        req = make_leaf_request(q, schema, [("ph", 3000)], max_hits=3000)
        g, e = gpu.leaf_search(req), cpu.leaf_search(req)
        assert g.get("num_hits", 0) == e.get("num_hits", 0), q
        assert sorted(h.get("doc_id", 0) for h in g.get("partial_hits", [])) \
            == sorted(h.get("doc_id", 0) for h in e.get("partial_hits", [])), q
    # memoized: second run hits the bitmap cache and stays identical
    req = make_leaf_request(queries[0], schema, [("ph", 3000)], max_hits=10)
    r1 = gpu.leaf_search(req)
    r2 = gpu.leaf_search(req)
    assert r1.get("num_hits") == r2.get("num_hits")


def test_terms_order_by_subagg_on_gpu(searchers):
    # terms order by a sub-aggregation value through the HIP path
    gpu, cpu = searchers
    for order in ({"st.avg": "desc"}, {"st.max": "asc"},
                  {"avg_t": "desc"}):
        aggs = {"t": {"terms": {"field": "tenant_name", "size": 8,
                                "order": order},
                      "aggs": {"st": {"stats": {"field": "tenant_id"}},
                               "avg_t": {"avg": {"field": "tenant_id"}}}}}
        req = make_leaf_request({"type": "match_all"}, SCHEMA, [(SID, NDOCS)],
                                max_hits=0, aggregation=aggs)
        g = gpu.leaf_search(req)
        e = cpu.leaf_search(req)
        gj = gpu.finalize_agg_json(g["intermediate_aggregation_result"], aggs)
        ej = cpu.finalize_agg_json(e["intermediate_aggregation_result"], aggs)
        assert [b["key"] for b in gj["t"]["buckets"]] == \
            [b["key"] for b in ej["t"]["buckets"]], order


# ------------------------------------------------ per-split failure as data
def test_per_split_failure_is_data_not_exception():
    """One split failing must not fail the call: the reference reports it
    inside failed_splits (leaf.rs:2143-2148) and the others still answer.
    Here: a phrase query over one split WITH positions and one WITHOUT."""
    schema = {"timestamp_field": None, "fields": [
        {"name": "body", "type": "text", "tokenizer": "default",
         "record": "position", "fieldnorms": True}]}
    wa = splitgen.SplitWriter(schema, "sa", store_docs=False)
    wa.add_documents([{"body": "alpha beta"}, {"body": "beta alpha"}])
    schema_nf = {"timestamp_field": None, "fields": [
        {"name": "body", "type": "text", "tokenizer": "default",
         "record": "freq", "fieldnorms": True}]}
    wb = splitgen.SplitWriter(schema_nf, "sb", store_docs=False)
    wb.add_documents([{"body": "alpha beta"}])
    gpu = GpuSearcher(device=0)
    gpu.add_split("sa", wa.finalize())
    gpu.add_split("sb", wb.finalize())
    q = {"type": "full_text", "field": "body", "text": "alpha beta",
         "params": {"mode": {"type": "phrase"}}}
    r = gpu.leaf_search(make_leaf_request(q, schema, [("sa", 2), ("sb", 1)],
                                          max_hits=5))
    assert r["num_hits"] == 1  # sa's "alpha beta" doc
    fails = r.get("failed_splits", [])
    assert len(fails) == 1 and fails[0]["split_id"] == "sb"
    assert "position" in fails[0]["error"]
    assert r.get("num_successful_splits", 0) == 1
    assert r.get("num_attempted_splits", 0) == 2
