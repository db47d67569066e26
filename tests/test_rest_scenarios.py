"""Replay the reference's rest-api-tests golden scenarios through the HTTP
shim (quickwit_amd/rest.py) with run_tests.py's checking semantics
(tests/rest_replay.py). The same suite runs against the oracle engine here
(CPU) and against the HIP product on the GPU box — the expected JSON is the
reference's own (tests/golden/rest_scenarios.json, extracted by
tests/golden/extract_goldens.py).

As of round 2 the nine core suites replay in full (272/272 steps, zero
skips) and a tenth extra ES-compat slice replays 118/131 with 13
declared skips (slop>0, multi-token phrase prefix, regex, cross-suite
_stats state);
the skip machinery remains so a future regression reports a reason instead
of a bare failure. Every step must match the reference byte-for-byte at
run_tests.py's granularity.
"""
import json
import os

import pytest

from rest_replay import agg_kinds, replay_suite

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

UNSUPPORTED_AGGS = {"percentiles": None,
                    "cardinality": None,
                    "extended_stats": None,
                    "composite": None,
                    "range": None,
                    "avg": None, "stats": None, "sum": None, "min": None,
                    "max": None, "value_count": None,
                    "date_histogram": None, "histogram": None, "terms": None}
MULTI_VALUED_FIELDS = set()  # multi-valued str fast columns now supported
NUMERIC_TERMS_FIELDS = set()  # terms over numeric fast columns supported


MIXED_TYPE_SORT_FIELDS = set()  # mixed-type dynamic columns supported (r2)


def skip_step(i, step):
    if "json" not in step:
        return None
    body = step["json"]
    if not isinstance(body, dict):
        return None
    for s in body.get("sort", []):
        field = s if isinstance(s, str) else next(iter(s))
        if field in MIXED_TYPE_SORT_FIELDS:
            return f"mixed-type dynamic column {field}: later round"

    for kind, field in agg_kinds(body.get("aggs")):
        reason = UNSUPPORTED_AGGS.get(kind, f"unknown agg kind {kind}")
        if reason:
            return reason
        if field in MULTI_VALUED_FIELDS:
            return f"multi-valued field {field}: later round"
        if kind == "terms" and field in NUMERIC_TERMS_FIELDS:
            return "terms over numeric fast field: later round"
    return None


def load_suite(name):
    with open(os.path.join(REPO, "tests", "golden", "rest_scenarios.json")) as f:
        return json.load(f)["suites"][name]


def make_client(searcher_factory):
    from fastapi.testclient import TestClient

    from quickwit_amd.rest import create_app
    return TestClient(create_app(searcher_factory))


def run_aggregations(searcher_factory):
    steps = load_suite("aggregations")
    client = make_client(searcher_factory)
    ran, skipped = replay_suite(client, steps, skip_step)
    # the setup/teardown + the in-scope golden search steps must all run
    assert ran >= 18, (ran, skipped)
    return ran, skipped


def test_rest_aggregations_suite_oracle():
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    ran, skipped = run_aggregations(OracleSearcher)
    for _, reason in skipped:
        assert "later round" in reason, reason


@pytest.mark.gpu
def test_rest_aggregations_suite_gpu():
    from quickwit_amd.api import GpuSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_aggregations(lambda: GpuSearcher(device=0))


def run_sort_orders(searcher_factory):
    steps = load_suite("sort_orders")
    client = make_client(searcher_factory)
    ran, skipped = replay_suite(client, steps)
    assert not skipped and ran == len(steps)


def test_rest_sort_orders_suite_oracle():
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_sort_orders(OracleSearcher)


@pytest.mark.gpu
def test_rest_sort_orders_suite_gpu():
    from quickwit_amd.api import GpuSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_sort_orders(lambda: GpuSearcher(device=0))


def run_search_after(searcher_factory):
    steps = load_suite("search_after")
    client = make_client(searcher_factory)
    ran, skipped = replay_suite(client, steps, skip_step)
    assert ran >= 18, (ran, skipped)
    for _, reason in skipped:
        assert "later round" in reason, reason


def test_rest_search_after_suite_oracle():
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_search_after(OracleSearcher)


@pytest.mark.gpu
def test_rest_search_after_suite_gpu():
    from quickwit_amd.api import GpuSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_search_after(lambda: GpuSearcher(device=0))


def run_es_compatibility(searcher_factory):
    steps = load_suite("es_compatibility")
    client = make_client(searcher_factory)
    ran, skipped = replay_suite(client, steps, skip_step)
    assert ran >= 92, (ran, skipped)
    for _, reason in skipped:
        assert "later round" in reason, reason


def test_rest_es_compatibility_suite_oracle():
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_es_compatibility(OracleSearcher)


@pytest.mark.gpu
def test_rest_es_compatibility_suite_gpu():
    from quickwit_amd.api import GpuSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_es_compatibility(lambda: GpuSearcher(device=0))


def run_qw_search_api(searcher_factory):
    steps = load_suite("qw_search_api")
    client = make_client(searcher_factory)
    ran, skipped = replay_suite(client, steps)
    assert not skipped and ran == len(steps)


def test_rest_qw_search_api_suite_oracle():
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_qw_search_api(OracleSearcher)


@pytest.mark.gpu
def test_rest_qw_search_api_suite_gpu():
    from quickwit_amd.api import GpuSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_qw_search_api(lambda: GpuSearcher(device=0))


def run_simple_suite(name, searcher_factory):
    steps = load_suite(name)
    client = make_client(searcher_factory)
    ran, skipped = replay_suite(client, steps)
    assert not skipped and ran == len(steps)


@pytest.mark.parametrize("suite", ["default_search_fields", "multi_splits"])
def test_rest_simple_suites_oracle(suite):
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_simple_suite(suite, OracleSearcher)


@pytest.mark.gpu
@pytest.mark.parametrize("suite", ["default_search_fields", "multi_splits"])
def test_rest_simple_suites_gpu(suite):
    from quickwit_amd.api import GpuSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_simple_suite(suite, lambda: GpuSearcher(device=0))


def test_rest_terms_with_subaggs_oracle():
    # terms buckets with stats subs through the ES _search surface
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    client = make_client(OracleSearcher)
    r = client.request("POST", "/api/v1/indexes", json={
        "version": "0.8", "index_id": "tsub",
        "doc_mapping": {"mode": "dynamic",
                        "dynamic_mapping": {"fast": True,
                                            "tokenizer": "default"}}})
    assert r.status_code == 200, r.text
    nd = "\n".join(
        json.dumps({"svc": ["api", "ing", "jan"][i % 3],
                    "lat": float(10 * (i % 7))}) for i in range(60)) + "\n"
    r = client.request("POST", "/api/v1/tsub/ingest", content=nd)
    assert r.status_code == 200, r.text
    r = client.request("GET", "/api/v1/_elastic/tsub/_search", json={
        "size": 0,
        "aggs": {"by_svc": {"terms": {"field": "svc", "size": 10,
                                      "order": {"_key": "asc"}},
                            "aggs": {"lat_avg": {"avg": {"field": "lat"}},
                                     "lat_st": {"stats": {"field": "lat"}}}}}})
    assert r.status_code == 200, r.text
    buckets = r.json()["aggregations"]["by_svc"]["buckets"]
    assert [b["key"] for b in buckets] == ["api", "ing", "jan"]
    assert sum(b["doc_count"] for b in buckets) == 60
    for b in buckets:
        assert b["lat_st"]["count"] == b["doc_count"]
        assert abs(b["lat_avg"]["value"] -
                   b["lat_st"]["sum"] / b["lat_st"]["count"]) < 1e-9


def run_concat(searcher_factory):
    steps = load_suite("concat_fields")
    client = make_client(searcher_factory)
    ran, skipped = replay_suite(client, steps)
    assert not skipped and ran == len(steps)


def test_rest_concat_fields_suite_oracle():
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_concat(OracleSearcher)


@pytest.mark.gpu
def test_rest_concat_fields_suite_gpu():
    from quickwit_amd.api import GpuSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_concat(lambda: GpuSearcher(device=0))


def run_tag_fields(searcher_factory):
    steps = load_suite("tag_fields")
    client = make_client(searcher_factory)
    ran, skipped = replay_suite(client, steps)
    assert not skipped and ran == len(steps)


def test_rest_tag_fields_suite_oracle():
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_tag_fields(OracleSearcher)


@pytest.mark.gpu
def test_rest_tag_fields_suite_gpu():
    from quickwit_amd.api import GpuSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_tag_fields(lambda: GpuSearcher(device=0))


def skip_extra(i, step):
    # declared skips inside the extra ES-compat slice:
    if step.get("endpoint") == "_stats":
        # the reference runs 0020-stats after ALL its scenario files in one
        # session; the cross-index _all totals depend on indexes created by
        # files outside this extracted slice
        return "global _stats totals depend on cross-suite session state"
    body = step.get("json")
    if isinstance(body, dict):
        qy = body.get("query", {})
        mp = qy.get("match_phrase")
        if isinstance(mp, dict):
            [(_f, spec)] = mp.items()
            if isinstance(spec, dict) and spec.get("slop"):
                return "phrase slop > 0: declared out (DESIGN.md §7)"
        mm = qy.get("multi_match")
        if isinstance(mm, dict):
            if mm.get("slop"):
                return "phrase slop > 0: declared out (DESIGN.md §7)"
            if mm.get("type") == "phrase_prefix":
                return "multi-token phrase prefix: declared out"
        mpp = qy.get("match_phrase_prefix")
        if isinstance(mpp, dict):
            [(_f, spec)] = mpp.items()
            text = str(spec["query"] if isinstance(spec, dict)
                       else spec).strip()
            if " " in text:
                return "multi-token phrase prefix: declared out"
        if "regexp" in qy:
            return "regex queries: declared out (DESIGN.md §7)"
    return None


def run_es_compat_extra(searcher_factory):
    steps = load_suite("es_compat_extra")
    client = make_client(searcher_factory)
    ran, skipped = replay_suite(client, steps, skip_extra)
    assert ran >= 118, (ran, skipped)
    assert len(skipped) <= 13, skipped
    for _, reason in skipped:
        assert "declared" in reason or "session state" in reason, reason
    return ran, skipped


def test_rest_es_compat_extra_suite_oracle():
    from quickwit_amd.api import OracleSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_es_compat_extra(OracleSearcher)


@pytest.mark.gpu
def test_rest_es_compat_extra_suite_gpu():
    from quickwit_amd.api import GpuSearcher
    import __graft_entry__
    __graft_entry__.build()
    run_es_compat_extra(lambda: GpuSearcher(device=0))
