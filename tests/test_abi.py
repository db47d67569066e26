"""C-ABI surface checks that need no GPU (the library must load and export
every symbol include/quickwit_amd.h declares; the product path must fail
LOUDLY without a GPU — QW_ERR_NO_GPU, never a CPU fallback), plus CPU-side
parity of the ctx-less entry points (merge, agg finalize) against the oracle:
qw_merge_leaf_responses must reproduce the oracle's cross-split merge
(reference merge_fruits semantics, collector.rs:832-861)."""
import ctypes
import json
import os
import re
import subprocess

import pytest

from quickwit_amd import proto, splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "libquickwit_amd.so")


@pytest.fixture(scope="module", autouse=True)
def build_all():
    import __graft_entry__
    __graft_entry__.build()


def test_malformed_request_bytes_never_crash():
    """§8b boundary contract: errors come back as status codes + messages,
    never faults. The C++ pb decoder (csrc/pb.h, shared by product and
    oracle) must survive truncated, corrupted and random request bytes —
    returning an error or an empty result, in-process."""
    import random

    from quickwit_amd import proto, splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    s = OracleSearcher()
    s.add_split("s", splitgen.generate_split(0, 500, seed=1))
    good = proto.encode("LeafSearchRequest", make_leaf_request(
        {"type": "match_all"}, splitgen.HDFS_SCHEMA, [("s", 500)],
        max_hits=5))
    rng = random.Random(7)
    cases = [b"", b"\xff", b"\xff" * 16, good[:-3], good[:7],
             bytes(rng.randrange(256) for _ in range(64))]
    for _ in range(120):
        b = bytearray(good)
        for _ in range(rng.randrange(1, 5)):
            b[rng.randrange(len(b))] = rng.randrange(256)
        cases.append(bytes(b))
    for c in cases:
        try:
            s.leaf_search_raw(c)  # error status surfaces as an exception
        except Exception:
            pass


def test_corrupted_split_containers_never_crash():
    """qw_ctx_add_split must reject truncated/corrupted QWA1 images with a
    status code (wild meta offsets must not drive pointer arithmetic);
    a split that slips past validation may return garbage results but
    must not fault."""
    import random

    from quickwit_amd import splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    good = splitgen.generate_split(0, 500, seed=1)
    rng = random.Random(11)
    cases = [b"", b"QWAMDSP1", good[:100], good[:-10], b"\x00" * 200,
             bytes(rng.randrange(256) for _ in range(512))]
    for _ in range(80):
        b = bytearray(good)
        for _ in range(rng.randrange(1, 6)):
            b[rng.randrange(len(b))] = rng.randrange(256)
        cases.append(bytes(b))
    for _ in range(40):  # footer bytes drive the meta offsets — hit them
        b = bytearray(good)
        b[len(b) - 24 + rng.randrange(16)] = rng.randrange(256)
        cases.append(bytes(b))
    req = make_leaf_request({"type": "match_all"}, splitgen.HDFS_SCHEMA,
                            [("x", 500)], max_hits=5)
    for c in cases:
        s = OracleSearcher()
        try:
            s.add_split("x", c)
            s.leaf_search(req)
        except Exception:
            pass


def test_deeply_nested_query_ast_is_error_not_stack_overflow():
    """A pathological query_ast nested 200k levels deep must come back as
    a per-split failure ('recursion limit exceeded', mirroring
    serde_json's 128 default in the reference), not a stack-overflow
    fault — which is exactly what it was before the minijson depth cap."""
    import json

    from quickwit_amd import proto, splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    s = OracleSearcher()
    s.add_split("d", splitgen.generate_split(0, 200, seed=1))
    base = make_leaf_request({"type": "match_all"}, splitgen.HDFS_SCHEMA,
                             [("d", 200)], max_hits=5)
    for depth, ok in ((50, True), (200, False), (200_000, False)):
        ast = ('{"type":"bool","must":[' * depth
               + '{"type":"term","field":"body","value":"w00001"}'
               + ']}' * depth)
        req = json.loads(json.dumps(base))
        req["search_request"]["query_ast"] = ast
        resp = proto.decode(
            "LeafSearchResponse",
            s.leaf_search_raw(proto.encode("LeafSearchRequest", req)))
        failed = resp.get("failed_splits", [])
        if ok:
            assert not failed and resp.get("num_hits", 0) > 0, depth
        else:
            assert failed and "recursion" in failed[0]["error"], (depth,
                                                                  failed)


def test_deep_paren_nesting_in_query_grammar_is_error():
    """Same stack-overflow class through the query-grammar parser:
    200k nested parens in user_input must fail the split, not the
    process."""
    from quickwit_amd import proto, splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    s = OracleSearcher()
    s.add_split("g", splitgen.generate_split(0, 200, seed=1))
    for depth, ok in ((100, True), (200_000, False)):
        txt = "(" * depth + "body:w00001" + ")" * depth
        q = {"type": "user_input", "user_text": txt,
             "default_fields": ["body"]}
        resp = proto.decode(
            "LeafSearchResponse",
            s.leaf_search_raw(proto.encode(
                "LeafSearchRequest",
                make_leaf_request(q, splitgen.HDFS_SCHEMA, [("g", 200)],
                                  max_hits=5))))
        failed = resp.get("failed_splits", [])
        if ok:
            assert not failed and resp.get("num_hits", 0) > 0, depth
        else:
            assert failed and "recursion" in failed[0]["error"], (depth,
                                                                  failed)


def test_degenerate_histogram_intervals_error_not_hang():
    """interval <= 0 is rejected at parse (tantivy errors on it) and a
    pathologically tiny interval trips the 65000-bucket gap-fill cap
    (tantivy's AggregationLimits default) at finalize — before the fix,
    interval=1e-12 looped over ~1e15 empty buckets."""
    from quickwit_amd import proto, splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    s = OracleSearcher()
    s.add_split("a", splitgen.generate_split(0, 500, seed=1))

    def leaf(aggs):
        req = make_leaf_request({"type": "match_all"}, splitgen.HDFS_SCHEMA,
                                [("a", 500)], max_hits=0, aggregation=aggs)
        return proto.decode(
            "LeafSearchResponse",
            s.leaf_search_raw(proto.encode("LeafSearchRequest", req)))

    for aggs in (
        {"h": {"date_histogram": {"field": "timestamp",
                                  "fixed_interval": "0ms"}}},
        {"h": {"date_histogram": {"field": "timestamp",
                                  "fixed_interval": "-5ms"}}},
        {"h": {"histogram": {"field": "tenant_id", "interval": 0}}},
    ):
        failed = leaf(aggs).get("failed_splits", [])
        assert failed and "interval" in failed[0]["error"], aggs

    tiny = {"h": {"histogram": {"field": "tenant_id", "interval": 1e-12}}}
    resp = leaf(tiny)
    assert not resp.get("failed_splits")  # sparse collection is fine
    import pytest as _pytest
    with _pytest.raises(RuntimeError):   # gap-fill cap trips at finalize
        s.finalize_agg_json(resp["intermediate_aggregation_result"], tiny)

    ok = {"h": {"date_histogram": {"field": "timestamp",
                                   "fixed_interval": "3600000ms"}}}
    resp = leaf(ok)
    j = s.finalize_agg_json(resp["intermediate_aggregation_result"], ok)
    assert len(j["h"]["buckets"]) > 0  # normal path unaffected


def test_more_than_two_sort_fields_rejected():
    """search.proto:269: at most two sort fields; a third is an error on
    both engines (the oracle previously accepted it silently)."""
    from quickwit_amd import proto, splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    s = OracleSearcher()
    s.add_split("a", splitgen.generate_split(0, 500, seed=1))
    req = make_leaf_request(
        {"type": "match_all"}, splitgen.HDFS_SCHEMA, [("a", 500)],
        max_hits=3,
        sort_fields=[{"field_name": "timestamp", "sort_order": 1},
                     {"field_name": "tenant_id", "sort_order": 0},
                     {"field_name": "severity_text", "sort_order": 1}])
    resp = proto.decode(
        "LeafSearchResponse",
        s.leaf_search_raw(proto.encode("LeafSearchRequest", req)))
    failed = resp.get("failed_splits", [])
    assert failed and "two sort fields" in failed[0]["error"]


def test_unknown_split_is_per_split_failure_not_call_error():
    """A split the ctx cannot open is a failed_splits entry inside the
    response (leaf.rs:2143-2148) and the other splits still answer —
    previously it failed the whole call with QW_ERR_NOT_FOUND."""
    from quickwit_amd import proto, splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    s = OracleSearcher()
    s.add_split("a", splitgen.generate_split(0, 500, seed=1))
    req = make_leaf_request({"type": "match_all"}, splitgen.HDFS_SCHEMA,
                            [("a", 500), ("ghost", 100)], max_hits=3)
    resp = proto.decode(
        "LeafSearchResponse",
        s.leaf_search_raw(proto.encode("LeafSearchRequest", req)))
    assert resp["num_hits"] == 500
    assert resp["num_attempted_splits"] == 2
    assert resp["num_successful_splits"] == 1
    failed = resp.get("failed_splits", [])
    assert [f["split_id"] for f in failed] == ["ghost"]
    assert "unknown split" in failed[0]["error"]


def test_agg_field_type_mismatch_is_error_absent_is_empty():
    """tantivy semantics: an aggregation over a PRESENT column of an
    incompatible type is an error (date_histogram on u64, stats on str);
    a column ABSENT from the split contributes empty results (columns
    open optionally — the aggregations golden relies on this)."""
    from quickwit_amd import proto, splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    s = OracleSearcher()
    s.add_split("a", splitgen.generate_split(0, 500, seed=1))

    def failed(aggs):
        req = make_leaf_request({"type": "match_all"}, splitgen.HDFS_SCHEMA,
                                [("a", 500)], max_hits=0, aggregation=aggs)
        resp = proto.decode(
            "LeafSearchResponse",
            s.leaf_search_raw(proto.encode("LeafSearchRequest", req)))
        return [f["error"] for f in resp.get("failed_splits", [])]

    assert "non-datetime" in failed(
        {"h": {"date_histogram": {"field": "tenant_id",
                                  "fixed_interval": "1000ms"}}})[0]
    assert "non-numeric" in failed(
        {"m": {"stats": {"field": "tenant_name"}}})[0]
    assert "non-numeric" in failed(
        {"h": {"histogram": {"field": "tenant_name", "interval": 5}}})[0]
    # absent columns: empty results, no failure
    assert failed({"h": {"date_histogram": {"field": "nope",
                                            "fixed_interval": "1000ms"}}}) \
        == []
    assert failed({"m": {"stats": {"field": "nope"}}}) == []
    # value_count works on any column type
    assert failed({"m": {"value_count": {"field": "tenant_name"}}}) == []


def test_term_on_fast_only_column_and_negative_u64_bounds():
    """Two query-plan parity fixes pinned: (1) a term query on a fast-only
    (non-text) column is equality over the fast field — previously it
    silently matched nothing; (2) a negative range bound against a u64
    column clamps at the domain edge (lower -> 0, upper -> empty) instead
    of wrapping to a huge unsigned value."""
    from quickwit_amd import proto, splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    s = OracleSearcher()
    s.add_split("a", splitgen.generate_split(0, 500, seed=1))

    def hits(q):
        req = make_leaf_request(q, splitgen.HDFS_SCHEMA, [("a", 500)],
                                max_hits=3)
        resp = proto.decode(
            "LeafSearchResponse",
            s.leaf_search_raw(proto.encode("LeafSearchRequest", req)))
        return (resp.get("num_hits", 0),
                [f["error"] for f in resp.get("failed_splits", [])])

    n_0_100, _ = hits({"type": "range", "field": "tenant_id",
                       "lower_bound": {"included": 0},
                       "upper_bound": {"included": 100}})
    n_neg, _ = hits({"type": "range", "field": "tenant_id",
                     "lower_bound": {"included": -50},
                     "upper_bound": {"included": 100}})
    assert n_neg == n_0_100 > 0
    assert hits({"type": "range", "field": "tenant_id",
                 "upper_bound": {"included": -5}})[0] == 0

    n_range, _ = hits({"type": "range", "field": "tenant_id",
                       "lower_bound": {"included": 17},
                       "upper_bound": {"included": 17}})
    n_term, _ = hits({"type": "term", "field": "tenant_id", "value": "17"})
    assert n_term == n_range > 0
    assert hits({"type": "term", "field": "tenant_name",
                 "value": "t0001"})[0] > 0
    assert hits({"type": "term", "field": "tenant_id", "value": "-3"})[0] == 0
    _, failed = hits({"type": "term", "field": "tenant_id", "value": "abc"})
    assert failed and "invalid term value" in failed[0]
    # term_set shares the same leaf helper
    n_set, _ = hits({"type": "term_set", "terms_per_field":
                     {"tenant_name": ["t0001", "t0002"]}})
    n_bool, _ = hits({"type": "bool", "should": [
        {"type": "term", "field": "tenant_name", "value": "t0001"},
        {"type": "term", "field": "tenant_name", "value": "t0002"}]})
    assert n_set == n_bool > 0


def test_corrupted_merge_inputs_never_crash():
    """The rank-0 merge path consumes response bytes and QAGG1 blobs that
    crossed the wire: qw_merge_leaf_responses and qw_finalize_agg_to_json
    must survive mutated and truncated inputs with a status code."""
    import ctypes
    import json
    import random

    from quickwit_amd import proto, splitgen
    from quickwit_amd.api import OracleSearcher, make_leaf_request
    from quickwit_amd.merge import _Buf, _get_lib, merge_leaf_responses

    aggs = {"h": {"date_histogram": {"field": "timestamp",
                                     "fixed_interval": "3600000ms"}},
            "t": {"terms": {"field": "tenant_name", "size": 5},
                  "aggs": {"st": {"stats": {"field": "tenant_id"}}}}}
    s = OracleSearcher()
    s.add_split("s", splitgen.generate_split(0, 2000, seed=3))
    req = make_leaf_request({"type": "match_all"}, splitgen.HDFS_SCHEMA,
                            [("s", 2000)], max_hits=5, aggregation=aggs)
    resp_pb = s.leaf_search_raw(proto.encode("LeafSearchRequest", req))
    sreq_pb = proto.encode("SearchRequest", req["search_request"])
    rng = random.Random(17)

    for _ in range(60):
        b = bytearray(resp_pb)
        for _ in range(rng.randrange(1, 5)):
            b[rng.randrange(len(b))] = rng.randrange(256)
        try:
            merge_leaf_responses(sreq_pb, [resp_pb, bytes(b)])
        except Exception:
            pass
    for cut in (0, 1, 7, len(resp_pb) // 2, len(resp_pb) - 2):
        try:
            merge_leaf_responses(sreq_pb, [resp_pb[:cut]])
        except Exception:
            pass

    blob = proto.decode("LeafSearchResponse",
                        resp_pb)["intermediate_aggregation_result"]
    lib = _get_lib()
    lib.qw_finalize_agg_to_json.argtypes = [
        ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p,
        ctypes.POINTER(_Buf)]
    muts = []
    for _ in range(60):
        b = bytearray(blob)
        for _ in range(rng.randrange(1, 5)):
            b[rng.randrange(len(b))] = rng.randrange(256)
        muts.append(bytes(b))
    muts += [blob[:c] for c in (0, 1, 5, len(blob) // 2, len(blob) - 1)]
    for m in muts:
        buf = _Buf()
        if lib.qw_finalize_agg_to_json(m, len(m), json.dumps(aggs).encode(),
                                       ctypes.byref(buf)) == 0:
            lib.qw_buf_free(ctypes.byref(buf))


def header_symbols():
    hdr = open(os.path.join(REPO, "include", "quickwit_amd.h")).read()
    return sorted(set(re.findall(r"\b(qw_[a-z_0-9]+)\s*\(", hdr)) - {"qw_buf"})


def test_exports_every_header_symbol():
    lib = ctypes.CDLL(LIB)
    missing = [s for s in header_symbols() if not hasattr(lib, s)]
    assert not missing, f"symbols declared in quickwit_amd.h but not exported: {missing}"


def test_version_string():
    lib = ctypes.CDLL(LIB)
    lib.qw_version.restype = ctypes.c_char_p
    v = lib.qw_version().decode()
    assert "gfx950" in v


def test_no_gpu_fails_loudly():
    """the product never falls back to CPU."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present; covered by gpu-marked tests")
    lib = ctypes.CDLL(LIB)
    lib.qw_ctx_create.restype = ctypes.c_void_p
    lib.qw_ctx_create.argtypes = [ctypes.c_char_p]
    ctx = lib.qw_ctx_create(b'{"device": 0}')
    assert ctx
    lib.qw_ctx_add_split.argtypes = [ctypes.c_void_p, ctypes.c_char_p,
                                     ctypes.c_char_p, ctypes.c_size_t]
    rc = lib.qw_ctx_add_split(ctx, b"s", b"x" * 128, 128)
    assert rc == -5, f"expected QW_ERR_NO_GPU (-5), got {rc}"
    lib.qw_ctx_free.argtypes = [ctypes.c_void_p]
    lib.qw_ctx_free(ctx)


class _Buf(ctypes.Structure):
    _fields_ = [("data", ctypes.POINTER(ctypes.c_uint8)), ("len", ctypes.c_size_t)]


def _merge(lib, sreq_pb, resp_pbs):
    lib.qw_merge_leaf_responses.argtypes = [
        ctypes.c_char_p, ctypes.c_size_t,
        ctypes.POINTER(ctypes.c_char_p), ctypes.POINTER(ctypes.c_size_t),
        ctypes.c_size_t, ctypes.POINTER(_Buf)]
    arr = (ctypes.c_char_p * len(resp_pbs))(*resp_pbs)
    lens = (ctypes.c_size_t * len(resp_pbs))(*[len(r) for r in resp_pbs])
    buf = _Buf()
    rc = lib.qw_merge_leaf_responses(sreq_pb, len(sreq_pb), arr, lens,
                                     len(resp_pbs), ctypes.byref(buf))
    assert rc == 0
    out = ctypes.string_at(buf.data, buf.len)
    lib.qw_buf_free.argtypes = [ctypes.POINTER(_Buf)]
    lib.qw_buf_free(ctypes.byref(buf))
    return proto.decode("LeafSearchResponse", out)


def test_merge_matches_oracle_cross_split_merge():
    """Merging per-split responses with the product's qw_merge_leaf_responses
    must equal the oracle searching both splits in one call."""
    lib = ctypes.CDLL(LIB)
    schema = splitgen.HDFS_SCHEMA
    datas = [splitgen.generate_split(i, 5_000, seed=7) for i in range(2)]
    sids = [f"synthetic-7-{i:04d}" for i in range(2)]

    both = OracleSearcher()
    singles = [OracleSearcher() for _ in range(2)]
    for i in range(2):
        both.add_split(sids[i], datas[i])
        singles[i].add_split(sids[i], datas[i])

    query = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i} for i in range(3)]}
    aggs = {"per_tenant": {"terms": {"field": "tenant_name", "size": 5}},
            "per_hour": {"date_histogram":
                         {"field": "timestamp", "fixed_interval": "86400000ms"}}}
    kw = dict(max_hits=7, aggregation=aggs,
              sort_fields=[{"field_name": "_score", "sort_order": 1}])
    req_both = make_leaf_request(query, schema,
                                 [(sids[i], 5_000) for i in range(2)], **kw)
    expected = both.leaf_search(req_both)

    resp_pbs = []
    for i in range(2):
        r = make_leaf_request(query, schema, [(sids[i], 5_000)], **kw)
        resp_pbs.append(singles[i].leaf_search_raw(proto.encode("LeafSearchRequest", r)))
    sreq_pb = proto.encode("SearchRequest", req_both["search_request"])
    merged = _merge(lib, sreq_pb, resp_pbs)

    assert merged["num_hits"] == expected["num_hits"]
    assert ([(h["split_id"], h["doc_id"]) for h in merged["partial_hits"]] ==
            [(h["split_id"], h["doc_id"]) for h in expected["partial_hits"]])
    # aggregation blobs merge to the same finalized JSON
    fin_m = _finalize(lib, merged["intermediate_aggregation_result"], aggs)
    fin_e = both.finalize_agg_json(expected["intermediate_aggregation_result"], aggs)
    assert fin_m == fin_e


def _finalize(lib, blob, aggs):
    lib.qw_finalize_agg_to_json.argtypes = [
        ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p, ctypes.POINTER(_Buf)]
    buf = _Buf()
    rc = lib.qw_finalize_agg_to_json(blob, len(blob), json.dumps(aggs).encode(),
                                     ctypes.byref(buf))
    assert rc == 0
    out = ctypes.string_at(buf.data, buf.len)
    lib.qw_buf_free(ctypes.byref(buf))
    return json.loads(out)


def test_gpu_searcher_binding_resolves():
    """GpuSearcher's ctypes binding must resolve every product symbol even
    without a GPU (ctx creation is lazy; add_split then fails with
    QW_ERR_NO_GPU — covered above)."""
    from quickwit_amd.api import GpuSearcher
    import torch
    s = GpuSearcher(device=0)  # binds all symbols; no device touch
    if not torch.cuda.is_available():
        with pytest.raises(RuntimeError, match="-5"):
            s.add_split("s", b"x" * 128)
