"""Multi-rank merge path on CPU (gloo, world_size=2) — the exact exchange
bench.py runs over RCCL on the 8-GPU node: each rank searches
its own split, then merge.distributed_merge exchanges PACKED TENSORS only
(32 B top-K hit records allgather, dense histogram bucket arrays
SUM/MIN/MAX-reduced, sparse aggs + stats as a sideband byte tensor) and
rank 0 merges through qw_merge_leaf_responses. No pickled objects cross
ranks on the data path. Oracle searchers stand in for the GPU path here
(same protobuf surfaces); nccl/RCCL only swaps the transport."""
import os

import pytest
import torch.multiprocessing as mp

from quickwit_amd import proto, splitgen

NDOCS = 5_000

AGGS = {"per_hour": {"date_histogram": {"field": "timestamp",
                                        "fixed_interval": "3600000ms"}},
        "per_tenant": {"terms": {"field": "tenant_name", "size": 10}},
        "stats_hour": {"date_histogram": {"field": "timestamp",
                                          "fixed_interval": "7200000ms"},
                       "aggs": {"ten": {"stats": {"field": "tenant_id"}}}}}

AGGS2 = {
    "card_t": {"cardinality": {"field": "tenant_name"}},
    "card_ts": {"cardinality": {"field": "timestamp"}},
    "comp": {"composite": {"size": 200, "sources": [
        {"t": {"terms": {"field": "tenant_name"}}},
        {"i": {"histogram": {"field": "tenant_id", "interval": 5}}}]}},
    "pct": {"percentiles": {"field": "tenant_id", "percents": [50, 95]}},
}

CASES = {
    "bm25": dict(
        q={"type": "bool", "should": [
            {"type": "term", "field": "body", "value": "w%05d" % i}
            for i in range(3)]},
        sort=[{"field_name": "_score", "sort_order": 1}], aggs=None),
    "agg": dict(q={"type": "match_all"}, sort=None, aggs=AGGS),
    "agg2": dict(q={"type": "match_all"}, sort=None, aggs=AGGS2),
    "field_sort": dict(
        q={"type": "term", "field": "severity_text", "value": "INFO"},
        sort=[{"field_name": "timestamp", "sort_order": 1},
              {"field_name": "tenant_id", "sort_order": 0}], aggs=None),
}


def _make_req(case, splits):
    from quickwit_amd.api import make_leaf_request
    c = CASES[case]
    return make_leaf_request(c["q"], splitgen.HDFS_SCHEMA, splits, max_hits=15,
                             sort_fields=c["sort"], aggregation=c["aggs"])


def _rank_main(rank, world, case, port, result):
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from quickwit_amd.api import OracleSearcher
    from quickwit_amd.merge import distributed_merge

    sid = f"synthetic-3-{rank:04d}"
    data = splitgen.generate_split(rank, NDOCS, seed=3)
    s = OracleSearcher()
    s.add_split(sid, data)
    req = _make_req(case, [(sid, NDOCS)])
    resp_pb = s.leaf_search_raw(proto.encode("LeafSearchRequest", req))

    sreq_pb = proto.encode("SearchRequest", req["search_request"])
    import torch
    merged = distributed_merge(sreq_pb, resp_pb, [sid],
                               device=torch.device("cpu"))
    if rank == 0:
        result.put(merged)
    dist.destroy_process_group()


def _run_case(case, port):
    import __graft_entry__
    __graft_entry__.build()
    ctx = mp.get_context("spawn")
    result = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, case, port, result))
             for r in range(2)]
    for p in procs:
        p.start()
    merged_pb = result.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    return proto.decode("LeafSearchResponse", merged_pb)


def _single_call_expected(case):
    from quickwit_amd.api import OracleSearcher
    both = OracleSearcher()
    splits = []
    for r in range(2):
        sid = f"synthetic-3-{r:04d}"
        both.add_split(sid, splitgen.generate_split(r, NDOCS, seed=3))
        splits.append((sid, NDOCS))
    return both, both.leaf_search(_make_req(case, splits))


def test_two_rank_packed_merge_bm25_equals_single_call():
    merged = _run_case("bm25", 29511)
    _, expected = _single_call_expected("bm25")
    assert merged["num_hits"] == expected["num_hits"]
    assert ([(h["split_id"], h["doc_id"]) for h in merged["partial_hits"]] ==
            [(h["split_id"], h["doc_id"]) for h in expected["partial_hits"]])
    for mh, eh in zip(merged["partial_hits"], expected["partial_hits"]):
        assert mh["sort_value"] == eh["sort_value"]


def test_two_rank_packed_merge_field_sort_equals_single_call():
    # two-key sort: both sort_value and sort_value2 survive the 32B records
    merged = _run_case("field_sort", 29512)
    _, expected = _single_call_expected("field_sort")
    assert merged["num_hits"] == expected["num_hits"]
    assert ([(h["split_id"], h["doc_id"]) for h in merged["partial_hits"]] ==
            [(h["split_id"], h["doc_id"]) for h in expected["partial_hits"]])
    for mh, eh in zip(merged["partial_hits"], expected["partial_hits"]):
        assert mh.get("sort_value") == eh.get("sort_value")
        assert mh.get("sort_value2") == eh.get("sort_value2")


def test_two_rank_packed_merge_aggs_equal_single_call():
    # date_histogram rides the dense SUM/MIN/MAX reduce (incl. a stats sub);
    # the terms agg rides the sideband + C-ABI merge
    merged = _run_case("agg", 29513)
    both, expected = _single_call_expected("agg")
    gj = both.finalize_agg_json(merged["intermediate_aggregation_result"], AGGS)
    ej = both.finalize_agg_json(expected["intermediate_aggregation_result"],
                                AGGS)
    assert merged["num_hits"] == expected["num_hits"]
    assert gj == ej


def test_two_rank_packed_merge_sideband_aggs_equal_single_call():
    """Cardinality (exact distinct-set union), composite (canonical-key
    merge across split-local ord spaces) and percentiles (DDSketch bucket
    add) are NOT dense-eligible: they cross ranks only through the sideband
    byte tensor and the rank-0 C-ABI merge. Must equal one oracle call
    over both splits."""
    merged = _run_case("agg2", 29515)
    both, expected = _single_call_expected("agg2")
    gj = both.finalize_agg_json(merged["intermediate_aggregation_result"],
                                AGGS2)
    ej = both.finalize_agg_json(expected["intermediate_aggregation_result"],
                                AGGS2)
    assert merged["num_hits"] == expected["num_hits"]
    assert gj == ej
    # sanity on shape: all three kinds actually produced values
    assert gj["card_t"]["value"] > 0
    assert gj["comp"]["buckets"]
    assert "50.0" in gj["pct"]["values"] or "50" in gj["pct"]["values"]


def test_qagg_blob_roundtrip():
    # codec sanity without any ranks: parse+serialize is the identity on a
    # real product blob
    import __graft_entry__
    __graft_entry__.build()
    from quickwit_amd import qagg
    from quickwit_amd.api import OracleSearcher

    s = OracleSearcher()
    s.add_split("s", splitgen.generate_split(0, NDOCS, seed=3))
    resp = s.leaf_search(_make_req("agg", [("s", NDOCS)]))
    blob = resp["intermediate_aggregation_result"]
    entries = qagg.parse_blob(blob)
    assert [e.name for e in entries] == ["per_hour", "per_tenant", "stats_hour"]
    assert entries[0].dense_eligible and entries[2].dense_eligible
    assert not entries[1].dense_eligible  # terms
    assert qagg.serialize_blob(entries) == blob
    # bucket re-encode (not raw passthrough) is also the identity
    for e in entries:
        if not e.dense_eligible:
            e.buckets = None
    assert qagg.serialize_blob(entries) == blob


def _rank_main_multiseg(rank, world, port, result):
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import torch

    from quickwit_amd.api import OracleSearcher
    from quickwit_amd.merge import distributed_merge

    # each rank owns a 2-segment QWA2 split
    a = splitgen.generate_split(10 * rank, 4000, seed=5)
    b = splitgen.generate_split(10 * rank + 1, 3000, seed=5)
    sid = f"ms-{rank}"
    s = OracleSearcher()
    s.add_split(sid, splitgen.concat_segments([a, b], sid))
    req = _make_req("bm25", [(sid, 7000)])
    resp_pb = s.leaf_search_raw(proto.encode("LeafSearchRequest", req))
    sreq_pb = proto.encode("SearchRequest", req["search_request"])
    merged = distributed_merge(sreq_pb, resp_pb, [sid],
                               device=torch.device("cpu"))
    if rank == 0:
        result.put(merged)
    dist.destroy_process_group()


PHRASE_SCHEMA = {"timestamp_field": None, "fields": [
    {"name": "body", "type": "text", "tokenizer": "default",
     "record": "position", "fieldnorms": True}]}
PHRASE_SCHEMA_NOPOS = {"timestamp_field": None, "fields": [
    {"name": "body", "type": "text", "tokenizer": "default",
     "record": "freq", "fieldnorms": True}]}


def _phrase_split(rank):
    schema = PHRASE_SCHEMA if rank == 0 else PHRASE_SCHEMA_NOPOS
    w = splitgen.SplitWriter(schema, f"pf-{rank}", store_docs=False)
    w.add_documents([{"body": "alpha beta"}, {"body": "beta alpha"}])
    return w.finalize()


def _rank_main_failure(rank, world, port, result):
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import torch

    from quickwit_amd.api import OracleSearcher, make_leaf_request
    from quickwit_amd.merge import distributed_merge

    sid = f"pf-{rank}"
    s = OracleSearcher()
    s.add_split(sid, _phrase_split(rank))
    q = {"type": "full_text", "field": "body", "text": "alpha beta",
         "params": {"mode": {"type": "phrase"}}}
    req = make_leaf_request(q, PHRASE_SCHEMA, [(sid, 2)], max_hits=10)
    resp_pb = s.leaf_search_raw(proto.encode("LeafSearchRequest", req))
    sreq_pb = proto.encode("SearchRequest", req["search_request"])
    merged = distributed_merge(sreq_pb, resp_pb, [sid],
                               device=torch.device("cpu"))
    if rank == 0:
        result.put(merged)
    dist.destroy_process_group()


def test_two_rank_merge_carries_failed_splits_as_data():
    """Per-split failures are DATA inside the response (leaf.rs:2143-2148),
    and the packed exchange must carry them across ranks through the
    sideband bytes: rank 1's positions-less split fails the phrase query,
    rank 0's split answers — the merged response reports both, exactly
    like one call over both splits."""
    import __graft_entry__
    __graft_entry__.build()
    ctx = mp.get_context("spawn")
    result = ctx.Queue()
    procs = [ctx.Process(target=_rank_main_failure, args=(r, 2, 29516, result))
             for r in range(2)]
    for p in procs:
        p.start()
    merged = proto.decode("LeafSearchResponse", result.get(timeout=180))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    from quickwit_amd.api import OracleSearcher, make_leaf_request
    both = OracleSearcher()
    for r in range(2):
        both.add_split(f"pf-{r}", _phrase_split(r))
    q = {"type": "full_text", "field": "body", "text": "alpha beta",
         "params": {"mode": {"type": "phrase"}}}
    expected = both.leaf_search(make_leaf_request(
        q, PHRASE_SCHEMA, [("pf-0", 2), ("pf-1", 2)], max_hits=10))

    assert merged.get("num_hits") == expected.get("num_hits") == 1
    assert ([h["split_id"] for h in merged["partial_hits"]] ==
            [h["split_id"] for h in expected["partial_hits"]] == ["pf-0"])
    mf = merged.get("failed_splits", [])
    ef = expected.get("failed_splits", [])
    assert [f["split_id"] for f in mf] == [f["split_id"] for f in ef] == \
        ["pf-1"]
    assert merged.get("num_successful_splits") == \
        expected.get("num_successful_splits")
    assert merged.get("num_attempted_splits") == \
        expected.get("num_attempted_splits")


def _rank_main_agg_failure(rank, world, port, result):
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    import torch

    from quickwit_amd.api import OracleSearcher, make_leaf_request
    from quickwit_amd.merge import distributed_merge

    s = OracleSearcher()
    s.add_split("af-0", splitgen.generate_split(0, NDOCS, seed=3))
    # rank 1 queries a split its ctx does not hold: per-split failure,
    # EMPTY agg blob — the dense reduce must not deadlock on the shape
    # mismatch and falls back to the sideband for every rank
    sid = "af-0" if rank == 0 else "af-missing"
    req = make_leaf_request({"type": "match_all"}, splitgen.HDFS_SCHEMA,
                            [(sid, NDOCS)], max_hits=5, aggregation=AGGS)
    resp_pb = s.leaf_search_raw(proto.encode("LeafSearchRequest", req))
    sreq_pb = proto.encode("SearchRequest", req["search_request"])
    merged = distributed_merge(sreq_pb, resp_pb, [sid],
                               device=torch.device("cpu"))
    if rank == 0:
        result.put(merged)
    dist.destroy_process_group()


def test_two_rank_merge_with_one_empty_agg_blob_does_not_deadlock():
    """Rank 1's only split fails, so its response carries NO aggregation
    blob while rank 0's does. The dense bucket reduce derives its tensor
    shape from the local blob; without the structure-agreement reduce
    this mismatched collective deadlocked. Merged result == rank 0's
    aggs + rank 1's failure as data."""
    import __graft_entry__
    __graft_entry__.build()
    ctx = mp.get_context("spawn")
    result = ctx.Queue()
    procs = [ctx.Process(target=_rank_main_agg_failure,
                         args=(r, 2, 29517, result)) for r in range(2)]
    for p in procs:
        p.start()
    merged = proto.decode("LeafSearchResponse", result.get(timeout=180))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    from quickwit_amd.api import OracleSearcher, make_leaf_request
    solo = OracleSearcher()
    solo.add_split("af-0", splitgen.generate_split(0, NDOCS, seed=3))
    expected = solo.leaf_search(make_leaf_request(
        {"type": "match_all"}, splitgen.HDFS_SCHEMA, [("af-0", NDOCS)],
        max_hits=5, aggregation=AGGS))
    assert merged["num_hits"] == expected["num_hits"]
    assert [f["split_id"] for f in merged.get("failed_splits", [])] == \
        ["af-missing"]
    gj = solo.finalize_agg_json(merged["intermediate_aggregation_result"],
                                AGGS)
    ej = solo.finalize_agg_json(expected["intermediate_aggregation_result"],
                                AGGS)
    assert gj == ej


def test_two_rank_packed_merge_multisegment_splits():
    """Packed 32B hit records carry (split_ord, segment_ord) across ranks:
    two ranks each holding a 2-segment QWA2 split must merge exactly like
    one oracle call over all four segments' splits."""
    import __graft_entry__
    __graft_entry__.build()
    ctx = mp.get_context("spawn")
    result = ctx.Queue()
    procs = [ctx.Process(target=_rank_main_multiseg,
                         args=(r, 2, 29514, result)) for r in range(2)]
    for p in procs:
        p.start()
    merged = proto.decode("LeafSearchResponse", result.get(timeout=180))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    from quickwit_amd.api import OracleSearcher
    both = OracleSearcher()
    splits = []
    for r in range(2):
        a = splitgen.generate_split(10 * r, 4000, seed=5)
        b = splitgen.generate_split(10 * r + 1, 3000, seed=5)
        sid = f"ms-{r}"
        both.add_split(sid, splitgen.concat_segments([a, b], sid))
        splits.append((sid, 7000))
    expected = both.leaf_search(_make_req("bm25", splits))
    assert merged["num_hits"] == expected["num_hits"]
    assert ([(h["split_id"], h.get("segment_ord", 0), h["doc_id"])
             for h in merged["partial_hits"]] ==
            [(h["split_id"], h.get("segment_ord", 0), h["doc_id"])
             for h in expected["partial_hits"]])
