"""Aggregation ground truth vs independent Python: date_histogram bucket
counts (floor semantics), cardinality (exact distinct), and composite
after-key pagination (full key-tuple ordering with missing_bucket) are
recomputed from the documents alone and must equal the finalized JSON."""
import random
from collections import Counter

from quickwit_amd import proto, splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

SCHEMA = {"timestamp_field": "ts", "fields": [
    {"name": "ts", "type": "datetime", "fast": True},
    {"name": "svc", "type": "str", "fast": True},
    {"name": "code", "type": "u64", "fast": True},
    {"name": "txt", "type": "text", "tokenizer": "default",
     "record": "basic", "fieldnorms": False}]}

SVCS = ["api", "etl", "web", None]


def _mk(rng, ndocs):
    docs = []
    for _ in range(ndocs):
        d = {"ts": 1_700_000_000 + rng.randrange(0, 50_000),
             "code": rng.choice([200, 204, 404, 500]), "txt": "x"}
        s = rng.choice(SVCS)
        if s is not None:
            d["svc"] = s
        docs.append(d)
    return docs


def _searcher(docs):
    w = splitgen.SplitWriter(SCHEMA, "gt", store_docs=False)
    w.add_documents(docs)
    s = OracleSearcher()
    s.add_split("gt", w.finalize())
    return s


def _agg(s, ndocs, aggs):
    req = make_leaf_request({"type": "match_all"}, SCHEMA, [("gt", ndocs)],
                            max_hits=0, aggregation=aggs)
    resp = proto.decode(
        "LeafSearchResponse",
        s.leaf_search_raw(proto.encode("LeafSearchRequest", req)))
    assert not resp.get("failed_splits"), resp.get("failed_splits")
    return s.finalize_agg_json(resp["intermediate_aggregation_result"],
                               aggs)


def test_date_histogram_and_cardinality_ground_truth():
    rng = random.Random(31)
    for round_i in range(6):
        ndocs = rng.randrange(1, 400)
        docs = _mk(rng, ndocs)
        s = _searcher(docs)
        iv_s = rng.choice([1000, 3600, 7200])  # seconds

        aggs = {"h": {"date_histogram": {"field": "ts",
                                         "fixed_interval":
                                         f"{iv_s * 1000}ms"}},
                "c1": {"cardinality": {"field": "svc"}},
                "c2": {"cardinality": {"field": "code"}}}
        j = _agg(s, ndocs, aggs)

        counts = Counter((d["ts"] // iv_s) * iv_s * 1000 for d in docs)
        # gap-fill: every bucket between min and max key appears
        keys = sorted(counts)
        want = [(float(k), counts.get(k, 0))
                for k in range(keys[0], keys[-1] + 1, iv_s * 1000)]
        got = [(b["key"], b["doc_count"]) for b in j["h"]["buckets"]]
        assert got == want, (round_i, got[:4], want[:4])

        assert j["c1"]["value"] == float(len(
            {d["svc"] for d in docs if "svc" in d}))
        assert j["c2"]["value"] == float(len({d["code"] for d in docs}))


def test_composite_pagination_ground_truth():
    rng = random.Random(37)
    ndocs = 300
    docs = _mk(rng, ndocs)
    s = _searcher(docs)
    base = {"sources": [
        {"s": {"terms": {"field": "svc", "missing_bucket": True}}},
        {"r": {"histogram": {"field": "code", "interval": 100}}}]}

    counts = Counter()
    for d in docs:
        counts[(d.get("svc"), (d["code"] // 100) * 100)] += 1
    # missing bucket (None) sorts FIRST, then key asc
    want = sorted(counts.items(),
                  key=lambda kv: ((kv[0][0] is not None, kv[0][0] or ""),
                                  kv[0][1]))

    got, after = [], None
    for _ in range(20):
        spec = dict(base, size=3)
        if after is not None:
            spec["after"] = after
        j = _agg(s, ndocs, {"c": {"composite": spec}})
        bs = j["c"]["buckets"]
        if not bs:
            break
        for b in bs:
            got.append(((b["key"]["s"], int(b["key"]["r"])), b["doc_count"]))
        ak = j["c"].get("after_key")
        if ak is None:
            break
        after = {"s": ("str:" + ak["s"]) if ak["s"] is not None else None,
                 "r": f'f64:{ak["r"]}'}
    assert got == want, (got[:5], want[:5])
