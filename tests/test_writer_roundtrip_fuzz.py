"""Randomized writer→search round trip against INDEPENDENT Python ground
truth: random schemas and documents go through SplitWriter, and the
oracle's answers (term counts, range counts, terms-agg buckets, stats)
must equal values computed directly from the documents in Python. This
checks the whole container encode/decode + query path with an oracle that
is NOT the implementation under test (unlike the GPU fuzz, which compares
the two engines against each other)."""
import math
import random
from collections import Counter

from quickwit_amd import proto, splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

WORDS = ["alpha", "beta", "gamma", "delta", "epsilon", "zeta", "eta",
         "theta", "iota", "kappa"]
STRS = ["red", "green", "blue", "cyan", "magenta", None]


def _corpus(rng, ndocs):
    docs = []
    for _ in range(ndocs):
        d = {"txt": " ".join(rng.choice(WORDS)
                             for _ in range(rng.randrange(0, 8)))}
        if rng.random() < 0.9:
            d["num"] = rng.randrange(-50, 51)
        if rng.random() < 0.8:
            s = rng.choice(STRS)
            if s is not None:
                d["tag"] = s
        if rng.random() < 0.7:
            d["val"] = round(rng.uniform(-5.0, 5.0), 3)
        docs.append(d)
    return docs


SCHEMA = {"timestamp_field": None, "fields": [
    {"name": "txt", "type": "text", "tokenizer": "default",
     "record": "freq", "fieldnorms": True},
    {"name": "num", "type": "i64", "fast": True},
    {"name": "tag", "type": "str", "fast": True},
    {"name": "val", "type": "f64", "fast": True}]}


def test_random_corpora_vs_python_ground_truth():
    rng = random.Random(20260915)
    for round_i in range(12):
        ndocs = rng.randrange(1, 400)
        docs = _corpus(rng, ndocs)
        w = splitgen.SplitWriter(SCHEMA, "rt", store_docs=False)
        w.add_documents(docs)
        s = OracleSearcher()
        s.add_split("rt", w.finalize())

        def hits(q, aggs=None):
            req = make_leaf_request(q, SCHEMA, [("rt", ndocs)],
                                    max_hits=0, aggregation=aggs)
            resp = proto.decode(
                "LeafSearchResponse",
                s.leaf_search_raw(proto.encode("LeafSearchRequest", req)))
            assert not resp.get("failed_splits"), resp.get("failed_splits")
            return resp

        # term counts == python containment counts
        for wd in rng.sample(WORDS, 4):
            want = sum(1 for d in docs if wd in d["txt"].split())
            got = hits({"type": "term", "field": "txt",
                        "value": wd}).get("num_hits", 0)
            assert got == want, (round_i, wd, got, want)

        # i64 range counts (negative bounds exercise signed handling)
        lo = rng.randrange(-60, 40)
        hi = lo + rng.randrange(0, 60)
        want = sum(1 for d in docs if "num" in d and lo <= d["num"] <= hi)
        got = hits({"type": "range", "field": "num",
                    "lower_bound": {"included": lo},
                    "upper_bound": {"included": hi}}).get("num_hits", 0)
        assert got == want, (round_i, lo, hi)

        # str term equality via the fast column
        tag = rng.choice([t for t in STRS if t])
        want = sum(1 for d in docs if d.get("tag") == tag)
        got = hits({"type": "term", "field": "tag",
                    "value": tag}).get("num_hits", 0)
        assert got == want, (round_i, tag)

        # terms agg buckets == python Counter (count desc, key asc ties)
        resp = hits({"type": "match_all"},
                    aggs={"t": {"terms": {"field": "tag", "size": 10}}})
        j = s.finalize_agg_json(resp["intermediate_aggregation_result"],
                                {"t": {"terms": {"field": "tag",
                                                 "size": 10}}})
        counts = Counter(d["tag"] for d in docs if "tag" in d)
        want_buckets = sorted(counts.items(), key=lambda kv: (-kv[1], kv[0]))
        got_buckets = [(b["key"], b["doc_count"]) for b in j["t"]["buckets"]]
        assert got_buckets == want_buckets, (round_i, got_buckets,
                                             want_buckets)

        # stats over f64 == python (within fp tolerance)
        resp = hits({"type": "match_all"},
                    aggs={"m": {"stats": {"field": "val"}}})
        j = s.finalize_agg_json(resp["intermediate_aggregation_result"],
                                {"m": {"stats": {"field": "val"}}})
        vals = [d["val"] for d in docs if "val" in d]
        st = j["m"]
        assert st["count"] == len(vals), round_i
        if vals:
            assert math.isclose(st["sum"], sum(vals), rel_tol=1e-9,
                                abs_tol=1e-9), round_i
            assert st["min"] == min(vals) and st["max"] == max(vals), round_i

        # field presence == python key counts
        for fld in ("num", "tag", "val"):
            want = sum(1 for d in docs if fld in d)
            got = hits({"type": "field_presence",
                        "field": fld}).get("num_hits", 0)
            assert got == want, (round_i, fld)
