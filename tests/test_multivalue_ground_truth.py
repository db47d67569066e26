"""Multi-valued str fast columns vs independent Python ground truth:
terms-agg doc_counts count each doc once per DISTINCT value it holds
(sum of bucket doc_counts == sum over docs of |distinct values|, the
semantics the reference's multi-valued columnar aggs produce), term
equality matches membership, and field presence counts docs with >= 1
value."""
import random
from collections import Counter

from quickwit_amd import proto, splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

VALS = ["ap", "db", "web", "cache", "etl"]

# multi-valued columns are declared the way the REST shim's dynamic
# mapping produces them (string arrays): a tokenized text index + a
# multi-valued str fast column under one name (splitgen.py, rest.py:278)
SCHEMA = {"timestamp_field": None, "fields": [
    {"name": "txt", "type": "text", "tokenizer": "default",
     "record": "basic", "fieldnorms": False},
    {"name": "svc", "type": "text", "tokenizer": "default",
     "record": "basic", "fieldnorms": False, "fast": True, "multi": True}]}


def test_multivalued_terms_membership_presence():
    rng = random.Random(23)
    for round_i in range(8):
        ndocs = rng.randrange(1, 300)
        docs = []
        for _ in range(ndocs):
            d = {"txt": "x"}
            r = rng.random()
            if r < 0.2:
                pass  # absent
            elif r < 0.4:
                d["svc"] = rng.choice(VALS)  # scalar form
            else:
                k = rng.randrange(1, 4)
                d["svc"] = [rng.choice(VALS) for _ in range(k)]  # dups ok
            docs.append(d)
        w = splitgen.SplitWriter(SCHEMA, "mv", store_docs=False)
        w.add_documents(docs)
        s = OracleSearcher()
        s.add_split("mv", w.finalize())

        def vals_of(d):
            v = d.get("svc")
            if v is None:
                return set()
            return {v} if isinstance(v, str) else set(v)

        def leaf(q, aggs=None):
            req = make_leaf_request(q, SCHEMA, [("mv", ndocs)], max_hits=0,
                                    aggregation=aggs)
            resp = proto.decode(
                "LeafSearchResponse",
                s.leaf_search_raw(proto.encode("LeafSearchRequest", req)))
            assert not resp.get("failed_splits"), resp.get("failed_splits")
            return resp

        # membership via term equality on the multi-valued column
        for v in VALS:
            want = sum(1 for d in docs if v in vals_of(d))
            got = leaf({"type": "term", "field": "svc",
                        "value": v}).get("num_hits", 0)
            assert got == want, (round_i, v, got, want)

        # presence = docs holding at least one value
        want = sum(1 for d in docs if vals_of(d))
        got = leaf({"type": "field_presence",
                    "field": "svc"}).get("num_hits", 0)
        assert got == want, round_i

        # terms agg: one count per (doc, distinct value)
        aggs = {"t": {"terms": {"field": "svc", "size": 10}}}
        resp = leaf({"type": "match_all"}, aggs)
        j = s.finalize_agg_json(resp["intermediate_aggregation_result"],
                                aggs)
        counts = Counter()
        for d in docs:
            for v in vals_of(d):
                counts[v] += 1
        want_buckets = sorted(counts.items(), key=lambda kv: (-kv[1], kv[0]))
        got_buckets = [(b["key"], b["doc_count"]) for b in j["t"]["buckets"]]
        assert got_buckets == want_buckets, (round_i, got_buckets,
                                             want_buckets)
        assert j["t"]["sum_other_doc_count"] == 0  # size covers all values
