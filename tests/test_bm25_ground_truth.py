"""Independent Python BM25 vs the oracle over random corpora: scores are
recomputed here from the documents alone (Lucene SmallFloat int4 fieldnorm
quantization + the golden-pinned BM25 formula, tests.rs:600-691) and must
match the oracle's _score sort values. Also pins multi-field sort order
and search_after pagination against plain Python sorting."""
import math
import random

from quickwit_amd import splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

WORDS = ["alpha", "beta", "gamma", "delta", "epsilon", "zeta"]

SCHEMA = {"timestamp_field": None, "fields": [
    {"name": "txt", "type": "text", "tokenizer": "default",
     "record": "freq", "fieldnorms": True},
    {"name": "num", "type": "i64", "fast": True}]}


def fieldnorm_id(length):
    # Lucene SmallFloat int4: identity < 8, then 3-bit mantissa + implicit
    # bit << exponent; encode = largest id whose decode <= length
    def decode(i):
        return i if i < 8 else ((i & 7) | 8) << ((i >> 3) - 1)
    lo = 0
    for i in range(256):
        if decode(i) <= length:
            lo = i
    return lo, decode(lo)


def py_bm25(docs, query_words):
    """Per-doc BM25 sum over query words, f32 per-posting arithmetic like
    both engines (W in f64 then f32; tf/(tf+K) in f32)."""
    import numpy as np
    n = len(docs)
    toks = [d["txt"].split() for d in docs]
    total = sum(len(t) for t in toks)
    avgdl = total / n if n else 0.0
    scores = [0.0] * n
    for wd in query_words:
        df = sum(1 for t in toks if wd in t)
        if df == 0:
            continue
        idf = math.log(1.0 + (n - df + 0.5) / (df + 0.5))
        w = np.float32(idf * 2.2)
        for i, t in enumerate(toks):
            tf = t.count(wd)
            if not tf:
                continue
            _, quant_len = fieldnorm_id(len(t))
            k = np.float32(1.2 * (0.25 + 0.75 * quant_len /
                                  (avgdl if avgdl > 0 else 1.0)))
            tf32 = np.float32(tf)
            scores[i] += float(w * (tf32 / (tf32 + k)))
    return scores


def test_bm25_scores_match_independent_python():
    rng = random.Random(7)
    for round_i in range(8):
        ndocs = rng.randrange(2, 300)
        docs = [{"txt": " ".join(rng.choice(WORDS)
                                 for _ in range(rng.randrange(1, 30))),
                 "num": rng.randrange(100)} for _ in range(ndocs)]
        w = splitgen.SplitWriter(SCHEMA, "bm", store_docs=False)
        w.add_documents(docs)
        s = OracleSearcher()
        s.add_split("bm", w.finalize())
        qwords = rng.sample(WORDS, rng.randrange(1, 4))
        q = {"type": "bool", "should": [
            {"type": "term", "field": "txt", "value": x} for x in qwords]}
        r = s.leaf_search(make_leaf_request(
            q, SCHEMA, [("bm", ndocs)], max_hits=ndocs,
            sort_fields=[{"field_name": "_score", "sort_order": 1}]))
        want = py_bm25(docs, qwords)
        matching = {i for i, sc in enumerate(want) if sc > 0}
        got = {h.get("doc_id", 0): h["sort_value"]["f64"]
               for h in r.get("partial_hits", [])}
        assert set(got) == matching, (round_i, qwords)
        for d, sc in got.items():
            assert math.isclose(sc, want[d], rel_tol=1e-5, abs_tol=1e-7), \
                (round_i, d, sc, want[d])


def test_two_field_sort_and_pagination_match_python():
    rng = random.Random(11)
    ndocs = 250
    docs = [{"txt": rng.choice(WORDS)} for _ in range(ndocs)]
    for d in docs:
        if rng.random() < 0.8:
            d["num"] = rng.randrange(0, 20)  # many ties + some missing
    w = splitgen.SplitWriter(SCHEMA, "so", store_docs=False)
    w.add_documents(docs)
    s = OracleSearcher()
    s.add_split("so", w.finalize())

    # reference order (sorting.md:14-26): primary num DESC with None LAST,
    # tie-break doc_id in the primary sort direction (descending)
    def key(i):
        v = docs[i].get("num")
        return (0 if v is not None else 1,
                -(v if v is not None else 0), -i)
    want = sorted(range(ndocs), key=key)

    page, cursor, got = 40, None, []
    for _ in range(10):
        req = make_leaf_request({"type": "match_all"}, SCHEMA,
                                [("so", ndocs)], max_hits=page,
                                sort_fields=[{"field_name": "num",
                                              "sort_order": 1}])
        if cursor is not None:
            req["search_request"]["search_after"] = cursor
        r = s.leaf_search(req)
        hits = r.get("partial_hits", [])
        if not hits:
            break
        got.extend(h.get("doc_id", 0) for h in hits)
        cursor = hits[-1]
    assert got == want
