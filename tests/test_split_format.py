"""QWA1 writer/reader round-trip tests (CPU-only)."""
import numpy as np
import pytest

from quickwit_amd import splitgen
from quickwit_amd.fieldnorm import FIELDNORM_TABLE, id_to_norm, norm_to_id
from quickwit_amd.splitread import Split


def test_fieldnorm_table_monotone_identity():
    assert list(FIELDNORM_TABLE[:16]) == list(range(16))
    assert all(FIELDNORM_TABLE[i] < FIELDNORM_TABLE[i + 1] for i in range(255))
    # floor rounding
    assert norm_to_id([0, 1, 15, 16, 17, 100, 10**6]).tolist() == [
        0, 1, 15, 16, 16, norm_to_id([100])[0], norm_to_id([10**6])[0]]
    assert id_to_norm(norm_to_id([100]))[0] <= 100
    # round-trip on exactly representable values
    for i in range(256):
        assert norm_to_id([FIELDNORM_TABLE[i]])[0] == i


SCHEMA = {
    "timestamp_field": None,
    "fields": [
        {"name": "body", "type": "text", "tokenizer": "default", "record": "freq",
         "fieldnorms": True},
        {"name": "sev", "type": "text", "tokenizer": "raw", "record": "basic",
         "fieldnorms": True},
        {"name": "num", "type": "u64", "fast": True},
        {"name": "when", "type": "datetime", "fast": True},
        {"name": "who", "type": "str", "fast": True},
    ],
}


def test_small_split_roundtrip():
    docs = [
        {"body": "the cat sat", "sev": "ERROR", "num": 7, "when": "2015-01-01T12:10:30Z",
         "who": "bob"},
        {"body": "the the THE dog", "sev": "INFO", "num": 3, "who": "alice"},
        {"body": "cat", "num": 9, "when": 1420070400, "who": "bob"},
        {"sev": "ERROR"},
    ]
    w = splitgen.SplitWriter(SCHEMA, "s1")
    w.add_documents(docs)
    s = Split(w.finalize())
    assert s.meta["num_docs"] == 4

    assert s.terms("body") == ["cat", "dog", "sat", "the"]
    d, tf = s.postings("body", "the")
    assert d.tolist() == [0, 1] and tf.tolist() == [1, 3]
    d, tf = s.postings("body", "cat")
    assert d.tolist() == [0, 2] and tf.tolist() == [1, 1]
    d, tf = s.postings("body", "missing")
    assert len(d) == 0

    # raw tokenizer: no lowercasing, record basic -> tf == 1
    d, tf = s.postings("sev", "ERROR")
    assert d.tolist() == [0, 3] and tf.tolist() == [1, 1]

    # fieldnorms = token counts (identity region)
    assert s.fieldnorms("body").tolist() == [3, 4, 1, 0]

    vals, present = s.fast_column("num")
    assert vals.tolist() == [7, 3, 9, 0]
    assert present.tolist() == [True, True, True, False]

    vals, present = s.fast_column("when")
    assert vals[0] == 1420114230000 and vals[2] == 1420070400000
    assert present.tolist() == [True, False, True, False]

    assert s.str_dict("who") == ["alice", "bob"]
    vals, present = s.fast_column("who")
    assert vals.tolist() == [1, 0, 1, 0]
    assert present.tolist() == [True, True, True, False]


def test_bulk_matches_doc_writer():
    rng = np.random.default_rng(0)
    n = 5000
    mat = rng.integers(0, 50, size=(n, 7)).astype(np.int32)
    vocab = ["tok%02d" % i for i in range(50)]
    schema = {"timestamp_field": None, "fields": [
        {"name": "body", "type": "text", "tokenizer": "default", "record": "freq",
         "fieldnorms": True}]}
    bulk = splitgen.build_split_from_columns(
        schema, "x", n, {"body": mat}, {"body": vocab}, {})
    docs = [{"body": " ".join(vocab[t] for t in row)} for row in mat]
    w = splitgen.SplitWriter(schema, "x", store_docs=False)
    w.add_documents(docs)
    slow = w.finalize()
    assert bulk == slow


def test_multiblock_postings():
    # one term appearing in >128 docs exercises multi-block packing
    n = 1000
    docs = [{"body": "common" + (" rare" if i % 37 == 0 else "")} for i in range(n)]
    w = splitgen.SplitWriter(
        {"timestamp_field": None, "fields": [
            {"name": "body", "type": "text", "tokenizer": "default", "record": "freq",
             "fieldnorms": True}]}, "m")
    w.add_documents(docs)
    s = Split(w.finalize())
    d, tf = s.postings("body", "common")
    assert d.tolist() == list(range(n))
    d, tf = s.postings("body", "rare")
    assert d.tolist() == list(range(0, n, 37))


def test_synthetic_generator_deterministic():
    a = splitgen.generate_split(0, 2000, seed=42)
    b = splitgen.generate_split(0, 2000, seed=42)
    assert a == b
    c = splitgen.generate_split(1, 2000, seed=42)
    assert c != a
    s = Split(a)
    assert s.meta["num_docs"] == 2000
    d, _ = s.postings("severity_text", "ERROR")
    assert 20 < len(d) < 140  # ~2.9%
    ts, pres = s.fast_column("timestamp")
    assert pres is None
    assert ts.min() >= splitgen.T0_EPOCH_S * 1000


def test_native_generator_matches_independent_readers():
    """The OpenMP generator (csrc/splitgen_native.cpp) must be readable
    identically by the independent numpy reader and the C++ oracle: doc ids
    sorted, df/count/bm25 agreement, severity/tenant distributions sane."""
    import math

    from quickwit_amd.api import OracleSearcher, make_leaf_request
    from quickwit_amd.fieldnorm import id_to_norm

    if splitgen._native_gen() is None:
        pytest.skip("libqwsplitgen.so not built")
    N = 150_000
    data = splitgen.generate_split(3, N, seed=7)
    sp = Split(data)
    assert sp.meta["num_docs"] == N
    assert sp.meta["split_id"] == "synthetic-7-0003"

    terms = ["w%05d" % i for i in (0, 9, 10, 11, 9999)]
    posts = {t: sp.postings("body", t) for t in terms}
    for t, (d, tf) in posts.items():
        assert (np.diff(d.astype(np.int64)) > 0).all()
        assert tf.min() >= 1
    # df close to N*q_t (Bernoulli margin 5 sigma)
    p = splitgen.zipf_probs(splitgen.BODY_VOCAB_SIZE)
    for rank in (0, 9, 10, 11):
        q = 1 - (1 - p[rank]) ** splitgen.BODY_TOKENS_PER_DOC
        df = len(posts["w%05d" % rank][0])
        assert abs(df - N * q) < 5 * math.sqrt(N * q * (1 - q)) + 10, rank

    # numpy-reader BM25 vs the oracle on the same bytes
    norms = sp.fieldnorms("body")
    assert (norms == 10).all()
    dl = float(id_to_norm(np.array([10]))[0])
    avgdl = sp.fields["body"]["total_tokens"] / N
    score = np.zeros(N, np.float64)
    k1, b = 1.2, 0.75
    for t in ["w%05d" % i for i in (9, 10, 11)]:
        d, tf = posts[t]
        idf = math.log(1 + (N - len(d) + 0.5) / (len(d) + 0.5))
        K = np.float32(k1 * (1 - b + b * dl / avgdl))
        tff = tf.astype(np.float32)
        np.add.at(score, d, (np.float32(idf * (k1 + 1)) * tff / (tff + K)).astype(np.float32))
    cpu = OracleSearcher()
    cpu.add_split("s", data)
    q3 = {"type": "bool", "should": [
        {"type": "term", "field": "body", "value": "w%05d" % i} for i in (9, 10, 11)]}
    req = make_leaf_request(q3, splitgen.HDFS_SCHEMA, [("s", N)], max_hits=5,
                            sort_fields=[{"field_name": "_score", "sort_order": 1}])
    r = cpu.leaf_search(req)
    assert r["num_hits"] == int((score > 0).sum())
    best = np.argsort(-score, kind="stable")[:5]
    for h, nd in zip(r["partial_hits"], best):
        assert math.isclose(h["sort_value"]["f64"], score[nd],
                            rel_tol=1e-5, abs_tol=1e-9)

    # severity postings partition the doc space
    sev_dfs = sp._sec("severity_text", "doc_freq", "<u4")
    assert int(sev_dfs.sum()) == N
    info, _ = sp.postings("severity_text", "INFO")
    r2 = cpu.leaf_search(make_leaf_request(
        {"type": "term", "field": "severity_text", "value": "INFO"},
        splitgen.HDFS_SCHEMA, [("s", N)], max_hits=0))
    assert r2["num_hits"] == len(info)

    # fast columns: range count via reader == oracle
    ten, present = sp.fast_column("tenant_id")
    assert present is None
    r3 = cpu.leaf_search(make_leaf_request(
        {"type": "range", "field": "tenant_id",
         "lower_bound": {"included": 100}, "upper_bound": {"excluded": 300}},
        splitgen.HDFS_SCHEMA, [("s", N)], max_hits=0))
    assert r3["num_hits"] == int(((ten >= 100) & (ten < 300)).sum())

    # determinism
    assert splitgen.generate_split(3, 50_000, seed=7) == \
        splitgen.generate_split(3, 50_000, seed=7)
