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
