"""Multi-token phrase queries (slop 0) on the CPU oracle, pinned against a
python brute force over the raw token lists (PhraseQuery semantics restated
from full_text_query.rs:113-137's phrase mode; positions = token index from
the tokenizer). Phrases are const-score: rejected under _score sorting."""
import random

import pytest

from quickwit_amd import proto, splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

SCHEMA = {"timestamp_field": None, "fields": [
    {"name": "body", "type": "text", "tokenizer": "default",
     "record": "position", "fieldnorms": True}]}
VOCAB = ["alpha", "beta", "gamma", "delta", "eps", "zeta"]
NDOCS = 700


@pytest.fixture(scope="module", autouse=True)
def build_all():
    import __graft_entry__
    __graft_entry__.build()


def corpus():
    rng = random.Random(1234)
    return [" ".join(rng.choice(VOCAB) for _ in range(rng.randint(3, 12)))
            for _ in range(NDOCS)]


@pytest.fixture(scope="module")
def setup():
    docs = corpus()
    w = splitgen.SplitWriter(SCHEMA, "ph", store_docs=False)
    w.add_documents([{"body": d} for d in docs])
    data = w.finalize()
    cpu = OracleSearcher()
    cpu.add_split("ph", data)
    return docs, data, cpu


def brute_phrase(docs, toks):
    out = []
    for d, text in enumerate(docs):
        t = text.split()
        if any(t[p:p + len(toks)] == toks
               for p in range(len(t) - len(toks) + 1)):
            out.append(d)
    return out


def phrase_query(toks):
    return {"type": "full_text", "field": "body", "text": " ".join(toks),
            "params": {"mode": {"type": "phrase"}}}


PHRASES = [["alpha", "beta"], ["beta", "beta"], ["gamma", "delta", "eps"],
           ["zeta", "alpha", "zeta"], ["eps", "eps", "eps", "eps"],
           ["alpha", "alpha"], ["delta", "gamma", "beta", "alpha"]]


@pytest.mark.parametrize("toks", PHRASES, ids=lambda t: "_".join(t))
def test_phrase_matches_brute_force(setup, toks):
    docs, _data, cpu = setup
    exp = brute_phrase(docs, toks)
    r = cpu.leaf_search(make_leaf_request(
        phrase_query(toks), SCHEMA, [("ph", NDOCS)], max_hits=NDOCS))
    got = sorted(h.get("doc_id", 0) for h in r.get("partial_hits", []))
    assert r.get("num_hits", 0) == len(exp), toks
    assert got == exp, toks


def test_phrase_absent_token_matches_nothing(setup):
    _docs, _data, cpu = setup
    r = cpu.leaf_search(make_leaf_request(
        phrase_query(["alpha", "missingtoken"]), SCHEMA, [("ph", NDOCS)],
        max_hits=5))
    assert r.get("num_hits", 0) == 0


def test_phrase_in_bool_and_must_not(setup):
    docs, _data, cpu = setup
    ph = brute_phrase(docs, ["alpha", "beta"])
    with_g = [d for d in range(NDOCS) if "gamma" in docs[d].split()]
    q = {"type": "bool",
         "filter": [phrase_query(["alpha", "beta"])],
         "must": [{"type": "term", "field": "body", "value": "gamma"}]}
    r = cpu.leaf_search(make_leaf_request(q, SCHEMA, [("ph", NDOCS)],
                                          max_hits=NDOCS))
    exp = sorted(set(ph) & set(with_g))
    assert r.get("num_hits", 0) == len(exp)
    q2 = {"type": "bool",
          "must": [{"type": "term", "field": "body", "value": "gamma"}],
          "must_not": [phrase_query(["alpha", "beta"])]}
    r2 = cpu.leaf_search(make_leaf_request(q2, SCHEMA, [("ph", NDOCS)],
                                           max_hits=0))
    assert r2.get("num_hits", 0) == len(set(with_g) - set(ph))


def test_phrase_under_score_sort_rejected(setup):
    _docs, _data, cpu = setup
    req = make_leaf_request(phrase_query(["alpha", "beta"]), SCHEMA,
                            [("ph", NDOCS)], max_hits=5,
                            sort_fields=[{"field_name": "_score",
                                          "sort_order": 1}])
    r = cpu.leaf_search(req)
    assert r.get("failed_splits"), r  # const-score: rejected, not mis-scored


def test_phrase_needs_positions():
    schema = {"timestamp_field": None, "fields": [
        {"name": "body", "type": "text", "tokenizer": "default",
         "record": "freq", "fieldnorms": True}]}
    w = splitgen.SplitWriter(schema, "nf", store_docs=False)
    w.add_documents([{"body": "alpha beta"}])
    cpu = OracleSearcher()
    cpu.add_split("nf", w.finalize())
    r = cpu.leaf_search(make_leaf_request(
        phrase_query(["alpha", "beta"]), schema, [("nf", 1)], max_hits=5))
    assert r.get("failed_splits")
    assert "position" in r["failed_splits"][0]["error"]


def test_query_grammar_quoted_phrase(setup):
    docs, _data, cpu = setup
    exp = brute_phrase(docs, ["gamma", "delta"])
    q = {"type": "user_input", "user_text": 'body:"gamma delta"',
         "default_fields": ["body"]}
    r = cpu.leaf_search(make_leaf_request(q, SCHEMA, [("ph", NDOCS)],
                                          max_hits=NDOCS))
    assert r.get("num_hits", 0) == len(exp)


def test_rest_match_phrase_end_to_end(setup):
    from fastapi.testclient import TestClient

    from quickwit_amd.rest import create_app
    docs, _data, _cpu = setup
    client = TestClient(create_app(OracleSearcher))
    r = client.post("/api/v1/indexes", json={
        "version": "0.7", "index_id": "ph",
        "doc_mapping": {"field_mappings": [
            {"name": "body", "type": "text", "record": "position"}]}})
    assert r.status_code == 200, r.text
    ndjson = "\n".join(
        __import__("json").dumps({"body": d}) for d in docs)
    r = client.post("/api/v1/ph/ingest", content=ndjson)
    assert r.status_code == 200, r.text
    r = client.post("/api/v1/_elastic/ph/_search", json={
        "query": {"match_phrase": {"body": "alpha beta"}}, "size": 0})
    assert r.status_code == 200, r.text
    assert r.json()["hits"]["total"]["value"] == \
        len(brute_phrase(docs, ["alpha", "beta"]))
