"""Unicode tokenization agreement between the split writer (Python,
quickwit_amd/unicode_tables.py) and the C++ query compiler
(csrc/qw_unicode.h) — both GENERATED from one script
(tools/gen_unicode_tables.py), restating tantivy's default chain
SimpleTokenizer [char::is_alphanumeric] + RemoveLongFilter(40 bytes) +
LowerCaser. A query tokenized differently from the index is the bug class
this pins (full_text/phrase over non-ASCII text returned 0 hits before
round 2's fix)."""
import random

import pytest

from quickwit_amd import splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

SCHEMA = {"timestamp_field": None, "fields": [
    {"name": "body", "type": "text", "tokenizer": "default",
     "record": "position", "fieldnorms": True}]}


def _search(body_docs, query_text, mode="phrase", searcher=None):
    w = splitgen.SplitWriter(SCHEMA, "u", store_docs=False)
    w.add_documents([{"body": b} for b in body_docs])
    s = searcher or OracleSearcher()
    s.add_split("u", w.finalize())
    q = {"type": "full_text", "field": "body", "text": query_text,
         "params": {"mode": {"type": mode}}}
    r = s.leaf_search(make_leaf_request(q, SCHEMA,
                                        [("u", len(body_docs))], max_hits=10))
    return r.get("num_hits", 0)


def test_unicode_phrase_and_case_folding():
    assert _search(["héllo wörld héllo"], "héllo wörld") == 1
    assert _search(["héllo wörld héllo"], "HÉLLO WÖRLD") == 1
    assert _search(["Привет мир"], "привет мир") == 1
    assert _search(["日本語のテキスト"], "日本語のテキスト") == 1
    assert _search(["Größe"], "größe", mode="bool") == 1
    # unicode punctuation splits on both sides: the phrase needle
    # "foo…bar" tokenizes to [foo, bar] and matches the adjacent pair
    assert _search(["foo…bar baz"], "foo…bar") == 1
    assert _search(["foo bar… baz"], "foo…bar") == 1
    assert _search(["foo baz bar"], "foo…bar") == 0


def test_long_token_filter_counts_utf8_bytes():
    # 14 CJK chars = 42 UTF-8 bytes > 40: dropped on BOTH sides
    # (tantivy's RemoveLongFilter limits byte length)
    long_tok = "語" * 14
    assert len(long_tok.encode()) > 40
    assert splitgen.tokenize(long_tok, "default") == []
    assert _search([long_tok + " short"], long_tok, mode="bool") == 0
    assert _search([long_tok + " short"], "short", mode="bool") == 1
    ok_tok = "語" * 13  # 39 bytes: kept
    assert splitgen.tokenize(ok_tok, "default") == [ok_tok]
    assert _search([ok_tok], ok_tok, mode="bool") == 1


def _random_text(rng):
    scripts = [
        lambda: chr(rng.randrange(0x61, 0x7B)),        # a-z
        lambda: chr(rng.randrange(0x41, 0x5B)),        # A-Z
        lambda: chr(rng.randrange(0xC0, 0x100)),       # Latin-1 letters + × ÷
        lambda: chr(rng.randrange(0x400, 0x450)),      # Cyrillic
        lambda: chr(rng.randrange(0x3B1, 0x3CA)),      # Greek lowercase
        lambda: chr(rng.randrange(0x4E00, 0x4E80)),    # CJK
        lambda: rng.choice(" .,…—·;\t"),               # separators
    ]
    return "".join(rng.choice(scripts)()
                   for _ in range(rng.randrange(3, 30)))


def test_random_unicode_index_query_agreement():
    """For random mixed-script strings: whatever the WRITER tokenizes a
    text to, a bool-mode full_text query with the same text must match the
    doc (same tokens on the query side), and each individual token must be
    findable as a term. 150 seeded cases."""
    rng = random.Random(88)
    texts = [t for t in (_random_text(rng) for _ in range(300))
             if splitgen.tokenize(t, "default")][:150]
    w = splitgen.SplitWriter(SCHEMA, "r", store_docs=False)
    w.add_documents([{"body": t} for t in texts])
    s = OracleSearcher()
    s.add_split("r", w.finalize())
    for i, t in enumerate(texts):
        q = {"type": "full_text", "field": "body", "text": t,
             "params": {"mode": {"type": "bool",
                                 "operator": "and"}}}
        r = s.leaf_search(make_leaf_request(q, SCHEMA, [("r", len(texts))],
                                            max_hits=0))
        assert r.get("num_hits", 0) >= 1, (i, t,
                                           splitgen.tokenize(t, "default"))


def test_unicode_wildcards():
    """Wildcard patterns fold through the field's tokenizer chain like the
    reference (wildcard_query.rs:111-160): default-analyzer fields
    lowercase the pattern with the shared table; `?` consumes one
    CODEPOINT (the reference compiles to a regex over chars); raw fields
    fold only under case_insensitive."""
    schema = {"timestamp_field": None, "fields": [
        {"name": "body", "type": "text", "tokenizer": "default",
         "record": "freq", "fieldnorms": True},
        {"name": "tag", "type": "text", "tokenizer": "raw",
         "record": "basic", "fieldnorms": False}]}
    w = splitgen.SplitWriter(schema, "wc", store_docs=False)
    w.add_documents([{"body": "hello héllo world wörld",
                      "tag": "Héllo-World"}])
    s = OracleSearcher()
    s.add_split("wc", w.finalize())
    cases = [
        (dict(field="body", value="HÉL*"), 1),
        (dict(field="body", value="WÖR*"), 1),
        (dict(field="body", value="h?llo"), 1),   # ? matches é (one cp)
        (dict(field="body", value="h??llo"), 0),  # é is ONE cp, not 2 bytes
        (dict(field="tag", value="Héllo-*"), 1),  # raw: byte-exact
        (dict(field="tag", value="héllo-*"), 0),
        (dict(field="tag", value="HÉLLO-*", case_insensitive=True), 1),
    ]
    for kw, want in cases:
        q = dict({"type": "wildcard"}, **kw)
        r = s.leaf_search(make_leaf_request(q, schema, [("wc", 1)],
                                            max_hits=5))
        assert r.get("num_hits", 0) == want, kw


@pytest.mark.gpu
def test_unicode_phrase_parity_gpu():
    from quickwit_amd.api import GpuSearcher
    docs = ["héllo wörld héllo", "Привет мир wörld", "日本語のテキスト",
            "plain ascii text", "GRÖSSE héllo"]
    for qt, mode in (("héllo wörld", "phrase"), ("привет мир", "phrase"),
                     ("HÉLLO", "bool"), ("日本語のテキスト", "bool"),
                     ("größe", "bool")):
        g = _search(docs, qt, mode=mode, searcher=GpuSearcher(device=0))
        e = _search(docs, qt, mode=mode)
        assert g == e, (qt, mode, g, e)
        assert g >= 1, (qt, mode)
