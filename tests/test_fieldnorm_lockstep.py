"""fieldnorm.py (writer) and csrc/fieldnorm.h (both engines) must encode
identically for every length — the header says 'must stay in lockstep'.
The C++ side is reached through the oracle: a one-doc split of length L
queried with a 1-term scored query exposes K = k1*(1-b+b*decode(id)/avgdl),
from which decode(encode(L)) is recovered exactly. Checked at every
boundary of the SmallFloat int4 table plus random lengths."""
import math
import random

import numpy as np

from quickwit_amd import fieldnorm, splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

SCHEMA = {"timestamp_field": None, "fields": [
    {"name": "txt", "type": "text", "tokenizer": "default",
     "record": "freq", "fieldnorms": True}]}


def py_decode(i):
    return i if i < 8 else ((i & 7) | 8) << ((i >> 3) - 1)


def test_python_table_matches_smallfloat_spec():
    # fieldnorm.py against an independent restatement of the int4 scheme
    ids = fieldnorm.norm_to_id(np.arange(0, 5000))
    for length in range(0, 5000):
        i = int(ids[length])
        assert py_decode(i) <= length, length
        assert i == 255 or py_decode(i + 1) > length, length
    assert int(fieldnorm.id_to_norm(np.array([255]))[0]) == py_decode(255)


def test_cpp_encode_matches_python_via_scores():
    rng = random.Random(5)
    # boundary lengths around every table step + random ones
    lengths = {1, 2, 7, 8, 9, 15, 16, 17, 23, 24, 31, 32, 39, 40, 47, 48}
    lengths |= {rng.randrange(1, 3000) for _ in range(20)}
    for L in sorted(lengths):
        doc = {"txt": "hit " + " ".join("pad%d" % i for i in range(L - 1))}
        w = splitgen.SplitWriter(SCHEMA, "fn", store_docs=False)
        w.add_documents([doc])
        s = OracleSearcher()
        s.add_split("fn", w.finalize())
        r = s.leaf_search(make_leaf_request(
            {"type": "term", "field": "txt", "value": "hit"}, SCHEMA,
            [("fn", 1)], max_hits=1,
            sort_fields=[{"field_name": "_score", "sort_order": 1}]))
        score = r["partial_hits"][0]["sort_value"]["f64"]
        # one doc: N=1, df=1 -> idf = ln(1 + 0.5/1.5); avgdl = L
        idf = math.log(1.0 + 0.5 / 1.5)
        w32 = np.float32(idf * 2.2)
        quant = py_decode(int(fieldnorm.norm_to_id(np.array([L]))[0]))
        k = np.float32(1.2 * (0.25 + 0.75 * quant / L))
        want = float(w32 * (np.float32(1.0) / (np.float32(1.0) + k)))
        assert math.isclose(score, want, rel_tol=1e-6), (L, score, want)
