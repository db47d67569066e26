"""date_histogram bucketing edges: pre-1970 (negative) timestamps must
floor toward -inf (ES/tantivy semantics), not truncate toward zero — the
classic C integer-division bug. The kernel's reciprocal-multiply floor
division (kernels.hip, int_fast path) and the oracle's f64 floor must
agree bit-exactly on the bucket keys."""
import pytest

from quickwit_amd import splitgen
from quickwit_amd.api import OracleSearcher, make_leaf_request

SCHEMA = {"timestamp_field": "timestamp", "fields": [
    {"name": "timestamp", "type": "datetime", "fast": True},
    {"name": "body", "type": "text", "tokenizer": "default",
     "record": "basic", "fieldnorms": False}]}

TS_SECONDS = (-7201, -7200, -3600, -1, 0, 1, 3599, 3600, 7199)
AGGS = {"h": {"date_histogram": {"field": "timestamp",
                                 "fixed_interval": "3600000ms"}}}
# floor(ts/3600)*3600000 ms keys
EXPECT = {-10800000.0: 1, -7200000.0: 1, -3600000.0: 2, 0.0: 3,
          3600000.0: 2}


def _split():
    w = splitgen.SplitWriter(SCHEMA, "neg", store_docs=False)
    w.add_documents([{"timestamp": t, "body": "x"} for t in TS_SECONDS])
    return w.finalize()


def _buckets(searcher):
    searcher.add_split("neg", _split())
    r = searcher.leaf_search(make_leaf_request(
        {"type": "match_all"}, SCHEMA, [("neg", len(TS_SECONDS))],
        max_hits=0, aggregation=AGGS))
    j = searcher.finalize_agg_json(r["intermediate_aggregation_result"],
                                  AGGS)
    return {b["key"]: b["doc_count"] for b in j["h"]["buckets"]}


def test_negative_timestamp_floor_oracle():
    assert _buckets(OracleSearcher()) == EXPECT


@pytest.mark.gpu
def test_negative_timestamp_floor_gpu():
    from quickwit_amd.api import GpuSearcher
    assert _buckets(GpuSearcher(device=0)) == EXPECT
