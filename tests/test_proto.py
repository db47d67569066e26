"""Wire-format tests for the minimal protobuf codec.

The byte fixtures below are hand-derived from the protobuf wire spec and pin
the encoding that the C++ side (csrc/pb.h) must also produce; field numbers
cite quickwit-proto/protos/quickwit/search.proto.
"""
from quickwit_amd import proto


def test_varint_wire_fixture():
    # SearchRequest{max_hits: 300} -> field 6 varint: tag 0x30, 300 = 0xAC 0x02
    assert proto.encode("SearchRequest", {"max_hits": 300}) == b"\x30\xac\x02"
    # query_ast (field 13, wiretype 2): tag = 13<<3|2 = 106 = 0x6a
    assert proto.encode("SearchRequest", {"query_ast": "ab"}) == b"\x6a\x02ab"


def test_negative_int64_ten_bytes():
    b = proto.encode("SplitIdAndFooterOffsets", {"timestamp_start": -1})
    # field 4 varint: tag 0x20 then ten 0xff..0x01
    assert b == b"\x20" + b"\xff" * 9 + b"\x01"
    d = proto.decode("SplitIdAndFooterOffsets", b)
    assert d["timestamp_start"] == -1


def test_double_fixture():
    b = proto.encode("SortByValue", {"f64": 1.5})
    assert b == b"\x19\x00\x00\x00\x00\x00\x00\xf8\x3f"  # tag 3<<3|1=0x19


def test_roundtrip_leaf_search_request():
    req = {
        "search_request": {
            "index_id_patterns": ["idx"],
            "query_ast": '{"type":"match_all"}',
            "max_hits": 10,
            "sort_fields": [{"field_name": "_score", "sort_order": 1}],
            "count_hits": 0,
        },
        "leaf_requests": [
            {
                "doc_mapper_ord": 0,
                "index_uri_ord": 0,
                "split_offsets": [
                    {"split_id": "s1", "split_footer_start": 100, "split_footer_end": 200,
                     "num_docs": 42},
                    {"split_id": "s2", "split_footer_start": 1, "split_footer_end": 2,
                     "timestamp_start": -5, "num_docs": 1},
                ],
            }
        ],
        "doc_mappers": ["{}"],
        "index_uris": ["ram:///x"],
    }
    b = proto.encode("LeafSearchRequest", req)
    d = proto.decode("LeafSearchRequest", b)
    assert d["search_request"]["query_ast"] == req["search_request"]["query_ast"]
    assert d["search_request"]["max_hits"] == 10
    assert d["search_request"]["sort_fields"][0] == {"field_name": "_score", "sort_order": 1}
    so = d["leaf_requests"][0]["split_offsets"]
    assert so[0]["split_id"] == "s1" and so[0]["num_docs"] == 42
    assert so[1]["timestamp_start"] == -5
    # proto3 zero-default scalars are absent after round-trip
    assert "count_hits" not in d["search_request"]


def test_roundtrip_leaf_search_response():
    resp = {
        "num_hits": 12345,
        "partial_hits": [
            {"sort_value": {"f64": 0.5}, "sort_value2": None, "split_id": "s1",
             "segment_ord": 0, "doc_id": 7},
            {"sort_value": {"u64": 0}, "split_id": "s2", "doc_id": 0},
        ],
        "failed_splits": [{"error": "boom", "split_id": "s3", "retryable_error": True}],
        "num_attempted_splits": 2,
        "num_successful_splits": 2,
        "intermediate_aggregation_result": b"\x01\x02\x00\xff",
        "resource_stats": {"localexec_num_splits": 2, "wall_time_microsecs": 999,
                           "search_pool_cpu_threads": 8},
    }
    d = proto.decode("LeafSearchResponse", proto.encode("LeafSearchResponse", resp))
    assert d["num_hits"] == 12345
    assert d["partial_hits"][0]["sort_value"]["f64"] == 0.5
    assert d["partial_hits"][0]["doc_id"] == 7
    # oneof member with zero value IS emitted (presence semantics)
    assert d["partial_hits"][1]["sort_value"] == {"u64": 0}
    assert d["failed_splits"][0]["retryable_error"] is True
    assert d["intermediate_aggregation_result"] == b"\x01\x02\x00\xff"
    assert d["resource_stats"]["search_pool_cpu_threads"] == 8


def _rand_value(rng, kind, depth):
    if kind == "u64":
        return rng.choice([0, 1, 127, 128, 300, 2**32, 2**64 - 1,
                           rng.getrandbits(64)])
    if kind == "u32":
        return rng.choice([0, 1, 255, 2**32 - 1, rng.getrandbits(32)])
    if kind == "i64":
        return rng.choice([0, -1, 1, -(2**63), 2**63 - 1,
                           rng.getrandbits(63) - 2**62])
    if kind == "bool":
        return rng.random() < 0.5
    if kind == "enum":
        return rng.randrange(3)
    if kind == "f64":
        return rng.choice([0.0, -0.0, 1.5, -3.25e300, 5e-324,
                           float("inf"), rng.uniform(-1e9, 1e9)])
    if kind == "str":
        return rng.choice(["", "a", "héllo wörld", "w%05d" % rng.randrange(99999),
                           "é世界" * rng.randrange(4)])
    if kind == "bytes":
        return bytes(rng.randrange(256) for _ in range(rng.randrange(12)))
    assert kind.startswith("msg:")
    return _rand_msg(rng, kind[4:], depth + 1)


def _rand_msg(rng, msg, depth):
    d = {}
    for no, (name, kind) in proto.SCHEMAS[msg].items():
        if rng.random() < 0.45 or depth > 4:
            continue
        if kind.startswith("*"):
            d[name] = [_rand_value(rng, kind[1:], depth)
                       for _ in range(rng.randrange(4))]
        else:
            d[name] = _rand_value(rng, kind, depth)
    return d


def test_randomized_encode_decode_fixpoint():
    """For every message type: decode(encode(d)) loses only proto3
    zero-defaults, so encode∘decode is the IDENTITY on encoder-produced
    bytes, and decode is stable from there on. 200 seeded random
    instances per type, covering varint boundaries, 10-byte negative
    i64, unicode, empty/absent repeated fields and nested messages."""
    import random
    rng = random.Random(20260915)
    for msg in proto.SCHEMAS:
        for _ in range(200):
            d = _rand_msg(rng, msg, 0)
            b1 = proto.encode(msg, d)
            d1 = proto.decode(msg, b1)
            b2 = proto.encode(msg, d1)
            assert b2 == b1, (msg, d)
            assert proto.decode(msg, b2) == d1, (msg, d)


def test_unknown_fields_skipped():
    # an unknown varint field (no 99) must be skipped, not crash
    extra = proto.encode("SearchRequest", {"max_hits": 5})
    tag = bytearray()
    proto._enc_varint(tag, 99 << 3 | 0)
    unknown = bytes(tag) + b"\x07"
    d = proto.decode("SearchRequest", unknown + extra)
    assert d["max_hits"] == 5
