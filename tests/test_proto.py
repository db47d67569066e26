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


def test_unknown_fields_skipped():
    # an unknown varint field (no 99) must be skipped, not crash
    extra = proto.encode("SearchRequest", {"max_hits": 5})
    tag = bytearray()
    proto._enc_varint(tag, 99 << 3 | 0)
    unknown = bytes(tag) + b"\x07"
    d = proto.decode("SearchRequest", unknown + extra)
    assert d["max_hits"] == 5
