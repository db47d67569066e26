"""Minimal protobuf wire codec for the quickwit search.proto messages.

Message/field numbers are verbatim from
quickwit/quickwit-proto/protos/quickwit/search.proto (SearchRequest :207,
LeafSearchRequest :362, LeafRequestRef :512, SplitIdAndFooterOffsets :524,
PartialHit :578, SortByValue :607, LeafSearchResponse :618, SortField :286,
SplitSearchError :350, SplitResourceStats :383, LeafResourceStats :425).
No protoc exists in this image, so messages are described by small schema
tables and encoded/decoded by hand; the C++ side (quickwit_amd/csrc/pb.h)
mirrors this exactly and cross-language byte fixtures pin the wire format
(tests/test_proto.py).

Messages are plain dicts; missing keys = default/absent. Wire types used:
varint (int/uint/sint-free: only non-zigzag ints appear in these messages),
64-bit (double), length-delimited (string/bytes/embedded).
"""
import struct

# ---- schema tables: field_no -> (name, kind)
# kinds: u64/i64/u32/bool/enum (varint), f64 (fixed64), str, bytes,
#        msg:<Name>, + "*" prefix = repeated, "?" = optional presence tracked
SCHEMAS = {
    "SortByValue": {  # oneof sort_value
        1: ("u64", "u64"),
        2: ("i64", "i64"),
        3: ("f64", "f64"),
        4: ("boolean", "bool"),
    },
    "SortField": {
        1: ("field_name", "str"),
        2: ("sort_order", "enum"),  # ASC=0 DESC=1
        3: ("sort_datetime_format", "enum"),
    },
    "PartialHit": {
        10: ("sort_value", "msg:SortByValue"),
        11: ("sort_value2", "msg:SortByValue"),
        2: ("split_id", "str"),
        3: ("segment_ord", "u32"),
        4: ("doc_id", "u32"),
    },
    "SearchRequest": {
        1: ("index_id_patterns", "*str"),
        13: ("query_ast", "str"),
        4: ("start_timestamp", "i64"),
        5: ("end_timestamp", "i64"),
        6: ("max_hits", "u64"),
        7: ("start_offset", "u64"),
        11: ("aggregation_request", "str"),
        12: ("snippet_fields", "*str"),
        14: ("sort_fields", "*msg:SortField"),
        16: ("search_after", "msg:PartialHit"),
        17: ("count_hits", "enum"),  # COUNT_ALL=0 UNDERESTIMATE=1
        18: ("ignore_missing_indexes", "bool"),
        19: ("skip_aggregation_finalization", "bool"),
    },
    "SplitIdAndFooterOffsets": {
        1: ("split_id", "str"),
        2: ("split_footer_start", "u64"),
        3: ("split_footer_end", "u64"),
        4: ("timestamp_start", "i64"),
        5: ("timestamp_end", "i64"),
        6: ("num_docs", "u64"),
    },
    "LeafRequestRef": {
        1: ("doc_mapper_ord", "u32"),
        2: ("index_uri_ord", "u32"),
        3: ("split_offsets", "*msg:SplitIdAndFooterOffsets"),
    },
    "LeafSearchRequest": {
        1: ("search_request", "msg:SearchRequest"),
        7: ("leaf_requests", "*msg:LeafRequestRef"),
        8: ("doc_mappers", "*str"),
        9: ("index_uris", "*str"),
    },
    "FetchDocsRequest": {      # search.proto:674
        1: ("partial_hits", "*msg:PartialHit"),
        3: ("split_offsets", "*msg:SplitIdAndFooterOffsets"),
        4: ("index_uri", "str"),
        6: ("doc_mapper", "str"),
    },
    "LeafHit": {               # search.proto:553
        1: ("leaf_json", "str"),
        2: ("partial_hit", "msg:PartialHit"),
        3: ("leaf_snippet_json", "str"),
    },
    "FetchDocsResponse": {     # search.proto:695
        1: ("hits", "*msg:LeafHit"),
    },
    "ListTermsRequest": {      # search.proto:700
        1: ("index_id_patterns", "*str"),
        3: ("field", "str"),
        4: ("start_timestamp", "i64"),
        5: ("end_timestamp", "i64"),
        6: ("max_hits", "u64"),
        7: ("start_key", "bytes"),
        8: ("end_key", "bytes"),
    },
    "LeafListTermsRequest": {  # search.proto:732
        1: ("list_terms_request", "msg:ListTermsRequest"),
        2: ("split_offsets", "*msg:SplitIdAndFooterOffsets"),
        3: ("index_uri", "str"),
    },
    "LeafListTermsResponse": {  # search.proto:745
        1: ("num_hits", "u64"),
        2: ("terms", "*bytes"),
        3: ("failed_splits", "*msg:SplitSearchError"),
        4: ("num_attempted_splits", "u64"),
    },
    "SplitSearchError": {
        1: ("error", "str"),
        2: ("split_id", "str"),
        3: ("retryable_error", "bool"),
    },
    "SplitResourceStats": {
        1: ("split_num_docs", "u64"),
        2: ("input_memory_bytes", "u64"),
        3: ("download_num_bytes", "u64"),
        4: ("download_num_requests", "u64"),
        5: ("matched_num_docs", "u64"),
        6: ("wait_for_search_permit_microsecs", "u64"),
        7: ("warmup_microsecs", "u64"),
        8: ("wait_for_cpu_pool_microsecs", "u64"),
        9: ("cpu_search_microsecs", "u64"),
    },
    "LeafResourceStats": {
        1: ("partial_result_cache_num_splits", "u64"),
        2: ("partial_result_cache_num_docs", "u64"),
        3: ("localexec_num_splits", "u64"),
        4: ("localexec_num_docs", "u64"),
        5: ("split_resources_worst", "msg:SplitResourceStats"),
        6: ("split_resources_sum", "msg:SplitResourceStats"),
        7: ("min_wait_for_search_permit_microsecs", "u64"),
        8: ("min_wait_for_cpu_pool_microsecs", "u64"),
        9: ("wall_time_microsecs", "u64"),
        15: ("search_pool_cpu_threads", "u64"),
    },
    "LeafSearchResponse": {
        1: ("num_hits", "u64"),
        2: ("partial_hits", "*msg:PartialHit"),
        3: ("failed_splits", "*msg:SplitSearchError"),
        4: ("num_attempted_splits", "u64"),
        7: ("num_successful_splits", "u64"),
        6: ("intermediate_aggregation_result", "bytes"),
        9: ("resource_stats", "msg:LeafResourceStats"),
    },
}

_NAME_TO_NO = {
    m: {name: (no, kind) for no, (name, kind) in fields.items()}
    for m, fields in SCHEMAS.items()
}

_VARINT_KINDS = {"u64", "i64", "u32", "bool", "enum"}


def _enc_varint(out: bytearray, v: int):
    if v < 0:
        v &= (1 << 64) - 1  # int64 negative -> 10-byte varint
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def _dec_varint(buf, pos):
    shift = 0
    val = 0
    while True:
        b = buf[pos]
        pos += 1
        val |= (b & 0x7F) << shift
        if not b & 0x80:
            return val, pos
        shift += 7
        if shift > 70:
            raise ValueError("varint too long")


def _encode_field(out, no, kind, v):
    if kind in _VARINT_KINDS:
        if kind == "bool":
            v = 1 if v else 0
        _enc_varint(out, no << 3 | 0)
        _enc_varint(out, int(v))
    elif kind == "f64":
        _enc_varint(out, no << 3 | 1)
        out += struct.pack("<d", v)
    elif kind == "str":
        b = v.encode("utf-8")
        _enc_varint(out, no << 3 | 2)
        _enc_varint(out, len(b))
        out += b
    elif kind == "bytes":
        _enc_varint(out, no << 3 | 2)
        _enc_varint(out, len(v))
        out += bytes(v)
    elif kind.startswith("msg:"):
        b = encode(kind[4:], v)
        _enc_varint(out, no << 3 | 2)
        _enc_varint(out, len(b))
        out += b
    else:
        raise ValueError(kind)


def encode(msg_name: str, d: dict) -> bytes:
    """Encode dict -> wire bytes, fields in ascending field-number order.
    Zero/empty scalar values are skipped only if the key is absent —
    presence in the dict means 'emit' (optional-field semantics)."""
    out = bytearray()
    for no in sorted(SCHEMAS[msg_name]):
        name, kind = SCHEMAS[msg_name][no]
        if name not in d or d[name] is None:
            continue
        v = d[name]
        if kind.startswith("*"):
            for item in v:
                _encode_field(out, no, kind[1:], item)
        else:
            # proto3 default-skipping for plain scalars
            if kind in _VARINT_KINDS and not kind_is_optional(msg_name, name) and int(v) == 0:
                continue
            if kind == "str" and v == "" and not kind_is_optional(msg_name, name):
                continue
            _encode_field(out, no, kind, v)
    return bytes(out)


_OPTIONAL = {
    ("LeafHit", "leaf_snippet_json"),
    ("ListTermsRequest", "start_timestamp"),
    ("ListTermsRequest", "end_timestamp"),
    ("ListTermsRequest", "max_hits"),
    ("ListTermsRequest", "start_key"),
    ("ListTermsRequest", "end_key"),
    ("SearchRequest", "start_timestamp"),
    ("SearchRequest", "end_timestamp"),
    ("SearchRequest", "aggregation_request"),
    ("SearchRequest", "scroll_ttl_secs"),
    ("SortField", "sort_datetime_format"),
    ("SplitIdAndFooterOffsets", "timestamp_start"),
    ("SplitIdAndFooterOffsets", "timestamp_end"),
    ("LeafResourceStats", "min_wait_for_search_permit_microsecs"),
    ("LeafResourceStats", "min_wait_for_cpu_pool_microsecs"),
    # oneof members are emitted when present, including zero values
    ("SortByValue", "u64"),
    ("SortByValue", "i64"),
    ("SortByValue", "f64"),
    ("SortByValue", "boolean"),
}


def kind_is_optional(msg, name):
    return (msg, name) in _OPTIONAL


def _to_signed64(v):
    return v - (1 << 64) if v >= (1 << 63) else v


def decode(msg_name: str, buf: bytes) -> dict:
    schema = SCHEMAS[msg_name]
    d = {}
    pos = 0
    n = len(buf)
    while pos < n:
        tag, pos = _dec_varint(buf, pos)
        no, wt = tag >> 3, tag & 7
        ent = schema.get(no)
        if ent is None:  # skip unknown field
            if wt == 0:
                _, pos = _dec_varint(buf, pos)
            elif wt == 1:
                pos += 8
            elif wt == 2:
                ln, pos = _dec_varint(buf, pos)
                pos += ln
            elif wt == 5:
                pos += 4
            else:
                raise ValueError(f"unknown wire type {wt}")
            continue
        name, kind = ent
        rep = kind.startswith("*")
        k = kind[1:] if rep else kind
        if k in _VARINT_KINDS:
            v, pos = _dec_varint(buf, pos)
            if k == "i64":
                v = _to_signed64(v)
            elif k == "bool":
                v = bool(v)
        elif k == "f64":
            (v,) = struct.unpack_from("<d", buf, pos)
            pos += 8
        else:
            ln, pos = _dec_varint(buf, pos)
            raw = buf[pos : pos + ln]
            pos += ln
            if k == "str":
                v = raw.decode("utf-8")
            elif k == "bytes":
                v = bytes(raw)
            else:
                v = decode(k[4:], raw)
        if rep:
            d.setdefault(name, []).append(v)
        else:
            d[name] = v
    return d
