// Minimal protobuf wire codec for the quickwit search.proto messages
// (boundary plumbing shared by product and oracle — DESIGN.md §1/§4).
// Field numbers verbatim from quickwit-proto/protos/quickwit/search.proto:
// SearchRequest :207, LeafSearchRequest :362, LeafRequestRef :512,
// SplitIdAndFooterOffsets :524, PartialHit :578, SortByValue :607,
// LeafSearchResponse :618, SortField :286, SplitSearchError :350,
// SplitResourceStats :383, LeafResourceStats :425.
// Mirrors quickwit_amd/proto.py; wire fixtures in tests/test_proto.py pin both.
#pragma once
#include <cstdint>
#include <cstring>
#include <optional>
#include <stdexcept>
#include <string>
#include <vector>

namespace pb {

// ------------------------------------------------------------ wire helpers
struct Writer {
    std::string out;
    void varint(uint64_t v) {
        while (true) {
            uint8_t b = v & 0x7F;
            v >>= 7;
            if (v) out.push_back(char(b | 0x80));
            else { out.push_back(char(b)); return; }
        }
    }
    void tag(uint32_t no, uint32_t wt) { varint(uint64_t(no) << 3 | wt); }
    void u64_field(uint32_t no, uint64_t v, bool always = false) {
        if (!v && !always) return;
        tag(no, 0);
        varint(v);
    }
    void i64_field(uint32_t no, int64_t v, bool always = false) {
        if (!v && !always) return;
        tag(no, 0);
        varint(uint64_t(v));
    }
    void bool_field(uint32_t no, bool v, bool always = false) {
        if (!v && !always) return;
        tag(no, 0);
        varint(v ? 1 : 0);
    }
    void f64_field(uint32_t no, double v) {
        tag(no, 1);
        uint64_t bits;
        memcpy(&bits, &v, 8);
        for (int i = 0; i < 8; ++i) out.push_back(char(bits >> (8 * i)));
    }
    void str_field(uint32_t no, const std::string& s, bool always = false) {
        if (s.empty() && !always) return;
        tag(no, 2);
        varint(s.size());
        out += s;
    }
    void bytes_field(uint32_t no, const std::string& s) { str_field(no, s, true); }
    void msg_field(uint32_t no, const std::string& sub) {
        tag(no, 2);
        varint(sub.size());
        out += sub;
    }
};

struct Reader {
    const uint8_t* p;
    const uint8_t* end;
    Reader(const uint8_t* data, size_t n) : p(data), end(data + n) {}
    bool done() const { return p >= end; }
    uint64_t varint() {
        uint64_t v = 0;
        int shift = 0;
        while (true) {
            if (p >= end) throw std::runtime_error("pb: truncated varint");
            uint8_t b = *p++;
            v |= uint64_t(b & 0x7F) << shift;
            if (!(b & 0x80)) return v;
            shift += 7;
            if (shift > 70) throw std::runtime_error("pb: varint too long");
        }
    }
    uint32_t read_tag(uint32_t* no) {
        uint64_t t = varint();
        *no = uint32_t(t >> 3);
        return uint32_t(t & 7);
    }
    double f64() {
        if (end - p < 8) throw std::runtime_error("pb: truncated f64");
        uint64_t bits = 0;
        for (int i = 0; i < 8; ++i) bits |= uint64_t(p[i]) << (8 * i);
        p += 8;
        double d;
        memcpy(&d, &bits, 8);
        return d;
    }
    std::string bytes() {
        uint64_t n = varint();
        if (uint64_t(end - p) < n) throw std::runtime_error("pb: truncated bytes");
        std::string s(reinterpret_cast<const char*>(p), n);
        p += n;
        return s;
    }
    void skip(uint32_t wt) {
        switch (wt) {
            case 0: varint(); break;
            case 1:
                if (end - p < 8) throw std::runtime_error("pb: trunc");
                p += 8;
                break;
            case 2: bytes(); break;
            case 5:
                if (end - p < 4) throw std::runtime_error("pb: trunc");
                p += 4;
                break;
            default: throw std::runtime_error("pb: bad wire type");
        }
    }
};

// ------------------------------------------------------------ message structs
struct SortByValue {  // search.proto:607 (oneof)
    enum Kind { NONE, U64, I64, F64, BOOL } kind = NONE;
    uint64_t u64 = 0;
    int64_t i64 = 0;
    double f64 = 0;
    bool boolean = false;
    std::string encode() const {
        Writer w;
        switch (kind) {
            case U64: w.u64_field(1, u64, true); break;
            case I64: w.i64_field(2, i64, true); break;
            case F64: w.f64_field(3, f64); break;
            case BOOL: w.bool_field(4, boolean, true); break;
            case NONE: break;
        }
        return w.out;
    }
    static SortByValue decode(Reader r) {
        SortByValue v;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 0) { v.kind = U64; v.u64 = r.varint(); }
            else if (no == 2 && wt == 0) { v.kind = I64; v.i64 = int64_t(r.varint()); }
            else if (no == 3 && wt == 1) { v.kind = F64; v.f64 = r.f64(); }
            else if (no == 4 && wt == 0) { v.kind = BOOL; v.boolean = r.varint() != 0; }
            else r.skip(wt);
        }
        return v;
    }
};

struct PartialHit {  // search.proto:578
    SortByValue sort_value, sort_value2;
    std::string split_id;
    uint32_t segment_ord = 0;
    uint32_t doc_id = 0;
    std::string encode() const {
        Writer w;
        w.str_field(2, split_id);
        w.u64_field(3, segment_ord);
        w.u64_field(4, doc_id);
        if (sort_value.kind != SortByValue::NONE) w.msg_field(10, sort_value.encode());
        if (sort_value2.kind != SortByValue::NONE) w.msg_field(11, sort_value2.encode());
        return w.out;
    }
    static PartialHit decode(Reader r) {
        PartialHit h;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 2 && wt == 2) h.split_id = r.bytes();
            else if (no == 3 && wt == 0) h.segment_ord = uint32_t(r.varint());
            else if (no == 4 && wt == 0) h.doc_id = uint32_t(r.varint());
            else if (no == 10 && wt == 2) {
                std::string b = r.bytes();
                h.sort_value = SortByValue::decode(Reader((const uint8_t*)b.data(), b.size()));
            } else if (no == 11 && wt == 2) {
                std::string b = r.bytes();
                h.sort_value2 = SortByValue::decode(Reader((const uint8_t*)b.data(), b.size()));
            } else r.skip(wt);
        }
        return h;
    }
};

struct SortField {  // search.proto:286
    std::string field_name;
    int32_t sort_order = 0;  // ASC=0 DESC=1 (search.proto:295)
    std::optional<int32_t> sort_datetime_format;
    std::string encode() const {
        Writer w;
        w.str_field(1, field_name);
        w.u64_field(2, uint64_t(sort_order));
        if (sort_datetime_format) w.u64_field(3, uint64_t(*sort_datetime_format), true);
        return w.out;
    }
    static SortField decode(Reader r) {
        SortField f;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 2) f.field_name = r.bytes();
            else if (no == 2 && wt == 0) f.sort_order = int32_t(r.varint());
            else if (no == 3 && wt == 0) f.sort_datetime_format = int32_t(r.varint());
            else r.skip(wt);
        }
        return f;
    }
};

struct SearchRequest {  // search.proto:207
    std::vector<std::string> index_id_patterns;  // 1
    std::string query_ast;                       // 13
    std::optional<int64_t> start_timestamp;      // 4
    std::optional<int64_t> end_timestamp;        // 5
    uint64_t max_hits = 0;                       // 6
    uint64_t start_offset = 0;                   // 7
    std::optional<std::string> aggregation_request;  // 11
    std::vector<std::string> snippet_fields;     // 12
    std::vector<SortField> sort_fields;          // 14
    std::optional<PartialHit> search_after;      // 16
    int32_t count_hits = 0;                      // 17 COUNT_ALL=0 UNDERESTIMATE=1
    std::string encode() const {
        Writer w;
        for (auto& s : index_id_patterns) w.str_field(1, s, true);
        if (start_timestamp) w.i64_field(4, *start_timestamp, true);
        if (end_timestamp) w.i64_field(5, *end_timestamp, true);
        w.u64_field(6, max_hits);
        w.u64_field(7, start_offset);
        if (aggregation_request) w.str_field(11, *aggregation_request, true);
        for (auto& s : snippet_fields) w.str_field(12, s, true);
        w.str_field(13, query_ast);
        for (auto& f : sort_fields) w.msg_field(14, f.encode());
        if (search_after) w.msg_field(16, search_after->encode());
        w.u64_field(17, uint64_t(count_hits));
        return w.out;
    }
    static SearchRequest decode(Reader r) {
        SearchRequest q;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            switch (no) {
                case 1: q.index_id_patterns.push_back(r.bytes()); break;
                case 4: q.start_timestamp = int64_t(r.varint()); break;
                case 5: q.end_timestamp = int64_t(r.varint()); break;
                case 6: q.max_hits = r.varint(); break;
                case 7: q.start_offset = r.varint(); break;
                case 11: q.aggregation_request = r.bytes(); break;
                case 12: q.snippet_fields.push_back(r.bytes()); break;
                case 13: q.query_ast = r.bytes(); break;
                case 14: {
                    std::string b = r.bytes();
                    q.sort_fields.push_back(
                        SortField::decode(Reader((const uint8_t*)b.data(), b.size())));
                    break;
                }
                case 16: {
                    std::string b = r.bytes();
                    q.search_after =
                        PartialHit::decode(Reader((const uint8_t*)b.data(), b.size()));
                    break;
                }
                case 17: q.count_hits = int32_t(r.varint()); break;
                default: r.skip(wt);
            }
        }
        return q;
    }
};

struct SplitIdAndFooterOffsets {  // search.proto:524
    std::string split_id;
    uint64_t split_footer_start = 0, split_footer_end = 0;
    std::optional<int64_t> timestamp_start, timestamp_end;
    uint64_t num_docs = 0;
    std::string encode() const {
        Writer w;
        w.str_field(1, split_id);
        w.u64_field(2, split_footer_start);
        w.u64_field(3, split_footer_end);
        if (timestamp_start) w.i64_field(4, *timestamp_start, true);
        if (timestamp_end) w.i64_field(5, *timestamp_end, true);
        w.u64_field(6, num_docs);
        return w.out;
    }
    static SplitIdAndFooterOffsets decode(Reader r) {
        SplitIdAndFooterOffsets s;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            switch (no) {
                case 1: s.split_id = r.bytes(); break;
                case 2: s.split_footer_start = r.varint(); break;
                case 3: s.split_footer_end = r.varint(); break;
                case 4: s.timestamp_start = int64_t(r.varint()); break;
                case 5: s.timestamp_end = int64_t(r.varint()); break;
                case 6: s.num_docs = r.varint(); break;
                default: r.skip(wt);
            }
        }
        return s;
    }
};

struct LeafRequestRef {  // search.proto:512
    uint32_t doc_mapper_ord = 0, index_uri_ord = 0;
    std::vector<SplitIdAndFooterOffsets> split_offsets;
    std::string encode() const {
        Writer w;
        w.u64_field(1, doc_mapper_ord);
        w.u64_field(2, index_uri_ord);
        for (auto& s : split_offsets) w.msg_field(3, s.encode());
        return w.out;
    }
    static LeafRequestRef decode(Reader r) {
        LeafRequestRef l;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 0) l.doc_mapper_ord = uint32_t(r.varint());
            else if (no == 2 && wt == 0) l.index_uri_ord = uint32_t(r.varint());
            else if (no == 3 && wt == 2) {
                std::string b = r.bytes();
                l.split_offsets.push_back(SplitIdAndFooterOffsets::decode(
                    Reader((const uint8_t*)b.data(), b.size())));
            } else r.skip(wt);
        }
        return l;
    }
};

struct LeafSearchRequest {  // search.proto:362
    SearchRequest search_request;
    std::vector<LeafRequestRef> leaf_requests;
    std::vector<std::string> doc_mappers;
    std::vector<std::string> index_uris;
    std::string encode() const {
        Writer w;
        w.msg_field(1, search_request.encode());
        for (auto& l : leaf_requests) w.msg_field(7, l.encode());
        for (auto& s : doc_mappers) w.str_field(8, s, true);
        for (auto& s : index_uris) w.str_field(9, s, true);
        return w.out;
    }
    static LeafSearchRequest decode(const uint8_t* data, size_t n) {
        Reader r(data, n);
        LeafSearchRequest q;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 2) {
                std::string b = r.bytes();
                q.search_request =
                    SearchRequest::decode(Reader((const uint8_t*)b.data(), b.size()));
            } else if (no == 7 && wt == 2) {
                std::string b = r.bytes();
                q.leaf_requests.push_back(
                    LeafRequestRef::decode(Reader((const uint8_t*)b.data(), b.size())));
            } else if (no == 8 && wt == 2) q.doc_mappers.push_back(r.bytes());
            else if (no == 9 && wt == 2) q.index_uris.push_back(r.bytes());
            else r.skip(wt);
        }
        return q;
    }
};

struct SplitSearchError {  // search.proto:350
    std::string error, split_id;
    bool retryable_error = false;
    std::string encode() const {
        Writer w;
        w.str_field(1, error);
        w.str_field(2, split_id);
        w.bool_field(3, retryable_error);
        return w.out;
    }
    static SplitSearchError decode(Reader r) {
        SplitSearchError e;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 2) e.error = r.bytes();
            else if (no == 2 && wt == 2) e.split_id = r.bytes();
            else if (no == 3 && wt == 0) e.retryable_error = r.varint() != 0;
            else r.skip(wt);
        }
        return e;
    }
};

struct ListTermsRequest {  // search.proto:700
    std::vector<std::string> index_id_patterns;      // 1
    std::string field;                               // 3
    std::optional<int64_t> start_timestamp;          // 4
    std::optional<int64_t> end_timestamp;            // 5
    std::optional<uint64_t> max_hits;                // 6
    std::optional<std::string> start_key, end_key;   // 7 incl, 8 excl
    std::string encode() const {
        Writer w;
        for (auto& s : index_id_patterns) w.str_field(1, s, true);
        w.str_field(3, field);
        if (start_timestamp) w.i64_field(4, *start_timestamp, true);
        if (end_timestamp) w.i64_field(5, *end_timestamp, true);
        if (max_hits) w.u64_field(6, *max_hits, true);
        if (start_key) w.str_field(7, *start_key, true);
        if (end_key) w.str_field(8, *end_key, true);
        return w.out;
    }
    static ListTermsRequest decode(Reader r) {
        ListTermsRequest q;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            switch (no) {
                case 1: q.index_id_patterns.push_back(r.bytes()); break;
                case 3: q.field = r.bytes(); break;
                case 4: q.start_timestamp = int64_t(r.varint()); break;
                case 5: q.end_timestamp = int64_t(r.varint()); break;
                case 6: q.max_hits = r.varint(); break;
                case 7: q.start_key = r.bytes(); break;
                case 8: q.end_key = r.bytes(); break;
                default: r.skip(wt);
            }
        }
        return q;
    }
};

struct SplitResourceStats {  // search.proto:383
    uint64_t split_num_docs = 0, input_memory_bytes = 0, download_num_bytes = 0,
             download_num_requests = 0, matched_num_docs = 0,
             wait_for_search_permit_microsecs = 0, warmup_microsecs = 0,
             wait_for_cpu_pool_microsecs = 0, cpu_search_microsecs = 0;
    std::string encode() const {
        Writer w;
        w.u64_field(1, split_num_docs);
        w.u64_field(2, input_memory_bytes);
        w.u64_field(3, download_num_bytes);
        w.u64_field(4, download_num_requests);
        w.u64_field(5, matched_num_docs);
        w.u64_field(6, wait_for_search_permit_microsecs);
        w.u64_field(7, warmup_microsecs);
        w.u64_field(8, wait_for_cpu_pool_microsecs);
        w.u64_field(9, cpu_search_microsecs);
        return w.out;
    }
    static SplitResourceStats decode(Reader r) {
        SplitResourceStats s;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            uint64_t* f[10] = {nullptr, &s.split_num_docs, &s.input_memory_bytes,
                               &s.download_num_bytes, &s.download_num_requests,
                               &s.matched_num_docs, &s.wait_for_search_permit_microsecs,
                               &s.warmup_microsecs, &s.wait_for_cpu_pool_microsecs,
                               &s.cpu_search_microsecs};
            if (wt == 0 && no >= 1 && no <= 9) *f[no] = r.varint();
            else r.skip(wt);
        }
        return s;
    }
};

struct LeafResourceStats {  // search.proto:425
    uint64_t partial_result_cache_num_splits = 0, partial_result_cache_num_docs = 0;
    uint64_t localexec_num_splits = 0, localexec_num_docs = 0;
    SplitResourceStats split_resources_worst, split_resources_sum;
    std::optional<uint64_t> min_wait_for_search_permit_microsecs,
        min_wait_for_cpu_pool_microsecs;
    uint64_t wall_time_microsecs = 0;
    uint64_t search_pool_cpu_threads = 0;
    std::string encode() const {
        Writer w;
        w.u64_field(1, partial_result_cache_num_splits);
        w.u64_field(2, partial_result_cache_num_docs);
        w.u64_field(3, localexec_num_splits);
        w.u64_field(4, localexec_num_docs);
        w.msg_field(5, split_resources_worst.encode());
        w.msg_field(6, split_resources_sum.encode());
        if (min_wait_for_search_permit_microsecs)
            w.u64_field(7, *min_wait_for_search_permit_microsecs, true);
        if (min_wait_for_cpu_pool_microsecs)
            w.u64_field(8, *min_wait_for_cpu_pool_microsecs, true);
        w.u64_field(9, wall_time_microsecs);
        w.u64_field(15, search_pool_cpu_threads);
        return w.out;
    }
    static LeafResourceStats decode(Reader r) {
        LeafResourceStats s;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 0) s.partial_result_cache_num_splits = r.varint();
            else if (no == 2 && wt == 0) s.partial_result_cache_num_docs = r.varint();
            else if (no == 3 && wt == 0) s.localexec_num_splits = r.varint();
            else if (no == 4 && wt == 0) s.localexec_num_docs = r.varint();
            else if (no == 5 && wt == 2) {
                std::string b = r.bytes();
                s.split_resources_worst = SplitResourceStats::decode(
                    Reader((const uint8_t*)b.data(), b.size()));
            } else if (no == 6 && wt == 2) {
                std::string b = r.bytes();
                s.split_resources_sum = SplitResourceStats::decode(
                    Reader((const uint8_t*)b.data(), b.size()));
            } else if (no == 7 && wt == 0) s.min_wait_for_search_permit_microsecs = r.varint();
            else if (no == 8 && wt == 0) s.min_wait_for_cpu_pool_microsecs = r.varint();
            else if (no == 9 && wt == 0) s.wall_time_microsecs = r.varint();
            else if (no == 15 && wt == 0) s.search_pool_cpu_threads = r.varint();
            else r.skip(wt);
        }
        return s;
    }
};

struct LeafSearchResponse {  // search.proto:618
    uint64_t num_hits = 0;                           // 1
    std::vector<PartialHit> partial_hits;            // 2
    std::vector<SplitSearchError> failed_splits;     // 3
    uint64_t num_attempted_splits = 0;               // 4
    uint64_t num_successful_splits = 0;              // 7
    std::optional<std::string> intermediate_aggregation_result;  // 6
    std::optional<LeafResourceStats> resource_stats;             // 9
    std::string encode() const {
        Writer w;
        w.u64_field(1, num_hits);
        for (auto& h : partial_hits) w.msg_field(2, h.encode());
        for (auto& e : failed_splits) w.msg_field(3, e.encode());
        w.u64_field(4, num_attempted_splits);
        if (intermediate_aggregation_result) w.bytes_field(6, *intermediate_aggregation_result);
        w.u64_field(7, num_successful_splits);
        if (resource_stats) w.msg_field(9, resource_stats->encode());
        return w.out;
    }
    static LeafSearchResponse decode(const uint8_t* data, size_t n) {
        Reader r(data, n);
        LeafSearchResponse v;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 0) v.num_hits = r.varint();
            else if (no == 2 && wt == 2) {
                std::string b = r.bytes();
                v.partial_hits.push_back(
                    PartialHit::decode(Reader((const uint8_t*)b.data(), b.size())));
            } else if (no == 3 && wt == 2) {
                std::string b = r.bytes();
                v.failed_splits.push_back(
                    SplitSearchError::decode(Reader((const uint8_t*)b.data(), b.size())));
            } else if (no == 4 && wt == 0) v.num_attempted_splits = r.varint();
            else if (no == 6 && wt == 2) v.intermediate_aggregation_result = r.bytes();
            else if (no == 7 && wt == 0) v.num_successful_splits = r.varint();
            else if (no == 9 && wt == 2) {
                std::string b = r.bytes();
                v.resource_stats = LeafResourceStats::decode(
                    Reader((const uint8_t*)b.data(), b.size()));
            } else r.skip(wt);
        }
        return v;
    }
};

struct FetchDocsRequest {  // search.proto:674
    std::vector<PartialHit> partial_hits;              // 1
    std::vector<SplitIdAndFooterOffsets> split_offsets;  // 3
    std::string index_uri;                             // 4
    std::string doc_mapper;                            // 6
    std::string encode() const {
        Writer w;
        for (auto& h : partial_hits) w.msg_field(1, h.encode());
        for (auto& s : split_offsets) w.msg_field(3, s.encode());
        w.str_field(4, index_uri);
        w.str_field(6, doc_mapper);
        return w.out;
    }
    static FetchDocsRequest decode(const uint8_t* data, size_t n) {
        Reader r(data, n);
        FetchDocsRequest q;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 2) {
                std::string b = r.bytes();
                q.partial_hits.push_back(
                    PartialHit::decode(Reader((const uint8_t*)b.data(), b.size())));
            } else if (no == 3 && wt == 2) {
                std::string b = r.bytes();
                q.split_offsets.push_back(SplitIdAndFooterOffsets::decode(
                    Reader((const uint8_t*)b.data(), b.size())));
            } else if (no == 4 && wt == 2) q.index_uri = r.bytes();
            else if (no == 6 && wt == 2) q.doc_mapper = r.bytes();
            else r.skip(wt);
        }
        return q;
    }
};

struct LeafHit {  // search.proto:553
    std::string leaf_json;                      // 1
    PartialHit partial_hit;                     // 2
    std::optional<std::string> leaf_snippet_json;  // 3
    std::string encode() const {
        Writer w;
        w.str_field(1, leaf_json);
        w.msg_field(2, partial_hit.encode());
        if (leaf_snippet_json) w.str_field(3, *leaf_snippet_json, true);
        return w.out;
    }
    static LeafHit decode(Reader r) {
        LeafHit h;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 2) h.leaf_json = r.bytes();
            else if (no == 2 && wt == 2) {
                std::string b = r.bytes();
                h.partial_hit =
                    PartialHit::decode(Reader((const uint8_t*)b.data(), b.size()));
            } else if (no == 3 && wt == 2) h.leaf_snippet_json = r.bytes();
            else r.skip(wt);
        }
        return h;
    }
};

struct FetchDocsResponse {  // search.proto:695
    std::vector<LeafHit> hits;  // 1
    std::string encode() const {
        Writer w;
        for (auto& h : hits) w.msg_field(1, h.encode());
        return w.out;
    }
    static FetchDocsResponse decode(const uint8_t* data, size_t n) {
        Reader r(data, n);
        FetchDocsResponse q;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 2) {
                std::string b = r.bytes();
                q.hits.push_back(
                    LeafHit::decode(Reader((const uint8_t*)b.data(), b.size())));
            } else r.skip(wt);
        }
        return q;
    }
};

struct LeafListTermsRequest {  // search.proto:732
    ListTermsRequest list_terms_request;              // 1
    std::vector<SplitIdAndFooterOffsets> split_offsets;  // 2
    std::string index_uri;                            // 3
    std::string encode() const {
        Writer w;
        w.msg_field(1, list_terms_request.encode());
        for (auto& s : split_offsets) w.msg_field(2, s.encode());
        w.str_field(3, index_uri);
        return w.out;
    }
    static LeafListTermsRequest decode(const uint8_t* data, size_t n) {
        Reader r(data, n);
        LeafListTermsRequest q;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 2) {
                std::string b = r.bytes();
                q.list_terms_request = ListTermsRequest::decode(
                    Reader((const uint8_t*)b.data(), b.size()));
            } else if (no == 2 && wt == 2) {
                std::string b = r.bytes();
                q.split_offsets.push_back(SplitIdAndFooterOffsets::decode(
                    Reader((const uint8_t*)b.data(), b.size())));
            } else if (no == 3 && wt == 2) q.index_uri = r.bytes();
            else r.skip(wt);
        }
        return q;
    }
};

struct LeafListTermsResponse {  // search.proto:745
    uint64_t num_hits = 0;                       // 1
    std::vector<std::string> terms;              // 2
    std::vector<SplitSearchError> failed_splits; // 3
    uint64_t num_attempted_splits = 0;           // 4
    std::string encode() const {
        Writer w;
        w.u64_field(1, num_hits);
        for (auto& t : terms) w.bytes_field(2, t);
        for (auto& f : failed_splits) w.msg_field(3, f.encode());
        w.u64_field(4, num_attempted_splits);
        return w.out;
    }
    static LeafListTermsResponse decode(const uint8_t* data, size_t n) {
        Reader r(data, n);
        LeafListTermsResponse q;
        while (!r.done()) {
            uint32_t no, wt = r.read_tag(&no);
            if (no == 1 && wt == 0) q.num_hits = r.varint();
            else if (no == 2 && wt == 2) q.terms.push_back(r.bytes());
            else if (no == 3 && wt == 2) {
                std::string b = r.bytes();
                q.failed_splits.push_back(SplitSearchError::decode(
                    Reader((const uint8_t*)b.data(), b.size())));
            } else if (no == 4 && wt == 0) q.num_attempted_splits = r.varint();
            else r.skip(wt);
        }
        return q;
    }
};

}  // namespace pb
