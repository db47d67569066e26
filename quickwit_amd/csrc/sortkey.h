// Monotonic sort-value <-> u64 maps and the PartialHit comparator.
// Restates tantivy-common's MonotonicallyMappableToU64 (f64_to_u64 /
// i64_to_u64) and the quickwit ordering rules of
// docs/internals/sorting.md:14-26 + quickwit-search/src/top_k_collector.rs:
//   - keys compared in the requested SortOrder, None (missing) always last;
//   - ties broken by GlobalDocId (split_id, segment_ord, doc_id) in the
//     order of the first sort field (Desc by default).
#pragma once
#include <algorithm>
#include <climits>
#include <cstdint>
#include <cstring>
#include <string>

#include "pb.h"

namespace qw {

inline uint64_t f64_to_u64(double v) {
    uint64_t bits;
    memcpy(&bits, &v, 8);
    return (bits & (1ULL << 63)) ? ~bits : bits | (1ULL << 63);
}
inline double u64_to_f64(uint64_t u) {
    uint64_t bits = (u & (1ULL << 63)) ? u & ~(1ULL << 63) : ~u;
    double d;
    memcpy(&d, &bits, 8);
    return d;
}
inline uint64_t i64_to_u64(int64_t v) { return uint64_t(v) ^ (1ULL << 63); }
inline int64_t u64_to_i64(uint64_t u) { return int64_t(u ^ (1ULL << 63)); }

// SortByValue -> comparable u64 key (same typed mapping the reference applies
// in collector.rs:180-205). has=false means None.
struct SortKey {
    bool has = false;
    uint64_t key = 0;
};

inline SortKey sort_key_of(const pb::SortByValue& v) {
    SortKey k;
    switch (v.kind) {
        case pb::SortByValue::NONE: break;
        case pb::SortByValue::U64: k = {true, v.u64}; break;
        case pb::SortByValue::I64: k = {true, i64_to_u64(v.i64)}; break;
        case pb::SortByValue::F64: k = {true, f64_to_u64(v.f64)}; break;
        case pb::SortByValue::BOOL: k = {true, v.boolean ? 1ULL : 0ULL}; break;
    }
    return k;
}

// returns -1/0/+1: a before/equal/after b under (order1, order2); order 0=Asc
// 1=Desc (search.proto:295). Missing values sort last under either order
// (sorting.md:19-21).
inline int cmp_key(const SortKey& a, const SortKey& b, int order) {
    if (a.has != b.has) return a.has ? -1 : 1;  // None last
    if (!a.has) return 0;
    if (a.key == b.key) return 0;
    bool a_first = (order == 1) ? (a.key > b.key) : (a.key < b.key);
    return a_first ? -1 : 1;
}

// Cross-type numeric comparison for MIXED dynamic columns: hits from one
// field may carry different SortByValue kinds (u64 beyond i64, i64, f64,
// bool) — the reference's SortValue ordering compares these numerically
// (bool as 0/1). Same-kind pairs keep the exact u64-mapped compare.
inline int cmp_value(const pb::SortByValue& a, const pb::SortByValue& b,
                     int order) {
    if (a.kind == b.kind || a.kind == pb::SortByValue::NONE ||
        b.kind == pb::SortByValue::NONE)
        return cmp_key(sort_key_of(a), sort_key_of(b), order);
    auto sign = [](const pb::SortByValue& v) -> int {
        switch (v.kind) {
            case pb::SortByValue::U64: return v.u64 > 0 ? 1 : 0;
            case pb::SortByValue::I64: return v.i64 > 0 ? 1 : (v.i64 < 0 ? -1 : 0);
            case pb::SortByValue::F64: return v.f64 > 0 ? 1 : (v.f64 < 0 ? -1 : 0);
            default: return v.boolean ? 1 : 0;
        }
    };
    int sa = sign(a), sb = sign(b);
    int c;
    if (sa != sb) c = sa < sb ? -1 : 1;
    else {
        // same sign, different kinds: exact for int-vs-int, via long double
        // against floats (declared approximation beyond 2^53 — DESIGN §7)
        auto mag = [](const pb::SortByValue& v) -> long double {
            switch (v.kind) {
                case pb::SortByValue::U64: return (long double)v.u64;
                case pb::SortByValue::I64: return (long double)v.i64;
                case pb::SortByValue::F64: return (long double)v.f64;
                default: return v.boolean ? 1.0L : 0.0L;
            }
        };
        long double ma = mag(a), mb = mag(b);
        c = ma == mb ? 0 : (ma < mb ? -1 : 1);
    }
    if (c == 0) return 0;
    return order == 1 ? -c : c;
}

// Full PartialHit comparator for leaf top-K and cross-split merge.
inline bool hit_before(const pb::PartialHit& a, const pb::PartialHit& b, int order1,
                       int order2) {
    int c = cmp_value(a.sort_value, b.sort_value, order1);
    if (c) return c < 0;
    c = cmp_value(a.sort_value2, b.sort_value2, order2);
    if (c) return c < 0;
    // GlobalDocId tie-break in the first field's order (sorting.md:14-17)
    int sc = a.split_id.compare(b.split_id);
    if (sc) return order1 == 1 ? sc > 0 : sc < 0;
    if (a.segment_ord != b.segment_ord)
        return order1 == 1 ? a.segment_ord > b.segment_ord : a.segment_ord < b.segment_ord;
    if (a.doc_id != b.doc_id)
        return order1 == 1 ? a.doc_id > b.doc_id : a.doc_id < b.doc_id;
    return false;
}

// field type tags for cursor conversion (SortFieldType analog).
// MIXED = a dynamic field whose docs carry several value types (u64/i64/
// f64/bool): device keys and cursor conversion use the f64-monotonic map
// over the numeric value (bool as 0/1); response values stay typed.
enum class SortFieldKind { NONE, SCORE, U64, I64, DATETIME, F64, STR, MIXED };

// typed SortByValue -> the MIXED column's f64-sortable key domain
inline SortKey mixed_key_of(const pb::SortByValue& v) {
    switch (v.kind) {
        case pb::SortByValue::NONE: return {};
        case pb::SortByValue::U64: return {true, f64_to_u64(double(v.u64))};
        case pb::SortByValue::I64: return {true, f64_to_u64(double(v.i64))};
        case pb::SortByValue::F64: return {true, f64_to_u64(v.f64)};
        default: return {true, f64_to_u64(v.boolean ? 1.0 : 0.0)};
    }
}

// converted search_after cursor value, in the target field's u64-mapped
// domain (SearchAfterSegment::new + convert_to_u64_ff_val,
// collector.rs:214-340). `disabled` = the cursor sorts before every possible
// value in the requested order: the whole search_after filter is off.
struct CursorKey {
    bool disabled = false;
    SortKey key;  // has=false: None (cursor past all valued docs)
};

inline CursorKey convert_cursor_key(const pb::SortByValue& v, SortFieldKind ft,
                                    int order /*0 asc 1 desc*/) {
    CursorKey out;
    if (v.kind == pb::SortByValue::NONE) return out;  // None cursor value
    const bool asc = order == 0;
    auto some = [&](uint64_t k) { out.key = {true, k}; };
    switch (ft) {
        case SortFieldKind::NONE:
            return out;  // unknown sort field: values are all None
        case SortFieldKind::SCORE:
            some(sort_key_of(v).key);
            return out;
        case SortFieldKind::STR:
            // ord cursor arrives as U64 from our own responses
            some(sort_key_of(v).key);
            return out;
        case SortFieldKind::U64:
            switch (v.kind) {
                case pb::SortByValue::U64: some(v.u64); break;
                case pb::SortByValue::I64:
                    if (v.i64 < 0 && asc) out.disabled = true;
                    else if (v.i64 < 0) some(0);  // desc: matches nothing after
                    else some(uint64_t(v.i64));
                    break;
                case pb::SortByValue::F64:
                    if ((v.f64 < 0.0 && asc) ||
                        (v.f64 > 18446744073709551615.0 && !asc))
                        out.disabled = true;
                    else
                        some(v.f64 < 0.0 ? 0
                             : v.f64 >= 18446744073709551615.0
                                 ? ~0ull
                                 : uint64_t(v.f64));
                    break;
                default: some(v.boolean ? 1 : 0); break;
            }
            return out;
        case SortFieldKind::I64:
        case SortFieldKind::DATETIME:
            switch (v.kind) {
                case pb::SortByValue::I64: some(i64_to_u64(v.i64)); break;
                case pb::SortByValue::U64: {
                    uint64_t val = v.u64;
                    if (!asc && val > uint64_t(INT64_MAX)) out.disabled = true;
                    else
                        some(i64_to_u64(int64_t(
                            std::min<uint64_t>(val, uint64_t(INT64_MAX)))));
                    break;
                }
                case pb::SortByValue::F64: {
                    double val = v.f64;
                    if ((val < -9.223372036854776e18 && asc) ||
                        (val > 9.223372036854775e18 && !asc))
                        out.disabled = true;
                    else {
                        int64_t vi = val <= -9.223372036854776e18 ? INT64_MIN
                                     : val >= 9.223372036854775e18
                                         ? INT64_MAX
                                         : int64_t(val);
                        some(i64_to_u64(vi));
                    }
                    break;
                }
                default: some(i64_to_u64(v.boolean ? 1 : 0)); break;
            }
            return out;
        case SortFieldKind::F64:
        case SortFieldKind::MIXED:
            switch (v.kind) {
                case pb::SortByValue::F64: some(f64_to_u64(v.f64)); break;
                case pb::SortByValue::U64: some(f64_to_u64(double(v.u64))); break;
                case pb::SortByValue::I64: some(f64_to_u64(double(v.i64))); break;
                default: some(f64_to_u64(v.boolean ? 1.0 : 0.0)); break;
            }
            return out;
    }
    return out;
}

// search_after filter (top_k_collector.rs:663-700 + SearchAfterSegment
// :821-872): a hit is kept iff the comparison chain
// (sort_value, sort_value2, then — only when the cursor carries a doc
// address — split_id, segment_ord, doc_id) is strictly Less under
// SortOrder::compare (Desc: natural, Asc: reversed; Some > None).
inline bool after_cursor(const pb::PartialHit& h, const pb::PartialHit& c,
                         const CursorKey& k1, const CursorKey& k2, int order1,
                         int order2, bool mixed1 = false, bool mixed2 = false) {
    if (k1.disabled) return true;  // cursor before all values: filter off
    auto cmpo = [](const SortKey& a, const SortKey& b, int order) -> int {
        if (a.has && b.has) {
            if (a.key == b.key) return 0;
            int n = a.key > b.key ? 1 : -1;
            return order == 1 ? n : -n;
        }
        if (a.has) return 1;   // (Some, None) -> Greater
        if (b.has) return -1;  // (None, Some) -> Less
        return 0;
    };
    // MIXED sort fields: compare in the f64-key domain on both sides (the
    // cursor was converted there; hit values stay typed in the response)
    int r = cmpo(mixed1 ? mixed_key_of(h.sort_value) : sort_key_of(h.sort_value),
                 k1.key, order1);
    if (r) return r < 0;
    r = cmpo(mixed2 ? mixed_key_of(h.sort_value2) : sort_key_of(h.sort_value2),
             k2.disabled ? SortKey{} : k2.key, order2);
    if (r) return r < 0;
    if (c.split_id.empty()) return false;  // equal values, no doc tiebreak
    int cs = h.split_id.compare(c.split_id);
    if (order1 != 1) cs = -cs;
    if (cs) return cs < 0;
    if (h.segment_ord != c.segment_ord) {
        bool n = h.segment_ord > c.segment_ord;
        return order1 == 1 ? !n : n;
    }
    if (h.doc_id != c.doc_id) {
        bool n = h.doc_id > c.doc_id;
        return order1 == 1 ? !n : n;
    }
    return false;  // identical to the cursor: excluded (not inclusive)
}

}  // namespace qw
