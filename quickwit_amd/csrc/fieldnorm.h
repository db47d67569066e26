// Fieldnorm 1-byte quantization — Lucene SmallFloat int4 scheme restated
// (tantivy 0.27 @ 86641f7 follows it; dependency not vendored — SURVEY.md §8c).
// Identity for 0..15, then 3-bit mantissa + implicit bit << exponent, floor
// rounding on encode. Pinned by tests/golden/bm25_sort.json at small lengths;
// must stay in lockstep with quickwit_amd/fieldnorm.py.
#pragma once
#include <cstdint>

namespace qw {

constexpr uint64_t fieldnorm_decode(uint8_t id) {
    return id < 8 ? id : (uint64_t)((id & 7) | 8) << ((id >> 3) - 1);
}

struct FieldnormTable {
    uint64_t v[256];
    constexpr FieldnormTable() : v() {
        for (int i = 0; i < 256; ++i) v[i] = fieldnorm_decode(uint8_t(i));
    }
};
inline constexpr FieldnormTable FIELDNORM_TABLE{};

inline uint8_t fieldnorm_encode(uint64_t len) {
    // largest id with decode(id) <= len (table is strictly increasing)
    int lo = 0, hi = 255;
    while (lo < hi) {
        int mid = (lo + hi + 1) >> 1;
        if (FIELDNORM_TABLE.v[mid] <= len) lo = mid;
        else hi = mid - 1;
    }
    return uint8_t(lo);
}

}  // namespace qw
