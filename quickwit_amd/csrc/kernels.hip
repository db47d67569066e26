// HIP/CDNA4 (gfx950) kernels for the leaf-search hot path — DESIGN.md §5.
//
// All kernels are HBM-bound integer/byte work (posting decode, boolean
// combination, BM25 FMA, column predicate scans, bucket histograms); MFMA is
// deliberately unused (nothing here is a dense contraction — BASELINE.json
// north star). Shapes restated from the reference hot loops:
//   - 128-doc bitpacked posting blocks + skip data (tantivy `bitpacking`,
//     SURVEY.md §8a "posting block decode");
//   - union/intersection combination (tantivy boolean scorers);
//   - BM25 fused with decode (leaf.rs:929 searcher.search inner loop);
//   - collect_block counting + top-K threshold (collector.rs:523-562,
//     top_k_collector.rs) — here: per-tile ballot popcount + exact u64
//     (sortkey, doc-tiebreak) selection;
//   - fast-field predicate + aggregation column scans (range_query.rs:86,
//     tantivy aggregation) — coalesced column gathers + LDS histograms.
//
// Execution geometry: one workgroup (256 threads = 4 wave64) owns a tile of
// TILE_DOCS=8192 consecutive doc ids; posting blocks overlapping the tile are
// decoded wave-per-block with funnel-shift bit extraction and a 64-lane
// shuffle prefix-sum for the doc-id deltas.
//
// The main kernel is TEMPLATE-SPECIALIZED over the query shape
// (NS = score/should-count array, NB = must/must_not bitsets, NA = LDS agg
// arrays, NC = candidate collection) with DYNAMIC LDS sized to exactly the
// sections the shape uses. LDS is the occupancy limiter on this path
// (160 KiB/CU): the monolithic struct was ~53 KiB => 3 workgroups/CU for
// every workload; specialized shapes run 4 (BM25+collect, 34 KiB) to 9+
// (agg-only, 17 KiB) workgroups/CU, which is what hides the gather latency.
#include <hip/hip_runtime.h>

#include "gpu_types.h"

namespace qw {

#pragma pack(push, 1)
struct SkipEntryDev {  // mirrors qsplit.h SkipEntry (device-safe copy)
    uint32_t first_doc;
    uint32_t last_doc;
    uint32_t word_off;
    uint8_t id_bits;
    uint8_t tf_bits;
    uint16_t count;
};
#pragma pack(pop)

// ---------------------------------------------------------------- helpers
__device__ __forceinline__ uint32_t lane_id() { return threadIdx.x & 63u; }

__device__ __forceinline__ uint64_t extract_bits(const uint32_t* base, uint64_t bitpos,
                                                 uint32_t w) {
    uint64_t word = bitpos >> 5;
    uint32_t sh = uint32_t(bitpos & 31);
    uint64_t v = (uint64_t(base[word]) | (uint64_t(base[word + 1]) << 32)) >> sh;
    return v & ((1ull << w) - 1ull);  // w <= 32
}

// extract the adjacent pair (j, j+1) of w-bit fields in one 64-bit window
// (valid for w <= 16: window holds >= 33 bits past any 32-bit alignment)
__device__ __forceinline__ void extract_bits_pair(const uint32_t* base, uint32_t j,
                                                  uint32_t w, uint32_t* g0,
                                                  uint32_t* g1) {
    if (w <= 16) {
        uint64_t bitpos = uint64_t(j) * w;
        uint64_t word = bitpos >> 5;
        uint32_t sh = uint32_t(bitpos & 31);
        uint64_t v = (uint64_t(base[word]) | (uint64_t(base[word + 1]) << 32)) >> sh;
        uint32_t mask = (1u << w) - 1u;
        *g0 = uint32_t(v) & mask;
        *g1 = uint32_t(v >> w) & mask;
    } else {
        *g0 = uint32_t(extract_bits(base, uint64_t(j) * w, w));
        *g1 = uint32_t(extract_bits(base, uint64_t(j + 1) * w, w));
    }
}

__device__ __forceinline__ uint32_t wave_incl_scan_u32(uint32_t v) {
    #pragma unroll
    for (int d = 1; d < 64; d <<= 1) {
        uint32_t n = __shfl_up(v, d, 64);
        if ((threadIdx.x & 63u) >= uint32_t(d)) v += n;
    }
    return v;
}

__device__ __forceinline__ uint32_t f32_sortable(float f) {
    uint32_t b = __float_as_uint(f);
    return (b & 0x80000000u) ? ~b : (b | 0x80000000u);
}

__device__ __forceinline__ uint64_t f64_sortable(double d) {
    uint64_t b = __double_as_longlong(d);
    return (b & 0x8000000000000000ull) ? ~b : (b | 0x8000000000000000ull);
}

// ------------------------------------------------------------- LDS layout
// Byte offsets of the sections a given query shape uses; total is the
// dynamic-LDS size the host passes at launch (keep tile_lds_bytes in sync).
template <bool NS, bool NB, bool NA, bool NC>
struct SmemMap {
    static constexpr uint32_t score_off = 0;                       // f32[TILE_DOCS]
    static constexpr uint32_t score_end = NS ? TILE_DOCS * 4 : 0;
    static constexpr uint32_t bits_acc_off = score_end;            // u32[256]
    static constexpr uint32_t bits_not_off = bits_acc_off + (NB ? TILE_DOCS / 8 : 0);
    static constexpr uint32_t bits_end = bits_not_off + (NB ? TILE_DOCS / 8 : 0);
    static constexpr uint32_t bits_m_off = bits_end;               // u32[256]
    static constexpr uint32_t word_pref_off = bits_m_off + (NC ? TILE_DOCS / 8 : 0);
    static constexpr uint32_t coll_end = word_pref_off + (NC ? TILE_DOCS / 8 : 0);
    static constexpr uint32_t agg_hist_off = coll_end;             // u32[2048] x2
    static constexpr uint32_t agg_terms_off = agg_hist_off + (NA ? AGG_LDS_BUCKETS * 4 : 0);
    static constexpr uint32_t agg_end = agg_terms_off + (NA ? AGG_LDS_BUCKETS * 4 : 0);
    static constexpr uint32_t tail_off = agg_end;  // agg_matched[4] wave_base[4] cand_base
    static constexpr uint32_t total = tail_off + 4 * 4 + 4 * 4 + 4;
};

constexpr uint32_t tile_lds_bytes(bool ns, bool nb, bool na, bool nc) {
    return (ns ? TILE_DOCS * 4 : 0) + (nb ? 2 * TILE_DOCS / 8 : 0) +
           (nc ? 2 * TILE_DOCS / 8 : 0) + (na ? 2 * AGG_LDS_BUCKETS * 4 : 0) + 36;
}

// decode every block of `t` overlapping the tile; accumulate into score[] /
// set bits in `bitset`. SCORE: add BM25 weight (or +1 count) into score.
template <bool SCORE>
__device__ void decode_term_tile(const QueryDev& q, const TermDev& t, uint32_t tile,
                                 uint32_t tile_lo, uint32_t tile_hi, float* score,
                                 uint32_t* bitset, bool scoring,
                                 const float* ktab_lds = nullptr,
                                 const uint8_t* norms_lds = nullptr) {
    const uint32_t* ranges = (const uint32_t*)t.ranges_addr;
    uint32_t blo = ranges[tile], bhi = ranges[q.n_tiles + tile];
    if (blo >= bhi) return;
    const SkipEntryDev* __restrict__ skip = (const SkipEntryDev*)(q.split + t.skip_off);
    const uint32_t* __restrict__ payload = (const uint32_t*)(q.split + t.payload_off);
    const uint8_t* __restrict__ norms = t.norms_off ? q.split + t.norms_off : nullptr;
    // LDS-staged fieldnorms for this tile (valid when the term's norms are
    // the staged section) and K tables: per-posting lookups leave HBM
    const bool nlds = norms_lds && t.norms_off &&
                      t.norms_off == q.norms_stage_off;
    const float* __restrict__ ktab =
        (ktab_lds ? ktab_lds : (const float*)(q.scratch + q.ktabs_off)) +
        256 * t.ktab_idx;
    uint32_t wave = threadIdx.x >> 6;
    uint32_t lane = lane_id();
    // one 128-doc block per 32-lane HALF-wave (4 elements/sub-lane): twice
    // the blocks in flight per wave and a 5-step scan instead of 6 — the
    // dependent shuffle chain is the decode critical path
    uint32_t half = lane >> 5;   // which block of the pair this lane works on
    uint32_t sl = lane & 31;     // sub-lane within the half-wave
    uint32_t j0 = 4 * sl;
    uint32_t seg = sl >> 3;
    uint32_t stride2 = 2 * (TILE_THREADS / 64);

    // 4 consecutive w-bit fields starting at bit `pos` of the 128-bit
    // window (lo, hi); valid for w <= 16 (pos <= 31, so pos + 4w <= 95)
    auto ext128 = [](uint64_t lo, uint64_t hi, uint32_t pos,
                     uint32_t w) -> uint32_t {
        uint64_t v;
        if (pos < 64) {
            v = lo >> pos;
            if (pos) v |= hi << (64 - pos);
        } else v = hi >> (pos - 64);
        return uint32_t(v) & ((1u << w) - 1u);
    };

    // SOFTWARE PIPELINE: the per-block decode chain is three dependent
    // global loads (skip entry -> payload words -> fieldnorm gather).
    // Iteration i consumes the skip entry, segment anchor and gap/tf
    // windows prefetched during iteration i-1, overlapping the two leading
    // loads with the previous block's extract/scan/score tail.
    SkipEntryDev e_pf{};
    uint64_t g_lo = 0, g_hi = 0, t_lo = 0, t_hi = 0;
    uint32_t base_pf = 0;
    const bool ntd = q.nt_decode != 0;
    auto ld2 = [&](const uint32_t* p2) -> uint64_t {
        // posting streams are read once: optionally bypass L2 so it stays
        // warm for the per-posting fieldnorm gathers (QW_NT_DECODE)
        if (ntd)
            return uint64_t(__builtin_nontemporal_load(p2)) |
                   (uint64_t(__builtin_nontemporal_load(p2 + 1)) << 32);
        return uint64_t(p2[0]) | (uint64_t(p2[1]) << 32);
    };
    auto prefetch = [&](uint32_t blk) {
        e_pf = skip[blk];
        const uint32_t* idb = payload + e_pf.word_off;
        uint32_t w = e_pf.id_bits;
        base_pf = seg == 0 ? e_pf.first_doc : idb[int(seg) - 4];
        if (w <= 16) {
            uint64_t word = (uint64_t(j0) * w) >> 5;
            g_lo = ld2(idb + word);
            g_hi = ld2(idb + word + 2);
        }
        if (SCORE && e_pf.tf_bits) {
            const uint32_t* tfb = idb + 2 * ((128u * w + 63u) / 64u);
            uint32_t tw = e_pf.tf_bits;
            if (tw <= 16) {
                uint64_t word = (uint64_t(j0) * tw) >> 5;
                t_lo = ld2(tfb + word);
                t_hi = ld2(tfb + word + 2);
            }
        }
    };
    {
        uint32_t first_blk = blo + 2 * wave + half;
        if (first_blk < bhi) prefetch(first_blk);
    }
    for (uint32_t blk0 = blo + 2 * wave; blk0 < bhi; blk0 += stride2) {
        uint32_t blk = blk0 + half;
        bool live = blk < bhi;
        SkipEntryDev e = e_pf;
        uint64_t glo = g_lo, ghi = g_hi, tlo = t_lo, thi = t_hi;
        uint32_t abase = base_pf;
        if (live && blk + stride2 < bhi) prefetch(blk + stride2);
        live = live && !(e.first_doc >= tile_hi || e.last_doc < tile_lo);
        uint32_t g0 = 0, g1 = 0, g2 = 0, g3 = 0;
        const uint32_t* idbase = payload + e.word_off;
        uint32_t w = e.id_bits;
        if (live) {
            if (w <= 16) {
                uint32_t posb = (j0 * w) & 31u;
                g0 = ext128(glo, ghi, posb, w);
                g1 = ext128(glo, ghi, posb + w, w);
                g2 = ext128(glo, ghi, posb + 2 * w, w);
                g3 = ext128(glo, ghi, posb + 3 * w, w);
            } else {
                extract_bits_pair(idbase, j0, w, &g0, &g1);
                extract_bits_pair(idbase, j0 + 2, w, &g2, &g3);
            }
        }
        // inclusive scan of the per-lane gap sums WITHIN the 32-element
        // segment only (8 sub-lanes): the block's v2 anchor words
        // [pad, doc31, doc63, doc95] sit just before the gaps and give the
        // segment base, cutting the dependent shuffle chain from 5 to 3
        uint32_t sum = g0 + g1 + g2 + g3;
        #pragma unroll
        for (int d = 1; d < 8; d <<= 1) {
            uint32_t n = __shfl_up(sum, d, 8);
            if ((sl & 7u) >= uint32_t(d)) sum += n;
        }
        if (!live) continue;
        uint32_t base = abase;  // prefetched segment anchor
        uint32_t doc3 = base + sum;
        uint32_t doc2 = doc3 - g3;
        uint32_t doc1 = doc2 - g2;
        uint32_t doc0 = doc1 - g1;
        float s0 = 1.f, s1 = 1.f, s2 = 1.f, s3 = 1.f;
        if (SCORE && scoring) {
            uint32_t tf0 = 1, tf1 = 1, tf2 = 1, tf3 = 1;
            if (e.tf_bits) {
                const uint32_t* tfbase = idbase + 2 * ((128u * e.id_bits + 63u) / 64u);
                uint32_t tw = e.tf_bits;
                if (tw <= 16) {
                    uint32_t posb = (j0 * tw) & 31u;
                    tf0 = ext128(tlo, thi, posb, tw) + 1u;
                    tf1 = ext128(tlo, thi, posb + tw, tw) + 1u;
                    tf2 = ext128(tlo, thi, posb + 2 * tw, tw) + 1u;
                    tf3 = ext128(tlo, thi, posb + 3 * tw, tw) + 1u;
                } else {
                    extract_bits_pair(tfbase, j0, tw, &tf0, &tf1);
                    extract_bits_pair(tfbase, j0 + 2, tw, &tf2, &tf3);
                    ++tf0; ++tf1; ++tf2; ++tf3;
                }
            }
            // BM25: W * tf / (tf + K[normid]) — same op order as the oracle.
            // Out-of-tile docs (boundary blocks) read a clamped index; their
            // scores are discarded by the range guard below.
            auto nid = [&](uint32_t doc) -> uint32_t {
                if (nlds) {
                    uint32_t li = doc - tile_lo;
                    return norms_lds[li < TILE_DOCS ? li : 0];
                }
                return norms ? norms[min(doc, q.num_docs - 1)] : 1;
            };
            float K0 = ktab[nid(doc0)];
            float K1 = ktab[nid(doc1)];
            float K2 = ktab[nid(doc2)];
            float K3 = ktab[nid(doc3)];
            s0 = t.weight * (float(tf0) / (float(tf0) + K0));
            s1 = t.weight * (float(tf1) / (float(tf1) + K1));
            s2 = t.weight * (float(tf2) / (float(tf2) + K2));
            s3 = t.weight * (float(tf3) / (float(tf3) + K3));
        }
        if (SCORE) {
            #pragma unroll
            for (int el = 0; el < 4; ++el) {
                uint32_t doc = el == 0 ? doc0 : el == 1 ? doc1 : el == 2 ? doc2 : doc3;
                float s = el == 0 ? s0 : el == 1 ? s1 : el == 2 ? s2 : s3;
                if (j0 + el < e.count && doc >= tile_lo && doc < tile_hi)
                    atomicAdd(&score[doc - tile_lo], s);
            }
        } else {
            // bitset role: a lane's 4 consecutive docs often share a 32-doc
            // word on dense terms (gaps of 2-3) — merge the bits locally so
            // the same-address LDS atomicOr serialization drops ~4x
            uint32_t pw = 0xFFFFFFFFu, pm = 0;
            #pragma unroll
            for (int el = 0; el < 4; ++el) {
                uint32_t doc = el == 0 ? doc0 : el == 1 ? doc1 : el == 2 ? doc2 : doc3;
                if (j0 + el < e.count && doc >= tile_lo && doc < tile_hi) {
                    uint32_t li = doc - tile_lo;
                    uint32_t wd = li >> 5;
                    if (wd == pw) pm |= 1u << (li & 31);
                    else {
                        if (pm) atomicOr(&bitset[pw], pm);
                        pw = wd;
                        pm = 1u << (li & 31);
                    }
                }
            }
            if (pm) atomicOr(&bitset[pw], pm);
        }
    }
}

__device__ __forceinline__ bool eval_pred(const QueryDev& q, const PredDev& p,
                                          uint32_t doc) {
    bool ok = true;
    if (p.nulls_off) {
        const uint64_t* nulls = (const uint64_t*)(q.split + p.nulls_off);
        ok = (nulls[doc >> 6] >> (doc & 63)) & 1;
    }
    if (p.type == PRED_BITSET) {
        const uint32_t* bm = (const uint32_t*)p.abs_bitmap;
        ok = (bm[doc >> 5] >> (doc & 31)) & 1;
        return (p.flags & PRED_NEGATED) ? !ok : ok;
    }
    if (ok && p.type == PRED_RANGE_F64) {
        double v = ((const double*)(q.split + p.values_off))[doc];
        double lo = __longlong_as_double(p.lo), hi = __longlong_as_double(p.hi);
        if (p.flags & PRED_LO_INCLUDED) ok &= v >= lo;
        if (p.flags & PRED_LO_EXCLUDED) ok &= v > lo;
        if (p.flags & PRED_HI_INCLUDED) ok &= v <= hi;
        if (p.flags & PRED_HI_EXCLUDED) ok &= v < hi;
        return (p.flags & PRED_NEGATED) ? !ok : ok;
    }
    if (p.offsets_off) {
        // multi-valued str column: presence = any value; range = any match
        const uint32_t* offs = (const uint32_t*)(q.split + p.offsets_off);
        uint32_t s = offs[doc], e = offs[doc + 1];
        if (p.type == PRED_PRESENCE) ok = e > s;
        else {
            const uint8_t* col = q.split + p.values_off;
            ok = false;
            for (uint32_t pos = s; pos < e && !ok; ++pos) {
                uint64_t v = p.value_width == 1   ? col[pos]
                             : p.value_width == 2 ? ((const uint16_t*)col)[pos]
                                                  : ((const uint32_t*)col)[pos];
                bool m2 = true;
                if (p.flags & PRED_LO_INCLUDED) m2 &= v >= uint64_t(p.lo);
                if (p.flags & PRED_LO_EXCLUDED) m2 &= v > uint64_t(p.lo);
                if (p.flags & PRED_HI_INCLUDED) m2 &= v <= uint64_t(p.hi);
                if (p.flags & PRED_HI_EXCLUDED) m2 &= v < uint64_t(p.hi);
                ok |= m2;
            }
        }
        return (p.flags & PRED_NEGATED) ? !ok : ok;
    }
    if (ok && p.type != PRED_PRESENCE) {
        if (p.type == PRED_RANGE_U64) {
            const uint8_t* col = q.split + p.values_off;
            uint64_t v;
            switch (p.value_width) {  // str ord columns are 1/2/4 wide
                case 1: v = col[doc]; break;
                case 2: v = ((const uint16_t*)col)[doc]; break;
                case 4: v = ((const uint32_t*)col)[doc]; break;
                default: v = ((const uint64_t*)col)[doc]; break;
            }
            if (p.flags & PRED_LO_INCLUDED) ok &= v >= uint64_t(p.lo);
            if (p.flags & PRED_LO_EXCLUDED) ok &= v > uint64_t(p.lo);
            if (p.flags & PRED_HI_INCLUDED) ok &= v <= uint64_t(p.hi);
            if (p.flags & PRED_HI_EXCLUDED) ok &= v < uint64_t(p.hi);
        } else {
            int64_t v = ((const int64_t*)(q.split + p.values_off))[doc];
            if (p.flags & PRED_LO_INCLUDED) ok &= v >= p.lo;
            if (p.flags & PRED_LO_EXCLUDED) ok &= v > p.lo;
            if (p.flags & PRED_HI_INCLUDED) ok &= v <= p.hi;
            if (p.flags & PRED_HI_EXCLUDED) ok &= v < p.hi;
        }
    }
    return (p.flags & PRED_NEGATED) ? !ok : ok;
}

__device__ __forceinline__ double agg_value(const QueryDev& q, uint64_t values_off,
                                            uint32_t width, uint32_t is_i64,
                                            uint32_t doc) {
    const uint8_t* col = q.split + values_off;
    switch (width) {
        case 1: return double(col[doc]);
        case 2: return double(((const uint16_t*)col)[doc]);
        case 4: return double(((const uint32_t*)col)[doc]);
        default:
            return is_i64 == 2 ? ((const double*)col)[doc]
                   : is_i64   ? double(((const int64_t*)col)[doc])
                              : double(((const uint64_t*)col)[doc]);
    }
}

// accumulate one value into a 40B stats slot {count,sum,min_s,max_s,sum_sq}
__device__ __forceinline__ void stats_slot_add(uint8_t* slot, double v) {
    atomicAdd((unsigned long long*)slot, 1ull);
    atomicAdd((double*)(slot + 8), v);
    atomicMin((unsigned long long*)(slot + 16),
              (unsigned long long)f64_sortable(v));
    atomicMax((unsigned long long*)(slot + 24),
              (unsigned long long)f64_sortable(v));
    atomicAdd((double*)(slot + 32), v * v);
}

// per-term stats sub-aggs: bucket = the term ord (mirrors the histo subs)
__device__ __forceinline__ void terms_subs_add(const QueryDev& q, const AggDev& a,
                                               uint64_t ord, uint32_t d) {
    for (uint32_t si = 0; si < a.n_sub; ++si) {
        if (!a.sub_width[si]) continue;  // missing sub column
        if (a.sub_nulls_off[si]) {
            const uint64_t* sn = (const uint64_t*)(q.split + a.sub_nulls_off[si]);
            if (!((sn[d >> 6] >> (d & 63)) & 1)) continue;
        }
        double sv = agg_value(q, a.sub_values_off[si], a.sub_width[si],
                              a.sub_is_i64[si], d);
        stats_slot_add(q.results + a.sub_out + (ord * a.n_sub + si) * 40, sv);
    }
}

// DDSketch slot for value v: 0 = zero bucket (v < 1e-9), else 1 + index of
// the first boundary >= v in the host-computed gamma^k table (identical
// doubles to the oracle's perc_key_for => bit-identical bucketing)
__device__ __forceinline__ uint32_t perc_slot(const QueryDev& q, const AggDev& a,
                                              double v) {
    if (v < 1e-9) return 0;
    const double* B = (const double*)(q.scratch + a.p_bound_off);
    uint32_t lo = 0, hi = a.p_n_keys - 1;
    while (lo < hi) {
        uint32_t mid = (lo + hi) >> 1;
        if (v <= B[mid]) hi = mid;
        else lo = mid + 1;
    }
    return lo + 1;
}

__device__ __forceinline__ uint64_t agg_ord(const QueryDev& q, uint64_t values_off,
                                            uint32_t width, uint32_t doc) {
    const uint8_t* col = q.split + values_off;
    switch (width) {
        case 1: return col[doc];
        case 2: return ((const uint16_t*)col)[doc];
        default: return ((const uint32_t*)col)[doc];
    }
}

// primary sort key of a matched doc for WIDE candidate records: the same
// monotonic u64 maps the host comparator uses (sortkey.h); None (missing
// fast-field value) maps to 0 so it sorts last under either order — exact
// ordering among boundary ties is re-established on host (sorting.md:14-26).
__device__ __forceinline__ uint64_t wide_sort_key(const QueryDev& q, uint32_t d,
                                                  float sc) {
    if (q.sort_src == 0) return 0;  // unknown sort field: None for every doc
    if (q.sort_nulls_off) {
        const uint64_t* nulls = (const uint64_t*)(q.split + q.sort_nulls_off);
        if (!((nulls[d >> 6] >> (d & 63)) & 1)) return 0;
    }
    uint64_t S = 0;
    if (q.sort_src == 1) {
        S = uint64_t(f32_sortable(sc)) << 32;
    } else {
        const uint8_t* col = q.split + q.sort_values_off;
        switch (q.sort_width) {
            case 1: S = col[d]; break;
            case 2: S = ((const uint16_t*)col)[d]; break;
            case 4: S = ((const uint32_t*)col)[d]; break;
            default: S = ((const uint64_t*)col)[d]; break;
        }
        if (q.sort_src == 3) S ^= (1ull << 63);  // i64_to_u64
        else if (q.sort_src == 4) S = f64_sortable(__longlong_as_double(S));
    }
    return q.sort_asc ? ~S : S;
}

// histogram bucket index relative to base_index, or -1 when out of range /
// null. int_fast: exact 64-bit floor-division (host proved it bit-equal to
// the oracle's floor((v-offset)/interval) double path for this column's
// value range — product.cpp plan_aggs); else the double path verbatim.
__device__ __forceinline__ int64_t histo_bucket(const QueryDev& q, const AggDev& a,
                                                uint32_t doc) {
    if (a.int_fast) {
        const int64_t* col = (const int64_t*)(q.split + a.values_off);
        int64_t num = col[doc] - a.i_offset;
        // floor division by positive i_interval via reciprocal multiply +
        // exact integer correction (cheaper than i64 div on CDNA4)
        int64_t idx = int64_t(floor(double(num) * a.inv_interval));
        if (idx * a.i_interval > num) --idx;
        else if ((idx + 1) * a.i_interval <= num) ++idx;
        return idx - a.base_index;
    }
    double v = agg_value(q, a.values_off, a.value_width, a.value_is_i64, doc);
    return int64_t(floor((v - a.offset) / a.interval)) - a.base_index;
}

// ---------------------------------------------------------------- main kernel
// One instantiation per query shape; LDS sections per SmemMap. do_aggs /
// collect of the old interface became NA / NC.
template <bool NS, bool NB, bool NA, bool NC>
__global__ void __launch_bounds__(TILE_THREADS) k_leaf_tile_t(QueryDev q,
                                                              uint32_t tile_base,
                                                              uint32_t tile_end,
                                                              uint32_t do_count) {
    using M = SmemMap<NS, NB, NA, NC>;
    extern __shared__ uint8_t smem[];
    float* sc_score = (float*)(smem + M::score_off);
    uint32_t* sc_bits_acc = (uint32_t*)(smem + M::bits_acc_off);
    uint32_t* sc_bits_not = (uint32_t*)(smem + M::bits_not_off);
    uint32_t* sc_bits_m = (uint32_t*)(smem + M::bits_m_off);
    uint32_t* sc_word_pref = (uint32_t*)(smem + M::word_pref_off);
    uint32_t* sc_agg_hist = (uint32_t*)(smem + M::agg_hist_off);
    uint32_t* sc_agg_terms = (uint32_t*)(smem + M::agg_terms_off);
    uint32_t* sc_agg_matched = (uint32_t*)(smem + M::tail_off);
    uint32_t* sc_wave_base = sc_agg_matched + 4;
    uint32_t* sc_cand_base = sc_wave_base + 4;
    // runtime-optional LDS staging (appended after the compile-time map;
    // host adds the bytes to the launch's dynamic-LDS size): per-tile
    // fieldnorms u8[TILE_DOCS] + BM25 K tables f32[256*n_ktabs]
    uint8_t* sc_norms = nullptr;
    float* sc_ktabs = nullptr;
    if (NS) {
        uint32_t soff = M::total;
        if (q.norms_stage_off) {
            sc_norms = smem + soff;
            soff += TILE_DOCS;
        }
        if (q.scoring && q.n_ktabs) {
            sc_ktabs = (float*)(smem + soff);
            const float* g = (const float*)(q.scratch + q.ktabs_off);
            for (uint32_t i = threadIdx.x; i < 256u * q.n_ktabs;
                 i += TILE_THREADS)
                sc_ktabs[i] = g[i];
        }
    }

    const TermDev* __restrict__ terms = (const TermDev*)(q.scratch + q.terms_off);
    const PredDev* __restrict__ preds = (const PredDev*)(q.scratch + q.preds_off);
    const AggDev* __restrict__ aggs = (const AggDev*)(q.scratch + q.aggs_off);
    uint32_t* tile_counts = (uint32_t*)(q.results + q.tile_counts_off);
    uint32_t* cand_count = (uint32_t*)(q.results + q.cand_count_off);
    uint64_t* cand = (uint64_t*)(q.results + q.cand_off);

    // agg LDS arrays accumulate across ALL tiles this workgroup owns and
    // flush ONCE at the end (global flush atomics scale with gridDim, not
    // n_tiles — the host caps the agg grid so a workgroup owns several
    // tiles). u32 counts cannot overflow: a workgroup owns < 2^31 docs.
    if (NA && q.n_aggs) {
        for (uint32_t i = threadIdx.x; i < AGG_LDS_BUCKETS; i += TILE_THREADS) {
            sc_agg_hist[i] = 0;
            sc_agg_terms[i] = 0;
        }
        if (threadIdx.x < 4) sc_agg_matched[threadIdx.x] = 0;
    }
    // agg_fast: hoist the (uniform) descriptor fields into registers once
    const int64_t* af_col = nullptr;
    const uint8_t* af_tcol = nullptr;
    int64_t af_ioff = 0, af_ivl = 1, af_base = 0;
    double af_inv = 0;
    uint32_t af_twidth = 0, af_hrep = 0, af_trep = 0, af_par = 0;
    bool af_terms = false;
    if (NA && q.agg_fast) {
        const AggDev& a0 = aggs[0];
        af_par = lane_id() & 1u;  // copy index for 2-way replication
        if (a0.kind == AGGD_HISTO) {
            af_col = (const int64_t*)(q.split + a0.values_off);
            af_ioff = a0.i_offset;
            af_ivl = a0.i_interval;
            af_inv = a0.inv_interval;
            af_base = a0.base_index;
            af_hrep = a0.lds_rep;
            if (q.n_aggs > 1) {
                af_terms = true;
                af_tcol = q.split + aggs[1].values_off;
                af_twidth = aggs[1].value_width;
                af_trep = aggs[1].lds_rep;
            }
        } else {  // terms-only straight-line path (af_col stays null)
            af_terms = true;
            af_tcol = q.split + a0.values_off;
            af_twidth = a0.value_width;
            af_trep = a0.lds_rep;
        }
    }

    for (uint32_t tile = tile_base + blockIdx.x; tile < tile_end; tile += gridDim.x) {
        uint32_t tile_lo = tile * TILE_DOCS;
        uint32_t tile_hi = min(tile_lo + TILE_DOCS, q.num_docs);
        bool have_should = q.msm > 0;  // shoulds REQUIRED (msm>=1); msm==0 =>
                                       // shoulds optional, only add scores
        // ---- zero LDS (+ stage this tile's fieldnorms)
        if (NS)
            for (uint32_t i = threadIdx.x; i < TILE_DOCS; i += TILE_THREADS)
                sc_score[i] = 0.f;
        if (NS && sc_norms) {
            const uint32_t* g32 =
                (const uint32_t*)(q.split + q.norms_stage_off + tile_lo);
            uint32_t* d32 = (uint32_t*)sc_norms;
            uint32_t words = (tile_hi - tile_lo + 3) >> 2;  // split img has
            for (uint32_t i = threadIdx.x; i < words; i += TILE_THREADS)
                d32[i] = g32[i];                            // +64B pad
        }
        if (NB)
            for (uint32_t i = threadIdx.x; i < TILE_DOCS / 32; i += TILE_THREADS) {
                sc_bits_acc[i] = 0;
                sc_bits_not[i] = 0;
            }
        if (NC)
            for (uint32_t i = threadIdx.x; i < TILE_DOCS / 32; i += TILE_THREADS)
                sc_bits_m[i] = 0;
        __syncthreads();

        // ---- should terms: score / count union
        if (NS)
            for (uint32_t t = 0; t < q.n_terms; ++t)
                if (terms[t].role == ROLE_SHOULD)
                    decode_term_tile<true>(q, terms[t], tile, tile_lo, tile_hi,
                                           sc_score, nullptr, q.scoring,
                                           sc_ktabs, sc_norms);
        __syncthreads();

        // ---- must groups: union the group's terms into a temp bitset,
        // AND the groups (q.n_must = group count; a group is one plain term
        // or a wildcard/term_set expansion — OR within, AND across)
        if (NB) {
            for (uint32_t g = 0; g < q.n_must; ++g) {
                // temp bitset lives in bits_not while must_nots are not yet done
                uint32_t* tmp = sc_bits_not;
                for (uint32_t i = threadIdx.x; i < TILE_DOCS / 32; i += TILE_THREADS)
                    tmp[i] = 0;
                __syncthreads();
                for (uint32_t t = 0; t < q.n_terms; ++t) {
                    if (terms[t].role != ROLE_MUST || terms[t].grp != g) continue;
                    decode_term_tile<false>(q, terms[t], tile, tile_lo, tile_hi,
                                            nullptr, tmp, false);
                    if (NS && q.scoring && terms[t].weight != 0.f)
                        decode_term_tile<true>(q, terms[t], tile, tile_lo, tile_hi,
                                               sc_score, nullptr, true,
                                               sc_ktabs, sc_norms);
                }
                __syncthreads();
                for (uint32_t i = threadIdx.x; i < TILE_DOCS / 32; i += TILE_THREADS)
                    sc_bits_acc[i] = g == 0 ? tmp[i] : (sc_bits_acc[i] & tmp[i]);
                __syncthreads();
            }
            if (q.n_must_not) {
                for (uint32_t i = threadIdx.x; i < TILE_DOCS / 32; i += TILE_THREADS)
                    sc_bits_not[i] = 0;
                __syncthreads();
                for (uint32_t t = 0; t < q.n_terms; ++t)
                    if (terms[t].role == ROLE_MUST_NOT)
                        decode_term_tile<false>(q, terms[t], tile, tile_lo, tile_hi,
                                                nullptr, sc_bits_not, false);
                __syncthreads();
            }
        }

        // ---- pure-aggregation tile (match_all, no predicates, no
        // collection): tight loop, every doc matches, count is the tile size
        if (NA && !NS && !NB && !NC && q.agg_fast && q.match_all && !q.n_preds) {
            #pragma unroll 4
            for (uint32_t d = tile_lo + threadIdx.x; d < tile_hi; d += TILE_THREADS) {
                if (af_col) {
                    int64_t num =
                        (q.agg_nt ? __builtin_nontemporal_load(&af_col[d])
                                  : af_col[d]) -
                        af_ioff;
                    int64_t idx = int64_t(floor(double(num) * af_inv));
                    if (idx * af_ivl > num) --idx;
                    else if ((idx + 1) * af_ivl <= num) ++idx;
                    uint32_t hb = uint32_t(idx - af_base);
                    atomicAdd(&sc_agg_hist[af_hrep == 2 ? hb * 2 + af_par : hb],
                              1u);
                }
                if (af_terms) {
                    uint64_t o = af_twidth == 2 ? ((const uint16_t*)af_tcol)[d]
                                 : (af_twidth == 1 ? af_tcol[d]
                                                   : ((const uint32_t*)af_tcol)[d]);
                    atomicAdd(&sc_agg_terms[af_trep == 2 ? o * 2 + af_par : o], 1u);
                }
            }
            if (do_count && threadIdx.x == 0) tile_counts[tile] = tile_hi - tile_lo;
            __syncthreads();
            continue;
        }

        // ---- epilogue: matched test, count, candidates, aggregations
        uint32_t local_count = 0;
        for (uint32_t d = tile_lo + threadIdx.x; d < tile_hi; d += TILE_THREADS) {
            uint32_t li = d - tile_lo;
            bool m;
            float sc = 0.f;
            if (NS && have_should) {
                // shoulds required: scoring path assumes msm==1 and all
                // weights > 0 (host rejects anything else — product.cpp)
                if (q.scoring) {
                    sc = sc_score[li];
                    m = sc > 0.f;
                } else m = sc_score[li] >= float(q.msm);
                if (NB && m && q.n_must) m = (sc_bits_acc[li >> 5] >> (li & 31)) & 1;
            } else {
                // base = match_all (no must terms; preds filter below) or the
                // must-term intersection; optional shoulds only add scores
                m = q.match_all ? true
                                : (NB && q.n_must > 0 &&
                                   ((sc_bits_acc[li >> 5] >> (li & 31)) & 1));
                if (NS && q.scoring && m) sc = sc_score[li];
            }
            if (NB && m && q.n_must_not) m = !((sc_bits_not[li >> 5] >> (li & 31)) & 1);
            for (uint32_t p = 0; p < q.n_preds; ++p) {
                const PredDev& pr = preds[p];
                if (m || (pr.flags & PRED_EAGER))
                    m = eval_pred(q, pr, d) && m;
            }

            if (m) {
                ++local_count;
                if (NA && q.agg_fast) {
                    // straight-line: exact floor-div histogram bucket (bounds
                    // cover the column's [min,max] => no range check) +
                    // optional ord-count table, both LDS
                    if (af_col) {
                        int64_t num = af_col[d] - af_ioff;
                        int64_t idx = int64_t(floor(double(num) * af_inv));
                        if (idx * af_ivl > num) --idx;
                        else if ((idx + 1) * af_ivl <= num) ++idx;
                        uint32_t hb = uint32_t(idx - af_base);
                        // same lds_rep mapping as the pure-agg tile loop:
                        // the end-of-kernel flush sums interleaved pairs
                        // when lds_rep==2, so indices must match there too
                        atomicAdd(
                            &sc_agg_hist[af_hrep == 2 ? hb * 2 + af_par : hb],
                            1u);
                    }
                    if (af_terms) {
                        uint64_t o = af_twidth == 2
                                         ? ((const uint16_t*)af_tcol)[d]
                                         : (af_twidth == 1
                                                ? af_tcol[d]
                                                : ((const uint32_t*)af_tcol)[d]);
                        atomicAdd(
                            &sc_agg_terms[af_trep == 2 ? o * 2 + af_par : o],
                            1u);
                    }
                } else if (NA)
                    for (uint32_t ai = 0; ai < q.n_aggs; ++ai) {
                        const AggDev& a = aggs[ai];
                        if (a.nulls_off) {
                            const uint64_t* nulls = (const uint64_t*)(q.split + a.nulls_off);
                            if (!((nulls[d >> 6] >> (d & 63)) & 1)) continue;
                        }
                        if (a.kind == AGGD_TERMS) {
                            if (!a.n_buckets) continue;  // missing/non-str column
                            if (a.offsets_off) {
                                // multi-valued: one count per (doc, value) in
                                // both the per-bucket counts AND matched
                                // (e2-s below) — matched must sum the same
                                // units as the buckets so host-side
                                // sum_other_doc_count = matched - Σ(kept
                                // bucket counts) stays consistent
                                const uint32_t* offs =
                                    (const uint32_t*)(q.split + a.offsets_off);
                                uint32_t s = offs[d], e2 = offs[d + 1];
                                if (s == e2) continue;  // no value
                                const uint8_t* col = q.split + a.values_off;
                                for (uint32_t pos = s; pos < e2; ++pos) {
                                    uint64_t o =
                                        a.value_width == 1 ? col[pos]
                                        : a.value_width == 2
                                            ? ((const uint16_t*)col)[pos]
                                            : ((const uint32_t*)col)[pos];
                                    if (a.lds_slot == 1)
                                        atomicAdd(
                                            &sc_agg_terms[a.lds_rep == 2
                                                              ? o * 2 +
                                                                    (lane_id() & 1u)
                                                              : o],
                                            1u);
                                    else
                                        atomicAdd((unsigned long long*)(q.results +
                                                                        a.counts_out) +
                                                      o,
                                                  1ull);
                                    if (a.n_sub) terms_subs_add(q, a, o, d);
                                }
                                if (ai < 4)
                                    atomicAdd(&sc_agg_matched[ai], e2 - s);
                                else
                                    atomicAdd((unsigned long long*)(q.results +
                                                                    a.matched_out),
                                              (unsigned long long)(e2 - s));
                                continue;
                            }
                            uint64_t o = agg_ord(q, a.values_off, a.value_width, d);
                            if (a.lds_slot == 1)
                                atomicAdd(&sc_agg_terms[a.lds_rep == 2
                                                            ? o * 2 + (lane_id() & 1u)
                                                            : o],
                                          1u);
                            else
                                atomicAdd((unsigned long long*)(q.results + a.counts_out) + o,
                                          1ull);
                            if (a.n_sub) terms_subs_add(q, a, o, d);
                            // docs-with-value == matched docs for a
                            // non-nullable column: host uses num_hits instead
                            if (a.nulls_off) {
                                if (ai < 4)  // LDS count, flushed at the end
                                    atomicAdd(&sc_agg_matched[ai], 1u);
                                else
                                    atomicAdd((unsigned long long*)(q.results +
                                                                    a.matched_out),
                                              1ull);
                            }
                        } else if (a.kind == AGGD_TERMS_NUM) {
                            // numeric-column terms: global open-addressing
                            // hash of the value's sortable bits (layout in
                            // gpu_types.h). Null docs were skipped above.
                            if (!a.n_buckets) continue;
                            uint64_t raw =
                                ((const uint64_t*)(q.split + a.values_off))[d];
                            uint64_t v = raw;
                            if (a.value_is_i64 == 1) v = raw ^ (1ull << 63);
                            else if (a.value_is_i64 == 2)
                                v = f64_sortable(__longlong_as_double(raw));
                            unsigned long long* tab =
                                (unsigned long long*)(q.results + a.counts_out);
                            uint32_t slots = (a.n_buckets - 2) >> 1;
                            const unsigned long long SENT = ~0ull;
                            if (v == SENT) {
                                atomicAdd(&tab[2 * slots], 1ull);
                            } else {
                                uint64_t h = v * 0x9E3779B97F4A7C15ull;
                                h ^= h >> 32;
                                uint32_t sl = uint32_t(h) & (slots - 1);
                                bool done = false;
                                for (uint32_t pr = 0; pr < slots; ++pr) {
                                    unsigned long long cur = tab[2 * sl];
                                    if (cur == SENT)
                                        cur = atomicCAS(&tab[2 * sl], SENT,
                                                        (unsigned long long)v);
                                    if (cur == SENT ||
                                        cur == (unsigned long long)v) {
                                        atomicAdd(&tab[2 * sl + 1], 1ull);
                                        done = true;
                                        break;
                                    }
                                    sl = (sl + 1) & (slots - 1);
                                }
                                if (!done) tab[2 * slots + 1] = 1ull;  // overflow
                            }
                            if (a.nulls_off) {
                                if (ai < 4)
                                    atomicAdd(&sc_agg_matched[ai], 1u);
                                else
                                    atomicAdd((unsigned long long*)(q.results +
                                                                    a.matched_out),
                                              1ull);
                            }
                        } else if (a.kind == AGGD_COMP) {
                            // composite: pack per-source components into a
                            // <=63-bit key (ordering == composite tuple asc),
                            // count via the same hash table as TERMS_NUM
                            if (!a.n_buckets) continue;
                            uint64_t key = 0;
                            bool okc = true;
                            for (uint32_t si = 0; si < a.n_sub; ++si) {
                                uint64_t comp = 0;
                                bool present = true;
                                if (a.sub_nulls_off[si]) {
                                    const uint64_t* nu =
                                        (const uint64_t*)(q.split +
                                                          a.sub_nulls_off[si]);
                                    present = (nu[d >> 6] >> (d & 63)) & 1;
                                }
                                uint64_t miss = (a.c_missing >> si) & 1;
                                if (!present) {
                                    if (!miss) { okc = false; break; }
                                } else if ((a.c_histo >> si) & 1) {
                                    double v = agg_value(q, a.sub_values_off[si],
                                                         a.sub_width[si],
                                                         a.sub_is_i64[si], d);
                                    int64_t idx =
                                        int64_t(floor((v - a.c_offset[si]) /
                                                      a.c_interval[si])) -
                                        a.c_base[si];
                                    comp = uint64_t(idx) + miss;
                                } else {
                                    comp = agg_ord(q, a.sub_values_off[si],
                                                   a.sub_width[si], d) + miss;
                                }
                                key |= comp << a.c_shift[si];
                            }
                            if (!okc) continue;
                            unsigned long long* tab =
                                (unsigned long long*)(q.results + a.counts_out);
                            uint32_t slots = (a.n_buckets - 2) >> 1;
                            const unsigned long long SENT = ~0ull;
                            uint64_t h = key * 0x9E3779B97F4A7C15ull;
                            h ^= h >> 32;
                            uint32_t sl = uint32_t(h) & (slots - 1);
                            bool done = false;
                            for (uint32_t pr = 0; pr < slots; ++pr) {
                                unsigned long long cur = tab[2 * sl];
                                if (cur == SENT)
                                    cur = atomicCAS(&tab[2 * sl], SENT,
                                                    (unsigned long long)key);
                                if (cur == SENT ||
                                    cur == (unsigned long long)key) {
                                    atomicAdd(&tab[2 * sl + 1], 1ull);
                                    done = true;
                                    break;
                                }
                                sl = (sl + 1) & (slots - 1);
                            }
                            if (!done) tab[2 * slots + 1] = 1ull;  // overflow
                        } else if (a.kind == AGGD_PERC) {
                            if (!a.n_buckets) continue;
                            double v = agg_value(q, a.values_off, a.value_width,
                                                 a.value_is_i64, d);
                            uint32_t sj = perc_slot(q, a, v);
                            atomicAdd((unsigned long long*)(q.results +
                                                            a.counts_out) +
                                          sj,
                                      1ull);
                        } else if (a.kind == AGGD_METRIC) {
                            if (!a.values_off) continue;
                            double v = agg_value(q, a.values_off, a.value_width,
                                                 a.value_is_i64, d);
                            uint8_t* slot = q.results + a.counts_out;
                            atomicAdd((unsigned long long*)slot, 1ull);
                            atomicAdd((double*)(slot + 8), v);
                            atomicMin((unsigned long long*)(slot + 16),
                                      (unsigned long long)f64_sortable(v));
                            atomicMax((unsigned long long*)(slot + 24),
                                      (unsigned long long)f64_sortable(v));
                            atomicAdd((double*)(slot + 32), v * v);
                        } else if (a.kind == AGGD_RANGE) {
                            if (!a.n_ranges) continue;
                            double v = agg_value(q, a.values_off, a.value_width,
                                                 a.value_is_i64, d);
                            for (uint32_t ri = 0; ri < a.n_ranges; ++ri) {
                                bool in = true;
                                if ((a.r_has_from >> ri) & 1) in &= v >= a.r_from[ri];
                                if ((a.r_has_to >> ri) & 1) in &= v < a.r_to[ri];
                                if (in)
                                    atomicAdd((unsigned long long*)(q.results +
                                                                    a.counts_out) +
                                                  ri,
                                              1ull);
                            }
                        } else {
                            int64_t idx = histo_bucket(q, a, d);
                            if (idx < 0 || idx >= int64_t(a.n_buckets)) continue;
                            if (a.lds_slot == 0)
                                atomicAdd(&sc_agg_hist[a.lds_rep == 2
                                                           ? uint32_t(idx) * 2 +
                                                                 (lane_id() & 1u)
                                                           : uint32_t(idx)],
                                          1u);
                            else
                                atomicAdd((unsigned long long*)(q.results + a.counts_out) +
                                              idx, 1ull);
                            for (uint32_t si = 0; si < a.n_sub; ++si) {
                                if (!a.sub_width[si]) continue;  // missing col
                                if (a.sub_nulls_off[si]) {
                                    const uint64_t* sn =
                                        (const uint64_t*)(q.split + a.sub_nulls_off[si]);
                                    if (!((sn[d >> 6] >> (d & 63)) & 1)) continue;
                                }
                                double sv = agg_value(q, a.sub_values_off[si],
                                                      a.sub_width[si], a.sub_is_i64[si], d);
                                if (si == a.p_si) {
                                    // percentiles sub: DDSketch key count
                                    // (word 0 = zero bucket, binary search
                                    // over the host boundary table)
                                    uint32_t sj = perc_slot(q, a, sv);
                                    atomicAdd((unsigned long long*)(q.results +
                                                                    a.p_out) +
                                                  uint64_t(idx) *
                                                      (a.p_n_keys + 1) +
                                                  sj,
                                              1ull);
                                    continue;
                                }
                                uint8_t* slot = q.results + a.sub_out +
                                                (uint64_t(idx) * a.n_sub + si) * 40;
                                atomicAdd((unsigned long long*)slot, 1ull);
                                atomicAdd((double*)(slot + 8), sv);
                                atomicMin((unsigned long long*)(slot + 16),
                                          (unsigned long long)f64_sortable(sv));
                                atomicMax((unsigned long long*)(slot + 24),
                                          (unsigned long long)f64_sortable(sv));
                                atomicAdd((double*)(slot + 32), sv * sv);
                            }
                        }
                    }
            }
            // phase A of collection: record matches in an LDS bitset (no
            // global traffic; one ballot + direct word store per wave —
            // Guideline 12: never a per-wave global atomic in the hot loop)
            if (NC) {
                uint64_t mb = __ballot(m);
                uint32_t wbase = (li - lane_id()) >> 5;
                if (lane_id() == 0) sc_bits_m[wbase] = uint32_t(mb);
                else if (lane_id() == 32) sc_bits_m[wbase + 1] = uint32_t(mb >> 32);
            }
        }
        // ---- per-tile count + phase B of collection (one global atomic per
        // TILE reserves contiguous cand space; ranks from an LDS bitset
        // prefix-scan; writes come out doc-ordered => coalesced)
        if (NC && q.bitmap_out) {
            __syncthreads();  // bits_m final
            ((uint32_t*)q.bitmap_out)[uint64_t(tile) * (TILE_DOCS / 32) +
                                      threadIdx.x] = sc_bits_m[threadIdx.x];
        }
        if (NC) {
            __syncthreads();  // bits_m/score final
            uint32_t w = threadIdx.x;  // one 32-doc word per thread (256 words)
            uint32_t cnt = __popc(sc_bits_m[w]);
            uint32_t incl = wave_incl_scan_u32(cnt);
            uint32_t wv = threadIdx.x >> 6;
            if (lane_id() == 63) sc_wave_base[wv] = incl;
            __syncthreads();
            uint32_t wave_off = 0;
            for (uint32_t i = 0; i < wv; ++i) wave_off += sc_wave_base[i];
            sc_word_pref[w] = wave_off + incl - cnt;
            uint32_t tile_total = sc_wave_base[0] + sc_wave_base[1] +
                                  sc_wave_base[2] + sc_wave_base[3];
            if (threadIdx.x == 0) {
                *sc_cand_base = atomicAdd(cand_count, tile_total);
                if (do_count) tile_counts[tile] = tile_total;
            }
            __syncthreads();
            uint64_t base = *sc_cand_base;
            // one 32-doc word per thread, iterating SET bits only (a full
            // grid-stride re-walk of the tile cost ~as much as the decode
            // itself at ~25% match density); per-thread writes are
            // consecutive addresses, coalescing at cache-line granularity
            {
                uint32_t wd = threadIdx.x;
                uint32_t bits = sc_bits_m[wd];
                uint64_t rank = sc_word_pref[wd];
                while (bits) {
                    uint32_t b = __ffs(bits) - 1u;
                    bits &= bits - 1u;
                    uint32_t li = wd * 32u + b;
                    uint32_t d = tile_lo + li;
                    float sc = (NS && q.scoring) ? sc_score[li] : 0.f;
                    uint64_t pos = base + rank++;
                    if (pos >= q.cand_cap) break;
                    if (q.wide_cand) {
                        uint64_t key = wide_sort_key(q, d, sc);
                        cand[2 * pos] = key;
                        cand[2 * pos + 1] =
                            (uint64_t(__float_as_uint(sc)) << 32) | d;
                    } else {
                        uint32_t kh = q.scoring ? f32_sortable(sc) : 0u;
                        if (q.sort_asc) kh = ~kh;
                        uint32_t kl = q.sort_asc ? ~d : d;
                        cand[pos] = (uint64_t(kh) << 32) | kl;
                    }
                }
            }
        } else if (do_count) {
            __syncthreads();
            #pragma unroll
            for (int dlt = 32; dlt; dlt >>= 1) local_count += __shfl_down(local_count, dlt, 64);
            if (lane_id() == 0) sc_wave_base[threadIdx.x >> 6] = local_count;
            __syncthreads();
            if (threadIdx.x == 0)
                tile_counts[tile] = sc_wave_base[0] + sc_wave_base[1] +
                                    sc_wave_base[2] + sc_wave_base[3];
        }
        __syncthreads();
    }

    // ---- flush LDS agg arrays + matched counters (once per workgroup)
    if (NA && q.n_aggs) {
        __syncthreads();
        for (uint32_t ai = 0; ai < q.n_aggs; ++ai) {
            const AggDev& a = aggs[ai];
            if (a.lds_slot > 1) continue;
            uint32_t* src = a.lds_slot == 1 ? sc_agg_terms : sc_agg_hist;
            // stagger the bucket order per workgroup: concurrent WGs would
            // otherwise sweep the buckets in lockstep and serialize on the
            // same global words
            uint32_t rot = (blockIdx.x * 131u) % (a.n_buckets ? a.n_buckets : 1u);
            for (uint32_t i0 = threadIdx.x; i0 < a.n_buckets;
                 i0 += TILE_THREADS) {
                uint32_t i = i0 + rot;
                if (i >= a.n_buckets) i -= a.n_buckets;
                uint32_t v = a.lds_rep == 2 ? src[2 * i] + src[2 * i + 1] : src[i];
                if (v)
                    atomicAdd((unsigned long long*)(q.results + a.counts_out) + i,
                              (unsigned long long)v);
            }
        }
        if (threadIdx.x < q.n_aggs && threadIdx.x < 4) {
            const AggDev& a = aggs[threadIdx.x];
            if ((a.kind == AGGD_TERMS || a.kind == AGGD_TERMS_NUM) &&
                sc_agg_matched[threadIdx.x])
                atomicAdd((unsigned long long*)(q.results + a.matched_out),
                          (unsigned long long)sc_agg_matched[threadIdx.x]);
        }
    }
}

// host-side dispatcher over the 16 shape instantiations; returns the dynamic
// LDS bytes it launched with (for occupancy reporting)
inline uint32_t launch_leaf_tile(bool ns, bool nb, bool na, bool nc, dim3 grid,
                                 hipStream_t stream, const QueryDev& q,
                                 uint32_t tile_base, uint32_t tile_end,
                                 uint32_t do_count) {
    uint32_t lds = tile_lds_bytes(ns, nb, na, nc);
    // runtime-optional staging sections (norms tile + K tables): pay LDS
    // only for what this query stages
    if (ns && q.norms_stage_off) lds += TILE_DOCS;
    if (ns && q.scoring && q.n_ktabs) lds += 1024 * q.n_ktabs;
    #define QW_CASE(NS_, NB_, NA_, NC_)                                           \
        if (ns == NS_ && nb == NB_ && na == NA_ && nc == NC_) {                   \
            hipLaunchKernelGGL((k_leaf_tile_t<NS_, NB_, NA_, NC_>), grid,         \
                               dim3(TILE_THREADS), lds, stream, q, tile_base,     \
                               tile_end, do_count);                               \
            return lds;                                                           \
        }
    QW_CASE(false, false, false, false)
    QW_CASE(false, false, false, true)
    QW_CASE(false, false, true, false)
    QW_CASE(false, false, true, true)
    QW_CASE(false, true, false, false)
    QW_CASE(false, true, false, true)
    QW_CASE(false, true, true, false)
    QW_CASE(false, true, true, true)
    QW_CASE(true, false, false, false)
    QW_CASE(true, false, false, true)
    QW_CASE(true, false, true, false)
    QW_CASE(true, false, true, true)
    QW_CASE(true, true, false, false)
    QW_CASE(true, true, false, true)
    QW_CASE(true, true, true, false)
    QW_CASE(true, true, true, true)
    #undef QW_CASE
    return 0;  // unreachable
}

// bitmap combine for the recursive boolean evaluator (product.cpp
// bitmap_eval): dst op= src over u32 words
extern "C" __global__ void k_bitmap_combine(uint32_t* dst, const uint32_t* src,
                                            uint64_t words, uint32_t op) {
    for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < words;
         i += uint64_t(gridDim.x) * blockDim.x) {
        uint32_t d = dst[i], s = src[i];
        switch (op) {
            case 0: d &= s; break;   // AND
            case 1: d |= s; break;   // OR
            case 2: d &= ~s; break;  // AND NOT
            default: d = s; break;   // COPY
        }
        dst[i] = d;
    }
}

// ----------------------------------------------------- top-K selection passes
// histogram of the top 12 bits (after `shift`) of candidates matching
// (key >> prefix_shift) == prefix
extern "C" __global__ void k_cand_hist(const uint64_t* cand, const uint32_t* n_ptr,
                                       uint64_t prefix, uint32_t prefix_bits,
                                       uint32_t* hist) {
    __shared__ uint32_t lh[TOPK_BINS];
    for (uint32_t i = threadIdx.x; i < TOPK_BINS; i += blockDim.x) lh[i] = 0;
    __syncthreads();
    uint32_t n = *n_ptr;  // device-side count: pass 0 is enqueued without a
                          // host round trip for the candidate count
    uint32_t shift = 64 - prefix_bits - 12;
    // per-thread run cache: score keys concentrate into few bins, so
    // consecutive samples often repeat a bin — batch the LDS atomics
    uint32_t last_bin = 0xFFFFFFFFu, acc = 0;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * blockDim.x) {
        uint64_t k = cand[i];
        if (prefix_bits && (k >> (64 - prefix_bits)) != prefix) continue;
        uint32_t b = uint32_t(k >> shift) & (TOPK_BINS - 1);
        if (b == last_bin) ++acc;
        else {
            if (acc) atomicAdd(&lh[last_bin], acc);
            last_bin = b;
            acc = 1;
        }
    }
    if (acc) atomicAdd(&lh[last_bin], acc);
    __syncthreads();
    for (uint32_t i = threadIdx.x; i < TOPK_BINS; i += blockDim.x)
        if (lh[i]) atomicAdd(&hist[i], lh[i]);
}

// wide-record (16B {u64 key, u64 aux}) variants of the selection passes
extern "C" __global__ void k_cand_hist_w(const uint64_t* cand, const uint32_t* n_ptr,
                                         uint64_t prefix, uint32_t prefix_bits,
                                         uint32_t* hist) {
    __shared__ uint32_t lh[TOPK_BINS];
    for (uint32_t i = threadIdx.x; i < TOPK_BINS; i += blockDim.x) lh[i] = 0;
    __syncthreads();
    uint32_t n = *n_ptr;
    uint32_t shift = 64 - prefix_bits - 12;
    uint32_t last_bin = 0xFFFFFFFFu, acc = 0;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * blockDim.x) {
        uint64_t k = cand[2 * i];
        if (prefix_bits && (k >> (64 - prefix_bits)) != prefix) continue;
        uint32_t b = uint32_t(k >> shift) & (TOPK_BINS - 1);
        if (b == last_bin) ++acc;
        else {
            if (acc) atomicAdd(&lh[last_bin], acc);
            last_bin = b;
            acc = 1;
        }
    }
    if (acc) atomicAdd(&lh[last_bin], acc);
    __syncthreads();
    for (uint32_t i = threadIdx.x; i < TOPK_BINS; i += blockDim.x)
        if (lh[i]) atomicAdd(&hist[i], lh[i]);
}

extern "C" __global__ void k_cand_compact_w(const uint64_t* cand, uint32_t n,
                                            uint64_t floor_key, uint64_t ceil_key,
                                            uint64_t* out, uint32_t* out_count,
                                            uint32_t cap) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * blockDim.x) {
        uint64_t k = cand[2 * i];
        bool take = k >= floor_key && k <= ceil_key;
        uint64_t mask = __ballot(take);
        uint32_t nw = __popcll(mask);
        if (!nw) continue;
        uint32_t leader = __ffsll((unsigned long long)mask) - 1;
        uint32_t base;
        if (lane_id() == leader) base = atomicAdd(out_count, nw);
        base = __shfl(base, leader, 64);
        if (take) {
            uint32_t off = base + __popcll(mask & ((1ull << lane_id()) - 1ull));
            if (off < cap) {
                out[2 * off] = k;
                out[2 * off + 1] = cand[2 * i + 1];
            }
        }
    }
}

// compact candidates with key >= floor_key into out (bounded by cap)
extern "C" __global__ void k_cand_compact(const uint64_t* cand, uint32_t n,
                                          uint64_t floor_key, uint64_t ceil_key,
                                          uint64_t* out, uint32_t* out_count,
                                          uint32_t cap) {
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * blockDim.x) {
        uint64_t k = cand[i];
        bool take = k >= floor_key && k <= ceil_key;
        uint64_t mask = __ballot(take);
        uint32_t nw = __popcll(mask);
        if (!nw) continue;
        uint32_t leader = __ffsll((unsigned long long)mask) - 1;
        uint32_t base;
        if (lane_id() == leader) base = atomicAdd(out_count, nw);
        base = __shfl(base, leader, 64);
        if (take) {
            uint32_t off = __popcll(mask & ((1ull << lane_id()) - 1ull));
            if (base + off < cap) out[base + off] = k;
        }
    }
}

}  // namespace qw

namespace qw {

// ------------------------------------------------------------ phrase eval
// Multi-token phrase (slop 0) into a device HitSet bitmap: one THREAD per
// posting of token 0 (the driver term). Per posting: decode its doc id via
// the block's 32-element anchors (<=32 gap extracts), its position-stream
// offset (tf prefix scan over the block), then for each other token a
// binary search over its skip entries + an in-block scan for the doc, and
// finally the consecutive-position chain over the raw u32 position lists.
// Phrases are rare-path (niche vs the flagship decode kernels) — scalar
// per-thread work is the right shape; throughput is bounded by the driver
// term's df, not the split size.
struct PhraseTokDev {
    uint64_t skip_off;       // byte off of this term's first SkipEntry
    uint64_t payload_off;    // field payload byte off
    uint64_t pos_start_off;  // byte off of pos_start u32 AT the term's first block
    uint64_t positions_off;  // field positions byte off
    uint32_t n_blocks;
    uint32_t df;
};
constexpr uint32_t PHRASE_MAX_TOKS = 8;
struct PhraseDev {
    uint32_t n_toks;
    uint32_t num_docs;
    PhraseTokDev tok[PHRASE_MAX_TOKS];
};

__device__ __forceinline__ uint32_t phrase_gap(const uint32_t* gw, uint32_t j,
                                               uint32_t w) {
    return uint32_t(extract_bits(gw, uint64_t(j) * w, w));
}

// doc id of posting j in block e (anchors shorten the scan to <=32 gaps)
__device__ static uint32_t phrase_doc_at(const uint32_t* payload,
                                         const SkipEntryDev& e, uint32_t j) {
    const uint32_t* gw = payload + e.word_off;
    uint32_t seg = j >> 5;
    // anchors live in the 2 u64 words before the gaps, u32 layout
    // [pad, doc31, doc63, doc95]: anchor for segment s is (gw-4)[s]
    uint32_t doc = seg == 0 ? e.first_doc : (gw - 4)[seg];
    for (uint32_t i = seg == 0 ? 1 : seg * 32; i <= j; ++i)
        doc += phrase_gap(gw, i, e.id_bits);
    return doc;
}

// tf of posting j + sum of tfs of postings [0, j) in block e
__device__ static void phrase_tf_prefix(const uint32_t* payload,
                                        const SkipEntryDev& e, uint32_t j,
                                        uint32_t* tf_out, uint32_t* pre_out) {
    if (!e.tf_bits) {  // record without tf: tf = 1 everywhere
        *tf_out = 1;
        *pre_out = j;
        return;
    }
    // tf fields start after the gap words (2*id_bits u64 = this many u32s)
    const uint32_t* tfw = payload + e.word_off + ((128u * e.id_bits + 63u) / 64u) * 2;
    uint32_t pre = 0;
    for (uint32_t i = 0; i < j; ++i)
        pre += phrase_gap(tfw, i, e.tf_bits) + 1u;
    *tf_out = phrase_gap(tfw, j, e.tf_bits) + 1u;
    *pre_out = pre;
}

extern "C" __global__ void k_phrase_bitmap(const uint8_t* split, PhraseDev p,
                                           uint32_t* bitmap) {
    const PhraseTokDev& t0 = p.tok[0];
    const SkipEntryDev* skip0 = (const SkipEntryDev*)(split + t0.skip_off);
    const uint32_t* pay0 = (const uint32_t*)(split + t0.payload_off);
    const uint32_t* ps0 = (const uint32_t*)(split + t0.pos_start_off);
    const uint32_t* P0 = (const uint32_t*)(split + t0.positions_off);
    for (uint32_t t = blockIdx.x * blockDim.x + threadIdx.x; t < t0.df;
         t += gridDim.x * blockDim.x) {
        uint32_t b = t >> 7, j = t & 127u;
        SkipEntryDev e0 = skip0[b];
        if (j >= e0.count) continue;
        uint32_t doc = phrase_doc_at(pay0, e0, j);
        uint32_t tf0, pre0;
        phrase_tf_prefix(pay0, e0, j, &tf0, &pre0);
        uint32_t off0 = ps0[b] + pre0;

        // locate `doc` in every other token; collect (pos_off, tf)
        uint32_t offi[PHRASE_MAX_TOKS], ni[PHRASE_MAX_TOKS];
        bool ok = true;
        for (uint32_t i = 1; i < p.n_toks && ok; ++i) {
            const PhraseTokDev& ti = p.tok[i];
            const SkipEntryDev* skipi = (const SkipEntryDev*)(split + ti.skip_off);
            // first block with last_doc >= doc
            uint32_t lo = 0, hi = ti.n_blocks;
            while (lo < hi) {
                uint32_t mid = (lo + hi) >> 1;
                if (skipi[mid].last_doc < doc) lo = mid + 1;
                else hi = mid;
            }
            if (lo == ti.n_blocks || skipi[lo].first_doc > doc) {
                ok = false;
                break;
            }
            SkipEntryDev ei = skipi[lo];
            const uint32_t* payi = (const uint32_t*)(split + ti.payload_off);
            const uint32_t* gw = payi + ei.word_off;
            uint32_t cur = ei.first_doc, pre = 0, jj = 0, tfj = 0;
            bool found = false;
            const uint32_t* tfw =
                ei.tf_bits
                    ? payi + ei.word_off + ((128u * ei.id_bits + 63u) / 64u) * 2
                    : nullptr;
            for (; jj < ei.count; ++jj) {
                if (jj) cur += phrase_gap(gw, jj, ei.id_bits);
                tfj = tfw ? phrase_gap(tfw, jj, ei.tf_bits) + 1u : 1u;
                if (cur == doc) {
                    found = true;
                    break;
                }
                if (cur > doc) break;
                pre += tfj;
            }
            if (!found) {
                ok = false;
                break;
            }
            offi[i] = ((const uint32_t*)(split + ti.pos_start_off))[lo] + pre;
            ni[i] = tfj;
        }
        if (!ok) continue;

        // consecutive-position chain: exists a in tok0 positions with
        // (a + i) in tok_i positions for all i
        bool hit = false;
        for (uint32_t a = 0; a < tf0 && !hit; ++a) {
            uint32_t pos0 = P0[off0 + a];
            bool chain = true;
            for (uint32_t i = 1; i < p.n_toks && chain; ++i) {
                const uint32_t* Pi =
                    (const uint32_t*)(split + p.tok[i].positions_off);
                uint32_t lo = offi[i], hi = offi[i] + ni[i], want = pos0 + i;
                chain = false;
                while (lo < hi) {
                    uint32_t mid = (lo + hi) >> 1;
                    uint32_t v = Pi[mid];
                    if (v == want) {
                        chain = true;
                        break;
                    }
                    if (v < want) lo = mid + 1;
                    else hi = mid;
                }
            }
            hit = chain;
        }
        if (hit) atomicOr(&bitmap[doc >> 5], 1u << (doc & 31u));
    }
}

}  // namespace qw
