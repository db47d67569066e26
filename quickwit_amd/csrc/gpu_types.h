// Device-visible descriptors shared between the host orchestration
// (product.cpp) and the HIP kernels (kernels.hip). All offsets are BYTE
// offsets into the split's device image (the QWA1 file uploaded verbatim to
// HBM — DESIGN.md §3/§5).
#pragma once
#include <cstdint>

namespace qw {

// docs per workgroup tile — build-time switch (-DQW_TILE_DOCS=4096) so the
// LDS-footprint/occupancy tradeoff can be measured as whole-library variants
#ifndef QW_TILE_DOCS
#define QW_TILE_DOCS 8192
#endif
constexpr uint32_t TILE_DOCS = QW_TILE_DOCS;
constexpr uint32_t TILE_THREADS = 256; // 4 waves
// BM25 K tables staged to LDS when the query has at most this many scored
// fields (1 KiB each)
constexpr uint32_t KTAB_LDS_MAX = 4;
constexpr uint32_t AGG_LDS_BUCKETS = 2048;
constexpr uint32_t TOPK_BINS = 4096;

// role of a term in the flattened boolean
enum TermRole : uint32_t { ROLE_SHOULD = 0, ROLE_MUST = 1, ROLE_MUST_NOT = 2 };

struct TermDev {
    uint32_t role;
    uint32_t grp;          // must-group: terms in one group OR together, the
                           // groups AND (wildcard/term_set under must)
    uint32_t n_blocks;
    uint64_t skip_off;     // byte offset of this term's first SkipEntry
    uint64_t payload_off;  // byte offset of the FIELD's payload section
    uint64_t norms_off;    // byte offset of the field's fieldnorms (0 = none)
    float weight;          // BM25 W = idf*(1+k1)*boost (0 when not scoring)
    uint32_t ktab_idx;     // index into the query's K tables (per field)
    uint64_t ranges_addr;  // ABSOLUTE device VA: u32 lo[ntiles], u32
                           // hi[ntiles] block-index ranges for this term
                           // (cached per (split, term) — the ranges depend
                           // only on the split's static block structure)
};

enum PredType : uint32_t {
    PRED_RANGE_U64 = 0,
    PRED_RANGE_I64 = 1,  // i64 / datetime(ms)
    PRED_PRESENCE = 2,
    PRED_BITSET = 3,     // device-resident HitSet (predicate cache, §8f.2)
    PRED_RANGE_F64 = 4,  // f64 column; lo/hi carry bit-cast doubles
};
enum PredFlags : uint32_t {
    PRED_NEGATED = 1,       // must_not predicate
    PRED_LO_INCLUDED = 2,
    PRED_LO_EXCLUDED = 4,
    PRED_HI_INCLUDED = 8,
    PRED_HI_EXCLUDED = 16,
    // evaluate for EVERY doc of the tile, not just still-matching lanes:
    // when the positive clauses are dense (host estimate >= ~15%), the
    // column lines are touched anyway and the unpredicated load keeps the
    // epilogue coalesced and convergent
    PRED_EAGER = 32,
};

struct PredDev {
    uint32_t type;
    uint32_t flags;
    int64_t lo, hi;        // bound values (u64 preds reinterpret the bits)
    uint64_t values_off;   // column byte offset (0 for pure presence w/o col)
    uint64_t nulls_off;    // null bitmap byte offset (0 = non-nullable)
    uint32_t value_width;  // 8 for u64/i64; 1/2/4 for str ords
    uint64_t abs_bitmap;   // PRED_BITSET: absolute device VA of u32 words
                           // laid out tile-major (TILE_DOCS/32 words per tile)
    uint64_t offsets_off;  // multi-valued column: u32[num_docs+1] prefix into
                           // values (any-value-matches semantics); 0 = single
};

enum AggKindDev : uint32_t {
    AGGD_HISTO = 0, AGGD_TERMS = 1, AGGD_RANGE = 2, AGGD_METRIC = 3,
    // terms over a NUMERIC fast column: open-addressing [key,count] u64-pair
    // hash table in the counts_out region (slots = (n_buckets-2)/2, power of
    // two; trailing two words = count for the key ~0 sentinel value +
    // overflow flag), keyed by the value's order-preserving sortable bits
    AGGD_TERMS_NUM = 4,
    // composite aggregation: same hash-table layout, keyed by the per-source
    // components packed MSB-first into <=63 bits (so the key can never hit
    // the ~0 sentinel); per-source params in the c_* fields + the sub_*
    // column arrays
    AGGD_COMP = 5,
    // top-level percentiles: DDSketch key counts, (p_n_keys+1) u64 words at
    // counts_out (word 0 = zero bucket, word j = key p_k_lo+j-1)
    AGGD_PERC = 6
};
constexpr uint32_t AGG_MAX_RANGES = 16;

struct AggDev {
    uint32_t kind;
    uint32_t n_buckets;     // histo: bucket count from [lo_idx..hi_idx];
                            // terms: cardinality
    int64_t base_index;     // histo: floor((min-offset)/interval) at bucket 0
    double interval;        // histo (ms for date_histogram)
    double offset;          // histo
    // exact integer fast path (set by plan_aggs when interval/offset are
    // integral and the column range makes floor-div provably bit-equal to
    // the double path — see histo_bucket in kernels.hip)
    uint32_t int_fast;
    uint32_t lds_slot;      // 0 = agg_hist, 1 = agg_terms, 0xFF = global atomics
    uint32_t lds_rep;       // 1 or 2: interleaved LDS histogram copies
                            // (lane-parity split halves same-bucket conflicts)
    int64_t i_interval;
    int64_t i_offset;
    double inv_interval;    // 1.0 / i_interval
    uint64_t values_off;    // column byte offset
    uint64_t nulls_off;
    uint32_t value_width;   // 8, or ord width for terms
    uint32_t value_is_i64;  // 0 = u64, 1 = i64/datetime, 2 = f64
    uint64_t counts_out;    // byte offset into result scratch: u64[n_buckets]
    uint64_t matched_out;   // terms: u64 counter of docs-with-value
    uint64_t offsets_off;   // terms over a multi-valued column: u32 prefix
    // AGGD_RANGE: [from, to) per range; NaN-free sentinels via has-masks
    uint32_t n_ranges;
    uint32_t r_has_from, r_has_to;  // bitmasks over ranges
    double r_from[AGG_MAX_RANGES], r_to[AGG_MAX_RANGES];
    // one optional stats sub-agg set per bucket: {u64 cnt, f64 sum, u64 min_s,
    // u64 max_s (sortable-mapped), f64 sum_sq} × n_sub (40 B), bucket-major.
    // AGGD_METRIC uses ONE such slot at counts_out.
    uint32_t n_sub;
    uint64_t sub_out;        // byte offset: n_buckets * n_sub * 32 bytes
    uint64_t sub_values_off[4];  // sub-agg source columns (<=4);
                                 // AGGD_COMP: the composite SOURCE columns
    uint64_t sub_nulls_off[4];
    uint32_t sub_width[4];
    uint32_t sub_is_i64[4];
    // AGGD_COMP per-source packing: component = ord+miss (terms) or
    // bucket_idx-base+miss (histogram), shifted into the 63-bit key
    uint32_t c_shift[4];
    uint32_t c_histo;        // bitmask: source i is a histogram
    uint32_t c_missing;      // bitmask: missing_bucket (else null doc drops)
    double c_interval[4], c_offset[4];
    int64_t c_base[4];       // histogram: floor((col_min-offset)/interval)
    // percentiles (DDSketch restatement): sub index carrying a sketch
    // (0xFF = none), sketch region, key range, and the scratch offset of the
    // gamma^k boundary doubles the kernel binary-searches (host-computed so
    // bucketing is bit-identical to the oracle)
    uint32_t p_si;
    uint64_t p_out;
    int32_t p_k_lo;
    uint32_t p_n_keys;
    uint64_t p_bound_off;
};

struct QueryDev {
    const uint8_t* split;    // device base of the split image
    const uint8_t* scratch;  // device base of the query scratch (ranges etc.)
    uint8_t* results;        // device base of the results scratch
    uint32_t num_docs;
    uint32_t n_tiles;
    uint32_t n_terms;
    uint32_t n_must;      // count of must GROUPS (see TermDev::grp)
    uint32_t n_must_not;  // count of ROLE_MUST_NOT terms
    uint32_t n_preds;
    uint32_t n_aggs;
    uint32_t msm;           // minimum should match (0 = no should clauses)
    uint32_t scoring;       // 1 = BM25 scores wanted
    uint32_t match_all;     // 1 = base matches everything (no should clauses)
    uint32_t collect_hits;  // 1 = write candidates
    uint32_t sort_asc;      // 1 = ascending primary sort (flip candidate keys)
    // primary sort-key source for candidate records (collector.rs:403-414
    // sort-key extraction). narrow records (8B: key32|doc) serve _score /
    // doc-order sorts; wide records (16B: u64 key, u32 score_bits|u32 doc)
    // serve fast-field sorts and two-field sorts — exact order is
    // re-established on host over the survivors.
    uint32_t wide_cand;     // 1 = 16B candidate records
    uint32_t sort_src;      // 0 none/doc, 1 _score, 2 u64/ord column,
                            // 3 i64 column, 4 f64 column
    uint32_t sort_width;    // column width (1/2/4/8), sort_src>=2
    uint64_t sort_values_off;
    uint64_t sort_nulls_off;  // 0 = non-nullable (missing -> None, sorts last)
    // 1 = straight-line agg path: agg[0] is an int_fast LDS histogram over a
    // non-nullable 8B column with no sub-aggs (bucket bounds derived from the
    // column min/max => no range check), optional agg[1] LDS terms over a
    // non-nullable ord column. Descriptors hoisted to registers.
    uint32_t agg_fast;
    uint32_t agg_nt;        // experiment: nontemporal loads in the pure-agg
                            // loop (QW_AGG_NT=1; streaming columns bypass L2)
    uint32_t nt_decode;     // experiment: nontemporal loads for the posting
                            // skip/gap/tf streams (QW_NT_DECODE=1; they are
                            // read once — keep L2 for the fieldnorm gathers)
    uint64_t bitmap_out;    // absolute device VA: store each tile's match
                            // bitset words here (predicate-cache fill); 0=off
    uint64_t terms_off;     // scratch offsets of descriptor arrays
    uint64_t preds_off;
    uint64_t aggs_off;
    uint64_t ktabs_off;     // scratch: n_fields * 256 f32 BM25 K tables
    // LDS staging for the scoring decode path (cuts the per-posting critical
    // path from two dependent global loads — norm byte, then K float — to
    // two LDS reads). 0 disables; sections appended after the compile-time
    // SmemMap, host adds the bytes to the dynamic-LDS size.
    uint32_t n_ktabs;          // K tables to copy to LDS (0 = read global)
    uint32_t _pad_ktab;
    uint64_t norms_stage_off;  // common fieldnorms byte offset of all scored
                               // terms (0 = heterogenous fields, read global)
    // result offsets (all in `results`)
    uint64_t tile_counts_off;  // u32[n_tiles]
    uint64_t cand_count_off;   // u32 (atomic)
    uint64_t cand_off;         // {u32 key, u32 doc}[cap]
    uint64_t cand_cap;
    uint64_t hist_off;         // u32[TOPK_BINS] histogram of candidate keys
};

}  // namespace qw
