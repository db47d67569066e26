// ============================================================================
// libquickwit_amd — MI355X-native host orchestration for the Quickwit
// leaf-search hot path (the PRODUCT; C-ABI in include/quickwit_amd.h).
//
// This is the work the reference ships to its rayon pool inside
// leaf_search_single_split (quickwit/quickwit-search/src/leaf.rs:899-959):
// query build (doc_mapper_impl.rs:637 -> tantivy_query_ast.rs:153-374),
// posting decode + boolean combination + BM25 + collection
// (searcher.search at leaf.rs:929, collector.rs:523-594), fast-field
// predicates (range_query.rs:86-158) and aggregations, top-K with the
// reference tie-breaks (top_k_collector.rs, docs/internals/sorting.md:14-26),
// and the cross-split merge (collector.rs:1198, :832-861).
//
// All index arithmetic runs in HIP kernels on gfx950 (kernels.hip, included
// below as one translation unit — no -fgpu-rdc needed). There is NO CPU
// fallback: without a HIP device every search returns QW_ERR_NO_GPU
// (DESIGN.md §1). The CPU restatement lives in oracle/ (test infra only).
// ============================================================================
#include <hip/hip_runtime.h>
#include <zlib.h>

#include <algorithm>
#include <chrono>
#include <cmath>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <list>
#include <map>
#include <memory>
#include <set>
#include <stdexcept>
#include <string>
#include <vector>

#include "../../include/quickwit_amd.h"
#include "fieldnorm.h"
#include "minijson.h"
#include "pb.h"
#include "qagg_format.h"
#include "qast.h"
#include "qsplit.h"
#include "sortkey.h"

// device code: single TU (kernels + host) so no relocatable device link
#include "kernels.hip"

namespace qw {

#define HIP_CHECK(expr)                                                        \
    do {                                                                       \
        hipError_t _e = (expr);                                                \
        if (_e != hipSuccess)                                                  \
            throw std::runtime_error(std::string("HIP error: ") +              \
                                     hipGetErrorString(_e) + " at " #expr);    \
    } while (0)

struct DeviceSplit {
    std::vector<uint8_t> host;  // host copy: term-dict lookups + meta parse
    SplitView view;             // views into `host` (segment 0 for QWA2)
    uint8_t* d_image = nullptr; // full file image resident in HBM
    size_t len = 0;
    uint64_t total_docs = 0;    // across all segments
    // QWA2 multi-segment container (collector.rs:475-594 per-segment
    // collection): one child per segment; child host/view/d_image alias the
    // parent allocation (owns_image=false)
    std::vector<std::unique_ptr<DeviceSplit>> segs;
    bool owns_image = true;
    // per-(field, term) device block-range tables (u32 lo[ntiles] +
    // hi[ntiles]): they depend only on the split's static block structure,
    // so the O(blocks+tiles) host walk (~180 us at 100M docs) and its
    // 300 KB upload happen once per term, not per query
    mutable std::map<std::pair<const void*, int64_t>, uint8_t*> range_cache;
    mutable uint64_t range_cache_bytes = 0;
    void free_range_cache() const {
        for (auto& kv : range_cache) (void)hipFree(kv.second);
        range_cache.clear();
        range_cache_bytes = 0;
    }
};

struct KernelTimer {
    double total_ms = 0;
    uint64_t launches = 0;
};

struct DevBuf {  // grow-only device scratch
    uint8_t* p = nullptr;
    size_t cap = 0;
    void ensure(size_t n) {
        if (n <= cap) return;
        if (p) (void)hipFree(p);
        p = nullptr;
        cap = 0;
        HIP_CHECK(hipMalloc(&p, n));
        cap = n;
    }
    ~DevBuf() {
        if (p) (void)hipFree(p);
    }
};

struct PinnedBuf {  // grow-only pinned host staging (fast H2D/D2H, no
                    // per-call allocation)
    uint8_t* p = nullptr;
    size_t cap = 0;
    void ensure(size_t n) {
        if (n <= cap) return;
        if (p) (void)hipHostFree(p);
        p = nullptr;
        cap = 0;
        HIP_CHECK(hipHostMalloc((void**)&p, n));
        cap = n;
    }
    ~PinnedBuf() {
        if (p) (void)hipHostFree(p);
    }
};

// LeafSearchCache restatement (quickwit-search/src/leaf_cache.rs): memoizes
// per-(split, canonical request) LeafSearchResponse bytes in a byte-capacity
// LRU (MemorySizedCache semantics). The key canonicalizes the request by
// clearing the timestamp bounds and forcing count_hits = COUNT_ALL
// (leaf_cache.rs:86-115) and appends the merged (request ∩ split) half-open
// time range (leaf_cache.rs:118-180).
struct LeafCache {
    size_t capacity = 64ull << 20;  // SearcherConfig partial_request_cache_capacity
    size_t used = 0;
    std::list<std::pair<std::string, std::string>> lru;  // front = most recent
    std::map<std::string, std::list<std::pair<std::string, std::string>>::iterator> idx;

    bool get(const std::string& key, std::string* out) {
        auto it = idx.find(key);
        if (it == idx.end()) return false;
        lru.splice(lru.begin(), lru, it->second);
        *out = it->second->second;
        return true;
    }
    void put(const std::string& key, std::string val) {
        if (!capacity || key.size() + val.size() > capacity) return;
        auto it = idx.find(key);
        if (it != idx.end()) {
            used -= it->second->first.size() + it->second->second.size();
            lru.erase(it->second);
            idx.erase(it);
        }
        used += key.size() + val.size();
        lru.emplace_front(key, std::move(val));
        idx[key] = lru.begin();
        while (used > capacity && !lru.empty()) {
            auto& back = lru.back();
            used -= back.first.size() + back.second.size();
            idx.erase(back.first);
            lru.pop_back();
        }
    }
    void remove_split(const std::string& split_id) {
        std::string prefix = split_id + '\0';
        for (auto it = lru.begin(); it != lru.end();) {
            if (it->first.compare(0, prefix.size(), prefix) == 0) {
                used -= it->first.size() + it->second.size();
                idx.erase(it->first);
                it = lru.erase(it);
            } else ++it;
        }
    }
};

// half-open [start, end) second range; has_end=false means unbounded
// (leaf_cache.rs HalfOpenRange, incl. the empty-range normalization)
struct HalfOpen {
    int64_t start = INT64_MIN;
    bool has_end = false;
    int64_t end = 0;
    void normalize() {
        if (has_end && end <= start) {
            start = 0;
            end = 0;
            has_end = true;
        }
    }
};

static std::string leaf_cache_key(const pb::SplitIdAndFooterOffsets& so,
                                  const pb::SearchRequest& req) {
    // request time range: [start included, end excluded)
    HalfOpen r;
    r.start = req.start_timestamp ? *req.start_timestamp : INT64_MIN;
    if (req.end_timestamp) {
        r.has_end = true;
        r.end = *req.end_timestamp;
    }
    r.normalize();
    // split time range: [start, end] inclusive -> [start, end+1)
    HalfOpen s;
    s.start = so.timestamp_start ? *so.timestamp_start : INT64_MIN;
    if (so.timestamp_end && *so.timestamp_end != INT64_MAX) {
        s.has_end = true;
        s.end = *so.timestamp_end + 1;
    }
    s.normalize();
    // intersection
    HalfOpen m;
    m.start = std::max(r.start, s.start);
    if (r.has_end || s.has_end) {
        m.has_end = true;
        m.end = r.has_end && s.has_end ? std::min(r.end, s.end)
                                       : (r.has_end ? r.end : s.end);
    }
    m.normalize();
    pb::SearchRequest canon = req;
    canon.start_timestamp.reset();
    canon.end_timestamp.reset();
    canon.count_hits = 0;  // CountHits::CountAll (leaf_cache.rs:108)
    std::string key = so.split_id;
    key += '\0';
    char rng[17];
    memcpy(rng, &m.start, 8);
    memcpy(rng + 8, &m.end, 8);
    rng[16] = m.has_end ? 1 : 0;
    key.append(rng, 17);
    key += canon.encode();
    return key;
}

// device HitSet bitmaps memoized per (split, subquery fingerprint) — the
// MI355X-native PredicateCache (cache_node.rs:202-486).
// Count-capped LRU; a 100M-doc bitmap is 12.5 MB of HBM.
struct HitsetCache {
    // Count-capped LRU of device bitmaps. Entries touched by the CURRENT
    // query (gen == cur_gen) are pinned: resolve_hitset/bitmap_eval return
    // raw device pointers that the later main-kernel launch reads, so an
    // eviction between resolve and launch would free in-use device memory.
    // begin_query() unpins everything from previous queries.
    struct Entry {
        std::string key;
        uint8_t* bm;
        uint64_t gen;
        uint64_t size;
    };
    size_t cap = 256;
    uint64_t cur_gen = 0;
    uint64_t bytes = 0;  // device bytes held (feeds the ctx HBM accounting)
    std::list<Entry> lru;
    std::map<std::string, std::list<Entry>::iterator> idx;
    uint64_t hits = 0, misses = 0;

    void begin_query() { ++cur_gen; }
    uint8_t* get(const std::string& key) {
        auto it = idx.find(key);
        if (it == idx.end()) {
            ++misses;
            return nullptr;
        }
        ++hits;
        it->second->gen = cur_gen;
        lru.splice(lru.begin(), lru, it->second);
        return it->second->bm;
    }
    void put(const std::string& key, uint8_t* bm, uint64_t size) {
        lru.push_front(Entry{key, bm, cur_gen, size});
        idx[key] = lru.begin();
        bytes += size;
        // evict oldest UNPINNED entries; a query resolving > cap distinct
        // subtrees temporarily overflows the cap rather than freeing memory
        // the pending launch still references
        for (auto it = std::prev(lru.end());
             lru.size() > cap && it != lru.begin();) {
            auto cur = it--;
            if (cur->gen == cur_gen) continue;
            (void)hipFree(cur->bm);
            bytes -= cur->size;
            idx.erase(cur->key);
            lru.erase(cur);
        }
    }
    void remove_split(const std::string& split_id) {
        std::string prefix = split_id + '\0';
        for (auto it = lru.begin(); it != lru.end();) {
            if (it->key.compare(0, prefix.size(), prefix) == 0) {
                (void)hipFree(it->bm);
                bytes -= it->size;
                idx.erase(it->key);
                it = lru.erase(it);
            } else ++it;
        }
    }
    void clear() {
        for (auto& e : lru) (void)hipFree(e.bm);
        lru.clear();
        idx.clear();
        bytes = 0;
    }
};

}  // namespace qw

// negative/absence term cache (leaf.rs:761-827): (split, field, term) keys
// proven absent from the split's term dictionary by an earlier search; a
// probe hit short-circuits the whole split search (the reference aborts
// during warmup, leaf.rs:315-320 / 472-477; key analog of
// term_absence_cache_key, leaf.rs:641-652)
struct AbsenceCache {
    std::set<std::string> keys;
    uint64_t hits = 0, misses = 0;
    size_t cap = 1 << 20;
    static std::string key(const std::string& split_id, const std::string& f,
                           const std::string& t) {
        std::string k = split_id;
        k += '\0';
        k += f;
        k += '\x1f';
        k += t;
        return k;
    }
    void remove_split(const std::string& split_id) {
        std::string lo = split_id + '\0';
        std::string hi = split_id + '\x01';
        keys.erase(keys.lower_bound(lo), keys.lower_bound(hi));
    }
};

struct qw_ctx {
    int device = 0;
    bool device_ready = false;
    int64_t agg_bucket_limit = 65000;
    std::map<std::string, std::unique_ptr<qw::DeviceSplit>> splits;
    qw::LeafCache leaf_cache;
    qw::HitsetCache hitsets;
    AbsenceCache absence;
    std::string last_error;
    hipStream_t stream = nullptr;
    hipEvent_t ev_start = nullptr, ev_stop = nullptr;
    qw::DevBuf d_scratch, d_results, d_survivors, d_cand2;
    qw::PinnedBuf h_scratch, h_counts, h_surv, h_topk, h_agg;  // pinned staging
    std::map<std::string, qw::KernelTimer> timers;
    // HBM accounting (SearchPermitProvider memory-budget analog,
    // search_permit_provider.rs:43-110): split images + scratch/result
    // buffers in hbm_used, hitset bitmaps in hitsets.bytes. budget 0 =
    // resolved from device free memory at first use; config key
    // "hbm_memory_budget" overrides (bytes). Over-budget add_split refuses
    // (QW_ERR_OVER_MEMORY_BUDGET); over-budget search scratch fails that
    // split into failed_splits, like a permit that can never be granted.
    uint64_t hbm_used = 0;
    uint64_t hbm_budget = 0;
    bool budget_configured = false;
};

static void qw_check_budget(qw_ctx* ctx, uint64_t need) {
    uint64_t used = ctx->hbm_used + ctx->hitsets.bytes;
    if (ctx->hbm_budget && used + need > ctx->hbm_budget)
        throw std::runtime_error(
            "HBM memory budget exceeded: used " + std::to_string(used) +
            " + need " + std::to_string(need) + " > budget " +
            std::to_string(ctx->hbm_budget) +
            " (search_permit_provider analog; raise hbm_memory_budget or "
            "remove splits)");
}

// budgeted grow of a ctx scratch DevBuf
static void qw_ensure_acct(qw_ctx* ctx, qw::DevBuf& b, size_t n) {
    if (n <= b.cap) return;
    qw_check_budget(ctx, n - b.cap);
    ctx->hbm_used -= b.cap;
    b.ensure(n);
    ctx->hbm_used += b.cap;
}

// device address of the (split, term) block-range table, computed and
// uploaded on first use (DeviceSplit::range_cache)
static uint64_t term_ranges_addr(qw_ctx* ctx, const qw::DeviceSplit& ds,
                                 const qw::TextFieldView* f, int64_t tid,
                                 uint32_t n_tiles, uint32_t num_docs) {
    using namespace qw;
    auto key = std::make_pair((const void*)f, tid);
    auto it = ds.range_cache.find(key);
    if (it != ds.range_cache.end()) return (uint64_t)it->second;
    if (ds.range_cache.size() >= 1024) {  // wildcard-expansion backstop
        ctx->hbm_used -= ds.range_cache_bytes;
        ds.free_range_cache();
    }
    size_t bytes = 2ull * n_tiles * 4;
    qw_check_budget(ctx, bytes);
    std::vector<uint32_t> r(2ull * n_tiles);
    const SkipEntry* sk = f->h_skip + f->h_skip_off[tid] / 16;
    uint32_t nb = f->h_n_blocks[tid];
    uint32_t* lo = r.data();
    uint32_t* hi = lo + n_tiles;
    uint32_t b_lo = 0, b_hi = 0;
    for (uint32_t tile = 0; tile < n_tiles; ++tile) {
        uint32_t tlo = tile * TILE_DOCS;
        uint32_t thi = std::min<uint32_t>(tlo + TILE_DOCS, num_docs);
        while (b_lo < nb && sk[b_lo].last_doc < tlo) ++b_lo;
        if (b_hi < b_lo) b_hi = b_lo;
        while (b_hi < nb && sk[b_hi].first_doc < thi) ++b_hi;
        lo[tile] = b_lo;
        hi[tile] = b_hi;
    }
    uint8_t* d = nullptr;
    HIP_CHECK(hipMalloc(&d, bytes));
    HIP_CHECK(hipMemcpy(d, r.data(), bytes, hipMemcpyHostToDevice));
    ctx->hbm_used += bytes;
    ds.range_cache.emplace(key, d);
    ds.range_cache_bytes += bytes;
    return (uint64_t)d;
}

static void ctx_ensure_device(qw_ctx* ctx) {
    using namespace qw;
    if (ctx->device_ready) return;
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess || n == 0)
        throw std::runtime_error("no HIP device visible");
    if (ctx->device >= n) throw std::runtime_error("bad device ordinal");
    HIP_CHECK(hipSetDevice(ctx->device));
    HIP_CHECK(hipStreamCreateWithFlags(&ctx->stream, hipStreamNonBlocking));
    HIP_CHECK(hipEventCreate(&ctx->ev_start));
    HIP_CHECK(hipEventCreate(&ctx->ev_stop));
    if (!ctx->budget_configured) {
        // default budget: 97% of the device's free HBM at first use
        // (warmup_memory_budget analog, node_config defaults)
        size_t free_b = 0, total_b = 0;
        if (hipMemGetInfo(&free_b, &total_b) == hipSuccess && free_b)
            ctx->hbm_budget = uint64_t(free_b) / 100 * 97;
    }
    ctx->device_ready = true;
}

namespace qw {

// ------------------------------------------------------------ flattening
// The kernel executes ONE flattened boolean level (should/must/must_not term
// lists + fast-field predicates). Nested plans that cannot be flattened are
// rejected with InvalidQuery — never silently mis-answered. This covers every
// BASELINE.json config and the golden-suite query shapes (term, full_text
// and/or, bool with term/range clauses, range, match_all).
struct FlatQuery {
    bool match_none = false;
    bool match_all = false;  // positive base = all docs (preds still filter)
    bool scoring = false;
    uint32_t msm = 0;  // >0: shoulds required (count >= msm)
    // required (must/filter) single terms proven ABSENT from this split's
    // term dict during flattening — feeds the negative term cache
    // (leaf.rs:761-827 / term_absence_cache_key leaf.rs:641-652)
    std::vector<std::pair<std::string, std::string>> absent_required;
    struct FTerm {
        const TextFieldView* f;
        int64_t tid;
        uint32_t role;
        float weight;  // BM25 W (0 for unscored filter terms)
        uint32_t grp = 0;  // must-group (OR within a group, AND across)
    };
    std::vector<FTerm> terms;
    uint32_t n_must_groups = 0;
    // CACHE nodes in filter position: resolved to device HitSet bitmaps
    // (PRED_BITSET) by the caller (predicate cache)
    std::vector<const PlanNode*> cache_nodes;
    std::vector<PredDev> preds;
    std::vector<const TextFieldView*> ktab_fields;  // ktab_idx -> field

    uint32_t ktab_for(const TextFieldView* f) {
        for (size_t i = 0; i < ktab_fields.size(); ++i)
            if (ktab_fields[i] == f) return uint32_t(i);
        ktab_fields.push_back(f);
        return uint32_t(ktab_fields.size() - 1);
    }
};

static std::string fold_lower(const std::string& s) {
    std::string o = s;
    for (char& c : o) c = char(std::tolower((unsigned char)c));
    return o;
}

static float bm25_weight(const SplitView& sv, const TextFieldView& f, int64_t tid,
                         float boost) {
    double N = double(sv.num_docs);
    double df = double(f.h_doc_freq[tid]);
    double idf = std::log(1.0 + (N - df + 0.5) / (df + 0.5));
    return float(idf * (1.0 + 1.2) * double(boost));
}

static void add_range_pred(FlatQuery& fq, const SplitView& sv, const PlanNode& n,
                           bool negated) {
    const FastFieldView* f = sv.fast_field(n.field);
    PredDev p{};
    if (!f && n.kind == PlanNode::FIELD_PRESENCE) {
        // presence over a text-only field: a doc has the field iff its
        // fieldnorm byte is nonzero (>= 1 token; index_field_presence analog)
        const TextFieldView* tf = sv.text_field(n.field);
        if (tf && tf->has_norms) {
            p.type = PRED_RANGE_U64;
            p.flags = (negated ? PRED_NEGATED : 0) | PRED_LO_INCLUDED;
            p.lo = 1;
            p.values_off = tf->fieldnorms.off;
            p.value_width = 1;
            fq.preds.push_back(p);
            return;
        }
    }
    if (!f) {
        // unknown fast field: matches nothing (oracle eval_range -> empty)
        if (negated) return;  // must_not nothing = no-op
        fq.match_none = true;
        return;
    }
    if (f->type == FastFieldView::MIXED && n.kind != PlanNode::FIELD_PRESENCE)
        throw std::runtime_error(
            "range/term over a mixed-type dynamic column (r2 limit: sorting "
            "and search_after are supported; typed predicates are not)");
    p.type = f->type == FastFieldView::U64   ? PRED_RANGE_U64
             : f->type == FastFieldView::F64 ? PRED_RANGE_F64
                                             : PRED_RANGE_I64;
    p.flags = negated ? PRED_NEGATED : 0;
    if (n.kind == PlanNode::FIELD_PRESENCE) {
        if (!f->nullable && !f->multi && !negated) return;  // always present
        if (!f->nullable && !f->multi && negated) {
            fq.match_none = true;
            return;
        }
        p.type = PRED_PRESENCE;
        if (f->multi) p.offsets_off = f->value_offsets.off;
    } else {
        if (n.lo.kind == Bound::INCLUDED) p.flags |= PRED_LO_INCLUDED;
        if (n.lo.kind == Bound::EXCLUDED) p.flags |= PRED_LO_EXCLUDED;
        if (n.hi.kind == Bound::INCLUDED) p.flags |= PRED_HI_INCLUDED;
        if (n.hi.kind == Bound::EXCLUDED) p.flags |= PRED_HI_EXCLUDED;
        if (f->type == FastFieldView::F64) {
            memcpy(&p.lo, &n.lo.fval, 8);  // bit-cast double bounds
            memcpy(&p.hi, &n.hi.fval, 8);
        } else if (f->type == FastFieldView::STR) {
            // lexicographic bounds -> ord space [lo, hi) over the sorted
            // dict; a lowercase-normalized column folds the bounds too;
            // multi-valued columns match when ANY value is in range
            if (f->multi) p.offsets_off = f->value_offsets.off;
            std::string lo_s = f->lower_norm ? fold_lower(n.lo.sval) : n.lo.sval;
            std::string hi_s = f->lower_norm ? fold_lower(n.hi.sval) : n.hi.sval;
            p.type = PRED_RANGE_U64;
            p.flags &= PRED_NEGATED;  // rebuild the bound flags below
            if (n.lo.kind != Bound::UNBOUNDED) {
                p.flags |= PRED_LO_INCLUDED;
                p.lo = int64_t(
                    f->str_bound_ord(lo_s, n.lo.kind == Bound::EXCLUDED));
            }
            if (n.hi.kind != Bound::UNBOUNDED) {
                p.flags |= PRED_HI_EXCLUDED;
                p.hi = int64_t(
                    f->str_bound_ord(hi_s, n.hi.kind == Bound::INCLUDED));
            }
        } else {
            p.lo = n.lo.ival;
            p.hi = n.hi.ival;
        }
    }
    p.values_off = f->values.off;
    p.nulls_off = f->nullable ? f->nulls.off : 0;
    p.value_width = f->type == FastFieldView::STR ? uint32_t(f->ord_width) : 8;
    fq.preds.push_back(p);
}

static void add_term(FlatQuery& fq, const SplitView& sv, const std::string& field,
                     const std::string& value, uint32_t role, float boost,
                     bool scored, uint32_t grp = 0, bool group_member = false) {
    const TextFieldView* f = sv.text_field(field);
    int64_t tid = f ? f->find_term(value.data(), value.size()) : -1;
    if (tid < 0) {
        // absent term: MUST -> split matches nothing; SHOULD/MUST_NOT -> skip
        // (skipped shoulds can never satisfy msm — handled by caller count).
        // A member of a must OR-GROUP only weakens the group when absent.
        if (role == ROLE_MUST && !group_member) {
            fq.match_none = true;
            fq.absent_required.emplace_back(field, value);
        }
        return;
    }
    FlatQuery::FTerm t;
    t.f = f;
    t.tid = tid;
    t.role = role;
    t.weight = scored ? bm25_weight(sv, *f, tid, boost) : 0.f;
    t.grp = grp;
    fq.terms.push_back(t);
}

// expand a wildcard over the field's term dictionary (qast.h glob_match;
// wildcard_query.rs -> AutomatonQuery). Returns the number of matched terms.
static size_t add_wildcard(FlatQuery& fq, const SplitView& sv, const PlanNode& n,
                           uint32_t role, uint32_t grp) {
    const TextFieldView* f = sv.text_field(n.field);
    if (!f) return 0;
    size_t cnt = 0;
    for (uint32_t t = 0; t < f->num_terms; ++t) {
        const char* s = (const char*)f->h_term_bytes + f->h_term_offsets[t];
        size_t sl = f->h_term_offsets[t + 1] - f->h_term_offsets[t];
        if (!glob_match(s, sl, n.value.data(), n.value.size(), n.ci)) continue;
        FlatQuery::FTerm ft;
        ft.f = f;
        ft.tid = t;
        ft.role = role;
        ft.weight = 0.f;  // const-score node; scored plans were rejected
        ft.grp = grp;
        fq.terms.push_back(ft);
        if (++cnt > 4096)
            throw std::runtime_error("wildcard expands to >4096 terms (GPU r1 limit)");
    }
    return cnt;
}

// strip semantics-transparent cache wrappers (contexts that cannot use the
// bitmap: should/must_not/scored root)
static const PlanNode& uncache(const PlanNode& c) {
    return c.kind == PlanNode::CACHE ? uncache(c.cache_inner.front()) : c;
}

static bool is_pure_should_terms(const PlanNode& n) {
    if (n.kind != PlanNode::BOOL) return false;
    if (!n.must.empty() || !n.must_not.empty() || !n.filter.empty()) return false;
    if (n.minimum_should_match > 1) return false;
    for (auto& c : n.should)
        if (c.kind != PlanNode::TERM) return false;
    return true;
}

static bool is_pure_must_terms(const PlanNode& n) {
    if (n.kind != PlanNode::BOOL) return false;
    if (!n.should.empty() || !n.must_not.empty() || !n.filter.empty()) return false;
    for (auto& c : n.must)
        if (c.kind != PlanNode::TERM) return false;
    return true;
}

static void add_positive(FlatQuery& fq, const SplitView& sv, const PlanNode& c,
                         bool scored, float boost);

static void flatten_bool(FlatQuery& fq, const SplitView& sv, const PlanNode& n,
                         float boost);

// a MUST/FILTER clause
static void add_positive(FlatQuery& fq, const SplitView& sv, const PlanNode& c,
                         bool scored, float boost) {
    switch (c.kind) {
        case PlanNode::CACHE:
            fq.cache_nodes.push_back(&c);
            break;
        case PlanNode::TERM:
            add_term(fq, sv, c.field, c.value, ROLE_MUST, boost * c.boost, scored,
                     fq.n_must_groups++);
            break;
        case PlanNode::WILDCARD:
            if (add_wildcard(fq, sv, c, ROLE_MUST, fq.n_must_groups++) == 0)
                fq.match_none = true;  // must clause matching nothing
            break;
        case PlanNode::RANGE:
        case PlanNode::FIELD_PRESENCE:
            add_range_pred(fq, sv, c, false);
            break;
        case PlanNode::MATCH_ALL:
            break;  // no constraint
        case PlanNode::MATCH_NONE:
            fq.match_none = true;
            break;
        case PlanNode::BOOL:
            if (is_pure_must_terms(c)) {
                for (auto& m : c.must)
                    add_term(fq, sv, m.field, m.value, ROLE_MUST,
                             boost * c.boost * m.boost, scored,
                             fq.n_must_groups++);
            } else if (is_pure_should_terms(c)) {
                // OR-of-terms under must/filter: one must-GROUP (the kernel
                // ORs the group's bitsets, then ANDs the groups)
                uint32_t grp = fq.n_must_groups++;
                size_t before = fq.terms.size();
                for (auto& s : c.should)
                    add_term(fq, sv, s.field, s.value, ROLE_MUST,
                             boost * c.boost * s.boost, scored, grp, true);
                if (fq.terms.size() == before) fq.match_none = true;
            } else {
                throw std::runtime_error(
                    "nested boolean inside must/filter not flattenable (GPU r1)");
            }
            break;
        case PlanNode::PHRASE:
            throw std::runtime_error(
                "phrase clause not flattenable (device-bitmap path)");
    }
}

static void flatten_bool(FlatQuery& fq, const SplitView& sv, const PlanNode& n,
                         float boost) {
    boost *= n.boost;
    if (n.must.empty() && n.filter.empty() && n.should.empty() &&
        n.must_not.empty() && n.minimum_should_match <= 0) {
        // empty bool is match_all by ES convention (tantivy_query_ast.rs:193)
        fq.match_all = true;
        return;
    }
    bool has_req = !n.must.empty() || !n.filter.empty();
    for (auto& c : n.must) add_positive(fq, sv, c, fq.scoring, boost);
    for (auto& c : n.filter) add_positive(fq, sv, c, false, boost);

    // shoulds: flatten TERM / pure-OR bool children into SHOULD terms
    size_t should_clauses = 0;
    for (auto& c_raw : n.should) {
        const PlanNode& c = uncache(c_raw);
        if (c.kind == PlanNode::TERM) {
            add_term(fq, sv, c.field, c.value, ROLE_SHOULD, boost * c.boost,
                     fq.scoring);
            ++should_clauses;
        } else if (is_pure_should_terms(c)) {
            // OR-of-terms child: each term is an independent SHOULD here; one
            // clause satisfied iff any term matches — valid for msm<=1 only
            for (auto& t : c.should)
                add_term(fq, sv, t.field, t.value, ROLE_SHOULD,
                         boost * c.boost * t.boost, fq.scoring);
            ++should_clauses;
        } else if (c.kind == PlanNode::WILDCARD) {
            add_wildcard(fq, sv, c, ROLE_SHOULD, 0);
            ++should_clauses;
        } else if (c.kind == PlanNode::MATCH_NONE) {
            // contributes nothing
        } else {
            throw std::runtime_error("should clause not flattenable (GPU r1)");
        }
    }

    for (auto& c_raw : n.must_not) {
        const PlanNode& c = uncache(c_raw);
        if (c.kind == PlanNode::TERM)
            add_term(fq, sv, c.field, c.value, ROLE_MUST_NOT, 1.f, false);
        else if (c.kind == PlanNode::RANGE || c.kind == PlanNode::FIELD_PRESENCE)
            add_range_pred(fq, sv, c, true);
        else if (c.kind == PlanNode::WILDCARD)
            add_wildcard(fq, sv, c, ROLE_MUST_NOT, 0);
        else if (is_pure_should_terms(c))
            // not(a or b) = not a and not b: must_not is a union-then-subtract
            for (auto& t : c.should)
                add_term(fq, sv, t.field, t.value, ROLE_MUST_NOT, 1.f, false);
        else if (c.kind == PlanNode::MATCH_NONE) {
        } else if (c.kind == PlanNode::MATCH_ALL) {
            fq.match_none = true;
        } else
            throw std::runtime_error("must_not clause not flattenable (GPU r1)");
    }

    int64_t msm = n.minimum_should_match;
    // unset OR non-positive: shoulds are optional beside musts, but with no
    // positive clause at least one should must match (bool_query.rs:34-45 —
    // the es_compat msm=0/negative cases pin this)
    if (msm <= 0) msm = has_req ? 0 : 1;
    bool nested_or = false;
    for (auto& c : n.should) nested_or |= c.kind != PlanNode::TERM;
    if (msm > 1 && nested_or)
        throw std::runtime_error("minimum_should_match>1 over nested clauses");
    if (should_clauses == 0 && msm > 0 && !n.should.empty()) {
        fq.match_none = true;  // all required shoulds were absent terms
        return;
    }
    if (!has_req && n.should.empty()) {
        if (n.must_not.empty()) {
            fq.match_none = true;  // no clause at all matches nothing
            return;
        }
        // must_not only: implicit match_all base
        // (tantivy_query_ast.rs:310-322 pushes match_all)
        fq.match_all = true;
        fq.msm = 0;
        return;
    }
    fq.msm = uint32_t(msm);
    uint32_t n_must_terms = 0, n_should_terms = 0;
    for (auto& t : fq.terms) {
        n_must_terms += t.role == ROLE_MUST;
        n_should_terms += t.role == ROLE_SHOULD;
    }
    if (msm > 0 && n_should_terms == 0) {
        fq.match_none = true;  // required shoulds but every term absent
        return;
    }
    if (fq.scoring && msm > 0 && n_must_terms > 0)
        throw std::runtime_error("scored msm>0 with must terms not supported (GPU r1)");
    if (fq.scoring && msm > 1)
        throw std::runtime_error("scored minimum_should_match>1 not supported (GPU r1)");
    // base: match_all when no must terms (pred-only/should-optional base)
    fq.match_all = n_must_terms == 0 && (has_req || msm == 0);
    if (!has_req && msm == 0 && n.should.empty()) fq.match_none = true;
}

static FlatQuery flatten(const SplitView& sv, const PlanNode& plan, bool scoring) {
    FlatQuery fq;
    fq.scoring = scoring;
    if (scoring && plan_has_const_score(plan))
        throw std::runtime_error(
            "term_set/wildcard/phrase under _score sorting not supported "
            "(const-score semantics, qast.h)");
    if (plan.kind == PlanNode::PHRASE)
        throw std::runtime_error(
            "phrase not flattenable (device-bitmap path)");
    switch (plan.kind) {
        case PlanNode::MATCH_ALL:
            fq.match_all = true;
            break;
        case PlanNode::MATCH_NONE:
            fq.match_none = true;
            break;
        case PlanNode::TERM:
            add_term(fq, sv, plan.field, plan.value, ROLE_SHOULD, plan.boost, scoring);
            if (fq.terms.empty()) fq.match_none = true;
            fq.msm = 1;
            break;
        case PlanNode::RANGE:
        case PlanNode::FIELD_PRESENCE:
            fq.match_all = true;
            add_range_pred(fq, sv, plan, false);
            break;
        case PlanNode::WILDCARD:
            add_wildcard(fq, sv, plan, ROLE_SHOULD, 0);
            if (fq.terms.empty()) fq.match_none = true;
            fq.msm = 1;
            break;
        case PlanNode::CACHE:
            if (!scoring) {  // root cache: match_all base + cached bitmap
                fq.match_all = true;
                fq.cache_nodes.push_back(&plan);
            } else {
                return flatten(sv, uncache(plan), scoring);
            }
            break;
        case PlanNode::BOOL:
            flatten_bool(fq, sv, plan, 1.f);
            break;
    }
    if (fq.scoring)
        for (auto& t : fq.terms)
            if (t.role == ROLE_SHOULD && fq.msm > 0 && t.weight <= 0.f)
                throw std::runtime_error("non-positive BM25 weight (boost<=0?)");
    return fq;
}

// ------------------------------------------------------------ agg planning
struct AggPlan {
    std::vector<AggDef> defs;
    std::vector<AggDev> devs;            // one per def
    std::vector<const FastFieldView*> fields;
    // composite aggs: per-source column views (null for other kinds)
    std::vector<std::array<const FastFieldView*, 4>> comp_fields;
    // percentiles: per-agg gamma^k boundary doubles (appended to scratch;
    // empty for aggs without a sketch)
    std::vector<std::vector<double>> pbounds;
    size_t out_bytes = 0;                // total result bytes (counts + subs)
    std::vector<uint8_t> init;           // initial contents of the out region
};

static void col_min_max(const FastFieldView& f, double* mn, double* mx) {
    if (f.type == FastFieldView::U64) {
        *mn = double(uint64_t(f.min_value));
        *mx = double(uint64_t(f.max_value));
    } else if (f.type == FastFieldView::F64) {
        *mn = f.fmin;
        *mx = f.fmax;
    } else {
        *mn = double(f.min_value);
        *mx = double(f.max_value);
    }
}

static AggPlan plan_aggs(const SplitView& sv, const std::string& agg_json,
                         int64_t bucket_limit, uint64_t out_base) {
    AggPlan ap;
    ap.defs = parse_agg_request(agg_json);
    {
        auto agg_col_type = [&](const std::string& name) -> int {
            const FastFieldView* cf = sv.fast_field(name);
            return cf ? int(cf->type) : -1;
        };
        validate_agg_fields(ap.defs, agg_col_type);
    }
    uint64_t off = out_base;
    for (const AggDef& d : ap.defs) {
        const FastFieldView* f = sv.fast_field(d.field);
        ap.fields.push_back(f);
        ap.comp_fields.push_back({nullptr, nullptr, nullptr, nullptr});
        ap.pbounds.emplace_back();
        AggDev a{};
        a.lds_slot = 0xFF;
        a.p_si = 0xFF;
        if (d.kind == AggDef::COMPOSITE) {
            a.kind = AGGD_COMP;
            bool all_present = true;
            for (const CompSource& cs : d.comp) {
                const FastFieldView* sf = sv.fast_field(cs.field);
                if (!sf) { all_present = false; continue; }
                if (!cs.is_histo && sf->type != FastFieldView::STR)
                    throw std::runtime_error(
                        "composite terms source over a numeric column "
                        "(r1 limit)");
                if (cs.is_histo && sf->type == FastFieldView::STR)
                    throw std::runtime_error(
                        "composite histogram source over a str column");
                if (sf->multi)
                    throw std::runtime_error(
                        "composite source over a multi-valued column "
                        "(r1 limit)");
            }
            if (all_present) {
                if (sv.num_docs > (1u << 21))
                    throw std::runtime_error(
                        "composite aggregation on a >2M-doc split (r1 limit)");
                // pack sources MSB-first; component value count per source
                // bounds its bit width (comp_key_bits in qagg_format.h keeps
                // assembly decode in lockstep)
                uint32_t acc = 0;
                for (int si = int(d.comp.size()) - 1; si >= 0; --si) {
                    const CompSource& cs = d.comp[si];
                    const FastFieldView* sf = sv.fast_field(cs.field);
                    ap.comp_fields.back()[si] = sf;
                    uint64_t miss = cs.missing_bucket ? 1 : 0;
                    uint64_t n_values;
                    if (cs.is_histo) {
                        double mn, mx;
                        col_min_max(*sf, &mn, &mx);
                        int64_t b0 = int64_t(
                            std::floor((mn - cs.offset) / cs.interval));
                        int64_t b1 = int64_t(
                            std::floor((mx - cs.offset) / cs.interval));
                        a.c_base[si] = b0;
                        a.c_interval[si] = cs.interval;
                        a.c_offset[si] = cs.offset;
                        a.c_histo |= 1u << si;
                        n_values = uint64_t(b1 - b0 + 1) + miss;
                    } else {
                        n_values = uint64_t(sf->cardinality) + miss;
                    }
                    if (miss) a.c_missing |= 1u << si;
                    a.sub_values_off[si] = sf->values.off;
                    a.sub_nulls_off[si] = sf->nullable ? sf->nulls.off : 0;
                    a.sub_width[si] = cs.is_histo ? 8 : uint32_t(sf->ord_width);
                    a.sub_is_i64[si] =
                        sf->type == FastFieldView::U64   ? 0
                        : sf->type == FastFieldView::F64 ? 2
                                                         : 1;
                    a.c_shift[si] = acc;
                    acc += comp_key_bits(n_values);
                }
                if (acc > 63)
                    throw std::runtime_error(
                        "composite key exceeds 63 bits (r1 limit)");
                a.n_sub = uint32_t(d.comp.size());
                uint64_t want = uint64_t(sv.num_docs) * 2;
                uint32_t slots = 1024;
                while (slots < want) slots <<= 1;
                a.n_buckets = slots * 2 + 2;
            } else {
                a.n_buckets = 0;  // a source column is absent: empty result
            }
        } else if (d.kind == AggDef::TERMS || d.kind == AggDef::CARDINALITY) {
            a.kind = AGGD_TERMS;
            if (d.kind == AggDef::TERMS)
                for (auto& s : d.sub)
                    if (s.kind == MetricAgg::PERCENTILES)
                        throw std::runtime_error(
                            "percentiles under terms (r1 limit)");
            if (f && f->type == FastFieldView::STR) {
                a.n_buckets = f->cardinality;
                a.values_off = f->values.off;
                a.nulls_off = f->nullable ? f->nulls.off : 0;
                a.value_width = uint32_t(f->ord_width);
                if (f->multi) a.offsets_off = f->value_offsets.off;
                if (d.kind == AggDef::TERMS && !d.sub.empty()) {
                    a.n_sub = uint32_t(std::min<size_t>(d.sub.size(), 4));
                    if (d.sub.size() > 4)
                        throw std::runtime_error(
                            ">4 metric sub-aggregations");
                    for (size_t si = 0; si < d.sub.size(); ++si) {
                        const FastFieldView* sf =
                            sv.fast_field(d.sub[si].field);
                        if (sf && sf->type != FastFieldView::STR) {
                            a.sub_values_off[si] = sf->values.off;
                            a.sub_nulls_off[si] =
                                sf->nullable ? sf->nulls.off : 0;
                            a.sub_width[si] = 8;
                            a.sub_is_i64[si] =
                                sf->type == FastFieldView::U64   ? 0
                                : sf->type == FastFieldView::F64 ? 2
                                                                 : 1;
                        } else {
                            a.sub_values_off[si] = 0;
                            a.sub_width[si] = 0;
                        }
                    }
                }
            } else if (f && !f->multi) {
                // terms over a numeric fast column (tantivy term_agg keys by
                // the column value): device hash table sized so distinct
                // values can never fill it (<= num_docs). Bounded at 4M slots
                // — beyond that the per-doc worst-case probe cost is a
                // hazard, and no golden/benchmark path needs it.
                if (sv.num_docs > (1u << 21))
                    throw std::runtime_error(
                        "terms aggregation over a numeric fast field on a "
                        ">2M-doc split (r1 limit)");
                a.kind = AGGD_TERMS_NUM;
                uint64_t want = uint64_t(sv.num_docs) * 2;
                uint32_t slots = 1024;
                while (slots < want) slots <<= 1;
                a.n_buckets = slots * 2 + 2;
                a.values_off = f->values.off;
                a.nulls_off = f->nullable ? f->nulls.off : 0;
                a.value_width = 8;
                a.value_is_i64 = f->type == FastFieldView::U64   ? 0
                                 : f->type == FastFieldView::F64 ? 2
                                                                 : 1;
            } else {
                a.n_buckets = 0;  // no such column: zero buckets
            }
        } else if (d.kind == AggDef::METRIC &&
                   d.metric.kind == MetricAgg::PERCENTILES) {
            a.kind = AGGD_PERC;
            if (f && f->type != FastFieldView::STR && !f->multi) {
                double mn, mx;
                col_min_max(*f, &mn, &mx);
                if (mn < 0)
                    throw std::runtime_error(
                        "percentiles over negative values (r1 limit)");
                int32_t k_lo = 0, k_hi = 0;
                if (mx >= PERC_MIN_VALUE) {
                    k_lo = perc_key_for(std::max(mn, PERC_MIN_VALUE));
                    k_hi = perc_key_for(mx);
                }
                a.p_k_lo = k_lo;
                a.p_n_keys = uint32_t(k_hi - k_lo + 1);
                a.n_buckets = a.p_n_keys + 1;  // + zero bucket word
                a.values_off = f->values.off;
                a.nulls_off = f->nullable ? f->nulls.off : 0;
                a.value_width = 8;
                a.value_is_i64 = f->type == FastFieldView::U64   ? 0
                                 : f->type == FastFieldView::F64 ? 2
                                                                 : 1;
                perc_boundaries(k_lo, k_hi, ap.pbounds.back());
            }
        } else if (d.kind == AggDef::METRIC) {
            a.kind = AGGD_METRIC;
            if (f && f->type != FastFieldView::STR && !f->multi) {
                a.n_buckets = 1;  // one 40B stats slot at counts_out
                a.values_off = f->values.off;
                a.nulls_off = f->nullable ? f->nulls.off : 0;
                a.value_width = 8;
                a.value_is_i64 = f->type == FastFieldView::U64   ? 0
                                 : f->type == FastFieldView::F64 ? 2
                                                                 : 1;
            }
        } else if (d.kind == AggDef::RANGE) {
            a.kind = AGGD_RANGE;
            if (f && f->type != FastFieldView::STR && !f->multi) {
                if (d.ranges.size() > AGG_MAX_RANGES)
                    throw std::runtime_error(">16 ranges in range aggregation");
                a.n_buckets = uint32_t(d.ranges.size());
                a.n_ranges = a.n_buckets;
                a.values_off = f->values.off;
                a.nulls_off = f->nullable ? f->nulls.off : 0;
                a.value_width = 8;
                a.value_is_i64 = f->type == FastFieldView::U64   ? 0
                                 : f->type == FastFieldView::F64 ? 2
                                                                 : 1;
                for (size_t ri = 0; ri < d.ranges.size(); ++ri) {
                    if (d.ranges[ri].has_from) a.r_has_from |= 1u << ri;
                    if (d.ranges[ri].has_to) a.r_has_to |= 1u << ri;
                    a.r_from[ri] = d.ranges[ri].from;
                    a.r_to[ri] = d.ranges[ri].to;
                }
            }
        } else {
            a.kind = AGGD_HISTO;
            if (f && f->type != FastFieldView::STR) {
                a.interval = d.interval;
                a.offset = d.offset;
                a.values_off = f->values.off;
                a.nulls_off = f->nullable ? f->nulls.off : 0;
                a.value_width = 8;
                a.value_is_i64 = f->type == FastFieldView::U64   ? 0
                                 : f->type == FastFieldView::F64 ? 2
                                                                 : 1;
                double mn = double(f->min_value), mx = double(f->max_value);
                if (f->type == FastFieldView::U64) {
                    mn = double(uint64_t(f->min_value));
                    mx = double(uint64_t(f->max_value));
                } else if (f->type == FastFieldView::F64) {
                    mn = f->fmin;
                    mx = f->fmax;
                }
                int64_t b0 = int64_t(std::floor((mn - d.offset) / d.interval));
                int64_t b1 = int64_t(std::floor((mx - d.offset) / d.interval));
                a.base_index = b0;
                // exact integer fast path (kernels.hip histo_bucket): valid
                // when interval/offset are integral, the i64 load reads the
                // true value (u64 columns must fit i64), and |v-offset| stays
                // below 2^53 so the double division the oracle does can never
                // land within one rounding error of an integer other than at
                // exact multiples (error <= |v-off|*2^-53 < 1 < 1/interval
                // distance to the nearest non-multiple boundary).
                {
                    double iv = d.interval, of = d.offset;
                    bool integral = iv >= 1.0 && std::floor(iv) == iv &&
                                    std::floor(of) == of;
                    bool u64_ok = (f->type != FastFieldView::U64 ||
                                   uint64_t(f->max_value) < (1ull << 63)) &&
                                  f->type != FastFieldView::F64;
                    double span = std::max(std::fabs(mn - of), std::fabs(mx - of));
                    if (integral && u64_ok && span < 9.0e15 && iv < 9.0e15 &&
                        std::fabs(of) < 9.0e15 && std::fabs(mn) < 9.0e15 &&
                        std::fabs(mx) < 9.0e15) {
                        a.int_fast = 1;
                        a.i_interval = int64_t(iv);
                        a.i_offset = int64_t(of);
                        a.inv_interval = 1.0 / iv;
                    }
                }
                int64_t nb = b1 - b0 + 1;
                if (nb < 0 || nb > bucket_limit)
                    throw std::runtime_error(
                        "aggregation bucket limit exceeded (AggregationLimitsGuard)");
                a.n_buckets = uint32_t(nb);
                a.n_sub = uint32_t(std::min<size_t>(d.sub.size(), 4));
                if (d.sub.size() > 4)
                    throw std::runtime_error(">4 metric sub-aggregations");
                for (size_t si = 0; si < d.sub.size(); ++si) {
                    const FastFieldView* sf = sv.fast_field(d.sub[si].field);
                    if (sf && sf->type != FastFieldView::STR) {
                        a.sub_values_off[si] = sf->values.off;
                        a.sub_nulls_off[si] = sf->nullable ? sf->nulls.off : 0;
                        a.sub_width[si] = 8;
                        a.sub_is_i64[si] = sf->type == FastFieldView::U64   ? 0
                                           : sf->type == FastFieldView::F64 ? 2
                                                                            : 1;
                        if (d.sub[si].kind == MetricAgg::PERCENTILES) {
                            if (a.p_si != 0xFF)
                                throw std::runtime_error(
                                    ">1 percentiles sub-aggregation "
                                    "(r1 limit)");
                            double mn, mx;
                            col_min_max(*sf, &mn, &mx);
                            if (mn < 0)
                                throw std::runtime_error(
                                    "percentiles over negative values "
                                    "(r1 limit)");
                            int32_t k_lo = 0, k_hi = 0;
                            if (mx >= PERC_MIN_VALUE) {
                                k_lo = perc_key_for(
                                    std::max(mn, PERC_MIN_VALUE));
                                k_hi = perc_key_for(mx);
                            }
                            a.p_si = uint32_t(si);
                            a.p_k_lo = k_lo;
                            a.p_n_keys = uint32_t(k_hi - k_lo + 1);
                            if (uint64_t(a.n_buckets) * (a.p_n_keys + 1) >
                                (1ull << 25))
                                throw std::runtime_error(
                                    "percentiles sketch region too large "
                                    "(r1 limit)");
                            perc_boundaries(k_lo, k_hi, ap.pbounds.back());
                        }
                    } else {
                        a.sub_values_off[si] = 0;  // missing column: no values
                        a.sub_width[si] = 0;
                    }
                }
            } else {
                a.n_buckets = 0;
            }
        }
        // layout: counts u64[n_buckets] (METRIC: one 40 B stats slot) |
        // matched u64 | subs n_buckets*n_sub*40
        a.counts_out = off;
        off += a.kind == AGGD_METRIC ? 40 : uint64_t(a.n_buckets) * 8;
        a.matched_out = off;
        off += 8;
        a.sub_out = off;
        // AGGD_COMP uses n_sub for its sources and n_buckets for the hash
        // table words — no per-bucket stats region
        if (a.kind != AGGD_COMP) off += uint64_t(a.n_buckets) * a.n_sub * 40;
        if (a.p_si != 0xFF) {
            a.p_out = off;
            off += uint64_t(a.n_buckets) * (a.p_n_keys + 1) * 8;
        }
        ap.devs.push_back(a);
    }
    // LDS slot assignment: the first histogram agg that fits gets the LDS
    // histogram array, the first terms agg that fits gets the LDS count
    // table; everything else accumulates with global atomics.
    bool slot0 = false, slot1 = false;
    for (AggDev& a : ap.devs) {
        a.lds_rep = 1;
        if (a.kind == AGGD_HISTO && !slot0 && a.n_buckets &&
            a.n_buckets <= AGG_LDS_BUCKETS) {
            a.lds_slot = 0;
            slot0 = true;
        } else if (a.kind == AGGD_TERMS && !slot1 && a.n_buckets &&
                   a.n_buckets <= AGG_LDS_BUCKETS) {
            a.lds_slot = 1;
            slot1 = true;
        }
        // two interleaved copies when they fit the LDS array: lane-parity
        // split halves same-bucket atomic serialization (Zipf ords)
        if (a.lds_slot <= 1 && a.n_buckets * 2 <= AGG_LDS_BUCKETS) a.lds_rep = 2;
    }
    ap.out_bytes = off - out_base;
    // init pattern: zeros except sub min slots = f64_sortable(+inf) pattern max
    ap.init.assign(ap.out_bytes, 0);
    for (size_t i = 0; i < ap.devs.size(); ++i) {
        const AggDev& a = ap.devs[i];
        uint64_t ones = ~0ull;
        if (a.kind == AGGD_METRIC && a.n_buckets) {
            uint64_t slot = a.counts_out - out_base;
            memcpy(&ap.init[slot + 16], &ones, 8);  // min: max sortable
        }
        if ((a.kind == AGGD_TERMS_NUM || a.kind == AGGD_COMP) && a.n_buckets) {
            // hash table: every key word starts at the empty sentinel ~0
            uint32_t slots = (a.n_buckets - 2) >> 1;
            for (uint32_t s2 = 0; s2 < slots; ++s2)
                memcpy(&ap.init[a.counts_out - out_base + uint64_t(s2) * 16],
                       &ones, 8);
        }
        // AGGD_COMP reuses n_sub for its sources and has NO stats sub region
        if (a.kind == AGGD_COMP) continue;
        for (uint64_t b = 0; b < a.n_buckets; ++b)
            for (uint32_t s = 0; s < a.n_sub; ++s) {
                if (s == a.p_si) continue;  // percentiles slot: stays zero
                uint64_t slot = a.sub_out - out_base + (b * a.n_sub + s) * 40;
                memcpy(&ap.init[slot + 16], &ones, 8);  // min slot: max sortable
                // max slot stays 0 (= most negative sortable)
            }
    }
    return ap;
}

// ------------------------------------------------------------ per-split search
struct SplitResult {
    uint64_t num_hits = 0;
    std::vector<pb::PartialHit> hits;
    IntermediateAggResults aggs;
    bool has_aggs = false;
    uint64_t micros = 0;
    std::string error;
    // required terms this SEGMENT proved absent; the caller populates the
    // absence cache only with terms absent from EVERY segment of the split
    std::vector<std::pair<std::string, std::string>> absent_required;
};

// sort specs, mirroring the oracle's (collector.rs:403-414 sort-key
// extraction; sorting.md None-last + GlobalDocId tie-break semantics)
struct SortSpec {
    enum Comp { DOC_ID, SCORE, FAST_FIELD } comp = DOC_ID;
    const FastFieldView* ff = nullptr;  // null: unknown field -> None values
    int order = 1;                      // 0 asc 1 desc
};

static pb::SortByValue sort_value_of(const SortSpec& s, uint32_t doc, float score) {
    pb::SortByValue v;
    switch (s.comp) {
        case SortSpec::DOC_ID:
            break;
        case SortSpec::SCORE:
            v.kind = pb::SortByValue::F64;
            v.f64 = double(score);
            break;
        case SortSpec::FAST_FIELD: {
            const FastFieldView* f = s.ff;
            if (!f || !f->present(doc)) break;
            if (f->type == FastFieldView::MIXED) {
                // echo the ORIGINAL typed value (true stays a bool, a u64
                // beyond 2^63 stays exact) — tags/raw sections
                uint64_t r = f->mixed_raw(doc);
                switch (f->mixed_tag(doc)) {
                    case 1:
                        v.kind = pb::SortByValue::I64;
                        v.i64 = int64_t(r);
                        break;
                    case 2:
                        v.kind = pb::SortByValue::F64;
                        memcpy(&v.f64, &r, 8);
                        break;
                    case 3:
                        v.kind = pb::SortByValue::BOOL;
                        v.boolean = r != 0;
                        break;
                    default:
                        v.kind = pb::SortByValue::U64;
                        v.u64 = r;
                }
            } else if (f->type == FastFieldView::U64) {
                v.kind = pb::SortByValue::U64;
                v.u64 = f->u64(doc);
            } else if (f->type == FastFieldView::DATETIME) {
                v.kind = pb::SortByValue::I64;
                v.i64 = f->i64(doc) * 1000000;  // ms -> ns (sorting.md default
                                                // output unix_timestamp_nanos)
            } else if (f->type == FastFieldView::STR) {
                v.kind = pb::SortByValue::U64;
                v.u64 = f->ord(doc);
            } else if (f->type == FastFieldView::F64) {
                v.kind = pb::SortByValue::F64;
                v.f64 = f->f64(doc);
            } else {
                v.kind = pb::SortByValue::I64;
                v.i64 = f->i64(doc);
            }
            break;
        }
    }
    return v;
}

// Evaluate a CACHE node's inner subtree into a device HitSet bitmap
// (tile-major u32 words) and memoize it per (split, fingerprint). The
// assembly mirrors the hot-path one minus scoring/aggs/top-K: the query
// shape becomes an unscored flattened boolean whose per-tile match bitset
// is stored via QueryDev::bitmap_out.
static uint8_t* resolve_hitset(qw_ctx* ctx, const DeviceSplit& ds,
                               const PlanNode& inner, const Schema& schema);

// combine helper: dst op= src (op per k_bitmap_combine)
static void bitmap_combine(qw_ctx* ctx, uint8_t* dst, const uint8_t* src,
                           size_t bm_bytes, uint32_t op) {
    uint64_t words = bm_bytes / 4;
    uint32_t grid = uint32_t(std::min<uint64_t>(2048, (words + 255) / 256));
    hipLaunchKernelGGL(k_bitmap_combine, dim3(grid), dim3(256), 0, ctx->stream,
                       (uint32_t*)dst, (const uint32_t*)src, words, op);
}

// Recursive boolean evaluation over device bitmaps — the fallback for plan
// shapes the single-pass flattened kernel cannot express (nested booleans
// under must_not, presence/range in should position, …). Each sub-plan's
// bitmap is memoized in the (split, fingerprint) hitset cache, so repeated
// subtrees and repeated queries pay once. Unscored semantics only.
static uint8_t* bitmap_eval(qw_ctx* ctx, const DeviceSplit& ds,
                            const PlanNode& node, const Schema& schema) {
    const SplitView& sv = ds.view;
    try {
        return resolve_hitset(ctx, ds, node, schema);  // flattenable fast path
    } catch (const std::exception&) {
        if (node.kind != PlanNode::BOOL) throw;
    }
    if (node.minimum_should_match > 1)
        throw std::runtime_error(
            "minimum_should_match>1 over nested clauses (GPU r1)");
    uint32_t n_tiles = (sv.num_docs + TILE_DOCS - 1) / TILE_DOCS;
    size_t bm_bytes = size_t(n_tiles) * (TILE_DOCS / 32) * 4;
    std::string key = sv.split_id;
    key += '\0';
    key += std::to_string(sv.segment_ord);
    key += '\x1e';
    plan_fingerprint(node, key);
    uint8_t* bm = nullptr;
    qw_check_budget(ctx, bm_bytes);
    HIP_CHECK(hipMalloc(&bm, bm_bytes));
    bool has_req = !node.must.empty() || !node.filter.empty();
    bool first = true;
    for (auto* clauses : {&node.must, &node.filter})
        for (const PlanNode& c : *clauses) {
            uint8_t* cb = bitmap_eval(ctx, ds, c, schema);
            bitmap_combine(ctx, bm, cb, bm_bytes, first ? 3u : 0u);
            first = false;
        }
    int64_t msm = node.minimum_should_match;
    if (msm <= 0) msm = has_req ? 0 : 1;
    if (!node.should.empty() && msm > 0) {
        uint8_t* su = nullptr;
        qw_check_budget(ctx, bm_bytes);
        HIP_CHECK(hipMalloc(&su, bm_bytes));
        bool sfirst = true;
        for (const PlanNode& c : node.should) {
            uint8_t* cb = bitmap_eval(ctx, ds, c, schema);
            bitmap_combine(ctx, su, cb, bm_bytes, sfirst ? 3u : 1u);
            sfirst = false;
        }
        if (first) {
            bitmap_combine(ctx, bm, su, bm_bytes, 3u);
            first = false;
        } else bitmap_combine(ctx, bm, su, bm_bytes, 0u);
        HIP_CHECK(hipStreamSynchronize(ctx->stream));
        (void)hipFree(su);
    }
    if (first) {
        // no positive clause: implicit match_all base (must_not-only)
        HIP_CHECK(hipMemsetAsync(bm, 0xFF, bm_bytes, ctx->stream));
        first = false;
    }
    for (const PlanNode& c : node.must_not) {
        uint8_t* cb = bitmap_eval(ctx, ds, c, schema);
        bitmap_combine(ctx, bm, cb, bm_bytes, 2u);
    }
    // mask the tail beyond num_docs (match_all memset covers padding bits)
    HIP_CHECK(hipStreamSynchronize(ctx->stream));
    {
        uint32_t ndocs = sv.num_docs;
        uint32_t full_words = ndocs / 32;
        size_t total_words = bm_bytes / 4;
        if (full_words < total_words) {
            std::vector<uint32_t> tail(total_words - full_words, 0);
            if (ndocs % 32)
                tail[0] = 0;  // recompute partial word below
            // read-modify host-side for the single partial word
            if (ndocs % 32) {
                uint32_t w = 0;
                HIP_CHECK(hipMemcpy(&w, bm + full_words * 4, 4,
                                    hipMemcpyDeviceToHost));
                w &= (1u << (ndocs % 32)) - 1u;
                tail[0] = w;
            }
            HIP_CHECK(hipMemcpy(bm + full_words * 4, tail.data(),
                                tail.size() * 4, hipMemcpyHostToDevice));
        }
    }
    ctx->hitsets.put(key, bm, bm_bytes);
    return bm;
}

static uint8_t* resolve_hitset(qw_ctx* ctx, const DeviceSplit& ds,
                               const PlanNode& inner, const Schema& schema) {
    const SplitView& sv = ds.view;
    std::string key = sv.split_id;
    key += '\0';
    key += std::to_string(sv.segment_ord);
    key += '\x1e';
    plan_fingerprint(inner, key);
    if (uint8_t* bm = ctx->hitsets.get(key)) return bm;

    uint32_t n_tiles = (sv.num_docs + TILE_DOCS - 1) / TILE_DOCS;
    size_t bm_bytes = size_t(n_tiles) * (TILE_DOCS / 32) * 4;
    uint8_t* bm = nullptr;
    qw_check_budget(ctx, bm_bytes);
    HIP_CHECK(hipMalloc(&bm, bm_bytes));

    if (inner.kind == PlanNode::PHRASE) {
        // multi-token phrase (slop 0, full_text_query.rs phrase mode):
        // k_phrase_bitmap — one thread per posting of the driver token,
        // consecutive-position chain over the field's position streams
        const TextFieldView* f = sv.text_field(inner.field);
        bool none = !f;
        if (f && !f->has_positions) {
            (void)hipFree(bm);
            throw std::runtime_error(
                "phrase query needs record: position on field " + inner.field);
        }
        if (inner.phrase_toks.size() > PHRASE_MAX_TOKS) {
            (void)hipFree(bm);
            throw std::runtime_error("phrase longer than 8 tokens (r2 limit)");
        }
        PhraseDev p{};
        p.n_toks = uint32_t(inner.phrase_toks.size());
        p.num_docs = sv.num_docs;
        for (size_t i = 0; i < inner.phrase_toks.size() && !none; ++i) {
            const std::string& tok = inner.phrase_toks[i];
            int64_t tid = f->find_term(tok.data(), tok.size());
            if (tid < 0) {
                none = true;  // absent token: phrase matches nothing
                break;
            }
            PhraseTokDev& td = p.tok[i];
            td.skip_off = f->skip.off + f->h_skip_off[tid];
            td.payload_off = f->payload.off;
            td.pos_start_off =
                f->pos_start.off + (f->h_skip_off[tid] / 16) * 4;
            td.positions_off = f->positions.off;
            td.n_blocks = f->h_n_blocks[tid];
            td.df = f->h_doc_freq[tid];
        }
        HIP_CHECK(hipMemsetAsync(bm, 0, bm_bytes, ctx->stream));
        if (!none && p.tok[0].df) {
            uint32_t grid =
                std::min<uint32_t>(2048, (p.tok[0].df + 255) / 256);
            hipLaunchKernelGGL(k_phrase_bitmap, dim3(grid), dim3(256), 0,
                               ctx->stream, ds.d_image, p, (uint32_t*)bm);
        }
        HIP_CHECK(hipStreamSynchronize(ctx->stream));
        HIP_CHECK(hipGetLastError());
        ctx->hitsets.put(key, bm, bm_bytes);
        return bm;
    }

    FlatQuery fq;
    try {
        fq = flatten(sv, inner, false);
    } catch (...) {
        (void)hipFree(bm);
        throw;
    }
    for (const PlanNode* cn : fq.cache_nodes) {  // nested cache nodes
        PredDev p{};
        p.type = PRED_BITSET;
        p.abs_bitmap =
            (uint64_t)resolve_hitset(ctx, ds, cn->cache_inner.front(), schema);
        fq.preds.push_back(p);
    }
    if (fq.match_none) {
        HIP_CHECK(hipMemsetAsync(bm, 0, bm_bytes, ctx->stream));
        HIP_CHECK(hipStreamSynchronize(ctx->stream));
        ctx->hitsets.put(key, bm, bm_bytes);
        return bm;
    }

    // scratch: TermDev[] | PredDev[] | zero ktab | per-term block ranges
    std::vector<TermDev> terms(fq.terms.size());
    uint32_t n_must_not = 0;
    for (size_t i = 0; i < fq.terms.size(); ++i) {
        const FlatQuery::FTerm& t = fq.terms[i];
        TermDev& d = terms[i];
        d.role = t.role;
        d.grp = t.grp;
        d.n_blocks = t.f->h_n_blocks[t.tid];
        d.skip_off = t.f->skip.off + t.f->h_skip_off[t.tid];
        d.payload_off = t.f->payload.off;
        d.norms_off = 0;
        d.weight = 0.f;
        d.ktab_idx = 0;
        n_must_not += t.role == ROLE_MUST_NOT;
    }
    size_t off_terms = 0;
    size_t off_preds = off_terms + terms.size() * sizeof(TermDev);
    size_t off_ktabs = off_preds + fq.preds.size() * sizeof(PredDev);
    size_t scratch_bytes = off_ktabs + 256 * 4;
    for (size_t i = 0; i < fq.terms.size(); ++i) {
        const FlatQuery::FTerm& t = fq.terms[i];
        terms[i].ranges_addr =
            term_ranges_addr(ctx, ds, t.f, t.tid, n_tiles, sv.num_docs);
    }
    std::vector<uint8_t> scratch(scratch_bytes, 0);
    memcpy(scratch.data() + off_terms, terms.data(), terms.size() * sizeof(TermDev));
    memcpy(scratch.data() + off_preds, fq.preds.data(),
           fq.preds.size() * sizeof(PredDev));
    qw_ensure_acct(ctx, ctx->d_scratch, scratch_bytes);
    qw_ensure_acct(ctx, ctx->d_results, 64);
    HIP_CHECK(hipMemcpyAsync(ctx->d_scratch.p, scratch.data(), scratch_bytes,
                             hipMemcpyHostToDevice, ctx->stream));
    HIP_CHECK(hipMemsetAsync(ctx->d_results.p, 0, 64, ctx->stream));

    uint32_t n_should = 0;
    for (auto& t : fq.terms) n_should += t.role == ROLE_SHOULD;
    QueryDev q{};
    q.split = ds.d_image;
    q.scratch = ctx->d_scratch.p;
    q.results = ctx->d_results.p;
    q.num_docs = sv.num_docs;
    q.n_tiles = n_tiles;
    q.n_terms = uint32_t(terms.size());
    q.n_must = fq.n_must_groups;
    q.n_must_not = n_must_not;
    q.n_preds = uint32_t(fq.preds.size());
    q.msm = fq.msm;
    q.match_all = fq.match_all ? 1 : 0;
    q.collect_hits = 1;
    q.terms_off = off_terms;
    q.preds_off = off_preds;
    q.ktabs_off = off_ktabs;
    q.tile_counts_off = 0;  // unused (do_count = 0)
    q.cand_count_off = 0;   // phase-B reservation counter only
    q.cand_off = 0;
    q.cand_cap = 0;  // no candidate writes, bitmap only
    q.bitmap_out = (uint64_t)bm;
    bool ns = n_should > 0;
    bool nb = fq.n_must_groups > 0 || n_must_not > 0;
    launch_leaf_tile(ns, nb, false, true, dim3(n_tiles), ctx->stream, q, 0u,
                     n_tiles, 0u);
    HIP_CHECK(hipGetLastError());
    HIP_CHECK(hipStreamSynchronize(ctx->stream));
    ctx->hitsets.put(key, bm, bm_bytes);
    return bm;
}

static uint32_t f32_sortable_h(float f) {
    uint32_t b;
    memcpy(&b, &f, 4);
    return (b & 0x80000000u) ? ~b : (b | 0x80000000u);
}

static void record_kernel_time(qw_ctx* ctx, const char* name, float ms) {
    KernelTimer& t = ctx->timers[name];
    t.total_ms += ms;
    t.launches += 1;
}

// single terms in REQUIRED position (must/filter chains, unwrapping cache
// nodes) — the candidates the absence cache can short-circuit on; restates
// the reference's required-term extraction (quickwit-query required_terms
// feeding leaf.rs:769-778)
static void collect_required_terms(
    const PlanNode& n, std::vector<std::pair<std::string, std::string>>& out) {
    switch (n.kind) {
        case PlanNode::TERM:
            out.emplace_back(n.field, n.value);
            break;
        case PlanNode::PHRASE:
            // every phrase token is required (the phrase cannot match if
            // any token is absent)
            for (const std::string& t : n.phrase_toks)
                out.emplace_back(n.field, t);
            break;
        case PlanNode::BOOL:
            for (const PlanNode& c : n.must) collect_required_terms(c, out);
            for (const PlanNode& c : n.filter) collect_required_terms(c, out);
            break;
        case PlanNode::CACHE:
            if (!n.cache_inner.empty())
                collect_required_terms(n.cache_inner.front(), out);
            break;
        default:
            break;
    }
}

static SplitResult search_split_gpu(qw_ctx* ctx, const DeviceSplit& ds,
                                    const pb::SearchRequest& req, const Schema& schema) {
    SplitResult out;
    auto t0 = std::chrono::steady_clock::now();
    static const bool timing = getenv("QW_TIMING") != nullptr;
    auto tprev = t0;
    auto mark = [&](const char* what) {
        if (!timing) return;
        auto now = std::chrono::steady_clock::now();
        fprintf(stderr, "[qw_timing] %-18s %8.1f us\n", what,
                std::chrono::duration_cast<std::chrono::nanoseconds>(now - tprev)
                        .count() /
                    1e3);
        tprev = now;
    };
    const SplitView& sv = ds.view;
    ctx->hitsets.begin_query();  // unpin previous query's bitmap entries

    PlanNode plan = parse_query_ast(req.query_ast, schema);
    if ((req.start_timestamp || req.end_timestamp) && !schema.timestamp_field.empty()) {
        // leaf.rs:977 rewrite_request: [start,end) seconds over the ts column
        PlanNode ts;
        ts.kind = PlanNode::RANGE;
        ts.field = schema.timestamp_field;
        if (req.start_timestamp) {
            ts.lo.kind = Bound::INCLUDED;
            ts.lo.ival = *req.start_timestamp * 1000;
        }
        if (req.end_timestamp) {
            ts.hi.kind = Bound::EXCLUDED;
            ts.hi.ival = *req.end_timestamp * 1000;
        }
        PlanNode b;
        b.kind = PlanNode::BOOL;
        b.filter.push_back(std::move(ts));
        b.must.push_back(std::move(plan));
        plan = std::move(b);
    }

    // ---- negative/absence term cache probe (leaf.rs:761-827): a required
    // term proven absent by an earlier search short-circuits this split —
    // replace the plan with MATCH_NONE so the standard empty-response path
    // (shaped agg blob, stats) runs with no kernel and no dict lookups
    {
        std::vector<std::pair<std::string, std::string>> reqs;
        collect_required_terms(plan, reqs);
        bool hit = false;
        for (auto& ft : reqs)
            if (ctx->absence.keys.count(
                    AbsenceCache::key(sv.split_id, ft.first, ft.second))) {
                hit = true;
                break;
            }
        if (!reqs.empty()) (hit ? ctx->absence.hits : ctx->absence.misses)++;
        if (hit) {
            PlanNode none;
            none.kind = PlanNode::MATCH_NONE;
            plan = std::move(none);
        }
    }

    if (req.sort_fields.size() > 2)
        throw std::runtime_error("more than two sort fields (search.proto:269)");
    std::vector<SortSpec> specs;
    for (auto& sf : req.sort_fields) {
        SortSpec s;
        s.order = sf.sort_order;
        if (sf.field_name == "_score") s.comp = SortSpec::SCORE;
        else {
            s.comp = SortSpec::FAST_FIELD;
            s.ff = sv.fast_field(sf.field_name);
        }
        specs.push_back(s);
    }
    bool scoring = false;
    for (auto& s : specs) scoring |= s.comp == SortSpec::SCORE;
    int order1 = specs.empty() ? 1 : specs[0].order;
    int order2 = specs.size() > 1 ? specs[1].order : 1;
    // narrow (8B) candidate records embed the f32 _score key + doc tiebreak:
    // exact device-side selection. fast-field or two-field sorts use wide
    // (16B) records selected by primary key only; exact order over the
    // survivors is re-established on host (kernels.hip wide_sort_key).
    bool wide = (!specs.empty() && specs[0].comp == SortSpec::FAST_FIELD) ||
                specs.size() > 1;

    FlatQuery fq;
    try {
        fq = flatten(sv, plan, scoring);
    } catch (const std::exception& e) {
        std::string msg = e.what();
        if (scoring || msg.find("flatten") == std::string::npos) throw;
        // unscored plan the single-pass kernel cannot express: evaluate the
        // whole boolean recursively into a device bitmap and search
        // match_all + PRED_BITSET (results identical; scores don't exist)
        uint8_t* bmp = bitmap_eval(ctx, ds, plan, schema);
        fq = FlatQuery{};
        fq.scoring = false;
        fq.match_all = true;
        PredDev p{};
        p.type = PRED_BITSET;
        p.abs_bitmap = (uint64_t)bmp;
        fq.preds.push_back(p);
    }
    // report required terms flattening proved absent; the caller inserts
    // into the absence cache (for multi-segment splits only terms absent
    // from EVERY segment qualify)
    out.absent_required = fq.absent_required;
    // CACHE nodes in filter position -> device HitSet bitmaps (PRED_BITSET)
    for (const PlanNode* cn : fq.cache_nodes) {
        PredDev p{};
        p.type = PRED_BITSET;
        p.abs_bitmap = (uint64_t)bitmap_eval(ctx, const_cast<DeviceSplit&>(ds),
                                             cn->cache_inner.front(), schema);
        fq.preds.push_back(p);
    }
    uint64_t leaf_max_hits = req.max_hits + req.start_offset;

    if (fq.match_none || sv.num_docs == 0) {
        if (req.aggregation_request) {
            // empty-but-shaped blob, like the oracle over an empty match set
            for (const AggDef& d : parse_agg_request(*req.aggregation_request)) {
                AggResult r;
                r.name = d.name;
                r.kind = d.kind == AggDef::TERMS         ? 3
                         : d.kind == AggDef::CARDINALITY ? 3
                         : d.kind == AggDef::COMPOSITE   ? 3
                         : d.kind == AggDef::RANGE       ? 4
                         : d.kind == AggDef::METRIC
                             ? (d.metric.kind == MetricAgg::PERCENTILES ? 6 : 5)
                         : d.kind == AggDef::HISTOGRAM ? 2
                                                       : 1;
                if (d.kind == AggDef::COMPOSITE) r.key_kind = 9;
                for (auto& s : d.sub) {
                    r.sub_names.push_back(s.name);
                    r.sub_kinds.push_back(
                        s.kind == MetricAgg::PERCENTILES ? 1 : 0);
                }
                out.aggs.aggs.push_back(std::move(r));
            }
            out.has_aggs = true;
        }
        out.micros = uint64_t(std::chrono::duration_cast<std::chrono::microseconds>(
                                  std::chrono::steady_clock::now() - t0)
                                  .count());
        return out;
    }

    AggPlan ap;
    bool do_aggs = req.aggregation_request.has_value();

    uint32_t n_tiles = (sv.num_docs + TILE_DOCS - 1) / TILE_DOCS;
    bool pure_match_all = fq.match_all && fq.preds.empty() && fq.terms.empty();
    // hits trivially enumerable only under doc-id order (leaf.rs default)
    const pb::PartialHit* after =
        req.search_after ? &*req.search_after : nullptr;
    // typed cursor conversion into the sort fields' u64 domains
    // (convert_to_u64_ff_val, collector.rs:214-340)
    auto kind_of = [&](const SortSpec& s) {
        if (s.comp == SortSpec::SCORE) return SortFieldKind::SCORE;
        if (s.comp != SortSpec::FAST_FIELD || !s.ff) return SortFieldKind::NONE;
        switch (s.ff->type) {
            case FastFieldView::U64: return SortFieldKind::U64;
            case FastFieldView::I64: return SortFieldKind::I64;
            case FastFieldView::DATETIME: return SortFieldKind::DATETIME;
            case FastFieldView::F64: return SortFieldKind::F64;
            case FastFieldView::MIXED: return SortFieldKind::MIXED;
            default: return SortFieldKind::STR;
        }
    };
    CursorKey ck1, ck2;
    if (after) {
        ck1 = specs.empty() ? CursorKey{}
                            : convert_cursor_key(after->sort_value,
                                                 kind_of(specs[0]), order1);
        ck2 = specs.size() < 2 ? CursorKey{}
                               : convert_cursor_key(after->sort_value2,
                                                    kind_of(specs[1]), order2);
        if (ck1.disabled) after = nullptr;  // cursor before all values
    }
    bool trivial_hits = pure_match_all && specs.empty() && !after;
    // candidate collection: needed unless hits are trivially enumerable
    bool collect = leaf_max_hits > 0 && !trivial_hits && !fq.match_none;
    bool need_kernel = !pure_match_all || do_aggs || collect;

    // ---- scratch assembly (descriptors + ktabs + block ranges), one H2D
    std::vector<TermDev> terms(fq.terms.size());
    uint32_t n_must = 0, n_must_not = 0;
    std::vector<float> ktabs(fq.ktab_fields.size() * 0);  // filled below

    // K tables per field actually referenced by scored terms
    std::vector<const TextFieldView*> ktab_fields;
    auto ktab_idx_of = [&](const TextFieldView* f) {
        for (size_t i = 0; i < ktab_fields.size(); ++i)
            if (ktab_fields[i] == f) return uint32_t(i);
        ktab_fields.push_back(f);
        return uint32_t(ktab_fields.size() - 1);
    };
    for (size_t i = 0; i < fq.terms.size(); ++i) {
        const FlatQuery::FTerm& t = fq.terms[i];
        TermDev& d = terms[i];
        d.role = t.role;
        d.n_blocks = t.f->h_n_blocks[t.tid];
        d.skip_off = t.f->skip.off + t.f->h_skip_off[t.tid];
        d.payload_off = t.f->payload.off;
        d.norms_off = t.f->has_norms ? t.f->fieldnorms.off : 0;
        d.weight = t.weight;
        d.grp = t.grp;
        d.ktab_idx = fq.scoring ? ktab_idx_of(t.f) : 0;
        n_must += t.role == ROLE_MUST;
        n_must_not += t.role == ROLE_MUST_NOT;
    }
    ktabs.resize(std::max<size_t>(1, ktab_fields.size()) * 256, 0.f);
    for (size_t k = 0; k < ktab_fields.size(); ++k) {
        const TextFieldView* f = ktab_fields[k];
        double avgdl = f->total_tokens > 0 && sv.num_docs > 0
                           ? double(f->total_tokens) / double(sv.num_docs)
                           : 0.0;
        const double K1 = 1.2, B = 0.75;
        for (int i = 0; i < 256; ++i) {
            double fn = double(FIELDNORM_TABLE.v[i]);
            ktabs[k * 256 + i] =
                float(K1 * (1.0 - B + B * fn / (avgdl > 0 ? avgdl : 1.0)));
        }
    }

    // scratch layout
    size_t off_terms = 0;
    size_t off_preds = off_terms + terms.size() * sizeof(TermDev);
    size_t off_aggs = off_preds + fq.preds.size() * sizeof(PredDev);
    // aggs descriptors appended after planning (below)
    // results layout
    size_t r_tile_counts = 0;
    size_t r_cand_count = r_tile_counts + size_t(n_tiles) * 4;
    r_cand_count = (r_cand_count + 63) & ~size_t(63);
    size_t r_hist = r_cand_count + 64;  // cand_count u32 + survivors count u32
    size_t r_agg = r_hist + TOPK_BINS * 4;
    r_agg = (r_agg + 63) & ~size_t(63);

    if (do_aggs)
        ap = plan_aggs(sv, *req.aggregation_request, ctx->agg_bucket_limit, r_agg);
    size_t r_cand = r_agg + ((ap.out_bytes + 63) & ~size_t(63));
    size_t cand_cap = collect ? sv.num_docs : 0;
    size_t cand_rec = wide ? 16 : 8;
    size_t results_bytes = r_cand + cand_cap * cand_rec;

    size_t off_ktabs = off_aggs + ap.devs.size() * sizeof(AggDev);
    size_t scratch_bytes = off_ktabs + ktabs.size() * 4;
    // percentiles boundary tables (gamma^k doubles the kernel searches)
    scratch_bytes = (scratch_bytes + 7) & ~size_t(7);
    for (size_t i = 0; i < ap.devs.size(); ++i)
        if (!ap.pbounds[i].empty()) {
            ap.devs[i].p_bound_off = scratch_bytes;
            scratch_bytes += ap.pbounds[i].size() * 8;
        }

    // dense positive base => eager predicate evaluation (kernels.hip):
    // estimate the pre-predicate match density from the term dfs
    {
        double density = 1.0;
        if (!fq.match_all && !fq.terms.empty()) {
            double n = double(std::max<uint32_t>(sv.num_docs, 1));
            double mind = 1.0, sum = 0.0;
            bool has_must = false;
            for (const FlatQuery::FTerm& t : fq.terms) {
                double dfr = double(t.f->h_doc_freq[t.tid]) / n;
                if (t.role == ROLE_MUST) {
                    mind = std::min(mind, dfr);
                    has_must = true;
                } else if (t.role == ROLE_SHOULD) sum += dfr;
            }
            density = has_must ? mind
                      : fq.msm > 0 ? std::min(1.0, sum)
                                   : 1.0;
        }
        if (density >= 0.15)
            for (PredDev& pr : fq.preds)
                if (pr.type == PRED_RANGE_U64 || pr.type == PRED_RANGE_I64 ||
                    pr.type == PRED_RANGE_F64)
                    pr.flags |= PRED_EAGER;
    }

    mark("plan");
    // per-term block ranges come from the split-level device cache
    // (term_ranges_addr): first query pays the walk+upload, repeats don't
    ctx->h_scratch.ensure(scratch_bytes);
    uint8_t* scratch = ctx->h_scratch.p;
    for (size_t i = 0; i < fq.terms.size(); ++i) {
        const FlatQuery::FTerm& t = fq.terms[i];
        terms[i].ranges_addr =
            term_ranges_addr(ctx, ds, t.f, t.tid, n_tiles, sv.num_docs);
    }
    mark("ranges");

    memcpy(scratch + off_terms, terms.data(), terms.size() * sizeof(TermDev));
    memcpy(scratch + off_preds, fq.preds.data(),
           fq.preds.size() * sizeof(PredDev));
    memcpy(scratch + off_aggs, ap.devs.data(), ap.devs.size() * sizeof(AggDev));
    memcpy(scratch + off_ktabs, ktabs.data(), ktabs.size() * 4);
    for (size_t i = 0; i < ap.devs.size(); ++i)
        if (!ap.pbounds[i].empty())
            memcpy(scratch + ap.devs[i].p_bound_off,
                   ap.pbounds[i].data(), ap.pbounds[i].size() * 8);

    uint64_t matched = 0;
    mark("assembly");
    std::vector<uint64_t> top_keys;  // survivors, sorted best-first
    std::function<void(uint64_t)> rerun_select;  // search_after retry hook
    uint64_t sel_band_n = 0, sel_kwant = 0;
    // prefetched selection pass-0 histogram, in pinned staging (async D2H
    // to pageable memory degrades to a blocking staged copy)
    ctx->h_surv.ensure(TOPK_BINS * 4);
    uint32_t* hist0 = (uint32_t*)ctx->h_surv.p;

    if (need_kernel) {
        qw_ensure_acct(ctx, ctx->d_scratch, scratch_bytes);
        qw_ensure_acct(ctx, ctx->d_results, results_bytes);
        HIP_CHECK(hipMemcpyAsync(ctx->d_scratch.p, scratch, scratch_bytes,
                                 hipMemcpyHostToDevice, ctx->stream));
        HIP_CHECK(hipMemsetAsync(ctx->d_results.p, 0, r_agg, ctx->stream));
        if (do_aggs && ap.out_bytes)
            HIP_CHECK(hipMemcpyAsync(ctx->d_results.p + r_agg, ap.init.data(),
                                     ap.out_bytes, hipMemcpyHostToDevice, ctx->stream));

        QueryDev q{};
        q.split = ds.d_image;
        q.scratch = ctx->d_scratch.p;
        q.results = ctx->d_results.p;
        q.num_docs = sv.num_docs;
        q.n_tiles = n_tiles;
        q.n_terms = uint32_t(terms.size());
        q.n_must = fq.n_must_groups;  // group count (TermDev::grp)
        q.n_must_not = n_must_not;
        q.n_preds = uint32_t(fq.preds.size());
        q.n_aggs = uint32_t(ap.devs.size());
        q.msm = fq.msm;
        q.scoring = fq.scoring ? 1 : 0;
        q.match_all = fq.match_all ? 1 : 0;
        q.collect_hits = collect ? 1 : 0;
        q.sort_asc = (!specs.empty() && specs[0].comp != SortSpec::DOC_ID &&
                      order1 == 0)
                         ? 1
                         : 0;
        q.wide_cand = wide ? 1 : 0;
        if (wide) {
            const SortSpec& s0 = specs[0];
            if (s0.comp == SortSpec::SCORE) {
                q.sort_src = 1;
            } else if (s0.ff) {
                const FastFieldView* f = s0.ff;
                if (f->multi)
                    throw std::runtime_error(
                        "sort by a multi-valued fast field not supported");
                q.sort_values_off = f->values.off;
                q.sort_nulls_off = f->nullable ? f->nulls.off : 0;
                if (f->type == FastFieldView::STR) {
                    q.sort_src = 2;
                    q.sort_width = uint32_t(f->ord_width);
                } else {
                    // MIXED columns store f64-monotonic u64 keys: ascending
                    // u64 order IS the numeric order -> device treats them
                    // like a u64 column
                    q.sort_src = f->type == FastFieldView::U64    ? 2
                                 : f->type == FastFieldView::MIXED ? 2
                                 : f->type == FastFieldView::F64  ? 4
                                                                  : 3;
                    q.sort_width = 8;
                }
            } else {
                q.sort_src = 0;  // unknown field: every key is None
            }
        }
        q.terms_off = off_terms;
        q.preds_off = off_preds;
        q.aggs_off = off_aggs;
        q.ktabs_off = off_ktabs;
        // LDS staging of fieldnorms + K tables for the scoring decode path
        // (kernels.hip): enabled when every scored term's norms live in ONE
        // section (single-field queries — the flagship shape) and the K
        // tables fit the LDS budget. QW_NO_LDS_NORMS / QW_NO_LDS_KTAB are
        // perf-experiment kill switches.
        if (fq.scoring) {
            // measured on MI355X at the 100M flagship: norms staging LOSES
            // (the +8KB LDS drops occupancy 4->3 WGs/CU while the norm
            // gathers already hit L2 — union covers ~25% of docs, ~16 hits
            // per cacheline); ktab staging is noise. Both stay available as
            // opt-in experiment switches (gpurun_out/r02_exp_variants.log).
            static const bool want_norms = getenv("QW_LDS_NORMS") != nullptr;
            static const bool want_ktab = getenv("QW_LDS_KTAB") != nullptr;
            uint64_t common = 0;
            bool uniform = true;
            for (const TermDev& t : terms)
                if (t.norms_off) {
                    if (!common) common = t.norms_off;
                    else if (common != t.norms_off) uniform = false;
                }
            q.norms_stage_off = (want_norms && uniform) ? common : 0;
            q.n_ktabs = (want_ktab && ktab_fields.size() <= KTAB_LDS_MAX)
                            ? uint32_t(ktab_fields.size())
                            : 0;
        }
        q.tile_counts_off = r_tile_counts;
        q.cand_count_off = r_cand_count;
        q.cand_off = r_cand;
        q.cand_cap = cand_cap;
        q.hist_off = r_hist;

        const char* kname = !fq.terms.empty() ? "union_bm25"
                            : do_aggs         ? "column_agg"
                                              : "range_filter";
        // query-shape flags select the template instantiation whose dynamic
        // LDS holds exactly the sections this shape touches (kernels.hip)
        uint32_t n_should = uint32_t(terms.size()) - n_must - n_must_not;
        bool ns = n_should > 0 || (fq.scoring && !terms.empty());
        bool nb = n_must > 0 || n_must_not > 0;
        bool na = do_aggs && !ap.devs.empty();
        // agg workloads: capped grid so the once-per-workgroup LDS flush
        // stays cheap (each WG owns several tiles). For PURE column scans
        // (no posting decode) the grid also scales DOWN with the split
        // size — the flush costs grid x buckets global atomics regardless
        // of docs, which dominated small splits (256 WGs x 4 waves still
        // fill the chip's 1024 resident-wave slots). Decode-carrying
        // kernels keep the wide grid: fewer WGs serialize the per-tile
        // decode loops and cost more than the flush saves (measured,
        // config5 ablation).
        uint32_t grid = n_tiles;
        if (na)
            grid = fq.terms.empty()
                       ? std::min<uint32_t>(
                             {n_tiles, 2048u,
                              std::max<uint32_t>(256u, n_tiles / 8)})
                       : std::min<uint32_t>(n_tiles, 2048u);
        // straight-line agg path (kernels.hip): histo[0] int_fast LDS
        // non-nullable no-subs (+ optional terms[1] LDS non-nullable)
        if (na && !ap.devs.empty() && ap.devs[0].kind == AGGD_HISTO &&
            ap.devs[0].lds_slot == 0 && ap.devs[0].int_fast &&
            ap.devs[0].n_sub == 0 && ap.devs[0].nulls_off == 0 &&
            ap.devs.size() <= 2 &&
            (ap.devs.size() < 2 ||
             (ap.devs[1].kind == AGGD_TERMS && ap.devs[1].lds_slot == 1 &&
              ap.devs[1].nulls_off == 0 && ap.devs[1].offsets_off == 0 &&
              ap.devs[1].n_buckets > 0)))
            q.agg_fast = 1;
        // terms-only straight-line path (config5's shape): one LDS terms
        // agg, non-nullable single-valued ord column, no subs
        if (na && ap.devs.size() == 1 && ap.devs[0].kind == AGGD_TERMS &&
            ap.devs[0].lds_slot == 1 && ap.devs[0].n_sub == 0 &&
            ap.devs[0].nulls_off == 0 && ap.devs[0].offsets_off == 0 &&
            ap.devs[0].n_buckets > 0)
            q.agg_fast = 1;
        static const bool agg_nt = getenv("QW_AGG_NT") != nullptr;
        q.agg_nt = agg_nt ? 1 : 0;
        static const bool nt_decode = getenv("QW_NT_DECODE") != nullptr;
        q.nt_decode = nt_decode ? 1 : 0;
        HIP_CHECK(hipEventRecord(ctx->ev_start, ctx->stream));
        launch_leaf_tile(ns, nb, na, collect, dim3(grid), ctx->stream, q, 0u,
                         n_tiles, 1u);
        HIP_CHECK(hipEventRecord(ctx->ev_stop, ctx->stream));
        HIP_CHECK(hipGetLastError());

        // enqueue top-K histogram pass 0 behind the main kernel (count read
        // device-side) so its result arrives with the same synchronize
        bool hist0_valid = false;
        if (collect && !after) {
            uint64_t* d_cand0 = (uint64_t*)(ctx->d_results.p + r_cand);
            uint32_t* d_hist0 = (uint32_t*)(ctx->d_results.p + r_hist);
            uint32_t* d_n0 = (uint32_t*)(ctx->d_results.p + r_cand_count);
            HIP_CHECK(hipMemsetAsync(d_hist0, 0, TOPK_BINS * 4, ctx->stream));
            // candidate count is device-side only at this point: size the
            // grid for the worst case (every doc a candidate)
            uint32_t hgrid =
                std::min<uint32_t>(512, (sv.num_docs + 4095) / 4096);
            if (wide)
                hipLaunchKernelGGL(k_cand_hist_w, dim3(hgrid), dim3(256), 0,
                                   ctx->stream, d_cand0, d_n0, 0ull, 0u, d_hist0);
            else
                hipLaunchKernelGGL(k_cand_hist, dim3(hgrid), dim3(256), 0,
                                   ctx->stream, d_cand0, d_n0, 0ull, 0u, d_hist0);
            HIP_CHECK(hipMemcpyAsync(hist0, d_hist0, TOPK_BINS * 4,
                                     hipMemcpyDeviceToHost, ctx->stream));
            hist0_valid = true;
        }

        // ---- download counts (into pinned staging: no pageable D2H)
        ctx->h_counts.ensure(size_t(n_tiles) * 4 + 64);
        uint32_t* tile_counts = (uint32_t*)ctx->h_counts.p;
        uint32_t* pc = (uint32_t*)(ctx->h_counts.p + size_t(n_tiles) * 4);
        HIP_CHECK(hipMemcpyAsync(tile_counts, ctx->d_results.p + r_tile_counts,
                                 size_t(n_tiles) * 4, hipMemcpyDeviceToHost,
                                 ctx->stream));
        if (collect)
            HIP_CHECK(hipMemcpyAsync(pc, ctx->d_results.p + r_cand_count, 4,
                                     hipMemcpyDeviceToHost, ctx->stream));
        if (do_aggs && ap.out_bytes) {
            // aggregation results ride the same async batch + sync (a
            // blocking copy after hit building cost ~40 us per split)
            ctx->h_agg.ensure(ap.out_bytes + 64);
            HIP_CHECK(hipMemcpyAsync(ctx->h_agg.p, ctx->d_results.p + r_agg,
                                     ap.out_bytes, hipMemcpyDeviceToHost,
                                     ctx->stream));
        }
        HIP_CHECK(hipStreamSynchronize(ctx->stream));
        uint32_t cand_n = collect ? *pc : 0;
        mark("main+pass0+sync");
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, ctx->ev_start, ctx->ev_stop));
        record_kernel_time(ctx, kname, ms);
        for (uint32_t i = 0; i < n_tiles; ++i) matched += tile_counts[i];

        // ---- top-K selection over candidates (device histogram refinement
        // + host exact sort of the survivors; top_k_collector.rs semantics)
        if (collect && cand_n > 0) {
            uint32_t rw = wide ? 2 : 1;  // u64 words per candidate record
            uint64_t* d_cand = (uint64_t*)(ctx->d_results.p + r_cand);
            uint32_t* d_hist = (uint32_t*)(ctx->d_results.p + r_hist);
            uint32_t* d_scount = (uint32_t*)(ctx->d_results.p + r_cand_count) + 1;
            // search_after: pre-compact to the cursor's primary-value band
            // (inclusive upper bound in device-key space; ties and the
            // (v2, split, seg, doc) chain are settled by the exact host
            // filter at hit building, with a growing-K retry on shortfall)
            if (after) {
                uint64_t ceil_key = ~0ull;
                const SortKey& k1 = ck1.key;  // converted to the field domain
                if (wide) {
                    if (!k1.has) ceil_key = 0;  // cursor in the None region
                    else {
                        uint64_t S = k1.key;
                        if (!specs.empty() && specs[0].comp == SortSpec::SCORE) {
                            uint32_t kh = f32_sortable_h(float(u64_to_f64(S)));
                            S = (uint64_t(kh) << 32) | 0xFFFFFFFFull;
                        } else if (!specs.empty() && specs[0].ff &&
                                   specs[0].ff->type == FastFieldView::DATETIME) {
                            int64_t ns = u64_to_i64(S);  // cursor carries ns
                            int64_t ms = ns / 1000000 - ((ns % 1000000) < 0 ? 1 : 0);
                            S = i64_to_u64(ms);
                        }
                        // I64/F64: same u64 map as the device key;
                        // U64/STR ord: identity
                        ceil_key = (order1 == 0) ? ~S : S;
                    }
                } else {
                    uint32_t kh = 0;
                    if (fq.scoring && k1.has)
                        kh = f32_sortable_h(float(u64_to_f64(k1.key)));
                    bool nasc = !specs.empty() &&
                                specs[0].comp != SortSpec::DOC_ID && order1 == 0;
                    if (nasc) kh = ~kh;
                    ceil_key = (uint64_t(kh) << 32) | 0xFFFFFFFFull;
                }
                qw_ensure_acct(ctx, ctx->d_cand2, size_t(cand_n) * 8 * rw + 16);
                uint32_t* d_c2n = (uint32_t*)(ctx->d_results.p + r_cand_count) + 2;
                HIP_CHECK(hipMemsetAsync(d_c2n, 0, 4, ctx->stream));
                uint32_t pgrid = std::min<uint32_t>(2048, (cand_n + 255) / 256);
                if (wide)
                    hipLaunchKernelGGL(k_cand_compact_w, dim3(pgrid), dim3(256), 0,
                                       ctx->stream, d_cand, cand_n, 0ull, ceil_key,
                                       (uint64_t*)ctx->d_cand2.p, d_c2n, cand_n);
                else
                    hipLaunchKernelGGL(k_cand_compact, dim3(pgrid), dim3(256), 0,
                                       ctx->stream, d_cand, cand_n, 0ull, ceil_key,
                                       (uint64_t*)ctx->d_cand2.p, d_c2n, cand_n);
                uint32_t n2 = 0;
                HIP_CHECK(hipMemcpyAsync(&n2, d_c2n, 4, hipMemcpyDeviceToHost,
                                         ctx->stream));
                HIP_CHECK(hipStreamSynchronize(ctx->stream));
                d_cand = (uint64_t*)ctx->d_cand2.p;
                cand_n = n2;
            }
        }
        if (collect && cand_n > 0) {
            uint64_t* d_cand = after ? (uint64_t*)ctx->d_cand2.p
                                     : (uint64_t*)(ctx->d_results.p + r_cand);
            uint32_t* d_hist = (uint32_t*)(ctx->d_results.p + r_hist);
            uint32_t* d_scount = (uint32_t*)(ctx->d_results.p + r_cand_count) + 1;
            uint32_t rw = wide ? 2 : 1;  // u64 words per candidate record
            sel_band_n = cand_n;
            // capture the block-local selection state BY VALUE: the lambda is
            // re-invoked by the search_after growing-K retry (below) after
            // this block's scope has ended
            rerun_select = [&, d_cand, rw, cand_n, d_hist, d_scount,
                            hist0_valid](uint64_t Kwant) {
            uint64_t K = std::min<uint64_t>(Kwant, cand_n);
            uint64_t prefix = 0;
            uint32_t prefix_bits = 0;
            uint64_t survivors = cand_n, above = 0, Krem = K;
            uint64_t floor_key = 0;
            std::vector<uint32_t> hist(TOPK_BINS);
            HIP_CHECK(hipEventRecord(ctx->ev_start, ctx->stream));
            while (true) {
                // refine until the band is small: the D2H + host sort of
                // the survivors is on the critical path (a 65k-u64 band
                // cost ~90 us at K=1000; one more 4096-bin pass is ~20 us)
                if (survivors <= std::max<uint64_t>(4 * K, 16384) || prefix_bits >= 48) {
                    if (prefix_bits == 0) floor_key = 0;
                    break;
                }
                if (prefix_bits == 0 && hist0_valid) {
                    memcpy(hist.data(), hist0,
                           TOPK_BINS * 4);  // prefetched with the main sync
                } else {
                    HIP_CHECK(hipMemsetAsync(d_hist, 0, TOPK_BINS * 4, ctx->stream));
                    // grid small enough that the per-WG 4096-bin LDS flush
                    // (global atomics ∝ grid) stays cheap, yet fills the chip
                    uint32_t hgrid = std::min<uint32_t>(512, (cand_n + 4095) / 4096);
                    uint32_t* d_n = after
                        ? (uint32_t*)(ctx->d_results.p + r_cand_count) + 2
                        : (uint32_t*)(ctx->d_results.p + r_cand_count);
                    if (wide)
                        hipLaunchKernelGGL(k_cand_hist_w, dim3(hgrid), dim3(256), 0,
                                           ctx->stream, d_cand, d_n, prefix,
                                           prefix_bits, d_hist);
                    else
                        hipLaunchKernelGGL(k_cand_hist, dim3(hgrid), dim3(256), 0,
                                           ctx->stream, d_cand, d_n, prefix,
                                           prefix_bits, d_hist);
                    HIP_CHECK(hipMemcpyAsync(hist.data(), d_hist, TOPK_BINS * 4,
                                             hipMemcpyDeviceToHost, ctx->stream));
                    HIP_CHECK(hipStreamSynchronize(ctx->stream));
                }
                uint64_t cum = 0;
                int b = TOPK_BINS - 1;
                for (; b >= 0; --b) {
                    if (cum + hist[b] >= Krem) break;
                    cum += hist[b];
                }
                if (b < 0) b = 0;
                uint32_t shift = 64 - prefix_bits - 12;
                floor_key = (prefix_bits ? (prefix << (64 - prefix_bits)) : 0) |
                            (uint64_t(uint32_t(b)) << shift);
                above += cum;
                survivors = above + hist[b];
                Krem = Krem - cum;
                prefix = (prefix << 12) | uint64_t(uint32_t(b));
                prefix_bits += 12;
            }
            // #keys >= floor_key is known EXACTLY from the histogram walk, so
            // no round trip for the compact count: one async D2H of survivors
            qw_ensure_acct(ctx, ctx->d_survivors, survivors * 8 * rw + 16);
            HIP_CHECK(hipMemsetAsync(d_scount, 0, 4, ctx->stream));
            uint32_t cgrid = std::min<uint32_t>(2048, (cand_n + 255) / 256);
            if (wide)
                hipLaunchKernelGGL(k_cand_compact_w, dim3(cgrid), dim3(256), 0,
                                   ctx->stream, d_cand, cand_n, floor_key, ~0ull,
                                   (uint64_t*)ctx->d_survivors.p, d_scount,
                                   uint32_t(survivors));
            else
                hipLaunchKernelGGL(k_cand_compact, dim3(cgrid), dim3(256), 0,
                                   ctx->stream, d_cand, cand_n, floor_key, ~0ull,
                                   (uint64_t*)ctx->d_survivors.p, d_scount,
                                   uint32_t(survivors));
            top_keys.resize(survivors * rw);
            ctx->h_topk.ensure(survivors * 8 * rw + 16);
            HIP_CHECK(hipMemcpyAsync(ctx->h_topk.p, ctx->d_survivors.p,
                                     survivors * 8 * rw, hipMemcpyDeviceToHost,
                                     ctx->stream));
            HIP_CHECK(hipEventRecord(ctx->ev_stop, ctx->stream));
            HIP_CHECK(hipStreamSynchronize(ctx->stream));
            memcpy(top_keys.data(), ctx->h_topk.p, survivors * 8 * rw);
            float tms = 0;
            HIP_CHECK(hipEventElapsedTime(&tms, ctx->ev_start, ctx->ev_stop));
            record_kernel_time(ctx, "topk_select", tms);
            if (!wide) {
                std::sort(top_keys.begin(), top_keys.end(),
                          std::greater<uint64_t>());
                if (top_keys.size() > K && !after) top_keys.resize(K);
            }
            };  // rerun_select
            sel_kwant = std::min<uint64_t>(
                leaf_max_hits + (after ? 1024 : 0), cand_n);
            rerun_select(sel_kwant);
            mark("topk_select");
        }
    } else {
        matched = sv.num_docs;  // pure match_all, no aggs
    }
    if (pure_match_all) matched = sv.num_docs;

    out.num_hits = matched;

    // ---- build PartialHits
    auto mk_hit = [&](uint32_t doc, float score) {
        pb::PartialHit h;
        h.split_id = sv.split_id;
        h.segment_ord = sv.segment_ord;
        h.doc_id = doc;
        if (!specs.empty()) h.sort_value = sort_value_of(specs[0], doc, score);
        if (specs.size() > 1) h.sort_value2 = sort_value_of(specs[1], doc, score);
        return h;
    };
    if (leaf_max_hits > 0) {
        if (trivial_hits) {
            // hits enumerable without the kernel: doc-id order (desc default)
            uint64_t k = std::min<uint64_t>(leaf_max_hits, sv.num_docs);
            for (uint64_t i = 0; i < k; ++i) {
                uint32_t doc = order1 == 1 ? uint32_t(sv.num_docs - 1 - i)
                                           : uint32_t(i);
                out.hits.push_back(mk_hit(doc, 0.f));
            }
        } else {
            // build hits from the survivors; with search_after, the exact
            // cursor filter runs before truncation and the selection widens
            // (growing K) until the page is full or the band is exhausted
            for (;;) {
                out.hits.clear();
                out.hits.reserve(top_keys.size() / (wide ? 2 : 1));
                if (wide) {
                    // survivors carry (selection key, score|doc); the exact
                    // reference order ((sort_value, sort_value2, GlobalDocId),
                    // sorting.md:14-26) is re-established over the survivors
                    size_t n = top_keys.size() / 2;
                    out.hits.reserve(n);
                    for (size_t i = 0; i < n; ++i) {
                        uint64_t aux = top_keys[2 * i + 1];
                        uint32_t doc = uint32_t(aux);
                        float score = 0.f;
                        if (fq.scoring) {
                            uint32_t b = uint32_t(aux >> 32);
                            memcpy(&score, &b, 4);
                        }
                        out.hits.push_back(mk_hit(doc, score));
                    }
                } else {
                    bool asc = !specs.empty() && order1 == 0;
                    for (uint64_t key : top_keys) {
                        uint32_t kh = uint32_t(key >> 32), kl = uint32_t(key);
                        if (asc) {
                            kh = ~kh;
                            kl = ~kl;
                        }
                        float score = 0.f;
                        if (fq.scoring) {
                            uint32_t b =
                                (kh & 0x80000000u) ? (kh & ~0x80000000u) : ~kh;
                            memcpy(&score, &b, 4);
                        }
                        out.hits.push_back(mk_hit(kl, score));
                    }
                }
                if (after) {
                    const pb::PartialHit& c = *after;
                    out.hits.erase(
                        std::remove_if(out.hits.begin(), out.hits.end(),
                                       [&](const pb::PartialHit& h) {
                                           return !after_cursor(
                                               h, c, ck1, ck2, order1, order2,
                                               !specs.empty() &&
                                                   kind_of(specs[0]) ==
                                                       SortFieldKind::MIXED,
                                               specs.size() > 1 &&
                                                   kind_of(specs[1]) ==
                                                       SortFieldKind::MIXED);
                                       }),
                        out.hits.end());
                }
                auto cmp = [&](const pb::PartialHit& a, const pb::PartialHit& b) {
                    return hit_before(a, b, order1, order2);
                };
                size_t k = std::min<size_t>(leaf_max_hits, out.hits.size());
                std::partial_sort(out.hits.begin(), out.hits.begin() + k,
                                  out.hits.end(), cmp);
                out.hits.resize(k);
                if (!after || !rerun_select ||
                    out.hits.size() >= std::min<uint64_t>(leaf_max_hits, sel_band_n) ||
                    sel_kwant >= sel_band_n)
                    break;
                sel_kwant = std::min<uint64_t>(sel_band_n, sel_kwant * 8);
                rerun_select(sel_kwant);
            }
        }
    }
    mark("build_hits");

    // ---- aggregation download + assembly (QAGG1 intermediate)
    if (do_aggs) {
        // downloaded asynchronously with the counts batch above (pinned)
        ctx->h_agg.ensure(ap.out_bytes + 64);
        const uint8_t* agg_out_p = ctx->h_agg.p;
        for (size_t i = 0; i < ap.defs.size(); ++i) {
            const AggDef& d = ap.defs[i];
            const AggDev& a = ap.devs[i];
            const FastFieldView* f = ap.fields[i];
            AggResult r;
            r.name = d.name;
            for (auto& s : d.sub) r.sub_names.push_back(s.name);
            const uint8_t* base = agg_out_p + (a.counts_out - r_agg);
            const uint64_t* counts = (const uint64_t*)base;
            if (d.kind == AggDef::TERMS || d.kind == AggDef::CARDINALITY) {
                r.kind = 3;
                if (a.n_buckets) {
                    if (a.nulls_off || a.offsets_off) {
                        uint64_t m = 0;
                        memcpy(&m, agg_out_p + (a.matched_out - r_agg), 8);
                        r.terms_matched_docs = m;
                    } else {
                        // non-nullable column: every matched doc has a value
                        r.terms_matched_docs = matched;
                    }
                    if (a.kind == AGGD_TERMS_NUM) {
                        r.key_kind = f->type == FastFieldView::U64   ? 1
                                     : f->type == FastFieldView::F64 ? 3
                                                                     : 2;
                        uint32_t slots = (a.n_buckets - 2) >> 1;
                        if (counts[2 * uint64_t(slots) + 1])
                            throw std::runtime_error(
                                "numeric terms hash table overflow");
                        for (uint64_t s2 = 0; s2 < slots; ++s2) {
                            uint64_t key = counts[2 * s2], c = counts[2 * s2 + 1];
                            if (key != ~0ull && c)
                                r.term_counts.emplace_back(num_term_key(key), c);
                        }
                        if (uint64_t sc2 = counts[2 * uint64_t(slots)])
                            r.term_counts.emplace_back(num_term_key(~0ull), sc2);
                        std::sort(r.term_counts.begin(), r.term_counts.end());
                    } else {
                        const uint8_t* subs =
                            agg_out_p + (a.sub_out - r_agg);
                        for (uint32_t o = 0; o < a.n_buckets; ++o)
                            if (counts[o]) {
                                r.term_counts.emplace_back(f->dict_entry(o),
                                                           counts[o]);
                                if (a.n_sub) {
                                    std::vector<StatsPayload> ts(d.sub.size());
                                    for (uint32_t s = 0; s < a.n_sub; ++s) {
                                        const uint8_t* slot =
                                            subs +
                                            (uint64_t(o) * a.n_sub + s) * 40;
                                        StatsPayload sp2;
                                        uint64_t mn, mx;
                                        memcpy(&sp2.count, slot, 8);
                                        memcpy(&sp2.sum, slot + 8, 8);
                                        memcpy(&mn, slot + 16, 8);
                                        memcpy(&mx, slot + 24, 8);
                                        memcpy(&sp2.sum_sq, slot + 32, 8);
                                        if (sp2.count) {
                                            sp2.min = u64_to_f64(mn);
                                            sp2.max = u64_to_f64(mx);
                                        }
                                        ts[s] = sp2;
                                    }
                                    r.term_subs.push_back(std::move(ts));
                                }
                            }
                    }
                    if (d.kind == AggDef::TERMS)
                        truncate_terms_split(
                            r, effective_split_size(d.size, d.split_size),
                            d.order_target, d.order_asc, &d.sub);
                }
            } else if (d.kind == AggDef::COMPOSITE) {
                // decode packed 63-bit keys into the canonical per-source
                // byte encoding (split-local ords -> term strings) so the
                // cross-split QAGG merge compares composite tuples directly
                r.kind = 3;
                r.key_kind = 9;
                if (a.n_buckets) {
                    uint32_t slots = (a.n_buckets - 2) >> 1;
                    if (counts[2 * uint64_t(slots) + 1])
                        throw std::runtime_error(
                            "composite hash table overflow");
                    // recompute per-source widths exactly as plan_aggs did
                    uint32_t width[4] = {0, 0, 0, 0};
                    for (size_t si = 0; si < d.comp.size(); ++si) {
                        const CompSource& cs = d.comp[si];
                        const FastFieldView* sf = ap.comp_fields[i][si];
                        uint64_t miss = cs.missing_bucket ? 1 : 0;
                        uint64_t nv;
                        if (cs.is_histo) {
                            double mn, mx;
                            col_min_max(*sf, &mn, &mx);
                            int64_t b0 = int64_t(
                                std::floor((mn - cs.offset) / cs.interval));
                            int64_t b1 = int64_t(
                                std::floor((mx - cs.offset) / cs.interval));
                            nv = uint64_t(b1 - b0 + 1) + miss;
                        } else {
                            nv = uint64_t(sf->cardinality) + miss;
                        }
                        width[si] = comp_key_bits(nv);
                    }
                    auto decode_key = [&](uint64_t key) {
                        std::string ck;
                        for (size_t si = 0; si < d.comp.size(); ++si) {
                            const CompSource& cs = d.comp[si];
                            uint64_t comp = (key >> a.c_shift[si]) &
                                            ((1ull << width[si]) - 1);
                            uint64_t miss = cs.missing_bucket ? 1 : 0;
                            if (miss && comp == 0) {
                                comp_encode_null(ck);
                            } else if (cs.is_histo) {
                                double v = double(a.c_base[si] +
                                                  int64_t(comp - miss)) *
                                               cs.interval +
                                           cs.offset;
                                comp_encode_f64(ck, v);
                            } else {
                                comp_encode_str(
                                    ck, ap.comp_fields[i][si]->dict_entry(
                                            uint32_t(comp - miss)));
                            }
                        }
                        return ck;
                    };
                    for (uint64_t s2 = 0; s2 < slots; ++s2) {
                        uint64_t key = counts[2 * s2], c = counts[2 * s2 + 1];
                        if (key != ~0ull && c)
                            r.term_counts.emplace_back(decode_key(key), c);
                    }
                    std::sort(r.term_counts.begin(), r.term_counts.end());
                }
            } else if (d.kind == AggDef::METRIC &&
                       d.metric.kind == MetricAgg::PERCENTILES) {
                r.kind = 6;
                if (a.n_buckets) {
                    r.sketch.zero = counts[0];
                    for (uint32_t j = 1; j < a.n_buckets; ++j)
                        if (counts[j])
                            r.sketch.counts[a.p_k_lo + int32_t(j) - 1] =
                                counts[j];
                }
            } else if (d.kind == AggDef::METRIC) {
                r.kind = 5;
                if (a.n_buckets) {
                    const uint8_t* slot = agg_out_p + (a.counts_out - r_agg);
                    StatsPayload sp2;
                    uint64_t mn, mx;
                    memcpy(&sp2.count, slot, 8);
                    memcpy(&sp2.sum, slot + 8, 8);
                    memcpy(&mn, slot + 16, 8);
                    memcpy(&mx, slot + 24, 8);
                    memcpy(&sp2.sum_sq, slot + 32, 8);
                    if (sp2.count) {
                        sp2.min = u64_to_f64(mn);
                        sp2.max = u64_to_f64(mx);
                    }
                    r.metric = sp2;
                }
            } else if (d.kind == AggDef::RANGE) {
                r.kind = 4;
                for (uint32_t ri = 0; ri < a.n_buckets; ++ri) {
                    if (!counts[ri]) continue;
                    AggBucket b;
                    b.key = double(ri);  // range index; finalize maps back
                    b.doc_count = counts[ri];
                    r.buckets.push_back(std::move(b));
                }
            } else {
                r.kind = d.kind == AggDef::DATE_HISTOGRAM ? 1 : 2;
                for (auto& sdef : d.sub)
                    r.sub_kinds.push_back(
                        sdef.kind == MetricAgg::PERCENTILES ? 1 : 0);
                const uint8_t* subs = agg_out_p + (a.sub_out - r_agg);
                const uint64_t* pwords =
                    a.p_si != 0xFF
                        ? (const uint64_t*)(agg_out_p + (a.p_out - r_agg))
                        : nullptr;
                for (uint32_t bi = 0; bi < a.n_buckets; ++bi) {
                    if (!counts[bi]) continue;
                    AggBucket b;
                    b.key = double(a.base_index + int64_t(bi)) * d.interval + d.offset;
                    b.doc_count = counts[bi];
                    b.sub.resize(d.sub.size());
                    b.psub.resize(d.sub.size());
                    if (pwords) {
                        const uint64_t* pw =
                            pwords + uint64_t(bi) * (a.p_n_keys + 1);
                        SketchPayload& pp = b.psub[a.p_si];
                        pp.zero = pw[0];
                        for (uint32_t j = 1; j <= a.p_n_keys; ++j)
                            if (pw[j])
                                pp.counts[a.p_k_lo + int32_t(j) - 1] = pw[j];
                    }
                    for (uint32_t s = 0; s < a.n_sub; ++s) {
                        if (s == a.p_si) continue;
                        const uint8_t* slot = subs + (uint64_t(bi) * a.n_sub + s) * 40;
                        StatsPayload sp2;
                        uint64_t mn, mx;
                        memcpy(&sp2.count, slot, 8);
                        memcpy(&sp2.sum, slot + 8, 8);
                        memcpy(&mn, slot + 16, 8);
                        memcpy(&mx, slot + 24, 8);
                        memcpy(&sp2.sum_sq, slot + 32, 8);
                        if (sp2.count) {
                            sp2.min = u64_to_f64(mn);
                            sp2.max = u64_to_f64(mx);
                        }
                        b.sub[s] = sp2;
                    }
                    r.buckets.push_back(std::move(b));
                }
            }
            out.aggs.aggs.push_back(std::move(r));
        }
        out.has_aggs = true;
    }
    mark("aggs_assembly");

    out.micros = uint64_t(std::chrono::duration_cast<std::chrono::microseconds>(
                              std::chrono::steady_clock::now() - t0)
                              .count());
    return out;
}

// ------------------------------------------------------------ pruning
// Restates CanSplitDoBetter (leaf.rs:1337-1553), simplify_search_request
// (leaf.rs:1666-1710) and the request-level part of rewrite_request
// (leaf.rs:977-990, remove_redundant_timestamp_range :1106-1211). The
// FindTraceIdsAggregation variant (Jaeger traces) is out of scope.
// Missing split timestamps read as 0 (prost's Option accessor default, the
// exact behavior of split.timestamp_start()/timestamp_end() in the reference).
// one split = one or more segments (collector.rs:475-594 per-segment
// collection): run the per-segment search and merge within the split with
// the exact (sort_value, sort_value2, split, segment_ord, doc) order
static SplitResult search_device_split(qw_ctx* ctx, const DeviceSplit& ds,
                                       const pb::SearchRequest& req,
                                       const Schema& schema) {
    using namespace qw;
    std::vector<const DeviceSplit*> segs;
    if (ds.segs.empty()) segs.push_back(&ds);
    else
        for (auto& sp : ds.segs) segs.push_back(sp.get());
    SplitResult out;
    std::map<std::pair<std::string, std::string>, size_t> absent;
    for (const DeviceSplit* seg : segs) {
        SplitResult r = search_split_gpu(ctx, *seg, req, schema);
        out.num_hits += r.num_hits;
        for (auto& h : r.hits) out.hits.push_back(std::move(h));
        if (r.has_aggs) {
            if (!out.has_aggs) {
                out.aggs = std::move(r.aggs);
                out.has_aggs = true;
            } else out.aggs.merge(r.aggs);
        }
        out.micros += r.micros;
        for (auto& ab : r.absent_required) absent[ab]++;
    }
    // absence cache: only terms absent from EVERY segment of the split
    // qualify (the reference early-aborts when warmup finds the term in no
    // segment, leaf.rs:315-320)
    const std::string& sid = segs[0]->view.split_id;
    for (auto& kv : absent)
        if (kv.second == segs.size() &&
            ctx->absence.keys.size() < ctx->absence.cap)
            ctx->absence.keys.insert(
                AbsenceCache::key(sid, kv.first.first, kv.first.second));
    uint64_t leaf_max = req.max_hits + req.start_offset;
    if (segs.size() > 1 && leaf_max > 0 && !out.hits.empty()) {
        int order1 = req.sort_fields.empty() ? 1 : req.sort_fields[0].sort_order;
        int order2 =
            req.sort_fields.size() > 1 ? req.sort_fields[1].sort_order : 1;
        size_t k = std::min<uint64_t>(leaf_max, out.hits.size());
        std::partial_sort(out.hits.begin(), out.hits.begin() + k,
                          out.hits.end(),
                          [&](const pb::PartialHit& a, const pb::PartialHit& b) {
                              return hit_before(a, b, order1, order2);
                          });
        out.hits.resize(k);
    }
    return out;
}

static int64_t split_ts_start(const pb::SplitIdAndFooterOffsets& s) {
    return s.timestamp_start ? *s.timestamp_start : 0;
}
static int64_t split_ts_end(const pb::SplitIdAndFooterOffsets& s) {
    return s.timestamp_end ? *s.timestamp_end : 0;
}

static int64_t div_ceil_i64(int64_t lhs, int64_t rhs) {  // quickwit-common lib.rs:210
    int64_t d = lhs / rhs, r = lhs % rhs;
    return ((r > 0 && rhs > 0) || (r < 0 && rhs < 0)) ? d + 1 : d;
}

struct SplitFilter {
    enum Kind { UNINFORMATIVE, SPLIT_ID_HIGHER, TS_HIGHER, TS_LOWER } kind =
        UNINFORMATIVE;
    bool has = false;     // a worst-of-top-K hit has been recorded
    std::string wid;      // SPLIT_ID_HIGHER: worst split id
    int64_t wts = 0;      // TS_*: worst timestamp in seconds

    static SplitFilter from_request(const pb::SearchRequest& req,
                                    const std::string& ts_field) {
        SplitFilter f;
        if (req.sort_fields.empty()) {
            f.kind = SPLIT_ID_HIGHER;
        } else if (!ts_field.empty() && req.sort_fields[0].field_name == ts_field) {
            f.kind = req.sort_fields[0].sort_order == 1 ? TS_HIGHER : TS_LOWER;
        }
        return f;
    }

    bool can_be_better(const pb::SplitIdAndFooterOffsets& s) const {
        if (!has) return true;
        switch (kind) {
            case SPLIT_ID_HIGHER: return s.split_id >= wid;
            case TS_HIGHER: return split_ts_end(s) >= wts;
            case TS_LOWER: return split_ts_start(s) <= wts;
            default: return true;
        }
    }

    // record the worst of the (now full) top-K (leaf.rs:1516-1542)
    void record_new_worst_hit(const pb::PartialHit& h) {
        switch (kind) {
            case SPLIT_ID_HIGHER:
                has = true;
                wid = h.split_id;
                break;
            case TS_HIGHER:
                if (h.sort_value.kind == pb::SortByValue::I64) {
                    has = true;
                    wts = div_ceil_i64(h.sort_value.i64, 1000000000);  // round UP
                }
                break;
            case TS_LOWER:
                if (h.sort_value.kind == pb::SortByValue::I64) {
                    has = true;
                    wts = h.sort_value.i64 / 1000000000;  // truncate (Rust /)
                }
                break;
            default: break;
        }
    }
};

static bool is_simple_all_query(const pb::SearchRequest& r) {  // leaf.rs:1315
    if (r.aggregation_request) return false;
    if (r.search_after) return false;
    if (r.start_timestamp || r.end_timestamp) return false;
    try {
        mj::ValuePtr v = mj::parse(r.query_ast);
        const mj::Value* t = v->get("type");
        return t && t->kind == mj::Value::STR && t->s == "match_all";
    } catch (...) {
        return false;
    }
}

static void disable_search_request_hits(pb::SearchRequest& r) {  // leaf.rs:1705
    r.max_hits = 0;
    r.start_offset = 0;
    r.sort_fields.clear();
    r.search_after.reset();
}

}  // namespace qw

// ============================================================== C ABI
static thread_local std::string g_tls_error;

static void set_err(qw_ctx* ctx, const std::string& msg) {
    if (ctx) ctx->last_error = msg;
    else g_tls_error = msg;
}

static void fill_buf(qw_buf* out, const std::string& s) {
    out->data = (uint8_t*)malloc(s.size() ? s.size() : 1);
    memcpy(out->data, s.data(), s.size());
    out->len = s.size();
}

extern "C" {

const char* qw_version(void) { return "quickwit_amd 0.1 gfx950"; }

qw_ctx* qw_ctx_create(const char* config_json) {
    auto* ctx = new qw_ctx();
    try {
        if (config_json && *config_json) {
            mj::ValuePtr cfg = mj::parse(config_json);
            if (const mj::Value* d = cfg->get("device")) ctx->device = int(d->as_i64());
            if (const mj::Value* b = cfg->get("aggregation_bucket_limit"))
                ctx->agg_bucket_limit = b->as_i64();
            if (const mj::Value* c = cfg->get("partial_request_cache_capacity"))
                ctx->leaf_cache.capacity = size_t(c->as_i64());
            if (const mj::Value* mb = cfg->get("hbm_memory_budget")) {
                ctx->hbm_budget = uint64_t(mb->as_i64());
                ctx->budget_configured = true;
            }
        }
        return ctx;
    } catch (const std::exception& e) {
        g_tls_error = e.what();
        delete ctx;
        return nullptr;
    }
}

void qw_ctx_free(qw_ctx* ctx) {
    if (!ctx) return;
    if (ctx->device_ready) {
        (void)hipSetDevice(ctx->device);
        for (auto& kv : ctx->splits) {
            kv.second->free_range_cache();
            for (auto& sg : kv.second->segs) sg->free_range_cache();
            if (kv.second->d_image) (void)hipFree(kv.second->d_image);
        }
        ctx->hitsets.clear();
        if (ctx->ev_start) (void)hipEventDestroy(ctx->ev_start);
        if (ctx->ev_stop) (void)hipEventDestroy(ctx->ev_stop);
        if (ctx->stream) (void)hipStreamDestroy(ctx->stream);
    }
    delete ctx;
}

const char* qw_last_error(const qw_ctx* ctx) {
    return ctx ? ctx->last_error.c_str() : g_tls_error.c_str();
}

int32_t qw_ctx_add_split(qw_ctx* ctx, const char* split_id, const uint8_t* data,
                         size_t len) {
    using namespace qw;
    try {
        ctx_ensure_device(ctx);
    } catch (const std::exception& e) {
        set_err(ctx, e.what());
        return QW_ERR_NO_GPU;
    }
    try {
        auto ds = std::make_unique<DeviceSplit>();
        ds->host.assign(data, data + len);
        if (len >= 88 && memcmp(data, "QWAMDSP2", 8) == 0) {
            // QWA2 multi-segment container: outer meta lists per-segment
            // QWA1 images; children alias the parent host/device image
            if (memcmp(data + len - 8, "QWA1FOOT", 8) != 0)
                throw std::runtime_error("bad QWA2 footer");
            uint64_t moff, mlen;
            memcpy(&moff, data + len - 24, 8);
            memcpy(&mlen, data + len - 16, 8);
            if (moff + mlen > len) throw std::runtime_error("bad QWA2 footer");
            mj::ValuePtr meta = mj::parse((const char*)data + moff, mlen);
            uint32_t ord = 0;
            for (auto& sgv : meta->at("segments")->arr) {
                uint64_t off = uint64_t(sgv->at("off")->as_i64());
                uint64_t slen = uint64_t(sgv->at("len")->as_i64());
                if (off + slen > len)
                    throw std::runtime_error("QWA2: segment out of bounds");
                auto child = std::make_unique<DeviceSplit>();
                child->owns_image = false;
                child->view.parse(ds->host.data() + off, slen);
                if (child->view.version < 2)
                    throw std::runtime_error("QWA2 segment: v1 container");
                child->view.split_id = split_id;
                child->view.segment_ord = ord++;
                child->len = slen;
                child->total_docs = child->view.num_docs;
                ds->total_docs += child->view.num_docs;
                ds->segs.push_back(std::move(child));
            }
            if (ds->segs.empty())
                throw std::runtime_error("QWA2: no segments");
            ds->view = ds->segs[0]->view;  // pointers alias ds->host
        } else {
            ds->view.parse(ds->host.data(), len);
            // the caller's split id is authoritative (the request names
            // splits by SplitIdAndFooterOffsets.split_id and the response
            // must echo it); the container's own id is only a default
            ds->view.split_id = split_id;
            if (ds->view.version < 2)
                throw std::runtime_error(
                    "QWA1 v1 container lacks posting segment anchors "
                    "(regenerate the split with the current writer)");
            ds->total_docs = ds->view.num_docs;
        }
        ds->len = len;
        HIP_CHECK(hipSetDevice(ctx->device));
        try {
            qw_check_budget(ctx, len + 64);
        } catch (const std::exception& e) {
            set_err(ctx, e.what());
            return QW_ERR_OVER_MEMORY_BUDGET;
        }
        HIP_CHECK(hipMalloc(&ds->d_image, len + 64));  // +64: decode overread pad
        HIP_CHECK(hipMemcpy(ds->d_image, data, len, hipMemcpyHostToDevice));
        ctx->hbm_used += len + 64;
        {
            // children point into the parent device image (their views'
            // offsets are relative to their own segment base)
            size_t ci = 0;
            uint64_t moff2, mlen2;
            if (!ds->segs.empty()) {
                mj::ValuePtr meta2;
                memcpy(&moff2, data + len - 24, 8);
                memcpy(&mlen2, data + len - 16, 8);
                meta2 = mj::parse((const char*)data + moff2, mlen2);
                for (auto& sgv : meta2->at("segments")->arr)
                    ds->segs[ci++]->d_image =
                        ds->d_image + uint64_t(sgv->at("off")->as_i64());
            }
        }
        ctx->splits[split_id] = std::move(ds);
        return QW_OK;
    } catch (const std::exception& e) {
        set_err(ctx, e.what());
        return QW_ERR_BAD_SPLIT;
    }
}

int32_t qw_ctx_remove_split(qw_ctx* ctx, const char* split_id) {
    auto it = ctx->splits.find(split_id);
    if (it == ctx->splits.end()) return QW_ERR_NOT_FOUND;
    ctx->hbm_used -= it->second->range_cache_bytes;
    it->second->free_range_cache();
    for (auto& sg : it->second->segs) {
        ctx->hbm_used -= sg->range_cache_bytes;
        sg->free_range_cache();
    }
    if (it->second->d_image) {
        (void)hipFree(it->second->d_image);
        ctx->hbm_used -= it->second->len + 64;
    }
    ctx->splits.erase(it);
    ctx->leaf_cache.remove_split(split_id);
    ctx->hitsets.remove_split(split_id);
    ctx->absence.remove_split(split_id);
    return QW_OK;
}

int32_t qw_leaf_search(qw_ctx* ctx, const uint8_t* req_pb, size_t req_len,
                       qw_buf* out) {
    using namespace qw;
    try {
        ctx_ensure_device(ctx);
    } catch (const std::exception& e) {
        set_err(ctx, e.what());
        return QW_ERR_NO_GPU;
    }
    try {
        HIP_CHECK(hipSetDevice(ctx->device));
        pb::LeafSearchRequest lreq = pb::LeafSearchRequest::decode(req_pb, req_len);
        const pb::SearchRequest& req = lreq.search_request;
        if (lreq.doc_mappers.empty()) throw std::runtime_error("missing doc_mapper");
        Schema schema = Schema::parse(lreq.doc_mappers[0]);

        struct Task {
            pb::SplitIdAndFooterOffsets so;
            const DeviceSplit* ds;
            pb::SearchRequest req;  // per-split request (may be demoted)
        };
        std::vector<Task> tasks;
        std::vector<pb::SplitSearchError> pre_failed;
        for (auto& lr : lreq.leaf_requests)
            for (auto& so : lr.split_offsets) {
                auto it = ctx->splits.find(so.split_id);
                if (it == ctx->splits.end()) {
                    // a split that cannot be opened is a per-split failure
                    // inside the response (leaf.rs:2143-2148), not a
                    // whole-call error — the other splits still answer
                    pb::SplitSearchError se;
                    se.error = "unknown split: " + so.split_id;
                    se.split_id = so.split_id;
                    se.retryable_error = true;
                    pre_failed.push_back(std::move(se));
                    continue;
                }
                tasks.push_back({so, it->second.get(), req});
            }

        // ---- pruning: split ordering + upfront count-only demotion
        // (CanSplitDoBetter::optimize, leaf.rs:1403-1514)
        SplitFilter filter = SplitFilter::from_request(req, schema.timestamp_field);
        switch (filter.kind) {
            case SplitFilter::SPLIT_ID_HIGHER:
                std::sort(tasks.begin(), tasks.end(), [](const Task& a, const Task& b) {
                    return b.so.split_id < a.so.split_id;
                });
                break;
            case SplitFilter::TS_HIGHER:
                std::sort(tasks.begin(), tasks.end(), [](const Task& a, const Task& b) {
                    return split_ts_end(b.so) < split_ts_end(a.so);
                });
                break;
            case SplitFilter::TS_LOWER:
                std::sort(tasks.begin(), tasks.end(), [](const Task& a, const Task& b) {
                    return split_ts_start(a.so) < split_ts_start(b.so);
                });
                break;
            default: break;
        }
        if (is_simple_all_query(req)) {
            uint64_t num_requested = req.start_offset + req.max_hits;
            size_t mrs = 0;  // number of splits guaranteed to fill the top-K
            uint64_t psum = 0;
            for (const Task& t : tasks) {
                psum += t.so.num_docs;
                if (psum >= num_requested) break;
                ++mrs;
            }
            mrs += 1;
            if (filter.kind == SplitFilter::SPLIT_ID_HIGHER) {
                for (size_t i = mrs; i < tasks.size(); ++i)
                    disable_search_request_hits(tasks[i].req);
            } else if (filter.kind == SplitFilter::TS_LOWER) {
                int64_t biggest_end = INT64_MIN;
                for (size_t i = 0; i < std::min(mrs, tasks.size()); ++i)
                    biggest_end = std::max(biggest_end, split_ts_end(tasks[i].so));
                for (size_t i = mrs; i < tasks.size(); ++i)
                    if (split_ts_start(tasks[i].so) > biggest_end)
                        disable_search_request_hits(tasks[i].req);
            } else if (filter.kind == SplitFilter::TS_HIGHER) {
                int64_t smallest_start = INT64_MAX;
                for (size_t i = 0; i < std::min(mrs, tasks.size()); ++i)
                    smallest_start = std::min(smallest_start, split_ts_start(tasks[i].so));
                for (size_t i = mrs; i < tasks.size(); ++i)
                    if (split_ts_end(tasks[i].so) < smallest_start)
                        disable_search_request_hits(tasks[i].req);
            }
        }
        // per-split rewrite (leaf.rs:977-990): drop sort fields on count-only
        // requests; drop request-level ts bounds covered by the split range
        for (Task& t : tasks) {
            if (t.req.max_hits == 0) t.req.sort_fields.clear();
            if (!schema.timestamp_field.empty()) {
                if (t.req.start_timestamp && t.so.timestamp_start &&
                    *t.req.start_timestamp <= *t.so.timestamp_start)
                    t.req.start_timestamp.reset();
                if (t.req.end_timestamp && t.so.timestamp_end &&
                    *t.req.end_timestamp >= *t.so.timestamp_end + 1)
                    t.req.end_timestamp.reset();
            }
        }

        pb::LeafSearchResponse resp;
        for (auto& se : pre_failed) {
            resp.num_attempted_splits++;
            resp.failed_splits.push_back(std::move(se));
        }
        IntermediateAggResults merged_aggs;
        bool any_aggs = false;
        std::vector<pb::PartialHit> all_hits;
        pb::LeafResourceStats rstats;
        rstats.search_pool_cpu_threads = 1;  // one host thread per GPU
        uint64_t worst = 0;
        int order1 = req.sort_fields.empty() ? 1 : req.sort_fields[0].sort_order;
        int order2 = req.sort_fields.size() > 1 ? req.sort_fields[1].sort_order : 1;
        auto cmp = [&](const pb::PartialHit& a, const pb::PartialHit& b) {
            return hit_before(a, b, order1, order2);
        };
        uint64_t leaf_max = req.max_hits + req.start_offset;
        for (Task& t : tasks) {
            // second simplify pass against the live filter state
            // (simplify_search_request, leaf.rs:1666-1703)
            if (!filter.can_be_better(t.so)) disable_search_request_hits(t.req);
            if (t.req.max_hits == 0 && !t.req.aggregation_request &&
                t.req.count_hits != 0 /* != CountHits::CountAll */)
                continue;  // pruned before warmup — not attempted
            resp.num_attempted_splits++;
            // partial-result cache probe (leaf.rs:667 / leaf_cache.rs)
            std::string ckey;
            bool cacheable = ctx->leaf_cache.capacity > 0;
            if (cacheable) {
                ckey = leaf_cache_key(t.so, t.req);
                std::string hit;
                if (ctx->leaf_cache.get(ckey, &hit)) {
                    pb::LeafSearchResponse cr = pb::LeafSearchResponse::decode(
                        (const uint8_t*)hit.data(), hit.size());
                    resp.num_successful_splits++;
                    resp.num_hits += cr.num_hits;
                    for (auto& h : cr.partial_hits) all_hits.push_back(std::move(h));
                    if (cr.intermediate_aggregation_result) {
                        IntermediateAggResults ir = IntermediateAggResults::decode(
                            (const uint8_t*)cr.intermediate_aggregation_result->data(),
                            cr.intermediate_aggregation_result->size());
                        if (!any_aggs) {
                            merged_aggs = std::move(ir);
                            any_aggs = true;
                        } else merged_aggs.merge(ir);
                    }
                    // cache-hit stats (leaf_cache.rs:73-79 overwrite-on-put)
                    rstats.partial_result_cache_num_splits += 1;
                    rstats.partial_result_cache_num_docs += t.so.num_docs;
                    if (leaf_max > 0 && all_hits.size() >= leaf_max) {
                        size_t k = std::min<size_t>(leaf_max, all_hits.size());
                        std::partial_sort(all_hits.begin(), all_hits.begin() + k,
                                          all_hits.end(), cmp);
                        all_hits.resize(k);
                        if (all_hits.size() == leaf_max)
                            filter.record_new_worst_hit(all_hits.back());
                    }
                    continue;
                }
            }
            SplitResult r;
            try {
                r = search_device_split(ctx, *t.ds, t.req, schema);
            } catch (const std::exception& e) {
                // per-split failure is data, not an exception (leaf.rs:2143)
                pb::SplitSearchError se;
                se.error = e.what();
                se.split_id = t.so.split_id;
                se.retryable_error = true;
                resp.failed_splits.push_back(std::move(se));
                continue;
            }
            resp.num_successful_splits++;
            resp.num_hits += r.num_hits;
            if (cacheable) {  // cache fill on success (leaf.rs:966)
                pb::LeafSearchResponse cv;
                cv.num_hits = r.num_hits;
                cv.partial_hits = r.hits;
                if (r.has_aggs) cv.intermediate_aggregation_result = r.aggs.encode();
                ctx->leaf_cache.put(ckey, cv.encode());
            }
            for (auto& h : r.hits) all_hits.push_back(std::move(h));
            if (r.has_aggs) {
                if (!any_aggs) {
                    merged_aggs = std::move(r.aggs);
                    any_aggs = true;
                } else merged_aggs.merge(r.aggs);
            }
            // incremental top-K + worst-hit feedback (leaf.rs:2310-2317)
            if (leaf_max > 0 && all_hits.size() >= leaf_max) {
                size_t k = std::min<size_t>(leaf_max, all_hits.size());
                std::partial_sort(all_hits.begin(), all_hits.begin() + k,
                                  all_hits.end(), cmp);
                all_hits.resize(k);
                if (all_hits.size() == leaf_max)
                    filter.record_new_worst_hit(all_hits.back());
            }
            pb::SplitResourceStats ss;
            ss.split_num_docs = t.ds->total_docs;
            ss.matched_num_docs = r.num_hits;
            ss.cpu_search_microsecs = r.micros;  // GPU wall (cpu_search analog)
            rstats.localexec_num_splits++;
            rstats.localexec_num_docs += ss.split_num_docs;
            rstats.split_resources_sum.split_num_docs += ss.split_num_docs;
            rstats.split_resources_sum.matched_num_docs += ss.matched_num_docs;
            rstats.split_resources_sum.cpu_search_microsecs += ss.cpu_search_microsecs;
            if (ss.cpu_search_microsecs >= worst) {
                worst = ss.cpu_search_microsecs;
                rstats.split_resources_worst = ss;
            }
        }
        size_t k = std::min<size_t>(leaf_max, all_hits.size());
        std::partial_sort(all_hits.begin(), all_hits.begin() + k, all_hits.end(), cmp);
        all_hits.resize(k);
        resp.partial_hits = std::move(all_hits);
        if (any_aggs) resp.intermediate_aggregation_result = merged_aggs.encode();
        resp.resource_stats = rstats;
        fill_buf(out, resp.encode());
        return QW_OK;
    } catch (const std::exception& e) {
        set_err(ctx, e.what());
        std::string msg = e.what();
        if (msg.find("not supported") != std::string::npos ||
            msg.find("not flattenable") != std::string::npos)
            return QW_ERR_INVALID_QUERY;
        return QW_ERR_INTERNAL;
    }
}

int32_t qw_fetch_docs(qw_ctx* ctx, const uint8_t* req_pb, size_t req_len,
                      qw_buf* out) {
    using namespace qw;
    try {
        pb::FetchDocsRequest freq = pb::FetchDocsRequest::decode(req_pb, req_len);
        // group hits by split (fetch_docs.rs:56-60 sorts GlobalDocAddress);
        // one block decompression serves consecutive docs of the same block
        std::vector<size_t> order(freq.partial_hits.size());
        for (size_t i = 0; i < order.size(); ++i) order[i] = i;
        std::sort(order.begin(), order.end(), [&](size_t a, size_t b) {
            const pb::PartialHit& ha = freq.partial_hits[a];
            const pb::PartialHit& hb = freq.partial_hits[b];
            if (ha.split_id != hb.split_id) return ha.split_id < hb.split_id;
            if (ha.segment_ord != hb.segment_ord)
                return ha.segment_ord < hb.segment_ord;
            return ha.doc_id < hb.doc_id;
        });
        pb::FetchDocsResponse resp;
        std::string cur_split;
        uint32_t cur_seg = ~0u;
        const DeviceSplit* ds = nullptr;
        const SplitView* svp = nullptr;
        std::vector<uint8_t> block;       // decompressed block cache
        int64_t cached_block = -1;
        for (size_t oi : order) {
            const pb::PartialHit& h = freq.partial_hits[oi];
            if (h.split_id != cur_split || h.segment_ord != cur_seg) {
                auto it = ctx->splits.find(h.split_id);
                if (it == ctx->splits.end())
                    throw std::runtime_error("unknown split: " + h.split_id);
                ds = it->second.get();
                cur_split = h.split_id;
                cur_seg = h.segment_ord;
                if (ds->segs.empty()) {
                    if (h.segment_ord != 0)
                        throw std::runtime_error("bad segment ord");
                    svp = &ds->view;
                } else {
                    if (h.segment_ord >= ds->segs.size())
                        throw std::runtime_error("bad segment ord");
                    svp = &ds->segs[h.segment_ord]->view;
                }
                cached_block = -1;
            }
            const DocStoreView& d = svp->docstore;
            if (!d.present)
                throw std::runtime_error("split has no docstore: " + h.split_id);
            if (h.doc_id >= svp->num_docs)
                throw std::runtime_error("doc id out of range");
            // binary search the block whose [first, next_first) covers doc
            uint32_t lo = 0, hi = d.n_blocks;
            while (lo + 1 < hi) {
                uint32_t mid = (lo + hi) / 2;
                if (d.firsts[mid] <= h.doc_id) lo = mid;
                else hi = mid;
            }
            if (int64_t(lo) != cached_block) {
                uint32_t nb = d.firsts[lo + 1] - d.firsts[lo];
                // uncompressed size upper bound: grow until inflate fits
                uLongf dst_len = uLongf(
                    std::max<uint64_t>(64 * 1024, (d.offs[lo + 1] - d.offs[lo]) * 8));
                for (;;) {
                    block.resize(dst_len);
                    uLongf got = dst_len;
                    int rc = uncompress(block.data(), &got, d.blocks + d.offs[lo],
                                        uLong(d.offs[lo + 1] - d.offs[lo]));
                    if (rc == Z_OK) {
                        block.resize(got);
                        break;
                    }
                    if (rc != Z_BUF_ERROR)
                        throw std::runtime_error("docstore block inflate failed");
                    dst_len *= 2;
                }
                if (block.size() < nb * 4)
                    throw std::runtime_error("docstore block too short");
                cached_block = int64_t(lo);
            }
            uint32_t nb = d.firsts[lo + 1] - d.firsts[lo];
            const uint32_t* lens = (const uint32_t*)block.data();
            uint32_t local = h.doc_id - d.firsts[lo];
            uint64_t off = nb * 4ull;
            for (uint32_t i = 0; i < local; ++i) off += lens[i];
            if (off + lens[local] > block.size())
                throw std::runtime_error("docstore block corrupt");
            pb::LeafHit lh;
            lh.leaf_json.assign((const char*)block.data() + off, lens[local]);
            lh.partial_hit = h;
            resp.hits.push_back(std::move(lh));
        }
        fill_buf(out, resp.encode());
        return QW_OK;
    } catch (const std::exception& e) {
        set_err(ctx, e.what());
        return QW_ERR_INTERNAL;
    }
}

int32_t qw_leaf_list_terms(qw_ctx* ctx, const uint8_t* req_pb, size_t req_len,
                           qw_buf* out) {
    using namespace qw;
    try {
        pb::LeafListTermsRequest lreq = pb::LeafListTermsRequest::decode(req_pb, req_len);
        const pb::ListTermsRequest& req = lreq.list_terms_request;
        pb::LeafListTermsResponse resp;
        // per split: sorted dict range scan, limited (list_terms.rs:246-290)
        std::vector<std::vector<std::string>> per_split;
        for (auto& so : lreq.split_offsets) {
            auto it = ctx->splits.find(so.split_id);
            if (it == ctx->splits.end()) {
                pb::SplitSearchError se;
                se.split_id = so.split_id;
                se.error = "unknown split: " + so.split_id;
                se.retryable_error = true;
                resp.failed_splits.push_back(std::move(se));
                continue;
            }
            const DeviceSplit* dsp = it->second.get();
            std::vector<const SplitView*> views;
            if (dsp->segs.empty()) views.push_back(&dsp->view);
            else
                for (auto& sg : dsp->segs) views.push_back(&sg->view);
            const TextFieldView* f0 = views[0]->text_field(req.field);
            if (!f0) {
                pb::SplitSearchError se;
                se.split_id = so.split_id;
                se.error = "couldn't get field named \"" + req.field +
                           "\" from schema to list terms";
                se.retryable_error = false;
                resp.failed_splits.push_back(std::move(se));
                continue;
            }
            resp.num_attempted_splits++;
            // each segment dict is sorted: one run per segment, k-merged +
            // deduped below with the other splits' runs
            for (const SplitView* svv : views) {
                const TextFieldView* f = svv->text_field(req.field);
                if (!f) continue;
                std::vector<std::string> terms;
                for (uint32_t t = 0; t < f->num_terms; ++t) {
                    const char* s = (const char*)f->h_term_bytes + f->h_term_offsets[t];
                    size_t sl = f->h_term_offsets[t + 1] - f->h_term_offsets[t];
                    std::string term(s, sl);
                    if (req.start_key && term < *req.start_key) continue;  // ge
                    if (req.end_key && term >= *req.end_key) break;        // lt
                    terms.push_back(std::move(term));
                    if (req.max_hits && terms.size() >= *req.max_hits) break;
                }
                per_split.push_back(std::move(terms));
            }
        }
        // k-merge + dedup + global limit (list_terms.rs:291-300,392-400)
        std::vector<size_t> pos(per_split.size(), 0);
        while (true) {
            if (req.max_hits && resp.terms.size() >= *req.max_hits) break;
            const std::string* best = nullptr;
            for (size_t i = 0; i < per_split.size(); ++i)
                if (pos[i] < per_split[i].size() &&
                    (!best || per_split[i][pos[i]] < *best))
                    best = &per_split[i][pos[i]];
            if (!best) break;
            std::string v = *best;
            for (size_t i = 0; i < per_split.size(); ++i)
                while (pos[i] < per_split[i].size() && per_split[i][pos[i]] == v)
                    ++pos[i];
            resp.terms.push_back(std::move(v));
        }
        resp.num_hits = resp.terms.size();
        fill_buf(out, resp.encode());
        return QW_OK;
    } catch (const std::exception& e) {
        set_err(ctx, e.what());
        return QW_ERR_INTERNAL;
    }
}

void qw_buf_free(qw_buf* buf) {
    if (!buf) return;
    free(buf->data);
    buf->data = nullptr;
    buf->len = 0;
}

int32_t qw_merge_leaf_responses(const uint8_t* search_request_pb,
                                size_t search_request_len, const uint8_t** response_pbs,
                                const size_t* response_lens, size_t n,
                                qw_buf* merged_out) {
    using namespace qw;
    try {
        pb::SearchRequest req = pb::SearchRequest::decode(
            pb::Reader(search_request_pb, search_request_len));
        pb::LeafSearchResponse m;
        IntermediateAggResults aggs;
        bool any_aggs = false;
        std::vector<pb::PartialHit> all_hits;
        for (size_t i = 0; i < n; ++i) {
            pb::LeafSearchResponse r =
                pb::LeafSearchResponse::decode(response_pbs[i], response_lens[i]);
            m.num_hits += r.num_hits;
            m.num_attempted_splits += r.num_attempted_splits;
            m.num_successful_splits += r.num_successful_splits;
            for (auto& h : r.partial_hits) all_hits.push_back(std::move(h));
            for (auto& f : r.failed_splits) m.failed_splits.push_back(std::move(f));
            if (r.intermediate_aggregation_result) {
                IntermediateAggResults ir = IntermediateAggResults::decode(
                    (const uint8_t*)r.intermediate_aggregation_result->data(),
                    r.intermediate_aggregation_result->size());
                if (!any_aggs) {
                    aggs = std::move(ir);
                    any_aggs = true;
                } else aggs.merge(ir);
            }
            if (r.resource_stats) {
                if (!m.resource_stats) m.resource_stats = pb::LeafResourceStats{};
                m.resource_stats->partial_result_cache_num_splits +=
                    r.resource_stats->partial_result_cache_num_splits;
                m.resource_stats->partial_result_cache_num_docs +=
                    r.resource_stats->partial_result_cache_num_docs;
                m.resource_stats->localexec_num_splits +=
                    r.resource_stats->localexec_num_splits;
                m.resource_stats->localexec_num_docs +=
                    r.resource_stats->localexec_num_docs;
                m.resource_stats->split_resources_sum.cpu_search_microsecs +=
                    r.resource_stats->split_resources_sum.cpu_search_microsecs;
                m.resource_stats->split_resources_sum.split_num_docs +=
                    r.resource_stats->split_resources_sum.split_num_docs;
                m.resource_stats->split_resources_sum.matched_num_docs +=
                    r.resource_stats->split_resources_sum.matched_num_docs;
            }
        }
        int order1 = req.sort_fields.empty() ? 1 : req.sort_fields[0].sort_order;
        int order2 = req.sort_fields.size() > 1 ? req.sort_fields[1].sort_order : 1;
        auto cmp = [&](const pb::PartialHit& a, const pb::PartialHit& b) {
            return hit_before(a, b, order1, order2);
        };
        size_t k = std::min<size_t>(req.max_hits + req.start_offset, all_hits.size());
        std::partial_sort(all_hits.begin(), all_hits.begin() + k, all_hits.end(), cmp);
        all_hits.resize(k);
        m.partial_hits = std::move(all_hits);
        if (any_aggs) m.intermediate_aggregation_result = aggs.encode();
        fill_buf(merged_out, m.encode());
        return QW_OK;
    } catch (const std::exception& e) {
        set_err(nullptr, e.what());
        return QW_ERR_INTERNAL;
    }
}

int32_t qw_finalize_agg_to_json(const uint8_t* blob, size_t len,
                                const char* agg_request_json, qw_buf* json_out) {
    using namespace qw;
    try {
        IntermediateAggResults ir = IntermediateAggResults::decode(blob, len);
        std::vector<AggDef> defs = parse_agg_request(agg_request_json);
        fill_buf(json_out, finalize_aggs_json(ir, defs));
        return QW_OK;
    } catch (const std::exception& e) {
        set_err(nullptr, e.what());
        return QW_ERR_INTERNAL;
    }
}

int32_t qw_kernel_stats(qw_ctx* ctx, const char* kernel_name, double* total_ms,
                        uint64_t* launches) {
    auto it = ctx->timers.find(kernel_name);
    if (it == ctx->timers.end()) {
        static const char* known[] = {"union_bm25", "range_filter", "column_agg",
                                      "topk_select"};
        bool ok = false;
        for (const char* k : known) ok |= kernel_name == std::string(k);
        if (!ok) return QW_ERR_NOT_FOUND;
        *total_ms = 0;
        *launches = 0;
        return QW_OK;
    }
    *total_ms = it->second.total_ms;
    *launches = it->second.launches;
    return QW_OK;
}

void qw_kernel_stats_reset(qw_ctx* ctx) { ctx->timers.clear(); }

int32_t qw_ctx_memory_stats(qw_ctx* ctx, uint64_t* used_bytes,
                            uint64_t* budget_bytes, uint64_t* splits_bytes) {
    if (!ctx) return QW_ERR_INVALID_ARGUMENT;
    if (used_bytes) *used_bytes = ctx->hbm_used + ctx->hitsets.bytes;
    if (budget_bytes) *budget_bytes = ctx->hbm_budget;
    if (splits_bytes) {
        uint64_t sb = 0;
        for (auto& kv : ctx->splits) sb += kv.second->len + 64;
        *splits_bytes = sb;
    }
    return QW_OK;
}

int32_t qw_absence_cache_stats(qw_ctx* ctx, uint64_t* hits, uint64_t* misses,
                               uint64_t* entries) {
    if (!ctx) return QW_ERR_INVALID_ARGUMENT;
    if (hits) *hits = ctx->absence.hits;
    if (misses) *misses = ctx->absence.misses;
    if (entries) *entries = uint64_t(ctx->absence.keys.size());
    return QW_OK;
}

int32_t qw_ctx_device_sync(qw_ctx* ctx) {
    try {
        ctx_ensure_device(ctx);
        HIP_CHECK(hipDeviceSynchronize());
        return QW_OK;
    } catch (const std::exception& e) {
        set_err(ctx, e.what());
        return QW_ERR_NO_GPU;
    }
}

}  // extern "C"
