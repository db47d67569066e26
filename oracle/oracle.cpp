// ============================================================================
// ORACLE — TEST INFRASTRUCTURE ONLY.
//
// CPU restatement of Quickwit's per-split leaf-search semantics
// (quickwit/quickwit-search/src/leaf.rs:655-970 and the tantivy 0.27 @
// 86641f7 execution it delegates to — tantivy is a non-vendored dependency,
// so its algorithms are restated from its published behavior and pinned by
// the reference's own golden vectors, tests/golden/*).
//
// Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
// call into this library. The product path (libquickwit_amd.so) never routes
// through it and fails loudly without a GPU.
//
// What is restated where:
//  - posting decode: our QWA1 blocks, scalar loop here;
//  - boolean combination: tantivy BooleanQuery/union/intersection semantics
//    (tantivy_query_ast.rs:153-374 for construction; minimum_should_match
//    default: >=1 should when no must/filter clause);
//  - BM25: Lucene BM25 with the (k1+1) factor, k1=1.2 b=0.75,
//    idf = ln(1+(N-df+0.5)/(df+0.5)), fieldnorm-quantized |d| — pinned by
//    tests.rs:600-691 golden scores (tests/golden/bm25_sort.json);
//  - top-K + tie-breaks: top_k_collector.rs + docs/internals/sorting.md:14-26
//    ((sort_value, sort_value2), None last, GlobalDocId tie in order1);
//  - aggregations: tantivy aggregation semantics pinned by
//    rest-api-tests/scenarii/aggregations (tests/golden/aggregations.json);
//  - per-request orchestration: leaf_search_single_split → one result per
//    split, merged with IncrementalCollector semantics (collector.rs:1198).
//
// OpenMP parallelism is across splits only (one thread per split), mirroring
// the reference's rayon pool with single-threaded per-split closures
// (quickwit-search/src/lib.rs:100-104). This parallel variant is the timed
// CPU baseline of bench.py (cpu_baseline.kind = "port").
// ============================================================================
#include <omp.h>

#include <algorithm>
#include <chrono>
#include <cmath>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

#include "../quickwit_amd/csrc/fieldnorm.h"
#include "../quickwit_amd/csrc/minijson.h"
#include "../quickwit_amd/csrc/pb.h"
#include "../quickwit_amd/csrc/qagg_format.h"
#include "../quickwit_amd/csrc/qast.h"
#include "../quickwit_amd/csrc/qsplit.h"
#include "../quickwit_amd/csrc/sortkey.h"

namespace qw {
namespace oracle {

// --------------------------------------------------------------- decode
// Scalar bit-unpack of one posting block (QWA1).
static void decode_block(const uint32_t* payload, const SkipEntry& e, uint32_t* docs,
                         uint32_t* tfs) {
    const uint32_t* base = payload + e.word_off;
    uint32_t w = e.id_bits;
    uint64_t mask = w >= 64 ? ~0ULL : ((1ULL << w) - 1);
    uint32_t doc = e.first_doc;
    for (uint32_t j = 0; j < e.count; ++j) {
        uint64_t bitpos = uint64_t(j) * w;
        uint64_t word = bitpos >> 5;
        uint32_t sh = uint32_t(bitpos & 31);
        uint64_t v = (uint64_t(base[word]) | (uint64_t(base[word + 1]) << 32)) >> sh;
        if (sh + w > 64) v |= uint64_t(base[word + 2]) << (64 - sh);
        doc += uint32_t(v & mask);
        docs[j] = doc;
    }
    if (e.tf_bits == 0) {
        for (uint32_t j = 0; j < e.count; ++j) tfs[j] = 1;
        return;
    }
    const uint32_t* tbase = base + 2 * ((128 * uint32_t(e.id_bits) + 63) / 64);
    w = e.tf_bits;
    mask = (1ULL << w) - 1;
    for (uint32_t j = 0; j < e.count; ++j) {
        uint64_t bitpos = uint64_t(j) * w;
        uint64_t word = bitpos >> 5;
        uint32_t sh = uint32_t(bitpos & 31);
        uint64_t v = (uint64_t(tbase[word]) | (uint64_t(tbase[word + 1]) << 32)) >> sh;
        if (sh + w > 64) v |= uint64_t(tbase[word + 2]) << (64 - sh);
        tfs[j] = uint32_t(v & mask) + 1;
    }
}

struct Postings {
    std::vector<uint32_t> docs;
    std::vector<uint32_t> tfs;
};

static Postings decode_term(const TextFieldView& f, int64_t tid) {
    Postings p;
    uint32_t df = f.h_doc_freq[tid];
    uint32_t nblk = f.h_n_blocks[tid];
    p.docs.resize(df);
    p.tfs.resize(df);
    const SkipEntry* skip = f.h_skip + f.h_skip_off[tid] / 16;
    uint32_t out = 0;
    for (uint32_t b = 0; b < nblk; ++b) {
        decode_block(f.h_payload, skip[b], p.docs.data() + out, p.tfs.data() + out);
        out += skip[b].count;
    }
    return p;
}

// --------------------------------------------------------------- match algebra
struct Match {
    bool all = false;  // matches every doc (scores empty)
    std::vector<uint32_t> docs;
    std::vector<float> scores;  // empty when not scored
    size_t size(uint32_t num_docs) const { return all ? num_docs : docs.size(); }
};

static Match intersect(const Match& a, const Match& b) {
    if (a.all) return b;
    if (b.all) return a;
    Match r;
    bool sa = !a.scores.empty(), sb = !b.scores.empty();
    if (sa || sb) r.scores.reserve(std::min(a.docs.size(), b.docs.size()));
    r.docs.reserve(std::min(a.docs.size(), b.docs.size()));
    size_t i = 0, j = 0;
    while (i < a.docs.size() && j < b.docs.size()) {
        if (a.docs[i] < b.docs[j]) ++i;
        else if (b.docs[j] < a.docs[i]) ++j;
        else {
            r.docs.push_back(a.docs[i]);
            if (sa || sb)
                r.scores.push_back((sa ? a.scores[i] : 0.f) + (sb ? b.scores[j] : 0.f));
            ++i;
            ++j;
        }
    }
    return r;
}

static Match subtract(Match a, const Match& b, uint32_t num_docs) {
    if (b.all) return Match{};
    if (b.docs.empty()) return a;
    if (a.all) {
        Match r;
        r.docs.reserve(num_docs - b.docs.size());
        size_t j = 0;
        for (uint32_t d = 0; d < num_docs; ++d) {
            while (j < b.docs.size() && b.docs[j] < d) ++j;
            if (j < b.docs.size() && b.docs[j] == d) continue;
            r.docs.push_back(d);
        }
        return r;
    }
    Match r;
    r.docs.reserve(a.docs.size());
    bool sa = !a.scores.empty();
    if (sa) r.scores.reserve(a.docs.size());
    size_t j = 0;
    for (size_t i = 0; i < a.docs.size(); ++i) {
        uint32_t d = a.docs[i];
        while (j < b.docs.size() && b.docs[j] < d) ++j;
        if (j < b.docs.size() && b.docs[j] == d) continue;
        r.docs.push_back(d);
        if (sa) r.scores.push_back(a.scores[i]);
    }
    return r;
}

// k-way union; keeps docs matched by >= msm clauses; sums scores of matching
// scoring clauses (tantivy BufferedUnionScorer semantics). match_all
// clauses count toward msm for EVERY doc and contribute no score (our
// restatement scores match_all as 0, like the unscored AllQuery leg).
static Match union_over(const std::vector<const Match*>& ms, size_t msm);

static Match union_n(const std::vector<Match>& ms, size_t msm, uint32_t num_docs) {
    if (msm == 0) msm = 1;
    size_t n_all = 0;
    std::vector<const Match*> rest;
    for (auto& m : ms) {
        if (m.all) ++n_all;
        else rest.push_back(&m);
    }
    if (n_all == 0) return union_over(rest, msm);
    if (n_all >= msm) {
        // every doc matches; keep per-doc scores of the non-all clauses
        bool scored = false;
        for (auto* m : rest) scored |= !m->scores.empty();
        Match r;
        if (!scored) {
            r.all = true;
            return r;
        }
        Match ru = union_over(rest, 1);
        r.docs.resize(num_docs);
        for (uint32_t d = 0; d < num_docs; ++d) r.docs[d] = d;
        r.scores.assign(num_docs, 0.f);
        for (size_t i = 0; i < ru.docs.size(); ++i)
            r.scores[ru.docs[i]] =
                ru.scores.empty() ? 0.f : ru.scores[i];
        return r;
    }
    return union_over(rest, msm - n_all);
}

static Match union_over(const std::vector<const Match*>& msp, size_t msm) {
    Match r;
    if (msm == 0) msm = 1;
    size_t k = msp.size();
    std::vector<size_t> idx(k, 0);
    bool scored = false;
    for (auto* m : msp) scored |= !m->scores.empty();
    while (true) {
        uint32_t best = UINT32_MAX;
        for (size_t c = 0; c < k; ++c) {
            const Match& m = *msp[c];
            if (idx[c] < m.docs.size()) best = std::min(best, m.docs[idx[c]]);
        }
        if (best == UINT32_MAX) break;
        size_t cnt = 0;
        float sc = 0;
        for (size_t c = 0; c < k; ++c) {
            const Match& m = *msp[c];
            if (idx[c] < m.docs.size() && m.docs[idx[c]] == best) {
                ++cnt;
                if (!m.scores.empty()) sc += m.scores[idx[c]];
                ++idx[c];
            }
        }
        if (cnt >= msm) {
            r.docs.push_back(best);
            if (scored) r.scores.push_back(sc);
        }
    }
    return r;
}

// --------------------------------------------------------------- evaluator
struct SplitSearcher {
    const SplitView& sv;
    const Schema& schema;
    bool scoring;

    Match eval(const PlanNode& n) const {
        switch (n.kind) {
            case PlanNode::MATCH_ALL: {
                Match m;
                m.all = true;
                return m;
            }
            case PlanNode::MATCH_NONE:
                return Match{};
            case PlanNode::TERM:
                return eval_term(n);
            case PlanNode::RANGE:
                return eval_range(n);
            case PlanNode::FIELD_PRESENCE:
                return eval_presence(n);
            case PlanNode::BOOL:
                return eval_bool(n);
            case PlanNode::WILDCARD:
                return eval_wildcard(n);
            case PlanNode::CACHE:
                // semantics-transparent (cache_node.rs); memoization is a
                // product-side device-bitmap concern
                return eval(n.cache_inner.at(0));
            case PlanNode::PHRASE:
                return eval_phrase(n);
        }
        return Match{};
    }

    // multi-token phrase, slop 0 (PhraseQuery semantics restated from
    // full_text_query.rs:113-137's phrase mode): a doc matches iff some
    // position p has token i at p+i for all i. Needs record: position.
    // Unscored (const-score like wildcard/term_set — the product rejects
    // phrases under _score sorting the same way).
    Match eval_phrase(const PlanNode& n) const {
        Match m;
        const TextFieldView* f = sv.text_field(n.field);
        if (!f) return m;
        if (!f->has_positions)
            throw std::runtime_error(
                "phrase query needs record: position on field " + n.field);
        size_t k = n.phrase_toks.size();
        struct TP {
            Postings p;
            std::vector<uint32_t> pos_off;  // per posting into h_positions
        };
        std::vector<TP> tps(k);
        for (size_t i = 0; i < k; ++i) {
            const std::string& tok = n.phrase_toks[i];
            int64_t tid = f->find_term(tok.data(), tok.size());
            if (tid < 0) return m;  // absent token: phrase matches nothing
            tps[i].p = decode_term(*f, tid);
            uint32_t df = f->h_doc_freq[tid];
            tps[i].pos_off.resize(df);
            uint32_t first_blk = uint32_t(f->h_skip_off[tid] / 16);
            uint32_t nblk = f->h_n_blocks[tid];
            uint32_t out = 0;
            for (uint32_t b = 0; b < nblk; ++b) {
                uint32_t start = f->h_pos_start[first_blk + b];
                uint32_t cnt = f->h_skip[first_blk + b].count;
                for (uint32_t j = 0; j < cnt; ++j) {
                    tps[i].pos_off[out] = start;
                    start += tps[i].p.tfs[out];
                    ++out;
                }
            }
        }
        // doc intersection driven by token 0, then the position chain
        std::vector<size_t> idx(k, 0);
        const uint32_t* P = f->h_positions;
        for (size_t i0 = 0; i0 < tps[0].p.docs.size(); ++i0) {
            uint32_t doc = tps[0].p.docs[i0];
            bool all = true;
            for (size_t i = 1; i < k && all; ++i) {
                auto& d = tps[i].p.docs;
                while (idx[i] < d.size() && d[idx[i]] < doc) ++idx[i];
                all = idx[i] < d.size() && d[idx[i]] == doc;
            }
            if (!all) continue;
            bool hit = false;
            uint32_t o0 = tps[0].pos_off[i0], n0 = tps[0].p.tfs[i0];
            for (uint32_t a = 0; a < n0 && !hit; ++a) {
                uint32_t p0 = P[o0 + a];
                bool chain = true;
                for (size_t i = 1; i < k && chain; ++i) {
                    uint32_t oi = tps[i].pos_off[idx[i]];
                    uint32_t ni = tps[i].p.tfs[idx[i]];
                    const uint32_t* lo2 = P + oi;
                    const uint32_t* hi2 = lo2 + ni;
                    chain = std::binary_search(lo2, hi2, p0 + uint32_t(i));
                }
                hit = chain;
            }
            if (hit) m.docs.push_back(doc);
        }
        return m;
    }

    // union of every dict term matching the glob (wildcard_query.rs ->
    // AutomatonQuery over the term dictionary; const score, see qast.h)
    Match eval_wildcard(const PlanNode& n) const {
        Match m;
        const TextFieldView* f = sv.text_field(n.field);
        if (!f) return m;
        std::vector<uint32_t> docs;
        for (uint32_t t = 0; t < f->num_terms; ++t) {
            const char* s = (const char*)f->h_term_bytes + f->h_term_offsets[t];
            size_t sl = f->h_term_offsets[t + 1] - f->h_term_offsets[t];
            if (!glob_match(s, sl, n.value.data(), n.value.size(), n.ci)) continue;
            Postings p = decode_term(*f, t);
            docs.insert(docs.end(), p.docs.begin(), p.docs.end());
        }
        std::sort(docs.begin(), docs.end());
        docs.erase(std::unique(docs.begin(), docs.end()), docs.end());
        m.docs = std::move(docs);
        return m;
    }

    Match eval_term(const PlanNode& n) const {
        Match m;
        const TextFieldView* f = sv.text_field(n.field);
        if (!f) return m;
        int64_t tid = f->find_term(n.value.data(), n.value.size());
        if (tid < 0) return m;
        Postings p = decode_term(*f, tid);
        m.docs = std::move(p.docs);
        if (scoring) {
            // BM25 (restated; golden-pinned — see file header). Weight in
            // f64, per-posting arithmetic in f32 exactly as the GPU kernel
            // computes it: score = W * tf / (tf + K[normid]).
            double N = double(sv.num_docs);
            double df = double(f->h_doc_freq[tid]);
            double idf = std::log(1.0 + (N - df + 0.5) / (df + 0.5));
            const double K1 = 1.2, B = 0.75;
            float W = float(idf * (1.0 + K1) * double(n.boost));
            double avgdl = f->total_tokens > 0 && sv.num_docs > 0
                               ? double(f->total_tokens) / double(sv.num_docs)
                               : 0.0;
            float Ktab[256];
            for (int i = 0; i < 256; ++i) {
                double fn = double(FIELDNORM_TABLE.v[i]);
                Ktab[i] = float(K1 * (1.0 - B + B * fn / (avgdl > 0 ? avgdl : 1.0)));
            }
            m.scores.resize(m.docs.size());
            const uint8_t* norms = f->h_fieldnorms;
            for (size_t i = 0; i < m.docs.size(); ++i) {
                float tf = float(p.tfs[i]);
                float K = norms ? Ktab[norms[m.docs[i]]] : Ktab[fieldnorm_encode(1)];
                m.scores[i] = W * (tf / (tf + K));
            }
        }
        return m;
    }

    bool range_test(const FastFieldView& f, uint32_t d, const Bound& lo,
                    const Bound& hi) const {
        if (!f.present(d)) return false;
        if (f.type == FastFieldView::U64) {
            uint64_t v = f.u64(d);
            if (lo.kind != Bound::UNBOUNDED) {
                uint64_t b = uint64_t(lo.ival);
                if (lo.kind == Bound::INCLUDED ? v < b : v <= b) return false;
            }
            if (hi.kind != Bound::UNBOUNDED) {
                uint64_t b = uint64_t(hi.ival);
                if (hi.kind == Bound::INCLUDED ? v > b : v >= b) return false;
            }
            return true;
        }
        if (f.type == FastFieldView::F64) {
            double v = f.f64(d);
            if (lo.kind != Bound::UNBOUNDED &&
                (lo.kind == Bound::INCLUDED ? v < lo.fval : v <= lo.fval))
                return false;
            if (hi.kind != Bound::UNBOUNDED &&
                (hi.kind == Bound::INCLUDED ? v > hi.fval : v >= hi.fval))
                return false;
            return true;
        }
        if (f.type == FastFieldView::STR) {
            // lexicographic bounds; lowercase-normalized columns fold bounds
            auto fold = [&](const std::string& s) {
                if (!f.lower_norm) return s;
                std::string o = s;
                for (char& c : o) c = char(std::tolower((unsigned char)c));
                return o;
            };
            std::string ls = fold(lo.sval), hs = fold(hi.sval);
            auto in_range = [&](const std::string& v) {
                if (lo.kind != Bound::UNBOUNDED &&
                    (lo.kind == Bound::INCLUDED ? v < ls : v <= ls))
                    return false;
                if (hi.kind != Bound::UNBOUNDED &&
                    (hi.kind == Bound::INCLUDED ? v > hs : v >= hs))
                    return false;
                return true;
            };
            if (f.multi) {  // any value in range matches
                uint32_t n = f.n_vals(d);
                for (uint32_t i = 0; i < n; ++i)
                    if (in_range(f.dict_entry(f.ord_at(d, i)))) return true;
                return false;
            }
            return in_range(f.dict_entry(f.ord(d)));
        }
        int64_t v = f.i64(d);
        if (lo.kind != Bound::UNBOUNDED &&
            (lo.kind == Bound::INCLUDED ? v < lo.ival : v <= lo.ival))
            return false;
        if (hi.kind != Bound::UNBOUNDED &&
            (hi.kind == Bound::INCLUDED ? v > hi.ival : v >= hi.ival))
            return false;
        return true;
    }

    Match eval_range(const PlanNode& n) const {
        Match m;
        const FastFieldView* f = sv.fast_field(n.field);
        if (!f) return m;
        if (f->type == FastFieldView::MIXED)
            throw std::runtime_error(
                "range/term over a mixed-type dynamic column (r2 limit)");
        for (uint32_t d = 0; d < sv.num_docs; ++d)
            if (range_test(*f, d, n.lo, n.hi)) m.docs.push_back(d);
        return m;
    }

    Match eval_presence(const PlanNode& n) const {
        Match m;
        const FastFieldView* f = sv.fast_field(n.field);
        if (!f) {
            // text-only field: present iff the fieldnorm byte is nonzero
            const TextFieldView* tf = sv.text_field(n.field);
            if (tf && tf->has_norms)
                for (uint32_t d = 0; d < sv.num_docs; ++d)
                    if (tf->h_fieldnorms[d]) m.docs.push_back(d);
            return m;
        }
        if (!f->nullable && !f->multi) {
            m.all = true;
            return m;
        }
        for (uint32_t d = 0; d < sv.num_docs; ++d)
            if (f->present(d)) m.docs.push_back(d);
        return m;
    }

    Match eval_bool(const PlanNode& n) const {
        if (n.must.empty() && n.filter.empty() && n.should.empty() &&
            n.must_not.empty() && n.minimum_should_match <= 0) {
            Match m;  // empty bool = match_all (tantivy_query_ast.rs:193)
            m.all = true;
            return m;
        }
        std::vector<Match> req;
        for (auto& c : n.must) req.push_back(eval(c));
        std::vector<Match> filt;
        for (auto& c : n.filter) {
            Match m = eval(c);
            m.scores.clear();  // filter clauses never score
            filt.push_back(std::move(m));
        }
        std::vector<Match> shoulds;
        for (auto& c : n.should) shoulds.push_back(eval(c));

        Match base;
        bool has_req = !req.empty() || !filt.empty();
        if (has_req) {
            base.all = true;
            // sum must scores: intersect accumulates scores
            for (auto& m : req) base = intersect(base, m);
            for (auto& m : filt) base = intersect(base, m);
            int64_t msm = n.minimum_should_match < 0 ? 0 : n.minimum_should_match;
            if (!shoulds.empty()) {
                bool should_all = false;
                for (auto& s : shoulds) should_all |= s.all;
                if (msm > 0) {
                    if (should_all && shoulds.size() == 1) {
                        // match_all should: always satisfied
                    } else {
                        Match su = union_n(shoulds, size_t(msm), sv.num_docs);
                        Match su_noscore = su;
                        su_noscore.scores.clear();
                        base = intersect(base, su_noscore);
                    }
                }
                if (scoring) {
                    // add should scores for docs already in base (merge-join)
                    Match su = union_n(shoulds, 1, sv.num_docs);
                    if (!su.scores.empty()) {
                        if (base.all) {
                            // materialize: an all-base (match_all filters)
                            // has no doc list to carry scores on
                            base.all = false;
                            base.docs.resize(sv.num_docs);
                            for (uint32_t d = 0; d < sv.num_docs; ++d)
                                base.docs[d] = d;
                        }
                        if (base.scores.empty()) base.scores.resize(base.docs.size(), 0.f);
                        size_t j = 0;
                        for (size_t i = 0; i < base.docs.size(); ++i) {
                            while (j < su.docs.size() && su.docs[j] < base.docs[i]) ++j;
                            if (j < su.docs.size() && su.docs[j] == base.docs[i])
                                base.scores[i] += su.scores[j];
                        }
                    }
                }
            }
        } else {
            size_t msm = n.minimum_should_match < 0 ? 1 : size_t(n.minimum_should_match);
            base = union_n(shoulds, msm, sv.num_docs);  // handles match_all
                                                        // clauses (see above)
            if (shoulds.empty()) {
                base = Match{};
                // must_not with no positive clause: implicit match_all
                // (tantivy_query_ast.rs:310-322 pushes match_all)
                if (!n.must_not.empty()) base.all = true;
            }
        }
        if (!n.must_not.empty()) {
            std::vector<Match> nots;
            bool not_all = false;
            for (auto& c : n.must_not) {
                Match m = eval(c);
                not_all |= m.all;
                nots.push_back(std::move(m));
            }
            if (not_all) return Match{};  // must_not match_all excludes everything
            Match nu = union_n(nots, 1, sv.num_docs);
            base = subtract(std::move(base), nu, sv.num_docs);
        }
        if (n.boost != 1.0f && !base.scores.empty())
            for (auto& s : base.scores) s *= n.boost;
        return base;
    }
};

// --------------------------------------------------------------- sort values
struct SortSpec {
    enum Comp { DOC_ID, SCORE, FAST_FIELD } comp = DOC_ID;
    const FastFieldView* ff = nullptr;
    int order = 1;  // 0 asc 1 desc
};

static pb::SortByValue sort_value_of(const SortSpec& s, const SplitView& sv, uint32_t doc,
                                     float score) {
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

// --------------------------------------------------------------- aggregation
static IntermediateAggResults collect_aggs(const std::vector<AggDef>& defs,
                                           const SplitView& sv, const Match& m) {
    IntermediateAggResults out;
    for (const AggDef& d : defs) {
        AggResult a;
        a.name = d.name;
        for (auto& s : d.sub) a.sub_names.push_back(s.name);
        const FastFieldView* f = sv.fast_field(d.field);
        std::vector<const FastFieldView*> subf;
        for (auto& s : d.sub) subf.push_back(sv.fast_field(s.field));
        auto sub_value = [&](size_t si, uint32_t doc, double* v) -> bool {
            const FastFieldView* sf = subf[si];
            if (!sf || !sf->present(doc)) return false;
            switch (sf->type) {
                case FastFieldView::U64: *v = double(sf->u64(doc)); return true;
                case FastFieldView::STR: return false;
                case FastFieldView::F64: *v = sf->f64(doc); return true;
                default: *v = double(sf->i64(doc)); return true;
            }
        };
        if (d.kind == AggDef::TERMS || d.kind == AggDef::CARDINALITY) {
            a.kind = 3;
            bool with_subs = d.kind == AggDef::TERMS && !d.sub.empty();
            if (with_subs) {
                for (auto& s : d.sub) {
                    if (s.kind == MetricAgg::PERCENTILES)
                        throw std::runtime_error(
                            "percentiles under terms (r1 limit)");
                    a.sub_names.push_back(s.name);
                    a.sub_kinds.push_back(0);
                }
            }
            if (f && f->type == FastFieldView::STR) {
                std::vector<uint64_t> counts(f->cardinality, 0);
                std::vector<std::vector<StatsPayload>> subs(
                    with_subs ? f->cardinality : 0);
                auto acc_subs = [&](uint32_t ord, uint32_t doc) {
                    if (!with_subs) return;
                    if (subs[ord].empty()) subs[ord].resize(d.sub.size());
                    for (size_t si = 0; si < d.sub.size(); ++si) {
                        double sval;
                        if (!sub_value(si, doc, &sval)) continue;
                        StatsPayload& sp = subs[ord][si];
                        sp.count++;
                        sp.sum += sval;
                        sp.min = std::min(sp.min, sval);
                        sp.max = std::max(sp.max, sval);
                        sp.sum_sq += sval * sval;
                    }
                };
                auto visit = [&](uint32_t doc) {
                    if (!f->present(doc)) return;
                    if (f->multi) {
                        uint32_t n = f->n_vals(doc);
                        for (uint32_t i = 0; i < n; ++i) {
                            counts[f->ord_at(doc, i)]++;  // distinct per doc
                            acc_subs(f->ord_at(doc, i), doc);
                        }
                        // sum_other base counts VALUE instances (equals the
                        // sum of all bucket doc_counts, like the reference)
                        a.terms_matched_docs += n;
                    } else {
                        counts[f->ord(doc)]++;
                        acc_subs(f->ord(doc), doc);
                        a.terms_matched_docs++;
                    }
                };
                if (m.all)
                    for (uint32_t doc = 0; doc < sv.num_docs; ++doc) visit(doc);
                else
                    for (uint32_t doc : m.docs) visit(doc);
                for (uint32_t o = 0; o < f->cardinality; ++o)
                    if (counts[o]) {
                        a.term_counts.emplace_back(f->dict_entry(o), counts[o]);
                        if (with_subs)
                            a.term_subs.push_back(
                                subs[o].empty()
                                    ? std::vector<StatsPayload>(d.sub.size())
                                    : std::move(subs[o]));
                    }
                // dict order is lexicographic => term_counts sorted by key;
                // per-split split_size truncation + error bound (qagg_format.h)
                if (d.kind == AggDef::TERMS)
                    truncate_terms_split(
                        a, effective_split_size(d.size, d.split_size),
                        d.order_target, d.order_asc, &d.sub);
            } else if (f && !f->multi) {
                // terms over a numeric fast column: count by the value's
                // order-preserving sortable bits, keys encoded big-endian so
                // map order == numeric order (mirrors the product's
                // AGGD_TERMS_NUM hash table; tantivy term_agg.rs keys by the
                // column value)
                if (sv.num_docs > (1u << 21))
                    throw std::runtime_error(
                        "terms aggregation over a numeric fast field on a "
                        ">2M-doc split (r1 limit)");
                a.key_kind = f->type == FastFieldView::U64   ? 1
                             : f->type == FastFieldView::F64 ? 3
                                                             : 2;
                std::map<uint64_t, uint64_t> counts;
                auto nvisit = [&](uint32_t doc) {
                    if (!f->present(doc)) return;
                    uint64_t v;
                    if (f->type == FastFieldView::U64) v = f->u64(doc);
                    else if (f->type == FastFieldView::F64)
                        v = f64_to_u64(f->f64(doc));
                    else v = i64_to_u64(f->i64(doc));
                    counts[v]++;
                    a.terms_matched_docs++;
                };
                if (m.all)
                    for (uint32_t doc = 0; doc < sv.num_docs; ++doc) nvisit(doc);
                else
                    for (uint32_t doc : m.docs) nvisit(doc);
                for (auto& kv : counts)
                    a.term_counts.emplace_back(num_term_key(kv.first), kv.second);
                if (d.kind == AggDef::TERMS)
                    truncate_terms_split(
                        a, effective_split_size(d.size, d.split_size),
                        d.order_target, d.order_asc, &d.sub);
            }
            out.aggs.push_back(std::move(a));
            continue;
        }
        if (d.kind == AggDef::METRIC && d.metric.kind == MetricAgg::PERCENTILES) {
            a.kind = 6;
            if (f && f->type != FastFieldView::STR && !f->multi) {
                auto pvisit = [&](uint32_t doc) {
                    if (!f->present(doc)) return;
                    double v;
                    if (f->type == FastFieldView::U64) v = double(f->u64(doc));
                    else if (f->type == FastFieldView::F64) v = f->f64(doc);
                    else v = double(f->i64(doc));
                    if (v < 0)
                        throw std::runtime_error(
                            "percentiles over negative values (r1 limit)");
                    if (v < PERC_MIN_VALUE) a.sketch.zero++;
                    else a.sketch.counts[perc_key_for(v)]++;
                };
                if (m.all)
                    for (uint32_t doc = 0; doc < sv.num_docs; ++doc) pvisit(doc);
                else
                    for (uint32_t doc : m.docs) pvisit(doc);
            }
            out.aggs.push_back(std::move(a));
            continue;
        }
        if (d.kind == AggDef::METRIC) {
            a.kind = 5;
            if (f && f->type != FastFieldView::STR && !f->multi) {
                auto mvisit = [&](uint32_t doc) {
                    if (!f->present(doc)) return;
                    double v;
                    if (f->type == FastFieldView::U64) v = double(f->u64(doc));
                    else if (f->type == FastFieldView::F64) v = f->f64(doc);
                    else v = double(f->i64(doc));
                    a.metric.count++;
                    a.metric.sum += v;
                    a.metric.min = std::min(a.metric.min, v);
                    a.metric.max = std::max(a.metric.max, v);
                    a.metric.sum_sq += v * v;
                };
                if (m.all)
                    for (uint32_t doc = 0; doc < sv.num_docs; ++doc) mvisit(doc);
                else
                    for (uint32_t doc : m.docs) mvisit(doc);
            }
            out.aggs.push_back(std::move(a));
            continue;
        }
        if (d.kind == AggDef::COMPOSITE) {
            // composite: one canonical byte-encoded tuple per doc (null /
            // str / f64 components — qagg_format.h comp_encode_*), counted
            // in a map whose order IS composite-key order (mirrors the
            // product's packed-u64 hash table + assembly decode)
            a.kind = 3;
            a.key_kind = 9;
            bool all_present = true;
            std::vector<const FastFieldView*> srcs;
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
                srcs.push_back(sf);
            }
            if (all_present) {
                if (sv.num_docs > (1u << 21))
                    throw std::runtime_error(
                        "composite aggregation on a >2M-doc split (r1 limit)");
                std::map<std::string, uint64_t> counts;
                auto cvisit = [&](uint32_t doc) {
                    std::string ck;
                    for (size_t si = 0; si < d.comp.size(); ++si) {
                        const CompSource& cs = d.comp[si];
                        const FastFieldView* sf = srcs[si];
                        if (!sf->present(doc)) {
                            if (!cs.missing_bucket) return;
                            comp_encode_null(ck);
                        } else if (cs.is_histo) {
                            double v;
                            if (sf->type == FastFieldView::U64)
                                v = double(sf->u64(doc));
                            else if (sf->type == FastFieldView::F64)
                                v = sf->f64(doc);
                            else v = double(sf->i64(doc));
                            double key = std::floor((v - cs.offset) /
                                                    cs.interval) *
                                             cs.interval +
                                         cs.offset;
                            comp_encode_f64(ck, key);
                        } else {
                            comp_encode_str(ck, sf->dict_entry(sf->ord(doc)));
                        }
                    }
                    counts[ck]++;
                };
                if (m.all)
                    for (uint32_t doc = 0; doc < sv.num_docs; ++doc) cvisit(doc);
                else
                    for (uint32_t doc : m.docs) cvisit(doc);
                for (auto& kv : counts)
                    a.term_counts.emplace_back(kv.first, kv.second);
            }
            out.aggs.push_back(std::move(a));
            continue;
        }
        if (d.kind == AggDef::RANGE) {
            a.kind = 4;
            if (f && f->type != FastFieldView::STR && !f->multi) {
                std::vector<uint64_t> counts(d.ranges.size(), 0);
                auto rvisit = [&](uint32_t doc) {
                    if (!f->present(doc)) return;
                    double v;
                    if (f->type == FastFieldView::U64) v = double(f->u64(doc));
                    else if (f->type == FastFieldView::F64) v = f->f64(doc);
                    else v = double(f->i64(doc));
                    for (size_t ri = 0; ri < d.ranges.size(); ++ri) {
                        const RangeSpec& r = d.ranges[ri];
                        if (r.has_from && v < r.from) continue;
                        if (r.has_to && v >= r.to) continue;
                        counts[ri]++;
                    }
                };
                if (m.all)
                    for (uint32_t doc = 0; doc < sv.num_docs; ++doc) rvisit(doc);
                else
                    for (uint32_t doc : m.docs) rvisit(doc);
                for (size_t ri = 0; ri < d.ranges.size(); ++ri) {
                    if (!counts[ri]) continue;
                    AggBucket b;
                    b.key = double(ri);  // range index; finalize maps back
                    b.doc_count = counts[ri];
                    a.buckets.push_back(std::move(b));
                }
            }
            out.aggs.push_back(std::move(a));
            continue;
        }
        a.kind = d.kind == AggDef::DATE_HISTOGRAM ? 1 : 2;
        for (auto& s : d.sub)
            a.sub_kinds.push_back(s.kind == MetricAgg::PERCENTILES ? 1 : 0);
        std::map<int64_t, AggBucket> buckets;  // key quantized to bucket index
        auto visit = [&](uint32_t doc) {
            if (!f || !f->present(doc)) return;
            double v;
            if (f->type == FastFieldView::U64) v = double(f->u64(doc));
            else if (f->type == FastFieldView::STR) return;
            else if (f->type == FastFieldView::F64) v = f->f64(doc);
            else v = double(f->i64(doc));
            int64_t bi = int64_t(std::floor((v - d.offset) / d.interval));
            AggBucket& b = buckets[bi];
            b.doc_count++;
            if (!d.sub.empty() && b.sub.empty()) {
                b.sub.resize(d.sub.size());
                b.psub.resize(d.sub.size());
            }
            for (size_t si = 0; si < d.sub.size(); ++si) {
                double sval;
                if (sub_value(si, doc, &sval)) {
                    if (d.sub[si].kind == MetricAgg::PERCENTILES) {
                        if (sval < 0)
                            throw std::runtime_error(
                                "percentiles over negative values (r1 limit)");
                        SketchPayload& pp = b.psub[si];
                        if (sval < PERC_MIN_VALUE) pp.zero++;
                        else pp.counts[perc_key_for(sval)]++;
                        continue;
                    }
                    StatsPayload& sp = b.sub[si];
                    sp.count++;
                    sp.sum += sval;
                    sp.min = std::min(sp.min, sval);
                    sp.max = std::max(sp.max, sval);
                    sp.sum_sq += sval * sval;
                }
            }
        };
        if (m.all)
            for (uint32_t doc = 0; doc < sv.num_docs; ++doc) visit(doc);
        else
            for (uint32_t doc : m.docs) visit(doc);
        for (auto& kv : buckets) {
            AggBucket b = kv.second;
            b.key = double(kv.first) * d.interval + d.offset;
            if (b.sub.empty() && !d.sub.empty()) b.sub.resize(d.sub.size());
            if (b.psub.empty() && !d.sub.empty()) b.psub.resize(d.sub.size());
            a.buckets.push_back(std::move(b));
        }
        out.aggs.push_back(std::move(a));
    }
    return out;
}

// --------------------------------------------------------------- per split
struct SplitResult {
    uint64_t num_hits = 0;
    std::vector<pb::PartialHit> hits;
    IntermediateAggResults aggs;
    bool has_aggs = false;
    uint64_t cpu_micros = 0;
    std::string error;
};

static SplitResult search_split(const SplitView& sv, const pb::SearchRequest& req,
                                const Schema& schema) {
    SplitResult out;
    auto t0 = std::chrono::steady_clock::now();
    PlanNode plan = parse_query_ast(req.query_ast, schema);
    // timestamp pruning filter: [start, end) in seconds
    // (leaf.rs:977 rewrite_request applies it via the timestamp fast field)
    if ((req.start_timestamp || req.end_timestamp) && !schema.timestamp_field.empty()) {
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

    // phrases are const-score (like wildcard/term_set): reject them under
    // _score sorting instead of mis-scoring — the product does the same
    std::function<bool(const PlanNode&)> has_phrase =
        [&](const PlanNode& pn) -> bool {
        if (pn.kind == PlanNode::PHRASE) return true;
        for (auto* v : {&pn.must, &pn.must_not, &pn.should, &pn.filter,
                        &pn.cache_inner})
            for (auto& c : *v)
                if (has_phrase(c)) return true;
        return false;
    };

    // sort specs (max 2, like the reference — search.proto:269)
    if (req.sort_fields.size() > 2)
        throw std::runtime_error(
            "more than two sort fields (search.proto:269)");
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
    if (scoring && has_phrase(plan))
        throw std::runtime_error(
            "phrase under _score sorting (const-score semantics, r2 limit)");

    if (scoring && plan_has_const_score(plan))
        throw std::runtime_error(
            "term_set/wildcard under _score sorting not supported (const-score "
            "semantics, qast.h)");
    SplitSearcher searcher{sv, schema, scoring};
    Match m = searcher.eval(plan);
    out.num_hits = m.size(sv.num_docs);

    uint64_t leaf_max_hits = req.max_hits + req.start_offset;
    if (leaf_max_hits > 0 && out.num_hits > 0) {
        int order1 = specs.empty() ? 1 : specs[0].order;
        int order2 = specs.size() > 1 ? specs[1].order : 1;
        auto mk_hit = [&](uint32_t doc, float score) {
            pb::PartialHit h;
            h.split_id = sv.split_id;
            h.segment_ord = sv.segment_ord;
            h.doc_id = doc;
            if (!specs.empty()) h.sort_value = sort_value_of(specs[0], sv, doc, score);
            if (specs.size() > 1) h.sort_value2 = sort_value_of(specs[1], sv, doc, score);
            return h;
        };
        std::vector<pb::PartialHit> cand;
        bool doc_order_shortcut = specs.empty() && !req.search_after;
        if (m.all) {
            // no clause work: hits are just doc ids; default order doc desc
            cand.reserve(std::min<uint64_t>(sv.num_docs, leaf_max_hits * 4 + 16));
            if (doc_order_shortcut) {
                // doc-id sort: take head/tail directly
                uint64_t k = std::min<uint64_t>(leaf_max_hits, sv.num_docs);
                for (uint64_t i = 0; i < k; ++i) {
                    uint32_t doc = order1 == 1 ? uint32_t(sv.num_docs - 1 - i) : uint32_t(i);
                    cand.push_back(mk_hit(doc, 0.f));
                }
            } else {
                for (uint32_t doc = 0; doc < sv.num_docs; ++doc)
                    cand.push_back(mk_hit(doc, 0.f));
            }
        } else if (doc_order_shortcut) {
            uint64_t k = std::min<uint64_t>(leaf_max_hits, m.docs.size());
            for (uint64_t i = 0; i < k; ++i) {
                uint32_t doc = order1 == 1 ? m.docs[m.docs.size() - 1 - i] : m.docs[i];
                cand.push_back(mk_hit(doc, 0.f));
            }
        } else {
            cand.reserve(m.docs.size());
            for (size_t i = 0; i < m.docs.size(); ++i)
                cand.push_back(mk_hit(m.docs[i], m.scores.empty() ? 0.f : m.scores[i]));
        }
        // search_after: keep only hits strictly after the cursor
        // (top_k_collector.rs:663-700; num_hits counting is unaffected)
        if (req.search_after) {
            const pb::PartialHit& c = *req.search_after;
            auto kind_of = [&](const SortSpec& s) {
                if (s.comp == SortSpec::SCORE) return SortFieldKind::SCORE;
                if (s.comp != SortSpec::FAST_FIELD || !s.ff)
                    return SortFieldKind::NONE;
                switch (s.ff->type) {
                    case FastFieldView::U64: return SortFieldKind::U64;
                    case FastFieldView::I64: return SortFieldKind::I64;
                    case FastFieldView::DATETIME: return SortFieldKind::DATETIME;
                    case FastFieldView::F64: return SortFieldKind::F64;
                    case FastFieldView::MIXED: return SortFieldKind::MIXED;
                    default: return SortFieldKind::STR;
                }
            };
            CursorKey k1 = specs.empty()
                               ? CursorKey{}
                               : convert_cursor_key(c.sort_value,
                                                    kind_of(specs[0]), order1);
            CursorKey k2 = specs.size() < 2
                               ? CursorKey{}
                               : convert_cursor_key(c.sort_value2,
                                                    kind_of(specs[1]), order2);
            if (!k1.disabled)
                cand.erase(
                    std::remove_if(
                        cand.begin(), cand.end(),
                        [&](const pb::PartialHit& h) {
                            return !after_cursor(
                                h, c, k1, k2, order1, order2,
                                !specs.empty() && kind_of(specs[0]) ==
                                                      SortFieldKind::MIXED,
                                specs.size() > 1 && kind_of(specs[1]) ==
                                                        SortFieldKind::MIXED);
                        }),
                    cand.end());
        }
        auto cmp = [&](const pb::PartialHit& a, const pb::PartialHit& b) {
            return hit_before(a, b, order1, order2);
        };
        size_t k = std::min<size_t>(leaf_max_hits, cand.size());
        std::partial_sort(cand.begin(), cand.begin() + k, cand.end(), cmp);
        cand.resize(k);
        out.hits = std::move(cand);
    }

    if (req.aggregation_request) {
        std::vector<AggDef> defs = parse_agg_request(*req.aggregation_request);
        auto agg_col_type = [&](const std::string& name) -> int {
            const FastFieldView* cf = sv.fast_field(name);
            return cf ? int(cf->type) : -1;
        };
        validate_agg_fields(defs, agg_col_type);
        out.aggs = collect_aggs(defs, sv, m);
        out.has_aggs = true;
    }
    out.cpu_micros = uint64_t(std::chrono::duration_cast<std::chrono::microseconds>(
                                  std::chrono::steady_clock::now() - t0)
                                  .count());
    return out;
}

}  // namespace oracle
}  // namespace qw

// ============================================================== C ABI
extern "C" {

typedef struct qw_oracle_ctx qw_oracle_ctx;

struct qw_oracle_ctx {
    struct SplitHolder {
        std::vector<uint8_t> data;
        qw::SplitView view;                  // single-segment (or segment 0)
        std::vector<qw::SplitView> seg_views;  // QWA2: one view per segment
    };
    std::map<std::string, std::unique_ptr<SplitHolder>> splits;
    std::string last_error;
};

typedef struct {
    uint8_t* data;
    size_t len;
} qw_oracle_buf;

qw_oracle_ctx* qw_oracle_create() { return new qw_oracle_ctx(); }
void qw_oracle_free(qw_oracle_ctx* ctx) { delete ctx; }
const char* qw_oracle_last_error(qw_oracle_ctx* ctx) { return ctx->last_error.c_str(); }
void qw_oracle_buf_free(qw_oracle_buf* b) {
    free(b->data);
    b->data = nullptr;
    b->len = 0;
}

// (the caller's split id is authoritative over the container's own id,
// matching the product's qw_ctx_add_split — responses echo the id the
// request named the split by)
int qw_oracle_add_split(qw_oracle_ctx* ctx, const char* split_id, const uint8_t* data,
                        size_t len) {
    try {
        auto h = std::make_unique<qw_oracle_ctx::SplitHolder>();
        h->data.assign(data, data + len);
        if (len >= 88 && memcmp(data, "QWAMDSP2", 8) == 0) {
            // QWA2 multi-segment container (same layout the product reads)
            uint64_t moff, mlen;
            memcpy(&moff, h->data.data() + len - 24, 8);
            memcpy(&mlen, h->data.data() + len - 16, 8);
            mj::ValuePtr meta =
                mj::parse((const char*)h->data.data() + moff, mlen);
            uint32_t ord = 0;
            for (auto& sgv : meta->at("segments")->arr) {
                uint64_t off = uint64_t(sgv->at("off")->as_i64());
                uint64_t slen = uint64_t(sgv->at("len")->as_i64());
                qw::SplitView v;
                v.parse(h->data.data() + off, slen);
                v.split_id = split_id;
                v.segment_ord = ord++;
                h->seg_views.push_back(std::move(v));
            }
            if (h->seg_views.empty())
                throw std::runtime_error("QWA2: no segments");
            h->view = h->seg_views[0];
        } else {
            h->view.parse(h->data.data(), h->data.size());
            h->view.split_id = split_id;
        }
        ctx->splits[split_id] = std::move(h);
        return 0;
    } catch (const std::exception& e) {
        ctx->last_error = e.what();
        return -6;
    }
}

static void fill_buf(qw_oracle_buf* out, const std::string& s) {
    out->data = (uint8_t*)malloc(s.size());
    memcpy(out->data, s.data(), s.size());
    out->len = s.size();
}

// LeafSearchRequest pb in -> LeafSearchResponse pb out. Splits run OpenMP-
// parallel (one thread each), then merge with IncrementalCollector semantics.
int qw_oracle_leaf_search(qw_oracle_ctx* ctx, const uint8_t* req_pb, size_t req_len,
                          qw_oracle_buf* out) {
    using namespace qw;
    using namespace qw::oracle;
    try {
        pb::LeafSearchRequest lreq = pb::LeafSearchRequest::decode(req_pb, req_len);
        pb::SearchRequest req = lreq.search_request;
        // count-only rewrite (leaf.rs:977-990, mirrored by the product's
        // per-split rewrite): no hits requested -> sort fields are dead
        if (req.max_hits == 0) req.sort_fields.clear();
        // one doc mapper (schema json) per index; r1 handles one index per call
        if (lreq.doc_mappers.empty()) throw std::runtime_error("missing doc_mapper");
        Schema schema = Schema::parse(lreq.doc_mappers[0]);

        struct Task {
            const qw::SplitView* view;  // one task per (split, segment)
            std::string split_id;
            bool first_seg;             // first segment of its split
        };
        std::vector<Task> tasks;
        std::vector<std::string> pre_failed;  // unknown splits -> data
        for (auto& lr : lreq.leaf_requests)
            for (auto& so : lr.split_offsets) {
                auto it = ctx->splits.find(so.split_id);
                if (it == ctx->splits.end()) {
                    // per-split failure as data (leaf.rs:2143-2148)
                    pre_failed.push_back(so.split_id);
                    continue;
                }
                const auto* holder = it->second.get();
                if (holder->seg_views.empty())
                    tasks.push_back({&holder->view, so.split_id, true});
                else {
                    bool first = true;
                    for (auto& v : holder->seg_views) {
                        tasks.push_back({&v, so.split_id, first});
                        first = false;
                    }
                }
            }

        std::vector<SplitResult> results(tasks.size());
#pragma omp parallel for schedule(dynamic)
        for (size_t i = 0; i < tasks.size(); ++i) {
            try {
                results[i] = search_split(*tasks[i].view, req, schema);
            } catch (const std::exception& e) {
                results[i].error = e.what();
            }
        }

        pb::LeafSearchResponse resp;
        for (auto& t : tasks)
            if (t.first_seg) resp.num_attempted_splits++;
        for (auto& sid : pre_failed) {
            resp.num_attempted_splits++;
            pb::SplitSearchError se;
            se.error = "unknown split: " + sid;
            se.split_id = sid;
            se.retryable_error = true;
            resp.failed_splits.push_back(std::move(se));
        }
        IntermediateAggResults merged_aggs;
        bool any_aggs = false;
        std::vector<pb::PartialHit> all_hits;
        pb::LeafResourceStats rstats;
        rstats.search_pool_cpu_threads = uint64_t(omp_get_max_threads());
        uint64_t worst_key = 0;
        for (size_t i = 0; i < tasks.size();) {
            size_t j = i + 1;  // [i, j) = this split's segment tasks
            while (j < tasks.size() && !tasks[j].first_seg) ++j;
            std::string err;
            uint64_t split_hits = 0, split_docs = 0, split_micros = 0;
            for (size_t t = i; t < j; ++t) {
                if (!results[t].error.empty() && err.empty())
                    err = results[t].error;
                split_hits += results[t].num_hits;
                split_docs += tasks[t].view->num_docs;
                split_micros += results[t].cpu_micros;
            }
            if (!err.empty()) {
                pb::SplitSearchError e;
                e.error = err;
                e.split_id = tasks[i].split_id;
                e.retryable_error = true;
                resp.failed_splits.push_back(std::move(e));
                i = j;
                continue;
            }
            resp.num_successful_splits++;
            resp.num_hits += split_hits;
            for (size_t t = i; t < j; ++t) {
                SplitResult& r = results[t];
                for (auto& h : r.hits) all_hits.push_back(std::move(h));
                if (r.has_aggs) {
                    if (!any_aggs) {
                        merged_aggs = std::move(r.aggs);
                        any_aggs = true;
                    } else merged_aggs.merge(r.aggs);
                }
            }
            pb::SplitResourceStats ss;
            ss.split_num_docs = split_docs;
            ss.matched_num_docs = split_hits;
            ss.cpu_search_microsecs = split_micros;
            rstats.localexec_num_splits++;
            rstats.localexec_num_docs += ss.split_num_docs;
            rstats.split_resources_sum.split_num_docs += ss.split_num_docs;
            rstats.split_resources_sum.matched_num_docs += ss.matched_num_docs;
            rstats.split_resources_sum.cpu_search_microsecs += ss.cpu_search_microsecs;
            if (ss.cpu_search_microsecs >= worst_key) {
                worst_key = ss.cpu_search_microsecs;
                rstats.split_resources_worst = ss;
            }
            i = j;
        }
        // cross-split merge (merge_fruits semantics, collector.rs:832-861)
        int order1 = req.sort_fields.empty() ? 1 : req.sort_fields[0].sort_order;
        int order2 = req.sort_fields.size() > 1 ? req.sort_fields[1].sort_order : 1;
        auto cmp = [&](const pb::PartialHit& a, const pb::PartialHit& b) {
            return qw::hit_before(a, b, order1, order2);
        };
        size_t k = std::min<size_t>(req.max_hits + req.start_offset, all_hits.size());
        std::partial_sort(all_hits.begin(), all_hits.begin() + k, all_hits.end(), cmp);
        all_hits.resize(k);
        resp.partial_hits = std::move(all_hits);
        if (any_aggs) resp.intermediate_aggregation_result = merged_aggs.encode();
        resp.resource_stats = rstats;
        fill_buf(out, resp.encode());
        return 0;
    } catch (const std::exception& e) {
        ctx->last_error = e.what();
        return -4;
    }
}

// finalize an intermediate agg blob to ES-shaped JSON (same code path the
// product exports as qw_finalize_agg_to_json — shared qagg_format.h)
int qw_oracle_finalize_agg(const uint8_t* blob, size_t len, const char* agg_req_json,
                           qw_oracle_buf* out) {
    try {
        qw::IntermediateAggResults ir = qw::IntermediateAggResults::decode(blob, len);
        std::vector<qw::AggDef> defs = qw::parse_agg_request(agg_req_json);
        fill_buf(out, qw::finalize_aggs_json(ir, defs));
        return 0;
    } catch (const std::exception& e) {
        (void)e;
        return -4;
    }
}

}  // extern "C"
