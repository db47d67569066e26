// Native QWA1 synthetic split generator (bench/test input tooling).
//
// Produces the same container format as quickwit_amd/splitgen.py
//  for the hdfs-logs-style synthetic corpus of SURVEY.md §8d:
// timestamp uniform over 30 days, tenant_id Zipf(1k), severity_text
// categorical {DEBUG 50%, INFO 40%, WARN 7%, ERROR 2.9%, FATAL 0.1%},
// body ~ 10 tokens/doc Zipf over a 10k vocabulary.
//
// The numpy generator materializes the (N,10) token matrix and sorts ~10N
// keys — minutes at 100M docs. Here each term's posting list is synthesized
// DIRECTLY (sorted by construction): doc membership is iid Bernoulli(q_t)
// with q_t = 1-(1-p_t)^10 (the exact marginal of the token-matrix model),
// sampled as Geometric(q_t) doc gaps; tf-1 | membership ~ Binomial(9, p_t)
// via a 9-entry inverse-CDF table. OpenMP over terms/doc-chunks; seeded
// xoshiro256++ streams fixed per (seed, split_ord, purpose) so output is
// independent of thread count. This is NOT the same byte stream as the
// numpy path (different RNG), but the same distributions; every consumer
// (oracle + product parity, bench) reads whichever split was generated.
//
// Build: g++ -O3 -fopenmp (no HIP, no GPU) -> libqwsplitgen.so.
#include <math.h>
#include <omp.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include <string>
#include <vector>

namespace {

constexpr int64_t BLOCK = 128;
constexpr int64_t ALIGN = 64;
constexpr int BODY_VOCAB = 10000;
constexpr int BODY_TOKENS = 10;
constexpr int N_TENANTS = 1000;
constexpr int64_t T0_EPOCH_S = 1700000000;
constexpr int64_t TS_RANGE_S = 30 * 86400;
constexpr int64_t DOC_CHUNK = 1 << 22;

// SEVERITIES draw order and probabilities (splitgen.py); vocab ord is the
// index in the SORTED vocab [DEBUG, ERROR, FATAL, INFO, WARN]
constexpr double SEV_P[5] = {0.50, 0.40, 0.07, 0.029, 0.001};  // DEBUG,INFO,WARN,ERROR,FATAL
constexpr int SEV_TO_ORD[5] = {0, 3, 4, 1, 2};
const char* SEV_VOCAB[5] = {"DEBUG", "ERROR", "FATAL", "INFO", "WARN"};

#pragma pack(push, 1)
struct Skip {
    uint32_t first_doc;
    uint32_t last_doc;
    uint32_t word_off;  // u32-word offset into the field payload (at gaps)
    uint8_t id_bits;
    uint8_t tf_bits;
    uint16_t count;
};
#pragma pack(pop)
static_assert(sizeof(Skip) == 16, "skip entry is 16B");

// ---------------------------------------------------------------- RNG
static inline uint64_t sm64(uint64_t& x) {
    uint64_t z = (x += 0x9E3779B97F4A7C15ull);
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    return z ^ (z >> 31);
}

struct Rng {  // xoshiro256++
    uint64_t s[4];
    void seed(int64_t seed, int64_t split_ord, uint64_t stream) {
        uint64_t x = uint64_t(seed) * 0xA24BAED4963EE407ull;
        x ^= uint64_t(split_ord) + 0x9E3779B97F4A7C15ull;
        x ^= stream * 0x9FB21C651E98DF25ull;
        for (int i = 0; i < 4; ++i) s[i] = sm64(x);
    }
    static inline uint64_t rotl(uint64_t v, int k) { return (v << k) | (v >> (64 - k)); }
    inline uint64_t next() {
        uint64_t r = rotl(s[0] + s[3], 23) + s[0];
        uint64_t t = s[1] << 17;
        s[2] ^= s[0];
        s[3] ^= s[1];
        s[1] ^= s[2];
        s[0] ^= s[3];
        s[2] ^= t;
        s[3] = rotl(s[3], 45);
        return r;
    }
    inline double next_double() {  // (0,1]: never 0 so log() is safe
        return double((next() >> 11) + 1) * 0x1.0p-53;
    }
    inline uint32_t next_u32() { return uint32_t(next() >> 32); }
};

// ------------------------------------------------------------- packing
struct PackedTerm {
    int64_t df = 0;
    std::vector<uint64_t> payload;  // anchors + gaps (+ tf-1), per block
    std::vector<Skip> skips;        // word_off local to this term's payload
};

static inline uint8_t bit_width(uint64_t m) {
    return m ? uint8_t(64 - __builtin_clzll(m)) : uint8_t(1);
}

static inline void pack_lane(uint64_t* w, int j, int width, uint64_t v) {
    int pos = j * width;
    int wd = pos >> 6, sh = pos & 63;
    w[wd] |= v << sh;
    if (sh + width > 64 && width > 1) w[wd + 1] |= v >> (64 - sh);
}

// docs: sorted doc ids; tf1 nullable (record: basic). Payload layout per
// block (format v2, splitgen.py _build_text_field): 2 anchor words
// [pad|doc31][doc63|doc95] then 2*id_bits words of gaps (gap[0]=0, base =
// skip.first_doc) then 2*tf_bits words of tf-1 when record=freq.
static void pack_term(const uint32_t* docs, const uint8_t* tf1, int64_t df,
                      PackedTerm& out) {
    out.df = df;
    int64_t nblk = (df + BLOCK - 1) / BLOCK;
    out.skips.resize(nblk);
    // pass 1: widths -> total words
    int64_t words = 0;
    for (int64_t b = 0; b < nblk; ++b) {
        int64_t s = b * BLOCK;
        int64_t cnt = (b == nblk - 1) ? df - s : BLOCK;
        uint64_t mg = 0, mt = 0;
        for (int64_t i = 1; i < cnt; ++i) {
            uint64_t g = docs[s + i] - docs[s + i - 1];
            if (g > mg) mg = g;
        }
        uint8_t idb = bit_width(mg);
        uint8_t tfb = 0;
        if (tf1) {
            for (int64_t i = 0; i < cnt; ++i)
                if (tf1[s + i] > mt) mt = tf1[s + i];
            tfb = bit_width(mt);
        }
        Skip& sk = out.skips[b];
        sk.first_doc = docs[s];
        sk.last_doc = docs[s + cnt - 1];
        sk.id_bits = idb;
        sk.tf_bits = tfb;
        sk.count = uint16_t(cnt);
        sk.word_off = uint32_t((words + 2) * 2);  // u32 units, at the gaps
        words += 2 + 2 * int64_t(idb) + 2 * int64_t(tfb);
    }
    out.payload.assign(words, 0);
    // pass 2: anchors + bitpack
    int64_t off = 0;
    for (int64_t b = 0; b < nblk; ++b) {
        int64_t s = b * BLOCK;
        const Skip& sk = out.skips[b];
        int64_t cnt = sk.count;
        uint64_t* w = out.payload.data() + off;
        uint64_t a1 = docs[s + (cnt - 1 < 31 ? cnt - 1 : 31)];
        uint64_t a2 = docs[s + (cnt - 1 < 63 ? cnt - 1 : 63)];
        uint64_t a3 = docs[s + (cnt - 1 < 95 ? cnt - 1 : 95)];
        w[0] = a1 << 32;
        w[1] = a2 | (a3 << 32);
        uint64_t* gw = w + 2;
        for (int64_t i = 1; i < cnt; ++i)
            pack_lane(gw, int(i), sk.id_bits, uint64_t(docs[s + i] - docs[s + i - 1]));
        if (tf1) {
            uint64_t* tw = gw + 2 * int64_t(sk.id_bits);
            for (int64_t i = 0; i < cnt; ++i)
                pack_lane(tw, int(i), sk.tf_bits, uint64_t(tf1[s + i]));
        }
        off += 2 + 2 * int64_t(sk.id_bits) + 2 * int64_t(sk.tf_bits);
    }
}

// --------------------------------------------------------- section layout
struct Layout {
    std::vector<uint8_t>* buf;
    int64_t pos = ALIGN;  // after magic+pad
    int64_t add(int64_t len, int64_t* off_out) {
        *off_out = pos;
        pos += len;
        if (pos % ALIGN) pos += ALIGN - pos % ALIGN;
        return *off_out;
    }
};

struct Sec {
    int64_t off = 0, len = 0;
};

static void json_sec(std::string& j, const char* name, const Sec& s, bool first) {
    char tmp[128];
    snprintf(tmp, sizeof tmp, "%s\"%s\":[%lld,%lld]", first ? "" : ",", name,
             (long long)s.off, (long long)s.len);
    j += tmp;
}

struct TextSecs {
    Sec term_offsets, term_bytes, posting_off, doc_freq, n_blocks, skip_off,
        skip, payload, fieldnorms;
};

}  // namespace

extern "C" {

int64_t qw_gen_split(int64_t split_ord, int64_t num_docs, int64_t seed,
                     uint8_t** out, size_t* out_len) {
    if (num_docs <= 0 || !out || !out_len) return -1;
    const int64_t N = num_docs;

    // ---- body term probabilities (Zipf s=1 over 10k)
    std::vector<double> p(BODY_VOCAB), q(BODY_VOCAB);
    {
        double H = 0;
        for (int t = 0; t < BODY_VOCAB; ++t) H += 1.0 / double(t + 1);
        for (int t = 0; t < BODY_VOCAB; ++t) {
            p[t] = (1.0 / double(t + 1)) / H;
            q[t] = 1.0 - pow(1.0 - p[t], double(BODY_TOKENS));
        }
    }

    // ---- body postings: per-term geometric gaps + binomial tf
    std::vector<PackedTerm> body(BODY_VOCAB);
#pragma omp parallel
    {
        std::vector<uint32_t> docs;
        std::vector<uint8_t> tf1;
#pragma omp for schedule(dynamic, 8)
        for (int t = 0; t < BODY_VOCAB; ++t) {
            Rng rg, rt;
            rg.seed(seed, split_ord, 0x100000000ull + 2 * uint64_t(t));
            rt.seed(seed, split_ord, 0x100000000ull + 2 * uint64_t(t) + 1);
            double inv_log1mq = 1.0 / log1p(-q[t]);
            docs.clear();
            docs.reserve(size_t(N * q[t] * 1.01) + 64);
            int64_t cur = -1;
            for (;;) {
                // Geometric(q): 1 + floor(ln(U)/ln(1-q)), U in (0,1]
                int64_t gap = 1 + int64_t(log(rg.next_double()) * inv_log1mq);
                if (gap < 1) gap = 1;  // fp edge
                cur += gap;
                if (cur >= N) break;
                docs.push_back(uint32_t(cur));
            }
            int64_t df = int64_t(docs.size());
            // tf-1 | membership ~ Binomial(9, p_t): 9 u32 CDF thresholds
            uint32_t thr[BODY_TOKENS - 1];
            int nthr = 0;
            {
                double cdf = 0, pt = p[t];
                double pk = pow(1.0 - pt, double(BODY_TOKENS - 1));  // k=0
                double ratio = pt / (1.0 - pt);
                for (int k = 0; k < BODY_TOKENS - 1; ++k) {
                    cdf += pk;
                    if (cdf >= 1.0 - 1e-12) break;
                    double v = cdf * 4294967296.0;
                    if (v >= 4294967295.0) break;
                    thr[nthr++] = uint32_t(v);
                    pk *= ratio * double(BODY_TOKENS - 1 - k) / double(k + 1);
                }
            }
            tf1.resize(df);
            for (int64_t i = 0; i < df; ++i) {
                uint32_t u = rt.next_u32();
                uint8_t v = 0;
                for (int k = nthr - 1; k >= 0; --k) {
                    if (u >= thr[k]) {
                        v = uint8_t(k + 1);
                        break;
                    }
                }
                tf1[i] = v;
            }
            pack_term(docs.data(), tf1.data(), df, body[t]);
        }
    }

    // ---- severity: one categorical draw per doc, postings per class
    int64_t n_chunks = (N + DOC_CHUNK - 1) / DOC_CHUNK;
    std::vector<uint8_t> sev(N);  // vocab ord per doc
    std::vector<int64_t> sev_cnt(size_t(n_chunks) * 5, 0);
    uint32_t sev_thr[4];
    {
        double c = 0;
        for (int i = 0; i < 4; ++i) {
            c += SEV_P[i];
            sev_thr[i] = uint32_t(c * 4294967296.0);
        }
    }
#pragma omp parallel for schedule(dynamic, 1)
    for (int64_t c = 0; c < n_chunks; ++c) {
        Rng r;
        r.seed(seed, split_ord, 0x200000000ull + uint64_t(c));
        int64_t lo = c * DOC_CHUNK, hi = lo + DOC_CHUNK < N ? lo + DOC_CHUNK : N;
        int64_t* cnt = &sev_cnt[size_t(c) * 5];
        for (int64_t d = lo; d < hi; ++d) {
            uint32_t u = r.next_u32();
            int k = 4;
            for (int i = 0; i < 4; ++i)
                if (u < sev_thr[i]) {
                    k = i;
                    break;
                }
            uint8_t o = uint8_t(SEV_TO_ORD[k]);
            sev[d] = o;
            ++cnt[o];
        }
    }
    int64_t sev_df[5] = {0, 0, 0, 0, 0};
    std::vector<int64_t> sev_base(size_t(n_chunks) * 5);
    for (int o = 0; o < 5; ++o)
        for (int64_t c = 0; c < n_chunks; ++c) {
            sev_base[size_t(c) * 5 + o] = sev_df[o];
            sev_df[o] += sev_cnt[size_t(c) * 5 + o];
        }
    std::vector<std::vector<uint32_t>> sev_docs(5);
    for (int o = 0; o < 5; ++o) sev_docs[o].resize(sev_df[o]);
#pragma omp parallel for schedule(dynamic, 1)
    for (int64_t c = 0; c < n_chunks; ++c) {
        int64_t lo = c * DOC_CHUNK, hi = lo + DOC_CHUNK < N ? lo + DOC_CHUNK : N;
        int64_t pos[5];
        for (int o = 0; o < 5; ++o) pos[o] = sev_base[size_t(c) * 5 + o];
        for (int64_t d = lo; d < hi; ++d) {
            int o = sev[d];
            sev_docs[o][pos[o]++] = uint32_t(d);
        }
    }
    std::vector<PackedTerm> sevp(5);
#pragma omp parallel for schedule(dynamic, 1)
    for (int o = 0; o < 5; ++o)
        pack_term(sev_docs[o].data(), nullptr, sev_df[o], sevp[o]);
    sev.clear();
    sev.shrink_to_fit();
    for (int o = 0; o < 5; ++o) {
        sev_docs[o].clear();
        sev_docs[o].shrink_to_fit();
    }

    // ---- layout
    auto pay_words = [](const std::vector<PackedTerm>& v) {
        int64_t w = 0;
        for (const auto& t : v) w += int64_t(t.payload.size());
        return w;
    };
    auto blk_count = [](const std::vector<PackedTerm>& v) {
        int64_t b = 0;
        for (const auto& t : v) b += int64_t(t.skips.size());
        return b;
    };
    int64_t body_words = pay_words(body), body_blocks = blk_count(body);
    int64_t sev_words = pay_words(sevp), sev_blocks = blk_count(sevp);

    Layout lay;
    Sec ts_values, ten_values, tn_values, tn_doff, tn_dbytes;
    TextSecs sevsec, bodysec;
    auto put = [&lay](Sec& s, int64_t len) {
        s.len = len;
        lay.add(len, &s.off);
    };
    // schema order: timestamp, tenant_id, tenant_name, severity_text, body
    put(ts_values, N * 8);
    put(ten_values, N * 8);
    put(tn_values, N * 2);  // ord_width 2 (cardinality 1000)
    put(tn_doff, (N_TENANTS + 1) * 4);
    put(tn_dbytes, N_TENANTS * 5);
    auto put_text = [&](TextSecs& ts, int nterms, int64_t term_bytes,
                        int64_t blocks, int64_t words) {
        put(ts.term_offsets, (nterms + 1) * 4);
        put(ts.term_bytes, term_bytes);
        put(ts.posting_off, int64_t(nterms) * 8);
        put(ts.doc_freq, int64_t(nterms) * 4);
        put(ts.n_blocks, int64_t(nterms) * 4);
        put(ts.skip_off, int64_t(nterms) * 8);
        put(ts.skip, blocks * 16);
        put(ts.payload, words * 8);
        put(ts.fieldnorms, N);
    };
    put_text(sevsec, 5, 23, sev_blocks, sev_words);
    put_text(bodysec, BODY_VOCAB, int64_t(BODY_VOCAB) * 6, body_blocks, body_words);
    int64_t meta_off = lay.pos;

    // ---- meta JSON (field order + keys as splitgen.py emits)
    char nb[9600];
    std::string meta;
    meta.reserve(1 << 20);
    auto text_json = [&](const char* name, const char* tok, const char* rec,
                         int64_t total_tokens, int nterms, const TextSecs& s) {
        snprintf(nb, sizeof nb,
                 "{\"fieldnorms\":true,\"name\":\"%s\",\"num_terms\":%d,"
                 "\"record\":\"%s\",\"sec\":{",
                 name, nterms, rec);
        meta += nb;
        json_sec(meta, "doc_freq", s.doc_freq, true);
        json_sec(meta, "fieldnorms", s.fieldnorms, false);
        json_sec(meta, "n_blocks", s.n_blocks, false);
        json_sec(meta, "payload", s.payload, false);
        json_sec(meta, "posting_off", s.posting_off, false);
        json_sec(meta, "skip", s.skip, false);
        json_sec(meta, "skip_off", s.skip_off, false);
        json_sec(meta, "term_bytes", s.term_bytes, false);
        json_sec(meta, "term_offsets", s.term_offsets, false);
        snprintf(nb, sizeof nb,
                 "},\"tokenizer\":\"%s\",\"total_tokens\":%lld,\"type\":\"text\"}",
                 tok, (long long)total_tokens);
        meta += nb;
    };

    // min/max of the numeric fast columns go into the meta JSON, so the
    // columns are generated into temporaries before the buffer exists
    std::vector<int64_t> ts_col(N);
    std::vector<uint64_t> ten_col(N);
    std::vector<uint16_t> tn_col(N);
    // tenant Zipf CDF (u32 thresholds, binary search)
    std::vector<uint32_t> ten_thr(N_TENANTS);
    {
        double H = 0;
        for (int i = 0; i < N_TENANTS; ++i) H += 1.0 / double(i + 1);
        double c = 0;
        for (int i = 0; i < N_TENANTS; ++i) {
            c += (1.0 / double(i + 1)) / H;
            double v = c * 4294967296.0;
            ten_thr[i] = v >= 4294967295.0 ? 0xFFFFFFFFu : uint32_t(v);
        }
        ten_thr[N_TENANTS - 1] = 0xFFFFFFFFu;
    }
    int64_t ts_min = INT64_MAX, ts_max = INT64_MIN;
    uint64_t ten_min = UINT64_MAX, ten_max = 0;
#pragma omp parallel for schedule(dynamic, 1) \
    reduction(min : ts_min) reduction(max : ts_max) \
    reduction(min : ten_min) reduction(max : ten_max)
    for (int64_t c = 0; c < n_chunks; ++c) {
        Rng rts, rtn;
        rts.seed(seed, split_ord, 0x300000000ull + uint64_t(c));
        rtn.seed(seed, split_ord, 0x400000000ull + uint64_t(c));
        int64_t lo = c * DOC_CHUNK, hi = lo + DOC_CHUNK < N ? lo + DOC_CHUNK : N;
        for (int64_t d = lo; d < hi; ++d) {
            uint64_t u = rts.next_u32();
            int64_t ts = (T0_EPOCH_S + int64_t((u * uint64_t(TS_RANGE_S)) >> 32)) * 1000;
            ts_col[d] = ts;
            if (ts < ts_min) ts_min = ts;
            if (ts > ts_max) ts_max = ts;
            uint32_t ut = rtn.next_u32();
            // branchy lower_bound over 1000-entry CDF
            int a = 0, b2 = N_TENANTS - 1;
            while (a < b2) {
                int m = (a + b2) >> 1;
                if (ut < ten_thr[m]) b2 = m;
                else a = m + 1;
            }
            ten_col[d] = uint64_t(a);
            tn_col[d] = uint16_t(a);
            if (uint64_t(a) < ten_min) ten_min = uint64_t(a);
            if (uint64_t(a) > ten_max) ten_max = uint64_t(a);
        }
    }

    // ---- JSON
    meta += "{\"fields\":[";
    snprintf(nb, sizeof nb,
             "{\"max_value\":%lld,\"min_value\":%lld,\"name\":\"timestamp\","
             "\"nullable\":false,\"sec\":{\"values\":[%lld,%lld]},"
             "\"type\":\"datetime\"},",
             (long long)ts_max, (long long)ts_min, (long long)ts_values.off,
             (long long)ts_values.len);
    meta += nb;
    snprintf(nb, sizeof nb,
             "{\"max_value\":%llu,\"min_value\":%llu,\"name\":\"tenant_id\","
             "\"nullable\":false,\"sec\":{\"values\":[%lld,%lld]},"
             "\"type\":\"u64\"},",
             (unsigned long long)ten_max, (unsigned long long)ten_min,
             (long long)ten_values.off, (long long)ten_values.len);
    meta += nb;
    snprintf(nb, sizeof nb,
             "{\"cardinality\":%d,\"name\":\"tenant_name\",\"nullable\":false,"
             "\"ord_width\":2,\"sec\":{\"dict_bytes\":[%lld,%lld],"
             "\"dict_offsets\":[%lld,%lld],\"values\":[%lld,%lld]},"
             "\"type\":\"str\"},",
             N_TENANTS, (long long)tn_dbytes.off, (long long)tn_dbytes.len,
             (long long)tn_doff.off, (long long)tn_doff.len,
             (long long)tn_values.off, (long long)tn_values.len);
    meta += nb;
    text_json("severity_text", "raw", "basic", N, 5, sevsec);
    meta += ",";
    text_json("body", "default", "freq", N * BODY_TOKENS, BODY_VOCAB, bodysec);
    snprintf(nb, sizeof nb,
             "],\"format\":\"QWA1\",\"num_docs\":%lld,"
             "\"split_id\":\"synthetic-%lld-%04lld\","
             "\"timestamp_field\":\"timestamp\",\"version\":2}",
             (long long)N, (long long)seed, (long long)split_ord);
    meta += nb;

    int64_t meta_len = int64_t(meta.size());
    int64_t total = meta_off + meta_len + 24;
    uint8_t* buf = (uint8_t*)calloc(1, size_t(total));
    if (!buf) return -2;

    // ---- fill buffer
    memcpy(buf, "QWAMDSP1", 8);
    memcpy(buf + ts_values.off, ts_col.data(), size_t(N) * 8);
    memcpy(buf + ten_values.off, ten_col.data(), size_t(N) * 8);
    memcpy(buf + tn_values.off, tn_col.data(), size_t(N) * 2);
    ts_col.clear(); ts_col.shrink_to_fit();
    ten_col.clear(); ten_col.shrink_to_fit();
    tn_col.clear(); tn_col.shrink_to_fit();
    {
        uint32_t* doff = (uint32_t*)(buf + tn_doff.off);
        char* db = (char*)(buf + tn_dbytes.off);
        for (int i = 0; i < N_TENANTS; ++i) {
            doff[i] = uint32_t(i * 5);
            char tmp[8];
            snprintf(tmp, sizeof tmp, "t%04d", i);
            memcpy(db + i * 5, tmp, 5);
        }
        doff[N_TENANTS] = uint32_t(N_TENANTS * 5);
    }

    auto fill_text = [&](const TextSecs& s, const std::vector<PackedTerm>& terms,
                         const char** vocab, const uint32_t* voff, int nterms,
                         uint8_t norm_id) {
        uint32_t* toff = (uint32_t*)(buf + s.term_offsets.off);
        char* tb = (char*)(buf + s.term_bytes.off);
        for (int t = 0; t <= nterms; ++t) toff[t] = voff[t];
        for (int t = 0; t < nterms; ++t)
            memcpy(tb + voff[t], vocab[t], voff[t + 1] - voff[t]);
        uint64_t* poff = (uint64_t*)(buf + s.posting_off.off);
        uint32_t* dfq = (uint32_t*)(buf + s.doc_freq.off);
        uint32_t* nbk = (uint32_t*)(buf + s.n_blocks.off);
        uint64_t* skoff = (uint64_t*)(buf + s.skip_off.off);
        std::vector<int64_t> wbase(nterms), bbase(nterms);
        int64_t w = 0, b = 0;
        for (int t = 0; t < nterms; ++t) {
            wbase[t] = w;
            bbase[t] = b;
            dfq[t] = uint32_t(terms[t].df);
            nbk[t] = uint32_t(terms[t].skips.size());
            poff[t] = terms[t].df ? uint64_t(w * 8) : 0;
            skoff[t] = uint64_t(b * 16);
            w += int64_t(terms[t].payload.size());
            b += int64_t(terms[t].skips.size());
        }
#pragma omp parallel for schedule(dynamic, 16)
        for (int t = 0; t < nterms; ++t) {
            const PackedTerm& pt = terms[t];
            if (pt.payload.size())
                memcpy(buf + s.payload.off + wbase[t] * 8, pt.payload.data(),
                       pt.payload.size() * 8);
            Skip* dst = (Skip*)(buf + s.skip.off) + bbase[t];
            uint32_t wadd = uint32_t(wbase[t] * 2);
            for (size_t i = 0; i < pt.skips.size(); ++i) {
                dst[i] = pt.skips[i];
                dst[i].word_off += wadd;
            }
        }
        memset(buf + s.fieldnorms.off, norm_id, size_t(N));
    };
    {
        uint32_t sev_off[6] = {0, 5, 10, 15, 19, 23};
        fill_text(sevsec, sevp, SEV_VOCAB, sev_off, 5, 1);  // norm_to_id(1)=1
        std::vector<std::string> body_vocab_s(BODY_VOCAB);
        std::vector<const char*> body_vocab_p(BODY_VOCAB);
        std::vector<uint32_t> body_voff(BODY_VOCAB + 1);
        for (int t = 0; t < BODY_VOCAB; ++t) {
            char tmp[8];
            snprintf(tmp, sizeof tmp, "w%05d", t);
            body_vocab_s[t] = tmp;
            body_vocab_p[t] = body_vocab_s[t].c_str();
            body_voff[t] = uint32_t(t * 6);
        }
        body_voff[BODY_VOCAB] = uint32_t(BODY_VOCAB * 6);
        fill_text(bodysec, body, body_vocab_p.data(), body_voff.data(),
                  BODY_VOCAB, 10);  // norm_to_id(10)=10
    }
    memcpy(buf + meta_off, meta.data(), meta.size());
    uint64_t foot[2] = {uint64_t(meta_off), uint64_t(meta_len)};
    memcpy(buf + meta_off + meta_len, foot, 16);
    memcpy(buf + meta_off + meta_len + 16, "QWA1FOOT", 8);

    *out = buf;
    *out_len = size_t(total);
    return 0;
}

void qw_gen_free(uint8_t* p) { free(p); }

// explicit worker count (overrides OMP_NUM_THREADS — torchrun pins workers
// to 1 thread, which would serialize 100M-doc generation)
void qw_gen_set_threads(int n) {
    if (n > 0) omp_set_num_threads(n);
}

int64_t qw_gen_split_to_file(int64_t split_ord, int64_t num_docs, int64_t seed,
                             const char* path) {
    uint8_t* buf = nullptr;
    size_t len = 0;
    int64_t rc = qw_gen_split(split_ord, num_docs, seed, &buf, &len);
    if (rc) return rc;
    FILE* f = fopen(path, "wb");
    if (!f) {
        free(buf);
        return -3;
    }
    size_t w = fwrite(buf, 1, len, f);
    fclose(f);
    free(buf);
    return w == len ? 0 : -4;
}

}  // extern "C"
