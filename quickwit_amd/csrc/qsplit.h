// QWA1 split container view: footer/meta parse + typed section pointers.
// Format spec: DESIGN.md §3 (written by quickwit_amd/splitgen.py). This is
// layout plumbing only — posting decode lives in the HIP kernels (product)
// and in oracle/oracle.cpp (restatement), independently.
#pragma once
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#include "minijson.h"

namespace qw {

#pragma pack(push, 1)
struct SkipEntry {  // 16 B, matches splitgen.SKIP_DTYPE
    uint32_t first_doc;
    uint32_t last_doc;
    uint32_t word_off;  // u32-word offset into the field's payload section
    uint8_t id_bits;
    uint8_t tf_bits;  // 0 = no tf section (record: basic)
    uint16_t count;
};
#pragma pack(pop)
static_assert(sizeof(SkipEntry) == 16, "skip entry layout");

struct Section {
    uint64_t off = 0, len = 0;
};

struct TextFieldView {
    std::string name;
    std::string tokenizer;  // "raw" | "default"
    bool record_freq = false;
    bool has_positions = false;  // record: "position" (phrase queries)
    bool has_norms = false;
    uint64_t total_tokens = 0;
    uint32_t num_terms = 0;
    Section term_offsets, term_bytes, posting_off, doc_freq, n_blocks, skip_off, skip,
        payload, fieldnorms, pos_start, positions;
    // host views
    const uint32_t* h_term_offsets = nullptr;
    const uint8_t* h_term_bytes = nullptr;
    const uint64_t* h_posting_off = nullptr;
    const uint32_t* h_doc_freq = nullptr;
    const uint32_t* h_n_blocks = nullptr;
    const uint64_t* h_skip_off = nullptr;
    const SkipEntry* h_skip = nullptr;
    const uint32_t* h_payload = nullptr;
    const uint8_t* h_fieldnorms = nullptr;
    const uint32_t* h_pos_start = nullptr;  // per block: cumulative tf
    const uint32_t* h_positions = nullptr;  // per (posting, occurrence)

    // binary search the sorted term dictionary; -1 if absent
    int64_t find_term(const char* t, size_t tlen) const {
        int64_t lo = 0, hi = int64_t(num_terms) - 1;
        while (lo <= hi) {
            int64_t mid = (lo + hi) >> 1;
            const char* s = (const char*)h_term_bytes + h_term_offsets[mid];
            size_t slen = h_term_offsets[mid + 1] - h_term_offsets[mid];
            int c = memcmp(s, t, slen < tlen ? slen : tlen);
            if (c == 0) c = (slen < tlen) ? -1 : (slen > tlen ? 1 : 0);
            if (c == 0) return mid;
            if (c < 0) lo = mid + 1;
            else hi = mid - 1;
        }
        return -1;
    }
};

struct FastFieldView {
    std::string name;
    // MIXED: a dynamic field whose docs carry several value types.
    // `values` holds f64-monotonic u64 sort keys (bool as 0/1); `tags`
    // (u8: 0=u64 1=i64 2=f64 3=bool) + `raw` (u64 bits) reproduce the
    // original typed value per doc.
    enum Type { U64, I64, DATETIME, STR, F64, MIXED } type = U64;
    bool nullable = false;
    bool multi = false;        // multi-valued str: per-doc ord lists
    bool lower_norm = false;   // str column written through a lowercase
                               // normalizer: query bounds fold too
    uint32_t cardinality = 0;  // str
    int ord_width = 0;         // str: 1/2/4
    int64_t min_value = 0, max_value = 0;
    double fmin = 0, fmax = 0;  // F64
    Section values, nulls, dict_offsets, dict_bytes, value_offsets;
    Section tags, raw;  // MIXED only
    const uint8_t* h_tags = nullptr;
    const uint64_t* h_raw = nullptr;
    const uint32_t* h_val_offsets = nullptr;  // multi: [num_docs+1] prefix
    const void* h_values = nullptr;
    const uint64_t* h_nulls = nullptr;  // bit d set = doc d has a value
    const uint32_t* h_dict_offsets = nullptr;
    const uint8_t* h_dict_bytes = nullptr;

    bool present(uint32_t doc) const {
        if (multi) return h_val_offsets[doc + 1] > h_val_offsets[doc];
        return !nullable || ((h_nulls[doc >> 6] >> (doc & 63)) & 1);
    }
    uint32_t n_vals(uint32_t doc) const {
        return h_val_offsets[doc + 1] - h_val_offsets[doc];
    }
    uint64_t ord_at(uint32_t doc, uint32_t i) const {  // multi
        uint32_t pos = h_val_offsets[doc] + i;
        switch (ord_width) {
            case 1: return ((const uint8_t*)h_values)[pos];
            case 2: return ((const uint16_t*)h_values)[pos];
            default: return ((const uint32_t*)h_values)[pos];
        }
    }
    uint64_t ord(uint32_t doc) const {
        switch (ord_width) {
            case 1: return ((const uint8_t*)h_values)[doc];
            case 2: return ((const uint16_t*)h_values)[doc];
            default: return ((const uint32_t*)h_values)[doc];
        }
    }
    int64_t i64(uint32_t doc) const { return ((const int64_t*)h_values)[doc]; }
    uint8_t mixed_tag(uint32_t doc) const { return h_tags[doc]; }
    uint64_t mixed_raw(uint32_t doc) const { return h_raw[doc]; }
    uint64_t u64(uint32_t doc) const { return ((const uint64_t*)h_values)[doc]; }
    double f64(uint32_t doc) const { return ((const double*)h_values)[doc]; }
    std::string dict_entry(uint64_t o) const {
        return std::string((const char*)h_dict_bytes + h_dict_offsets[o],
                           h_dict_offsets[o + 1] - h_dict_offsets[o]);
    }
    // first ord whose dict entry compares >= s (after=false) or > s
    // (after=true); == cardinality when none
    uint32_t str_bound_ord(const std::string& s, bool after) const {
        uint32_t lo = 0, hi = cardinality;
        while (lo < hi) {
            uint32_t mid = (lo + hi) / 2;
            std::string e = dict_entry(mid);
            bool go_right = after ? (e <= s) : (e < s);
            if (go_right) lo = mid + 1;
            else hi = mid;
        }
        return lo;
    }
};

// optional row store (fetch_docs phase 2): zlib blocks + block index
// (u32 first_doc[n+1] then u64 block_off[n+1]) — splitgen._build_docstore
struct DocStoreView {
    bool present = false;
    uint32_t n_blocks = 0;
    const uint32_t* firsts = nullptr;  // [n_blocks+1], last = num_docs
    const uint64_t* offs = nullptr;    // [n_blocks+1] into blocks
    const uint8_t* blocks = nullptr;
};

struct SplitView {
    std::string split_id;
    uint32_t segment_ord = 0;  // QWA2 multi-segment containers: this
                               // segment's ordinal within the split
    uint32_t version = 1;
    uint32_t num_docs = 0;
    std::string timestamp_field;
    std::vector<TextFieldView> text_fields;
    std::vector<FastFieldView> fast_fields;
    DocStoreView docstore;
    const uint8_t* base = nullptr;  // host copy of the file image
    size_t file_len = 0;

    const TextFieldView* text_field(const std::string& n) const {
        for (auto& f : text_fields)
            if (f.name == n) return &f;
        return nullptr;
    }
    const FastFieldView* fast_field(const std::string& n) const {
        for (auto& f : fast_fields)
            if (f.name == n) return &f;
        return nullptr;
    }

    static Section sec(const mj::Value* m, const char* key, bool required = true) {
        const mj::Value* v = m->get(key);
        if (!v || v->is_null()) {
            if (required) throw std::runtime_error(std::string("missing section ") + key);
            return {};
        }
        return {uint64_t(v->arr.at(0)->as_i64()), uint64_t(v->arr.at(1)->as_i64())};
    }

    void parse(const uint8_t* data, size_t len) {
        base = data;
        file_len = len;
        if (len < 64 + 24 || memcmp(data, "QWAMDSP1", 8) != 0 ||
            memcmp(data + len - 8, "QWA1FOOT", 8) != 0)
            throw std::runtime_error("bad QWA1 container");
        uint64_t meta_off, meta_len;
        memcpy(&meta_off, data + len - 24, 8);
        memcpy(&meta_len, data + len - 16, 8);
        if (meta_off + meta_len > len) throw std::runtime_error("bad QWA1 footer");
        mj::ValuePtr meta = mj::parse((const char*)data + meta_off, meta_len);
        split_id = meta->at("split_id")->s;
        if (const mj::Value* v = meta->get("version")) version = uint32_t(v->as_i64());
        num_docs = uint32_t(meta->at("num_docs")->as_i64());
        const mj::Value* tsf = meta->get("timestamp_field");
        if (tsf && !tsf->is_null()) timestamp_field = tsf->s;
        if (const mj::Value* ds = meta->get("docstore")) {
            docstore.present = true;
            docstore.n_blocks = uint32_t(ds->at("n_blocks")->as_i64());
            Section idx = sec(ds, "doc_index");
            Section blk = sec(ds, "doc_blocks");
            docstore.firsts = (const uint32_t*)(data + idx.off);
            docstore.offs = (const uint64_t*)(data + idx.off +
                                              (docstore.n_blocks + 1) * 4);
            docstore.blocks = data + blk.off;
        }
        for (auto& fv : meta->at("fields")->arr) {
            const mj::Value* f = fv.get();
            std::string ty = f->at("type")->s;
            const mj::Value* s = f->at("sec");
            if (ty == "text") {
                TextFieldView t;
                t.name = f->at("name")->s;
                t.tokenizer = f->at("tokenizer")->s;
                t.record_freq = f->at("record")->s != "basic";
                t.has_positions = f->at("record")->s == "position";
                t.has_norms = f->at("fieldnorms")->b;
                t.total_tokens = uint64_t(f->at("total_tokens")->as_i64());
                t.num_terms = uint32_t(f->at("num_terms")->as_i64());
                t.term_offsets = sec(s, "term_offsets");
                t.term_bytes = sec(s, "term_bytes");
                t.posting_off = sec(s, "posting_off");
                t.doc_freq = sec(s, "doc_freq");
                t.n_blocks = sec(s, "n_blocks");
                t.skip_off = sec(s, "skip_off");
                t.skip = sec(s, "skip");
                t.payload = sec(s, "payload");
                if (t.has_norms) t.fieldnorms = sec(s, "fieldnorms");
                if (t.has_positions) {
                    t.pos_start = sec(s, "pos_start");
                    t.positions = sec(s, "positions");
                }
                t.h_term_offsets = (const uint32_t*)(data + t.term_offsets.off);
                t.h_term_bytes = data + t.term_bytes.off;
                t.h_posting_off = (const uint64_t*)(data + t.posting_off.off);
                t.h_doc_freq = (const uint32_t*)(data + t.doc_freq.off);
                t.h_n_blocks = (const uint32_t*)(data + t.n_blocks.off);
                t.h_skip_off = (const uint64_t*)(data + t.skip_off.off);
                t.h_skip = (const SkipEntry*)(data + t.skip.off);
                t.h_payload = (const uint32_t*)(data + t.payload.off);
                if (t.has_norms) t.h_fieldnorms = data + t.fieldnorms.off;
                if (t.has_positions) {
                    t.h_pos_start = (const uint32_t*)(data + t.pos_start.off);
                    t.h_positions = (const uint32_t*)(data + t.positions.off);
                }
                text_fields.push_back(std::move(t));
            } else {
                FastFieldView ff;
                ff.name = f->at("name")->s;
                ff.type = ty == "u64"   ? FastFieldView::U64
                          : ty == "i64" ? FastFieldView::I64
                          : ty == "datetime" ? FastFieldView::DATETIME
                          : ty == "f64" ? FastFieldView::F64
                          : ty == "mixed" ? FastFieldView::MIXED
                                        : FastFieldView::STR;
                ff.nullable = f->at("nullable")->b;
                ff.values = sec(s, "values");
                if (ff.nullable) ff.nulls = sec(s, "nulls");
                if (ff.type == FastFieldView::STR) {
                    const mj::Value* nrm = f->get("normalizer");
                    ff.lower_norm = nrm && nrm->s == "lowercase";
                    const mj::Value* mv = f->get("multi");
                    ff.multi = mv && mv->b;
                    if (ff.multi) {
                        ff.value_offsets = sec(s, "value_offsets");
                        ff.h_val_offsets =
                            (const uint32_t*)(data + ff.value_offsets.off);
                    }
                    ff.cardinality = uint32_t(f->at("cardinality")->as_i64());
                    ff.ord_width = int(f->at("ord_width")->as_i64());
                    ff.dict_offsets = sec(s, "dict_offsets");
                    ff.dict_bytes = sec(s, "dict_bytes");
                    ff.h_dict_offsets = (const uint32_t*)(data + ff.dict_offsets.off);
                    ff.h_dict_bytes = data + ff.dict_bytes.off;
                } else if (ff.type == FastFieldView::MIXED) {
                    ff.tags = sec(s, "tags");
                    ff.raw = sec(s, "raw");
                    ff.h_tags = data + ff.tags.off;
                    ff.h_raw = (const uint64_t*)(data + ff.raw.off);
                } else if (ff.type == FastFieldView::F64) {
                    const mj::Value* mn = f->get("min_value");
                    const mj::Value* mx = f->get("max_value");
                    if (mn) ff.fmin = mn->num();
                    if (mx) ff.fmax = mx->num();
                } else {
                    const mj::Value* mn = f->get("min_value");
                    const mj::Value* mx = f->get("max_value");
                    if (mn) ff.min_value = mn->as_i64();
                    if (mx) ff.max_value = mx->as_i64();
                }
                ff.h_values = data + ff.values.off;
                if (ff.nullable) ff.h_nulls = (const uint64_t*)(data + ff.nulls.off);
                fast_fields.push_back(std::move(ff));
            }
        }
    }
};

}  // namespace qw
