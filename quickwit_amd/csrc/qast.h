// QueryAst JSON -> normalized query plan, + doc-mapper schema parse, +
// aggregation-request parse (ES-style aggs JSON subset).
//
// Semantics restated from the reference:
//  - QueryAst serde model: quickwit-query/src/query_ast/mod.rs:57-77
//    (#[serde(tag="type", rename_all="snake_case")]).
//  - full_text -> per-token term disjunction/conjunction with
//    IndexRecordOption::WithFreqs: full_text_query.rs:103-137.
//  - bool -> TantivyBoolQuery: tantivy_query_ast.rs:153-374; default
//    minimum_should_match semantics: with no must/filter clause at least one
//    should must match; an explicit minimum_should_match overrides
//    (bool_query.rs:34-45, tantivy_query_ast.rs:372).
//  - range over fast fields: range_query.rs:30-158 (Bound<JsonLiteral> with
//    Rust serde form {"included":..}/{"excluded":..}/"unbounded" — serde
//    renames Bound variants to snake_case inside quickwit's JSON).
//  - tokenizers: "raw" = one verbatim token; "default" = alphanumeric runs,
//    lowercased, tokens > 40 chars removed (tantivy default analyzer).
// The shared location (product + oracle both build plans from it) is a
// declared plumbing exception — results are pinned end-to-end by the golden
// suites.
#pragma once
#include <algorithm>
#include <cctype>
#include <cstdint>
#include <cstdio>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

#include "qw_unicode.h"

#include "minijson.h"

namespace qw {

// ---------------------------------------------------------------- schema
struct SchemaField {
    std::string name;
    std::string type;       // text | u64 | i64 | datetime | str
    std::string tokenizer;  // text only
    bool record_freq = false;
    bool fast = false;
};

struct Schema {
    std::vector<SchemaField> fields;
    std::vector<std::string> default_search_fields;
    std::string timestamp_field;

    const SchemaField* field(const std::string& n) const {
        for (auto& f : fields)
            if (f.name == n) return &f;
        return nullptr;
    }

    // Accepts this repo's schema JSON (splitgen.HDFS_SCHEMA shape):
    // {"timestamp_field": ..., "default_search_fields": [...]?, "fields":
    //  [{"name","type","tokenizer"?,"record"?,"fast"?}...]}
    static Schema parse(const std::string& json) {
        Schema s;
        mj::ValuePtr root = mj::parse(json);
        const mj::Value* tf = root->get("timestamp_field");
        if (tf && !tf->is_null()) s.timestamp_field = tf->s;
        const mj::Value* dsf = root->get("default_search_fields");
        if (dsf)
            for (auto& v : dsf->arr) s.default_search_fields.push_back(v->s);
        for (auto& fv : root->at("fields")->arr) {
            SchemaField f;
            f.name = fv->at("name")->s;
            f.type = fv->at("type")->s;
            const mj::Value* tok = fv->get("tokenizer");
            f.tokenizer = tok ? tok->s : "default";
            const mj::Value* rec = fv->get("record");
            f.record_freq = rec && (rec->s == "freq" || rec->s == "position");
            const mj::Value* fast = fv->get("fast");
            f.fast = fast && fast->b;
            s.fields.push_back(std::move(f));
        }
        return s;
    }
};

// UTF-8-aware default tokenizer; alnum/lowercase from the GENERATED
// shared tables (qw_unicode.h == quickwit_amd/unicode_tables.py) so the
// query side agrees with the split writer on every codepoint. A malformed
// UTF-8 byte is treated as a token separator.
inline void qw_utf8_append(std::string& s, uint32_t cp) {
    if (cp < 0x80) s.push_back(char(cp));
    else if (cp < 0x800) {
        s.push_back(char(0xC0 | (cp >> 6)));
        s.push_back(char(0x80 | (cp & 0x3F)));
    } else if (cp < 0x10000) {
        s.push_back(char(0xE0 | (cp >> 12)));
        s.push_back(char(0x80 | ((cp >> 6) & 0x3F)));
        s.push_back(char(0x80 | (cp & 0x3F)));
    } else {
        s.push_back(char(0xF0 | (cp >> 18)));
        s.push_back(char(0x80 | ((cp >> 12) & 0x3F)));
        s.push_back(char(0x80 | ((cp >> 6) & 0x3F)));
        s.push_back(char(0x80 | (cp & 0x3F)));
    }
}

// UTF-8-aware lowercase with the same shared table (wildcard patterns,
// case_insensitive folds); malformed bytes pass through unchanged
inline std::string qw_utf8_lower(const std::string& text) {
    std::string out;
    out.reserve(text.size());
    size_t i = 0, n = text.size();
    while (i < n) {
        unsigned char c = (unsigned char)text[i];
        if (c < 0x80) {
            out.push_back(char(std::tolower(c)));
            ++i;
            continue;
        }
        uint32_t cp = 0;
        size_t len = 0;
        if ((c & 0xE0) == 0xC0) { cp = c & 0x1F; len = 2; }
        else if ((c & 0xF0) == 0xE0) { cp = c & 0x0F; len = 3; }
        else if ((c & 0xF8) == 0xF0) { cp = c & 0x07; len = 4; }
        bool ok = len > 0 && i + len <= n;
        for (size_t k = 1; ok && k < len; ++k) {
            unsigned char cc = (unsigned char)text[i + k];
            if ((cc & 0xC0) != 0x80) ok = false;
            else cp = (cp << 6) | (cc & 0x3F);
        }
        if (!ok) { out.push_back(char(c)); ++i; continue; }
        qw_utf8_append(out, qw_lower_cp(cp));
        i += len;
    }
    return out;
}

inline std::vector<std::string> tokenize(const std::string& text,
                                         const std::string& tokenizer) {
    if (tokenizer == "raw") {
        if (text.empty()) return {};
        return {text};
    }
    std::vector<std::string> out;
    std::string cur;
    auto flush = [&]() {
        if (!cur.empty()) {
            if (cur.size() <= 40) out.push_back(cur);
            cur.clear();
        }
    };
    size_t i = 0, n = text.size();
    while (i < n) {
        unsigned char c = (unsigned char)text[i];
        if (c < 0x80) {
            if (std::isalnum(c)) cur.push_back(char(std::tolower(c)));
            else flush();
            ++i;
            continue;
        }
        // decode one UTF-8 sequence (2-4 bytes)
        uint32_t cp = 0;
        size_t len = 0;
        if ((c & 0xE0) == 0xC0) { cp = c & 0x1F; len = 2; }
        else if ((c & 0xF0) == 0xE0) { cp = c & 0x0F; len = 3; }
        else if ((c & 0xF8) == 0xF0) { cp = c & 0x07; len = 4; }
        bool ok = len > 0 && i + len <= n;
        for (size_t k = 1; ok && k < len; ++k) {
            unsigned char cc = (unsigned char)text[i + k];
            if ((cc & 0xC0) != 0x80) ok = false;
            else cp = (cp << 6) | (cc & 0x3F);
        }
        if (!ok) { flush(); ++i; continue; }
        if (qw_is_alnum_cp(cp)) qw_utf8_append(cur, qw_lower_cp(cp));
        else flush();
        i += len;
    }
    flush();
    return out;
}

// ---------------------------------------------------------------- plan
struct Bound {
    enum Kind { UNBOUNDED, INCLUDED, EXCLUDED } kind = UNBOUNDED;
    int64_t ival = 0;    // canonical value for u64/i64/datetime(ms) columns
    bool from_f64 = false;
    double fval = 0;
    std::string sval;    // str columns: lexicographic bound (ord-mapped)
};

struct PlanNode {
    enum Kind {
        MATCH_ALL, MATCH_NONE, TERM, BOOL, RANGE, FIELD_PRESENCE, WILDCARD,
        CACHE, PHRASE
    } kind = MATCH_ALL;
    // TERM / WILDCARD (value = glob pattern: '*' any run, '?' one char)
    std::string field;
    std::string value;
    bool ci = false;  // WILDCARD case_insensitive
    // BOOL
    std::vector<PlanNode> must, must_not, should, filter;
    int64_t minimum_should_match = -1;  // -1 = unset
    // RANGE / FIELD_PRESENCE
    Bound lo, hi;
    float boost = 1.0f;
    // term_set / wildcard match the union's DOC SET but score const 1.0
    // (tantivy TermSetQuery / AutomatonQuery const scorer) — the build
    // rejects these under _score sorting instead of mis-scoring them
    bool const_score = false;
    // CACHE: the wrapped subtree (cache_node.rs CacheNode.inner); evaluated
    // via a device-resident HitSet bitmap when in filter position
    std::vector<PlanNode> cache_inner;  // size <= 1
    // PHRASE: consecutive tokens (slop 0 — full_text_query.rs:113-137 phrase
    // mode; needs record: position). Matching is unscored (const-score
    // semantics like term_set/wildcard).
    std::vector<std::string> phrase_toks;
};

// stable fingerprint of a plan subtree — the (split, subquery) key of the
// device HitSet cache (the analog of CacheNode's PredicateCache key,
// cache_node.rs:202-486)
inline void plan_fingerprint(const PlanNode& n, std::string& out) {
    out += char('A' + int(n.kind));
    out += n.field;
    out += '\x1f';
    out += n.value;
    out += '\x1f';
    for (const std::string& t : n.phrase_toks) {
        out += t;
        out += '\x1c';
    }
    char buf[96];
    snprintf(buf, sizeof buf, "%d:%lld:%d:%lld:%lld:%.9g:%d|",
             int(n.lo.kind), (long long)n.lo.ival, int(n.hi.kind),
             (long long)n.hi.ival, (long long)n.minimum_should_match,
             double(n.boost), int(n.ci));
    out += buf;
    for (auto* v : {&n.must, &n.must_not, &n.should, &n.filter, &n.cache_inner}) {
        out += '[';
        for (auto& c : *v) plan_fingerprint(c, out);
        out += ']';
    }
}

// does any node require const-score semantics (term_set / wildcard)?
inline bool plan_has_const_score(const PlanNode& n) {
    if (n.const_score || n.kind == PlanNode::PHRASE) return true;
    for (auto* v : {&n.must, &n.must_not, &n.should, &n.filter, &n.cache_inner})
        for (auto& c : *v)
            if (plan_has_const_score(c)) return true;
    return false;
}

// decode one UTF-8 codepoint at byte index i (advances i); a malformed
// lead byte yields a per-byte sentinel so it only equals itself
inline uint32_t qw_utf8_next(const char* s, size_t n, size_t& i) {
    unsigned char c = (unsigned char)s[i];
    if (c < 0x80) { ++i; return c; }
    uint32_t cp = 0;
    size_t len = 0;
    if ((c & 0xE0) == 0xC0) { cp = c & 0x1F; len = 2; }
    else if ((c & 0xF0) == 0xE0) { cp = c & 0x0F; len = 3; }
    else if ((c & 0xF8) == 0xF0) { cp = c & 0x07; len = 4; }
    if (!len || i + len > n) { ++i; return 0x80000000u | c; }
    for (size_t k = 1; k < len; ++k) {
        unsigned char cc = (unsigned char)s[i + k];
        if ((cc & 0xC0) != 0x80) { ++i; return 0x80000000u | c; }
        cp = (cp << 6) | (cc & 0x3F);
    }
    i += len;
    return cp;
}

// glob match with '*' (any run) and '?' (single CODEPOINT — the reference
// compiles wildcards to a regex over chars, wildcard_query.rs), iterative
// backtracking over UTF-8; case_insensitive folds through the shared
// lowercase table on both sides
inline bool glob_match(const char* s, size_t sn, const char* p, size_t pn, bool ci) {
    auto fold = [&](uint32_t cp) -> uint32_t {
        if (!ci) return cp;
        if (cp < 0x80) return uint32_t(std::tolower(int(cp)));
        return cp < 0x80000000u ? qw_lower_cp(cp) : cp;
    };
    size_t si = 0, pi = 0, star_p = size_t(-1), star_s = 0;
    while (si < sn) {
        if (pi < pn && p[pi] == '*') {
            star_p = ++pi;
            star_s = si;
            continue;
        }
        if (pi < pn) {
            size_t pj = pi, sj = si;
            if (p[pi] == '?') {
                ++pj;
                qw_utf8_next(s, sn, sj);
                pi = pj;
                si = sj;
                continue;
            }
            uint32_t pc = qw_utf8_next(p, pn, pj);
            uint32_t sc = qw_utf8_next(s, sn, sj);
            if (fold(pc) == fold(sc)) {
                pi = pj;
                si = sj;
                continue;
            }
        }
        if (star_p != size_t(-1)) {
            qw_utf8_next(s, sn, star_s);  // widen the '*' by one codepoint
            si = star_s;
            pi = star_p;
            continue;
        }
        return false;
    }
    while (pi < pn && p[pi] == '*') ++pi;
    return pi == pn;
}

inline int64_t parse_datetime_ms(const mj::Value* lit);

// RFC3339 (subset: YYYY-MM-DD['T'HH:MM[:SS[.fff]]][Z|±hh:mm]) -> epoch ms
inline int64_t rfc3339_to_ms(const std::string& s);

inline Bound parse_bound(const mj::Value* v, const SchemaField& f,
                         bool is_upper = false) {
    Bound b;
    if (!v) return b;
    // serde's built-in std::ops::Bound impl: {"Included": lit} /
    // {"Excluded": lit} / "Unbounded" (capitalized; lowercase also accepted)
    const mj::Value* lit = nullptr;
    if (v->kind == mj::Value::STR && (v->s == "Unbounded" || v->s == "unbounded")) return b;
    const mj::Value* inc = v->get("Included");
    if (!inc) inc = v->get("included");
    const mj::Value* exc = v->get("Excluded");
    if (!exc) exc = v->get("excluded");
    if (inc) {
        b.kind = Bound::INCLUDED;
        lit = inc;
    } else if (exc) {
        b.kind = Bound::EXCLUDED;
        lit = exc;
    } else throw std::runtime_error("bad range bound");
    if (f.type == "str" || f.type == "text") {
        // str fast columns (incl. the raw fast column of a text+fast
        // field): lexicographic bound over the sorted ord dictionary
        if (lit->kind != mj::Value::STR)
            throw std::runtime_error("str range bound must be a string");
        b.sval = lit->s;
        return b;
    }
    if (f.type == "datetime") {
        b.ival = parse_datetime_ms(lit);
    } else if (lit->kind == mj::Value::INT) {
        b.ival = lit->i;
        b.fval = double(lit->i);
    } else if (lit->kind == mj::Value::DBL) {
        b.from_f64 = true;
        b.fval = lit->d;
        b.ival = int64_t(lit->d);
    } else if (lit->kind == mj::Value::STR) {
        b.ival = strtoll(lit->s.c_str(), nullptr, 10);
        b.fval = atof(lit->s.c_str());
    } else throw std::runtime_error("bad range literal");
    if (f.type == "u64" && (b.ival < 0 || (b.from_f64 && b.fval < 0))) {
        // negative bound against an unsigned column (tantivy coerces at
        // the column's domain edge): lower clamps to 0 (all docs satisfy
        // it), upper becomes the empty range x < 0
        if (is_upper) {
            b.kind = Bound::EXCLUDED;
            b.ival = 0;
            b.fval = 0;
            b.from_f64 = false;
        } else {
            b.kind = Bound::INCLUDED;
            b.ival = 0;
            b.fval = 0;
            b.from_f64 = false;
        }
    }
    return b;
}

// TERM leaf over any schema field: text fields use the inverted index;
// fast-only columns become an equality range over the fast field (the
// reference routes such terms through the fast-field machinery).
inline PlanNode term_leaf_plan(const std::string& field,
                               const std::string& value,
                               const Schema& schema) {
    const SchemaField* tf = schema.field(field);
    if (!tf) throw std::runtime_error("term on unknown field: " + field);
    PlanNode n;
    if (tf->type == "text") {
        n.kind = PlanNode::TERM;
        n.field = field;
        n.value = value;
        return n;
    }
    PlanNode r;
    r.kind = PlanNode::RANGE;
    r.field = field;
    Bound b;
    b.kind = Bound::INCLUDED;
    if (tf->type == "str") {
        b.sval = value;
    } else if (tf->type == "bool") {
        // doc-mapper bool: u64 0/1 column (splitgen stores bools that way)
        if (value == "true") b.ival = 1;
        else if (value == "false") b.ival = 0;
        else
            throw std::runtime_error(
                "invalid term value for bool field " + field);
        b.fval = double(b.ival);
    } else if (tf->type == "datetime") {
        mj::Value lit;
        lit.kind = mj::Value::STR;
        lit.s = value;
        b.ival = parse_datetime_ms(&lit);
    } else {
        char* endp = nullptr;
        errno = 0;
        if (tf->type == "f64") {
            b.from_f64 = true;
            b.fval = strtod(value.c_str(), &endp);
            b.ival = int64_t(b.fval);
        } else {
            long long v = strtoll(value.c_str(), &endp, 10);
            b.ival = v;
            b.fval = double(v);
        }
        if (errno != 0 || !endp || *endp != 0)
            throw std::runtime_error(
                "invalid term value for numeric field " + field);
        if (tf->type == "u64" && b.ival < 0) {
            PlanNode none;
            none.kind = PlanNode::MATCH_NONE;
            return none;
        }
    }
    r.lo = b;
    r.hi = b;
    return r;
}

inline PlanNode build_plan(const mj::Value* ast, const Schema& schema);

inline PlanNode full_text_plan(const std::string& field, const std::string& text,
                               const std::string& op, const Schema& schema,
                               bool zero_terms_all = false) {
    const SchemaField* f = schema.field(field);
    if (!f || f->type != "text")
        throw std::runtime_error("full_text on unknown/non-text field: " + field);
    std::vector<std::string> toks = tokenize(text, f->tokenizer);
    if (toks.empty()) {
        PlanNode n;
        // zero_terms_query: none (default) | all (full_text_query.rs params)
        n.kind = zero_terms_all ? PlanNode::MATCH_ALL : PlanNode::MATCH_NONE;
        return n;
    }
    if (toks.size() == 1) {
        PlanNode n;
        n.kind = PlanNode::TERM;
        n.field = field;
        n.value = toks[0];
        return n;
    }
    PlanNode b;
    b.kind = PlanNode::BOOL;
    for (auto& t : toks) {
        PlanNode n;
        n.kind = PlanNode::TERM;
        n.field = field;
        n.value = t;
        (op == "and" ? b.must : b.should).push_back(std::move(n));
    }
    return b;
}

// multi-token phrase: consecutive positions, slop 0 (PhraseQuery,
// full_text_query.rs phrase mode). Single-token phrases degrade to TERM;
// zero tokens follow zero_terms_query like full_text_plan.
inline PlanNode phrase_plan(const std::string& field, const std::string& text,
                            const Schema& schema, bool zero_terms_all = false) {
    const SchemaField* f = schema.field(field);
    if (!f || f->type != "text")
        throw std::runtime_error("phrase on unknown/non-text field: " + field);
    std::vector<std::string> toks = tokenize(text, f->tokenizer);
    PlanNode n;
    if (toks.empty()) {
        n.kind = zero_terms_all ? PlanNode::MATCH_ALL : PlanNode::MATCH_NONE;
        return n;
    }
    if (toks.size() == 1) {
        n.kind = PlanNode::TERM;
        n.field = field;
        n.value = toks[0];
        return n;
    }
    n.kind = PlanNode::PHRASE;
    n.field = field;
    n.phrase_toks = std::move(toks);
    return n;
}

// minimal user_input support: whitespace-separated clauses of `field:token`
// or bare tokens over default fields, implicit OR (the subset quickwit's
// query parser covers for plain log searches; anything else -> error =
// SearchError::InvalidQuery, never a silent wrong answer)
// ------------------------------------------------ query-language parser
// Restates the tantivy query grammar subset quickwit's user_input queries
// use (tantivy-query-grammar 0.26, external — pinned by the qw_search_api
// golden scenarios): clause sequences with +/-/NOT prefixes and AND/OR
// connectors, parentheses, `*`, field:*, field:value, field:"quoted",
// field:[a TO b] / {a TO b}, field:>=v >v <=v <v, field:IN [a b], ^boost.
// Out of scope r1: regex //, slop ~N, multi-token phrases (need positions).
struct QTok {
    enum Kind { LPAREN, RPAREN, AND, OR, NOT, PLUS, MINUS, LIT } kind = LIT;
    std::string field;   // LIT: optional field ("" = default fields)
    std::string text;    // LIT: raw literal text (quotes stripped)
    bool quoted = false;
    bool no_glob = false;  // backslash-escaped * or ? (literal, not glob)
    bool prefix = false;  // quoted phrase followed by '*' (phrase prefix)
    bool range_ge = false, range_gt = false, range_le = false, range_lt = false;
    bool bracket = false;  // [a TO b] / {a TO b}
    bool lo_incl = true, hi_incl = true;
    std::string lo, hi;    // bracket bounds ("*" = unbounded)
    bool set = false;      // IN [..]
    std::vector<std::string> set_vals;
    float boost = 1.0f;
};

inline std::vector<QTok> qlex(const std::string& s) {
    std::vector<QTok> out;
    size_t i = 0;
    auto read_quoted = [&](std::string* t, char q) {
        ++i;  // opening quote
        while (i < s.size() && s[i] != q) t->push_back(s[i++]);
        if (i >= s.size()) throw std::runtime_error("unterminated quote");
        ++i;
    };
    auto after_quote_star = [&]() {  // "phrase"* -> prefix
        if (i < s.size() && s[i] == '*') {
            ++i;
            return true;
        }
        return false;
    };
    while (i < s.size()) {
        char c = s[i];
        if (c == ' ' || c == '\t') { ++i; continue; }
        if (c == '(') { out.push_back({QTok::LPAREN}); ++i; continue; }
        if (c == ')') {
            QTok t{QTok::RPAREN};
            ++i;
            out.push_back(t);
            continue;
        }
        if (c == '+') { out.push_back({QTok::PLUS}); ++i; continue; }
        if (c == '-') { out.push_back({QTok::MINUS}); ++i; continue; }
        // word or field:...
        QTok t;
        std::string word;
        if (c == '"' || c == '\'') {
            read_quoted(&t.text, c);
            t.quoted = true;
            t.prefix = after_quote_star();
        } else {
            bool esc = false;
            while (i < s.size() && s[i] != ' ' && s[i] != '\t' && s[i] != '(' &&
                   s[i] != ')' && s[i] != ':' && s[i] != '^') {
                if (s[i] == '\\' && i + 1 < s.size()) {
                    esc = true;
                    ++i;  // escaped char, taken verbatim
                }
                word.push_back(s[i++]);
            }
            t.no_glob = esc;
            if (i < s.size() && s[i] == ':') {
                t.field = word;
                ++i;
                // value part
                if (i < s.size() && (s[i] == '"' || s[i] == '\'')) {
                    read_quoted(&t.text, s[i]);
                    t.quoted = true;
                    t.prefix = after_quote_star();
                } else if (i < s.size() && (s[i] == '[' || s[i] == '{')) {
                    t.bracket = true;
                    t.lo_incl = s[i] == '[';
                    ++i;
                    auto piece = [&]() {
                        // no quoting/escaping inside range bounds — the
                        // reference grammar rejects those forms (0004 golden
                        // cases expect 400)
                        std::string p;
                        while (i < s.size() && s[i] != ' ' && s[i] != ']' &&
                               s[i] != '}')
                            p.push_back(s[i++]);
                        return p;
                    };
                    t.lo = piece();
                    while (i < s.size() && s[i] == ' ') ++i;
                    std::string to = piece();
                    if (to != "TO") throw std::runtime_error("range needs TO");
                    while (i < s.size() && s[i] == ' ') ++i;
                    t.hi = piece();
                    if (i >= s.size() || (s[i] != ']' && s[i] != '}'))
                        throw std::runtime_error("unterminated range");
                    t.hi_incl = s[i] == ']';
                    ++i;
                } else if (i + 1 < s.size() && s[i] == '>' && s[i + 1] == '=') {
                    t.range_ge = true;
                    i += 2;
                    while (i < s.size() && s[i] != ' ') t.text.push_back(s[i++]);
                } else if (i + 1 < s.size() && s[i] == '<' && s[i + 1] == '=') {
                    t.range_le = true;
                    i += 2;
                    while (i < s.size() && s[i] != ' ') t.text.push_back(s[i++]);
                } else if (i < s.size() && s[i] == '>') {
                    t.range_gt = true;
                    ++i;
                    while (i < s.size() && s[i] != ' ') t.text.push_back(s[i++]);
                } else if (i < s.size() && s[i] == '<') {
                    t.range_lt = true;
                    ++i;
                    while (i < s.size() && s[i] != ' ') t.text.push_back(s[i++]);
                } else {
                    // plain value or IN [..]
                    std::string v;
                    while (i < s.size() && s[i] != ' ' && s[i] != ')' && s[i] != '^') {
                        if (s[i] == '\\' && i + 1 < s.size()) {
                            t.no_glob = true;
                            ++i;
                        }
                        v.push_back(s[i++]);
                    }
                    if (v == "IN") {
                        while (i < s.size() && s[i] == ' ') ++i;
                        if (i >= s.size() || s[i] != '[')
                            throw std::runtime_error("IN needs [..]");
                        ++i;
                        t.set = true;
                        std::string cur;
                        while (i < s.size() && s[i] != ']') {
                            if (s[i] == ' ') {
                                if (!cur.empty()) t.set_vals.push_back(cur);
                                cur.clear();
                            } else cur.push_back(s[i]);
                            ++i;
                        }
                        if (!cur.empty()) t.set_vals.push_back(cur);
                        if (i >= s.size()) throw std::runtime_error("unterminated IN");
                        ++i;
                    } else t.text = v;
                }
            } else {
                t.text = word;
                if (word == "AND") { out.push_back({QTok::AND}); continue; }
                if (word == "OR") { out.push_back({QTok::OR}); continue; }
                if (word == "NOT") { out.push_back({QTok::NOT}); continue; }
            }
        }
        if (i < s.size() && s[i] == '^') {
            ++i;
            std::string b;
            while (i < s.size() && s[i] != ' ') b.push_back(s[i++]);
            t.boost = float(atof(b.c_str()));
        }
        out.push_back(std::move(t));
    }
    return out;
}

inline PlanNode qtok_leaf(const QTok& t, const std::vector<std::string>& dfs,
                          const Schema& schema);

// string literal -> typed range bound for one schema field
inline Bound qbound(const std::string& text, Bound::Kind kind, const SchemaField& f) {
    Bound b;
    if (text.empty() || text == "*") return b;
    b.kind = kind;
    if (f.type == "bool") {
        if (text == "true") b.ival = 1;
        else if (text == "false") b.ival = 0;
        else throw std::runtime_error("invalid bool literal: " + text);
        b.fval = double(b.ival);
    } else if (f.type == "datetime") {
        mj::Value v;
        v.kind = mj::Value::STR;
        v.s = text;
        b.ival = parse_datetime_ms(&v);
    } else if (f.type == "str" || f.type == "text") {
        b.sval = text;
    } else if (f.type == "f64") {
        b.fval = atof(text.c_str());
        b.ival = int64_t(b.fval);
        b.from_f64 = true;
    } else {
        b.ival = strtoll(text.c_str(), nullptr, 10);
        b.fval = double(b.ival);
    }
    return b;
}

// parse a clause sequence (until RPAREN/end). The occur model restated from
// tantivy's grammar: leading +/-/NOT set Must/MustNot on the next leaf; an
// AND connector promotes BOTH neighbors to Must, OR keeps Should.
inline PlanNode qparse_clause(const std::vector<QTok>& toks, size_t* pos,
                              const std::vector<std::string>& dfs,
                              const Schema& schema, int depth = 0) {
    // paren nesting capped like the JSON parser (a pathological query is
    // an error, never a stack overflow; tests/test_abi.py)
    if (depth > 128)
        throw std::runtime_error(
            "query grammar: recursion limit exceeded");
    struct Item {
        int occur = 0;  // 0 default(Should), 1 Must, 2 MustNot
        PlanNode node;
    };
    std::vector<Item> items;
    bool saw_and = false;
    bool dangling = false;  // connector awaiting its right operand
    int pending = 0;
    while (*pos < toks.size()) {
        const QTok& t = toks[*pos];
        if (t.kind == QTok::RPAREN) { ++(*pos); break; }
        if (t.kind == QTok::AND) { saw_and = true; dangling = true; ++(*pos); continue; }
        if (t.kind == QTok::OR) { dangling = true; ++(*pos); continue; }
        if (t.kind == QTok::PLUS) { pending = 1; ++(*pos); continue; }
        if (t.kind == QTok::MINUS || t.kind == QTok::NOT) {
            pending = 2;
            ++(*pos);
            continue;
        }
        Item it;
        it.occur = pending;
        pending = 0;
        if (t.kind == QTok::LPAREN) {
            ++(*pos);
            it.node = qparse_clause(toks, pos, dfs, schema, depth + 1);
        } else {
            it.node = qtok_leaf(t, dfs, schema);
            ++(*pos);
        }
        items.push_back(std::move(it));
        dangling = false;
    }
    if (dangling)
        throw std::runtime_error(
            "query grammar: operator without right operand");
    if (items.empty()) {
        PlanNode n;
        n.kind = PlanNode::MATCH_ALL;
        return n;
    }
    if (items.size() == 1 && items[0].occur == 0) return std::move(items[0].node);
    PlanNode b;
    b.kind = PlanNode::BOOL;
    for (Item& it : items) {
        int occur = it.occur ? it.occur : (saw_and ? 1 : 0);
        if (occur == 1) b.must.push_back(std::move(it.node));
        else if (occur == 2) b.must_not.push_back(std::move(it.node));
        else b.should.push_back(std::move(it.node));
    }
    if (items.size() == 1 && items[0].occur == 2) {
        // single NOT clause: implicit match_all handled downstream
    }
    return b;
}

inline PlanNode qtok_leaf_impl(const QTok& t, const std::vector<std::string>& dfs_in,
                               const Schema& schema) {
    const std::vector<std::string>& dfs =
        dfs_in.empty() ? schema.default_search_fields : dfs_in;
    PlanNode n;
    bool is_range = t.bracket || t.range_ge || t.range_gt || t.range_le || t.range_lt;
    if (t.field.empty() && !t.quoted && t.text == "*") {
        n.kind = PlanNode::MATCH_ALL;
        return n;
    }
    if (!t.field.empty() && !t.quoted && !is_range && !t.set && t.text == "*") {
        // field:* -> Exists; an object/json parent is present iff any dotted
        // subfield is (index_field_presence semantics over expand_dots)
        if (!schema.field(t.field)) {
            PlanNode b;
            b.kind = PlanNode::BOOL;
            std::string prefix = t.field + ".";
            for (auto& f : schema.fields)
                if (f.name.compare(0, prefix.size(), prefix) == 0) {
                    PlanNode c;
                    c.kind = PlanNode::FIELD_PRESENCE;
                    c.field = f.name;
                    b.should.push_back(std::move(c));
                }
            if (!b.should.empty()) {
                b.boost = t.boost;
                return b;
            }
        }
        n.kind = PlanNode::FIELD_PRESENCE;
        n.field = t.field;
        n.boost = t.boost;
        return n;
    }
    if (t.set) {
        // field:IN [a b] -> TermSetQuery (user_input_query.rs:158-176)
        n.kind = PlanNode::BOOL;
        n.const_score = true;
        std::vector<std::string> fields =
            t.field.empty() ? dfs : std::vector<std::string>{t.field};
        if (fields.empty()) throw std::runtime_error("set query needs a field");
        for (auto& f : fields)
            for (auto& v : t.set_vals) {
                PlanNode c;
                c.kind = PlanNode::TERM;
                c.field = f;
                c.value = v;
                n.should.push_back(std::move(c));
            }
        if (n.should.empty()) n.kind = PlanNode::MATCH_NONE;
        n.boost = t.boost;
        return n;
    }
    if (is_range) {
        if (t.field.empty())
            throw std::runtime_error("range query without field is not supported");
        const SchemaField* f = schema.field(t.field);
        if (!f) throw std::runtime_error("range on unknown field: " + t.field);
        n.kind = PlanNode::RANGE;
        n.field = t.field;
        if (t.bracket) {
            n.lo = qbound(t.lo, t.lo_incl ? Bound::INCLUDED : Bound::EXCLUDED, *f);
            n.hi = qbound(t.hi, t.hi_incl ? Bound::INCLUDED : Bound::EXCLUDED, *f);
        } else {
            if (t.range_ge) n.lo = qbound(t.text, Bound::INCLUDED, *f);
            if (t.range_gt) n.lo = qbound(t.text, Bound::EXCLUDED, *f);
            if (t.range_le) n.hi = qbound(t.text, Bound::INCLUDED, *f);
            if (t.range_lt) n.hi = qbound(t.text, Bound::EXCLUDED, *f);
        }
        n.boost = t.boost;
        return n;
    }
    if (!t.field.empty()) {
        const SchemaField* f = schema.field(t.field);
        if (f && f->type != "text") {
            // term on a typed fast column -> point range (the reference's
            // FastFieldRangeQuery translation for fast-only fields)
            n.kind = PlanNode::RANGE;
            n.field = t.field;
            n.lo = qbound(t.text, Bound::INCLUDED, *f);
            n.hi = qbound(t.text, Bound::INCLUDED, *f);
            n.boost = t.boost;
            return n;
        }
        if (t.quoted) {
            // phrase: tokenize with the field's tokenizer; single-token
            // phrases (e.g. raw tokenizer keeps spaces) are plain terms,
            // multi-token phrases match consecutive positions (slop 0)
            std::vector<std::string> toks =
                tokenize(t.text, f ? f->tokenizer : "default");
            if (toks.empty()) {
                n.kind = PlanNode::MATCH_NONE;
                return n;
            }
            if (t.prefix) {
                // "phrase"* phrase-prefix: supported where it reduces to a
                // single-token prefix (e.g. raw tokenizer), as a wildcard
                if (toks.size() != 1)
                    throw std::runtime_error(
                        "multi-token phrase prefix not supported");
                if (toks[0].find('*') != std::string::npos ||
                    toks[0].find('?') != std::string::npos)
                    throw std::runtime_error(
                        "phrase prefix with wildcard metachars");
                n.kind = PlanNode::WILDCARD;
                n.const_score = true;
                n.field = t.field;
                n.value = toks[0] + "*";
                n.boost = t.boost;
                return n;
            }
            if (toks.size() > 1) {
                n.kind = PlanNode::PHRASE;
                n.field = t.field;
                n.phrase_toks = std::move(toks);
                n.boost = t.boost;
                return n;
            }
            n.kind = PlanNode::TERM;
            n.field = t.field;
            n.value = toks[0];
            n.boost = t.boost;
            return n;
        }
        if (!t.no_glob && (t.text.find('*') != std::string::npos ||
                           t.text.find('?') != std::string::npos)) {
            // field:pat* -> wildcard over the term dictionary, pattern
            // folded through the field's (non-raw) analyzer chain
            n.kind = PlanNode::WILDCARD;
            n.const_score = true;
            n.field = t.field;
            n.value = (f && f->tokenizer == "raw") ? t.text
                                                   : qw_utf8_lower(t.text);
            n.boost = t.boost;
            return n;
        }
        // unquoted value: a single grammar word; when the field's analyzer
        // splits it into several tokens (1.5 -> [1,5], AB-CD -> [ab,cd])
        // the reference's parser emits a slop-0 PhraseQuery, not an OR
        {
            std::vector<std::string> toks =
                tokenize(t.text, f ? f->tokenizer : "default");
            if (toks.empty()) {
                n.kind = PlanNode::MATCH_NONE;
                return n;
            }
            if (toks.size() > 1) {
                n.kind = PlanNode::PHRASE;
                n.field = t.field;
                n.phrase_toks = std::move(toks);
                n.boost = t.boost;
                return n;
            }
            n.kind = PlanNode::TERM;
            n.field = t.field;
            n.value = toks[0];
            n.boost = t.boost;
            return n;
        }
    }
    if (dfs.empty()) throw std::runtime_error("no default search fields");
    PlanNode b;
    b.kind = PlanNode::BOOL;
    bool has_glob = !t.no_glob &&
                    (t.text.find('*') != std::string::npos ||
                     t.text.find('?') != std::string::npos);
    // bare ">=N" / "<N" over the default fields: a range on each
    // (query_string with a numeric default_field — es_compat golden)
    {
        Bound::Kind lk = Bound::UNBOUNDED, hk = Bound::UNBOUNDED;
        size_t skip = 0;
        if (t.text.rfind(">=", 0) == 0) { lk = Bound::INCLUDED; skip = 2; }
        else if (t.text.rfind("<=", 0) == 0) { hk = Bound::INCLUDED; skip = 2; }
        else if (!t.text.empty() && t.text[0] == '>') { lk = Bound::EXCLUDED; skip = 1; }
        else if (!t.text.empty() && t.text[0] == '<') { hk = Bound::EXCLUDED; skip = 1; }
        if (skip) {
            std::string lit = t.text.substr(skip);
            for (auto& df : dfs) {
                const SchemaField* f = schema.field(df);
                if (!f || f->type == "text") continue;
                try {
                    PlanNode r;
                    r.kind = PlanNode::RANGE;
                    r.field = df;
                    if (lk != Bound::UNBOUNDED) r.lo = qbound(lit, lk, *f);
                    else r.hi = qbound(lit, hk, *f);
                    b.should.push_back(std::move(r));
                } catch (const std::exception&) {
                }
            }
            if (!b.should.empty()) {
                PlanNode r = b.should.size() == 1 ? std::move(b.should[0])
                                                  : std::move(b);
                r.boost *= t.boost;
                return r;
            }
            PlanNode none;
            none.kind = PlanNode::MATCH_NONE;
            return none;
        }
    }
    for (auto& df : dfs) {
        // lenient over the default-field list: configured names may be
        // dynamic fields absent from a given split's schema
        const SchemaField* f = schema.field(df);
        if (!f) continue;
        if (f->type != "text") {
            // numeric/bool/str fast default field: equality when the
            // literal parses for that type (lenient otherwise)
            if (has_glob) continue;
            try {
                b.should.push_back(term_leaf_plan(df, t.text, schema));
            } catch (const std::exception&) {
            }
            continue;
        }
        if (has_glob) {
            PlanNode w;
            w.kind = PlanNode::WILDCARD;
            w.const_score = true;
            w.field = df;
            w.value = qw_utf8_lower(t.text);
            b.should.push_back(std::move(w));
            continue;
        }
        b.should.push_back(full_text_plan(df, t.text, "or", schema));
    }
    if (b.should.empty()) {
        PlanNode none;
        none.kind = PlanNode::MATCH_NONE;
        return none;
    }
    PlanNode r = b.should.size() == 1 ? std::move(b.should[0]) : std::move(b);
    r.boost *= t.boost;
    return r;
}

inline PlanNode qtok_leaf(const QTok& t, const std::vector<std::string>& dfs,
                          const Schema& schema) {
    return qtok_leaf_impl(t, dfs, schema);
}

inline PlanNode user_input_plan(const std::string& text,
                                const std::vector<std::string>& default_fields,
                                const Schema& schema) {
    // tantivy query grammar (see QTok above); default occur = Should (the
    // reference's BooleanOperand::Or default, user_input_query.rs:125-130)
    std::vector<QTok> toks = qlex(text);
    size_t pos = 0;
    PlanNode n = qparse_clause(toks, &pos, default_fields, schema);
    if (pos < toks.size()) throw std::runtime_error("unbalanced parentheses");
    return n;
}

inline PlanNode build_plan(const mj::Value* ast, const Schema& schema) {
    std::string ty = ast->at("type")->s;
    PlanNode n;
    if (ty == "match_all") {
        n.kind = PlanNode::MATCH_ALL;
    } else if (ty == "match_none") {
        n.kind = PlanNode::MATCH_NONE;
    } else if (ty == "term") {
        return term_leaf_plan(ast->at("field")->s, ast->at("value")->s,
                              schema);
    } else if (ty == "full_text") {
        std::string op = "or";
        bool zta = false;
        const mj::Value* params = ast->get("params");
        if (params) {
            const mj::Value* z = params->get("zero_terms_query");
            zta = z && z->s == "all";
            const mj::Value* mode = params->get("mode");
            if (mode) {
                // FullTextMode serde: {"type":"bool","operator":"or"|"and"},
                // {"type":"phrase","slop":N} (full_text_query.rs:113-137,172)
                const mj::Value* mt = mode->get("type");
                if (mt && mt->s == "phrase") {
                    const mj::Value* slop = mode->get("slop");
                    if (slop && slop->as_i64() != 0)
                        throw std::runtime_error(
                            "phrase slop > 0 not supported (r2 limit)");
                    return phrase_plan(ast->at("field")->s,
                                       ast->at("text")->s, schema, zta);
                }
                if (mt && mt->s != "bool" && mt->s != "bool_prefix")
                    throw std::runtime_error("full_text mode not supported: " + mt->s);
                const mj::Value* o = mode->get("operator");
                if (o) op = o->s;
                std::transform(op.begin(), op.end(), op.begin(), ::tolower);
            }
        }
        return full_text_plan(ast->at("field")->s, ast->at("text")->s, op, schema,
                              zta);
    } else if (ty == "bool") {
        n.kind = PlanNode::BOOL;
        auto fill = [&](const char* key, std::vector<PlanNode>& out) {
            const mj::Value* arr = ast->get(key);
            if (arr)
                for (auto& c : arr->arr) out.push_back(build_plan(c.get(), schema));
        };
        fill("must", n.must);
        fill("must_not", n.must_not);
        fill("should", n.should);
        fill("filter", n.filter);
        const mj::Value* msm = ast->get("minimum_should_match");
        if (msm && !msm->is_null()) n.minimum_should_match = msm->as_i64();
    } else if (ty == "range") {
        n.kind = PlanNode::RANGE;
        n.field = ast->at("field")->s;
        const SchemaField* f = schema.field(n.field);
        if (!f || !f->fast)
            throw std::runtime_error("range on non-fast field: " + n.field);
        n.lo = parse_bound(ast->get("lower_bound"), *f, false);
        n.hi = parse_bound(ast->get("upper_bound"), *f, true);
    } else if (ty == "field_presence") {
        n.kind = PlanNode::FIELD_PRESENCE;
        n.field = ast->at("field")->s;
    } else if (ty == "boost") {
        PlanNode inner = build_plan(ast->at("underlying"), schema);
        inner.boost *= float(ast->at("boost")->num());
        return inner;
    } else if (ty == "term_set") {
        // TermSetQuery "matches the same document set as a union of the
        // equivalent TermQueries", untokenized values
        // (query_ast/term_set_query.rs:25-80); values arrive as a sorted set
        n.kind = PlanNode::BOOL;
        n.const_score = true;
        const mj::Value* tpf = ast->at("terms_per_field");
        for (auto& kv : tpf->obj) {
            if (!schema.field(kv.first))
                throw std::runtime_error("term_set on unknown field: " + kv.first);
            for (auto& v : kv.second->arr)
                n.should.push_back(term_leaf_plan(kv.first, v->s, schema));
        }
        if (n.should.empty()) n.kind = PlanNode::MATCH_NONE;
    } else if (ty == "cache") {
        // CacheNode (cache_node.rs:33-37): semantics-transparent wrapper;
        // the product memoizes the inner subtree's HitSet as a device
        // bitmap when the node sits in filter position
        n.kind = PlanNode::CACHE;
        n.cache_inner.push_back(build_plan(ast->at("inner"), schema));
    } else if (ty == "wildcard") {
        n.kind = PlanNode::WILDCARD;
        n.const_score = true;
        n.field = ast->at("field")->s;
        n.value = ast->at("value")->s;
        const mj::Value* ci = ast->get("case_insensitive");
        n.ci = ci && ci->b;
        const SchemaField* f = schema.field(n.field);
        if (!f) {
            const mj::Value* len = ast->get("lenient");
            if (len && len->b) {
                n.kind = PlanNode::MATCH_NONE;  // lenient: missing field OK
                return n;
            }
            throw std::runtime_error("wildcard on unknown field: " + n.field);
        }
        // the reference tokenizes the pattern's literal segments with the
        // field's tokenizer (wildcard_query.rs:111-160): for the default
        // (lowercasing) analyzer that lowercases them
        if (f->tokenizer != "raw") n.value = qw_utf8_lower(n.value);
    } else if (ty == "user_input") {
        std::vector<std::string> dfs;
        const mj::Value* d = ast->get("default_fields");
        if (d && !d->is_null())
            for (auto& v : d->arr) dfs.push_back(v->s);
        return user_input_plan(ast->at("user_text")->s, dfs, schema);
    } else {
        throw std::runtime_error("query ast type not supported: " + ty);
    }
    return n;
}

inline PlanNode parse_query_ast(const std::string& json, const Schema& schema) {
    return build_plan(mj::parse(json).get(), schema);
}

// ---------------------------------------------------------------- rfc3339
inline int64_t days_from_civil(int64_t y, unsigned m, unsigned d) {
    y -= m <= 2;
    const int64_t era = (y >= 0 ? y : y - 399) / 400;
    const unsigned yoe = unsigned(y - era * 400);
    const unsigned doy = (153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1;
    const unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
    return era * 146097 + int64_t(doe) - 719468;
}

inline int64_t rfc3339_to_ms(const std::string& s) {
    int y = 0, mo = 0, d = 0, h = 0, mi = 0;
    double sec = 0;
    int off_sign = 0, off_h = 0, off_m = 0;
    if (s.size() < 10) throw std::runtime_error("bad datetime: " + s);
    y = atoi(s.substr(0, 4).c_str());
    mo = atoi(s.substr(5, 2).c_str());
    d = atoi(s.substr(8, 2).c_str());
    size_t i = 10;
    if (i < s.size() && (s[i] == 'T' || s[i] == 't' || s[i] == ' ')) {
        h = atoi(s.substr(i + 1, 2).c_str());
        mi = atoi(s.substr(i + 4, 2).c_str());
        i += 6;
        if (i < s.size() && s[i] == ':') {
            size_t j = i + 1;
            while (j < s.size() && (isdigit((unsigned char)s[j]) || s[j] == '.')) ++j;
            sec = atof(s.substr(i + 1, j - i - 1).c_str());
            i = j;
        }
        if (i < s.size() && (s[i] == '+' || s[i] == '-')) {
            off_sign = s[i] == '+' ? 1 : -1;
            off_h = atoi(s.substr(i + 1, 2).c_str());
            off_m = atoi(s.substr(i + 4, 2).c_str());
        }
    }
    int64_t days = days_from_civil(y, unsigned(mo), unsigned(d));
    double t = double(days) * 86400.0 + h * 3600.0 + mi * 60.0 + sec -
               off_sign * (off_h * 3600.0 + off_m * 60.0);
    return int64_t(t * 1000.0 + (t >= 0 ? 0.5 : -0.5));
}

inline int64_t parse_datetime_ms(const mj::Value* lit) {
    if (lit->kind == mj::Value::STR) {
        // numeric string or rfc3339
        char* endp = nullptr;
        long long v = strtoll(lit->s.c_str(), &endp, 10);
        if (endp && *endp == 0) {
            long long a = v < 0 ? -v : v;
            return a < 1000000000000LL ? v * 1000 : v;  // s vs ms, like splitgen
        }
        if (lit->s.find('/') != std::string::npos) {
            // "YYYY/MM/DD" (quickwit-datetime lenient date format)
            std::string iso = lit->s;
            for (char& c : iso)
                if (c == '/') c = '-';
            return rfc3339_to_ms(iso);
        }
        return rfc3339_to_ms(lit->s);
    }
    int64_t v = lit->as_i64();
    int64_t a = v < 0 ? -v : v;
    return a < 1000000000000LL ? v * 1000 : v;
}

inline std::string ms_to_rfc3339(int64_t ms) {
    int64_t days = ms / 86400000;
    int64_t rem = ms % 86400000;
    if (rem < 0) {
        rem += 86400000;
        days -= 1;
    }
    // civil_from_days (Howard Hinnant)
    int64_t z = days + 719468;
    const int64_t era = (z >= 0 ? z : z - 146096) / 146097;
    const unsigned doe = unsigned(z - era * 146097);
    const unsigned yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
    const int64_t y = int64_t(yoe) + era * 400;
    const unsigned doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
    const unsigned mp = (5 * doy + 2) / 153;
    const unsigned d = doy - (153 * mp + 2) / 5 + 1;
    const unsigned m = mp + (mp < 10 ? 3 : -9);
    int64_t yy = y + (m <= 2);
    int sec_of_day = int(rem / 1000);
    int msec = int(rem % 1000);
    char buf[40];
    if (msec)
        snprintf(buf, sizeof buf, "%04lld-%02u-%02uT%02d:%02d:%02d.%03dZ", (long long)yy, m,
                 d, sec_of_day / 3600, (sec_of_day / 60) % 60, sec_of_day % 60, msec);
    else
        snprintf(buf, sizeof buf, "%04lld-%02u-%02uT%02d:%02d:%02dZ", (long long)yy, m, d,
                 sec_of_day / 3600, (sec_of_day / 60) % 60, sec_of_day % 60);
    return buf;
}

// ---------------------------------------------------------------- agg plan
struct MetricAgg {
    std::string name;
    enum Kind { STATS, AVG, SUM, MIN, MAX, COUNT, EXTENDED, PERCENTILES }
        kind = STATS;
    std::string field;
    // PERCENTILES (restated DDSketch, alpha=0.01 — the sketches-ddsketch
    // config tantivy's percentiles agg uses; fit pinned by the
    // aggregations golden's expected values): requested percents + keyed
    std::vector<double> percents{1, 5, 25, 50, 75, 95, 99};
    bool keyed = true;
};

struct RangeSpec {
    std::string key;
    bool has_from = false, has_to = false;
    double from = 0, to = 0;  // [from, to) — ES range-agg semantics
};

// one source of a composite aggregation (ES composite agg: a flattened
// cross-product of per-source buckets, paginated by composite key;
// tantivy supports terms + histogram sources)
struct CompSource {
    std::string name;
    std::string field;
    bool is_histo = false;
    double interval = 0, offset = 0;  // histogram source
    bool missing_bucket = false;      // terms source: null bucket, sorts first
    // `after` component for this source (resume strictly after this key):
    // kind 0 = none, 1 = null, 2 = str, 3 = number
    int after_kind = 0;
    std::string after_s;
    double after_n = 0;
};

struct AggDef {
    std::string name;
    enum Kind { DATE_HISTOGRAM, HISTOGRAM, TERMS, RANGE, METRIC, COMPOSITE,
                CARDINALITY } kind = DATE_HISTOGRAM;
    std::string field;
    std::vector<CompSource> comp;   // COMPOSITE
    bool has_after = false;         // COMPOSITE: `after` given
    std::vector<RangeSpec> ranges;  // RANGE
    MetricAgg metric;               // METRIC (top-level stats/avg/... agg)
    double interval = 0;  // ms for date_histogram; raw units for histogram
    double offset = 0;
    bool has_bounds = false;
    double bmin = 0, bmax = 0;
    uint32_t size = 10;          // terms
    // terms bucket order: "" = _count (default), "_key" = by term key;
    // order by a sub-aggregation value: rejected (r1 limit)
    std::string order_target;
    bool order_asc = false;      // _count default desc; explicit order sets it
    int64_t split_size = -1;     // terms: per-split truncation; -1 = default
                                 // (size*3/2+10, the ES shard_size default)
    int64_t min_doc_count = -1;  // -1 = default (0 for histos, 1 for terms)
    std::vector<MetricAgg> sub;
};

inline double parse_duration_ms(const std::string& s) {
    size_t i = 0;
    while (i < s.size() && (isdigit((unsigned char)s[i]) || s[i] == '.' || s[i] == '-' ||
                            s[i] == '+'))
        ++i;
    double v = atof(s.substr(0, i).c_str());
    std::string u = s.substr(i);
    if (u == "ms") return v;
    if (u == "s") return v * 1000;
    if (u == "m") return v * 60000;
    if (u == "h") return v * 3600000;
    if (u == "d") return v * 86400000;
    throw std::runtime_error("bad interval: " + s);
}

// Validate aggregation targets against the split's fast columns:
// wrong-typed PRESENT columns are errors (tantivy rejects e.g.
// date_histogram over a text column); absent columns are allowed and
// contribute empty results. col_type(name) returns
// the FastFieldView::Type as int, or -1 when the field has no fast column.
// Type codes: 0=U64 1=I64 2=DATETIME 3=STR 4=F64 5=MIXED.
template <class ColType>
inline void validate_agg_fields(const std::vector<AggDef>& defs,
                                ColType&& col_type) {
    // A column absent from this split is NOT an error: tantivy opens
    // columns optionally and an empty column contributes empty results
    // (the aggregations golden aggregates on fields some splits lack).
    // Only a PRESENT column of an incompatible type is an error.
    auto require = [&](const std::string& field) -> int {
        return col_type(field);  // -1 = absent, allowed everywhere
    };
    auto metric_ok = [&](const MetricAgg& m) {
        int t = require(m.field);
        // value_count works on any column; other metrics need numbers
        if (t >= 0 && m.kind != MetricAgg::COUNT && t == 3 /*STR*/)
            throw std::runtime_error(
                "metric aggregation on non-numeric field: " + m.field);
    };
    for (const AggDef& d : defs) {
        switch (d.kind) {
            case AggDef::DATE_HISTOGRAM: {
                int t = require(d.field);
                if (t >= 0 && t != 2 /*DATETIME*/)
                    throw std::runtime_error(
                        "date_histogram on non-datetime field: " + d.field);
                break;
            }
            case AggDef::HISTOGRAM:
            case AggDef::RANGE: {
                int t = require(d.field);
                if (t == 3 /*STR*/)
                    throw std::runtime_error(
                        "histogram/range on non-numeric field: " + d.field);
                break;
            }
            case AggDef::TERMS:
            case AggDef::CARDINALITY:
                require(d.field);
                break;
            case AggDef::METRIC:
                metric_ok(d.metric);
                break;
            case AggDef::COMPOSITE:
                for (const auto& cs : d.comp) {
                    int t = require(cs.field);
                    if (cs.is_histo && t == 3 /*STR*/)
                        throw std::runtime_error(
                            "composite histogram source on non-numeric "
                            "field: " + cs.field);
                }
                break;
        }
        for (const MetricAgg& m : d.sub) metric_ok(m);
    }
}

inline std::vector<AggDef> parse_agg_request(const std::string& json) {
    std::vector<AggDef> out;
    mj::ValuePtr root = mj::parse(json);
    for (auto& kv : root->obj) {
        AggDef a;
        a.name = kv.first;
        const mj::Value* body = kv.second.get();
        const mj::Value* spec = nullptr;
        if ((spec = body->get("date_histogram"))) {
            a.kind = AggDef::DATE_HISTOGRAM;
            a.field = spec->at("field")->s;
            const mj::Value* fi = spec->get("fixed_interval");
            if (!fi) throw std::runtime_error("date_histogram requires fixed_interval");
            a.interval = parse_duration_ms(fi->s);
            if (!(a.interval > 0))
                throw std::runtime_error(
                    "date_histogram fixed_interval must be > 0");
            if (const mj::Value* off = spec->get("offset"))
                a.offset = off->kind == mj::Value::STR ? parse_duration_ms(off->s)
                                                       : off->num();
            if (const mj::Value* eb = spec->get("extended_bounds")) {
                a.has_bounds = true;
                a.bmin = eb->at("min")->num();
                a.bmax = eb->at("max")->num();
            }
        } else if ((spec = body->get("histogram"))) {
            a.kind = AggDef::HISTOGRAM;
            a.field = spec->at("field")->s;
            a.interval = spec->at("interval")->num();
            if (!(a.interval > 0))
                throw std::runtime_error("histogram interval must be > 0");
            if (const mj::Value* off = spec->get("offset")) a.offset = off->num();
            if (const mj::Value* eb = spec->get("extended_bounds")) {
                a.has_bounds = true;
                a.bmin = eb->at("min")->num();
                a.bmax = eb->at("max")->num();
            }
        } else if ((spec = body->get("range"))) {
            a.kind = AggDef::RANGE;
            a.field = spec->at("field")->s;
            for (auto& rv : spec->at("ranges")->arr) {
                RangeSpec r;
                if (const mj::Value* f2 = rv->get("from")) {
                    r.has_from = true;
                    r.from = f2->num();
                }
                if (const mj::Value* t2 = rv->get("to")) {
                    r.has_to = true;
                    r.to = t2->num();
                }
                if (const mj::Value* k2 = rv->get("key")) r.key = k2->s;
                else {
                    char buf[64];
                    auto fmt = [&](bool has, double v) {
                        if (!has) return std::string("*");
                        snprintf(buf, sizeof buf, "%g", v);
                        return std::string(buf);
                    };
                    r.key = fmt(r.has_from, r.from) + "-" + fmt(r.has_to, r.to);
                }
                a.ranges.push_back(std::move(r));
            }
        } else if ((spec = body->get("composite"))) {
            // ES composite aggregation: sources = terms | histogram, paged
            // by composite key asc with optional `after` cursor; keys typed
            // "str:x" / "f64:1.5" in quickwit's after serialization
            a.kind = AggDef::COMPOSITE;
            a.size = 10;
            if (const mj::Value* sz = spec->get("size"))
                a.size = uint32_t(sz->as_i64());
            for (auto& sv2 : spec->at("sources")->arr) {
                if (sv2->obj.size() != 1)
                    throw std::runtime_error("composite source must have one name");
                CompSource cs;
                cs.name = sv2->obj.begin()->first;
                const mj::Value* sb = sv2->obj.begin()->second.get();
                const mj::Value* ss2 = nullptr;
                if ((ss2 = sb->get("terms"))) {
                    cs.is_histo = false;
                    if (const mj::Value* mb = ss2->get("missing_bucket"))
                        cs.missing_bucket = mb->b;
                } else if ((ss2 = sb->get("histogram"))) {
                    cs.is_histo = true;
                    cs.interval = ss2->at("interval")->num();
                    if (const mj::Value* off = ss2->get("offset"))
                        cs.offset = off->num();
                } else {
                    throw std::runtime_error(
                        "composite source must be terms or histogram");
                }
                cs.field = ss2->at("field")->s;
                a.comp.push_back(std::move(cs));
            }
            if (a.comp.empty() || a.comp.size() > 4)
                throw std::runtime_error("composite: 1..4 sources supported");
            if (const mj::Value* af = spec->get("after")) {
                a.has_after = true;
                for (CompSource& cs : a.comp) {
                    const mj::Value* v = af->get(cs.name);
                    if (!v) throw std::runtime_error(
                        "composite after missing source " + cs.name);
                    if (v->kind == mj::Value::NUL) cs.after_kind = 1;
                    else if (v->kind == mj::Value::STR) {
                        std::string s = v->s;
                        // typed after-key serialization: str:/f64:/i64:/u64:
                        if (s.rfind("str:", 0) == 0) {
                            cs.after_kind = 2;
                            cs.after_s = s.substr(4);
                        } else if (s.rfind("f64:", 0) == 0 ||
                                   s.rfind("i64:", 0) == 0 ||
                                   s.rfind("u64:", 0) == 0) {
                            cs.after_kind = 3;
                            cs.after_n = atof(s.c_str() + 4);
                        } else {
                            cs.after_kind = 2;
                            cs.after_s = s;
                        }
                    } else {
                        cs.after_kind = 3;
                        cs.after_n = v->num();
                    }
                }
            }
        } else if ((spec = body->get("cardinality"))) {
            // exact distinct count via the terms machinery (merged distinct
            // key set). DEVIATION from the reference's HyperLogLog++ sketch:
            // exact answers, matching ES/golden outputs wherever the sketch
            // is exact (small cardinalities); bounded by the per-split
            // distinct-set guards. DESIGN.md §7.
            a.kind = AggDef::CARDINALITY;
            a.field = spec->at("field")->s;
        } else if ((spec = body->get("terms"))) {
            a.kind = AggDef::TERMS;
            a.field = spec->at("field")->s;
            if (const mj::Value* sz = spec->get("size")) a.size = uint32_t(sz->as_i64());
            // one segment per split in QWA1 => split/segment/shard size
            // aliases all bound the same per-split truncation
            if (const mj::Value* ss = spec->get("split_size"))
                a.split_size = ss->as_i64();
            else if (const mj::Value* sg = spec->get("segment_size"))
                a.split_size = sg->as_i64();
            else if (const mj::Value* sh = spec->get("shard_size"))
                a.split_size = sh->as_i64();
            if (const mj::Value* mdc = spec->get("min_doc_count"))
                a.min_doc_count = mdc->as_i64();
            if (const mj::Value* od = spec->get("order")) {
                // ES: {"order": {"_key": "asc"}} (single criterion; arrays
                // and sub-aggregation targets: later round)
                const mj::Value* body2 = od;
                if (od->kind == mj::Value::ARR) {
                    if (od->arr.size() != 1)
                        throw std::runtime_error(
                            "terms order: multiple criteria (r1 limit)");
                    body2 = od->arr[0].get();
                }
                if (body2->obj.size() != 1)
                    throw std::runtime_error("terms order: one criterion");
                const std::string& tgt = body2->obj.begin()->first;
                const std::string dir = body2->obj.begin()->second->s;
                if (tgt == "_count") a.order_target = "";
                else if (tgt == "_key" || tgt == "_term") a.order_target = "_key";
                else a.order_target = tgt;  // sub-aggregation value path:
                                            // "sub" or "sub.stat"
                a.order_asc = dir == "asc";
            }
        } else {
            // top-level metric aggregations (stats/avg/sum/min/max/
            // value_count/extended_stats over every matched doc)
            const mj::Value* ms0 = nullptr;
            MetricAgg::Kind mk = MetricAgg::STATS;
            if ((ms0 = body->get("stats"))) mk = MetricAgg::STATS;
            else if ((ms0 = body->get("extended_stats"))) mk = MetricAgg::EXTENDED;
            else if ((ms0 = body->get("avg"))) mk = MetricAgg::AVG;
            else if ((ms0 = body->get("sum"))) mk = MetricAgg::SUM;
            else if ((ms0 = body->get("min"))) mk = MetricAgg::MIN;
            else if ((ms0 = body->get("max"))) mk = MetricAgg::MAX;
            else if ((ms0 = body->get("value_count"))) mk = MetricAgg::COUNT;
            else if ((ms0 = body->get("percentiles"))) mk = MetricAgg::PERCENTILES;
            else throw std::runtime_error("aggregation not supported: " + a.name);
            a.kind = AggDef::METRIC;
            a.field = ms0->at("field")->s;
            a.metric.kind = mk;
            a.metric.name = a.name;
            a.metric.field = a.field;
            if (mk == MetricAgg::PERCENTILES) {
                if (const mj::Value* ps = ms0->get("percents")) {
                    a.metric.percents.clear();
                    for (auto& pv : ps->arr) a.metric.percents.push_back(pv->num());
                }
                if (const mj::Value* kd = ms0->get("keyed"))
                    a.metric.keyed = kd->b;
            }
        }
        if (a.min_doc_count < 0) a.min_doc_count = (a.kind == AggDef::TERMS) ? 1 : 0;
        if (const mj::Value* subs = body->get("aggs")) {
            for (auto& skv : subs->obj) {
                MetricAgg m;
                m.name = skv.first;
                const mj::Value* sb = skv.second.get();
                const mj::Value* ms = nullptr;
                if ((ms = sb->get("stats"))) m.kind = MetricAgg::STATS;
                else if ((ms = sb->get("extended_stats"))) m.kind = MetricAgg::EXTENDED;
                else if ((ms = sb->get("avg"))) m.kind = MetricAgg::AVG;
                else if ((ms = sb->get("sum"))) m.kind = MetricAgg::SUM;
                else if ((ms = sb->get("min"))) m.kind = MetricAgg::MIN;
                else if ((ms = sb->get("max"))) m.kind = MetricAgg::MAX;
                else if ((ms = sb->get("value_count"))) m.kind = MetricAgg::COUNT;
                else if ((ms = sb->get("percentiles"))) m.kind = MetricAgg::PERCENTILES;
                else throw std::runtime_error("sub-aggregation not supported: " + m.name);
                m.field = ms->at("field")->s;
                if (m.kind == MetricAgg::PERCENTILES) {
                    if (const mj::Value* ps = ms->get("percents")) {
                        m.percents.clear();
                        for (auto& pv : ps->arr) m.percents.push_back(pv->num());
                    }
                    if (const mj::Value* kd = ms->get("keyed")) m.keyed = kd->b;
                }
                a.sub.push_back(std::move(m));
            }
        }
        out.push_back(std::move(a));
    }
    return out;
}

}  // namespace qw
