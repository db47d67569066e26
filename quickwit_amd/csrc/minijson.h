// Minimal JSON DOM parser/writer (boundary plumbing shared by the product
// host code and by oracle/ — codec only, no search semantics).
// Parses the subset quickwit emits for QueryAst / aggregation requests /
// config (objects, arrays, strings with escapes, doubles, ints, bool, null).
#pragma once
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace mj {

struct Value;
using ValuePtr = std::shared_ptr<Value>;

struct Value {
    enum Kind { NUL, BOOL, INT, DBL, STR, ARR, OBJ } kind = NUL;
    bool b = false;
    int64_t i = 0;      // INT: exact integers (also mirrored into d)
    double d = 0.0;     // DBL
    std::string s;
    std::vector<ValuePtr> arr;
    std::vector<std::pair<std::string, ValuePtr>> obj;  // insertion order kept

    bool is_null() const { return kind == NUL; }
    bool is_num() const { return kind == INT || kind == DBL; }
    double num() const { return kind == INT ? double(i) : d; }
    int64_t as_i64() const { return kind == INT ? i : int64_t(d); }

    const Value* get(const std::string& key) const {
        for (auto& kv : obj)
            if (kv.first == key) return kv.second.get();
        return nullptr;
    }
    const Value* at(const std::string& key) const {
        const Value* v = get(key);
        if (!v) throw std::runtime_error("missing json key: " + key);
        return v;
    }
};

class Parser {
  public:
    explicit Parser(const char* p, size_t n) : p_(p), end_(p + n) {}
    ValuePtr parse() {
        ValuePtr v = value();
        ws();
        if (p_ != end_) fail("trailing data");
        return v;
    }

  private:
    const char* p_;
    const char* end_;
    // serde_json's default recursion limit is 128; the reference's
    // query_ast parse errors on deeper input instead of overflowing the
    // stack — mirror that (a pathological request must be a status code,
    // never a fault; pinned by tests/test_abi.py)
    static constexpr int kMaxDepth = 128;
    int depth_ = 0;
    [[noreturn]] void fail(const char* msg) {
        throw std::runtime_error(std::string("json parse error: ") + msg);
    }
    void ws() {
        while (p_ < end_ && (*p_ == ' ' || *p_ == '\t' || *p_ == '\n' || *p_ == '\r')) ++p_;
    }
    char peek() {
        if (p_ >= end_) fail("eof");
        return *p_;
    }
    bool lit(const char* w) {
        size_t n = strlen(w);
        if (size_t(end_ - p_) >= n && !memcmp(p_, w, n)) {
            p_ += n;
            return true;
        }
        return false;
    }
    ValuePtr value() {
        ws();
        char c = peek();
        if ((c == '{' || c == '[') && ++depth_ > kMaxDepth)
            fail("recursion limit exceeded");
        struct DepthGuard {
            int* d;
            ~DepthGuard() { if (d) --*d; }
        } guard{(c == '{' || c == '[') ? &depth_ : nullptr};
        auto v = std::make_shared<Value>();
        if (c == '{') {
            v->kind = Value::OBJ;
            ++p_;
            ws();
            if (peek() == '}') { ++p_; return v; }
            while (true) {
                ws();
                if (peek() != '"') fail("expected key");
                std::string k = str();
                ws();
                if (peek() != ':') fail("expected :");
                ++p_;
                v->obj.emplace_back(std::move(k), value());
                ws();
                char t = peek();
                ++p_;
                if (t == '}') return v;
                if (t != ',') fail("expected , or }");
            }
        } else if (c == '[') {
            v->kind = Value::ARR;
            ++p_;
            ws();
            if (peek() == ']') { ++p_; return v; }
            while (true) {
                v->arr.push_back(value());
                ws();
                char t = peek();
                ++p_;
                if (t == ']') return v;
                if (t != ',') fail("expected , or ]");
            }
        } else if (c == '"') {
            v->kind = Value::STR;
            v->s = str();
            return v;
        } else if (lit("true")) {
            v->kind = Value::BOOL;
            v->b = true;
            return v;
        } else if (lit("false")) {
            v->kind = Value::BOOL;
            v->b = false;
            return v;
        } else if (lit("null")) {
            return v;
        }
        return number(v);
    }
    std::string str() {
        ++p_;  // opening quote
        std::string out;
        while (true) {
            if (p_ >= end_) fail("eof in string");
            char c = *p_++;
            if (c == '"') return out;
            if (c != '\\') {
                out.push_back(c);
                continue;
            }
            if (p_ >= end_) fail("eof in escape");
            char e = *p_++;
            switch (e) {
                case '"': out.push_back('"'); break;
                case '\\': out.push_back('\\'); break;
                case '/': out.push_back('/'); break;
                case 'b': out.push_back('\b'); break;
                case 'f': out.push_back('\f'); break;
                case 'n': out.push_back('\n'); break;
                case 'r': out.push_back('\r'); break;
                case 't': out.push_back('\t'); break;
                case 'u': {
                    if (end_ - p_ < 4) fail("bad \\u");
                    unsigned cp = 0;
                    for (int k = 0; k < 4; ++k) {
                        char h = *p_++;
                        cp <<= 4;
                        if (h >= '0' && h <= '9') cp |= h - '0';
                        else if (h >= 'a' && h <= 'f') cp |= h - 'a' + 10;
                        else if (h >= 'A' && h <= 'F') cp |= h - 'A' + 10;
                        else fail("bad hex");
                    }
                    // surrogate pairs
                    if (cp >= 0xD800 && cp <= 0xDBFF && end_ - p_ >= 6 && p_[0] == '\\' &&
                        p_[1] == 'u') {
                        unsigned lo = 0;
                        const char* q = p_ + 2;
                        for (int k = 0; k < 4; ++k) {
                            char h = q[k];
                            lo <<= 4;
                            if (h >= '0' && h <= '9') lo |= h - '0';
                            else if (h >= 'a' && h <= 'f') lo |= h - 'a' + 10;
                            else if (h >= 'A' && h <= 'F') lo |= h - 'A' + 10;
                            else { lo = 0xFFFFFFFF; break; }
                        }
                        if (lo >= 0xDC00 && lo <= 0xDFFF) {
                            cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
                            p_ += 6;
                        }
                    }
                    // utf-8 encode
                    if (cp < 0x80) out.push_back(char(cp));
                    else if (cp < 0x800) {
                        out.push_back(char(0xC0 | (cp >> 6)));
                        out.push_back(char(0x80 | (cp & 0x3F)));
                    } else if (cp < 0x10000) {
                        out.push_back(char(0xE0 | (cp >> 12)));
                        out.push_back(char(0x80 | ((cp >> 6) & 0x3F)));
                        out.push_back(char(0x80 | (cp & 0x3F)));
                    } else {
                        out.push_back(char(0xF0 | (cp >> 18)));
                        out.push_back(char(0x80 | ((cp >> 12) & 0x3F)));
                        out.push_back(char(0x80 | ((cp >> 6) & 0x3F)));
                        out.push_back(char(0x80 | (cp & 0x3F)));
                    }
                    break;
                }
                default: fail("bad escape");
            }
        }
    }
    ValuePtr number(ValuePtr v) {
        const char* start = p_;
        bool is_int = true;
        if (p_ < end_ && (*p_ == '-' || *p_ == '+')) ++p_;
        while (p_ < end_ && ((*p_ >= '0' && *p_ <= '9') || *p_ == '.' || *p_ == 'e' ||
                             *p_ == 'E' || *p_ == '-' || *p_ == '+')) {
            if (*p_ == '.' || *p_ == 'e' || *p_ == 'E') is_int = false;
            ++p_;
        }
        if (p_ == start) fail("bad value");
        std::string tok(start, p_ - start);
        if (is_int) {
            errno = 0;
            char* endp = nullptr;
            long long x = strtoll(tok.c_str(), &endp, 10);
            if (errno == 0 && endp && *endp == 0) {
                v->kind = Value::INT;
                v->i = x;
                v->d = double(x);
                return v;
            }
            // u64 overflow of i64: fall through to unsigned
            errno = 0;
            unsigned long long ux = strtoull(tok.c_str(), &endp, 10);
            if (errno == 0 && endp && *endp == 0) {
                v->kind = Value::INT;
                v->i = int64_t(ux);  // two's-complement carry; callers wanting u64 reinterpret
                v->d = double(ux);
                return v;
            }
        }
        v->kind = Value::DBL;
        v->d = strtod(tok.c_str(), nullptr);
        return v;
    }
};

inline ValuePtr parse(const std::string& s) { return Parser(s.data(), s.size()).parse(); }
inline ValuePtr parse(const char* s, size_t n) { return Parser(s, n).parse(); }

// ---- writer (used for finalized aggregation JSON)
inline void escape_to(std::string& out, const std::string& s) {
    out.push_back('"');
    for (char c : s) {
        switch (c) {
            case '"': out += "\\\""; break;
            case '\\': out += "\\\\"; break;
            case '\n': out += "\\n"; break;
            case '\r': out += "\\r"; break;
            case '\t': out += "\\t"; break;
            default:
                if ((unsigned char)c < 0x20) {
                    char buf[8];
                    snprintf(buf, sizeof buf, "\\u%04x", c);
                    out += buf;
                } else out.push_back(c);
        }
    }
    out.push_back('"');
}

// Serialize a double the way serde_json does for f64 (shortest round-trip);
// "%.17g" is round-trip-safe but not shortest — tests compare parsed values,
// not strings, so this only needs to round-trip.
inline void num_to(std::string& out, double d) {
    if (d == (double)(int64_t)d && d >= -9.2e18 && d <= 9.2e18) {
        char buf[32];
        snprintf(buf, sizeof buf, "%lld.0", (long long)d);
        out += buf;
    } else {
        char buf[40];
        snprintf(buf, sizeof buf, "%.17g", d);
        out += buf;
    }
}

}  // namespace mj
