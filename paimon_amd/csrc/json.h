// Minimal JSON parser/serializer for the plan descriptor — no dependencies.
#pragma once

#include <cmath>
#include <cstdint>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace pmh {

struct Json {
    enum Type { NUL, BOOL, NUM, STR, ARR, OBJ } type = NUL;
    bool b = false;
    double num = 0;
    std::string str;
    std::vector<Json> arr;
    std::map<std::string, Json> obj;

    bool has(const std::string &k) const { return type == OBJ && obj.count(k); }
    const Json &operator[](const std::string &k) const {
        static Json null_json;
        auto it = obj.find(k);
        return it == obj.end() ? null_json : it->second;
    }
    int64_t as_i64(int64_t dflt = 0) const {
        return type == NUM ? (int64_t)llround(num) : dflt;
    }
    bool as_bool(bool dflt = false) const { return type == BOOL ? b : dflt; }
    std::string as_str(const std::string &dflt = "") const {
        return type == STR ? str : dflt;
    }
};

class JsonParser {
  public:
    explicit JsonParser(const char *s) : p_(s) {}
    Json parse() {
        Json v = value();
        ws();
        if (*p_) throw std::runtime_error("trailing JSON content");
        return v;
    }

  private:
    const char *p_;
    void ws() {
        while (*p_ == ' ' || *p_ == '\t' || *p_ == '\n' || *p_ == '\r') p_++;
    }
    Json value() {
        ws();
        switch (*p_) {
        case '{': return object();
        case '[': return array();
        case '"': {
            Json j;
            j.type = Json::STR;
            j.str = string();
            return j;
        }
        case 't': expect("true"); return mk_bool(true);
        case 'f': expect("false"); return mk_bool(false);
        case 'n': expect("null"); return Json{};
        default: return number();
        }
    }
    static Json mk_bool(bool v) {
        Json j;
        j.type = Json::BOOL;
        j.b = v;
        return j;
    }
    void expect(const char *lit) {
        for (const char *q = lit; *q; q++, p_++)
            if (*p_ != *q) throw std::runtime_error("bad JSON literal");
    }
    std::string string() {
        p_++;  // opening quote
        std::string out;
        while (*p_ && *p_ != '"') {
            if (*p_ == '\\') {
                p_++;
                switch (*p_) {
                case 'n': out += '\n'; break;
                case 't': out += '\t'; break;
                case 'r': out += '\r'; break;
                case 'b': out += '\b'; break;
                case 'f': out += '\f'; break;
                case 'u': {
                    unsigned cp = 0;
                    for (int i = 0; i < 4; i++) {
                        p_++;
                        char c = *p_;
                        cp = cp * 16 + (c <= '9' ? c - '0' : (c | 32) - 'a' + 10);
                    }
                    if (cp < 0x80) out += (char)cp;
                    else if (cp < 0x800) {
                        out += (char)(0xC0 | (cp >> 6));
                        out += (char)(0x80 | (cp & 0x3F));
                    } else {
                        out += (char)(0xE0 | (cp >> 12));
                        out += (char)(0x80 | ((cp >> 6) & 0x3F));
                        out += (char)(0x80 | (cp & 0x3F));
                    }
                    break;
                }
                default: out += *p_;
                }
                p_++;
            } else {
                out += *p_++;
            }
        }
        if (*p_ != '"') throw std::runtime_error("unterminated string");
        p_++;
        return out;
    }
    Json number() {
        char *end = nullptr;
        Json j;
        j.type = Json::NUM;
        j.num = strtod(p_, &end);
        if (end == p_) throw std::runtime_error("bad JSON number");
        p_ = end;
        return j;
    }
    Json array() {
        p_++;
        Json j;
        j.type = Json::ARR;
        ws();
        if (*p_ == ']') {
            p_++;
            return j;
        }
        for (;;) {
            j.arr.push_back(value());
            ws();
            if (*p_ == ',') {
                p_++;
                continue;
            }
            if (*p_ == ']') {
                p_++;
                return j;
            }
            throw std::runtime_error("bad JSON array");
        }
    }
    Json object() {
        p_++;
        Json j;
        j.type = Json::OBJ;
        ws();
        if (*p_ == '}') {
            p_++;
            return j;
        }
        for (;;) {
            ws();
            if (*p_ != '"') throw std::runtime_error("bad JSON key");
            std::string k = string();
            ws();
            if (*p_ != ':') throw std::runtime_error("missing ':'");
            p_++;
            j.obj[k] = value();
            ws();
            if (*p_ == ',') {
                p_++;
                continue;
            }
            if (*p_ == '}') {
                p_++;
                return j;
            }
            throw std::runtime_error("bad JSON object");
        }
    }
};

inline std::string json_escape(const std::string &s) {
    std::string o;
    for (char c : s) {
        if (c == '"' || c == '\\') {
            o += '\\';
            o += c;
        } else if (c == '\n') {
            o += "\\n";
        } else {
            o += c;
        }
    }
    return o;
}

}  // namespace pmh
