// Minimal JSON parser + writer for the gateway (header-only).
// Parses into a small DOM; also provides targeted raw-body helpers used
// on the hot path (find/replace a top-level string field without
// reserializing — the zero-copy analog of the reference's model-rewrite
// in api/openai.rs:945-993).
#pragma once

#include <cmath>
#include <cstring>
#include <map>
#include <memory>
#include <string>
#include <string_view>
#include <variant>
#include <vector>

namespace gw {

class Json;
using JsonArray = std::vector<Json>;
using JsonObject = std::vector<std::pair<std::string, Json>>;  // ordered

class Json {
 public:
  using Value = std::variant<std::nullptr_t, bool, double, std::string,
                             JsonArray, JsonObject>;
  Json() : v_(nullptr) {}
  Json(std::nullptr_t) : v_(nullptr) {}
  Json(bool b) : v_(b) {}
  Json(double d) : v_(d) {}
  Json(int i) : v_((double)i) {}
  Json(long long i) : v_((double)i) {}
  Json(const char* s) : v_(std::string(s)) {}
  Json(std::string s) : v_(std::move(s)) {}
  Json(std::string_view s) : v_(std::string(s)) {}
  Json(JsonArray a) : v_(std::move(a)) {}
  Json(JsonObject o) : v_(std::move(o)) {}

  bool is_null() const { return std::holds_alternative<std::nullptr_t>(v_); }
  bool is_bool() const { return std::holds_alternative<bool>(v_); }
  bool is_num() const { return std::holds_alternative<double>(v_); }
  bool is_str() const { return std::holds_alternative<std::string>(v_); }
  bool is_arr() const { return std::holds_alternative<JsonArray>(v_); }
  bool is_obj() const { return std::holds_alternative<JsonObject>(v_); }

  bool as_bool(bool dflt = false) const {
    return is_bool() ? std::get<bool>(v_) : dflt;
  }
  double as_num(double dflt = 0) const {
    return is_num() ? std::get<double>(v_) : dflt;
  }
  long long as_int(long long dflt = 0) const {
    return is_num() ? (long long)std::llround(std::get<double>(v_)) : dflt;
  }
  const std::string& as_str() const {
    static const std::string kEmpty;
    return is_str() ? std::get<std::string>(v_) : kEmpty;
  }
  const JsonArray& as_arr() const {
    static const JsonArray kEmpty;
    return is_arr() ? std::get<JsonArray>(v_) : kEmpty;
  }
  JsonArray& arr() { return std::get<JsonArray>(v_); }
  const JsonObject& as_obj() const {
    static const JsonObject kEmpty;
    return is_obj() ? std::get<JsonObject>(v_) : kEmpty;
  }
  JsonObject& obj() { return std::get<JsonObject>(v_); }

  // object access
  const Json* find(std::string_view key) const {
    if (!is_obj()) return nullptr;
    for (auto& [k, val] : std::get<JsonObject>(v_))
      if (k == key) return &val;
    return nullptr;
  }
  const Json& operator[](std::string_view key) const {
    static const Json kNull;
    const Json* p = find(key);
    return p ? *p : kNull;
  }
  Json& set(std::string key, Json val) {
    if (!is_obj()) v_ = JsonObject{};
    auto& o = std::get<JsonObject>(v_);
    for (auto& [k, existing] : o)
      if (k == key) {
        existing = std::move(val);
        return existing;
      }
    o.emplace_back(std::move(key), std::move(val));
    return o.back().second;
  }
  void erase(std::string_view key) {
    if (!is_obj()) return;
    auto& o = std::get<JsonObject>(v_);
    for (auto it = o.begin(); it != o.end(); ++it)
      if (it->first == key) {
        o.erase(it);
        return;
      }
  }

  // ---- serialization ----
  void dump(std::string& out) const {
    switch (v_.index()) {
      case 0: out += "null"; break;
      case 1: out += std::get<bool>(v_) ? "true" : "false"; break;
      case 2: {
        double d = std::get<double>(v_);
        char buf[32];
        if (d == (long long)d && std::fabs(d) < 1e15)
          snprintf(buf, sizeof buf, "%lld", (long long)d);
        else
          snprintf(buf, sizeof buf, "%.17g", d);
        out += buf;
        break;
      }
      case 3: escape(std::get<std::string>(v_), out); break;
      case 4: {
        out += '[';
        bool first = true;
        for (auto& e : std::get<JsonArray>(v_)) {
          if (!first) out += ',';
          first = false;
          e.dump(out);
        }
        out += ']';
        break;
      }
      case 5: {
        out += '{';
        bool first = true;
        for (auto& [k, val] : std::get<JsonObject>(v_)) {
          if (!first) out += ',';
          first = false;
          escape(k, out);
          out += ':';
          val.dump(out);
        }
        out += '}';
        break;
      }
    }
  }
  std::string dump() const {
    std::string s;
    dump(s);
    return s;
  }

  static void escape(std::string_view s, std::string& out) {
    out += '"';
    for (unsigned char c : s) {
      switch (c) {
        case '"': out += "\\\""; break;
        case '\\': out += "\\\\"; break;
        case '\n': out += "\\n"; break;
        case '\r': out += "\\r"; break;
        case '\t': out += "\\t"; break;
        case '\b': out += "\\b"; break;
        case '\f': out += "\\f"; break;
        default:
          if (c < 0x20) {
            char buf[8];
            snprintf(buf, sizeof buf, "\\u%04x", c);
            out += buf;
          } else {
            out += (char)c;
          }
      }
    }
    out += '"';
  }

  // ---- parsing ----
  // Returns true on success.  On failure `err` holds a message.
  static bool parse(std::string_view s, Json& out, std::string* err = nullptr);

 private:
  Value v_;
};

namespace jsondetail {
struct Parser {
  const char* p;
  const char* end;
  int depth = 0;
  bool fail(std::string* err, const char* msg) {
    if (err) *err = msg;
    return false;
  }
  void ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r'))
      ++p;
  }
  bool value(Json& out, std::string* err) {
    if (++depth > 128) return fail(err, "too deep");
    ws();
    if (p >= end) return fail(err, "eof");
    bool ok = false;
    switch (*p) {
      case '{': ok = object(out, err); break;
      case '[': ok = array(out, err); break;
      case '"': {
        std::string s;
        ok = str(s, err);
        if (ok) out = Json(std::move(s));
        break;
      }
      case 't':
        if (end - p >= 4 && memcmp(p, "true", 4) == 0) {
          p += 4;
          out = Json(true);
          ok = true;
        } else
          return fail(err, "bad literal");
        break;
      case 'f':
        if (end - p >= 5 && memcmp(p, "false", 5) == 0) {
          p += 5;
          out = Json(false);
          ok = true;
        } else
          return fail(err, "bad literal");
        break;
      case 'n':
        if (end - p >= 4 && memcmp(p, "null", 4) == 0) {
          p += 4;
          out = Json(nullptr);
          ok = true;
        } else
          return fail(err, "bad literal");
        break;
      default: ok = number(out, err);
    }
    --depth;
    return ok;
  }
  bool number(Json& out, std::string* err) {
    char* endp = nullptr;
    double d = strtod(p, &endp);
    if (endp == p) return fail(err, "bad number");
    p = endp;
    out = Json(d);
    return true;
  }
  bool str(std::string& out, std::string* err) {
    ++p;  // opening quote
    while (p < end) {
      unsigned char c = *p;
      if (c == '"') {
        ++p;
        return true;
      }
      if (c == '\\') {
        if (p + 1 >= end) return fail(err, "bad escape");
        ++p;
        switch (*p) {
          case '"': out += '"'; break;
          case '\\': out += '\\'; break;
          case '/': out += '/'; break;
          case 'n': out += '\n'; break;
          case 'r': out += '\r'; break;
          case 't': out += '\t'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'u': {
            if (end - p < 5) return fail(err, "bad \\u");
            auto hex4 = [&](const char* q) -> int {
              int v = 0;
              for (int i = 0; i < 4; ++i) {
                char h = q[i];
                v <<= 4;
                if (h >= '0' && h <= '9') v |= h - '0';
                else if (h >= 'a' && h <= 'f') v |= h - 'a' + 10;
                else if (h >= 'A' && h <= 'F') v |= h - 'A' + 10;
                else return -1;
              }
              return v;
            };
            int cp = hex4(p + 1);
            if (cp < 0) return fail(err, "bad \\u");
            p += 4;
            if (cp >= 0xD800 && cp <= 0xDBFF && end - p >= 7 &&
                p[1] == '\\' && p[2] == 'u') {
              int lo = hex4(p + 3);
              if (lo >= 0xDC00 && lo <= 0xDFFF) {
                cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
                p += 6;
              }
            }
            // utf8 encode
            if (cp < 0x80) out += (char)cp;
            else if (cp < 0x800) {
              out += (char)(0xC0 | (cp >> 6));
              out += (char)(0x80 | (cp & 63));
            } else if (cp < 0x10000) {
              out += (char)(0xE0 | (cp >> 12));
              out += (char)(0x80 | ((cp >> 6) & 63));
              out += (char)(0x80 | (cp & 63));
            } else {
              out += (char)(0xF0 | (cp >> 18));
              out += (char)(0x80 | ((cp >> 12) & 63));
              out += (char)(0x80 | ((cp >> 6) & 63));
              out += (char)(0x80 | (cp & 63));
            }
            break;
          }
          default: return fail(err, "bad escape");
        }
        ++p;
      } else {
        out += (char)c;
        ++p;
      }
    }
    return fail(err, "unterminated string");
  }
  bool object(Json& out, std::string* err) {
    ++p;
    JsonObject o;
    ws();
    if (p < end && *p == '}') {
      ++p;
      out = Json(std::move(o));
      return true;
    }
    while (p < end) {
      ws();
      if (p >= end || *p != '"') return fail(err, "expected key");
      std::string key;
      if (!str(key, err)) return false;
      ws();
      if (p >= end || *p != ':') return fail(err, "expected :");
      ++p;
      Json val;
      if (!value(val, err)) return false;
      o.emplace_back(std::move(key), std::move(val));
      ws();
      if (p < end && *p == ',') {
        ++p;
        continue;
      }
      if (p < end && *p == '}') {
        ++p;
        out = Json(std::move(o));
        return true;
      }
      return fail(err, "expected , or }");
    }
    return fail(err, "unterminated object");
  }
  bool array(Json& out, std::string* err) {
    ++p;
    JsonArray a;
    ws();
    if (p < end && *p == ']') {
      ++p;
      out = Json(std::move(a));
      return true;
    }
    while (p < end) {
      Json val;
      if (!value(val, err)) return false;
      a.emplace_back(std::move(val));
      ws();
      if (p < end && *p == ',') {
        ++p;
        continue;
      }
      if (p < end && *p == ']') {
        ++p;
        out = Json(std::move(a));
        return true;
      }
      return fail(err, "expected , or ]");
    }
    return fail(err, "unterminated array");
  }
};
}  // namespace jsondetail

inline bool Json::parse(std::string_view s, Json& out, std::string* err) {
  jsondetail::Parser ps{s.data(), s.data() + s.size()};
  if (!ps.value(out, err)) return false;
  ps.ws();
  if (ps.p != ps.end) {
    if (err) *err = "trailing data";
    return false;
  }
  return true;
}

// ---------------------------------------------------------------------
// Hot-path raw helpers: operate on the raw body without building a DOM.
// ---------------------------------------------------------------------

// Find the value range of a top-level (depth-1) string field `"key":"..."`.
// Returns {start, len} of the value INSIDE the quotes, or {npos, 0}.
inline std::pair<size_t, size_t> find_raw_string_field(std::string_view body,
                                                       std::string_view key) {
  int depth = 0;
  bool in_str = false;
  bool key_match = false;
  size_t i = 0;
  const size_t n = body.size();
  while (i < n) {
    char c = body[i];
    if (in_str) {
      if (c == '\\') { i += 2; continue; }
      if (c == '"') in_str = false;
      ++i;
      continue;
    }
    switch (c) {
      case '"': {
        // string start; if depth==1 this may be a key
        size_t start = i + 1;
        size_t j = start;
        bool esc = false;
        while (j < n) {
          if (esc) { esc = false; ++j; continue; }
          if (body[j] == '\\') { esc = true; ++j; continue; }
          if (body[j] == '"') break;
          ++j;
        }
        if (j >= n) return {std::string_view::npos, 0};
        std::string_view sv = body.substr(start, j - start);
        i = j + 1;
        // peek for ':' at depth 1 -> key
        size_t k = i;
        while (k < n && (body[k] == ' ' || body[k] == '\t' ||
                         body[k] == '\n' || body[k] == '\r'))
          ++k;
        if (depth == 1 && k < n && body[k] == ':') {
          key_match = (sv == key);
          i = k + 1;
          // skip ws then, if value is a string and key matched, return it
          while (i < n && (body[i] == ' ' || body[i] == '\t' ||
                           body[i] == '\n' || body[i] == '\r'))
            ++i;
          if (key_match && i < n && body[i] == '"') {
            size_t vs = i + 1, vj = vs;
            bool e2 = false;
            while (vj < n) {
              if (e2) { e2 = false; ++vj; continue; }
              if (body[vj] == '\\') { e2 = true; ++vj; continue; }
              if (body[vj] == '"') break;
              ++vj;
            }
            return {vs, vj - vs};
          }
          key_match = false;
        }
        continue;
      }
      case '{': case '[': ++depth; ++i; break;
      case '}': case ']': --depth; ++i; break;
      default: ++i;
    }
  }
  return {std::string_view::npos, 0};
}

// Extract `"key":"value"` at top level from raw JSON (no unescaping of
// simple values; returns empty when missing).
inline std::string raw_string_field(std::string_view body,
                                    std::string_view key) {
  auto [pos, len] = find_raw_string_field(body, key);
  if (pos == std::string_view::npos) return "";
  return std::string(body.substr(pos, len));
}

}  // namespace gw
