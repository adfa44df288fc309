#include "cpilot/json.hpp"

#include <cctype>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <sstream>

namespace cpilot {

namespace {

class Parser {
 public:
  explicit Parser(const std::string& s) : s_(s) {}

  Json parseDocument() {
    skipWs();
    Json v = parseValue();
    skipWs();
    if (pos_ != s_.size()) fail("unexpected trailing characters");
    return v;
  }

 private:
  const std::string& s_;
  size_t pos_ = 0;
  int depth_ = 0;

  [[noreturn]] void fail(const std::string& msg) {
    throw JsonParseError(msg, pos_);
  }

  bool eof() const { return pos_ >= s_.size(); }
  char peek() const { return eof() ? '\0' : s_[pos_]; }
  char get() {
    if (eof()) fail("unexpected end of input");
    return s_[pos_++];
  }

  void skipWs() {
    while (!eof()) {
      char c = s_[pos_];
      if (c == ' ' || c == '\t' || c == '\n' || c == '\r' || c == '\v' ||
          c == '\f') {
        pos_++;
      } else if (c == '/' && pos_ + 1 < s_.size()) {
        if (s_[pos_ + 1] == '/') {
          pos_ += 2;
          while (!eof() && s_[pos_] != '\n') pos_++;
        } else if (s_[pos_ + 1] == '*') {
          pos_ += 2;
          while (pos_ + 1 < s_.size() &&
                 !(s_[pos_] == '*' && s_[pos_ + 1] == '/'))
            pos_++;
          if (pos_ + 1 >= s_.size()) fail("unterminated block comment");
          pos_ += 2;
        } else {
          break;
        }
      } else {
        break;
      }
    }
  }

  Json parseValue() {
    if (++depth_ > 500) fail("nesting too deep");
    Json v = parseValueInner();
    depth_--;
    return v;
  }

  Json parseValueInner() {
    if (eof()) fail("unexpected end of input");
    char c = peek();
    switch (c) {
      case '{':
        return parseObject();
      case '[':
        return parseArray();
      case '"':
      case '\'':
        return Json(parseString());
      default:
        break;
    }
    if (c == '-' || c == '+' || c == '.' || isdigit((unsigned char)c))
      return parseNumber();
    if (match("true")) return Json(true);
    if (match("false")) return Json(false);
    if (match("null")) return Json(nullptr);
    if (match("Infinity")) return Json(INFINITY);
    if (match("NaN")) return Json(NAN);
    fail("invalid character");
  }

  bool match(const char* word) {
    size_t n = strlen(word);
    if (s_.compare(pos_, n, word) != 0) return false;
    // must not be followed by an identifier char
    if (pos_ + n < s_.size() &&
        (isalnum((unsigned char)s_[pos_ + n]) || s_[pos_ + n] == '_'))
      return false;
    pos_ += n;
    return true;
  }

  Json parseObject() {
    get();  // '{'
    JsonObject obj;
    skipWs();
    if (peek() == '}') {
      get();
      return Json(std::move(obj));
    }
    while (true) {
      skipWs();
      std::string key = parseKey();
      skipWs();
      if (get() != ':') {
        pos_--;
        fail("expected ':' after object key");
      }
      skipWs();
      Json value = parseValue();
      // duplicate keys: last one wins (Go map unmarshal semantics)
      bool replaced = false;
      for (auto& kv : obj) {
        if (kv.first == key) {
          kv.second = std::move(value);
          replaced = true;
          break;
        }
      }
      if (!replaced) obj.emplace_back(std::move(key), std::move(value));
      skipWs();
      char c = get();
      if (c == ',') {
        skipWs();
        if (peek() == '}') {
          get();
          break;
        }
        continue;
      }
      if (c == '}') break;
      pos_--;
      fail("expected ',' or '}' in object");
    }
    return Json(std::move(obj));
  }

  std::string parseKey() {
    char c = peek();
    if (c == '"' || c == '\'') return parseString();
    // JSON5 identifier key: letters, digits, _, $ (not starting with digit)
    if (!(isalpha((unsigned char)c) || c == '_' || c == '$'))
      fail("expected object key");
    std::string key;
    while (!eof()) {
      c = s_[pos_];
      if (isalnum((unsigned char)c) || c == '_' || c == '$') {
        key += c;
        pos_++;
      } else {
        break;
      }
    }
    return key;
  }

  Json parseArray() {
    get();  // '['
    JsonArray arr;
    skipWs();
    if (peek() == ']') {
      get();
      return Json(std::move(arr));
    }
    while (true) {
      skipWs();
      arr.push_back(parseValue());
      skipWs();
      char c = get();
      if (c == ',') {
        skipWs();
        if (peek() == ']') {
          get();
          break;
        }
        continue;
      }
      if (c == ']') break;
      pos_--;
      fail("expected ',' or ']' in array");
    }
    return Json(std::move(arr));
  }

  std::string parseString() {
    char quote = get();
    std::string out;
    while (true) {
      if (eof()) fail("unterminated string");
      char c = get();
      if (c == quote) break;
      if (c == '\n') fail("unescaped newline in string");
      if (c != '\\') {
        out += c;
        continue;
      }
      if (eof()) fail("unterminated escape");
      char e = get();
      switch (e) {
        case 'n': out += '\n'; break;
        case 't': out += '\t'; break;
        case 'r': out += '\r'; break;
        case 'b': out += '\b'; break;
        case 'f': out += '\f'; break;
        case 'v': out += '\v'; break;
        case '0': out += '\0'; break;
        case '\n': break;  // JSON5 line continuation
        case '\r':
          if (peek() == '\n') get();
          break;
        case 'x': {
          unsigned v = parseHex(2);
          out += (char)v;
          break;
        }
        case 'u': {
          unsigned cp = parseHex(4);
          // surrogate pair
          if (cp >= 0xD800 && cp <= 0xDBFF && peek() == '\\') {
            size_t save = pos_;
            get();
            if (peek() == 'u') {
              get();
              unsigned lo = parseHex(4);
              if (lo >= 0xDC00 && lo <= 0xDFFF) {
                cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
              } else {
                pos_ = save;
              }
            } else {
              pos_ = save;
            }
          }
          appendUtf8(out, cp);
          break;
        }
        default:
          out += e;  // \", \\, \/, \' and any other char maps to itself
      }
    }
    return out;
  }

  unsigned parseHex(int n) {
    unsigned v = 0;
    for (int i = 0; i < n; i++) {
      char c = get();
      v <<= 4;
      if (c >= '0' && c <= '9') v += c - '0';
      else if (c >= 'a' && c <= 'f') v += c - 'a' + 10;
      else if (c >= 'A' && c <= 'F') v += c - 'A' + 10;
      else { pos_--; fail("invalid hex digit"); }
    }
    return v;
  }

  static void appendUtf8(std::string& out, unsigned cp) {
    if (cp < 0x80) {
      out += (char)cp;
    } else if (cp < 0x800) {
      out += (char)(0xC0 | (cp >> 6));
      out += (char)(0x80 | (cp & 0x3F));
    } else if (cp < 0x10000) {
      out += (char)(0xE0 | (cp >> 12));
      out += (char)(0x80 | ((cp >> 6) & 0x3F));
      out += (char)(0x80 | (cp & 0x3F));
    } else {
      out += (char)(0xF0 | (cp >> 18));
      out += (char)(0x80 | ((cp >> 12) & 0x3F));
      out += (char)(0x80 | ((cp >> 6) & 0x3F));
      out += (char)(0x80 | (cp & 0x3F));
    }
  }

  Json parseNumber() {
    size_t start = pos_;
    bool neg = false;
    if (peek() == '+' || peek() == '-') {
      neg = (peek() == '-');
      pos_++;
    }
    if (match("Infinity")) return Json(neg ? -INFINITY : INFINITY);
    if (match("NaN")) return Json(NAN);
    // hex
    if (peek() == '0' && pos_ + 1 < s_.size() &&
        (s_[pos_ + 1] == 'x' || s_[pos_ + 1] == 'X')) {
      pos_ += 2;
      int64_t v = 0;
      bool any = false;
      while (!eof() && isxdigit((unsigned char)peek())) {
        char c = get();
        v = v * 16 + (c <= '9' ? c - '0' : (tolower(c) - 'a' + 10));
        any = true;
      }
      if (!any) fail("invalid hex number");
      return Json(neg ? -v : v);
    }
    bool isFloat = false;
    while (!eof()) {
      char c = peek();
      if (isdigit((unsigned char)c)) {
        pos_++;
      } else if (c == '.' || c == 'e' || c == 'E') {
        isFloat = true;
        pos_++;
        if ((c == 'e' || c == 'E') && (peek() == '+' || peek() == '-')) pos_++;
      } else {
        break;
      }
    }
    std::string num = s_.substr(start, pos_ - start);
    if (num.empty() || num == "-" || num == "+") fail("invalid number");
    errno = 0;
    if (!isFloat) {
      char* end = nullptr;
      long long v = strtoll(num.c_str(), &end, 10);
      if (errno == 0 && end && *end == '\0') return Json((int64_t)v);
    }
    char* end = nullptr;
    double d = strtod(num.c_str(), &end);
    if (!end || *end != '\0') fail("invalid number");
    return Json(d);
  }
};

void dumpString(const std::string& s, std::string& out) {
  out += '"';
  for (char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\n': out += "\\n"; break;
      case '\r': out += "\\r"; break;
      case '\t': out += "\\t"; break;
      case '\b': out += "\\b"; break;
      case '\f': out += "\\f"; break;
      default:
        if ((unsigned char)c < 0x20) {
          char buf[8];
          snprintf(buf, sizeof(buf), "\\u%04x", c);
          out += buf;
        } else {
          out += c;
        }
    }
  }
  out += '"';
}

void dumpValue(const Json& v, std::string& out) {
  switch (v.type()) {
    case Json::Type::Null:
      out += "null";
      break;
    case Json::Type::Bool:
      out += v.boolean() ? "true" : "false";
      break;
    case Json::Type::Int: {
      char buf[32];
      snprintf(buf, sizeof(buf), "%lld", (long long)v.asInt());
      out += buf;
      break;
    }
    case Json::Type::Double: {
      double d = v.asDouble();
      if (std::isnan(d) || std::isinf(d)) {
        out += "null";
      } else {
        char buf[40];
        snprintf(buf, sizeof(buf), "%.17g", d);
        out += buf;
      }
      break;
    }
    case Json::Type::String:
      dumpString(v.str(), out);
      break;
    case Json::Type::Array: {
      out += '[';
      bool first = true;
      for (auto& e : v.array()) {
        if (!first) out += ',';
        first = false;
        dumpValue(e, out);
      }
      out += ']';
      break;
    }
    case Json::Type::Object: {
      out += '{';
      bool first = true;
      for (auto& kv : v.object()) {
        if (!first) out += ',';
        first = false;
        dumpString(kv.first, out);
        out += ':';
        dumpValue(kv.second, out);
      }
      out += '}';
      break;
    }
  }
}

}  // namespace

std::string Json::dump() const {
  std::string out;
  dumpValue(*this, out);
  return out;
}

Json parseJson5(const std::string& text) { return Parser(text).parseDocument(); }

std::string formatParseError(const std::string& text,
                             const JsonParseError& err) {
  // locate line/col of err.offset, show previous + offending line and a caret
  // (semantics of config/config.go:203-232)
  size_t line = 1, lineStart = 0;
  for (size_t i = 0; i < err.offset && i < text.size(); i++) {
    if (text[i] == '\n') {
      line++;
      lineStart = i + 1;
    }
  }
  size_t col = err.offset - lineStart;
  size_t prevStart = std::string::npos;
  if (lineStart > 0) {
    size_t p = text.rfind('\n', lineStart >= 2 ? lineStart - 2 : 0);
    prevStart = (p == std::string::npos) ? 0 : p + 1;
  }
  auto lineText = [&](size_t start) {
    size_t end = text.find('\n', start);
    if (end == std::string::npos) end = text.size();
    return text.substr(start, end - start);
  };
  std::ostringstream out;
  out << "parse error at line:col [" << line << ":" << col
      << "]: " << err.what() << "\n";
  if (prevStart != std::string::npos && prevStart < lineStart) {
    char buf[16];
    snprintf(buf, sizeof(buf), "%5zu: ", line - 1);
    out << buf << lineText(prevStart) << "\n";
  }
  char buf[16];
  snprintf(buf, sizeof(buf), "%5zu: ", line);
  out << buf << lineText(lineStart) << "\n";
  out << std::string(7 + col, '-') << "^";
  return out.str();
}

}  // namespace cpilot
