#include "cpilot/decode.hpp"

#include <cstdlib>

namespace cpilot {
namespace decode {

bool checkKeys(const Json& obj, const std::set<std::string>& allowed,
               std::string* err) {
  if (!obj.isObject()) return true;
  std::string bad;
  for (auto& kv : obj.object()) {
    if (allowed.count(kv.first) == 0) {
      if (!bad.empty()) bad += ", ";
      bad += kv.first;
    }
  }
  if (!bad.empty()) {
    *err = "invalid keys: " + bad;
    return false;
  }
  return true;
}

bool toInt(const Json& v, int* out) {
  if (v.isInt()) {
    *out = (int)v.asInt();
    return true;
  }
  if (v.isDouble()) {
    *out = (int)v.asDouble();
    return true;
  }
  if (v.isBool()) {
    *out = v.boolean() ? 1 : 0;
    return true;
  }
  if (v.isString()) {
    char* end = nullptr;
    long val = strtol(v.str().c_str(), &end, 10);
    if (end && *end == '\0' && !v.str().empty()) {
      *out = (int)val;
      return true;
    }
  }
  return false;
}

bool toString(const Json& v, std::string* out) {
  if (v.isString()) {
    *out = v.str();
    return true;
  }
  if (v.isInt()) {
    *out = std::to_string(v.asInt());
    return true;
  }
  if (v.isDouble()) {
    char buf[40];
    snprintf(buf, sizeof(buf), "%g", v.asDouble());
    *out = buf;
    return true;
  }
  if (v.isBool()) {
    *out = v.boolean() ? "1" : "0";
    return true;
  }
  return false;
}

bool toBool(const Json& v, bool* out) {
  if (v.isBool()) {
    *out = v.boolean();
    return true;
  }
  if (v.isInt()) {
    *out = v.asInt() != 0;
    return true;
  }
  if (v.isString()) {
    const std::string& s = v.str();
    if (s == "true" || s == "1" || s == "t" || s == "T" || s == "True") {
      *out = true;
      return true;
    }
    if (s == "false" || s == "0" || s == "f" || s == "F" || s == "False" ||
        s.empty()) {
      *out = false;
      return true;
    }
  }
  return false;
}

bool toStrings(const Json& v, std::vector<std::string>* out) {
  out->clear();
  if (v.isNull()) return true;
  if (v.isString()) {
    out->push_back(v.str());
    return true;
  }
  if (v.isArray()) {
    for (auto& e : v.array()) {
      std::string s;
      if (!toString(e, &s)) return false;
      out->push_back(std::move(s));
    }
    return true;
  }
  return false;
}

}  // namespace decode
}  // namespace cpilot
