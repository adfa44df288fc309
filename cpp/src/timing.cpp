#include "cpilot/timing.hpp"

#include <cctype>
#include <cmath>
#include <cstdlib>

namespace cpilot {

Duration parseGoDuration(const std::string& s) {
  // Go time.ParseDuration grammar: [-+]?([0-9]*(\.[0-9]*)?[a-z]+)+
  // "0" is allowed without a unit.
  if (s.empty()) throw std::runtime_error("time: invalid duration " + s);
  size_t i = 0;
  bool neg = false;
  if (s[i] == '+' || s[i] == '-') {
    neg = (s[i] == '-');
    i++;
  }
  if (s.compare(i, std::string::npos, "0") == 0) return Duration(0);
  if (i == s.size()) throw std::runtime_error("time: invalid duration " + s);
  double totalNs = 0;
  while (i < s.size()) {
    size_t start = i;
    while (i < s.size() && (isdigit((unsigned char)s[i]) || s[i] == '.')) i++;
    if (i == start)
      throw std::runtime_error("time: invalid duration " + s);
    double v = strtod(s.substr(start, i - start).c_str(), nullptr);
    // unit
    size_t ustart = i;
    while (i < s.size() && !isdigit((unsigned char)s[i]) && s[i] != '.') i++;
    std::string unit = s.substr(ustart, i - ustart);
    double mult;
    if (unit == "ns") mult = 1;
    else if (unit == "us" || unit == "µs" || unit == "μs") mult = 1e3;
    else if (unit == "ms") mult = 1e6;
    else if (unit == "s") mult = 1e9;
    else if (unit == "m") mult = 60e9;
    else if (unit == "h") mult = 3600e9;
    else
      throw std::runtime_error("time: unknown unit \"" + unit +
                               "\" in duration " + s);
    totalNs += v * mult;
  }
  if (neg) totalNs = -totalNs;
  return Duration((int64_t)totalNs);
}

Duration parseDuration(const Json& v) {
  if (v.isInt()) return std::chrono::seconds(v.asInt());
  if (v.isDouble()) {
    // the reference only accepts integer types here; a JSON5 float is an
    // error ("unexpected duration of type float64")
    throw std::runtime_error("unexpected duration of type float64");
  }
  if (v.isString()) {
    const std::string& s = v.str();
    // integer-only string = seconds (duration.go:53-55)
    char* end = nullptr;
    long val = strtol(s.c_str(), &end, 10);
    if (!s.empty() && end && *end == '\0')
      return std::chrono::seconds(val);
    return parseGoDuration(s);
  }
  throw std::runtime_error("unexpected duration type");
}

Duration getTimeout(const std::string& s) {
  if (s.empty()) return Duration(0);
  return parseDuration(Json(s));
}

std::string secondsString(int seconds) {
  return std::to_string(seconds) + "s";
}

}  // namespace cpilot
