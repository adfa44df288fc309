// Decode helpers matching the reference's mapstructure usage:
// WeaklyTypedInput coercions (strings <-> numbers) plus ErrorUnused
// (unknown object keys are hard errors).
// Parity: /root/reference/config/decode/decode.go:13-62.
#pragma once

#include <set>
#include <string>
#include <vector>

#include "cpilot/json.hpp"

namespace cpilot {
namespace decode {

// Unknown-key check: every key of `obj` must be in `allowed`.
// On failure sets err to "invalid keys: ..." like mapstructure ErrorUnused.
bool checkKeys(const Json& obj, const std::set<std::string>& allowed,
               std::string* err);

// weak int: accepts Int, Double (truncates), numeric String, Bool(0/1)
bool toInt(const Json& v, int* out);
// weak string: String, or number rendered as text
bool toString(const Json& v, std::string* out);
// weak bool: Bool, "true"/"false"/"1"/"0", 0/1
bool toBool(const Json& v, bool* out);
// ToStrings: null -> empty, string -> [s], array -> each element weakly
// stringified (decode.go:48-62)
bool toStrings(const Json& v, std::vector<std::string>* out);

}  // namespace decode
}  // namespace cpilot
