// Duration parsing with the reference's rules:
//  - bare integers (any numeric JSON value) are seconds
//  - strings with Go units ("1m30s", "500ms", "1.5h") use Go's grammar
//  - unit-less numeric strings are seconds
// Parity: /root/reference/config/timing/duration.go:13-58.
#pragma once

#include <chrono>
#include <stdexcept>
#include <string>

#include "cpilot/json.hpp"

namespace cpilot {

using Duration = std::chrono::nanoseconds;

// Parse a Go-style duration string: sequence of decimal[unit] where unit is
// one of ns, us, µs, μs, ms, s, m, h. Throws std::runtime_error on bad input.
Duration parseGoDuration(const std::string& s);

// ParseDuration semantics of timing.ParseDuration: numbers = seconds,
// numeric strings = seconds, unit strings = Go grammar.
Duration parseDuration(const Json& v);

// GetTimeout: empty string -> 0, otherwise parseDuration.
Duration getTimeout(const std::string& s);

// Render a duration the way Go's Duration.String() would for whole seconds
// ("5s"); used for Consul TTL strings.
std::string secondsString(int seconds);

}  // namespace cpilot
