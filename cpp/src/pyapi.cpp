// C API for the Python harness (loaded with ctypes as
// containerpilot_amd/_native.so). Exposes the pure config-pipeline
// components so Python tests exercise the same native code paths the
// daemon runs.
#include <cstdlib>
#include <cstring>
#include <string>

#include "cpilot/config.hpp"
#include "cpilot/discovery.hpp"
#include "cpilot/json.hpp"
#include "cpilot/timing.hpp"
#include "cpilot/tmpl.hpp"
#include "cpilot/version.hpp"

using namespace cpilot;

namespace {
char* dupString(const std::string& s) {
  char* out = (char*)malloc(s.size() + 1);
  memcpy(out, s.c_str(), s.size() + 1);
  return out;
}
}  // namespace

extern "C" {

const char* cp_version() { return kVersion; }

void cp_free(char* p) { free(p); }

// Render a config template against the current environment.
// Returns malloc'd result; on error returns nullptr and fills *errOut.
char* cp_render_template(const char* text, char** errOut) {
  try {
    return dupString(renderTemplate(text));
  } catch (const std::exception& e) {
    if (errOut) *errOut = dupString(e.what());
    return nullptr;
  }
}

// Parse a duration (JSON5 value text: bare int or quoted string).
// Returns nanoseconds, or -1 on error.
long long cp_parse_duration_ns(const char* text) {
  try {
    Json v = parseJson5(text);
    return (long long)parseDuration(v).count();
  } catch (const std::exception&) {
    return -1;
  }
}

// Convert JSON5 text to strict JSON. nullptr + *errOut on parse error.
char* cp_json5_to_json(const char* text, char** errOut) {
  try {
    return dupString(parseJson5(text).dump());
  } catch (const JsonParseError& e) {
    if (errOut) *errOut = dupString(formatParseError(text, e));
    return nullptr;
  } catch (const std::exception& e) {
    if (errOut) *errOut = dupString(e.what());
    return nullptr;
  }
}

// Validate a full (already rendered) config. Returns nullptr when valid,
// else a malloc'd error message.
char* cp_validate_config(const char* text) {
  std::string err;
  auto cfg = newConfig(text, &err);
  if (cfg) return nullptr;
  return dupString(err);
}

// Resolve the consul endpoint ("scheme://host:port") that the given
// `consul` config value + the current CONSUL_* environment produce
// (discovery/config.go:29-61 + api.DefaultConfig env handling).
// Returns nullptr + *errOut on config error.
char* cp_consul_endpoint(const char* consulJson, char** errOut) {
  try {
    Json raw = parseJson5(consulJson);
    std::string err;
    auto backend = ConsulBackend::create(&raw, &err);
    if (!backend) {
      if (errOut) *errOut = dupString(err);
      return nullptr;
    }
    return dupString(backend->scheme() + "://" + backend->address());
  } catch (const std::exception& e) {
    if (errOut) *errOut = dupString(e.what());
    return nullptr;
  }
}

}  // extern "C"
