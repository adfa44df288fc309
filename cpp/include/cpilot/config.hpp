// Top-level config pipeline: read file -> render template against env ->
// parse JSON5 -> decode {consul, logging, stopTimeout, jobs, watches,
// telemetry, control} -> per-package validation. Unknown top-level keys
// are hard errors.
// Parity: /root/reference/config/config.go:24-269.
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "cpilot/control.hpp"
#include "cpilot/discovery.hpp"
#include "cpilot/jobs.hpp"
#include "cpilot/log.hpp"
#include "cpilot/telemetry.hpp"
#include "cpilot/watches.hpp"

namespace cpilot {

struct AppConfig {
  std::unique_ptr<ConsulBackend> discovery;
  logging::Config logConfig;
  int stopTimeout = 5;  // seconds (config/config.go:45-48)
  std::vector<std::shared_ptr<JobConfig>> jobs;
  std::vector<std::shared_ptr<WatchConfig>> watches;
  std::shared_ptr<TelemetryConfig> telemetry;  // null when disabled
  ControlConfig control;
};

// Load + render + parse + validate. Returns nullptr and sets err on any
// failure (messages mirror the reference's).
std::unique_ptr<AppConfig> loadConfig(const std::string& path,
                                      std::string* err);

// Parse a rendered config string (exposed for tests).
std::unique_ptr<AppConfig> newConfig(const std::string& rendered,
                                     std::string* err);

// -template handling: render the config and return it (config.go:67-88).
bool renderConfigFile(const std::string& configPath,
                      const std::string& outPath, std::string* err);

}  // namespace cpilot
