// Logging: levels, three formats (default/text/json), stdout/stderr/file
// output with SIGUSR1 reopen for logrotate.
// Parity: /root/reference/config/logger/logging.go:39-129.
#pragma once

#include <cstdlib>

#include <cstdarg>
#include <string>

namespace cpilot {
namespace logging {

enum class Level { Debug = 0, Info, Warn, Error, Fatal };

struct Config {
  std::string level = "INFO";
  std::string format = "default";  // default | text | json
  std::string output = "stdout";   // stdout | stderr | <file path>
};

// Apply a logging config; returns false + sets err on invalid level/format.
bool init(const Config& cfg, std::string* err);

Level level();
void setLevel(Level l);

// Reopen the log file (SIGUSR1 / logrotate support). No-op for stdout/stderr.
void reopen();

void logf(Level l, const char* fmt, ...) __attribute__((format(printf, 2, 3)));
// Log with job/pid fields (the reference's logrus fields for job output).
void logFields(Level l, const std::string& job, int pid, const std::string& msg);

#define LOG_DEBUG(...)                                         \
  do {                                                         \
    if (::cpilot::logging::level() <= ::cpilot::logging::Level::Debug) \
      ::cpilot::logging::logf(::cpilot::logging::Level::Debug, __VA_ARGS__); \
  } while (0)
#define LOG_INFO(...) ::cpilot::logging::logf(::cpilot::logging::Level::Info, __VA_ARGS__)
#define LOG_WARN(...) ::cpilot::logging::logf(::cpilot::logging::Level::Warn, __VA_ARGS__)
#define LOG_ERROR(...) ::cpilot::logging::logf(::cpilot::logging::Level::Error, __VA_ARGS__)

}  // namespace logging
}  // namespace cpilot

namespace cpilot {
// env flag helper: set and not "0"/"" means on
inline bool cpilotDebugEnv(const char* name) {
  const char* v = getenv(name);
  return v && v[0] && !(v[0] == '0' && !v[1]);
}
}  // namespace cpilot
