#include "cpilot/log.hpp"

#include <fcntl.h>
#include <sys/time.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <ctime>
#include <mutex>

namespace cpilot {
namespace logging {

namespace {

struct State {
  Level level = Level::Info;
  std::string format = "default";
  int fd = 1;            // stdout
  std::string filePath;  // non-empty when logging to a file
  std::mutex mu;
};

State& state() {
  static State s;
  return s;
}

const char* levelName(Level l) {
  switch (l) {
    case Level::Debug: return "debug";
    case Level::Info: return "info";
    case Level::Warn: return "warning";
    case Level::Error: return "error";
    case Level::Fatal: return "fatal";
  }
  return "info";
}

// RFC3339Nano-ish timestamp in local time, like the reference's default
// formatter (logging.go:111)
std::string timestamp() {
  struct timeval tv;
  gettimeofday(&tv, nullptr);
  struct tm tm;
  localtime_r(&tv.tv_sec, &tm);
  char buf[64];
  size_t n = strftime(buf, sizeof(buf), "%Y-%m-%dT%H:%M:%S", &tm);
  char frac[16];
  snprintf(frac, sizeof(frac), ".%09ld",
           static_cast<long>(tv.tv_usec) * 1000L);
  char tz[8];
  strftime(tz, sizeof(tz), "%z", &tm);
  // %z gives +0000; RFC3339 wants +00:00
  std::string tzs(tz);
  if (tzs.size() == 5) tzs = tzs.substr(0, 3) + ":" + tzs.substr(3);
  return std::string(buf, n) + frac + tzs;
}

std::string jsonEscape(const std::string& s) {
  std::string out;
  for (char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\n': out += "\\n"; break;
      case '\t': out += "\\t"; break;
      case '\r': out += "\\r"; break;
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
  return out;
}

void writeLine(Level l, const std::string& fields, const std::string& msg) {
  State& s = state();
  std::string line;
  if (s.format == "json") {
    line = "{\"level\":\"" + std::string(levelName(l)) + "\",\"msg\":\"" +
           jsonEscape(msg) + "\",\"time\":\"" + timestamp() + "\"}\n";
  } else if (s.format == "text") {
    line = "time=\"" + timestamp() + "\" level=" + levelName(l) + " msg=\"" +
           jsonEscape(msg) + "\"\n";
  } else {
    // default: "<ts>[ fields] <msg>" (logging.go:97-114)
    line = timestamp() + fields + " " + msg + "\n";
  }
  std::lock_guard<std::mutex> lock(s.mu);
  ssize_t unused = write(s.fd, line.data(), line.size());
  (void)unused;
}

}  // namespace

bool init(const Config& cfg, std::string* err) {
  State& s = state();
  std::string lv = cfg.level.empty() ? "INFO" : cfg.level;
  std::string fm = cfg.format.empty() ? "default" : cfg.format;
  std::string out = cfg.output.empty() ? "stdout" : cfg.output;

  for (auto& c : lv) c = tolower(c);
  Level level;
  if (lv == "debug") level = Level::Debug;
  else if (lv == "info") level = Level::Info;
  else if (lv == "warn" || lv == "warning") level = Level::Warn;
  else if (lv == "error") level = Level::Error;
  else if (lv == "fatal" || lv == "panic") level = Level::Fatal;
  else {
    if (err) *err = "Unknown log level '" + cfg.level + "'";
    return false;
  }

  std::string fmLower = fm;
  for (auto& c : fmLower) c = tolower(c);
  if (fmLower != "default" && fmLower != "text" && fmLower != "json") {
    if (err) *err = "Unknown log format '" + cfg.format + "'";
    return false;
  }

  int fd;
  std::string filePath;
  std::string outLower = out;
  for (auto& c : outLower) c = tolower(c);
  if (outLower == "stdout") {
    fd = 1;
  } else if (outLower == "stderr") {
    fd = 2;
  } else {
    fd = open(out.c_str(), O_WRONLY | O_CREAT | O_APPEND, 0644);
    if (fd < 0) {
      if (err)
        *err = "Error initializing log file '" + out + "': " + strerror(errno);
      return false;
    }
    filePath = out;
  }

  std::lock_guard<std::mutex> lock(s.mu);
  if (!s.filePath.empty() && s.fd > 2) close(s.fd);
  s.level = level;
  s.format = fmLower;
  s.fd = fd;
  s.filePath = filePath;
  return true;
}

Level level() { return state().level; }
void setLevel(Level l) { state().level = l; }

void reopen() {
  State& s = state();
  std::lock_guard<std::mutex> lock(s.mu);
  if (s.filePath.empty()) return;
  int fd = open(s.filePath.c_str(), O_WRONLY | O_CREAT | O_APPEND, 0644);
  if (fd >= 0) {
    if (s.fd > 2) close(s.fd);
    s.fd = fd;
  }
}

void logf(Level l, const char* fmt, ...) {
  if (l < state().level) return;
  char buf[4096];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  writeLine(l, "", buf);
  if (l == Level::Fatal) _exit(1);
}

void logFields(Level l, const std::string& job, int pid,
               const std::string& msg) {
  if (l < state().level) return;
  std::string fields;
  if (!job.empty()) fields += " " + job;
  if (pid > 0) fields += " " + std::to_string(pid);
  writeLine(l, fields, msg);
}

}  // namespace logging
}  // namespace cpilot
