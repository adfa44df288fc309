#include "cpilot/command.hpp"

#include <fcntl.h>
#include <signal.h>
#include <sys/epoll.h>
#include <sys/wait.h>
#include <unistd.h>

#include <cctype>
#include <cstring>
#include <cstdio>
#include <map>

#include "cpilot/log.hpp"
#include "cpilot/spawner.hpp"

extern char** environ;

namespace cpilot {

namespace {

// CONTAINERPILOT_{NAME}_PID vars for running execs. Kept as an overlay
// merged into each spawn's env snapshot instead of calling setenv per
// exec: glibc's setenv leaks the replaced "NAME=value" string on every
// update (it may still be referenced), which at ~1000 execs/sec grew
// the daemon ~70 KB/s. Loop-thread only.
std::map<std::string, std::string>& pidEnvOverlay() {
  static std::map<std::string, std::string> overlay;
  return overlay;
}

// The base environment (environ minus overlay-shadowed names) is an
// immutable shared snapshot: at thousands of spawns/s, rebuilding ~50
// strings per spawn on the loop thread was measurable allocator churn.
// It only goes stale when environ itself changes (PutEnviron, reload
// IP/PID vars — commandEnvInvalidate()) or when the overlay KEY SET
// changes (value-only PID updates don't shadow anything new). The few
// overlay values are appended per spawn as extraEnv.
std::shared_ptr<const std::vector<std::string>>& baseEnvCache() {
  static std::shared_ptr<const std::vector<std::string>> cache;
  return cache;
}
bool& baseEnvDirty() {
  static bool dirty = true;
  return dirty;
}

std::shared_ptr<const std::vector<std::string>> baseEnvSnapshot() {
  if (!baseEnvDirty() && baseEnvCache()) return baseEnvCache();
  auto& overlay = pidEnvOverlay();
  auto env = std::make_shared<std::vector<std::string>>();
  for (char** e = ::environ; *e; e++) {
    const char* eq = strchr(*e, '=');
    if (eq && overlay.count(std::string(*e, eq - *e))) continue;
    env->emplace_back(*e);
  }
  baseEnvCache() = env;
  baseEnvDirty() = false;
  return baseEnvCache();
}

// The overlay is ALSO kept materialized as "K=V" strings, updated
// incrementally: with one PID var per in-flight exec (the check
// commands each have their own name), rebuilding this list per spawn
// was O(jobs) on the loop thread — measured 88 us/spawn at 2000 jobs,
// the cap on saturated throughput. spawn() serializes from a const
// reference to this vector synchronously, so no copy is ever taken.
std::vector<std::string>& overlayStrings() {
  static std::vector<std::string> strings;
  return strings;
}
std::map<std::string, size_t>& overlayIndex() {
  static std::map<std::string, size_t> index;  // key -> strings slot
  return index;
}

const std::vector<std::string>& overlayExtras() { return overlayStrings(); }

void overlaySet(const std::string& key, const std::string& value) {
  auto [it, inserted] = pidEnvOverlay().emplace(key, value);
  if (inserted) {
    overlayIndex()[key] = overlayStrings().size();
    overlayStrings().push_back(key + "=" + value);
    // the base snapshot only needs a rebuild if this name exists in
    // environ and must now be shadowed; CONTAINERPILOT_*_PID names
    // normally don't, so steady-state check traffic never rebuilds
    if (getenv(key.c_str())) baseEnvDirty() = true;
  } else {
    it->second = value;
    overlayStrings()[overlayIndex()[key]] = key + "=" + value;
  }
}

void overlayErase(const std::string& key) {
  if (pidEnvOverlay().erase(key)) {
    auto idxIt = overlayIndex().find(key);
    if (idxIt != overlayIndex().end()) {
      size_t slot = idxIt->second;
      overlayIndex().erase(idxIt);
      auto& strs = overlayStrings();
      if (slot + 1 != strs.size()) {
        // swap-pop; re-index the moved entry
        strs[slot] = std::move(strs.back());
        size_t eq = strs[slot].find('=');
        overlayIndex()[strs[slot].substr(0, eq)] = slot;
      }
      strs.pop_back();
    }
    if (getenv(key.c_str())) baseEnvDirty() = true;
  }
}

}  // namespace

void commandEnvInvalidate() { baseEnvDirty() = true; }

bool parseArgs(const Json& raw, std::string* execPath,
               std::vector<std::string>* args, std::string* err) {
  std::vector<std::string> all;
  if (raw.isString()) {
    // TrimSpace then split on single spaces (commands/args.go:14-18)
    std::string s = raw.str();
    size_t a = 0, b = s.size();
    while (a < b && isspace((unsigned char)s[a])) a++;
    while (b > a && isspace((unsigned char)s[b - 1])) b--;
    s = s.substr(a, b - a);
    if (!s.empty()) {
      std::string cur;
      for (char c : s) {
        if (c == ' ') {
          all.push_back(cur);
          cur.clear();
        } else {
          cur += c;
        }
      }
      all.push_back(cur);
    }
  } else if (raw.isArray()) {
    for (auto& e : raw.array()) {
      if (e.isString()) all.push_back(e.str());
      else all.push_back(e.dump());
    }
  } else if (!raw.isNull()) {
    *err = "unable to parse exec arguments";
    return false;
  }
  if (all.empty()) {
    *err = "received zero-length argument";
    return false;
  }
  *execPath = all[0];
  args->assign(all.begin() + 1, all.end());
  return true;
}

CommandPtr newCommand(const Json& rawExec, Duration timeout, bool raw,
                      const std::string& logField, std::string* err) {
  std::string execPath;
  std::vector<std::string> args;
  if (!parseArgs(rawExec, &execPath, &args, err)) return nullptr;
  return std::make_shared<Command>(execPath, args, timeout, raw, logField);
}

Command::Command(std::string execPath, std::vector<std::string> args,
                 Duration timeout, bool raw, std::string logField)
    : name_(execPath),
      exec_(std::move(execPath)),
      args_(std::move(args)),
      timeout_(timeout),
      raw_(raw),
      logField_(std::move(logField)) {}

std::string Command::envName() const {
  if (name_.empty()) return name_;
  // base name
  std::string name = name_;
  size_t slash = name.find_last_of('/');
  if (slash != std::string::npos) name = name.substr(slash + 1);
  // strip one extension
  size_t dot = name.find_last_of('.');
  if (dot != std::string::npos && dot > 0)
    name = name.substr(0, dot) + name.substr(name.size());
  // non-alphanumerics -> underscore, compact doubles
  std::string out;
  for (char c : name) {
    if (isalnum((unsigned char)c)) {
      out += (char)toupper((unsigned char)c);
    } else if (out.empty() || out.back() != '_') {
      out += '_';
    }
  }
  return out;
}

void Command::run(Loop& loop, std::shared_ptr<Bus> bus) {
  if (running_) {
    // the reference queues concurrent runs on a mutex
    // (commands/commands.go:92-93); we bound that queue at one
    LOG_DEBUG("%s already running, queueing run", name_.c_str());
    pendingRun_ = true;
    return;
  }
  spawn(loop, std::move(bus));
}

void Command::spawn(Loop& loop, std::shared_ptr<Bus> bus) {
  loop_ = &loop;
  LOG_DEBUG("%s.Run start", name_.c_str());

  int pipefds[2] = {-1, -1};
  if (!raw_) {
    if (pipe2(pipefds, O_CLOEXEC) != 0) {
      LOG_ERROR("unable to create pipe for %s: %s", name_.c_str(),
                strerror(errno));
      bus->publish(Event{EventCode::ExitFailed, name_});
      bus->publish(Event{EventCode::Error, strerror(errno)});
      return;
    }
  }

  running_ = true;
  pid_ = -1;
  pendingSignal_ = 0;
  tReq_ = Clock::now();
  auto self = shared_from_this();
  int readFd = pipefds[0];
  // snapshot env (+ the PID-var overlay) on this (the loop) thread; the
  // spawner must never read the live environ concurrently with setenv
  auto baseEnv = baseEnvSnapshot();
  // the spawner pool does the posix_spawnp so a burst of launches never
  // blocks event dispatch; completion lands back on the loop
  Spawner::SpawnCb compCb =
      [this, self, bus, readFd](pid_t pid, int err) {
        if (pid < 0) {
          LOG_ERROR("unable to start %s: %s", name_.c_str(), strerror(err));
          if (readFd >= 0) close(readFd);
          running_ = false;
          pendingRun_ = false;
          pendingSignal_ = 0;
          bus->publish(Event{EventCode::ExitFailed, name_});
          bus->publish(Event{EventCode::Error, strerror(err)});
          return;
        }
        pid_ = pid;
        tCb_ = Clock::now();
        if (pidEnvName_.empty())
          pidEnvName_ = "CONTAINERPILOT_" + envName() + "_PID";
        overlaySet(pidEnvName_, std::to_string(pid));

        if (!raw_) {
          logFd_ = readFd;
          fcntl(logFd_, F_SETFL, O_NONBLOCK);
          loop_->watchFd(logFd_, EPOLLIN | EPOLLHUP, [this, self](uint32_t) {
            char buf[4096];
            while (true) {
              ssize_t n = read(logFd_, buf, sizeof(buf));
              if (n > 0) {
                logBuf_.append(buf, n);
                size_t pos;
                while ((pos = logBuf_.find('\n')) != std::string::npos) {
                  logging::logFields(logging::Level::Info, logField_, pid_,
                                     logBuf_.substr(0, pos));
                  logBuf_.erase(0, pos + 1);
                }
              } else if (n == 0 || (n < 0 && errno != EAGAIN)) {
                if (!logBuf_.empty()) {
                  logging::logFields(logging::Level::Info, logField_, pid_,
                                     logBuf_);
                  logBuf_.clear();
                }
                loop_->unwatchFd(logFd_);
                close(logFd_);
                logFd_ = -1;
                return;
              } else {
                return;  // EAGAIN
              }
            }
          });
        }

        if (timeout_ > Duration(0)) {
          timeoutTimer_ = loop_->addTimeout(timeout_, [this, self] {
            LOG_WARN("%s timeout after %llds", name_.c_str(),
                     (long long)std::chrono::duration_cast<
                         std::chrono::seconds>(timeout_)
                         .count());
            timeoutTimer_ = 0;
            kill();
          });
        }

        // a term/kill that arrived while the spawn was in flight:
        // deliver before watchChild, which may synchronously run
        // onExit (and even a queued respawn) for an already-reaped pid
        if (pendingSignal_ != 0) {
          ::kill(-pid, pendingSignal_);
          pendingSignal_ = 0;
        }

        loop_->watchChild(pid, [this, self, bus](int status) {
          onExit(*loop_, bus, status);
        });
      };
  Spawner::global().spawn(loop, exec_, args_, std::move(baseEnv),
                          overlayExtras(), raw_ ? -1 : pipefds[1],
                          std::move(compCb));
}

void Command::onExit(Loop& loop, std::shared_ptr<Bus> bus, int status) {
  LOG_DEBUG("%s.Run end", name_.c_str());
  static const bool rttDebug = cpilotDebugEnv("CPILOT_LOOP_DEBUG");
  if (rttDebug) {
    auto now = Clock::now();
    double total =
        std::chrono::duration<double, std::milli>(now - tReq_).count();
    if (total > 50.0) {
      double execPhase =
          std::chrono::duration<double, std::milli>(now - tCb_).count();
      fprintf(stderr,
              "spawn rtt: %s total=%.0fms spawn_phase=%.0fms "
              "exec_phase=%.0fms\n",
              name_.c_str(), total, total - execPhase, execPhase);
    }
  }
  if (timeoutTimer_) {
    loop.cancelTimer(timeoutTimer_);
    timeoutTimer_ = 0;
  }
  if (logFd_ >= 0) {
    // drain any remaining output synchronously
    char buf[4096];
    ssize_t n;
    while ((n = read(logFd_, buf, sizeof(buf))) > 0) logBuf_.append(buf, n);
    size_t pos;
    while ((pos = logBuf_.find('\n')) != std::string::npos) {
      logging::logFields(logging::Level::Info, logField_, pid_,
                         logBuf_.substr(0, pos));
      logBuf_.erase(0, pos + 1);
    }
    if (!logBuf_.empty()) {
      logging::logFields(logging::Level::Info, logField_, pid_, logBuf_);
      logBuf_.clear();
    }
    loop.unwatchFd(logFd_);
    close(logFd_);
    logFd_ = -1;
  }

  if (!pidEnvName_.empty()) overlayErase(pidEnvName_);

  running_ = false;
  pid_ = -1;

  bool success = WIFEXITED(status) && WEXITSTATUS(status) == 0;
  if (success) {
    LOG_DEBUG("%s exited without error", name_.c_str());
    bus->publish(Event{EventCode::ExitSuccess, name_});
  } else {
    int code = WIFEXITED(status) ? WEXITSTATUS(status)
                                 : 128 + (WIFSIGNALED(status) ? WTERMSIG(status) : 0);
    LOG_ERROR("%s exited with error: exit status %d", name_.c_str(), code);
    bus->publish(Event{EventCode::ExitFailed, name_});
    bus->publish(Event{EventCode::Error,
                      name_ + ": exit status " + std::to_string(code)});
  }

  if (pendingRun_) {
    pendingRun_ = false;
    spawn(loop, bus);
  }
}

void Command::term() {
  LOG_DEBUG("%s.term", name_.c_str());
  if (pid_ > 0) {
    LOG_DEBUG("terminating command '%s' at pid: %d", name_.c_str(), pid_);
    ::kill(-pid_, SIGTERM);
  } else if (running_ && pendingSignal_ != SIGKILL) {
    pendingSignal_ = SIGTERM;  // spawn in flight
  }
}

void Command::kill() {
  LOG_DEBUG("%s.kill", name_.c_str());
  if (pid_ > 0) {
    LOG_DEBUG("killing command '%s' at pid: %d", name_.c_str(), pid_);
    ::kill(-pid_, SIGKILL);
  } else if (running_) {
    pendingSignal_ = SIGKILL;  // spawn in flight
  }
}

}  // namespace cpilot
