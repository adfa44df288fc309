// Process execution: fork/exec in its own process group, per-exec timeout
// (SIGKILL the group), SIGTERM on cancel, wrapped-or-raw log piping,
// CONTAINERPILOT_{NAME}_PID env, and exit-event publication.
// Parity: /root/reference/commands/commands.go:24-188, args.go:12-31.
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "cpilot/events.hpp"
#include "cpilot/json.hpp"
#include "cpilot/loop.hpp"
#include "cpilot/timing.hpp"

namespace cpilot {

// ParseArgs: string -> space-split; array -> strings.
// Returns false + err on zero-length args. (commands/args.go:12-31)
bool parseArgs(const Json& raw, std::string* exec,
               std::vector<std::string>* args, std::string* err);

class Command : public std::enable_shared_from_this<Command> {
 public:
  Command(std::string execPath, std::vector<std::string> args, Duration timeout,
          bool raw, std::string logField);
  Command(const Command&) = delete;
  Command& operator=(const Command&) = delete;

  const std::string& name() const { return name_; }
  void setName(std::string name) { name_ = std::move(name); }
  const std::string& execPath() const { return exec_; }

  // Name formatted for the CONTAINERPILOT_{NAME}_PID env var
  // (commands/commands.go:59-81)
  std::string envName() const;

  // Run asynchronously; publishes ExitSuccess/ExitFailed{name} (+Error) on
  // exit. If already running, one run is queued (the reference blocks a
  // goroutine on a lock; we bound the queue at 1). The bus is captured by
  // shared_ptr so exit events from a previous config generation's children
  // land on that generation's (drained) bus, never a dangling one.
  void run(Loop& loop, std::shared_ptr<Bus> bus);

  // SIGTERM / SIGKILL the process group (commands/commands.go:172-188)
  void term();
  void kill();

  bool running() const { return running_; }
  pid_t pid() const { return pid_; }

 private:
  void spawn(Loop& loop, std::shared_ptr<Bus> bus);
  void onExit(Loop& loop, std::shared_ptr<Bus> bus, int status);

  std::string name_;
  std::string exec_;
  std::vector<std::string> args_;
  Duration timeout_;
  bool raw_;              // raw log passthrough (no wrapping)
  std::string logField_;  // "job" or "check" label value for wrapped logs
  std::string pidEnvName_;  // cached CONTAINERPILOT_{NAME}_PID

  pid_t pid_ = -1;
  bool running_ = false;
  bool pendingRun_ = false;
  int pendingSignal_ = 0;  // term/kill requested while spawn in flight
  Loop* loop_ = nullptr;
  uint64_t timeoutTimer_ = 0;
  TimePoint tReq_{}, tCb_{};  // spawn-RTT probe (CPILOT_LOOP_DEBUG)
  int logFd_ = -1;
  std::string logBuf_;
};

using CommandPtr = std::shared_ptr<Command>;

// Build a Command from raw config (exec field). fields=false means raw
// logging. Returns nullptr + err on parse failure.
// Mark the cached base-environment snapshot stale. Must be called on
// the loop thread after any setenv/unsetenv (PutEnviron endpoint,
// reload-time CONTAINERPILOT_*_IP updates) so subsequent spawns see the
// new environment.
void commandEnvInvalidate();

CommandPtr newCommand(const Json& rawExec, Duration timeout, bool raw,
                      const std::string& logField, std::string* err);

}  // namespace cpilot
