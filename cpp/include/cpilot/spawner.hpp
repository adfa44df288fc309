// Dedicated spawner thread: posix_spawnp runs off the reactor so a burst
// of health-check launches never delays event dispatch. Spawn completions
// are posted back onto the loop; Loop::watchChild handles the
// SIGCHLD-before-completion race via its unclaimed-exit map. See the
// constructor comment for why this is exactly one thread.
#pragma once

#include <sys/types.h>

#include <condition_variable>
#include <deque>
#include <functional>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "cpilot/loop.hpp"

namespace cpilot {

class Spawner {
 public:
  // callback runs ON THE LOOP THREAD: (pid, errno) — pid < 0 on failure
  using SpawnCb = std::function<void(pid_t pid, int err)>;

  static Spawner& global();

  // stdioFd >= 0 is dup2'd onto the child's stdout+stderr and closed in
  // the parent after the spawn completes. env is a snapshot of the
  // environment taken ON THE LOOP THREAD — posix_spawn must never read
  // the live environ while the reactor setenv()s (data race -> EFAULT).
  void spawn(Loop& loop, std::string execPath, std::vector<std::string> args,
             std::vector<std::string> env, int stdioFd, SpawnCb cb);

 private:
  Spawner();
  void threadMain();

  struct Request {
    Loop* loop;
    std::string execPath;
    std::vector<std::string> args;
    std::vector<std::string> env;
    int stdioFd;
    SpawnCb cb;
  };

  std::vector<std::thread> threads_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<Request> queue_;
};

}  // namespace cpilot
