// Spawner pool: posix_spawnp runs off the reactor so bursts of
// health-check launches never delay event dispatch.
//
// Sizing: one posix_spawn("/bin/true") call costs ~0.2-0.5 ms on the
// parent side (vfork suspends only the calling thread until exec), so a
// single thread tops out at ~2-4k launches/s — below the >=10k
// published-events/s target (each completed check publishes two
// events). The pool defaults to nproc/4 clamped to [2,6] and can be
// pinned with CPILOT_SPAWNER_THREADS. Overload backpressure is
// preserved: a heartbeat tick whose check is still pending is skipped
// (single-instance exec semantics), and completions are batch-paced
// into the reactor by Loop::drainPosted's bounded batches, which is
// what kept a naive pool from blowing p99 dispatch latency (see
// profiles/capacity.md).
//
// Spawn completions are posted back onto the loop; Loop::watchChild
// handles the SIGCHLD-before-completion race via its unclaimed-exit map.
#pragma once

#include <sys/types.h>

#include <condition_variable>
#include <deque>
#include <functional>
#include <memory>
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

  // stdioFd >= 0 is dup2'd onto the child's stdout+stderr and closed
  // after the spawn completes. baseEnv is an immutable shared snapshot
  // of the environment taken ON THE LOOP THREAD (posix_spawn must never
  // read the live environ while the reactor setenv()s); extraEnv are
  // per-spawn overlay entries ("K=V") appended after it.
  void spawn(Loop& loop, std::string execPath, std::vector<std::string> args,
             std::shared_ptr<const std::vector<std::string>> baseEnv,
             std::vector<std::string> extraEnv, int stdioFd, SpawnCb cb);

  int threads() const { return (int)threads_.size(); }

 private:
  Spawner();
  void threadMain(int index);

  struct Request {
    Loop* loop;
    std::string execPath;
    std::vector<std::string> args;
    std::shared_ptr<const std::vector<std::string>> baseEnv;
    std::vector<std::string> extraEnv;
    int stdioFd;
    SpawnCb cb;
  };

  std::vector<std::thread> threads_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<Request> queue_;
};

}  // namespace cpilot
