// Spawner: hands process launches to a pool of pre-spawned helper
// processes over SOCK_SEQPACKET socketpairs (see spawnproto.hpp for the
// protocol and the measurements behind the design).
//
// Why not spawn from the daemon: vfork'd posix_spawns share the
// daemon's mm until exec, so concurrent spawns serialize on its mmap
// lock and stall the reactor's page faults; fork also copies the
// daemon's fd table (one log pipe per supervised job), making per-spawn
// cost grow with job count. Helpers are tiny exec'd processes: spawn
// cost is constant and parallel across helpers, and each helper reaps
// its own children, so the reactor never forks and never takes child
// SIGCHLD storms — exits stream back as messages the reader thread
// posts onto the requesting loop.
//
// Overload backpressure is preserved upstream: a heartbeat tick whose
// check is still pending is skipped (single-instance exec semantics),
// and completions/exits are batch-paced into the reactor by
// Loop::drainPosted's bounded batches.
#pragma once

#include <sys/types.h>

#include <cstdint>
#include <deque>
#include <functional>
#include <map>
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

  // stdioFd >= 0 is passed to the helper (SCM_RIGHTS), dup2'd onto the
  // child's stdout+stderr, and closed on this side after sending.
  // baseEnv is an immutable shared snapshot of the environment taken ON
  // THE LOOP THREAD; extraEnv are per-spawn overlay entries ("K=V")
  // appended after it.
  // extraEnv is serialized synchronously before spawn() returns, so a
  // reference to a loop-thread-owned vector is safe and avoids an
  // O(overlay) copy per spawn
  void spawn(Loop& loop, const std::string& execPath,
             const std::vector<std::string>& args,
             std::shared_ptr<const std::vector<std::string>> baseEnv,
             const std::vector<std::string>& extraEnv, int stdioFd,
             SpawnCb cb);

  int helperCount() const { return (int)helpers_.size(); }

  // Requests in flight (sent or queued, not yet replied). Used for
  // EARLY shedding: a health-check tick that would join a deep backlog
  // is skipped at tick time (same observable as the single-instance
  // skip — a shed check — but it keeps queues and round trips short,
  // so saturation degrades flat instead of collapsing).
  size_t backlog();
  bool overloaded();

 private:
  Spawner();
  struct Helper {
    int sock = -1;
    pid_t pid = -1;
    bool dead = false;
    // requests that hit EAGAIN (kernel buffer full); flushed by the
    // reader thread on POLLOUT (mu_ guards the deque, never held
    // across sendmsg)
    std::deque<std::pair<std::vector<char>, int>> overflow;  // buf, fd
    // serializes sendmsg on this socket between the loop thread and
    // the reader's overflow flush; held only around the syscall so a
    // long flush never blocks spawn admission (measured: a global
    // lock across flushes cost ~137 us per spawn at saturation)
    std::mutex sendMu;
  };
  struct Pending {
    Loop* loop;
    uint64_t loopId;
    SpawnCb cb;
    size_t helperIdx;
  };

  bool launchHelper(Helper* h);
  void flushOverflow(size_t idx);
  void readerMain();
  void handleMessage(size_t idx, const char* buf, size_t len);
  void helperDied(size_t idx);
  bool sendRequest(Helper& h, const std::vector<char>& buf, int fd);
  void wakeReader();

  std::string helperPath_;
  std::vector<std::unique_ptr<Helper>> helpers_;
  size_t nextHelper_ = 0;

  std::mutex mu_;  // guards pending_, overflow queues, helper respawn
  std::map<uint32_t, Pending> pending_;
  uint32_t nextReqId_ = 1;
  // pid -> (loop, incarnation id) for exit notices (reader thread only)
  std::map<pid_t, std::pair<Loop*, uint64_t>> pidLoops_;

  int wakeFds_[2] = {-1, -1};
  std::thread reader_;
};

}  // namespace cpilot
