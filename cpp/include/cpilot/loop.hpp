// Single-threaded epoll reactor: fd watchers, a timer heap backed by one
// timerfd, deferred callbacks, cross-thread task posting, and a child
// process table reaped on SIGCHLD.
//
// This replaces the reference's goroutine-per-component model
// (GOMAXPROCS=1, main.go:19): one reactor delivers the same observable
// semantics (every component sees every event in publish order) without
// locks on the hot path.
#pragma once

#include <sys/types.h>

#include <chrono>
#include <cstdint>
#include <deque>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <queue>
#include <vector>

namespace cpilot {

using Clock = std::chrono::steady_clock;
using TimePoint = Clock::time_point;
using Ns = std::chrono::nanoseconds;

// Debug helpers (CPILOT_LOOP_DEBUG): wrap a posted callback so its
// execution cost is accumulated under `tag`; the loop's 5 s phase
// report prints per-tag totals. No-ops (returns fn) when debug is off.
std::function<void()> timedItem(const char* tag, std::function<void()> fn);
void dumpItemTags();

// Reset the calling thread to default scheduling (SCHED_OTHER, nice 0).
// Defensive: background threads and spawned children must not inherit
// any elevated scheduling an operator may have started the daemon with.
void resetThreadScheduling();

class Loop {
 public:
  using FdCallback = std::function<void(uint32_t epollEvents)>;
  using TimerCallback = std::function<void()>;
  using ChildCallback = std::function<void(int waitStatus)>;

  Loop();
  ~Loop();

  Loop(const Loop&) = delete;
  Loop& operator=(const Loop&) = delete;

  // --- fds ---
  void watchFd(int fd, uint32_t events, FdCallback cb);
  void modifyFd(int fd, uint32_t events);
  void unwatchFd(int fd);

  // --- timers (ids are never 0) ---
  uint64_t addTimeout(Ns delay, TimerCallback cb);
  // initialDelay < 0 means "one full interval" (default phase)
  uint64_t addInterval(Ns interval, TimerCallback cb, Ns initialDelay = Ns(-1));
  void cancelTimer(uint64_t id);

  // --- deferred work (runs on the next loop iteration, FIFO) ---
  void defer(std::function<void()> fn);

  // --- cross-thread: safe to call from worker threads ---
  void post(std::function<void()> fn);

  // --- children ---
  // Registers interest in a child's exit. If the exit was already
  // recorded (a helper's exit notice can overtake the spawn-completion
  // callback), the stored status is delivered immediately.
  void watchChild(pid_t pid, ChildCallback cb);
  // call on SIGCHLD: waitpid(-1, WNOHANG) loop dispatching to callbacks
  // (direct children: spawn helpers; supervised processes are helper
  // children and arrive via notifyChildExit instead)
  void reapChildren();
  // dispatch an exit reported by a spawn helper (loop thread only; the
  // spawner posts these). Same claimed/unclaimed logic as reapChildren.
  void notifyChildExit(pid_t pid, int status);

  // Liveness-checked cross-thread post: delivers fn only if `loop` still
  // exists AND is the same incarnation (`id`): spawn-helper replies and
  // exit notices can outlive the Loop that requested the spawn, and a
  // NEW loop can be constructed at the SAME address (stack-allocated
  // per-test loops reuse frames), so pointer identity alone would
  // deliver a dead generation's callbacks into the wrong loop.
  static void postIfLive(Loop* loop, uint64_t id, std::function<void()> fn);
  uint64_t id() const { return id_; }

  void run();   // until stop()
  void stop();

  bool stopped() const { return stopped_; }

 private:
  struct Timer {
    uint64_t id;
    TimePoint deadline;
    Ns interval;  // zero for one-shot
    TimerCallback cb;
    bool canceled = false;
  };
  struct TimerCmp {
    bool operator()(const std::shared_ptr<Timer>& a,
                    const std::shared_ptr<Timer>& b) const {
      return a->deadline > b->deadline;
    }
  };

  void armTimerFd();
  void fireDueTimers();
  void drainDeferred();
  void drainPosted();

  uint64_t id_ = 0;  // unique per Loop incarnation
  int epfd_ = -1;
  int timerfd_ = -1;
  int wakeupFds_[2] = {-1, -1};  // pipe for post()
  bool stopped_ = false;

  std::map<int, FdCallback> fdCallbacks_;
  std::priority_queue<std::shared_ptr<Timer>,
                      std::vector<std::shared_ptr<Timer>>, TimerCmp>
      timers_;
  std::map<uint64_t, std::shared_ptr<Timer>> timersById_;
  uint64_t nextTimerId_ = 1;

  std::deque<std::function<void()>> deferred_;

  std::mutex postedMu_;
  std::deque<std::function<void()>> posted_;

  std::map<pid_t, ChildCallback> children_;
  std::map<pid_t, int> unclaimedExits_;
};

}  // namespace cpilot
