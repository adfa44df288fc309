#include "cpilot/spawner.hpp"

#include <signal.h>
#include <spawn.h>
#include <unistd.h>

#include <cstdlib>

extern char** environ;

namespace cpilot {

namespace {

int defaultThreadCount() {
  if (const char* v = getenv("CPILOT_SPAWNER_THREADS")) {
    int n = atoi(v);
    if (n >= 1 && n <= 64) return n;
  }
  long ncpu = sysconf(_SC_NPROCESSORS_ONLN);
  if (ncpu < 1) ncpu = 1;
  int n = (int)(ncpu / 4);
  if (n < 2) n = 2;
  if (n > 6) n = 6;
  return n;
}

}  // namespace

Spawner& Spawner::global() {
  // intentionally leaked: destroying the condvar/mutex at static
  // destruction while spawner threads wait on them deadlocks exit
  static Spawner* s = new Spawner();
  return *s;
}

Spawner::Spawner() {
  int n = defaultThreadCount();
  for (int i = 0; i < n; i++) {
    threads_.emplace_back([this, i] { threadMain(i); });
    threads_.back().detach();  // process-lifetime singleton
  }
}

void Spawner::spawn(Loop& loop, std::string execPath,
                    std::vector<std::string> args,
                    std::shared_ptr<const std::vector<std::string>> baseEnv,
                    std::vector<std::string> extraEnv, int stdioFd,
                    SpawnCb cb) {
  {
    std::lock_guard<std::mutex> l(mu_);
    queue_.push_back(Request{&loop, std::move(execPath), std::move(args),
                             std::move(baseEnv), std::move(extraEnv), stdioFd,
                             std::move(cb)});
  }
  // notify_all: predicates differ per thread (depth gates); notify_one
  // could wake only a gated thread that immediately re-sleeps
  cv_.notify_all();
}

void Spawner::threadMain(int index) {
  // Depth gating: thread 0 always serves; thread i only engages once
  // the queue backs up past 2*i requests. Concurrent vfork'd spawns
  // contend on the parent's mm (exec of a CLONE_VM child takes the
  // shared mmap lock), which was measured to double the reactor's p99
  // dispatch latency at LIGHT load — so extra spawn concurrency is
  // bought only when a backlog actually needs it.
  const size_t gate = (size_t)(2 * index);
  while (true) {
    Request req;
    {
      std::unique_lock<std::mutex> l(mu_);
      cv_.wait(l, [this, gate] { return queue_.size() > gate; });
      req = std::move(queue_.front());
      queue_.pop_front();
    }

    posix_spawnattr_t attr;
    posix_spawnattr_init(&attr);
    sigset_t empty;
    sigemptyset(&empty);
    posix_spawnattr_setsigmask(&attr, &empty);  // undo the signalfd mask
    posix_spawnattr_setpgroup(&attr, 0);        // own process group
    short flags = POSIX_SPAWN_SETPGROUP | POSIX_SPAWN_SETSIGMASK;
#ifdef POSIX_SPAWN_USEVFORK
    flags |= POSIX_SPAWN_USEVFORK;
#endif
    posix_spawnattr_setflags(&attr, flags);

    posix_spawn_file_actions_t actions;
    posix_spawn_file_actions_init(&actions);
    if (req.stdioFd >= 0) {
      posix_spawn_file_actions_adddup2(&actions, req.stdioFd, 1);
      posix_spawn_file_actions_adddup2(&actions, req.stdioFd, 2);
    }

    std::vector<char*> argv;
    argv.push_back(const_cast<char*>(req.execPath.c_str()));
    for (auto& a : req.args) argv.push_back(const_cast<char*>(a.c_str()));
    argv.push_back(nullptr);
    std::vector<char*> envp;
    if (req.baseEnv) {
      envp.reserve(req.baseEnv->size() + req.extraEnv.size() + 1);
      for (auto& e : *req.baseEnv) envp.push_back(const_cast<char*>(e.c_str()));
    }
    for (auto& e : req.extraEnv) envp.push_back(const_cast<char*>(e.c_str()));
    envp.push_back(nullptr);

    pid_t pid = -1;
    int rc = posix_spawnp(&pid, req.execPath.c_str(), &actions, &attr,
                          argv.data(), envp.data());
    posix_spawn_file_actions_destroy(&actions);
    posix_spawnattr_destroy(&attr);
    if (req.stdioFd >= 0) close(req.stdioFd);

    SpawnCb cb = std::move(req.cb);
    pid_t resultPid = (rc == 0) ? pid : -1;
    req.loop->post([cb, resultPid, rc] { cb(resultPid, rc); });
  }
}

}  // namespace cpilot
