#include "cpilot/spawner.hpp"

#include <signal.h>
#include <spawn.h>
#include <unistd.h>

extern char** environ;

namespace cpilot {

Spawner& Spawner::global() {
  // intentionally leaked: destroying the condvar/mutex at static
  // destruction while the spawner thread waits on them deadlocks exit
  static Spawner* s = new Spawner();
  return *s;
}

Spawner::Spawner() {
  // deliberately ONE thread: it serializes launches, which acts as
  // natural backpressure past saturation. A 4-thread pool was measured
  // to push ~2x the launch rate at 5x baseline load but let completion
  // bursts flood the reactor (p99 dispatch 1 ms -> 50+ ms); a supervisor
  // should shed overlapping checks (reference single-instance exec
  // semantics) rather than trade latency for past-saturation throughput.
  threads_.emplace_back([this] { threadMain(); });
  threads_.back().detach();  // process-lifetime singleton
}

void Spawner::spawn(Loop& loop, std::string execPath,
                    std::vector<std::string> args,
                    std::vector<std::string> env, int stdioFd, SpawnCb cb) {
  {
    std::lock_guard<std::mutex> l(mu_);
    queue_.push_back(Request{&loop, std::move(execPath), std::move(args),
                             std::move(env), stdioFd, std::move(cb)});
  }
  cv_.notify_one();
}

void Spawner::threadMain() {
  while (true) {
    Request req;
    {
      std::unique_lock<std::mutex> l(mu_);
      cv_.wait(l, [this] { return !queue_.empty(); });
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
    for (auto& e : req.env) envp.push_back(const_cast<char*>(e.c_str()));
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
