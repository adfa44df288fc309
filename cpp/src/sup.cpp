// PID-1 supervisor: re-exec ourselves as a worker child, stay behind as a
// minimal reaper and signal forwarder. Keeping the reaper a separate
// process means the worker's waitpid() never races the wait4(-1) loop.
// Parity: /root/reference/sup/sup.go:15-92.
#include "cpilot/sup.hpp"

#include <signal.h>
#include <sys/prctl.h>
#include <cerrno>
#include <sys/wait.h>
#include <unistd.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>

namespace cpilot {

int supRun(int argc, char** argv) {
  // As PID 1 orphans reparent to us by definition; when forced into sup
  // mode without PID 1 (CPILOT_FORCE_SUP — e.g. running under a
  // container init that doesn't reap, or the unprivileged sup tests),
  // become a subreaper so orphaned grandchildren still land here
  prctl(PR_SET_CHILD_SUBREAPER, 1, 0, 0, 0);
  pid_t worker = fork();
  if (worker < 0) {
    fprintf(stderr, "failed to start ContainerPilot worker process: %s\n",
            strerror(errno));
    return 1;
  }
  if (worker == 0) {
    // child: continue as the worker (not PID 1 anymore)
    return -1;  // sentinel: caller proceeds with normal main
  }

  // parent: block the signals we forward/reap and wait synchronously
  sigset_t mask;
  sigemptyset(&mask);
  sigaddset(&mask, SIGINT);
  sigaddset(&mask, SIGTERM);
  sigaddset(&mask, SIGHUP);
  sigaddset(&mask, SIGUSR1);
  sigaddset(&mask, SIGUSR2);
  sigaddset(&mask, SIGCHLD);
  sigprocmask(SIG_BLOCK, &mask, nullptr);

  int workerStatus = 0;
  bool workerExited = false;
  while (!workerExited) {
    siginfo_t si;
    if (sigwaitinfo(&mask, &si) < 0) {
      if (errno == EINTR) continue;
      break;
    }
    switch (si.si_signo) {
      case SIGINT:
      case SIGTERM:
      case SIGHUP:
      case SIGUSR1:
      case SIGUSR2:
        kill(worker, si.si_signo);  // pass-thru (sup/sup.go:32-57)
        break;
      case SIGCHLD: {
        // reap everything reparented to PID 1 (sup/sup.go:73-92)
        while (true) {
          int status = 0;
          pid_t pid = waitpid(-1, &status, WNOHANG);
          if (pid <= 0) break;
          if (pid == worker) {
            workerExited = true;
            workerStatus = status;
          }
        }
        break;
      }
    }
  }
  if (WIFEXITED(workerStatus)) return WEXITSTATUS(workerStatus);
  if (WIFSIGNALED(workerStatus)) return 128 + WTERMSIG(workerStatus);
  return 0;
}

}  // namespace cpilot
