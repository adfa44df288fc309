// cpilot-spawn-helper: a tiny exec'd process that does posix_spawn on
// behalf of the daemon and reaps the resulting children.
//
// See cpilot/spawnproto.hpp for why this exists (constant-cost, fully
// parallel spawning isolated from the daemon's mm and fd table).
//
// Lifecycle: reads kSpawnRequest datagrams on fd 3, posix_spawns each
// (own process group, optional stdio fd dup2'd onto stdout+stderr),
// answers with kSpawnReply, and streams kChildExit notices as its
// children are reaped. Exits on socket EOF (daemon gone); remaining
// children reparent to PID 1 (sup / container init) which reaps them.
#include <errno.h>
#include <fcntl.h>
#include <poll.h>
#include <signal.h>
#include <sched.h>
#include <spawn.h>
#include <sys/resource.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/wait.h>
#include <unistd.h>

#include <vector>

#include "cpilot/spawnproto.hpp"

using namespace cpilot::spawnproto;

namespace {

int selfPipe[2] = {-1, -1};

bool helperDebug() {
  const char* v = getenv("CPILOT_SPAWN_DEBUG");
  static bool on = v && v[0] && !(v[0] == '0' && !v[1]);
  return on;
}

void onSigchld(int) {
  char b = 1;
  ssize_t unused = write(selfPipe[1], &b, 1);
  (void)unused;
}

bool sendAll(int fd, const void* buf, size_t len) {
  while (true) {
    ssize_t n = send(fd, buf, len, MSG_NOSIGNAL);
    if (n == (ssize_t)len) return true;
    if (n < 0 && (errno == EINTR)) continue;
    if (n < 0 && errno == EAGAIN) {
      struct pollfd p{fd, POLLOUT, 0};
      poll(&p, 1, 1000);
      continue;
    }
    return false;  // daemon gone
  }
}

void reapAndNotify(int sock) {
  while (true) {
    int status = 0;
    pid_t pid = waitpid(-1, &status, WNOHANG);
    if (pid <= 0) return;
    if (helperDebug())
      fprintf(stderr, "[helper %d] reaped pid %d\n", getpid(), (int)pid);
    ExitNotice note{kChildExit, 0, (int32_t)pid, (int32_t)status};
    if (!sendAll(sock, &note, sizeof(note))) _exit(0);
  }
}

void handleRequest(int sock, const char* buf, size_t len, int stdioFd) {
  if (len < sizeof(RequestHeader)) return;
  RequestHeader hdr;
  memcpy(&hdr, buf, sizeof(hdr));
  const char* p = buf + sizeof(hdr);
  const char* end = buf + len;

  std::vector<const char*> argv, envp;
  for (uint32_t i = 0; i < hdr.argc + hdr.envc && p < end; i++) {
    if (i < hdr.argc)
      argv.push_back(p);
    else
      envp.push_back(p);
    p += strnlen(p, end - p) + 1;
  }
  Reply reply{kSpawnReply, hdr.reqId, -1, EINVAL};
  if (argv.size() == hdr.argc && envp.size() == hdr.envc && !argv.empty()) {
    argv.push_back(nullptr);
    envp.push_back(nullptr);

    posix_spawnattr_t attr;
    posix_spawnattr_init(&attr);
    sigset_t empty;
    sigemptyset(&empty);
    posix_spawnattr_setsigmask(&attr, &empty);
    posix_spawnattr_setpgroup(&attr, 0);  // own process group
    short flags = POSIX_SPAWN_SETPGROUP | POSIX_SPAWN_SETSIGMASK;
#ifdef POSIX_SPAWN_USEVFORK
    flags |= POSIX_SPAWN_USEVFORK;
#endif
    posix_spawnattr_setflags(&attr, flags);

    posix_spawn_file_actions_t actions;
    posix_spawn_file_actions_init(&actions);
    if (hdr.wantStdio && stdioFd >= 0) {
      posix_spawn_file_actions_adddup2(&actions, stdioFd, 1);
      posix_spawn_file_actions_adddup2(&actions, stdioFd, 2);
    }
    pid_t pid = -1;
    int rc = posix_spawnp(&pid, argv[0], &actions, &attr,
                          const_cast<char**>(argv.data()),
                          const_cast<char**>(envp.data()));
    posix_spawn_file_actions_destroy(&actions);
    posix_spawnattr_destroy(&attr);
    reply.pid = (rc == 0) ? (int32_t)pid : -1;
    reply.err = rc;
    if (helperDebug())
      fprintf(stderr, "[helper %d] spawned pid %d req %u\n", getpid(),
              (int)pid, hdr.reqId);
  }
  if (stdioFd >= 0) close(stdioFd);
  if (!sendAll(sock, &reply, sizeof(reply))) _exit(0);
}

}  // namespace

int cpilotSpawnHelperMain() {
  // the daemon's reactor may run SCHED_RR / negative nice; neither the
  // helper nor the processes it spawns should inherit that
  {
    struct sched_param sp;
    memset(&sp, 0, sizeof(sp));
    sched_setscheduler(0, SCHED_OTHER, &sp);
    setpriority(PRIO_PROCESS, 0, 0);
  }
  const int sock = kHelperFd;
  // The dup2 that placed the socket at fd 3 cleared CLOEXEC so it
  // survived OUR exec; restore it now or every process this helper
  // spawns inherits the control socket — which both leaks an fd into
  // user processes and keeps the socketpair alive after this helper
  // dies, so the daemon would never see EOF and never respawn it.
  fcntl(sock, F_SETFD, FD_CLOEXEC);
  // nonblocking: the drain loop must hit EAGAIN, not block, once the
  // pending wakeup bytes are consumed
  if (pipe2(selfPipe, O_NONBLOCK | O_CLOEXEC) != 0) _exit(1);

  struct sigaction sa;
  memset(&sa, 0, sizeof(sa));
  sa.sa_handler = onSigchld;
  sa.sa_flags = SA_RESTART | SA_NOCLDSTOP;
  sigaction(SIGCHLD, &sa, nullptr);
  signal(SIGPIPE, SIG_IGN);
  // the daemon blocks these for its signalfd; the helper must not
  // inherit a mask that blinds its SIGCHLD handler
  sigset_t none;
  sigemptyset(&none);
  sigprocmask(SIG_SETMASK, &none, nullptr);

  std::vector<char> buf(kMaxRequestBytes);
  char cmsgBuf[CMSG_SPACE(sizeof(int))];

  while (true) {
    struct pollfd fds[2] = {{sock, POLLIN, 0}, {selfPipe[0], POLLIN, 0}};
    int rc = poll(fds, 2, -1);
    if (helperDebug())
      fprintf(stderr, "[helper %d] poll rc=%d sock=%x pipe=%x\n", getpid(),
              rc, fds[0].revents, fds[1].revents);
    if (rc < 0) {
      if (errno == EINTR) {
        reapAndNotify(sock);
        continue;
      }
      _exit(1);
    }
    if (fds[1].revents & POLLIN) {
      char drain[256];
      while (read(selfPipe[0], drain, sizeof(drain)) > 0) {
      }
      reapAndNotify(sock);
    }
    if (fds[0].revents & (POLLIN | POLLHUP | POLLERR)) {
      struct iovec iov{buf.data(), buf.size()};
      struct msghdr msg;
      memset(&msg, 0, sizeof(msg));
      msg.msg_iov = &iov;
      msg.msg_iovlen = 1;
      msg.msg_control = cmsgBuf;
      msg.msg_controllen = sizeof(cmsgBuf);
      ssize_t n = recvmsg(sock, &msg, MSG_CMSG_CLOEXEC);
      if (helperDebug())
        fprintf(stderr, "[helper %d] recvmsg n=%zd errno=%d\n", getpid(), n,
                n < 0 ? errno : 0);
      if (n == 0) _exit(0);  // daemon closed: done
      if (n < 0) {
        if (errno == EINTR || errno == EAGAIN) continue;
        _exit(0);
      }
      int passedFd = -1;
      for (struct cmsghdr* c = CMSG_FIRSTHDR(&msg); c;
           c = CMSG_NXTHDR(&msg, c)) {
        if (c->cmsg_level == SOL_SOCKET && c->cmsg_type == SCM_RIGHTS)
          memcpy(&passedFd, CMSG_DATA(c), sizeof(int));
      }
      handleRequest(sock, buf.data(), (size_t)n, passedFd);
      // opportunistic reap keeps exit latency low under load
      reapAndNotify(sock);
    }
  }
}

#ifndef CPILOT_HELPER_NO_MAIN
int main() { return cpilotSpawnHelperMain(); }
#endif
