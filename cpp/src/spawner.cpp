#include "cpilot/spawner.hpp"

#include <fcntl.h>
#include <limits.h>
#include <poll.h>
#include <signal.h>
#include <spawn.h>
#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstdio>
#include <cstdlib>

#include "cpilot/log.hpp"
#include "cpilot/spawnproto.hpp"

extern char** environ;

namespace cpilot {

using namespace spawnproto;

namespace {

int defaultHelperCount() {
  if (const char* v = getenv("CPILOT_SPAWN_HELPERS")) {
    int n = atoi(v);
    if (n >= 1 && n <= 64) return n;
  }
  long ncpu = sysconf(_SC_NPROCESSORS_ONLN);
  if (ncpu < 1) ncpu = 1;
  int n = (int)(ncpu / 16);
  if (n < 2) n = 2;
  if (n > 8) n = 8;
  return n;
}

std::string findHelperBinary() {
  if (const char* v = getenv("CPILOT_SPAWN_HELPER")) return v;
  char buf[PATH_MAX];
  ssize_t n = readlink("/proc/self/exe", buf, sizeof(buf) - 1);
  if (n > 0) {
    buf[n] = 0;
    std::string dir(buf);
    size_t slash = dir.find_last_of('/');
    if (slash != std::string::npos) {
      std::string candidate = dir.substr(0, slash) + "/cpilot-spawn-helper";
      if (access(candidate.c_str(), X_OK) == 0) return candidate;
    }
  }
  return "cpilot-spawn-helper";  // PATH lookup as last resort
}

bool spawnDebug() {
  static bool on = cpilotDebugEnv("CPILOT_SPAWN_DEBUG");
  return on;
}

}  // namespace

Spawner& Spawner::global() {
  // intentionally leaked: the reader thread and helper processes live
  // for the process lifetime
  static Spawner* s = new Spawner();
  return *s;
}

Spawner::Spawner() {
  helperPath_ = findHelperBinary();
  if (pipe2(wakeFds_, O_NONBLOCK | O_CLOEXEC) != 0)
    LOG_ERROR("spawner: pipe2 failed: %s", strerror(errno));
  int n = defaultHelperCount();
  for (int i = 0; i < n; i++) helpers_.push_back(std::make_unique<Helper>());
  for (auto& h : helpers_) {
    if (!launchHelper(h.get())) {
      logging::logf(logging::Level::Fatal,
                    "spawner: cannot launch spawn helper '%s': %s",
                    helperPath_.c_str(), strerror(errno));
    }
  }
  reader_ = std::thread([this] { readerMain(); });
  reader_.detach();
}

bool Spawner::launchHelper(Helper* h) {
  int socks[2];
  if (socketpair(AF_UNIX, SOCK_SEQPACKET | SOCK_CLOEXEC, 0, socks) != 0)
    return false;
  // generous kernel buffers: requests carry the env (~3 KB); the
  // overflow queue handles the rest
  int bufsz = 1 << 20;
  setsockopt(socks[0], SOL_SOCKET, SO_SNDBUF, &bufsz, sizeof(bufsz));
  setsockopt(socks[1], SOL_SOCKET, SO_SNDBUF, &bufsz, sizeof(bufsz));
  fcntl(socks[0], F_SETFL, O_NONBLOCK);

  posix_spawn_file_actions_t actions;
  posix_spawn_file_actions_init(&actions);
  // dup2 clears CLOEXEC on the helper's copy
  posix_spawn_file_actions_adddup2(&actions, socks[1], kHelperFd);
  posix_spawnattr_t attr;
  posix_spawnattr_init(&attr);
  sigset_t empty;
  sigemptyset(&empty);
  posix_spawnattr_setsigmask(&attr, &empty);
  posix_spawnattr_setflags(&attr, POSIX_SPAWN_SETSIGMASK);

  char* argv[] = {const_cast<char*>("cpilot-spawn-helper"), nullptr};
  pid_t pid = -1;
  int rc = posix_spawn(&pid, helperPath_.c_str(), &actions, &attr, argv,
                       environ);
  posix_spawn_file_actions_destroy(&actions);
  posix_spawnattr_destroy(&attr);
  close(socks[1]);
  if (rc != 0) {
    close(socks[0]);
    errno = rc;
    return false;
  }
  h->sock = socks[0];
  h->pid = pid;
  h->dead = false;
  return true;
}

size_t Spawner::backlog() {
  std::lock_guard<std::mutex> l(mu_);
  return pending_.size();
}

// Drain a helper's overflow: pop a batch under mu_, send outside it
// (only the reader calls this, so within-overflow order is preserved;
// a concurrent direct send from the loop may interleave, which is fine
// — spawn requests of distinct commands are order-independent).
void Spawner::flushOverflow(size_t idx) {
  Helper& h = *helpers_[idx];
  while (true) {
    std::deque<std::pair<std::vector<char>, int>> batch;
    {
      std::lock_guard<std::mutex> l(mu_);
      if (h.dead || h.overflow.empty()) return;
      for (int i = 0; i < 16 && !h.overflow.empty(); i++) {
        batch.push_back(std::move(h.overflow.front()));
        h.overflow.pop_front();
      }
    }
    size_t sent = 0;
    {
      std::lock_guard<std::mutex> sl(h.sendMu);
      for (auto& [buf, fd] : batch) {
        if (!sendRequest(h, buf, fd)) break;
        if (fd >= 0) close(fd);
        sent++;
      }
    }
    if (sent < batch.size()) {
      // kernel buffer full again: put the unsent tail back at the front
      std::lock_guard<std::mutex> l(mu_);
      for (size_t i = batch.size(); i > sent; i--)
        h.overflow.push_front(std::move(batch[i - 1]));
      return;
    }
  }
}

bool Spawner::overloaded() {
  // ~16 outstanding per helper ≈ a few ms of queue at measured spawn
  // cost; beyond that a new check would only inflate round trips
  return backlog() > helpers_.size() * 16;
}

void Spawner::wakeReader() {
  char b = 1;
  ssize_t unused = write(wakeFds_[1], &b, 1);
  (void)unused;
}

void Spawner::spawn(Loop& loop, const std::string& execPath,
                    const std::vector<std::string>& args,
                    std::shared_ptr<const std::vector<std::string>> baseEnv,
                    const std::vector<std::string>& extraEnv, int stdioFd,
                    SpawnCb cb) {
  // serialize: header + argv strings + env strings, NUL-terminated
  RequestHeader hdr;
  memset(&hdr, 0, sizeof(hdr));
  hdr.type = kSpawnRequest;
  hdr.argc = 1 + (uint32_t)args.size();
  hdr.envc = (uint32_t)((baseEnv ? baseEnv->size() : 0) + extraEnv.size());
  hdr.wantStdio = stdioFd >= 0 ? 1 : 0;

  size_t size = sizeof(hdr) + execPath.size() + 1;
  for (auto& a : args) size += a.size() + 1;
  if (baseEnv)
    for (auto& e : *baseEnv) size += e.size() + 1;
  for (auto& e : extraEnv) size += e.size() + 1;
  if (size > kMaxRequestBytes) {
    if (stdioFd >= 0) close(stdioFd);
    loop.post([cb] { cb(-1, E2BIG); });
    return;
  }

  std::vector<char> buf;
  buf.reserve(size);
  auto append = [&buf](const std::string& s) {
    buf.insert(buf.end(), s.begin(), s.end());
    buf.push_back('\0');
  };
  buf.resize(sizeof(hdr));
  append(execPath);
  for (auto& a : args) append(a);
  if (baseEnv)
    for (auto& e : *baseEnv) append(e);
  for (auto& e : extraEnv) append(e);

  size_t idx = 0;
  uint32_t reqId = 0;
  bool queued = false;
  {
    std::lock_guard<std::mutex> l(mu_);
    reqId = nextReqId_++;
    hdr.reqId = reqId;
    memcpy(buf.data(), &hdr, sizeof(hdr));

    // round-robin over live helpers
    bool found = false;
    for (size_t i = 0; i < helpers_.size(); i++) {
      idx = (nextHelper_ + i) % helpers_.size();
      if (!helpers_[idx]->dead) {
        found = true;
        break;
      }
    }
    nextHelper_ = (idx + 1) % helpers_.size();
    if (!found) {
      if (stdioFd >= 0) close(stdioFd);
      loop.post([cb] { cb(-1, ECHILD); });
      return;
    }
    pending_[reqId] = Pending{&loop, loop.id(), std::move(cb), idx};
    if (!helpers_[idx]->overflow.empty()) {
      // backlog exists: append behind it (rough fairness)
      helpers_[idx]->overflow.emplace_back(std::move(buf), stdioFd);
      queued = true;
    }
  }
  if (spawnDebug())
    fprintf(stderr, "[spawner] req %u -> helper %zu\n", reqId, idx);
  if (queued) {
    wakeReader();
    return;
  }
  // send outside mu_: only this helper's sendMu serializes the syscall,
  // so the reader's overflow flushes never block spawn admission
  Helper& h = *helpers_[idx];
  bool sent;
  {
    std::lock_guard<std::mutex> sl(h.sendMu);
    sent = sendRequest(h, buf, stdioFd);
  }
  if (!sent) {
    // the fd stays open until actually sent
    std::lock_guard<std::mutex> l(mu_);
    h.overflow.emplace_back(std::move(buf), stdioFd);
    wakeReader();
    return;
  }
  if (stdioFd >= 0) close(stdioFd);
}

// Send one request datagram (+fd). Returns false on EAGAIN (caller
// queues to overflow); helper death is discovered by the reader.
bool Spawner::sendRequest(Helper& h, const std::vector<char>& buf, int fd) {
  struct iovec iov{const_cast<char*>(buf.data()), buf.size()};
  struct msghdr msg;
  memset(&msg, 0, sizeof(msg));
  msg.msg_iov = &iov;
  msg.msg_iovlen = 1;
  char cmsgBuf[CMSG_SPACE(sizeof(int))];
  if (fd >= 0) {
    memset(cmsgBuf, 0, sizeof(cmsgBuf));
    msg.msg_control = cmsgBuf;
    msg.msg_controllen = sizeof(cmsgBuf);
    struct cmsghdr* c = CMSG_FIRSTHDR(&msg);
    c->cmsg_level = SOL_SOCKET;
    c->cmsg_type = SCM_RIGHTS;
    c->cmsg_len = CMSG_LEN(sizeof(int));
    memcpy(CMSG_DATA(c), &fd, sizeof(int));
  }
  while (true) {
    ssize_t n = sendmsg(h.sock, &msg, MSG_NOSIGNAL);
    if (spawnDebug())
      fprintf(stderr, "[spawner] sendmsg fd=%d n=%zd errno=%d\n", h.sock, n,
              n < 0 ? errno : 0);
    if (n >= 0) return true;
    if (errno == EINTR) continue;
    return false;  // EAGAIN or helper gone (reader handles death)
  }
}

void Spawner::readerMain() {
  resetThreadScheduling();  // do not inherit the reactor's RT priority
  std::vector<char> buf(64);
  while (true) {
    std::vector<struct pollfd> fds;
    {
      std::lock_guard<std::mutex> l(mu_);
      for (auto& h : helpers_) {
        short ev = h->dead ? 0 : POLLIN;
        if (!h->dead && !h->overflow.empty()) ev |= POLLOUT;
        fds.push_back({h->sock, ev, 0});
      }
    }
    fds.push_back({wakeFds_[0], POLLIN, 0});
    int rc = poll(fds.data(), (nfds_t)fds.size(), 1000);
    if (spawnDebug() && rc != 0) {
      fprintf(stderr, "[spawner] poll rc=%d", rc);
      for (size_t i = 0; i < fds.size(); i++)
        fprintf(stderr, " fd%d=%x", fds[i].fd, fds[i].revents);
      fprintf(stderr, "\n");
    }
    if (rc < 0) {
      if (errno == EINTR) continue;
      return;
    }
    if (fds.back().revents & POLLIN) {
      char drain[256];
      while (read(wakeFds_[0], drain, sizeof(drain)) > 0) {
      }
    }
    for (size_t i = 0; i + 1 < fds.size(); i++) {
      if (fds[i].revents & POLLOUT) flushOverflow(i);
      if (fds[i].revents & (POLLIN | POLLHUP | POLLERR)) {
        while (true) {
          ssize_t n = recv(fds[i].fd, buf.data(), buf.size(), MSG_DONTWAIT);
          if (n > 0) {
            if (spawnDebug())
              fprintf(stderr, "[spawner] msg from helper %zu len=%zd\n", i, n);
            handleMessage(i, buf.data(), (size_t)n);
            continue;
          }
          if (n < 0 && (errno == EAGAIN || errno == EINTR)) break;
          if (spawnDebug())
            fprintf(stderr, "[spawner] helper %zu died n=%zd errno=%d\n",
                    i, n, errno);
          helperDied(i);  // EOF or error
          break;
        }
      }
    }
  }
}

void Spawner::handleMessage(size_t idx, const char* data, size_t len) {
  uint32_t type = 0;
  if (len >= sizeof(uint32_t)) memcpy(&type, data, sizeof(type));
  if (spawnDebug() && type != kSpawnReply && type != kChildExit)
    fprintf(stderr, "[spawner] DROP unknown msg type=%u len=%zu\n", type, len);
  if (type == kSpawnReply && len >= sizeof(Reply)) {
    Reply r;
    memcpy(&r, data, sizeof(r));
    Pending p;
    {
      std::lock_guard<std::mutex> l(mu_);
      auto it = pending_.find(r.reqId);
      if (it == pending_.end()) {
        if (spawnDebug())
          fprintf(stderr, "[spawner] DROP reply for unknown req %u\n",
                  r.reqId);
        return;
      }
      p = std::move(it->second);
      pending_.erase(it);
    }
    if (spawnDebug())
      fprintf(stderr, "[spawner] reply req %u pid %d\n", r.reqId, r.pid);
    if (r.pid > 0) pidLoops_[r.pid] = {p.loop, p.loopId};
    SpawnCb cb = std::move(p.cb);
    pid_t pid = r.pid;
    int err = r.err;
    Loop::postIfLive(p.loop, p.loopId,
                     timedItem("spawncb", [cb, pid, err] { cb(pid, err); }));
  } else if (type == kChildExit && len >= sizeof(ExitNotice)) {
    ExitNotice note;
    memcpy(&note, data, sizeof(note));
    auto it = pidLoops_.find(note.pid);
    if (it == pidLoops_.end()) {
      if (spawnDebug())
        fprintf(stderr, "[spawner] DROP exit notice for unknown pid %d\n",
                note.pid);
      return;
    }
    if (spawnDebug())
      fprintf(stderr, "[spawner] exit pid %d\n", note.pid);
    auto [loop, loopId] = it->second;
    pidLoops_.erase(it);
    pid_t pid = note.pid;
    int status = note.status;
    Loop::postIfLive(loop, loopId, timedItem("exit", [loop, pid, status] {
                       loop->notifyChildExit(pid, status);
                     }));
  }
}

void Spawner::helperDied(size_t idx) {
  std::vector<Pending> lost;
  {
    std::lock_guard<std::mutex> l(mu_);
    Helper& h = *helpers_[idx];
    if (h.dead) return;
    h.dead = true;
    LOG_ERROR("spawner: helper %d (pid %d) died; respawning",
              (int)idx, (int)h.pid);
    for (auto it = pending_.begin(); it != pending_.end();) {
      if (it->second.helperIdx == idx) {
        lost.push_back(std::move(it->second));
        it = pending_.erase(it);
      } else {
        ++it;
      }
    }
    for (auto& [obuf, ofd] : h.overflow)
      if (ofd >= 0) close(ofd);
    h.overflow.clear();
    close(h.sock);
    if (!launchHelper(&h)) {
      LOG_ERROR("spawner: helper respawn failed: %s", strerror(errno));
      h.dead = true;
    }
  }
  for (auto& p : lost) {
    SpawnCb cb = std::move(p.cb);
    Loop::postIfLive(p.loop, p.loopId, [cb] { cb(-1, ECHILD); });
  }
}

}  // namespace cpilot
