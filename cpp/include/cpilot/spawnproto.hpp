// Wire protocol between the daemon and its spawn-helper processes
// (SOCK_SEQPACKET socketpair, one per helper; message == datagram).
//
// Why helpers exist: spawning from the daemon itself does not scale.
// posix_spawn vforks share the parent's mm until exec, so concurrent
// spawns from a pool of threads serialize on the daemon's mmap lock and
// stall the reactor's own page faults (measured: p99 dispatch 50 ms and
// published-event throughput FALLING with job count on a 256-CPU EPYC);
// fork also copies the parent's fd table, so per-spawn cost grows with
// the number of supervised jobs (one log pipe each). A helper is a tiny
// exec'd process (~2 MB, 4 fds): its spawns are constant-cost and fully
// parallel across helpers, and it reaps its own children, so the
// daemon's reactor never forks and never handles child SIGCHLD storms.
#pragma once

#include <cstdint>

namespace cpilot {
namespace spawnproto {

// The helper inherits its socket on this fd.
constexpr int kHelperFd = 3;

enum MsgType : uint32_t {
  kSpawnRequest = 1,  // daemon -> helper, may carry a stdio fd (SCM_RIGHTS)
  kSpawnReply = 2,    // helper -> daemon
  kChildExit = 3,     // helper -> daemon
};

struct RequestHeader {
  uint32_t type;      // kSpawnRequest
  uint32_t reqId;
  uint32_t argc;      // count of argv strings that follow (>= 1: exec path)
  uint32_t envc;      // count of env strings that follow
  uint8_t wantStdio;  // 1: dup2 the passed fd onto child stdout+stderr
  uint8_t pad[3];
  // payload: argc + envc NUL-terminated strings, concatenated
};

struct Reply {
  uint32_t type;  // kSpawnReply
  uint32_t reqId;
  int32_t pid;    // < 0: spawn failed
  int32_t err;    // errno when pid < 0
};

struct ExitNotice {
  uint32_t type;  // kChildExit
  uint32_t pad;
  int32_t pid;
  int32_t status;  // waitpid status
};

// Requests larger than this are rejected client-side with E2BIG.
// 1 MiB fits the socketpair SNDBUF as a single SEQPACKET datagram and
// covers any realistic environment (the kernel's own execve limit is
// ~2 MiB total; environments that large fail there in the reference
// too).
constexpr size_t kMaxRequestBytes = 1024 * 1024;

}  // namespace spawnproto
}  // namespace cpilot
