// Control plane: HTTP server over a unix domain socket with the reference's
// five POST endpoints + GET ping, implementing the code's actual behavior
// (empty 200 body "\n", 422 on bad input, 405 on wrong method).
// Parity: /root/reference/control/{control,config,endpoints}.go.
#pragma once

#include <functional>
#include <memory>
#include <string>

#include "cpilot/events.hpp"
#include "cpilot/http.hpp"
#include "cpilot/json.hpp"
#include "cpilot/metrics.hpp"

namespace cpilot {

extern const char* kDefaultControlSocket;  // /var/run/containerpilot.socket

struct ControlConfig {
  std::string socketPath = "/var/run/containerpilot.socket";
};

bool newControlConfig(const Json* raw, ControlConfig* out, std::string* err);

class ControlServer {
 public:
  // reloadCb: invoked by POST /v3/reload after setting the bus reload flag
  ControlServer(Loop& loop, std::string socketPath);
  ~ControlServer();

  // Unlink stale socket (control/control.go:61-73), then bind. A failed
  // bind retries 10x1s on a LOOP TIMER (the reference retries inside a
  // goroutine, control/control.go:125-140) so event dispatch never
  // stalls behind a contended socket; after the last failed attempt the
  // daemon exits fatally like the reference's log.Fatal. Returns false
  // only for unretryable config errors (missing path, unremovable stale
  // socket).
  bool start(std::shared_ptr<Bus> bus, std::string* err);
  void stop();

  const std::string& socketPath() const { return socketPath_; }

 private:
  http::Response handle(const http::Request& req);
  bool tryListen(std::string* bindErr);
  void scheduleRetry(int attempt, const std::string& lastErr);

  Loop& loop_;
  std::string socketPath_;
  std::shared_ptr<Bus> bus_;
  std::unique_ptr<http::Server> server_;
  std::shared_ptr<prom::Family> requestCounter_;
  uint64_t retryTimer_ = 0;
};

}  // namespace cpilot
