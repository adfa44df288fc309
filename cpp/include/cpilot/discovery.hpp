// Consul service-discovery backend: registration, TTL heartbeats,
// deregistration, and polled health queries with compare-and-swap change
// detection. Blocking HTTP runs on a small worker pool; results are posted
// back onto the reactor so all state stays single-threaded.
// Parity: /root/reference/discovery/{discovery,consul,config,service}.go.
#pragma once

#include <condition_variable>
#include <deque>
#include <functional>
#include <map>
#include <memory>
#include <set>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "cpilot/http.hpp"
#include "cpilot/json.hpp"
#include "cpilot/loop.hpp"
#include "cpilot/metrics.hpp"

namespace cpilot {

struct ServiceEntry {
  std::string id;
  std::string address;
  int port = 0;
};

class ConsulBackend {
 public:
  // Build from the top-level "consul" config value: a URI string or a map
  // with {address, scheme, token, tls:{...}}. CONSUL_HTTP_TOKEN overrides
  // the token (discovery/consul.go:48-50). Returns nullptr + err on
  // "no discovery backend defined".
  static std::unique_ptr<ConsulBackend> create(const Json* raw,
                                               std::string* err);

  ~ConsulBackend();

  void start(Loop& loop);
  void stop();

  using DoneCb = std::function<void(bool ok, const std::string& err)>;
  using HealthCb =
      std::function<void(bool ok, std::vector<ServiceEntry> entries)>;
  // blocking-query variant also yields the new Consul index
  using HealthBlockingCb = std::function<void(
      bool ok, std::vector<ServiceEntry> entries, uint64_t index)>;

  // byte-compatible payload with the reference's registration
  // (discovery/service.go:93-110)
  void serviceRegister(const std::string& id, const std::string& name,
                       const std::vector<std::string>& tags, int port,
                       const std::string& address, bool enableTagOverride,
                       int ttlSeconds, const std::string& status,
                       const std::string& deregisterAfter, DoneCb cb);
  void updateTTL(const std::string& checkID, const std::string& output,
                 const std::string& status, DoneCb cb);
  void serviceDeregister(const std::string& id, DoneCb cb);
  // GET /v1/health/service/<name>?passing=1 (+tag,+dc)
  void healthService(const std::string& name, const std::string& tag,
                     const std::string& dc, HealthCb cb);

  // Consul blocking query: long-polls with ?index=<lastIndex>&wait=<N>s;
  // returns when membership changes or the wait elapses. The request is
  // cancellable so generation teardown never stalls behind the poll.
  void healthServiceBlocking(const std::string& name, const std::string& tag,
                             const std::string& dc, uint64_t lastIndex,
                             int waitSeconds, HealthBlockingCb cb);

  // change detection against the cached set (loop thread only)
  // (discovery/consul.go:102-125)
  bool compareAndSwap(const std::string& service,
                      std::vector<ServiceEntry> entries);

  const std::string& address() const { return address_; }
  const std::string& scheme() const { return scheme_; }

  std::shared_ptr<prom::Family> watchGauge() { return watchGauge_; }

 private:
  ConsulBackend() = default;
  void workerMain();
  // key != "": coalesce — if a task with the same key is already
  // queued, the new one is rejected (idempotent heartbeats/registrations
  // must not pile up behind a slow agent). The queue is also hard-capped.
  // Returns false when rejected; callers must complete their callback
  // path themselves (in-flight guards would otherwise stick).
  bool enqueue(const std::string& key, std::function<void()> task);

  std::string address_;  // host:port
  std::string scheme_ = "http";
  std::string token_;
  http::TlsOptions tls_;

  Loop* loop_ = nullptr;
  std::vector<std::thread> workers_;
  std::deque<std::pair<std::string, std::function<void()>>> tasks_;
  std::set<std::string> queuedKeys_;
  uint64_t dropped_ = 0;
  std::mutex mu_;
  std::condition_variable cv_;
  bool stopping_ = false;

  std::map<std::string, std::vector<ServiceEntry>> watched_;
  std::shared_ptr<prom::Family> watchGauge_;

  // blocking queries run on their own short-lived threads (a parked
  // long-poll must not occupy the shared worker pool, or N blocking
  // watches would starve TTL heartbeats). Shared with those threads so
  // stop() can cancel without use-after-free on backend teardown.
  struct BlockingReg {
    std::mutex mu;
    std::set<std::shared_ptr<http::CancelToken>> tokens;
    bool stopping = false;
  };
  std::shared_ptr<BlockingReg> blockingReg_ =
      std::make_shared<BlockingReg>();
};

// Per-job service registration state (discovery/service.go:12-110).
// Always held via shared_ptr: async registration callbacks posted back
// onto the reactor capture a weak_ptr so a callback that outlives its
// generation (reload while consul is unreachable keeps the HTTP request
// in flight past teardown) no-ops instead of touching freed memory.
struct ServiceDefinition
    : public std::enable_shared_from_this<ServiceDefinition> {
  std::string id;
  std::string name;
  int port = 0;
  int ttl = 0;
  std::vector<std::string> tags;
  std::string initialStatus;
  std::string ipAddress;
  bool enableTagOverride = false;
  std::string deregisterCriticalServiceAfter;
  ConsulBackend* consul = nullptr;

  bool wasRegistered = false;
  bool registerInFlight = false;

  void sendHeartbeat();
  void registerWithInitialStatus();
  void deregister();
  void markForMaintenance();

 private:
  void registerService(const std::string& status);
};

}  // namespace cpilot
