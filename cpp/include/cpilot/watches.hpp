// Watches: poll Consul for healthy instances of a service every `interval`
// seconds; on membership/address change publish StatusChanged plus
// StatusHealthy/StatusUnhealthy from source "watch.<name>".
// Parity: /root/reference/watches/{watches,config}.go.
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "cpilot/discovery.hpp"
#include "cpilot/events.hpp"
#include "cpilot/json.hpp"

namespace cpilot {

struct WatchConfig {
  std::string name;         // "watch.<service>"
  std::string serviceName;  // original name
  int poll = 0;             // seconds
  std::string tag;
  std::string dc;
  // extension: blocking=true uses Consul blocking queries (long-poll on
  // the service's index) — changes propagate in milliseconds instead of
  // the poll interval, and idle traffic drops to one request per wait
  // window. `interval` becomes the error-backoff period.
  bool blocking = false;
};

bool newWatchConfigs(const Json& rawWatches,
                     std::vector<std::shared_ptr<WatchConfig>>* out,
                     std::string* err);

class Watch : public std::enable_shared_from_this<Watch> {
 public:
  explicit Watch(const std::shared_ptr<WatchConfig>& cfg)
      : name_(cfg->name),
        serviceName_(cfg->serviceName),
        tag_(cfg->tag),
        dc_(cfg->dc),
        poll_(cfg->poll),
        blocking_(cfg->blocking) {}

  const std::string& name() const { return name_; }
  const std::string& serviceName() const { return serviceName_; }

  void run(Loop& loop, std::shared_ptr<Bus> bus, ConsulBackend* consul);
  void stop(Loop& loop);

 private:
  void tick();
  void issueBlocking();
  void onResult(bool ok, std::vector<ServiceEntry> entries);

  std::string name_, serviceName_, tag_, dc_;
  int poll_;
  bool blocking_ = false;
  uint64_t lastIndex_ = 0;
  Loop* loop_ = nullptr;
  std::shared_ptr<Bus> bus_;
  ConsulBackend* consul_ = nullptr;
  uint64_t timer_ = 0;
  bool inFlight_ = false;
  bool stopped_ = false;
};

}  // namespace cpilot
