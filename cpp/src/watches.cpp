#include "cpilot/watches.hpp"

#include "cpilot/decode.hpp"
#include "cpilot/ips.hpp"
#include "cpilot/log.hpp"

namespace cpilot {

bool newWatchConfigs(const Json& rawWatches,
                     std::vector<std::shared_ptr<WatchConfig>>* out,
                     std::string* err) {
  out->clear();
  if (rawWatches.isNull()) return true;
  if (!rawWatches.isArray()) {
    *err = "Watch configuration error: watches must be an array";
    return false;
  }
  for (auto& raw : rawWatches.array()) {
    if (!raw.isObject()) {
      *err = "Watch configuration error: watch must be an object";
      return false;
    }
    if (!decode::checkKeys(raw, {"name", "interval", "tag", "dc", "blocking"},
                           err)) {
      *err = "Watch configuration error: " + *err;
      return false;
    }
    auto cfg = std::make_shared<WatchConfig>();
    if (const Json* v = raw.find("name")) decode::toString(*v, &cfg->name);
    if (const Json* v = raw.find("interval")) decode::toInt(*v, &cfg->poll);
    if (const Json* v = raw.find("tag")) decode::toString(*v, &cfg->tag);
    if (const Json* v = raw.find("dc")) decode::toString(*v, &cfg->dc);
    if (const Json* v = raw.find("blocking")) {
      if (!decode::toBool(*v, &cfg->blocking)) {
        *err = "watch[" + cfg->name + "].blocking must be a bool";
        return false;
      }
    }

    if (!validateServiceName(cfg->name, err)) return false;
    cfg->serviceName = cfg->name;
    cfg->name = "watch." + cfg->name;  // watches/config.go:44-45
    if (cfg->poll < 1) {
      *err = "watch[" + cfg->serviceName + "].interval must be > 0";
      return false;
    }
    out->push_back(cfg);
  }
  return true;
}

void Watch::run(Loop& loop, std::shared_ptr<Bus> bus, ConsulBackend* consul) {
  loop_ = &loop;
  bus_ = std::move(bus);
  consul_ = consul;
  auto self = shared_from_this();
  if (blocking_) {
    issueBlocking();
    return;
  }
  timer_ = loop.addInterval(std::chrono::seconds(poll_),
                            [this, self] { tick(); });
}

void Watch::stop(Loop& loop) {
  stopped_ = true;
  if (timer_) {
    loop.cancelTimer(timer_);
    timer_ = 0;
  }
}

void Watch::onResult(bool ok, std::vector<ServiceEntry> entries) {
  if (!ok) {
    LOG_WARN("failed to query %s", serviceName_.c_str());
    return;
  }
  consul_->watchGauge()->set({serviceName_}, (double)entries.size());
  bool isHealthy = !entries.empty();
  bool didChange = consul_->compareAndSwap(serviceName_, entries);
  if (didChange) {
    bus_->publish(Event{EventCode::StatusChanged, name_});
    if (isHealthy)
      bus_->publish(Event{EventCode::StatusHealthy, name_});
    else
      bus_->publish(Event{EventCode::StatusUnhealthy, name_});
  }
}

void Watch::tick() {
  if (stopped_ || !consul_) return;
  if (inFlight_) return;  // one query at a time; skip this tick
  inFlight_ = true;
  auto self = shared_from_this();
  consul_->healthService(
      serviceName_, tag_, dc_,
      [this, self](bool ok, std::vector<ServiceEntry> entries) {
        inFlight_ = false;
        if (stopped_) return;
        onResult(ok, std::move(entries));
      });
}

void Watch::issueBlocking() {
  if (stopped_ || !consul_) return;
  auto self = shared_from_this();
  consul_->healthServiceBlocking(
      serviceName_, tag_, dc_, lastIndex_, 10,
      [this, self](bool ok, std::vector<ServiceEntry> entries,
                   uint64_t index) {
        if (stopped_) return;
        onResult(ok, std::move(entries));
        if (ok) {
          // consul index contract: reset when it goes backwards
          lastIndex_ = (index > 0 && index >= lastIndex_) ? index : 0;
          // immediate re-issue with a tiny floor so a hot agent can't
          // spin us
          timer_ = loop_->addTimeout(std::chrono::milliseconds(50),
                                     [this, self] { issueBlocking(); });
        } else {
          // error backoff: fall back to the configured interval
          timer_ = loop_->addTimeout(std::chrono::seconds(poll_),
                                     [this, self] { issueBlocking(); });
        }
      });
}

}  // namespace cpilot
