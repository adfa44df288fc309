// Telemetry: TCP HTTP server exposing Prometheus /metrics and JSON
// /status, user metric collectors fed by Metric events, and the synthetic
// always-healthy "containerpilot" job advertised in Consul (TTL 15s /
// heartbeat 5s).
// Parity: /root/reference/telemetry/*.go.
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "cpilot/events.hpp"
#include "cpilot/http.hpp"
#include "cpilot/jobs.hpp"
#include "cpilot/json.hpp"
#include "cpilot/metrics.hpp"
#include "cpilot/watches.hpp"

namespace cpilot {

struct MetricConfig {
  std::string ns, subsystem, name, help, type;
  std::string fullName;  // ns_subsystem_name joined with "_"
  prom::MetricType metricType = prom::MetricType::Counter;
  std::shared_ptr<prom::Family> collector;
};

bool newMetricConfigs(const Json& raw,
                      std::vector<std::shared_ptr<MetricConfig>>* out,
                      std::string* err);

// Per-metric event subscriber: parses "name|value" Metric events and
// records into its collector (telemetry/metrics.go:48-112).
class Metric : public Subscriber,
               public std::enable_shared_from_this<Metric> {
 public:
  explicit Metric(const std::shared_ptr<MetricConfig>& cfg) : cfg_(cfg) {}
  void run(std::shared_ptr<Bus> bus);
  void onEvent(const Event& event) override;
  Subscription subscription() const override {
    // Metric events carry "name|value" as their source, so interest is
    // by code; each collector filters by its own name in onEvent
    return {false, {}, {EventCode::Metric}};
  }
  const std::string& name() const { return cfg_->fullName; }

 private:
  void record(const std::string& value);
  std::shared_ptr<MetricConfig> cfg_;
  std::shared_ptr<Bus> bus_;
};

struct TelemetryConfig {
  int port = 9090;
  std::vector<std::string> interfaces;  // raw spec list
  std::vector<std::string> tags;
  std::string ipAddress;
  std::vector<std::shared_ptr<MetricConfig>> metricConfigs;
  std::shared_ptr<JobConfig> jobConfig;  // synthetic "containerpilot" job
};

// Returns true with *out == nullptr when raw is null (telemetry disabled).
bool newTelemetryConfig(const Json* raw, ConsulBackend* disc,
                        std::shared_ptr<TelemetryConfig>* out,
                        std::string* err);

class Telemetry {
 public:
  Telemetry(Loop& loop, const std::shared_ptr<TelemetryConfig>& cfg);

  void monitorJobs(const std::vector<std::shared_ptr<Job>>& jobs);
  void monitorWatches(const std::vector<std::shared_ptr<Watch>>& watches);

  // Bind, retrying 10x1s on a loop timer (never blocks the reactor);
  // exits fatally after the last failed attempt like the reference's
  // listenWithRetry + log.Fatal (telemetry/telemetry.go:77-91).
  bool start(std::string* err);
  void stop();

  std::vector<std::shared_ptr<Metric>>& metrics() { return metrics_; }

 private:
  http::Response handle(const http::Request& req);
  std::string statusJson();
  void scheduleRetry(int attempt, const std::string& lastErr);

  Loop& loop_;
  std::shared_ptr<TelemetryConfig> cfg_;
  std::unique_ptr<http::Server> server_;
  std::vector<std::shared_ptr<Metric>> metrics_;
  std::vector<std::shared_ptr<Job>> jobs_;
  std::vector<std::string> watchNames_;
  uint64_t retryTimer_ = 0;
};

}  // namespace cpilot
