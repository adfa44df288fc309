// A small Prometheus client: counter / gauge / histogram / summary
// collectors, label vectors, and text exposition for the /metrics endpoint.
// Fills the role prometheus/client_golang plays in the reference
// (events/bus.go:60-68, control/control.go:25-33, discovery/consul.go:14-22,
// telemetry/metrics_config.go:39-86).
#pragma once

#include <algorithm>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

namespace cpilot {
namespace prom {

enum class MetricType { Counter, Gauge, Histogram, Summary };

struct HistogramData {
  // Go client default buckets
  std::vector<double> bounds{.005, .01, .025, .05, .1, .25, .5, 1, 2.5, 5, 10};
  std::vector<uint64_t> counts;  // same size as bounds
  uint64_t count = 0;
  double sum = 0;
  HistogramData() : counts(bounds.size(), 0) {}
  void setBounds(std::vector<double> b) {
    bounds = std::move(b);
    counts.assign(bounds.size(), 0);
  }
  void observe(double v) {
    for (size_t i = 0; i < bounds.size(); i++)
      if (v <= bounds[i]) counts[i]++;
    count++;
    sum += v;
  }
};

struct SummaryData {
  // sliding window of samples; quantiles computed at scrape time.
  // (Go client default objectives are 0.5/0.9/0.99.)
  std::vector<double> window;
  size_t cap = 8192;
  uint64_t count = 0;
  double sum = 0;
  void observe(double v) {
    if (window.size() >= cap) window.erase(window.begin());
    window.push_back(v);
    count++;
    sum += v;
  }
  double quantile(double q) const {
    if (window.empty()) return 0;
    std::vector<double> s(window);
    std::sort(s.begin(), s.end());
    size_t idx = (size_t)(q * (s.size() - 1));
    return s[idx];
  }
};

struct Child {
  double value = 0;  // counter/gauge
  HistogramData hist;
  SummaryData summ;
};

// One metric family: name, help, type, and children keyed by label values.
class Family {
 public:
  Family(std::string name, std::string help, MetricType type,
         std::vector<std::string> labelNames = {})
      : name_(std::move(name)),
        help_(std::move(help)),
        type_(type),
        labelNames_(std::move(labelNames)) {}

  void inc(const std::vector<std::string>& labels = {}, double by = 1) {
    std::lock_guard<std::mutex> l(mu_);
    child(labels).value += by;
  }
  void set(const std::vector<std::string>& labels, double v) {
    std::lock_guard<std::mutex> l(mu_);
    child(labels).value = v;
  }
  void set(double v) { set({}, v); }
  void observe(const std::vector<std::string>& labels, double v) {
    std::lock_guard<std::mutex> l(mu_);
    Child& c = child(labels);
    if (type_ == MetricType::Histogram) c.hist.observe(v);
    else c.summ.observe(v);
  }
  void observe(double v) { observe({}, v); }
  // set histogram bucket bounds (before first observe)
  void setBuckets(std::vector<double> bounds) {
    std::lock_guard<std::mutex> l(mu_);
    bucketBounds_ = bounds;
    for (auto& kv : children_) kv.second.hist.setBounds(bounds);
  }

  void expose(std::string& out) const;
  const std::string& name() const { return name_; }
  MetricType type() const { return type_; }

 private:
  Child& child(const std::vector<std::string>& labels) {
    auto it = children_.find(labels);
    if (it != children_.end()) return it->second;
    Child& c = children_[labels];
    if (!bucketBounds_.empty()) c.hist.setBounds(bucketBounds_);
    return c;
  }
  std::vector<double> bucketBounds_;
  std::string name_, help_;
  MetricType type_;
  std::vector<std::string> labelNames_;
  mutable std::mutex mu_;
  std::map<std::vector<std::string>, Child> children_;
};

// Global registry (like prometheus.MustRegister / default registry)
class Registry {
 public:
  static Registry& global();

  // Registers a family and returns it. keepExisting=true returns the
  // already-registered family of that name (internal collectors persist
  // across config reloads, like the reference's package-level
  // prometheus.MustRegister in init()); keepExisting=false replaces it
  // (user metrics unregister-then-register on reload and reset,
  // metrics_config.go:83-85).
  std::shared_ptr<Family> registerFamily(const std::string& name,
                                         const std::string& help,
                                         MetricType type,
                                         std::vector<std::string> labelNames = {},
                                         bool keepExisting = true);
  std::string expose() const;

 private:
  mutable std::mutex mu_;
  std::vector<std::shared_ptr<Family>> families_;
};

}  // namespace prom
}  // namespace cpilot
