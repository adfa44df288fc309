#include "cpilot/metrics.hpp"

#include <cstdio>

namespace cpilot {
namespace prom {

namespace {
std::string escapeLabel(const std::string& s) {
  std::string out;
  for (char c : s) {
    if (c == '\\') out += "\\\\";
    else if (c == '"') out += "\\\"";
    else if (c == '\n') out += "\\n";
    else out += c;
  }
  return out;
}

std::string fmtDouble(double v) {
  if (v == (long long)v && v < 1e15 && v > -1e15) {
    char buf[32];
    snprintf(buf, sizeof(buf), "%lld", (long long)v);
    return buf;
  }
  char buf[40];
  snprintf(buf, sizeof(buf), "%.10g", v);
  return buf;
}

std::string labelString(const std::vector<std::string>& names,
                        const std::vector<std::string>& values,
                        const std::string& extraName = "",
                        const std::string& extraValue = "") {
  std::string out;
  bool any = false;
  for (size_t i = 0; i < names.size() && i < values.size(); i++) {
    if (any) out += ",";
    out += names[i] + "=\"" + escapeLabel(values[i]) + "\"";
    any = true;
  }
  if (!extraName.empty()) {
    if (any) out += ",";
    out += extraName + "=\"" + extraValue + "\"";
    any = true;
  }
  if (!any) return "";
  return "{" + out + "}";
}
}  // namespace

void Family::expose(std::string& out) const {
  std::lock_guard<std::mutex> l(mu_);
  const char* typeName;
  switch (type_) {
    case MetricType::Counter: typeName = "counter"; break;
    case MetricType::Gauge: typeName = "gauge"; break;
    case MetricType::Histogram: typeName = "histogram"; break;
    case MetricType::Summary: typeName = "summary"; break;
  }
  out += "# HELP " + name_ + " " + help_ + "\n";
  out += "# TYPE " + name_ + " " + std::string(typeName) + "\n";
  // a family with no children still exposes one zero-value child for
  // unlabeled collectors (prometheus exposes e.g. user metrics at 0)
  if (children_.empty() && labelNames_.empty() &&
      (type_ == MetricType::Counter || type_ == MetricType::Gauge)) {
    out += name_ + " 0\n";
    return;
  }
  for (auto& kv : children_) {
    const auto& labels = kv.first;
    const Child& c = kv.second;
    switch (type_) {
      case MetricType::Counter:
      case MetricType::Gauge:
        out += name_ + labelString(labelNames_, labels) + " " +
               fmtDouble(c.value) + "\n";
        break;
      case MetricType::Histogram: {
        uint64_t cumulative = 0;
        for (size_t i = 0; i < c.hist.bounds.size(); i++) {
          cumulative = c.hist.counts[i];
          out += name_ + "_bucket" +
                 labelString(labelNames_, labels, "le",
                             fmtDouble(c.hist.bounds[i])) +
                 " " + std::to_string(cumulative) + "\n";
        }
        out += name_ + "_bucket" +
               labelString(labelNames_, labels, "le", "+Inf") + " " +
               std::to_string(c.hist.count) + "\n";
        out += name_ + "_sum" + labelString(labelNames_, labels) + " " +
               fmtDouble(c.hist.sum) + "\n";
        out += name_ + "_count" + labelString(labelNames_, labels) + " " +
               std::to_string(c.hist.count) + "\n";
        break;
      }
      case MetricType::Summary: {
        for (double q : {0.5, 0.9, 0.99}) {
          out += name_ +
                 labelString(labelNames_, labels, "quantile", fmtDouble(q)) +
                 " " + fmtDouble(c.summ.quantile(q)) + "\n";
        }
        out += name_ + "_sum" + labelString(labelNames_, labels) + " " +
               fmtDouble(c.summ.sum) + "\n";
        out += name_ + "_count" + labelString(labelNames_, labels) + " " +
               std::to_string(c.summ.count) + "\n";
        break;
      }
    }
  }
}

Registry& Registry::global() {
  static Registry r;
  return r;
}

std::shared_ptr<Family> Registry::registerFamily(
    const std::string& name, const std::string& help, MetricType type,
    std::vector<std::string> labelNames, bool keepExisting) {
  std::lock_guard<std::mutex> l(mu_);
  for (auto it = families_.begin(); it != families_.end(); ++it) {
    if ((*it)->name() == name) {
      if (keepExisting && (*it)->type() == type) return *it;
      families_.erase(it);
      break;
    }
  }
  auto fam = std::make_shared<Family>(name, help, type, std::move(labelNames));
  families_.push_back(fam);
  return fam;
}

std::string Registry::expose() const {
  std::lock_guard<std::mutex> l(mu_);
  std::string out;
  for (auto& f : families_) f->expose(out);
  return out;
}

}  // namespace prom
}  // namespace cpilot
