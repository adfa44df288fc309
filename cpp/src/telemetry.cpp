#include "cpilot/telemetry.hpp"

#include <cstdlib>
#include <thread>

#include "cpilot/decode.hpp"
#include "cpilot/ips.hpp"
#include "cpilot/log.hpp"
#include "cpilot/version.hpp"

namespace cpilot {

// ---------------- metric configs ----------------

bool newMetricConfigs(const Json& raw,
                      std::vector<std::shared_ptr<MetricConfig>>* out,
                      std::string* err) {
  out->clear();
  if (raw.isNull()) return true;
  if (!raw.isArray()) {
    *err = "MetricConfig configuration error: metrics must be an array";
    return false;
  }
  for (auto& m : raw.array()) {
    if (!m.isObject()) {
      *err = "MetricConfig configuration error: metric must be an object";
      return false;
    }
    if (!decode::checkKeys(m, {"namespace", "subsystem", "name", "help", "type"},
                           err)) {
      *err = "MetricConfig configuration error: " + *err;
      return false;
    }
    auto cfg = std::make_shared<MetricConfig>();
    if (const Json* v = m.find("namespace")) decode::toString(*v, &cfg->ns);
    if (const Json* v = m.find("subsystem"))
      decode::toString(*v, &cfg->subsystem);
    if (const Json* v = m.find("name")) decode::toString(*v, &cfg->name);
    if (const Json* v = m.find("help")) decode::toString(*v, &cfg->help);
    if (const Json* v = m.find("type")) decode::toString(*v, &cfg->type);

    // fullName joins all three with "_" even when empty
    // (telemetry/metrics_config.go:42)
    cfg->fullName = cfg->ns + "_" + cfg->subsystem + "_" + cfg->name;

    // the registered collector name skips empty parts (prometheus
    // BuildFQName semantics)
    std::string fq;
    for (const std::string* part : {&cfg->ns, &cfg->subsystem, &cfg->name}) {
      if (part->empty()) continue;
      if (!fq.empty()) fq += "_";
      fq += *part;
    }

    if (cfg->type == "counter") cfg->metricType = prom::MetricType::Counter;
    else if (cfg->type == "gauge") cfg->metricType = prom::MetricType::Gauge;
    else if (cfg->type == "histogram")
      cfg->metricType = prom::MetricType::Histogram;
    else if (cfg->type == "summary")
      cfg->metricType = prom::MetricType::Summary;
    else {
      *err = "invalid metric type: " + cfg->type;
      return false;
    }
    // unregister-then-register so reloads survive
    // (telemetry/metrics_config.go:83-85)
    cfg->collector = prom::Registry::global().registerFamily(
        fq, cfg->help, cfg->metricType, {}, /*keepExisting=*/false);
    out->push_back(cfg);
  }
  return true;
}

// ---------------- metric runtime ----------------

void Metric::run(std::shared_ptr<Bus> bus) {
  bus_ = std::move(bus);
  bus_->subscribe(this);
}

void Metric::onEvent(const Event& event) {
  if (event.code == EventCode::Metric) {
    // "name|value" (telemetry/metrics.go:48-59)
    size_t pipe = event.source.find('|');
    if (pipe == std::string::npos) {
      LOG_ERROR("metric: invalid metric format: %s", event.source.c_str());
      return;
    }
    std::string key = event.source.substr(0, pipe);
    std::string val = event.source.substr(pipe + 1);
    if (key == cfg_->fullName) record(val);
    return;
  }
  if (event == GlobalShutdown || event == QuitByTest) {
    bus_->unsubscribe(this);
  }
}

void Metric::record(const std::string& value) {
  // TrimSpace + ParseFloat (telemetry/metrics.go:61-80)
  std::string v = value;
  size_t a = 0, b = v.size();
  while (a < b && isspace((unsigned char)v[a])) a++;
  while (b > a && isspace((unsigned char)v[b - 1])) b--;
  v = v.substr(a, b - a);
  char* end = nullptr;
  double val = strtod(v.c_str(), &end);
  if (v.empty() || !end || *end != '\0') {
    LOG_ERROR("metric produced non-numeric value: %s", value.c_str());
    return;
  }
  switch (cfg_->metricType) {
    case prom::MetricType::Counter: cfg_->collector->inc({}, val); break;
    case prom::MetricType::Gauge: cfg_->collector->set({}, val); break;
    case prom::MetricType::Histogram:
    case prom::MetricType::Summary:
      cfg_->collector->observe({}, val);
      break;
  }
}

// ---------------- telemetry config ----------------

bool newTelemetryConfig(const Json* raw, ConsulBackend* disc,
                        std::shared_ptr<TelemetryConfig>* out,
                        std::string* err) {
  *out = nullptr;
  if (raw == nullptr || raw->isNull()) return true;  // telemetry disabled
  if (!raw->isObject()) {
    *err = "telemetry configuration error: must be an object";
    return false;
  }
  if (!decode::checkKeys(*raw, {"port", "interfaces", "tags", "metrics"},
                         err)) {
    *err = "telemetry configuration error: " + *err;
    return false;
  }
  auto cfg = std::make_shared<TelemetryConfig>();
  if (const Json* v = raw->find("port")) {
    if (!decode::toInt(*v, &cfg->port)) {
      *err = "telemetry configuration error: port must be a number";
      return false;
    }
  }
  if (const Json* v = raw->find("interfaces")) {
    if (!decode::toStrings(*v, &cfg->interfaces)) {
      *err = "telemetry configuration error: bad interfaces";
      return false;
    }
  }
  if (const Json* v = raw->find("tags")) {
    if (!decode::toStrings(*v, &cfg->tags)) {
      *err = "telemetry configuration error: tags must be strings";
      return false;
    }
  }

  std::string ipErr;
  if (!getIP(cfg->interfaces, &cfg->ipAddress, &ipErr)) {
    *err = "telemetry validation error: " + ipErr;
    return false;
  }

  if (const Json* v = raw->find("metrics")) {
    if (!newMetricConfigs(*v, &cfg->metricConfigs, err)) return false;
  }

  // synthetic "containerpilot" job: TTL 15 / heartbeat 5, no check exec
  // (telemetry/telemetry_config.go:71-86)
  JsonObject jobRaw;
  jobRaw.emplace_back("name", Json("containerpilot"));
  JsonObject health;
  health.emplace_back("interval", Json((int64_t)5));
  health.emplace_back("ttl", Json((int64_t)15));
  jobRaw.emplace_back("health", Json(std::move(health)));
  if (!cfg->interfaces.empty()) {
    JsonArray ifaces;
    for (auto& i : cfg->interfaces) ifaces.push_back(Json(i));
    jobRaw.emplace_back("interfaces", Json(std::move(ifaces)));
  }
  jobRaw.emplace_back("port", Json((int64_t)cfg->port));
  {
    JsonArray tags;
    for (auto& t : cfg->tags) tags.push_back(Json(t));
    if (kVersion[0] != '\0') tags.push_back(Json(kVersion));
    if (!tags.empty()) jobRaw.emplace_back("tags", Json(std::move(tags)));
  }
  std::shared_ptr<JobConfig> jobCfg;
  std::string jobErr;
  if (!validateJobConfig(Json(std::move(jobRaw)), disc, &jobCfg, &jobErr)) {
    *err = "could not validate telemetry service: " + jobErr;
    return false;
  }
  cfg->jobConfig = jobCfg;
  *out = cfg;
  return true;
}

// ---------------- telemetry server ----------------

Telemetry::Telemetry(Loop& loop, const std::shared_ptr<TelemetryConfig>& cfg)
    : loop_(loop), cfg_(cfg) {
  for (auto& mc : cfg->metricConfigs)
    metrics_.push_back(std::make_shared<Metric>(mc));
}

void Telemetry::monitorJobs(const std::vector<std::shared_ptr<Job>>& jobs) {
  jobs_ = jobs;
}

void Telemetry::monitorWatches(
    const std::vector<std::shared_ptr<Watch>>& watches) {
  for (auto& w : watches) {
    std::string name = w->name();
    if (name.rfind("watch.", 0) == 0) name = name.substr(6);
    watchNames_.push_back(name);
  }
}

bool Telemetry::start(std::string* err) {
  (void)err;
  server_ = std::make_unique<http::Server>(
      loop_, [this](const http::Request& req) { return handle(req); });
  std::string bindErr;
  if (server_->listenTcp(cfg_->ipAddress, cfg_->port, &bindErr)) {
    LOG_INFO("telemetry: serving at %s:%d", cfg_->ipAddress.c_str(),
             cfg_->port);
    return true;
  }
  // retry from the reactor, never by sleeping on the loop thread
  LOG_WARN("telemetry: error listening at %s:%d: %s (retrying)",
           cfg_->ipAddress.c_str(), cfg_->port, bindErr.c_str());
  scheduleRetry(1, bindErr);
  return true;
}

void Telemetry::scheduleRetry(int attempt, const std::string& lastErr) {
  if (attempt >= 10) {
    logging::logf(logging::Level::Fatal,
                  "error listening to socket at %s:%d: %s",
                  cfg_->ipAddress.c_str(), cfg_->port, lastErr.c_str());
    return;
  }
  retryTimer_ = loop_.addTimeout(std::chrono::seconds(1), [this, attempt] {
    retryTimer_ = 0;
    if (!server_) return;  // stopped while the retry was pending
    std::string bindErr;
    if (server_->listenTcp(cfg_->ipAddress, cfg_->port, &bindErr)) {
      LOG_INFO("telemetry: serving at %s:%d", cfg_->ipAddress.c_str(),
               cfg_->port);
      return;
    }
    scheduleRetry(attempt + 1, bindErr);
  });
}

void Telemetry::stop() {
  if (retryTimer_) {
    loop_.cancelTimer(retryTimer_);
    retryTimer_ = 0;
  }
  if (server_) {
    server_->stop();
    server_.reset();
    LOG_DEBUG("telemetry: completed graceful shutdown of server");
  }
}

http::Response Telemetry::handle(const http::Request& req) {
  http::Response resp;
  if (req.path == "/metrics") {
    resp.contentType = "text/plain; version=0.0.4";
    resp.body = prom::Registry::global().expose();
    return resp;
  }
  if (req.path == "/status") {
    if (req.method != "GET") {
      resp.status = 405;
      resp.body = "Method Not Allowed\n";
      return resp;
    }
    resp.contentType = "application/json";
    resp.body = statusJson();
    return resp;
  }
  resp.status = 404;
  resp.body = "404 page not found\n";
  return resp;
}

std::string Telemetry::statusJson() {
  // shape of telemetry/status.go:15-33 via Go json.Marshal: nil slices
  // marshal as null
  std::string out = "{\"Version\":\"" + std::string(kVersion) + "\"";
  std::string jobsPart, servicesPart;
  bool anyJob = false, anyService = false;
  for (auto& job : jobs_) {
    const char* status = jobStatusString(job->getStatus());
    auto svc = job->service();
    if (svc && svc->port != 0) {
      if (anyService) servicesPart += ",";
      anyService = true;
      servicesPart += "{\"Name\":\"" + job->name() + "\",\"Address\":\"" +
                      svc->ipAddress + "\",\"Port\":" +
                      std::to_string(svc->port) + ",\"Status\":\"" + status +
                      "\"}";
    } else {
      if (anyJob) jobsPart += ",";
      anyJob = true;
      jobsPart += "{\"Name\":\"" + job->name() + "\",\"Status\":\"" + status +
                  "\"}";
    }
  }
  out += ",\"Jobs\":";
  out += anyJob ? "[" + jobsPart + "]" : "null";
  out += ",\"Services\":";
  out += anyService ? "[" + servicesPart + "]" : "null";
  out += ",\"Watches\":";
  if (watchNames_.empty()) {
    out += "null";
  } else {
    out += "[";
    for (size_t i = 0; i < watchNames_.size(); i++) {
      if (i) out += ",";
      out += "\"" + watchNames_[i] + "\"";
    }
    out += "]";
  }
  out += "}\n";
  return out;
}

}  // namespace cpilot
