#include "cpilot/control.hpp"

#include <unistd.h>

#include <cstdio>
#include <thread>

#include "cpilot/command.hpp"
#include "cpilot/decode.hpp"
#include "cpilot/log.hpp"

namespace cpilot {

const char* kDefaultControlSocket = "/var/run/containerpilot.socket";

bool newControlConfig(const Json* raw, ControlConfig* out, std::string* err) {
  out->socketPath = kDefaultControlSocket;
  if (raw == nullptr || raw->isNull()) return true;
  if (!raw->isObject()) {
    *err = "control config parsing error: must be an object";
    return false;
  }
  if (!decode::checkKeys(*raw, {"socket"}, err)) {
    *err = "control config parsing error: " + *err;
    return false;
  }
  if (const Json* v = raw->find("socket")) decode::toString(*v, &out->socketPath);
  return true;
}

ControlServer::ControlServer(Loop& loop, std::string socketPath)
    : loop_(loop), socketPath_(std::move(socketPath)) {
  requestCounter_ = prom::Registry::global().registerFamily(
      "containerpilot_control_http_requests",
      "count of requests to control socket, partitioned by path and HTTP code",
      prom::MetricType::Counter, {"code", "path"});
}

ControlServer::~ControlServer() { stop(); }

bool ControlServer::start(std::shared_ptr<Bus> bus, std::string* err) {
  bus_ = std::move(bus);
  if (socketPath_.empty()) {
    *err = "control: validate failed with control server not loading due to "
           "missing config";
    return false;
  }
  // unlink stale socket (control/control.go:61-73)
  if (access(socketPath_.c_str(), F_OK) == 0) {
    LOG_DEBUG("control: unlinking previous socket at %s", socketPath_.c_str());
    if (unlink(socketPath_.c_str()) != 0) {
      *err = "could not remove stale socket";
      return false;
    }
  }
  server_ = std::make_unique<http::Server>(
      loop_, [this](const http::Request& req) { return handle(req); });
  std::string bindErr;
  if (tryListen(&bindErr)) return true;
  // retry from the reactor instead of sleeping on it: a briefly
  // contended socket must not freeze timers/dispatch for up to 10 s
  LOG_WARN("control: error listening to socket at %s: %s (retrying)",
           socketPath_.c_str(), bindErr.c_str());
  scheduleRetry(1, bindErr);
  return true;
}

bool ControlServer::tryListen(std::string* bindErr) {
  if (server_->listenUnix(socketPath_, bindErr)) {
    LOG_DEBUG("control: listening to %s", socketPath_.c_str());
    LOG_INFO("control: serving at %s", socketPath_.c_str());
    return true;
  }
  return false;
}

void ControlServer::scheduleRetry(int attempt, const std::string& lastErr) {
  if (attempt >= 10) {
    // parity with the reference's log.Fatal after exhausted retries
    logging::logf(logging::Level::Fatal,
                  "error listening to socket at %s: %s", socketPath_.c_str(),
                  lastErr.c_str());
    return;
  }
  retryTimer_ = loop_.addTimeout(std::chrono::seconds(1), [this, attempt] {
    retryTimer_ = 0;
    if (!server_) return;  // stopped while the retry was pending
    std::string bindErr;
    if (tryListen(&bindErr)) return;
    scheduleRetry(attempt + 1, bindErr);
  });
}

void ControlServer::stop() {
  if (retryTimer_) {
    loop_.cancelTimer(retryTimer_);
    retryTimer_ = 0;
  }
  if (server_) {
    server_->stop();
    server_.reset();
    unlink(socketPath_.c_str());
    LOG_DEBUG("control: completed graceful shutdown of control server");
  }
}

namespace {

http::Response plainResponse(int status) {
  http::Response resp;
  resp.status = status;
  if (status == 200) {
    resp.body = "\n";  // io.WriteString(w, "\n") on empty 200
  } else {
    resp.body = std::string(http::statusText(status)) + "\n";  // http.Error
  }
  return resp;
}

// render a JSON value the way Go fmt %v renders the unmarshaled
// interface{} (endpoints.go:124-127): float64 integral values print
// without decimals
std::string metricValueString(const Json& v) {
  if (v.isString()) return v.str();
  if (v.isBool()) return v.boolean() ? "true" : "false";
  if (v.isInt()) return std::to_string(v.asInt());
  if (v.isDouble()) {
    char buf[40];
    snprintf(buf, sizeof(buf), "%g", v.asDouble());
    return buf;
  }
  return v.dump();
}

}  // namespace

http::Response ControlServer::handle(const http::Request& req) {
  http::Response resp;
  const std::string& path = req.path;

  if (path == "/v3/ping") {
    // GetPing has no method check (endpoints.go:133-138)
    resp = plainResponse(200);
    requestCounter_->inc({"200", path});
    return resp;
  }

  auto finish = [&](int status) {
    requestCounter_->inc({std::to_string(status), path});
    return plainResponse(status);
  };

  if (path != "/v3/environ" && path != "/v3/reload" && path != "/v3/metric" &&
      path != "/v3/maintenance/enable" && path != "/v3/maintenance/disable") {
    http::Response notFound;
    notFound.status = 404;
    notFound.body = "404 page not found\n";
    requestCounter_->inc({"404", path});
    return notFound;
  }
  if (req.method != "POST") return finish(405);

  if (path == "/v3/environ") {
    // PutEnviron (endpoints.go:57-72)
    try {
      Json doc = parseJson5(req.body);
      if (!doc.isObject()) return finish(422);
      for (auto& kv : doc.object()) {
        if (!kv.second.isString()) return finish(422);
        setenv(kv.first.c_str(), kv.second.str().c_str(), 1);
      }
      commandEnvInvalidate();  // future spawns see the new environ
    } catch (const std::exception&) {
      return finish(422);
    }
    return finish(200);
  }
  if (path == "/v3/reload") {
    LOG_DEBUG("control: reloading app via control plane");
    bus_->setReloadFlag();
    bus_->shutdown();
    LOG_DEBUG("control: reloaded app via control plane");
    return finish(200);
  }
  if (path == "/v3/metric") {
    // PostMetric (endpoints.go:111-129)
    try {
      Json doc = parseJson5(req.body);
      if (!doc.isObject()) return finish(422);
      for (auto& kv : doc.object()) {
        std::string eventVal = kv.first + "|" + metricValueString(kv.second);
        bus_->publish(Event{EventCode::Metric, eventVal});
      }
    } catch (const std::exception&) {
      return finish(422);
    }
    return finish(200);
  }
  if (path == "/v3/maintenance/enable") {
    bus_->publish(GlobalEnterMaintenance);
    return finish(200);
  }
  if (path == "/v3/maintenance/disable") {
    bus_->publish(GlobalExitMaintenance);
    return finish(200);
  }
  return finish(404);
}

}  // namespace cpilot
