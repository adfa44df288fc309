#include "cpilot/discovery.hpp"

#include <algorithm>
#include <cstdlib>
#include <thread>

#include "cpilot/decode.hpp"
#include "cpilot/http.hpp"
#include "cpilot/log.hpp"

namespace cpilot {

namespace {

// strip http:// or https:// prefix; bare addresses default to http
// (discovery/config.go:90-103)
void parseRawURI(const std::string& raw, std::string* address,
                 std::string* scheme) {
  *scheme = "http";
  *address = raw;
  if (raw.rfind("http://", 0) == 0) {
    *address = raw.substr(7);
  } else if (raw.rfind("https://", 0) == 0) {
    *address = raw.substr(8);
    *scheme = "https";
  }
}

std::string urlEncode(const std::string& s) {
  std::string out;
  for (unsigned char c : s) {
    if (isalnum(c) || c == '-' || c == '_' || c == '.' || c == '~' ||
        c == ':' || c == '@') {
      out += c;
    } else {
      char buf[4];
      snprintf(buf, sizeof(buf), "%%%02X", c);
      out += buf;
    }
  }
  return out;
}

}  // namespace

std::unique_ptr<ConsulBackend> ConsulBackend::create(const Json* raw,
                                                     std::string* err) {
  auto backend = std::unique_ptr<ConsulBackend>(new ConsulBackend());
  if (raw == nullptr || raw->isNull()) {
    *err = "no discovery backend defined";
    return nullptr;
  }
  // env overrides apply only where the config left the field unset,
  // mirroring consul api.DefaultConfig + the reference's override order
  // (discovery/config.go:29-61: env-derived defaults, then config values)
  bool explicitAddress = false;
  bool explicitScheme = false;
  if (raw->isString()) {
    parseRawURI(raw->str(), &backend->address_, &backend->scheme_);
    explicitAddress = !backend->address_.empty();
    explicitScheme = raw->str().rfind("http://", 0) == 0 ||
                     raw->str().rfind("https://", 0) == 0;
  } else if (raw->isObject()) {
    for (auto& kv : raw->object()) {
      if (kv.first == "address" && kv.second.isString()) {
        backend->address_ = kv.second.str();
        explicitAddress = !backend->address_.empty();
      } else if (kv.first == "scheme" && kv.second.isString()) {
        backend->scheme_ = kv.second.str();
        explicitScheme = true;
      }
      else if (kv.first == "token" && kv.second.isString())
        backend->token_ = kv.second.str();
      else if (kv.first == "tls") {
        // optional TLS settings (discovery/config.go:18-25)
        if (kv.second.isObject()) {
          for (auto& tkv : kv.second.object()) {
            if (tkv.first == "cafile" && tkv.second.isString())
              backend->tls_.caFile = tkv.second.str();
            else if (tkv.first == "capath" && tkv.second.isString())
              backend->tls_.caPath = tkv.second.str();
            else if (tkv.first == "clientcert" && tkv.second.isString())
              backend->tls_.certFile = tkv.second.str();
            else if (tkv.first == "clientkey" && tkv.second.isString())
              backend->tls_.keyFile = tkv.second.str();
            else if (tkv.first == "servername" && tkv.second.isString())
              backend->tls_.serverName = tkv.second.str();
            else if (tkv.first == "verify") {
              bool verify = false;
              if (decode::toBool(tkv.second, &verify))
                backend->tls_.insecureSkipVerify = !verify;
            } else {
              *err = "consul configuration error: invalid tls key: " +
                     tkv.first;
              return nullptr;
            }
          }
        }
      } else {
        *err = "consul configuration error: invalid key: " + kv.first;
        return nullptr;
      }
    }
  } else {
    *err = "no discovery backend defined";
    return nullptr;
  }
  if (const char* token = getenv("CONSUL_HTTP_TOKEN")) {
    if (token[0]) backend->token_ = token;
  }
  // CONSUL_HTTP_ADDR fills in the address whenever the config did not
  // pin one — whether consul was an empty string, an object without an
  // address, or an object with an empty address (api.DefaultConfig)
  if (const char* addr = getenv("CONSUL_HTTP_ADDR")) {
    std::string a(addr);
    if (!a.empty() && !explicitAddress) {
      std::string envScheme;
      std::string envAddr;
      parseRawURI(a, &envAddr, &envScheme);
      backend->address_ = envAddr;
      if (!explicitScheme) {
        backend->scheme_ = envScheme;
        // a scheme-qualified env address pins the scheme (the consul
        // client forces https when the address carries the prefix)
        explicitScheme = a.rfind("http://", 0) == 0 ||
                         a.rfind("https://", 0) == 0;
      }
    }
  }
  // CONSUL_HTTP_SSL switches the default scheme (api.DefaultConfig);
  // an explicit config scheme or a scheme-qualified URI/addr wins
  if (const char* ssl = getenv("CONSUL_HTTP_SSL")) {
    std::string s2(ssl);
    for (auto& ch : s2) ch = (char)tolower((unsigned char)ch);
    if (!explicitScheme && !s2.empty()) {
      if (s2 == "1" || s2 == "true") backend->scheme_ = "https";
      else if (s2 == "0" || s2 == "false") backend->scheme_ = "http";
    }
  }
  if (backend->address_.empty()) backend->address_ = "127.0.0.1:8500";
  // env overrides (discovery/config.go:29-51)
  if (const char* v = getenv("CONSUL_CACERT"))
    if (v[0]) backend->tls_.caFile = v;
  if (const char* v = getenv("CONSUL_CAPATH"))
    if (v[0]) backend->tls_.caPath = v;
  if (const char* v = getenv("CONSUL_CLIENT_CERT"))
    if (v[0]) backend->tls_.certFile = v;
  if (const char* v = getenv("CONSUL_CLIENT_KEY"))
    if (v[0]) backend->tls_.keyFile = v;
  if (const char* v = getenv("CONSUL_TLS_SERVER_NAME"))
    if (v[0]) backend->tls_.serverName = v;
  if (const char* v = getenv("CONSUL_HTTP_SSL_VERIFY")) {
    std::string s2(v);
    for (auto& ch : s2) ch = tolower(ch);
    if (s2 == "1" || s2 == "true") backend->tls_.insecureSkipVerify = false;
    else if (s2 == "0" || s2 == "false")
      backend->tls_.insecureSkipVerify = true;
  }
  backend->tls_.enabled = (backend->scheme_ == "https");
  backend->watchGauge_ = prom::Registry::global().registerFamily(
      "containerpilot_watch_instances",
      "gauge of instances found for each ContainerPilot watch, partitioned "
      "by service",
      prom::MetricType::Gauge, {"service"});
  return backend;
}

ConsulBackend::~ConsulBackend() { stop(); }

void ConsulBackend::start(Loop& loop) {
  loop_ = &loop;
  stopping_ = false;
  if (blockingReg_->stopping)
    blockingReg_ = std::make_shared<BlockingReg>();  // fresh after stop()
  for (int i = 0; i < 4; i++)
    workers_.emplace_back([this] { workerMain(); });
}

void ConsulBackend::stop() {
  {
    std::lock_guard<std::mutex> l(mu_);
    if (stopping_ && workers_.empty()) return;
    stopping_ = true;
  }
  {
    // interrupt in-flight blocking queries so teardown never waits out
    // a long-poll (they run on their own threads; see healthServiceBlocking)
    std::lock_guard<std::mutex> l(blockingReg_->mu);
    blockingReg_->stopping = true;
    for (auto& t : blockingReg_->tokens) t->cancel();
  }
  cv_.notify_all();
  for (auto& t : workers_) t.join();
  workers_.clear();
}

void ConsulBackend::workerMain() {
  resetThreadScheduling();  // do not inherit the reactor's RT priority
  while (true) {
    std::function<void()> task;
    {
      std::unique_lock<std::mutex> l(mu_);
      cv_.wait(l, [this] { return stopping_ || !tasks_.empty(); });
      if (stopping_ && tasks_.empty()) return;
      auto& front = tasks_.front();
      if (!front.first.empty()) queuedKeys_.erase(front.first);
      task = std::move(front.second);
      tasks_.pop_front();
    }
    task();
  }
}

bool ConsulBackend::enqueue(const std::string& key,
                            std::function<void()> task) {
  {
    std::lock_guard<std::mutex> l(mu_);
    if (stopping_) return false;
    // coalesce: a second heartbeat/registration for the same target is
    // redundant while one is still queued behind a slow agent
    if (!key.empty() && !queuedKeys_.insert(key).second) return false;
    // hard cap: a dead-slow agent must not grow the daemon unboundedly
    if (tasks_.size() >= 10000) {
      if (++dropped_ % 1000 == 1)
        LOG_WARN("consul task queue full, dropping requests (%llu dropped)",
                 (unsigned long long)dropped_);
      if (!key.empty()) queuedKeys_.erase(key);
      return false;
    }
    tasks_.emplace_back(key, std::move(task));
  }
  cv_.notify_one();
  return true;
}

void ConsulBackend::serviceRegister(
    const std::string& id, const std::string& name,
    const std::vector<std::string>& tags, int port, const std::string& address,
    bool enableTagOverride, int ttlSeconds, const std::string& status,
    const std::string& deregisterAfter, DoneCb cb) {
  JsonObject check;
  check.emplace_back("TTL", Json(std::to_string(ttlSeconds) + "s"));
  if (!status.empty()) check.emplace_back("Status", Json(status));
  check.emplace_back("Notes",
                     Json("TTL for " + name + " set by containerpilot"));
  if (!deregisterAfter.empty())
    check.emplace_back("DeregisterCriticalServiceAfter", Json(deregisterAfter));

  JsonObject payload;
  payload.emplace_back("ID", Json(id));
  payload.emplace_back("Name", Json(name));
  if (!tags.empty()) {
    JsonArray tagArr;
    for (auto& t : tags) tagArr.push_back(Json(t));
    payload.emplace_back("Tags", Json(std::move(tagArr)));
  }
  payload.emplace_back("Port", Json((int64_t)port));
  payload.emplace_back("Address", Json(address));
  if (enableTagOverride)
    payload.emplace_back("EnableTagOverride", Json(true));
  payload.emplace_back("Check", Json(std::move(check)));
  std::string body = Json(std::move(payload)).dump();

  std::string target = address_;
  std::string token = token_;
  http::TlsOptions tls = tls_;
  Loop* loop = loop_;
  bool accepted = enqueue("reg:" + id, [target, token, tls, body, cb, loop] {
    std::map<std::string, std::string> headers;
    if (!token.empty()) headers["X-Consul-Token"] = token;
    auto res = http::request(target, "PUT", "/v1/agent/service/register",
                             body, "application/json", headers, 10000, &tls);
    bool ok = res.ok && res.status == 200;
    std::string err = res.ok ? ("status " + std::to_string(res.status) + ": " +
                                res.body)
                             : res.error;
    loop->post(timedItem("consul", [cb, ok, err] { cb(ok, ok ? "" : err); }));
  });
  if (!accepted)
    loop_->post([cb] { cb(false, "request coalesced or queue full"); });
}

void ConsulBackend::updateTTL(const std::string& checkID,
                              const std::string& output,
                              const std::string& status, DoneCb cb) {
  // the Consul agent API maps "pass" -> passing (api.Agent.UpdateTTL)
  std::string st = status;
  if (st == "pass") st = "passing";
  else if (st == "warn") st = "warning";
  else if (st == "fail") st = "critical";
  JsonObject payload;
  payload.emplace_back("Status", Json(st));
  payload.emplace_back("Output", Json(output));
  std::string body = Json(std::move(payload)).dump();
  std::string path = "/v1/agent/check/update/" + urlEncode(checkID);

  std::string target = address_;
  std::string token = token_;
  http::TlsOptions tls = tls_;
  Loop* loop = loop_;
  bool accepted =
      enqueue("ttl:" + checkID, [target, token, tls, path, body, cb, loop] {
    std::map<std::string, std::string> headers;
    if (!token.empty()) headers["X-Consul-Token"] = token;
    auto res = http::request(target, "PUT", path, body, "application/json",
                             headers, 10000, &tls);
    bool ok = res.ok && res.status == 200;
    std::string err =
        res.ok ? ("status " + std::to_string(res.status) + ": " + res.body)
               : res.error;
    loop->post(timedItem("consul", [cb, ok, err] { cb(ok, ok ? "" : err); }));
  });
  if (!accepted) {
    // a heartbeat for this check is already queued; nothing to report
    (void)cb;
  }
}

void ConsulBackend::serviceDeregister(const std::string& id, DoneCb cb) {
  std::string path = "/v1/agent/service/deregister/" + urlEncode(id);
  std::string target = address_;
  std::string token = token_;
  http::TlsOptions tls = tls_;
  Loop* loop = loop_;
  bool accepted = enqueue("", [target, token, tls, path, cb, loop] {
    std::map<std::string, std::string> headers;
    if (!token.empty()) headers["X-Consul-Token"] = token;
    auto res = http::request(target, "PUT", path, "", "application/json",
                             headers, 10000, &tls);
    bool ok = res.ok && res.status == 200;
    std::string err =
        res.ok ? ("status " + std::to_string(res.status)) : res.error;
    loop->post(timedItem("consul", [cb, ok, err] { cb(ok, ok ? "" : err); }));
  });
  if (!accepted) loop_->post([cb] { cb(false, "queue full"); });
}

namespace {

std::vector<ServiceEntry> parseHealthEntries(const std::string& body,
                                             bool* ok) {
  std::vector<ServiceEntry> entries;
  try {
    Json doc = parseJson5(body);
    if (doc.isArray()) {
      for (auto& e : doc.array()) {
        const Json* svc = e.find("Service");
        if (!svc || !svc->isObject()) continue;
        ServiceEntry entry;
        if (const Json* v = svc->find("ID"))
          if (v->isString()) entry.id = v->str();
        if (const Json* v = svc->find("Address"))
          if (v->isString()) entry.address = v->str();
        if (const Json* v = svc->find("Port"))
          if (v->isNumber()) entry.port = (int)v->asInt();
        entries.push_back(std::move(entry));
      }
    }
  } catch (const std::exception&) {
    *ok = false;
  }
  return entries;
}

}  // namespace

void ConsulBackend::healthService(const std::string& name,
                                  const std::string& tag,
                                  const std::string& dc, HealthCb cb) {
  std::string path = "/v1/health/service/" + urlEncode(name) + "?passing=1";
  if (!tag.empty()) path += "&tag=" + urlEncode(tag);
  if (!dc.empty()) path += "&dc=" + urlEncode(dc);
  std::string target = address_;
  std::string token = token_;
  http::TlsOptions tls = tls_;
  Loop* loop = loop_;
  bool accepted = enqueue("health:" + name + "|" + tag + "|" + dc,
                          [target, token, tls, path, cb, loop] {
    std::map<std::string, std::string> headers;
    if (!token.empty()) headers["X-Consul-Token"] = token;
    auto res = http::request(target, "GET", path, "", "application/json",
                             headers, 10000, &tls);
    bool ok = res.ok && res.status == 200;
    std::vector<ServiceEntry> entries;
    if (ok) entries = parseHealthEntries(res.body, &ok);
    loop->post(timedItem("consul",
                         [cb, ok, entries = std::move(entries)]() mutable {
                           cb(ok, std::move(entries));
                         }));
  });
  if (!accepted) {
    loop_->post([cb] { cb(false, {}); });
  }
}

void ConsulBackend::healthServiceBlocking(const std::string& name,
                                          const std::string& tag,
                                          const std::string& dc,
                                          uint64_t lastIndex, int waitSeconds,
                                          HealthBlockingCb cb) {
  std::string path = "/v1/health/service/" + urlEncode(name) + "?passing=1";
  if (!tag.empty()) path += "&tag=" + urlEncode(tag);
  if (!dc.empty()) path += "&dc=" + urlEncode(dc);
  path += "&index=" + std::to_string(lastIndex) +
          "&wait=" + std::to_string(waitSeconds) + "s";
  auto token = std::make_shared<http::CancelToken>();
  auto reg = blockingReg_;
  {
    std::lock_guard<std::mutex> l(reg->mu);
    if (reg->stopping) {
      loop_->post([cb] { cb(false, {}, 0); });
      return;
    }
    reg->tokens.insert(token);
  }
  std::string target = address_;
  std::string tokenHdr = token_;
  http::TlsOptions tls = tls_;
  Loop* loop = loop_;
  // a parked long-poll must not occupy the shared worker pool (N
  // blocking watches would starve TTL heartbeats), so each query gets a
  // short-lived thread of its own; `this` is NOT captured — the thread
  // may outlive the backend and touches only the shared reg + the loop
  std::thread([reg, target, tokenHdr, tls, path, cb, loop, token,
               waitSeconds] {
    resetThreadScheduling();
    std::map<std::string, std::string> headers;
    if (!tokenHdr.empty()) headers["X-Consul-Token"] = tokenHdr;
    auto res = http::request(target, "GET", path, "", "application/json",
                             headers, (waitSeconds + 10) * 1000, &tls,
                             token.get());
    {
      std::lock_guard<std::mutex> l(reg->mu);
      reg->tokens.erase(token);
    }
    bool ok = res.ok && res.status == 200 && !token->cancelled();
    uint64_t index = 0;
    auto it = res.headers.find("x-consul-index");
    if (it != res.headers.end())
      index = strtoull(it->second.c_str(), nullptr, 10);
    std::vector<ServiceEntry> entries;
    if (ok) entries = parseHealthEntries(res.body, &ok);
    loop->post([cb, ok, index, entries = std::move(entries)]() mutable {
      cb(ok, std::move(entries), index);
    });
  }).detach();
}

bool ConsulBackend::compareAndSwap(const std::string& service,
                                   std::vector<ServiceEntry> entries) {
  std::vector<ServiceEntry> existing = watched_[service];
  watched_[service] = entries;
  // compareForChange (discovery/consul.go:112-125)
  if (existing.size() != entries.size()) return true;
  auto byID = [](const ServiceEntry& a, const ServiceEntry& b) {
    return a.id < b.id;
  };
  std::sort(existing.begin(), existing.end(), byID);
  std::sort(entries.begin(), entries.end(), byID);
  for (size_t i = 0; i < existing.size(); i++) {
    if (existing[i].address != entries[i].address ||
        existing[i].port != entries[i].port)
      return true;
  }
  return false;
}

// ---------------- ServiceDefinition ----------------

void ServiceDefinition::sendHeartbeat() {
  registerService("passing");
  std::string checkID = "service:" + id;
  consul->updateTTL(checkID, "ok", "pass", [](bool ok, const std::string& err) {
    if (!ok) LOG_WARN("service update TTL failed: %s", err.c_str());
  });
}

void ServiceDefinition::registerWithInitialStatus() {
  if (wasRegistered) return;
  std::string status;
  if (initialStatus == "passing") status = "passing";
  else if (initialStatus == "warning") status = "warning";
  else if (initialStatus == "critical") status = "critical";
  LOG_INFO("Registering service %s with initial status set to %s",
           name.c_str(), initialStatus.c_str());
  registerService(status);
}

void ServiceDefinition::registerService(const std::string& status) {
  if (wasRegistered || registerInFlight) return;
  registerInFlight = true;
  // weak capture: the completion is posted onto the reactor by a worker
  // thread and can land after a reload has freed this generation's
  // object graph (e.g. a 10 s HTTP timeout against an unreachable agent
  // straddling teardown) — it must then no-op, not dereference this
  std::weak_ptr<ServiceDefinition> weak = weak_from_this();
  consul->serviceRegister(
      id, name, tags, port, ipAddress, enableTagOverride, ttl, status,
      deregisterCriticalServiceAfter,
      [weak](bool ok, const std::string& err) {
        auto self = weak.lock();
        if (!self) return;  // generation torn down while in flight
        self->registerInFlight = false;
        if (!ok) {
          LOG_WARN("service registration failed: %s", err.c_str());
          return;
        }
        LOG_INFO("Service registered: %s", self->name.c_str());
        self->wasRegistered = true;
      });
}

void ServiceDefinition::deregister() {
  LOG_DEBUG("deregistering: %s", id.c_str());
  consul->serviceDeregister(id, [](bool ok, const std::string& err) {
    if (!ok) LOG_INFO("deregistering failed: %s", err.c_str());
  });
}

void ServiceDefinition::markForMaintenance() { deregister(); }

}  // namespace cpilot
