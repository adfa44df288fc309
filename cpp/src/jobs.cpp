#include "cpilot/jobs.hpp"

#include <unistd.h>

#include <cstring>

#include "cpilot/decode.hpp"
#include "cpilot/ips.hpp"
#include "cpilot/log.hpp"
#include "cpilot/spawner.hpp"

namespace cpilot {

const char* jobStatusString(JobStatus s) {
  switch (s) {
    case JobStatus::Healthy: return "healthy";
    case JobStatus::Unhealthy: return "unhealthy";
    case JobStatus::Maintenance: return "maintenance";
    case JobStatus::AlwaysHealthy: return "healthy";
    case JobStatus::Completed: return "completed";
    default: return "unknown";
  }
}

// ---------------- config ----------------

namespace {

constexpr auto kTaskMinDuration = std::chrono::milliseconds(1);

// strict: a non-bool value is an error (mapstructure semantics)
bool getRawBool(const Json* obj, const char* key, bool def, bool* out,
                std::string* err) {
  *out = def;
  if (!obj) return true;
  const Json* v = obj->find(key);
  if (!v) return true;
  if (!decode::toBool(*v, out)) {
    *err = std::string("cannot parse '") + key + "' as bool";
    return false;
  }
  return true;
}

// parse when{} (jobs/config.go:178-246)
bool validateWhen(const Json* when, std::shared_ptr<JobConfig>& cfg,
                  std::string* err) {
  if (!when || when->isNull()) {
    cfg->whenTimeout = Duration(0);
    cfg->whenEvent = GlobalStartup;
    cfg->whenStartsLimit = 1;
    return true;
  }
  if (!when->isObject()) {
    *err = "job[" + cfg->name + "].when must be an object";
    return false;
  }
  if (!decode::checkKeys(*when,
                         {"interval", "source", "once", "each", "timeout"},
                         err)) {
    *err = "job configuration error: " + *err;
    return false;
  }
  std::string frequency, source, once, each, timeout;
  if (const Json* v = when->find("interval")) decode::toString(*v, &frequency);
  if (const Json* v = when->find("source")) decode::toString(*v, &source);
  if (const Json* v = when->find("once")) decode::toString(*v, &once);
  if (const Json* v = when->find("each")) decode::toString(*v, &each);
  if (const Json* v = when->find("timeout")) decode::toString(*v, &timeout);

  int set = (!frequency.empty()) + (!once.empty()) + (!each.empty());
  if (set > 1) {
    *err = "job[" + cfg->name +
           "].when can have only one of 'interval', 'once', or 'each'";
    return false;
  }
  if (!frequency.empty()) {
    Duration freq;
    try {
      freq = parseDuration(Json(frequency));
    } catch (const std::exception& e) {
      *err = "unable to parse job[" + cfg->name + "].when.interval '" +
             frequency + "': " + e.what();
      return false;
    }
    if (freq < kTaskMinDuration) {
      *err = "job[" + cfg->name + "].when.interval '" + frequency +
             "' cannot be less than 1ms";
      return false;
    }
    cfg->freqInterval = freq;
    cfg->whenTimeout = Duration(0);
    cfg->whenEvent = GlobalStartup;
    cfg->whenStartsLimit = 1;
    return true;
  }
  // event-driven when (jobs/config.go:217-246)
  try {
    cfg->whenTimeout = getTimeout(timeout);
  } catch (const std::exception& e) {
    *err = "unable to parse job[" + cfg->name + "].when.timeout: " + e.what();
    return false;
  }
  EventCode code = EventCode::None;
  bool parseErr = false;
  if (!once.empty()) {
    parseErr = !eventCodeFromString(once, &code);
    cfg->whenStartsLimit = 1;
  }
  if (!each.empty() && once.empty()) {
    parseErr = !eventCodeFromString(each, &code);
    cfg->whenStartsLimit = kUnlimited;
  }
  if (parseErr) {
    *err = "unable to parse job[" + cfg->name + "].when.event: " +
           (once.empty() ? each : once) + " is not a valid event code";
    return false;
  }
  if (source == "SIGHUP" || source == "SIGUSR2") {
    code = EventCode::Signal;
    cfg->whenStartsLimit = kUnlimited;
  }
  cfg->whenEvent = Event{code, source};
  return true;
}

bool validateRestarts(const Json* restarts, const Json* when,
                      std::shared_ptr<JobConfig>& cfg, std::string* err) {
  // defaults (jobs/config.go:348-356)
  if (!restarts || restarts->isNull()) {
    cfg->restartLimit = (cfg->freqInterval > Duration(0)) ? kUnlimited : 0;
    return true;
  }
  std::string whenEach;
  if (when && when->isObject())
    if (const Json* v = when->find("each")) decode::toString(*v, &whenEach);

  auto fmtErr = [&](const std::string& why) {
    std::string rendered =
        restarts->isString() ? restarts->str() : restarts->dump();
    *err = "job[" + cfg->name + "].restarts field '" + rendered +
           "' invalid: " + why;
  };

  if (restarts->isString()) {
    const std::string& t = restarts->str();
    if (t == "unlimited") {
      if (!whenEach.empty()) {
        fmtErr(
            "may not be used when 'job.when.each' is set because it may "
            "result in infinite processes");
        return false;
      }
      cfg->restartLimit = kUnlimited;
    } else if (t == "never") {
      cfg->restartLimit = 0;
    } else {
      char* end = nullptr;
      long v = strtol(t.c_str(), &end, 10);
      if (end && *end == '\0' && !t.empty() && v >= 0) {
        cfg->restartLimit = (int)v;
      } else {
        fmtErr("accepts positive integers, \"unlimited\", or \"never\"");
        return false;
      }
    }
    return true;
  }
  if (restarts->isInt()) {
    if (restarts->asInt() < 0) {
      fmtErr("number must be positive integer");
      return false;
    }
    cfg->restartLimit = (int)restarts->asInt();
    return true;
  }
  if (restarts->isDouble()) {
    // undocumented truncation, jobs/config.go:375-389
    if (restarts->asDouble() < 0) {
      fmtErr("number must be positive integer");
      return false;
    }
    cfg->restartLimit = (int)restarts->asDouble();
    return true;
  }
  fmtErr("accepts positive integers, \"unlimited\", or \"never\"");
  return false;
}

// health{} block (jobs/config.go:297-344)
bool validateHealthCheck(const Json& raw, std::shared_ptr<JobConfig>& cfg,
                         std::string* err) {
  const Json* health = raw.find("health");
  if (cfg->port != 0 && (!health || health->isNull()) &&
      cfg->name != "containerpilot") {
    *err = "job[" + cfg->name + "].health must be set if 'port' is set";
    return false;
  }
  if (!health || health->isNull()) return true;
  if (!health->isObject()) {
    *err = "job[" + cfg->name + "].health must be an object";
    return false;
  }
  if (!decode::checkKeys(*health,
                         {"exec", "timeout", "interval", "ttl", "logging"},
                         err)) {
    *err = "job configuration error: " + *err;
    return false;
  }
  // interval: int = seconds (reference-compatible, jobs/config.go:74) or a
  // Go duration string (extension, so sub-second check cadences like the
  // baseline's 100ms are expressible)
  Duration heartbeat{0};
  int ttl = 0;
  if (const Json* v = health->find("interval")) {
    if (v->isString() && !v->str().empty() &&
        !isdigit((unsigned char)v->str()[0]) ) {
      *err = "job[" + cfg->name + "].health.interval must be > 0";
      return false;
    }
    try {
      heartbeat = parseDuration(*v);
    } catch (const std::exception&) {
      *err = "job[" + cfg->name + "].health.interval must be > 0";
      return false;
    }
  }
  if (const Json* v = health->find("ttl")) decode::toInt(*v, &ttl);
  if (heartbeat < std::chrono::milliseconds(1)) {
    *err = "job[" + cfg->name + "].health.interval must be > 0";
    return false;
  }
  if (ttl < 1) {
    *err = "job[" + cfg->name + "].health.ttl must be > 0";
    return false;
  }
  cfg->ttl = ttl;
  cfg->heartbeatInterval = heartbeat;

  Duration checkTimeout = cfg->heartbeatInterval;
  std::string timeoutStr;
  if (const Json* v = health->find("timeout"))
    decode::toString(*v, &timeoutStr);
  if (!timeoutStr.empty()) {
    try {
      checkTimeout = getTimeout(timeoutStr);
    } catch (const std::exception& e) {
      *err = "could not parse job[" + cfg->name + "].health.timeout '" +
             timeoutStr + "': " + e.what();
      return false;
    }
  }

  const Json* checkExec = health->find("exec");
  if (checkExec && !checkExec->isNull()) {
    std::string checkName = "check." + cfg->name;
    bool raw_ = false;
    std::string boolErr;
    if (!getRawBool(health->find("logging"), "raw", false, &raw_, &boolErr)) {
      *err = "job[" + cfg->name + "].health.logging: " + boolErr;
      return false;
    }
    std::string cmdErr;
    CommandPtr cmd =
        newCommand(*checkExec, checkTimeout, raw_, checkName, &cmdErr);
    if (!cmd) {
      *err = "unable to create job[" + cfg->name + "].health.exec: " + cmdErr;
      return false;
    }
    cmd->setName(checkName);
    cfg->healthCheckExec = cmd;
  }
  return true;
}

// discovery wiring (jobs/config.go:139-160, 400-440)
bool validateDiscovery(const Json& raw, ConsulBackend* disc,
                       std::shared_ptr<JobConfig>& cfg, std::string* err) {
  if (!validateHealthCheck(raw, cfg, err)) return false;
  if ((cfg->port == 0 || disc == nullptr) && !cfg->name.empty()) return true;

  // initial_status (jobs/config.go:162-176)
  if (!cfg->initialStatus.empty() && cfg->initialStatus != "passing" &&
      cfg->initialStatus != "warning" && cfg->initialStatus != "critical") {
    *err = "job[" + cfg->name +
           "].initialStatus must be one of 'passing', 'warning' or 'critical'";
    return false;
  }
  if (!validateServiceName(cfg->name, err)) return false;

  // addDiscoveryConfig (jobs/config.go:400-440)
  std::vector<std::string> interfaces;
  if (const Json* v = raw.find("interfaces")) {
    if (!decode::toStrings(*v, &interfaces)) {
      *err = "job[" + cfg->name + "].interfaces must be a string or array";
      return false;
    }
  }
  std::string ip, ipErr;
  if (!getIP(interfaces, &ip, &ipErr)) {
    *err = ipErr;
    return false;
  }
  char hostname[256] = {0};
  gethostname(hostname, sizeof(hostname) - 1);
  std::string id = cfg->name + "-" + hostname;

  bool enableTagOverride = false;
  std::string deregAfter;
  const Json* extras = raw.find("consul");
  if (extras && !extras->isNull()) {
    if (!extras->isObject()) {
      *err = "job[" + cfg->name + "].consul must be an object";
      return false;
    }
    if (!decode::checkKeys(
            *extras, {"enableTagOverride", "deregisterCriticalServiceAfter"},
            err)) {
      *err = "job configuration error: " + *err;
      return false;
    }
    if (const Json* v = extras->find("deregisterCriticalServiceAfter")) {
      decode::toString(*v, &deregAfter);
      if (!deregAfter.empty()) {
        try {
          parseGoDuration(deregAfter);
        } catch (const std::exception& e) {
          *err = "unable to parse job[" + cfg->name +
                 "].consul.deregisterCriticalServiceAfter: " + e.what();
          return false;
        }
      }
    }
    if (const Json* v = extras->find("enableTagOverride")) {
      if (!decode::toBool(*v, &enableTagOverride)) {
        *err = "job configuration error: cannot parse "
               "'enableTagOverride' as bool for job[" + cfg->name + "]";
        return false;
      }
    }
  }

  auto svc = std::make_shared<ServiceDefinition>();
  svc->id = id;
  svc->name = cfg->name;
  svc->port = cfg->port;
  svc->ttl = cfg->ttl;
  svc->tags = cfg->tags;
  svc->initialStatus = cfg->initialStatus;
  svc->ipAddress = ip;
  svc->deregisterCriticalServiceAfter = deregAfter;
  svc->enableTagOverride = enableTagOverride;
  svc->consul = disc;
  cfg->serviceDefinition = svc;
  return true;
}

bool validateExec(const Json& raw, std::shared_ptr<JobConfig>& cfg,
                  std::string* err) {
  std::string execTimeoutStr;
  if (const Json* v = raw.find("timeout")) decode::toString(*v, &execTimeoutStr);

  if (execTimeoutStr.empty() && cfg->freqInterval > Duration(0)) {
    // periodic tasks require a timeout (jobs/config.go:261-264)
    cfg->execTimeout = cfg->freqInterval;
  }
  if (!execTimeoutStr.empty()) {
    Duration execTimeout;
    try {
      execTimeout = getTimeout(execTimeoutStr);
    } catch (const std::exception& e) {
      *err = "unable to parse job[" + cfg->name + "].timeout '" +
             execTimeoutStr + "': " + e.what();
      return false;
    }
    if (execTimeout < std::chrono::milliseconds(1)) {
      *err = "job[" + cfg->name + "].timeout '" + execTimeoutStr +
             "' cannot be less than 1ms";
      return false;
    }
    cfg->execTimeout = execTimeout;
  }
  const Json* exec = raw.find("exec");
  if (exec && !exec->isNull()) {
    bool rawLog = false;
    std::string boolErr;
    if (!getRawBool(raw.find("logging"), "raw", false, &rawLog, &boolErr)) {
      *err = "job[" + cfg->name + "].logging: " + boolErr;
      return false;
    }
    std::string cmdErr;
    CommandPtr cmd =
        newCommand(*exec, cfg->execTimeout, rawLog, cfg->name, &cmdErr);
    if (!cmd) {
      *err = "unable to create job[" + cfg->name + "].exec: " + cmdErr;
      return false;
    }
    if (cfg->name.empty()) cfg->name = cmd->execPath();
    cmd->setName(cfg->name);
    cfg->exec = cmd;
  }
  return true;
}

}  // namespace

bool validateJobConfig(const Json& raw, ConsulBackend* disc,
                       std::shared_ptr<JobConfig>* out, std::string* err) {
  if (!raw.isObject()) {
    *err = "job configuration error: job must be an object";
    return false;
  }
  if (!decode::checkKeys(
          raw,
          {"name", "exec", "port", "initial_status", "interfaces", "tags",
           "consul", "health", "timeout", "restarts", "stopTimeout", "when",
           "logging"},
          err)) {
    *err = "job configuration error: " + *err;
    return false;
  }
  auto cfg = std::make_shared<JobConfig>();
  if (const Json* v = raw.find("name")) decode::toString(*v, &cfg->name);
  if (const Json* v = raw.find("port")) {
    if (!decode::toInt(*v, &cfg->port)) {
      *err = "job[" + cfg->name + "].port must be a number";
      return false;
    }
  }
  if (const Json* v = raw.find("initial_status"))
    decode::toString(*v, &cfg->initialStatus);
  if (const Json* v = raw.find("tags")) {
    if (!decode::toStrings(*v, &cfg->tags)) {
      *err = "job[" + cfg->name + "].tags must be an array of strings";
      return false;
    }
  }
  if (const Json* logging = raw.find("logging")) {
    if (!logging->isNull() && !logging->isObject()) {
      *err = "job[" + cfg->name + "].logging must be an object";
      return false;
    }
    if (logging->isObject() &&
        !decode::checkKeys(*logging, {"raw"}, err)) {
      *err = "job configuration error: " + *err;
      return false;
    }
  }

  // order mirrors Config.Validate (jobs/config.go:118-133)
  if (!validateDiscovery(raw, disc, cfg, err)) return false;
  if (!validateWhen(raw.find("when"), cfg, err)) return false;
  // stoppingTimeout (jobs/config.go:248-257)
  std::string stopTimeoutStr;
  if (const Json* v = raw.find("stopTimeout"))
    decode::toString(*v, &stopTimeoutStr);
  try {
    cfg->stoppingTimeout = getTimeout(stopTimeoutStr);
  } catch (const std::exception& e) {
    *err = "unable to parse job[" + cfg->name + "].stopTimeout '" +
           stopTimeoutStr + "': " + e.what();
    return false;
  }
  cfg->stoppingWaitEvent = NonEvent;
  if (!validateRestarts(raw.find("restarts"), raw.find("when"), cfg, err))
    return false;
  if (!validateExec(raw, cfg, err)) return false;
  *out = cfg;
  return true;
}

bool newJobConfigs(const Json& rawJobs, ConsulBackend* disc,
                   std::vector<std::shared_ptr<JobConfig>>* out,
                   std::string* err) {
  out->clear();
  if (rawJobs.isNull()) return true;
  if (!rawJobs.isArray()) {
    *err = "job configuration error: jobs must be an array";
    return false;
  }
  std::vector<std::pair<std::string, std::string>> stopDeps;  // source -> job
  for (auto& raw : rawJobs.array()) {
    std::shared_ptr<JobConfig> cfg;
    if (!validateJobConfig(raw, disc, &cfg, err)) return false;
    if (cfg->whenEvent.code == EventCode::Stopping)
      stopDeps.emplace_back(cfg->whenEvent.source, cfg->name);
    out->push_back(cfg);
  }
  // wire stopping dependencies (jobs/config.go:108-113,135-137)
  for (auto& cfg : *out) {
    for (auto& dep : stopDeps) {
      if (dep.first == cfg->name)
        cfg->stoppingWaitEvent = Event{EventCode::Stopped, dep.second};
    }
  }
  return true;
}

// ---------------- runtime ----------------

Job::Job(const std::shared_ptr<JobConfig>& cfg)
    : name_(cfg->name),
      exec_(cfg->exec),
      service_(cfg->serviceDefinition),
      healthCheckExec_(cfg->healthCheckExec),
      startEvent_(cfg->whenEvent),
      startTimeout_(cfg->whenTimeout),
      startsRemain_(cfg->whenStartsLimit),
      stoppingWaitEvent_(cfg->stoppingWaitEvent),
      stoppingTimeout_(cfg->stoppingTimeout),
      heartbeat_(cfg->heartbeatInterval),
      restartLimit_(cfg->restartLimit),
      restartsRemain_(cfg->restartLimit),
      frequency_(cfg->freqInterval) {
  heartbeatSource_ = name_ + ".heartbeat";
  runEverySource_ = name_ + ".run-every";
  healthCheckName_ =
      healthCheckExec_ ? healthCheckExec_->name() : ("check." + name_);
  stoppingTimeoutSource_ = name_ + ".stopping-timeout";
  std::hash<std::string> h;
  hName_ = h(name_);
  hHeartbeat_ = h(heartbeatSource_);
  hRunEvery_ = h(runEverySource_);
  hCheck_ = h(healthCheckName_);
  if (name_ == "containerpilot") {
    // hardcoded always-healthy telemetry job (jobs/jobs.go:82-87)
    status_ = JobStatus::AlwaysHealthy;
  }
}

Subscription Job::subscription() const {
  Subscription sub;
  sub.all = false;
  sub.sources = {name_,          healthCheckName_,
                 heartbeatSource_, runEverySource_,
                 stoppingTimeoutSource_, name_ + ".wait-timeout"};
  if (!startEvent_.source.empty()) sub.sources.push_back(startEvent_.source);
  if (stoppingWaitEvent_ != NonEvent)
    sub.sources.push_back(stoppingWaitEvent_.source);
  sub.codes = {EventCode::Quit, EventCode::Shutdown,
               EventCode::EnterMaintenance, EventCode::ExitMaintenance,
               EventCode::Signal};
  return sub;
}

void Job::run(Loop& loop, std::shared_ptr<Bus> bus,
              std::function<void()> completedCb) {
  loop_ = &loop;
  bus_ = std::move(bus);
  completedCb_ = std::move(completedCb);
  auto self = shared_from_this();

  // Stagger each job's periodic timers by a deterministic per-name phase
  // inside the first interval. Without this every job's check fires in
  // the same millisecond (timers are all created at GlobalStartup),
  // producing spawn/exit storms that batch the bus and blow out dispatch
  // latency; with it the load spreads evenly across the period.
  auto phase = [this](Duration interval) {
    uint64_t h = std::hash<std::string>{}(name_);
    return Duration(interval.count() / 2 +
                    (Ns::rep)(h % 1000) * interval.count() / 2000);
  };

  if (frequency_ > Duration(0)) {
    freqTimer_ = loop.addInterval(frequency_, [this, self] {
      LOG_DEBUG("timer: {TimerExpired %s.run-every}", name_.c_str());
      processEvent(Event{EventCode::TimerExpired, runEverySource_});
    }, phase(frequency_));
  }
  if (heartbeat_ > Duration(0)) {
    heartbeatTimer_ = loop.addInterval(heartbeat_, [this, self] {
      // heartbeat ticks for the telemetry job are not logged [GH-556]
      if (name_ != "containerpilot")
        LOG_DEBUG("timer: {TimerExpired %s.heartbeat}", name_.c_str());
      processEvent(Event{EventCode::TimerExpired, heartbeatSource_});
    }, phase(heartbeat_));
  }
  if (service_ && !service_->initialStatus.empty()) {
    // The reference retries initial-status registration at the top of
    // every event-loop pass (jobs/jobs.go:168-171); with indexed bus
    // delivery a quiet job may receive no events, so the retry runs on
    // a 1s timer until registration sticks (same observable behavior:
    // registration is retried until the agent answers).
    regRetryTimer_ = loop.addInterval(std::chrono::seconds(1), [this, self] {
      if (service_->wasRegistered) {
        loop_->cancelTimer(regRetryTimer_);
        regRetryTimer_ = 0;
        return;
      }
      checkRegistration();
    });
  }
  if (startTimeout_ > Duration(0)) {
    std::string timeoutName = name_ + ".wait-timeout";
    startTimeoutEvent_ = Event{EventCode::TimerExpired, timeoutName};
    startTimeoutTimer_ = loop.addTimeout(startTimeout_, [this, self] {
      startTimeoutTimer_ = 0;
      LOG_DEBUG("timeout: {TimerExpired %s.wait-timeout}", name_.c_str());
      processEvent(Event{EventCode::TimerExpired, name_ + ".wait-timeout"});
    });
  } else {
    startTimeoutEvent_ = NonEvent;
  }
}

void Job::kill() {
  if (exec_) exec_->kill();
}

void Job::onEvent(const Event& event) {
  processEventHashed(event, std::hash<std::string>{}(event.source));
}

void Job::onEventHashed(const Event& event, size_t srcHash) {
  processEventHashed(event, srcHash);
}

void Job::processEvent(const Event& event) {
  processEventHashed(event, std::hash<std::string>{}(event.source));
}

void Job::processEventHashed(const Event& event, size_t srcHash) {
  if (phase_ == Phase::Complete) return;
  // registration retry inside the event loop (jobs/jobs.go:168-171)
  checkRegistration();

  if (phase_ == Phase::StoppingWait) {
    // cleanup wait loop: only the awaited Stopped event or the stopping
    // timeout break it (jobs/jobs.go:397-407)
    if (event == stoppingWaitEvent_ ||
        (event.code == EventCode::TimerExpired &&
         event.source == stoppingTimeoutSource_)) {
      finishCleanup();
    }
    return;
  }

  if (event == QuitByTest) {
    cleanup();
    return;
  }
  if (dispatch(event, srcHash) == kHalt) cleanup();
}

Job::HandleResult Job::dispatch(const Event& event, size_t srcHash) {
  // match order mirrors the reference switch (jobs/jobs.go:195-232);
  // code then source-hash compared first so the common non-matching
  // case never touches the string buffers
  auto srcIs = [&](size_t h, const std::string& s) {
    return srcHash == h && event.source == s;
  };
  switch (event.code) {
    case EventCode::TimerExpired:
      if (srcIs(hHeartbeat_, heartbeatSource_))
        return onHeartbeatTimerExpired();
      if (startTimeoutEvent_ != NonEvent && event == startTimeoutEvent_)
        return onStartTimeoutExpired();
      if (srcIs(hRunEvery_, runEverySource_)) return onRunEveryTimerExpired();
      break;
    case EventCode::ExitFailed:
      if (srcIs(hCheck_, healthCheckName_)) return onHealthCheckFailed();
      if (srcIs(hName_, name_)) return onExecExit();
      break;
    case EventCode::ExitSuccess:
      if (srcIs(hCheck_, healthCheckName_)) return onHealthCheckPassed();
      if (srcIs(hName_, name_)) return onExecExit();
      break;
    case EventCode::Quit:
      if (srcIs(hName_, name_)) return onQuit();
      break;
    case EventCode::Shutdown:
      if (event == GlobalShutdown) return onQuit();
      break;
    case EventCode::EnterMaintenance:
      if (event == GlobalEnterMaintenance) return onEnterMaintenance();
      break;
    case EventCode::ExitMaintenance:
      if (event == GlobalExitMaintenance) return onExitMaintenance();
      break;
    case EventCode::Signal:
      if (event.source == "SIGHUP" || event.source == "SIGUSR2")
        return onSignalEvent(event.source);
      break;
    default:
      break;
  }
  if (event == startEvent_) return onStartEvent();
  return kContinue;
}

void Job::startJobExec() {
  startTimeoutEvent_ = NonEvent;
  if (startTimeoutTimer_) {
    loop_->cancelTimer(startTimeoutTimer_);
    startTimeoutTimer_ = 0;
  }
  setStatus(JobStatus::Unknown);
  if (exec_) exec_->run(*loop_, bus_);
}

Job::HandleResult Job::onHeartbeatTimerExpired() {
  JobStatus status = getStatus();
  if (status != JobStatus::Maintenance && status != JobStatus::Idle) {
    if (healthCheckExec_) {
      if (Spawner::global().overloaded()) {
        // early shed: joining a deep spawn backlog only inflates
        // round trips; skipping now is the same observable as the
        // single-instance skip (an overloaded daemon sheds checks)
        LOG_DEBUG("%s check shed: spawn backlog deep", name_.c_str());
      } else {
        healthCheckExec_->run(*loop_, bus_);
      }
    } else if (service_) {
      // non-checked but advertised services (telemetry endpoint)
      sendHeartbeat();
    }
  }
  return kContinue;
}

Job::HandleResult Job::onStartTimeoutExpired() {
  bus_->publish(Event{EventCode::TimerExpired, name_});
  // self-send Quit (jobs/jobs.go:259-264)
  auto self = shared_from_this();
  loop_->defer([this, self] {
    processEvent(Event{EventCode::Quit, name_});
  });
  return kContinue;
}

Job::HandleResult Job::onRunEveryTimerExpired() {
  if (!restartPermitted()) {
    LOG_DEBUG("interval expired but restart not permitted: %s", name_.c_str());
    startEvent_ = NonEvent;
    return kHalt;
  }
  restartsRemain_--;
  startJobExec();
  return kContinue;
}

Job::HandleResult Job::onHealthCheckFailed() {
  if (getStatus() != JobStatus::Maintenance) {
    setStatus(JobStatus::Unhealthy);
    bus_->publish(Event{EventCode::StatusUnhealthy, name_});
  }
  return kContinue;
}

Job::HandleResult Job::onHealthCheckPassed() {
  if (getStatus() != JobStatus::Maintenance) {
    setStatus(JobStatus::Healthy);
    bus_->publish(Event{EventCode::StatusHealthy, name_});
    sendHeartbeat();
  }
  return kContinue;
}

Job::HandleResult Job::onQuit() {
  restartsRemain_ = 0;
  if ((startEvent_.code == EventCode::Stopping ||
       startEvent_.code == EventCode::Stopped) &&
      exec_) {
    // pre-stop / post-stop jobs get one more start (jobs/jobs.go:295-308)
    if (startsRemain_ == kUnlimited) startsRemain_ = 1;
    return kContinue;
  }
  startsRemain_ = 0;
  startEvent_ = NonEvent;
  return kHalt;
}

Job::HandleResult Job::onEnterMaintenance() {
  setStatus(JobStatus::Maintenance);
  if (service_) service_->markForMaintenance();
  if (startEvent_ == GlobalEnterMaintenance) return onStartEvent();
  return kContinue;
}

Job::HandleResult Job::onExitMaintenance() {
  setStatus(JobStatus::Unknown);
  if (startEvent_ == GlobalExitMaintenance) return onStartEvent();
  return kContinue;
}

Job::HandleResult Job::onExecExit() {
  if (frequency_ > Duration(0)) return kContinue;  // periodic jobs ignore
  if (restartPermitted()) {
    restartsRemain_--;
    startJobExec();
    return kContinue;
  }
  if (startsRemain_ != 0) return kContinue;
  LOG_DEBUG("job exited but restart not permitted: %s", name_.c_str());
  startEvent_ = NonEvent;
  setStatus(JobStatus::Unknown);
  return kHalt;
}

Job::HandleResult Job::onSignalEvent(const std::string& sig) {
  if (startEvent_.code == EventCode::Signal && startEvent_.source == sig)
    startJobExec();
  return kContinue;
}

Job::HandleResult Job::onStartEvent() {
  if (startsRemain_ == 0) {
    startEvent_ = NonEvent;
    return kHalt;
  }
  if (startsRemain_ != kUnlimited) {
    startsRemain_--;
    if (startsRemain_ == 0 || restartsRemain_ == 0) {
      // prevent re-delivery while the exec is still running
      startEvent_ = NonEvent;
    }
  }
  startJobExec();
  return kContinue;
}

bool Job::restartPermitted() const {
  return restartLimit_ == kUnlimited || restartsRemain_ > 0;
}

void Job::setStatus(JobStatus s) {
  if (status_ != JobStatus::AlwaysHealthy) status_ = s;
}

void Job::checkRegistration() {
  if (service_ && !service_->initialStatus.empty())
    service_->registerWithInitialStatus();
}

void Job::sendHeartbeat() {
  if (service_) service_->sendHeartbeat();
}

// cleanup: fire Stopping, optionally wait for the stopping-dependency's
// Stopped event bounded by stoppingTimeout (jobs/jobs.go:388-416)
void Job::cleanup() {
  phase_ = Phase::StoppingWait;
  bus_->publish(Event{EventCode::Stopping, name_});
  if (stoppingWaitEvent_ != NonEvent) {
    if (stoppingTimeout_ > Duration(0)) {
      auto self = shared_from_this();
      stoppingTimer_ = loop_->addTimeout(stoppingTimeout_, [this, self] {
        stoppingTimer_ = 0;
        processEvent(
            Event{EventCode::TimerExpired, name_ + ".stopping-timeout"});
      });
    }
    return;  // wait in Phase::StoppingWait
  }
  finishCleanup();
}

void Job::finishCleanup() {
  phase_ = Phase::Complete;
  // cancel the job's context: stop timers, SIGTERM any running execs
  if (freqTimer_) loop_->cancelTimer(freqTimer_);
  if (heartbeatTimer_) loop_->cancelTimer(heartbeatTimer_);
  if (startTimeoutTimer_) loop_->cancelTimer(startTimeoutTimer_);
  if (stoppingTimer_) loop_->cancelTimer(stoppingTimer_);
  if (regRetryTimer_) loop_->cancelTimer(regRetryTimer_);
  freqTimer_ = heartbeatTimer_ = startTimeoutTimer_ = stoppingTimer_ =
      regRetryTimer_ = 0;
  if (exec_ && exec_->running()) exec_->term();
  if (healthCheckExec_ && healthCheckExec_->running()) healthCheckExec_->term();

  if (service_) service_->deregister();
  bus_->unsubscribe(this);
  complete_ = true;
  bus_->publish(Event{EventCode::Stopped, name_});
  if (completedCb_) completedCb_();
}

}  // namespace cpilot
