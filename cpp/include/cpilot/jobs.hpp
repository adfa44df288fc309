// Job configuration + runtime state machine.
//
// Config validation reproduces the reference's behavior table
// (jobs/config.go:91-446): `when` once/each/interval/timeout exclusivity,
// restarts ("unlimited"/"never"/int, each+unlimited forbidden, float
// truncation), health block requirements, Consul extras, stop-dependency
// wiring, and ServiceDefinition assembly (ID "<name>-<hostname>").
//
// The runtime reproduces jobs/jobs.go:27-421: start-on-event, restarts,
// periodic runs, health-check scheduling, maintenance mode, the pre/post-
// stop exception on shutdown, and stopping-wait chains — as an explicit
// state machine on the reactor instead of a goroutine per job.
#pragma once

#include <functional>
#include <memory>
#include <string>
#include <vector>

#include "cpilot/command.hpp"
#include "cpilot/discovery.hpp"
#include "cpilot/events.hpp"
#include "cpilot/json.hpp"
#include "cpilot/timing.hpp"

namespace cpilot {

constexpr int kUnlimited = -1;

enum class JobStatus {
  Idle = 0,
  Unknown,
  Healthy,
  Unhealthy,
  Maintenance,
  AlwaysHealthy,
  Completed,
};
// serialization: idle/unknown -> "unknown", alwaysHealthy -> "healthy"
// (jobs/status.go:17-34)
const char* jobStatusString(JobStatus s);

struct JobConfig {
  std::string name;
  CommandPtr exec;

  // service discovery
  int port = 0;
  std::string initialStatus;
  std::vector<std::string> tags;
  std::shared_ptr<ServiceDefinition> serviceDefinition;

  // health checking
  CommandPtr healthCheckExec;
  Duration heartbeatInterval{0};
  int ttl = 0;

  // timeouts and restarts
  Duration execTimeout{0};
  Duration stoppingTimeout{0};
  int restartLimit = 0;
  Duration freqInterval{0};

  // start/stop events
  Event whenEvent = GlobalStartup;
  Duration whenTimeout{0};
  int whenStartsLimit = 1;
  Event stoppingWaitEvent = NonEvent;
};

// Parse + validate the raw `jobs` array. disc may be null (no discovery).
// On error returns false and sets err (messages mirror the reference's).
bool newJobConfigs(const Json& rawJobs, ConsulBackend* disc,
                   std::vector<std::shared_ptr<JobConfig>>* out,
                   std::string* err);

// Validate one raw job object into a JobConfig (exposed for tests).
bool validateJobConfig(const Json& raw, ConsulBackend* disc,
                       std::shared_ptr<JobConfig>* out, std::string* err);

class Job : public Subscriber, public std::enable_shared_from_this<Job> {
 public:
  explicit Job(const std::shared_ptr<JobConfig>& cfg);

  const std::string& name() const { return name_; }
  JobStatus getStatus() const { return status_; }
  bool isComplete() const { return complete_; }
  std::shared_ptr<ServiceDefinition> service() const { return service_; }

  // Subscribe and start timers; completedCb fires once when the job
  // finishes its cleanup (core/app.go:121-140 completion watcher).
  void run(Loop& loop, std::shared_ptr<Bus> bus,
           std::function<void()> completedCb);

  // SIGKILL the exec's process group (kill sweep, core/app.go:152-155)
  void kill();

  void onEvent(const Event& event) override;
  void onEventHashed(const Event& event, size_t srcHash) override;
  // Declared interests for indexed bus delivery: the sources this job's
  // dispatch() can match (its own name, its check name, its start/stop
  // dependency sources, its timer sources) plus the broadcast codes
  // (Shutdown, maintenance, Signal, Quit). Any event outside this set
  // is a no-op in dispatch() by construction (jobs/jobs.go:195-232).
  Subscription subscription() const override;

 private:
  enum class Phase { Running, StoppingWait, Complete };
  using HandleResult = bool;  // true = halt
  static constexpr bool kContinue = false, kHalt = true;

  void processEvent(const Event& event);
  void processEventHashed(const Event& event, size_t srcHash);
  HandleResult dispatch(const Event& event, size_t srcHash);
  HandleResult onHeartbeatTimerExpired();
  HandleResult onStartTimeoutExpired();
  HandleResult onRunEveryTimerExpired();
  HandleResult onHealthCheckFailed();
  HandleResult onHealthCheckPassed();
  HandleResult onQuit();
  HandleResult onEnterMaintenance();
  HandleResult onExitMaintenance();
  HandleResult onExecExit();
  HandleResult onSignalEvent(const std::string& sig);
  HandleResult onStartEvent();
  bool restartPermitted() const;
  void startJobExec();
  void setStatus(JobStatus s);
  void checkRegistration();
  void sendHeartbeat();
  void cleanup();
  void finishCleanup();

  std::string name_;
  // precomputed event-match sources + their hashes (hot path: every bus
  // event hits dispatch() in every job; hashes keep the non-matching
  // common case to one cache line instead of string compares)
  std::string heartbeatSource_, runEverySource_, healthCheckName_,
      stoppingTimeoutSource_;
  size_t hName_ = 0, hHeartbeat_ = 0, hRunEvery_ = 0, hCheck_ = 0;
  CommandPtr exec_;
  JobStatus status_ = JobStatus::Idle;
  std::shared_ptr<ServiceDefinition> service_;
  CommandPtr healthCheckExec_;

  Event startEvent_;
  Duration startTimeout_{0};
  int startsRemain_ = 1;
  Event startTimeoutEvent_ = NonEvent;

  Event stoppingWaitEvent_ = NonEvent;
  Duration stoppingTimeout_{0};

  Duration heartbeat_{0};
  int restartLimit_ = 0;
  int restartsRemain_ = 0;
  Duration frequency_{0};

  bool complete_ = false;
  Phase phase_ = Phase::Running;

  Loop* loop_ = nullptr;
  std::shared_ptr<Bus> bus_;
  std::function<void()> completedCb_;
  uint64_t freqTimer_ = 0, heartbeatTimer_ = 0, startTimeoutTimer_ = 0,
           stoppingTimer_ = 0, regRetryTimer_ = 0;
};

}  // namespace cpilot
