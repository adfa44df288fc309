// App lifecycle: builds components from config, runs the per-generation
// event bus on the reactor, handles signals, reload generations, the
// all-jobs-complete shutdown, and the StopTimeout kill sweep.
// Parity: /root/reference/core/{app,signals}.go.
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "cpilot/config.hpp"
#include "cpilot/control.hpp"
#include "cpilot/events.hpp"
#include "cpilot/jobs.hpp"
#include "cpilot/loop.hpp"
#include "cpilot/telemetry.hpp"
#include "cpilot/watches.hpp"

namespace cpilot {

class App {
 public:
  App(std::string configPath, std::string statsOutPath = "",
      int benchSeconds = 0);
  ~App();

  // load + validate config, init logging, set CONTAINERPILOT_PID and
  // CONTAINERPILOT_{JOB}_IP env vars (core/app.go:45-89)
  bool init(std::string* err);

  // blocks until shutdown; returns process exit code
  int run();

 private:
  void setupSignals();
  void startGeneration();
  void teardownGeneration();
  void onJobComplete();
  void maybeFinishGeneration();
  void writeStats();
  void exportJobIPEnv();
  void waitStopTimeoutOrSignal(int seconds);

  std::string configPath_;
  std::string statsOutPath_;
  int benchSeconds_ = 0;

  Loop loop_;
  std::unique_ptr<AppConfig> cfg_;
  std::shared_ptr<Bus> bus_;
  std::unique_ptr<ControlServer> control_;
  std::unique_ptr<Telemetry> telemetry_;
  std::vector<std::shared_ptr<Job>> jobs_;
  std::vector<std::shared_ptr<Watch>> watches_;
  int signalFd_ = -1;
  bool finishing_ = false;
  uint64_t killSweepTimer_ = 0;

  // watches the bus for Shutdown so a generation with zero (or already
  // completed) jobs still finishes
  struct ShutdownWatcher : Subscriber {
    App* app = nullptr;
    void onEvent(const Event& event) override;
    Subscription subscription() const override {
      return {false, {}, {EventCode::Shutdown}};
    }
  };
  ShutdownWatcher shutdownWatcher_;

  // accumulated bench stats across generations
  uint64_t totalPublished_ = 0, totalDelivered_ = 0;
  std::vector<double> latencies_;
  TimePoint startTime_;
};

}  // namespace cpilot
