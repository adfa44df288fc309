#include "cpilot/app.hpp"

#include <malloc.h>
#include <poll.h>
#include <signal.h>
#include <sys/epoll.h>
#include <sys/signalfd.h>
#include <unistd.h>

#include <algorithm>
#include <cstdio>
#include <random>
#include <cstring>
#include <thread>

#include "cpilot/command.hpp"
#include "cpilot/log.hpp"

namespace cpilot {

App::App(std::string configPath, std::string statsOutPath, int benchSeconds)
    : configPath_(std::move(configPath)),
      statsOutPath_(std::move(statsOutPath)),
      benchSeconds_(benchSeconds) {}

App::~App() {
  if (signalFd_ >= 0) close(signalFd_);
}

bool App::init(std::string* err) {
  setenv("CONTAINERPILOT_PID", std::to_string(getpid()).c_str(), 1);
  cfg_ = loadConfig(configPath_, err);
  if (!cfg_) return false;
  if (!logging::init(cfg_->logConfig, err)) return false;

  exportJobIPEnv();
  return true;
}

// CONTAINERPILOT_{JOB}_IP for each advertised job (core/app.go:81-86,
// 92-97). Runs at startup AND after each reload: the reference re-runs
// NewApp per generation, so a reloaded config's new job set updates the
// exported IPs.
void App::exportJobIPEnv() {
  for (auto& jobCfg : cfg_->jobs) {
    if (jobCfg->serviceDefinition) {
      std::string key = jobCfg->name;
      for (auto& c : key) {
        c = (char)toupper((unsigned char)c);
        if (c == '-') c = '_';
      }
      key = "CONTAINERPILOT_" + key + "_IP";
      setenv(key.c_str(), jobCfg->serviceDefinition->ipAddress.c_str(), 1);
    }
  }
  commandEnvInvalidate();
}

void App::setupSignals() {
  sigset_t mask;
  sigemptyset(&mask);
  sigaddset(&mask, SIGCHLD);
  sigaddset(&mask, SIGTERM);
  sigaddset(&mask, SIGINT);
  sigaddset(&mask, SIGHUP);
  sigaddset(&mask, SIGUSR1);
  sigaddset(&mask, SIGUSR2);
  sigaddset(&mask, SIGPIPE);
  sigprocmask(SIG_BLOCK, &mask, nullptr);
  signalFd_ = signalfd(-1, &mask, SFD_NONBLOCK | SFD_CLOEXEC);
  loop_.watchFd(signalFd_, EPOLLIN, [this](uint32_t) {
    struct signalfd_siginfo si;
    while (read(signalFd_, &si, sizeof(si)) == sizeof(si)) {
      switch (si.ssi_signo) {
        case SIGCHLD:
          loop_.reapChildren();
          break;
        case SIGTERM:
        case SIGINT:
          // Terminate (core/signals.go:21-23)
          if (bus_) bus_->shutdown();
          break;
        case SIGHUP:
          if (bus_) bus_->publishSignal("SIGHUP");
          break;
        case SIGUSR2:
          if (bus_) bus_->publishSignal("SIGUSR2");
          break;
        case SIGUSR1:
          logging::reopen();  // logrotate (config/logger/logging.go:116-129)
          break;
        default:
          break;
      }
    }
  });
}

void App::ShutdownWatcher::onEvent(const Event& event) {
  if (event.code != EventCode::Shutdown) return;
  App* a = app;
  // a generation with zero (or already completed) jobs still finishes
  a->loop_.defer([a] {
    for (auto& job : a->jobs_) {
      if (!job->isComplete()) return;
    }
    a->maybeFinishGeneration();
  });
  // Bound the graceful drain: the reference relies on Docker's SIGKILL
  // when a job's stop chain hangs past stopTimeout (core/app.go:146-157
  // only sweeps after all jobs completed, so a hung pre-stop exec stalls
  // shutdown forever). We SIGKILL all job process groups after
  // stopTimeout so SIGTERM drains are always bounded.
  if (a->killSweepTimer_ == 0 && a->cfg_->stopTimeout > 0) {
    a->killSweepTimer_ = a->loop_.addTimeout(
        std::chrono::seconds(a->cfg_->stopTimeout), [a] {
          a->killSweepTimer_ = 0;
          if (a->finishing_) return;
          LOG_DEBUG("stop timeout exceeded, killing all job processes");
          for (auto& job : a->jobs_) {
            if (!job->isComplete()) job->kill();
          }
        });
  }
}

void App::startGeneration() {
  finishing_ = false;
  bus_ = std::make_shared<Bus>(loop_);
  shutdownWatcher_.app = this;
  bus_->subscribe(&shutdownWatcher_);
  cfg_->discovery->start(loop_);

  control_ = std::make_unique<ControlServer>(loop_, cfg_->control.socketPath);
  std::string err;
  if (!control_->start(bus_, &err)) {
    logging::logf(logging::Level::Fatal, "%s", err.c_str());
  }

  jobs_.clear();
  for (auto& jobCfg : cfg_->jobs)
    jobs_.push_back(std::make_shared<Job>(jobCfg));
  watches_.clear();
  for (auto& watchCfg : cfg_->watches)
    watches_.push_back(std::make_shared<Watch>(watchCfg));
  telemetry_ =
      cfg_->telemetry ? std::make_unique<Telemetry>(loop_, cfg_->telemetry)
                      : nullptr;
  if (telemetry_) {
    telemetry_->monitorJobs(jobs_);
    telemetry_->monitorWatches(watches_);
  }

  // subscribe all jobs before running any so no job misses another's
  // startup events (core/app.go:200-210)
  for (auto& job : jobs_) bus_->subscribe(job.get());
  for (auto& job : jobs_)
    job->run(loop_, bus_, [this] { onJobComplete(); });
  for (auto& watch : watches_)
    watch->run(loop_, bus_, cfg_->discovery.get());
  if (telemetry_) {
    for (auto& metric : telemetry_->metrics()) metric->run(bus_);
    std::string terr;
    if (!telemetry_->start(&terr))
      logging::logf(logging::Level::Fatal, "%s", terr.c_str());
  }

  if (benchSeconds_ > 0) {
    loop_.addTimeout(std::chrono::seconds(benchSeconds_), [this] {
      LOG_INFO("bench window complete, shutting down");
      if (bus_) bus_->shutdown();
    });
  }

  bus_->publish(GlobalStartup);
}

void App::onJobComplete() {
  for (auto& job : jobs_) {
    if (!job->isComplete()) return;
  }
  maybeFinishGeneration();
}

void App::maybeFinishGeneration() {
  if (finishing_) return;
  finishing_ = true;
  // let queued events drain before tearing the generation down
  loop_.defer([this] { teardownGeneration(); });
}

void App::teardownGeneration() {
  if (killSweepTimer_) {
    loop_.cancelTimer(killSweepTimer_);
    killSweepTimer_ = 0;
  }
  // aux components shut down after all jobs are complete
  // (core/app.go:104-140 completion watcher -> ctx cancel)
  if (control_) control_->stop();
  if (telemetry_) telemetry_->stop();
  for (auto& watch : watches_) watch->stop(loop_);
  cfg_->discovery->stop();

  totalPublished_ += bus_->publishedCount();
  totalDelivered_ += bus_->deliveredCount();
  const auto& window = bus_->latencyWindow();
  latencies_.insert(latencies_.end(), window.begin(), window.end());
  // bound the whole-run stats sample across many reload generations
  // (each generation contributes up to 64k samples)
  constexpr size_t kMaxSamples = 262144;
  if (latencies_.size() > kMaxSamples) {
    std::minstd_rand rng(42);
    std::shuffle(latencies_.begin(), latencies_.end(), rng);
    latencies_.resize(kMaxSamples / 2);
  }

  loop_.stop();
}

int App::run() {
  // NOTE on scheduling: an experiment ran the reactor thread at
  // SCHED_RR to shield its dispatch tail from the daemon's own child
  // churn. Measured result: where RT was actually granted it CUT
  // throughput ~4x under load (the RT loop starves the reader/helper
  // threads it depends on and trips RT throttling windows), and on the
  // target boxes CAP_SYS_NICE isn't granted anyway. The daemon runs at
  // default priority.
  // long-running daemon hygiene: cap glibc arenas (worker threads
  // otherwise each grow their own) and periodically return freed pages
  // to the OS so RSS tracks live data instead of allocator high-water.
  // 8 arenas, not 2: with the consul pool + spawner reader allocating
  // concurrently, 2 arenas contended with the reactor's own mallocs
  // (measured: ~40 us average per posted callback at high load)
  mallopt(M_ARENA_MAX, 8);
  loop_.addInterval(std::chrono::seconds(60), [] { malloc_trim(0); });

  setupSignals();
  startTime_ = Clock::now();
  if (getenv("CPILOT_MEMDEBUG")) {
    loop_.addInterval(std::chrono::seconds(10), [] {
      struct mallinfo2 mi = mallinfo2();
      LOG_WARN("memdebug: heap_alloc=%zu KB arena=%zu KB free=%zu KB",
               mi.uordblks / 1024, mi.arena / 1024, mi.fordblks / 1024);
    });
  }
  while (true) {
    startGeneration();
    loop_.run();

    if (!bus_->reloadFlag()) {
      if (cfg_->stopTimeout > 0) {
        LOG_DEBUG("killing all processes in %d seconds", cfg_->stopTimeout);
        // wait out stopTimeout (core/app.go:146-157), but escalate: a
        // second SIGTERM/SIGINT during this window triggers the kill
        // sweep immediately instead of being silently swallowed
        waitStopTimeoutOrSignal(cfg_->stopTimeout);
      }
      for (auto& job : jobs_) {
        LOG_INFO("killing processes for job %s", job->name().c_str());
        job->kill();
      }
      break;
    }
    // reload (core/app.go:183-196): rebuild the whole object graph;
    // on config error the daemon exits
    std::string err;
    auto newCfg = loadConfig(configPath_, &err);
    if (!newCfg) {
      LOG_ERROR("error initializing config: %s", err.c_str());
      break;
    }
    std::string logErr;
    if (!logging::init(newCfg->logConfig, &logErr)) {
      LOG_ERROR("error initializing config: %s", logErr.c_str());
      break;
    }
    cfg_ = std::move(newCfg);
    exportJobIPEnv();
  }
  writeStats();
  return 0;
}

void App::waitStopTimeoutOrSignal(int seconds) {
  auto deadline = Clock::now() + std::chrono::seconds(seconds);
  struct pollfd pfd {signalFd_, POLLIN, 0};
  while (true) {
    auto left = std::chrono::duration_cast<std::chrono::milliseconds>(
        deadline - Clock::now());
    if (left.count() <= 0) return;
    int rc = poll(&pfd, 1, (int)left.count());
    if (rc <= 0) {
      if (rc < 0 && errno == EINTR) continue;
      return;  // timeout
    }
    struct signalfd_siginfo si;
    while (read(signalFd_, &si, sizeof(si)) == sizeof(si)) {
      if (si.ssi_signo == SIGTERM || si.ssi_signo == SIGINT) {
        LOG_INFO("second signal received, killing processes immediately");
        return;
      }
      if (si.ssi_signo == SIGCHLD) loop_.reapChildren();
    }
  }
}

void App::writeStats() {
  if (statsOutPath_.empty()) return;
  double wall = std::chrono::duration<double>(Clock::now() - startTime_).count();
  std::sort(latencies_.begin(), latencies_.end());
  auto pct = [&](double q) -> double {
    if (latencies_.empty()) return 0;
    size_t idx = (size_t)(q * (latencies_.size() - 1));
    return latencies_[idx] * 1e6;  // microseconds
  };
  FILE* f = fopen(statsOutPath_.c_str(), "w");
  if (!f) return;
  fprintf(f,
          "{\"events_published\": %llu, \"events_delivered\": %llu, "
          "\"wall_seconds\": %.6f, \"events_per_sec\": %.1f, "
          "\"deliveries_per_sec\": %.1f, "
          "\"dispatch_p50_us\": %.1f, \"dispatch_p99_us\": %.1f, "
          "\"dispatch_p999_us\": %.1f, \"dispatch_max_us\": %.1f, "
          "\"latency_samples\": %zu}\n",
          (unsigned long long)totalPublished_,
          (unsigned long long)totalDelivered_, wall,
          wall > 0 ? totalPublished_ / wall : 0,
          wall > 0 ? totalDelivered_ / wall : 0, pct(0.50), pct(0.99),
          pct(0.999), latencies_.empty() ? 0 : latencies_.back() * 1e6,
          latencies_.size());
  fclose(f);
}

}  // namespace cpilot
