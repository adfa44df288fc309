// Bus capacity micro-benchmark: raw dispatch throughput of the event
// core with a realistic subscriber population (150 job-like state
// machines), no processes involved. This isolates the bus from the
// workload-fixed event rate of the stress config — bench.py measures
// the named BASELINE config; this measures the ceiling.
//
// Usage: cpilot_busbench [num_subscribers] [num_events]
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <memory>
#include <vector>

#include "cpilot/events.hpp"
#include "cpilot/loop.hpp"

using namespace cpilot;

namespace {

// a subscriber that does the same order of matching work as a Job's
// dispatch(): one switch on code + a string compare against its name
struct FakeJob : Subscriber {
  std::string name;
  std::string checkName;
  uint64_t hits = 0;
  explicit FakeJob(int i)
      : name("svc-" + std::to_string(i)),
        checkName("check.svc-" + std::to_string(i)) {}
  void onEvent(const Event& event) override {
    switch (event.code) {
      case EventCode::ExitSuccess:
        if (event.source == checkName) hits++;
        break;
      case EventCode::StatusHealthy:
        if (event.source == name) hits++;
        break;
      default:
        break;
    }
  }
};

}  // namespace

int main(int argc, char** argv) {
  int numSubs = argc > 1 ? atoi(argv[1]) : 150;
  long numEvents = argc > 2 ? atol(argv[2]) : 2'000'000;

  Loop loop;
  auto bus = std::make_shared<Bus>(loop);
  std::vector<std::unique_ptr<FakeJob>> jobs;
  for (int i = 0; i < numSubs; i++) {
    jobs.push_back(std::make_unique<FakeJob>(i));
    bus->subscribe(jobs.back().get());
  }

  long published = 0;
  auto t0 = Clock::now();
  // publish from a repeating deferred task so the loop's normal
  // publish->defer->drain path is exercised, batches of 64
  std::function<void()> pump = [&] {
    for (int i = 0; i < 64 && published < numEvents; i++) {
      int j = (int)(published % numSubs);
      if (published % 2 == 0)
        bus->publish(Event{EventCode::ExitSuccess,
                           "check.svc-" + std::to_string(j)});
      else
        bus->publish(Event{EventCode::StatusHealthy,
                           "svc-" + std::to_string(j)});
      published++;
    }
    if (published < numEvents)
      loop.defer(pump);
    else
      loop.defer([&] { loop.stop(); });
  };
  loop.defer(pump);
  loop.run();
  auto t1 = Clock::now();

  double secs = std::chrono::duration<double>(t1 - t0).count();
  uint64_t delivered = bus->deliveredCount();
  uint64_t hits = 0;
  for (auto& j : jobs) hits += j->hits;
  printf(
      "{\"subscribers\": %d, \"events_published\": %ld, "
      "\"events_delivered\": %llu, \"matched\": %llu, "
      "\"wall_seconds\": %.3f, \"published_per_sec\": %.0f, "
      "\"delivered_per_sec\": %.0f}\n",
      numSubs, published, (unsigned long long)delivered,
      (unsigned long long)hits, secs, published / secs, delivered / secs);
  return hits == (uint64_t)published ? 0 : 1;
}
