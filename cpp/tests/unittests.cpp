// Assert-style unit tests for the pure components (config pipeline,
// template engine, durations, event model, IP specs, arg parsing).
// Test coverage mirrors the reference's package unit tests
// (config/*_test.go, commands/commands_test.go, events/events_test.go).
// Run via `bin/cpilot_unittests`; exits non-zero on first failure.
#include <arpa/inet.h>
#include <netinet/in.h>

#include <poll.h>

#include <atomic>
#include <cassert>
#include <cstdio>
#include <thread>
#include <vector>
#include <cstdlib>
#include <cstring>
#include <string>

#include <signal.h>
#include <sys/epoll.h>
#include <sys/signalfd.h>
#include <unistd.h>

#include "cpilot/command.hpp"
#include "cpilot/config.hpp"
#include "cpilot/jobs.hpp"
#include "cpilot/decode.hpp"
#include "cpilot/events.hpp"
#include "cpilot/http.hpp"
#include "cpilot/ips.hpp"
#include "cpilot/json.hpp"
#include "cpilot/timing.hpp"
#include "cpilot/tmpl.hpp"

using namespace cpilot;

static int failures = 0;
#define CHECK(cond)                                                       \
  do {                                                                    \
    if (!(cond)) {                                                        \
      fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__, #cond);     \
      failures++;                                                         \
    }                                                                     \
  } while (0)
#define CHECK_EQ(a, b)                                                    \
  do {                                                                    \
    auto va = (a);                                                        \
    auto vb = (b);                                                        \
    if (!(va == vb)) {                                                    \
      fprintf(stderr, "FAIL %s:%d: %s != %s\n", __FILE__, __LINE__, #a,   \
              #b);                                                        \
      failures++;                                                         \
    }                                                                     \
  } while (0)

static void testJson5() {
  // comments, unquoted keys, trailing commas, single quotes
  Json v = parseJson5(R"({
    // line comment
    unquoted: 'single',
    "trailing": [1, 2, 3,],
    /* block comment */
    hex: 0xFF,
    float: 1.5e2,
    nested: {a: true, b: null},
  })");
  CHECK(v.isObject());
  CHECK_EQ(v.find("unquoted")->str(), std::string("single"));
  CHECK_EQ(v.find("trailing")->array().size(), (size_t)3);
  CHECK_EQ(v.find("hex")->asInt(), (int64_t)255);
  CHECK_EQ(v.find("float")->asDouble(), 150.0);
  CHECK(v.find("nested")->find("b")->isNull());

  // int vs double distinction (restarts truncation semantics)
  Json n = parseJson5("{a: 3, b: 1.2}");
  CHECK(n.find("a")->isInt());
  CHECK(n.find("b")->isDouble());

  // parse errors report offsets
  bool threw = false;
  try {
    parseJson5("{a: }");
  } catch (const JsonParseError& e) {
    threw = true;
    std::string msg = formatParseError("{a: }", e);
    CHECK(msg.find("parse error at line:col") == 0);
  }
  CHECK(threw);

  // serializer round-trip
  CHECK_EQ(parseJson5("{\"a\":[1,\"x\"],\"b\":true}").dump(),
           std::string("{\"a\":[1,\"x\"],\"b\":true}"));

  // duplicate keys: last wins (Go map unmarshal semantics)
  CHECK_EQ(parseJson5("{a: 1, a: 2}").find("a")->asInt(), (int64_t)2);

  // JSON5 numeric edge forms
  CHECK_EQ(parseJson5("[+5, .5, 5., -0x10]").array().size(), (size_t)4);
  CHECK(parseJson5("Infinity").isDouble());
  CHECK(parseJson5("[-Infinity]").array()[0].asDouble() < 0);
  CHECK_EQ(parseJson5("\"\\u00e9\"").str(), std::string("\xc3\xa9"));

  // pathological nesting is rejected, not a stack overflow
  std::string deep(100000, '[');
  bool threwDeep = false;
  try {
    parseJson5(deep);
  } catch (const JsonParseError&) {
    threwDeep = true;
  }
  CHECK(threwDeep);
}

static void testDurations() {
  // bare ints are seconds (timing/duration.go:33-51)
  CHECK(parseDuration(Json((int64_t)60)) == std::chrono::seconds(60));
  // numeric strings are seconds (duration.go:53-55)
  CHECK(parseDuration(Json("60")) == std::chrono::seconds(60));
  // unit strings use Go grammar
  CHECK(parseDuration(Json("1m")) == std::chrono::minutes(1));
  CHECK(parseDuration(Json("1m30s")) == std::chrono::seconds(90));
  CHECK(parseDuration(Json("500ms")) == std::chrono::milliseconds(500));
  CHECK(parseDuration(Json("1.5h")) == std::chrono::minutes(90));
  CHECK(parseGoDuration("100us") == std::chrono::microseconds(100));
  bool threw = false;
  try {
    parseDuration(Json("xx"));
  } catch (...) {
    threw = true;
  }
  CHECK(threw);
  // floats are an error (unexpected duration of type float64)
  threw = false;
  try {
    parseDuration(Json(1.5));
  } catch (...) {
    threw = true;
  }
  CHECK(threw);
}

static void testTemplate() {
  setenv("TEST_NAME", "eleven", 1);
  setenv("TEST_PARTS", "a:b:c", 1);
  unsetenv("TEST_MISSING");

  CHECK_EQ(renderTemplate("Hello, {{.TEST_NAME}}!"),
           std::string("Hello, eleven!"));
  // missingkey=zero
  CHECK_EQ(renderTemplate("[{{.TEST_MISSING}}]"), std::string("[]"));
  // default (template.go:129-140)
  CHECK_EQ(renderTemplate("{{.TEST_MISSING | default \"World\"}}"),
           std::string("World"));
  CHECK_EQ(renderTemplate("{{.TEST_NAME | default \"World\"}}"),
           std::string("eleven"));
  CHECK_EQ(renderTemplate("{{.TEST_MISSING | default 100}}"),
           std::string("100"));
  CHECK_EQ(renderTemplate("{{.TEST_MISSING | default 10.1}}"),
           std::string("10.1"));
  // env func
  CHECK_EQ(renderTemplate("{{ env \"TEST_NAME\" }}"), std::string("eleven"));
  // split | join
  CHECK_EQ(renderTemplate("{{.TEST_PARTS | split \":\" | join \".\"}}"),
           std::string("a.b.c"));
  // replaceAll / regexReplaceAll
  CHECK_EQ(renderTemplate("{{.TEST_NAME | replaceAll \"e\" \"_\"}}"),
           std::string("_l_v_n"));
  CHECK_EQ(
      renderTemplate("{{.TEST_NAME | regexReplaceAll \"[el]+\" \"_\"}}"),
      std::string("_v_n"));
  // loop + range
  CHECK_EQ(renderTemplate("{{ range loop 3 }}x{{ end }}"), std::string("xxx"));
  CHECK_EQ(renderTemplate("{{ range $i := loop 2 5 }}{{ $i }}{{ end }}"),
           std::string("234"));
  CHECK_EQ(renderTemplate("{{ range $i := loop 5 1 }}{{ $i }}{{ end }}"),
           std::string("5432"));
  // trim markers
  CHECK_EQ(renderTemplate("a {{- \"b\" -}} c"), std::string("abc"));
  // printf + nested calls
  CHECK_EQ(renderTemplate("{{ env (printf \"TEST_%s\" \"NAME\") }}"),
           std::string("eleven"));
  // if/else
  CHECK_EQ(renderTemplate("{{ if .TEST_NAME }}y{{ else }}n{{ end }}"),
           std::string("y"));
  CHECK_EQ(renderTemplate("{{ if .TEST_MISSING }}y{{ else }}n{{ end }}"),
           std::string("n"));

  // exact expectations from the reference's template_test.go:50-70
  setenv("COUNT", "3", 1);
  CHECK_EQ(renderTemplate("{{ loop 2 5 }}"), std::string("[2 3 4]"));
  CHECK_EQ(renderTemplate("{{ loop 10 1 }}"),
           std::string("[10 9 8 7 6 5 4 3 2]"));
  CHECK_EQ(renderTemplate("{{ loop 5 }}"), std::string("[0 1 2 3 4]"));
  CHECK_EQ(renderTemplate("{{ loop .COUNT }}"), std::string("[0 1 2]"));
  CHECK_EQ(renderTemplate("{{ loop 1 .COUNT }}"), std::string("[1 2]"));
  CHECK_EQ(renderTemplate("{{ range $i := loop 2 5 -}}i={{$i}},{{ end }}"),
           std::string("i=2,i=3,i=4,"));
  unsetenv("COUNT");

  // Go template builtins are available to reference configs
  setenv("STAGE", "prod", 1);
  CHECK_EQ(renderTemplate("{{ if eq .STAGE \"prod\" }}P{{ else }}D{{ end }}"),
           std::string("P"));
  CHECK_EQ(renderTemplate("{{ if ne .STAGE \"dev\" }}y{{ end }}"),
           std::string("y"));
  CHECK_EQ(renderTemplate("{{ if and .STAGE .TEST_NAME }}both{{ end }}"),
           std::string("both"));
  CHECK_EQ(renderTemplate("{{ or .TEST_MISSING \"fallback\" }}"),
           std::string("fallback"));
  CHECK_EQ(renderTemplate("{{ if not .TEST_MISSING }}empty{{ end }}"),
           std::string("empty"));
  CHECK_EQ(renderTemplate("{{ len \"abcd\" }}"), std::string("4"));
  CHECK_EQ(renderTemplate("{{ len (loop 3) }}"), std::string("3"));
  CHECK_EQ(renderTemplate("{{ index (loop 5 8) 1 }}"), std::string("6"));
  CHECK_EQ(renderTemplate("{{ if lt 1 2 }}y{{ end }}"), std::string("y"));
  CHECK_EQ(renderTemplate("{{ print \"a\" 1 2 }}"), std::string("a1 2"));

  // comments, assignment, with, else-if, two-var range
  CHECK_EQ(renderTemplate("a{{/* ignore me */}}b"), std::string("ab"));
  CHECK_EQ(renderTemplate("{{ $x := \"v\" }}{{ $x }}{{ $x }}"),
           std::string("vv"));
  CHECK_EQ(renderTemplate("{{ with .STAGE }}got-{{ . }}{{ end }}"),
           std::string("got-prod"));
  CHECK_EQ(renderTemplate("{{ with .TEST_MISSING }}y{{ else }}n{{ end }}"),
           std::string("n"));
  CHECK_EQ(renderTemplate(
               "{{ if eq .STAGE \"dev\" }}d{{ else if eq .STAGE \"prod\" "
               "}}p{{ else }}o{{ end }}"),
           std::string("p"));
  CHECK_EQ(renderTemplate(
               "{{ range $i, $v := loop 5 8 }}{{ $i }}:{{ $v }} {{ end }}"),
           std::string("0:5 1:6 2:7 "));
  unsetenv("STAGE");
}

static void testEvents() {
  EventCode code;
  CHECK(eventCodeFromString("exitSuccess", &code) &&
        code == EventCode::ExitSuccess);
  CHECK(eventCodeFromString("healthy", &code) &&
        code == EventCode::StatusHealthy);
  CHECK(eventCodeFromString("SIGHUP", &code) && code == EventCode::Signal);
  CHECK(!eventCodeFromString("bogus", &code));
  CHECK_EQ(std::string(eventCodeString(EventCode::StatusChanged)),
           std::string("StatusChanged"));
  Event a{EventCode::Startup, "global"};
  CHECK(a == GlobalStartup);
  CHECK(a != GlobalShutdown);
}

static void testParseArgs() {
  std::string exec, err;
  std::vector<std::string> args;
  CHECK(parseArgs(Json("/bin/to run"), &exec, &args, &err));
  CHECK_EQ(exec, std::string("/bin/to"));
  CHECK_EQ(args.size(), (size_t)1);
  CHECK(parseArgs(Json(JsonArray{Json("/bin/to"), Json("-a"), Json("-b")}),
                  &exec, &args, &err));
  CHECK_EQ(args.size(), (size_t)2);
  CHECK(!parseArgs(Json(""), &exec, &args, &err));
  CHECK_EQ(err, std::string("received zero-length argument"));

  // envName (commands/commands_test.go semantics)
  Command c1("/bin/to-run.sh", {}, Duration(0), false, "");
  CHECK_EQ(c1.envName(), std::string("TO_RUN"));
  Command c2("myjob", {}, Duration(0), false, "");
  CHECK_EQ(c2.envName(), std::string("MYJOB"));
}

static void testIps() {
  std::vector<InterfaceIP> ifaces;
  auto mk = [](const char* name, const char* ip) {
    InterfaceIP i;
    i.name = name;
    i.ip = ip;
    // fill bytes via getIP parse path helper: reuse static parse by spec
    struct in_addr a4;
    if (inet_pton(AF_INET, ip, &a4) == 1) {
      memset(i.bytes, 0, 16);
      i.bytes[10] = 0xff;
      i.bytes[11] = 0xff;
      memcpy(i.bytes + 12, &a4, 4);
      i.ipv6 = false;
    } else {
      struct in6_addr a6;
      inet_pton(AF_INET6, ip, &a6);
      memcpy(i.bytes, &a6, 16);
      i.ipv6 = true;
    }
    return i;
  };
  ifaces.push_back(mk("eth0", "10.2.0.5"));
  ifaces.push_back(mk("eth1", "192.168.1.10"));
  ifaces.push_back(mk("eth1", "192.168.1.11"));
  ifaces.push_back(mk("lo", "127.0.0.1"));

  std::string out, err;
  CHECK(getIP({"eth0"}, ifaces, &out, &err) && out == "10.2.0.5");
  CHECK(getIP({"eth1[1]"}, ifaces, &out, &err) && out == "192.168.1.11");
  CHECK(getIP({"192.168.1.0/24"}, ifaces, &out, &err) &&
        out == "192.168.1.10");
  CHECK(getIP({"inet"}, ifaces, &out, &err) && out == "10.2.0.5");
  CHECK(getIP({"static:192.168.1.100"}, ifaces, &out, &err) &&
        out == "192.168.1.100");
  CHECK(getIP({"bogus0"}, ifaces, &out, &err) == false);
  // default spec list: eth0:inet then inet
  CHECK(getIP({}, ifaces, &out, &err) && out == "10.2.0.5");

  // IPv6 specs
  ifaces.push_back(mk("eth2", "fe80::1234"));
  ifaces.push_back(mk("lo", "::1"));
  std::stable_sort(ifaces.begin(), ifaces.end(),
                   [](const InterfaceIP& a, const InterfaceIP& b) {
                     if (a.name != b.name) return a.name < b.name;
                     return memcmp(a.bytes, b.bytes, 16) < 0;
                   });
  CHECK(getIP({"eth2:inet6"}, ifaces, &out, &err) && out == "fe80::1234");
  CHECK(getIP({"inet6"}, ifaces, &out, &err) && out == "fe80::1234");
  // inet6 wildcard skips ::1 loopback; eth0 has no v6 address
  CHECK(getIP({"eth0:inet6"}, ifaces, &out, &err) == false);

  CHECK(validateServiceName("my-service", &err));
  CHECK(!validateServiceName("", &err));
  CHECK(!validateServiceName("-bad", &err));
  CHECK(!validateServiceName("Bad", &err));
}

static void testConfig() {
  std::string err;
  // minimal valid config
  auto cfg = newConfig(R"({"consul": "localhost:8500",
    jobs: [{name: "hello", exec: "echo hello"}]})", &err);
  CHECK(cfg != nullptr);
  if (cfg) {
    CHECK_EQ(cfg->jobs.size(), (size_t)1);
    CHECK_EQ(cfg->jobs[0]->name, std::string("hello"));
    CHECK_EQ(cfg->stopTimeout, 5);
  }

  // unknown top-level key
  CHECK(newConfig(R"({"consul": "x:8500", "bogus": 1})", &err) == nullptr);
  CHECK(err.find("unknown config keys") != std::string::npos);

  // missing consul
  CHECK(newConfig(R"({jobs: []})", &err) == nullptr);
  CHECK(err.find("no discovery backend defined") != std::string::npos);

  // job validation: port without health
  CHECK(newConfig(R"({"consul": "x:8500",
    jobs: [{name: "svc-a", port: 80}]})", &err) == nullptr);
  CHECK(err.find("health must be set if 'port' is set") != std::string::npos);

  // health requires interval + ttl
  CHECK(newConfig(R"({"consul": "x:8500",
    jobs: [{name: "svc-a", port: 80, health: {ttl: 10}}]})",
                  &err) == nullptr);
  CHECK(err.find("health.interval must be > 0") != std::string::npos);

  // when exclusivity
  CHECK(newConfig(R"({"consul": "x:8500",
    jobs: [{name: "j", exec: "x", when: {once: "startup", each: "changed"}}]})",
                  &err) == nullptr);
  CHECK(err.find("only one of") != std::string::npos);

  // restarts: unlimited with each forbidden
  CHECK(newConfig(R"({"consul": "x:8500",
    jobs: [{name: "j", exec: "x", restarts: "unlimited",
            when: {source: "watch.w", each: "changed"}}]})",
                  &err) == nullptr);
  CHECK(err.find("infinite processes") != std::string::npos);

  // restarts float truncation (jobs/config.go:375-389)
  auto cfg2 = newConfig(R"({"consul": "x:8500",
    jobs: [{name: "j", exec: "x", restarts: 1.2}]})", &err);
  CHECK(cfg2 != nullptr);
  if (cfg2) CHECK_EQ(cfg2->jobs[0]->restartLimit, 1);

  // restarts "never"
  auto cfg3 = newConfig(R"({"consul": "x:8500",
    jobs: [{name: "j", exec: "x", restarts: "never"}]})", &err);
  CHECK(cfg3 && cfg3->jobs[0]->restartLimit == 0);

  // interval job gets unlimited restarts by default + timeout=interval
  auto cfg4 = newConfig(R"({"consul": "x:8500",
    jobs: [{name: "p", exec: "x", when: {interval: "500ms"}}]})", &err);
  CHECK(cfg4 != nullptr);
  if (cfg4) {
    CHECK_EQ(cfg4->jobs[0]->restartLimit, kUnlimited);
    CHECK(cfg4->jobs[0]->freqInterval == std::chrono::milliseconds(500));
    CHECK(cfg4->jobs[0]->execTimeout == std::chrono::milliseconds(500));
  }

  // interval below 1ms rejected
  CHECK(newConfig(R"({"consul": "x:8500",
    jobs: [{name: "p", exec: "x", when: {interval: "100us"}}]})",
                  &err) == nullptr);

  // unnamed jobs are rejected (jobs/config_test.go:243-256)
  CHECK(newConfig(R"({"consul": "x:8500",
    jobs: [{exec: "/bin/echo hi"}]})", &err) == nullptr);
  CHECK(err.find("'name' must not be blank") != std::string::npos);
  // invalid name is permitted if there is no 'port' config
  auto cfg5 = newConfig(R"({"consul": "x:8500",
    jobs: [{name: "myjob_invalid_name", exec: "myexec"}]})", &err);
  CHECK(cfg5 != nullptr);

  // stopping dependency wiring (jobs/config.go:104-113)
  auto cfg6 = newConfig(R"({"consul": "x:8500", jobs: [
    {name: "main-app", exec: "sleep 1", stopTimeout: "2s"},
    {name: "pre-stop", exec: "echo bye",
     when: {source: "main-app", once: "stopping"}}]})", &err);
  CHECK(cfg6 != nullptr);
  if (cfg6) {
    CHECK(cfg6->jobs[0]->stoppingWaitEvent ==
          (Event{EventCode::Stopped, "pre-stop"}));
    CHECK(cfg6->jobs[1]->whenEvent ==
          (Event{EventCode::Stopping, "main-app"}));
  }

  // telemetry synthetic job appended (config/config.go:176-179)
  auto cfg7 = newConfig(R"({"consul": "x:8500",
    telemetry: {port: 19090, interfaces: ["lo"],
      metrics: [{namespace: "app", subsystem: "db", name: "queries",
                 help: "count", type: "counter"}]}})", &err);
  CHECK(cfg7 != nullptr);
  if (cfg7) {
    CHECK_EQ(cfg7->jobs.size(), (size_t)1);
    CHECK_EQ(cfg7->jobs.back()->name, std::string("containerpilot"));
    CHECK(cfg7->jobs.back()->heartbeatInterval == std::chrono::seconds(5));
    CHECK_EQ(cfg7->jobs.back()->ttl, 15);
    CHECK_EQ(cfg7->telemetry->metricConfigs.size(), (size_t)1);
    CHECK_EQ(cfg7->telemetry->metricConfigs[0]->fullName,
             std::string("app_db_queries"));
  }

  // watch config
  auto cfg8 = newConfig(R"({"consul": "x:8500",
    watches: [{name: "backend", interval: 3, tag: "prod"}]})", &err);
  CHECK(cfg8 != nullptr);
  if (cfg8) {
    CHECK_EQ(cfg8->watches[0]->name, std::string("watch.backend"));
    CHECK_EQ(cfg8->watches[0]->serviceName, std::string("backend"));
  }
  CHECK(newConfig(R"({"consul": "x:8500",
    watches: [{name: "backend"}]})", &err) == nullptr);
  CHECK(err.find("interval must be > 0") != std::string::npos);

  // blocking flag: bool (extension), bad values rejected
  auto cfgB = newConfig(R"({"consul": "x:8500",
    watches: [{name: "backend", interval: 5, blocking: true}]})", &err);
  CHECK(cfgB && cfgB->watches[0]->blocking);
  CHECK(newConfig(R"({"consul": "x:8500",
    watches: [{name: "backend", interval: 5, blocking: "nope"}]})",
                  &err) == nullptr);

  // unknown job field rejected (decode.go:15-17 ErrorUnused)
  CHECK(newConfig(R"({"consul": "x:8500",
    jobs: [{name: "j", exec: "x", bogusField: true}]})", &err) == nullptr);
  CHECK(err.find("invalid keys") != std::string::npos);
}

// ---------------------------------------------------------------
// Job state-machine scenarios on a real Loop+Bus (the counterpart of
// jobs/jobs_test.go:15-206's event-sequence assertions against the
// debug ring).

struct JobScenario {
  Loop loop;
  std::shared_ptr<Bus> bus;
  std::vector<std::shared_ptr<Job>> jobs;
  int sigfd = -1;

  JobScenario() {
    bus = std::make_shared<Bus>(loop);
    sigset_t mask;
    sigemptyset(&mask);
    sigaddset(&mask, SIGCHLD);
    sigprocmask(SIG_BLOCK, &mask, nullptr);
    sigfd = signalfd(-1, &mask, SFD_NONBLOCK | SFD_CLOEXEC);
    loop.watchFd(sigfd, EPOLLIN, [this](uint32_t) {
      struct signalfd_siginfo si;
      while (read(sigfd, &si, sizeof(si)) == sizeof(si)) {
      }
      loop.reapChildren();
    });
  }
  ~JobScenario() {
    if (sigfd >= 0) {
      loop.unwatchFd(sigfd);
      close(sigfd);
    }
  }

  std::shared_ptr<Job> addJob(const std::string& jobJson) {
    addJobs("[" + jobJson + "]");
    return jobs.back();
  }

  // parse a full jobs array through newJobConfigs so stopping
  // dependencies get wired (jobs/config.go:99-113)
  void addJobs(const std::string& jobsArrayJson) {
    Json raw = parseJson5(jobsArrayJson);
    std::vector<std::shared_ptr<JobConfig>> cfgs;
    std::string err;
    if (!newJobConfigs(raw, nullptr, &cfgs, &err)) {
      fprintf(stderr, "job config error: %s\n", err.c_str());
      abort();
    }
    for (auto& cfg : cfgs) {
      auto job = std::make_shared<Job>(cfg);
      jobs.push_back(job);
      bus->subscribe(job.get());
    }
  }

  // start all jobs, publish GlobalStartup, run the loop for runMs
  void run(int runMs) {
    for (auto& job : jobs)
      job->run(loop, bus, [] {});
    bus->publish(GlobalStartup);
    loop.addTimeout(std::chrono::milliseconds(runMs),
                    [this] { loop.stop(); });
    loop.run();
  }
};

// probe subscriber recording every event it sees
struct Probe : Subscriber {
  std::vector<Event> events;
  void onEvent(const Event& event) override { events.push_back(event); }
  int count(EventCode code, const std::string& source) const {
    int n = 0;
    for (auto& e : events)
      if (e.code == code && e.source == source) n++;
    return n;
  }
  int indexOf(EventCode code, const std::string& source) const {
    for (size_t i = 0; i < events.size(); i++)
      if (events[i].code == code && events[i].source == source) return (int)i;
    return -1;
  }
};

static void testJobLifecycleSequence() {
  // one-shot job: Startup -> ExitSuccess -> Stopping -> Stopped
  // (jobs_test.go:15-48)
  JobScenario sc;
  Probe probe;
  sc.bus->subscribe(&probe);
  sc.addJob(R"({name: "myjob", exec: "true"})");
  sc.run(1500);
  CHECK_EQ(probe.count(EventCode::ExitSuccess, "myjob"), 1);
  CHECK_EQ(probe.count(EventCode::Stopping, "myjob"), 1);
  CHECK_EQ(probe.count(EventCode::Stopped, "myjob"), 1);
  int iStart = probe.indexOf(EventCode::Startup, "global");
  int iExit = probe.indexOf(EventCode::ExitSuccess, "myjob");
  int iStopping = probe.indexOf(EventCode::Stopping, "myjob");
  int iStopped = probe.indexOf(EventCode::Stopped, "myjob");
  CHECK(iStart < iExit && iExit < iStopping && iStopping < iStopped);
}

static void testJobRestartCounting() {
  // restarts: 2 -> exactly 3 runs (jobs_test.go:127-164)
  JobScenario sc;
  Probe probe;
  sc.bus->subscribe(&probe);
  sc.addJob(R"({name: "flappy", exec: "true", restarts: 2})");
  sc.run(2500);
  CHECK_EQ(probe.count(EventCode::ExitSuccess, "flappy"), 3);
  CHECK_EQ(probe.count(EventCode::Stopped, "flappy"), 1);
}

static void testJobFailedExit() {
  JobScenario sc;
  Probe probe;
  sc.bus->subscribe(&probe);
  sc.addJob(R"({name: "failing", exec: "false", restarts: 1})");
  sc.run(2000);
  CHECK_EQ(probe.count(EventCode::ExitFailed, "failing"), 2);
}

static void testPeriodicJobRuns() {
  // when.interval fires repeatedly (jobs_test.go:166-206)
  JobScenario sc;
  Probe probe;
  sc.bus->subscribe(&probe);
  sc.addJob(R"({name: "tick", exec: "true", when: {interval: "150ms"}})");
  sc.run(1000);
  int runs = probe.count(EventCode::ExitSuccess, "tick");
  CHECK(runs >= 3 && runs <= 8);
}

static void testPreStopJobOnShutdown() {
  // pre-stop jobs get one more run during shutdown (jobs/jobs.go:295-312)
  JobScenario sc;
  Probe probe;
  sc.bus->subscribe(&probe);
  sc.addJobs(R"([
    {name: "main-app", exec: "sleep 30", stopTimeout: "2s"},
    {name: "pre-stop", exec: "true",
     when: {source: "main-app", once: "stopping"}}])");
  sc.loop.addTimeout(std::chrono::milliseconds(300),
                     [&sc] { sc.bus->shutdown(); });
  sc.run(3000);
  CHECK_EQ(probe.count(EventCode::ExitSuccess, "pre-stop"), 1);
  int iPre = probe.indexOf(EventCode::ExitSuccess, "pre-stop");
  int iStopped = probe.indexOf(EventCode::Stopped, "main-app");
  CHECK(iPre >= 0 && iStopped >= 0 && iPre < iStopped);
}

static void testWhenEachFiresRepeatedly() {
  JobScenario sc;
  Probe probe;
  sc.bus->subscribe(&probe);
  sc.addJob(R"({name: "onchange", exec: "true",
                when: {source: "watch.db", each: "changed"}})");
  // fire three synthetic change events, spaced out so the exec finishes
  for (int i = 1; i <= 3; i++) {
    sc.loop.addTimeout(std::chrono::milliseconds(200 * i), [&sc] {
      sc.bus->publish(Event{EventCode::StatusChanged, "watch.db"});
    });
  }
  sc.run(1500);
  CHECK_EQ(probe.count(EventCode::ExitSuccess, "onchange"), 3);
}

static void testDecodeEdges() {
  // weak coercions (config/decode/decode.go ToStrings/WeaklyTypedInput)
  std::vector<std::string> out;
  std::string err;
  CHECK(decode::toStrings(parseJson5("[1, \"a\", true, 2.5]"), &out));
  CHECK_EQ(out.size(), (size_t)4);
  CHECK_EQ(out[0], std::string("1"));
  CHECK_EQ(out[2], std::string("1"));  // bool -> "1"
  CHECK(decode::toStrings(parseJson5("\"solo\""), &out) && out.size() == 1);
  CHECK(decode::toStrings(parseJson5("null"), &out) && out.empty());

  int n = 0;
  CHECK(decode::toInt(parseJson5("\"42\""), &n) && n == 42);
  CHECK(decode::toInt(parseJson5("3.9"), &n) && n == 3);  // truncation
  CHECK(!decode::toInt(parseJson5("\"4x\""), &n));
  bool b = false;
  CHECK(decode::toBool(parseJson5("\"true\""), &b) && b);
  CHECK(!decode::toBool(parseJson5("\"nope\""), &b));

  // template nesting bomb rejects instead of overflowing
  std::string bomb;
  for (int i = 0; i < 5000; i++) bomb += "{{ if 1 }}";
  bool threw = false;
  try {
    renderTemplate(bomb);
  } catch (const std::exception&) {
    threw = true;
  }
  CHECK(threw);
}

static void testDebugRing() {
  // the bus's 10-slot circular debug buffer (events/bus.go:24-54):
  // keeps the last 10 events, drains oldest-first
  Loop loop;
  auto bus = std::make_shared<Bus>(loop);
  for (int i = 0; i < 13; i++)
    bus->publish(Event{EventCode::Metric, "ev-" + std::to_string(i)});
  auto drained = bus->debugEvents();
  CHECK_EQ(drained.size(), (size_t)10);
  CHECK_EQ(drained.front().source, std::string("ev-3"));
  CHECK_EQ(drained.back().source, std::string("ev-12"));
  // drained: a second read is empty
  CHECK_EQ(bus->debugEvents().size(), (size_t)0);
}

static void testMaintenanceMatrix() {
  // maintenance-mode matrix (jobs/jobs_test.go:208-269): health events
  // are suppressed during maintenance and resume after
  JobScenario sc;
  Probe probe;
  sc.bus->subscribe(&probe);
  sc.addJob(R"({name: "svc", exec: "sleep 10",
                health: {exec: "true", interval: "200ms", ttl: 5}})");
  sc.loop.addTimeout(std::chrono::milliseconds(600), [&sc] {
    sc.bus->publish(GlobalEnterMaintenance);
  });
  sc.loop.addTimeout(std::chrono::milliseconds(1400), [&sc] {
    sc.bus->publish(GlobalExitMaintenance);
  });
  int healthyAtEnter = -1, healthyAtExit = -1;
  sc.loop.addTimeout(std::chrono::milliseconds(1300), [&] {
    healthyAtEnter = probe.count(EventCode::StatusHealthy, "svc");
  });
  sc.loop.addTimeout(std::chrono::milliseconds(1450), [&] {
    healthyAtExit = probe.count(EventCode::StatusHealthy, "svc");
  });
  sc.run(2200);
  // some health events before maintenance
  CHECK(healthyAtEnter >= 1);
  // no healthy events published during the maintenance window
  CHECK_EQ(healthyAtExit, healthyAtEnter);
  // and they resume after exitMaintenance
  CHECK(probe.count(EventCode::StatusHealthy, "svc") > healthyAtExit);
}

static void testSignalJobMatrix() {
  // signal-driven jobs fire on their own signal only and repeatedly
  // (jobs/jobs.go:351-357)
  JobScenario sc;
  Probe probe;
  sc.bus->subscribe(&probe);
  sc.addJob(R"({name: "onhup", exec: "true", when: {source: "SIGHUP"}})");
  sc.addJob(R"({name: "onusr2", exec: "true", when: {source: "SIGUSR2"}})");
  for (int i = 1; i <= 2; i++) {
    sc.loop.addTimeout(std::chrono::milliseconds(200 * i),
                       [&sc] { sc.bus->publishSignal("SIGHUP"); });
  }
  sc.loop.addTimeout(std::chrono::milliseconds(500),
                     [&sc] { sc.bus->publishSignal("SIGUSR2"); });
  sc.run(1300);
  CHECK_EQ(probe.count(EventCode::ExitSuccess, "onhup"), 2);
  CHECK_EQ(probe.count(EventCode::ExitSuccess, "onusr2"), 1);
}

static void testStartTimeoutQuitsJob() {
  // when.timeout fires -> TimerExpired{job} published + job quits
  // (jobs/jobs.go:259-264)
  JobScenario sc;
  Probe probe;
  sc.bus->subscribe(&probe);
  sc.addJob(R"({name: "patient", exec: "true",
                when: {source: "ghost", once: "exitSuccess",
                       timeout: "300ms"}})");
  sc.run(1500);
  CHECK_EQ(probe.count(EventCode::ExitSuccess, "patient"), 0);
  CHECK_EQ(probe.count(EventCode::TimerExpired, "patient"), 1);
  CHECK_EQ(probe.count(EventCode::Stopped, "patient"), 1);
}

// ---- HTTP client framing + keep-alive pool (http.cpp) ----
//
// A scripted in-process server: accepts connections on a TCP port and
// answers each request with the next canned raw response, recording how
// many CONNECTIONS it saw — which is what proves (or disproves) reuse.
namespace {

struct ScriptedServer {
  int listenFd = -1;
  int port = 0;
  std::vector<std::string> responses;
  std::atomic<int> connections{0};
  std::atomic<int> requests{0};
  std::atomic<bool> stop{false};
  std::thread thread;

  void start() {
    listenFd = socket(AF_INET, SOCK_STREAM, 0);
    int one = 1;
    setsockopt(listenFd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    struct sockaddr_in addr;
    memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
    CHECK(bind(listenFd, (struct sockaddr*)&addr, sizeof(addr)) == 0);
    socklen_t len = sizeof(addr);
    getsockname(listenFd, (struct sockaddr*)&addr, &len);
    port = ntohs(addr.sin_port);
    CHECK(listen(listenFd, 8) == 0);
    thread = std::thread([this] { serve(); });
  }

  void serve() {
    while (!stop) {
      struct pollfd p{listenFd, POLLIN, 0};
      if (poll(&p, 1, 100) <= 0) continue;
      int fd = accept(listenFd, nullptr, nullptr);
      if (fd < 0) continue;
      struct timeval tv{0, 200000};  // shutdown() must not hang in recv
      setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
      connections++;
      // serve requests on this connection until close or script end
      while (!stop) {
        char buf[8192];
        std::string req;
        bool gotReq = false;
        bool peerGone = false;
        while (req.find("\r\n\r\n") == std::string::npos) {
          ssize_t n = recv(fd, buf, sizeof(buf), 0);
          if (n == 0) {
            peerGone = true;
            break;
          }
          if (n < 0) {
            if ((errno == EAGAIN || errno == EWOULDBLOCK) && !stop) continue;
            peerGone = true;
            break;
          }
          req.append(buf, n);
          gotReq = true;
        }
        if (peerGone) break;
        if (!gotReq || req.find("\r\n\r\n") == std::string::npos) break;
        int idx = requests++;
        if (idx >= (int)responses.size()) break;
        const std::string& resp = responses[idx];
        size_t off = 0;
        while (off < resp.size()) {
          ssize_t n = send(fd, resp.data() + off, resp.size() - off,
                           MSG_NOSIGNAL);
          if (n <= 0) break;
          off += n;
        }
        if (resp.find("Connection: close") != std::string::npos) break;
      }
      close(fd);
    }
  }

  void shutdown() {
    stop = true;
    if (thread.joinable()) thread.join();
    if (listenFd >= 0) close(listenFd);
  }
};

}  // namespace

static void testHttpClientFraming() {
  using cpilot::http::request;
  // content-length framing + keep-alive reuse: three requests over ONE
  // connection (the pool holds it between calls)
  {
    ScriptedServer srv;
    std::string ok =
        "HTTP/1.1 200 OK\r\nContent-Type: text/plain\r\n"
        "Content-Length: 5\r\n\r\nhello";
    srv.responses = {ok, ok, ok};
    srv.start();
    std::string target = "127.0.0.1:" + std::to_string(srv.port);
    for (int i = 0; i < 3; i++) {
      auto r = request(target, "GET", "/x", "");
      CHECK(r.ok);
      CHECK_EQ(r.status, 200);
      CHECK_EQ(r.body, std::string("hello"));
    }
    CHECK_EQ(srv.connections.load(), 1);  // reused, not re-connected
    CHECK_EQ(srv.requests.load(), 3);
    srv.shutdown();
  }
  // chunked transfer-encoding is decoded and the connection still reused
  {
    ScriptedServer srv;
    std::string chunked =
        "HTTP/1.1 200 OK\r\nTransfer-Encoding: chunked\r\n\r\n"
        "5\r\nhello\r\n6\r\n world\r\n0\r\n\r\n";
    srv.responses = {chunked, chunked};
    srv.start();
    std::string target = "127.0.0.1:" + std::to_string(srv.port);
    auto r1 = request(target, "GET", "/c", "");
    CHECK(r1.ok);
    CHECK_EQ(r1.body, std::string("hello world"));
    auto r2 = request(target, "GET", "/c", "");
    CHECK(r2.ok);
    CHECK_EQ(r2.body, std::string("hello world"));
    CHECK_EQ(srv.connections.load(), 1);
    srv.shutdown();
  }
  // Connection: close is honored — the next request reconnects
  {
    ScriptedServer srv;
    std::string closing =
        "HTTP/1.1 200 OK\r\nContent-Length: 2\r\n"
        "Connection: close\r\n\r\nok";
    srv.responses = {closing, closing};
    srv.start();
    std::string target = "127.0.0.1:" + std::to_string(srv.port);
    CHECK(request(target, "GET", "/a", "").ok);
    CHECK(request(target, "GET", "/a", "").ok);
    CHECK_EQ(srv.connections.load(), 2);
    srv.shutdown();
  }
  // truncated framed body (server closes mid-body) is an error, not a
  // short success
  {
    ScriptedServer srv;
    srv.responses = {
        "HTTP/1.1 200 OK\r\nContent-Length: 10\r\n"
        "Connection: close\r\n\r\nhalf"};
    srv.start();
    std::string target = "127.0.0.1:" + std::to_string(srv.port);
    auto r = request(target, "GET", "/t", "");
    CHECK(!r.ok);
    CHECK(r.error.find("truncated") != std::string::npos);
    srv.shutdown();
  }
  // stale pooled connection: server restarts between calls; the client
  // retries transparently on a fresh connection
  {
    std::string ok =
        "HTTP/1.1 200 OK\r\nContent-Length: 1\r\n\r\nA";
    int port;
    {
      ScriptedServer srv;
      srv.responses = {ok};
      srv.start();
      port = srv.port;
      std::string target = "127.0.0.1:" + std::to_string(port);
      CHECK(request(target, "GET", "/s", "").ok);  // pools the conn
      srv.shutdown();                              // kills it server-side
    }
    ScriptedServer srv2;
    srv2.responses = {ok};
    srv2.listenFd = socket(AF_INET, SOCK_STREAM, 0);
    int one = 1;
    setsockopt(srv2.listenFd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    struct sockaddr_in addr;
    memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
    addr.sin_port = htons(port);
    CHECK(bind(srv2.listenFd, (struct sockaddr*)&addr, sizeof(addr)) == 0);
    CHECK(listen(srv2.listenFd, 8) == 0);
    srv2.port = port;
    srv2.thread = std::thread([&srv2] { srv2.serve(); });
    std::string target = "127.0.0.1:" + std::to_string(port);
    auto r = request(target, "GET", "/s", "");
    CHECK(r.ok);
    CHECK_EQ(r.body, std::string("A"));
    srv2.shutdown();
  }
}

int main() {
  testJson5();
  testDurations();
  testTemplate();
  testEvents();
  testParseArgs();
  testIps();
  testConfig();
  testJobLifecycleSequence();
  testJobRestartCounting();
  testJobFailedExit();
  testPeriodicJobRuns();
  testPreStopJobOnShutdown();
  testWhenEachFiresRepeatedly();
  testDecodeEdges();
  testDebugRing();
  testMaintenanceMatrix();
  testSignalJobMatrix();
  testStartTimeoutQuitsJob();
  testHttpClientFraming();
  if (failures) {
    fprintf(stderr, "%d failures\n", failures);
    return 1;
  }
  printf("all unit tests passed\n");
  return 0;
}
