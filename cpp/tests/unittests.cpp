// Assert-style unit tests for the pure components (config pipeline,
// template engine, durations, event model, IP specs, arg parsing).
// Test coverage mirrors the reference's package unit tests
// (config/*_test.go, commands/commands_test.go, events/events_test.go).
// Run via `bin/cpilot_unittests`; exits non-zero on first failure.
#include <arpa/inet.h>
#include <netinet/in.h>

#include <cassert>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>

#include "cpilot/command.hpp"
#include "cpilot/config.hpp"
#include "cpilot/decode.hpp"
#include "cpilot/events.hpp"
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

  // unknown job field rejected (decode.go:15-17 ErrorUnused)
  CHECK(newConfig(R"({"consul": "x:8500",
    jobs: [{name: "j", exec: "x", bogusField: true}]})", &err) == nullptr);
  CHECK(err.find("invalid keys") != std::string::npos);
}

int main() {
  testJson5();
  testDurations();
  testTemplate();
  testEvents();
  testParseArgs();
  testIps();
  testConfig();
  if (failures) {
    fprintf(stderr, "%d failures\n", failures);
    return 1;
  }
  printf("all unit tests passed\n");
  return 0;
}
