// containerpilot entry point: PID-1 supervisor fork, flag parsing,
// one-shot subcommands over the control socket, or the App event loop.
// Parity: /root/reference/main.go, core/flags.go, subcommands/,
// client/client.go.
#include <unistd.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <map>
#include <string>
#include <vector>

#include "cpilot/app.hpp"
#include "cpilot/config.hpp"
#include "cpilot/http.hpp"
#include "cpilot/json.hpp"
#include "cpilot/log.hpp"
#include "cpilot/sup.hpp"
#include "cpilot/version.hpp"

using namespace cpilot;

namespace {

struct Flags {
  bool version = false;
  bool templateFlag = false;
  bool reload = false;
  bool ping = false;
  std::string config;
  std::string out;
  std::string maintenance;
  std::map<std::string, std::string> putMetrics;
  std::map<std::string, std::string> putEnv;
  // extensions beyond the reference CLI:
  std::string statsOut;   // -stats-out <file>: write bus stats JSON on exit
  int benchSeconds = 0;   // -bench-seconds N: shut down after N seconds
};

[[noreturn]] void usageExit() {
  fprintf(stderr,
          "Usage of containerpilot:\n"
          "  -config string\n"
          "        File path to JSON5 configuration file. Defaults to "
          "CONTAINERPILOT env var.\n"
          "  -maintenance string\n"
          "        Toggle maintenance mode ('enable' or 'disable')\n"
          "  -out string\n"
          "        File path for rendered config with '-template' "
          "(default stdout '-')\n"
          "  -ping\n        Check that the control socket is up.\n"
          "  -putenv value\n        Update environ: 'key=value'\n"
          "  -putmetric value\n        Update metrics: 'key=value'\n"
          "  -reload\n        Reload a running ContainerPilot process.\n"
          "  -template\n        Render template and quit.\n"
          "  -version\n        Show version identifier and quit.\n");
  exit(2);
}

bool parseKV(const std::string& val, std::map<std::string, std::string>* out) {
  size_t eq = val.find('=');
  if (eq == std::string::npos) {
    fprintf(stderr, "flag value '%s' was not in the format 'key=val'\n",
            val.c_str());
    return false;
  }
  (*out)[val.substr(0, eq)] = val.substr(eq + 1);
  return true;
}

bool parseFlags(int argc, char** argv, Flags* f) {
  for (int i = 1; i < argc; i++) {
    std::string arg = argv[i];
    // accept both -flag and --flag
    if (arg.size() > 1 && arg[0] == '-' && arg[1] == '-') arg = arg.substr(1);
    auto next = [&](const char* name) -> std::string {
      if (i + 1 >= argc) {
        fprintf(stderr, "flag needs an argument: %s\n", name);
        usageExit();
      }
      return argv[++i];
    };
    if (arg == "-version") f->version = true;
    else if (arg == "-template") f->templateFlag = true;
    else if (arg == "-reload") f->reload = true;
    else if (arg == "-ping") f->ping = true;
    else if (arg == "-config") f->config = next("-config");
    else if (arg == "-out") f->out = next("-out");
    else if (arg == "-maintenance") f->maintenance = next("-maintenance");
    else if (arg == "-putmetric") {
      if (!parseKV(next("-putmetric"), &f->putMetrics)) return false;
    } else if (arg == "-putenv") {
      if (!parseKV(next("-putenv"), &f->putEnv)) return false;
    } else if (arg == "-stats-out") f->statsOut = next("-stats-out");
    else if (arg == "-bench-seconds")
      f->benchSeconds = atoi(next("-bench-seconds").c_str());
    else if (arg == "-h" || arg == "-help") usageExit();
    else {
      fprintf(stderr, "flag provided but not defined: %s\n", arg.c_str());
      usageExit();
    }
  }
  return true;
}

// control-socket client (client/client.go:30-115): load the config to
// find the socket path, then POST/GET over it
bool initSocket(const std::string& configPath, std::string* socketPath,
                std::string* err) {
  auto cfg = loadConfig(configPath, err);
  if (!cfg) return false;
  *socketPath = cfg->control.socketPath;
  return true;
}

int controlPost(const std::string& configPath, const std::string& path,
                const std::string& body, const char* what) {
  std::string socketPath, err;
  if (!initSocket(configPath, &socketPath, &err)) {
    fprintf(stderr, "%s\n", err.c_str());
    return 1;
  }
  auto res = http::request("unix:" + socketPath, "POST", path, body);
  if (!res.ok || res.status != 200) {
    std::string why = res.ok ? "HTTP " + std::to_string(res.status) : res.error;
    fprintf(stderr, "%s: failed to run subcommand: %s\n", what, why.c_str());
    return 1;
  }
  return 0;
}

std::string mapToJson(const std::map<std::string, std::string>& m) {
  JsonObject obj;
  for (auto& kv : m) obj.emplace_back(kv.first, Json(kv.second));
  return Json(std::move(obj)).dump();
}

}  // namespace

int main(int argc, char** argv) {
  // PID-1: fork into the supervisor before anything else (main.go:25-28).
  // CPILOT_FORCE_SUP forces the same split without PID 1 (sup becomes a
  // child subreaper): for hosts whose init doesn't reap, and for
  // unprivileged tests of the reap path (no PID namespace needed).
  const char* forceSup = getenv("CPILOT_FORCE_SUP");
  if (getpid() == 1 ||
      (forceSup && forceSup[0] && !(forceSup[0] == '0' && !forceSup[1]))) {
    unsetenv("CPILOT_FORCE_SUP");  // the worker must not recurse
    int rc = supRun(argc, argv);
    if (rc >= 0) return rc;  // parent (supervisor) path
    // child falls through as the worker
  }

  Flags flags;
  if (!parseFlags(argc, argv, &flags)) return 2;

  if (flags.version) {
    printf("Version: %s\nGitHash: %s\n", kVersion, kGitHash);
    return 0;
  }
  if (flags.config.empty()) {
    const char* env = getenv("CONTAINERPILOT");
    if (env) flags.config = env;
  }
  if (flags.templateFlag) {
    std::string err;
    if (!renderConfigFile(flags.config, flags.out, &err)) {
      fprintf(stderr, "%s\n", err.c_str());
      return 1;
    }
    return 0;
  }
  if (flags.reload)
    return controlPost(flags.config, "/v3/reload", "", "-reload");
  if (!flags.maintenance.empty()) {
    if (flags.maintenance != "enable" && flags.maintenance != "disable") {
      fprintf(stderr,
              "-maintenance: expected 'enable' or 'disable', got '%s'\n",
              flags.maintenance.c_str());
      return 1;
    }
    return controlPost(flags.config, "/v3/maintenance/" + flags.maintenance,
                       "", "-maintenance");
  }
  if (!flags.putEnv.empty())
    return controlPost(flags.config, "/v3/environ", mapToJson(flags.putEnv),
                       "-putenv");
  if (!flags.putMetrics.empty())
    return controlPost(flags.config, "/v3/metric", mapToJson(flags.putMetrics),
                       "-putmetric");
  if (flags.ping) {
    std::string socketPath, err;
    if (!initSocket(flags.config, &socketPath, &err)) {
      fprintf(stderr, "%s\n", err.c_str());
      return 1;
    }
    auto res = http::request("unix:" + socketPath, "GET", "/v3/ping", "");
    if (!res.ok || res.status != 200) {
      std::string why =
          res.ok ? "HTTP " + std::to_string(res.status) : res.error;
      fprintf(stderr, "-ping: failed: %s\n", why.c_str());
      return 1;
    }
    printf("ok\n");
    return 0;
  }

  App app(flags.config, flags.statsOut, flags.benchSeconds);
  std::string err;
  if (!app.init(&err)) {
    // match the reference's fatal log on config errors (main.go:38-40)
    fprintf(stderr, "%s\n", err.c_str());
    return 1;
  }
  return app.run();
}
