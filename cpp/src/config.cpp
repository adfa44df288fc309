#include "cpilot/config.hpp"

#include <cstdio>
#include <fstream>
#include <sstream>

#include "cpilot/decode.hpp"
#include "cpilot/tmpl.hpp"

namespace cpilot {

namespace {

bool readFile(const std::string& path, std::string* out, std::string* err) {
  if (path.empty()) {
    *err = "-config flag is required";
    return false;
  }
  std::ifstream f(path, std::ios::binary);
  if (!f) {
    *err = "could not read config file: open " + path +
           ": no such file or directory";
    return false;
  }
  std::stringstream ss;
  ss << f.rdbuf();
  *out = ss.str();
  return true;
}

bool renderString(const std::string& text, std::string* out,
                  std::string* err) {
  try {
    *out = renderTemplate(text);
    return true;
  } catch (const std::exception& e) {
    *err = std::string("could not apply template to config: ") + e.what();
    return false;
  }
}

}  // namespace

std::unique_ptr<AppConfig> newConfig(const std::string& rendered,
                                     std::string* err) {
  Json doc;
  try {
    doc = parseJson5(rendered);
  } catch (const JsonParseError& e) {
    *err = formatParseError(rendered, e);
    return nullptr;
  } catch (const std::exception& e) {
    *err = std::string("could not parse configuration: ") + e.what();
    return nullptr;
  }
  if (!doc.isObject()) {
    *err = "could not parse configuration: config must be a JSON5 object";
    return nullptr;
  }
  // unknown top-level key check (config/config.go:254-267)
  std::string keyErr;
  if (!decode::checkKeys(doc,
                         {"consul", "logging", "stopTimeout", "jobs",
                          "watches", "telemetry", "control"},
                         &keyErr)) {
    std::string unknown = keyErr.substr(std::string("invalid keys: ").size());
    *err = "unknown config keys: [" + unknown + "]";
    return nullptr;
  }

  auto cfg = std::make_unique<AppConfig>();

  // logging
  if (const Json* v = doc.find("logging")) {
    if (!v->isNull()) {
      if (!v->isObject() ||
          !decode::checkKeys(*v, {"level", "format", "output"}, err))
        return nullptr;
      if (const Json* f = v->find("level"))
        decode::toString(*f, &cfg->logConfig.level);
      if (const Json* f = v->find("format"))
        decode::toString(*f, &cfg->logConfig.format);
      if (const Json* f = v->find("output"))
        decode::toString(*f, &cfg->logConfig.output);
    }
  }

  // stopTimeout (0 -> default 5s)
  if (const Json* v = doc.find("stopTimeout")) {
    int st = 0;
    if (!v->isNull() && !decode::toInt(*v, &st)) {
      *err = "could not parse configuration: bad stopTimeout";
      return nullptr;
    }
    cfg->stopTimeout = (st == 0) ? 5 : st;
  }

  // discovery backend (required, discovery/consul.go:33-58)
  cfg->discovery = ConsulBackend::create(doc.find("consul"), err);
  if (!cfg->discovery) return nullptr;

  // control
  if (!newControlConfig(doc.find("control"), &cfg->control, err)) {
    *err = "unable to parse control: " + *err;
    return nullptr;
  }

  // jobs
  const Json* jobsRaw = doc.find("jobs");
  if (jobsRaw) {
    std::string jerr;
    if (!newJobConfigs(*jobsRaw, cfg->discovery.get(), &cfg->jobs, &jerr)) {
      *err = "unable to parse jobs: " + jerr;
      return nullptr;
    }
  }

  // watches
  if (const Json* v = doc.find("watches")) {
    std::string werr;
    if (!newWatchConfigs(*v, &cfg->watches, &werr)) {
      *err = "unable to parse watches: " + werr;
      return nullptr;
    }
  }

  // telemetry (+ synthetic job appended, config/config.go:176-179)
  if (!newTelemetryConfig(doc.find("telemetry"), cfg->discovery.get(),
                          &cfg->telemetry, err))
    return nullptr;
  if (cfg->telemetry) cfg->jobs.push_back(cfg->telemetry->jobConfig);

  return cfg;
}

std::unique_ptr<AppConfig> loadConfig(const std::string& path,
                                      std::string* err) {
  std::string data;
  if (!readFile(path, &data, err)) return nullptr;
  std::string rendered;
  if (!renderString(data, &rendered, err)) return nullptr;
  return newConfig(rendered, err);
}

bool renderConfigFile(const std::string& configPath,
                      const std::string& outPath, std::string* err) {
  std::string data;
  if (!readFile(configPath, &data, err)) return false;
  std::string rendered;
  if (!renderString(data, &rendered, err)) return false;
  if (outPath == "-" || outPath.empty()) {
    fwrite(rendered.data(), 1, rendered.size(), stdout);
  } else {
    std::ofstream f(outPath, std::ios::binary);
    if (!f) {
      *err = "could not write config file: " + outPath;
      return false;
    }
    f << rendered;
  }
  return true;
}

}  // namespace cpilot
