#include "log.hpp"

#include <unistd.h>

#include <algorithm>
#include <cstdio>
#include <cstring>

#include "json.hpp"
#include "strutil.hpp"

namespace logx {

namespace {

struct State {
  std::atomic<int> level{static_cast<int>(Level::Info)};  // min over all directives
  std::atomic<int> default_level{static_cast<int>(Level::Info)};
  std::atomic<int> format{static_cast<int>(Format::Default)};
  std::atomic<bool> color{false};
  std::mutex write_mu;
  std::mutex counter_mu;
  std::map<std::string, int64_t> counters;
  // Per-target directives (path → level), longest path first. Written only
  // by init() (process startup, before worker threads); read lock-free.
  std::vector<std::pair<std::string, int>> directives;
};

State& state() {
  static State s;
  return s;
}

const char* level_name(Level l) {
  switch (l) {
    case Level::Trace: return "TRACE";
    case Level::Debug: return "DEBUG";
    case Level::Info: return "INFO";
    case Level::Warn: return "WARN";
    case Level::Error: return "ERROR";
    default: return "OFF";
  }
}

const char* level_name_lower(Level l) {
  switch (l) {
    case Level::Trace: return "trace";
    case Level::Debug: return "debug";
    case Level::Info: return "info";
    case Level::Warn: return "warn";
    case Level::Error: return "error";
    default: return "off";
  }
}

Level parse_level(const std::string& s, Level dflt) {
  std::string v = strutil::lower(strutil::trim(s));
  if (v == "trace") return Level::Trace;
  if (v == "debug") return Level::Debug;
  if (v == "info") return Level::Info;
  if (v == "warn" || v == "warning") return Level::Warn;
  if (v == "error") return Level::Error;
  if (v == "off" || v == "none") return Level::Off;
  return dflt;
}

}  // namespace

void init(Format format, const char* env_filter) {
  auto& s = state();
  s.format.store(static_cast<int>(format));
  Level dflt = Level::Info;
  std::vector<std::pair<std::string, int>> directives;
  const char* env = env_filter;
  if (!env || !*env) env = std::getenv("GPU_PRUNER_LOG");
  if (!env || !*env) env = std::getenv("RUST_LOG");  // drop-in with the reference's env knob
  if (env && *env) {
    // env_logger / tracing-subscriber EnvFilter directive list:
    //   "info,pruner::engine=debug,hyper=error"
    // bare level → default; path=level → that path and its ::-descendants.
    for (const std::string& raw : strutil::split(env, ',')) {
      std::string item = strutil::trim(raw);
      if (item.empty()) continue;
      size_t eq = item.find('=');
      if (eq == std::string::npos) {
        dflt = parse_level(item, dflt);
      } else {
        std::string path = strutil::trim(item.substr(0, eq));
        Level lv = parse_level(item.substr(eq + 1), Level::Info);
        if (!path.empty()) directives.emplace_back(path, static_cast<int>(lv));
      }
    }
  }
  // longest path first so the most specific directive wins
  std::sort(directives.begin(), directives.end(),
            [](const auto& a, const auto& b) { return a.first.size() > b.first.size(); });
  int min_lvl = static_cast<int>(dflt);
  for (const auto& [_, lv] : directives) min_lvl = std::min(min_lvl, lv);
  s.directives = std::move(directives);
  s.default_level.store(static_cast<int>(dflt));
  s.level.store(min_lvl);
  s.color.store(isatty(2) != 0);
}

Level level() { return static_cast<Level>(state().level.load(std::memory_order_relaxed)); }

bool enabled(Level lvl) { return static_cast<int>(lvl) >= state().level.load(std::memory_order_relaxed); }

bool enabled_for(Level lvl, const std::string& target) {
  auto& s = state();
  for (const auto& [path, dir_lvl] : s.directives) {
    // prefix match on module-path boundaries: "pruner" matches
    // "pruner::engine" but not "prunerx"
    if (target.size() >= path.size() && target.compare(0, path.size(), path) == 0 &&
        (target.size() == path.size() ||
         (target.size() >= path.size() + 2 && target[path.size()] == ':' &&
          target[path.size() + 1] == ':')))
      return static_cast<int>(lvl) >= dir_lvl;
  }
  return static_cast<int>(lvl) >= s.default_level.load(std::memory_order_relaxed);
}

void emit(Level lvl, const std::string& target, const std::string& msg) {
  emit_kv(lvl, target, msg, {});
}

void emit_kv(Level lvl, const std::string& target, const std::string& msg,
             const std::vector<std::pair<std::string, std::string>>& fields) {
  auto& s = state();
  if (!enabled_for(lvl, target)) return;
  Format fmt = static_cast<Format>(s.format.load(std::memory_order_relaxed));
  std::string line;
  std::string ts = strutil::rfc3339_micro_now();
  switch (fmt) {
    case Format::Json: {
      jsn::Value v = jsn::Value::object();
      v["timestamp"] = ts;
      v["level"] = level_name(lvl);
      v["target"] = target;
      jsn::Value f = jsn::Value::object();
      f["message"] = msg;
      for (const auto& [k, val] : fields) f[k] = val;
      v["fields"] = f;
      line = v.dump();
      break;
    }
    case Format::Pretty: {
      bool color = s.color.load(std::memory_order_relaxed);
      const char* c0 = "";
      const char* c1 = "";
      if (color) {
        switch (lvl) {
          case Level::Error: c0 = "\x1b[31m"; break;
          case Level::Warn: c0 = "\x1b[33m"; break;
          case Level::Info: c0 = "\x1b[32m"; break;
          default: c0 = "\x1b[36m"; break;
        }
        c1 = "\x1b[0m";
      }
      line = "  " + ts + " " + c0 + level_name(lvl) + c1 + " " + target + ":\n    " + msg;
      for (const auto& [k, val] : fields) line += "\n    " + k + ": " + val;
      break;
    }
    case Format::Default:
    default: {
      line = ts + " " + level_name(lvl) + " " + target + ": " + msg;
      for (const auto& [k, val] : fields) line += " " + k + "=" + val;
      break;
    }
  }
  line += '\n';
  std::lock_guard<std::mutex> lock(s.write_mu);
  std::fwrite(line.data(), 1, line.size(), stderr);
  std::fflush(stderr);
  (void)level_name_lower;
}

void counter_add(const std::string& name, int64_t delta) {
  auto& s = state();
  std::lock_guard<std::mutex> lock(s.counter_mu);
  s.counters[name] += delta;
}

void gauge_set(const std::string& name, int64_t value) {
  auto& s = state();
  std::lock_guard<std::mutex> lock(s.counter_mu);
  s.counters[name] = value;
}

int64_t counter_get(const std::string& name) {
  auto& s = state();
  std::lock_guard<std::mutex> lock(s.counter_mu);
  auto it = s.counters.find(name);
  return it == s.counters.end() ? 0 : it->second;
}

std::map<std::string, int64_t> counters_snapshot() {
  auto& s = state();
  std::lock_guard<std::mutex> lock(s.counter_mu);
  return s.counters;
}

void counters_reset_for_test() {
  auto& s = state();
  std::lock_guard<std::mutex> lock(s.counter_mu);
  s.counters.clear();
}

}  // namespace logx
