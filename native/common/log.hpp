// log.hpp — structured logging + named counters for the MI355X-native gpu-pruner.
//
// Mirrors the observability surface of the reference daemon (SURVEY.md §5.5,
// reference gpu-pruner/src/main.rs:138-271): three output formats
// (default / pretty / json), an env-driven level filter, and named counters
// attached to log events (monotonic_counter.query_successes etc.) that the
// OTLP exporter drains periodically.
#pragma once

#include <atomic>
#include <cstdint>
#include <map>
#include <mutex>
#include <string>
#include <vector>

namespace logx {

enum class Level : int { Trace = 0, Debug = 1, Info = 2, Warn = 3, Error = 4, Off = 5 };
enum class Format : int { Default = 0, Pretty = 1, Json = 2 };

struct Config {
  Level level = Level::Info;
  Format format = Format::Default;
  bool color = false;  // pretty format only
};

// Initialize global logger. `env_filter` comes from GPU_PRUNER_LOG / RUST_LOG
// and supports env_logger-style comma lists of per-target directives —
// "info,pruner::engine=debug,hyper=error" — where a bare level sets the
// default and `path=level` applies to `path` and its `::`-descendants, the
// longest matching path winning (reference parity: tracing-subscriber
// EnvFilter, main.rs:157-173). Flag-level format selection mirrors
// --log-format.
void init(Format format, const char* env_filter = nullptr);
Level level();
bool enabled(Level lvl);  // fast path: true if ANY target could log at lvl
// Per-target decision honoring directives.
bool enabled_for(Level lvl, const std::string& target);

// Core emit. `target` is the module path shown in logs (e.g. "pruner::engine").
void emit(Level lvl, const std::string& target, const std::string& msg);

// Emit with structured fields (rendered as key=value / JSON members).
void emit_kv(Level lvl, const std::string& target, const std::string& msg,
             const std::vector<std::pair<std::string, std::string>>& fields);

// The message expression is only evaluated when the level is enabled — hot
// loops log per pod, and the string concatenation would otherwise dominate
// filtered-out levels.
#define LOGX_AT(lvl, target, msg)                                          \
  do {                                                                     \
    if (::logx::enabled(lvl) && ::logx::enabled_for(lvl, target))          \
      ::logx::emit(lvl, target, msg);                                      \
  } while (0)
#define LOGT(target, msg) LOGX_AT(::logx::Level::Trace, target, msg)
#define LOGD(target, msg) LOGX_AT(::logx::Level::Debug, target, msg)
#define LOGI(target, msg) LOGX_AT(::logx::Level::Info, target, msg)
#define LOGW(target, msg) LOGX_AT(::logx::Level::Warn, target, msg)
#define LOGE(target, msg) LOGX_AT(::logx::Level::Error, target, msg)

// ---- counters ---------------------------------------------------------------
// Named monotonic counters and gauges, mirroring the reference's
// tracing-field-derived OTEL instruments:
//   monotonic_counter.query_successes / query_failures / scale_successes /
//   scale_failures, counter.query_returned_candidates /
//   query_returned_shutdown_events  
void counter_add(const std::string& name, int64_t delta);
void gauge_set(const std::string& name, int64_t value);
int64_t counter_get(const std::string& name);
std::map<std::string, int64_t> counters_snapshot();
void counters_reset_for_test();

}  // namespace logx
