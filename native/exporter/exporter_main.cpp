// exporter_main.cpp — mi355-exporter binary.
//
// Per-node DaemonSet companion of the gpu-pruner daemon: the first-party
// ROCm/gfx950 metrics source replacing the reference's external
// dcgm-exporter dependency (SURVEY.md §7 target architecture (a)).
// Serves Prometheus text exposition on --port (default 9400, dcgm-exporter's
// port) with DCGM-shaped series, pod attribution from the KFD process
// registry, and a /healthz endpoint.
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <csignal>
#include <cstdio>
#include <thread>

#include "../common/http_server.hpp"
#include "../common/log.hpp"
#include "../common/strutil.hpp"
#include "attrib.hpp"
#include "registry.hpp"
#include "sampler.hpp"

namespace {

std::atomic<bool> g_stop{false};

void on_signal(int) { g_stop.store(true); }

const char* HELP = R"(mi355-exporter — first-party ROCm GPU metrics exporter for MI355X nodes

Publishes per-GPU activity/power/memory series in Prometheus text format with
DCGM-compatible names and labels, attributing GPUs to pods via the amdgpu KFD
process registry + cgroups.

USAGE: mi355-exporter [OPTIONS]

OPTIONS:
  -p, --port <PORT>        listen port [default: 9400]
  -b, --bind <ADDR>        bind address [default: 0.0.0.0]
  -i, --interval <MS>      activity poll interval [default: 1000]
      --idle-epsilon <R>   windowed activity ratios below R report as exactly
                           0 (firmware housekeeping noise floor) [default: 0.005]
      --activity-window <S> sliding window (seconds) the activity ratio is
                           computed over; scrapes are idempotent — concurrent
                           scrapers all observe the same window [default: 30]
      --node-type <STR>    value for the node_type const label
      --hostname <STR>     override Hostname label (default: gethostname)
  -l, --log-format <FMT>   default | json | pretty
  -h, --help               this help
)";

}  // namespace


#include <malloc.h>

namespace {
// The 256-worker I/O pool scatters allocations across glibc's per-thread
// malloc arenas; each arena retains its high-water mark, growing RSS toward
// N_arenas x peak (measured: ~48 MB flat with 2 arenas vs ~170 MB and
// climbing with the default). Two arenas are plenty for an I/O-bound daemon.
void cap_malloc_arenas() {
#ifdef M_ARENA_MAX
  if (!std::getenv("MALLOC_ARENA_MAX")) mallopt(M_ARENA_MAX, 2);
#endif
}
}  // namespace

int main(int argc, char** argv) {
  cap_malloc_arenas();
  uint16_t port = 9400;
  std::string bind_addr = "0.0.0.0";
  int interval_ms = 1000;
  double idle_epsilon = 0.005;
  double activity_window_s = 30.0;
  std::string node_type;
  std::string hostname;
  logx::Format fmt = logx::Format::Default;

  for (int i = 1; i < argc; i++) {
    std::string a = argv[i];
    auto next = [&]() -> std::string {
      if (i + 1 >= argc) {
        std::fprintf(stderr, "missing value for %s\n", a.c_str());
        std::exit(2);
      }
      return argv[++i];
    };
    if (a == "-p" || a == "--port") port = static_cast<uint16_t>(std::stoi(next()));
    else if (a == "-b" || a == "--bind") bind_addr = next();
    else if (a == "-i" || a == "--interval") interval_ms = std::stoi(next());
    else if (a == "--idle-epsilon") idle_epsilon = std::stod(next());
    else if (a == "--activity-window") activity_window_s = std::stod(next());
    else if (a == "--node-type") node_type = next();
    else if (a == "--hostname") hostname = next();
    else if (a == "-l" || a == "--log-format") {
      std::string v = next();
      fmt = v == "json" ? logx::Format::Json
            : v == "pretty" ? logx::Format::Pretty
                            : logx::Format::Default;
    } else if (a == "--version" || a == "-V") {
      std::puts("mi355-exporter 0.1.0 (MI355X-native)");
      return 0;
    } else if (a == "-h" || a == "--help") {
      std::fputs(HELP, stdout);
      return 0;
    } else {
      std::fprintf(stderr, "unknown flag: %s\n%s", a.c_str(), HELP);
      return 2;
    }
  }

  logx::init(fmt);
  if (hostname.empty()) {
    char buf[256] = {0};
    if (const char* env = std::getenv("NODE_NAME"); env && *env) hostname = env;
    else if (::gethostname(buf, sizeof buf - 1) == 0) hostname = buf;
  }

  exporter::Sampler sampler(interval_ms, idle_epsilon, activity_window_s);
  try {
    sampler.init();
  } catch (const std::exception& e) {
    LOGE("exporter", std::string("sampler init failed: ") + e.what());
    return 1;
  }
  sampler.start();
  exporter::Attributor attributor;

  exporter::RenderOptions opts;
  opts.hostname = hostname;
  if (!node_type.empty()) opts.const_labels.emplace_back("node_type", node_type);

  http::Server server(bind_addr, port, [&](const http::ServerRequest& req) {
    http::ServerResponse resp;
    if (req.path == "/metrics") {
      // Idempotent scrape: snapshot() is read-only (fixed sliding window),
      // so concurrent scrapers (HA Prometheus pairs, debug curls) cannot
      // truncate the activity window another scraper observes.
      auto samples = sampler.snapshot();
      auto attribs = attributor.resolve_full(samples);
      resp.body = exporter::render_metrics(samples, attribs, opts);
      resp.content_type = "text/plain; version=0.0.4; charset=utf-8";
    } else if (req.path == "/healthz") {
      // liveness reflects the sampler: if EVERY device's activity reads are
      // failing, the exporter is not doing its job — let the kubelet
      // restart it (per-device health is also exported as
      // mi355_sampler_healthy for alerting)
      auto samples = sampler.snapshot();
      bool any_healthy = false;
      for (const auto& d : samples)
        if (d.healthy) any_healthy = true;
      if (any_healthy) {
        resp.body = "ok\n";
      } else {
        resp.status = 503;
        resp.body = "no device with working activity reads\n";
      }
    } else {
      resp.status = 404;
      resp.body = "not found; try /metrics\n";
    }
    return resp;
  });

  try {
    server.start();
  } catch (const std::exception& e) {
    LOGE("exporter", std::string("server start failed: ") + e.what());
    return 1;
  }
  LOGI("exporter", "mi355-exporter serving /metrics on " + bind_addr + ":" +
                       std::to_string(server.port()) + " (" +
                       std::to_string(sampler.device_count()) + " GPUs)");

  std::signal(SIGINT, on_signal);
  std::signal(SIGTERM, on_signal);
  while (!g_stop.load()) std::this_thread::sleep_for(std::chrono::milliseconds(200));

  LOGI("exporter", "shutting down");
  server.stop();
  sampler.stop();
  return 0;
}
