// main.cpp — gpu-pruner binary entry point (MI355X-native culler daemon).
//
// Shape mirrors the reference's main(): parse CLI, init
// logging + OTLP, run the producer/consumer daemon.
#include <cstdio>

#include "../common/log.hpp"
#include "config.hpp"
#include "daemon.hpp"
#include "otlp.hpp"


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
  std::vector<std::string> args(argv + 1, argv + argc);
  if (args.size() == 1 && (args[0] == "--version" || args[0] == "-V")) {
    std::puts("gpu-pruner 0.1.0 (MI355X-native)");
    return 0;
  }
  pruner::CliResult cli = pruner::parse_cli(args);
  if (cli.show_help) {
    std::fputs(pruner::cli_help().c_str(), stdout);
    return 0;
  }
  if (cli.error) {
    std::fprintf(stderr, "error: %s\n\n%s", cli.error->c_str(), pruner::cli_help().c_str());
    return 2;
  }

  logx::Format fmt = logx::Format::Default;
  switch (cli.config.log_format) {
    case pruner::LogFormatOpt::Json: fmt = logx::Format::Json; break;
    case pruner::LogFormatOpt::Pretty: fmt = logx::Format::Pretty; break;
    default: break;
  }
  logx::init(fmt);
  otlp::init("gpu-pruner");

  int rc = pruner::run_daemon(cli.config);

  otlp::shutdown();
  return rc;
}
