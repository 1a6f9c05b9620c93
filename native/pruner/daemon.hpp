// daemon.hpp — process orchestration: producer/consumer tasks, interval tick,
// consecutive-failure breaker.
//
// The reference runs two tokio tasks joined at exit (main.rs:284-372): a
// query task that ticks every --check-interval (daemon mode) or runs once,
// and a scale task draining a bounded channel. This is the same shape with
// two threads and qx::BoundedQueue; the breaker aborts the producer after
// more than --max-failures consecutive query failures (reference
// main.rs:136,310-321 pins >5).
#pragma once

#include "config.hpp"

namespace pruner {

// Runs the daemon until one-shot completion, breaker trip, or (daemon mode)
// forever. Returns the process exit code.
int run_daemon(const Config& cfg);

}  // namespace pruner
