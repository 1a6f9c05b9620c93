#!/bin/bash
# GPU profiling pass (run on the MI355X box via gpurun):
#  1. sampler calibration trace (idle / load / settle + acc-counter rate)
#  2. rocprofv3 kernel trace + stats of the gfx950 busy probe
#  3. sampler + exporter overhead measurements (poll latency, scrape latency,
#     probe slowdown with the sampler polling at 10 Hz)
# Outputs land in gpurun_out/ for merge-back; curated summaries are committed
# under profiles/.
set -x
REPO=$(pwd)
mkdir -p gpurun_out/prof

python scripts/calibrate_sampler.py > gpurun_out/calibration.log 2>&1

# rocprofv3 wants TMPDIR=/tmp and a /tmp cwd
cat > /tmp/probe_run.py <<'EOF'
import sys, time
sys.path.insert(0, "/root/repo")
from gpu_pruner_amd import probe
probe.run_for_ms(0, 2000)
print("probe done")
EOF
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof" -- \
    python /tmp/probe_run.py > "$REPO/gpurun_out/rocprof_probe.log" 2>&1
cd "$REPO"

python - > gpurun_out/overhead.log 2>&1 <<'EOF'
import statistics, subprocess, sys, time, urllib.request
sys.path.insert(0, ".")
from gpu_pruner_amd import _gpumon, probe

# 1) sampler poll latency (the per-GPU cost the DaemonSet pays at 1 Hz)
s = _gpumon.Sampler(poll_interval_ms=1000)
s.init()
lat = []
for _ in range(50):
    t0 = time.perf_counter(); s.poll_once(); lat.append(time.perf_counter() - t0)
print(f"sampler poll_once: p50={statistics.median(lat)*1000:.2f} ms "
      f"max={max(lat)*1000:.2f} ms (per node-poll, all GPUs)")

# 2) exporter scrape latency
proc = subprocess.Popen(["./bin/mi355-exporter", "-p", "19402", "-b", "127.0.0.1",
                         "-i", "500"], stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
time.sleep(2)
try:
    lat = []
    for _ in range(100):
        t0 = time.perf_counter()
        urllib.request.urlopen("http://127.0.0.1:19402/metrics", timeout=5).read()
        lat.append(time.perf_counter() - t0)
    print(f"exporter /metrics scrape: p50={statistics.median(lat)*1000:.2f} ms "
          f"max={max(lat)*1000:.2f} ms")
finally:
    proc.terminate(); proc.wait()

# 3) probe-kernel wall time with vs without 10 Hz sampling (workload
#    perturbation check — SURVEY.md §7 "Low-overhead sampling")
def timed_probe(ms):
    t0 = time.perf_counter(); probe.run_for_ms(0, ms); return time.perf_counter() - t0

base = min(timed_probe(1500) for _ in range(3))
s2 = _gpumon.Sampler(poll_interval_ms=100); s2.init(); s2.start()
with_sampling = min(timed_probe(1500) for _ in range(3))
s2.stop()
print(f"probe 1.5s wall: alone={base:.3f}s with-10Hz-sampling={with_sampling:.3f}s "
      f"overhead={(with_sampling/base-1)*100:.2f}%")
s.stop()
EOF

ls -la gpurun_out/prof/ >> gpurun_out/rocprof_probe.log 2>&1
find gpurun_out/prof -name '*stats*' -o -name '*.csv' | head -20
