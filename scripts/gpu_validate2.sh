#!/bin/bash
# Second GPU validation pass:
#  1. rocprofv3 PMC run (counters only, per pool rules): GRBM_GUI_ACTIVE on
#     the busy probe — the hardware counter DCGM_FI_PROF_GR_ENGINE_ACTIVE is
#     named after, tying the sampler's windowed ratio to silicon ground truth.
#  2. 3-minute daemon-mode soak: gpu-pruner --daemon-mode (1s ticks,
#     scale-down, OTLP on) against the synthetic backend; RSS sampled every
#     5 s to show steady-state memory.
#  3. Concurrent-scrape stress on the exporter (8 parallel scrapers, 15 s).
set -x
REPO=$(pwd)
mkdir -p gpurun_out/prof2

# ---- 1. PMC counters on the probe (no trace domains with --pmc) ----
cat > /tmp/probe_run2.py <<'EOF'
import sys
sys.path.insert(0, "/root/repo")
from gpu_pruner_amd import probe
probe.run_for_ms(0, 1500)
print("probe done")
EOF
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --pmc GRBM_GUI_ACTIVE GRBM_COUNT -d "$REPO/gpurun_out/prof2" \
    --output-format csv -- python /tmp/probe_run2.py \
    > "$REPO/gpurun_out/pmc_probe.log" 2>&1
cd "$REPO"
find gpurun_out/prof2 -name '*.csv' -exec head -5 {} \; >> gpurun_out/pmc_probe.log

# ---- 2. daemon soak with RSS tracking ----
python - > gpurun_out/daemon_soak.log 2>&1 <<'EOF'
import json, os, subprocess, sys, time
sys.path.insert(0, ".")
os.environ["PROMETHEUS_TOKEN"] = "t"
from gpu_pruner_amd import _pruner_core as core
from gpu_pruner_amd.fixtures import FakeOtlpCollector

backend = core.SyntheticBackend(n_pods=500)
backend.start()
col = FakeOtlpCollector().start()
env = dict(os.environ)
env["GPU_PRUNER_K8S_URL"] = backend.k8s_url
env["GPU_PRUNER_LOG"] = "warn"
env["OTEL_EXPORTER_OTLP_ENDPOINT"] = col.url
env["OTEL_METRIC_EXPORT_INTERVAL"] = "5000"
proc = subprocess.Popen(
    ["./bin/gpu-pruner", "--prometheus-url", backend.prom_url, "--daemon-mode",
     "--check-interval", "1", "--run-mode", "scale-down", "--metrics-port", "19495"],
    env=env, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
rss = []
try:
    t_end = time.time() + 180
    while time.time() < t_end:
        time.sleep(5)
        if proc.poll() is not None:
            print("FAIL daemon exited rc=", proc.returncode)
            break
        with open(f"/proc/{proc.pid}/status") as f:
            for line in f:
                if line.startswith("VmRSS"):
                    rss.append(int(line.split()[1]))
    import urllib.request
    metrics = urllib.request.urlopen("http://127.0.0.1:19495/metrics", timeout=3).read().decode()
    qs = [l for l in metrics.splitlines() if l.startswith("gpu_pruner_query_successes_total")]
    print("self-metrics:", qs)
finally:
    proc.terminate(); proc.wait()
    col.stop(); backend.stop()
print(f"RSS KiB over {len(rss)} samples: first={rss[0]} mid={rss[len(rss)//2]} last={rss[-1]}")
print("events posted:", backend.events_posted, "| spans:", len(col.span_names()),
      "| metric exports:", len(col.metrics))
# warmup allocations (pools, TLS contexts, arenas) stabilize in the first
# third; judge steady-state growth on the back half and absolute budget
half = rss[len(rss) // 2]
growth = (rss[-1] - half) / max(half, 1) * 100
ok = growth < 15 and rss[-1] < 128 * 1024  # inside the 128Mi container budget
print(f"SOAK2 {'PASS' if ok else 'FAIL'} (2nd-half growth {growth:.1f}%, last {rss[-1]} KiB)")
EOF

# ---- 3. concurrent scrape stress ----
python - > gpurun_out/scrape_stress.log 2>&1 <<'EOF'
import statistics, subprocess, sys, threading, time, urllib.request
proc = subprocess.Popen(["./bin/mi355-exporter", "-p", "19496", "-b", "127.0.0.1",
                         "-i", "250"], stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
time.sleep(2)
lat = []
lock = threading.Lock()
stop = time.time() + 15
def scraper():
    while time.time() < stop:
        t0 = time.perf_counter()
        urllib.request.urlopen("http://127.0.0.1:19496/metrics", timeout=5).read()
        with lock:
            lat.append(time.perf_counter() - t0)
threads = [threading.Thread(target=scraper) for _ in range(8)]
[t.start() for t in threads]
[t.join() for t in threads]
proc.terminate(); proc.wait()
lat.sort()
print(f"8-way concurrent scrapes: n={len(lat)} p50={statistics.median(lat)*1000:.2f} ms "
      f"p99={lat[int(len(lat)*0.99)]*1000:.2f} ms max={lat[-1]*1000:.2f} ms")
print("STRESS", "PASS" if lat[-1] < 1.0 else "CHECK")
EOF

tail -4 gpurun_out/daemon_soak.log
tail -2 gpurun_out/scrape_stress.log
grep -i -E 'GRBM|gui' gpurun_out/pmc_probe.log | head -6
