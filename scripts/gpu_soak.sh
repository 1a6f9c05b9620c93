#!/bin/bash
# 60 s closed-loop soak on an MI355X box: mi355-exporter + gpu-pruner in
# daemon mode (1 s ticks, scale-down) against the native synthetic backend,
# while the gfx950 busy probe cycles on/off. Verifies over many ticks that
# busy phases produce no scale actions and idle phases cull everything, with
# no crashes, leaks of ticks, or stuck utilization.
set -x
mkdir -p gpurun_out

python - > gpurun_out/soak.log 2>&1 <<'EOF'
import json, os, subprocess, sys, time, urllib.request
sys.path.insert(0, ".")
os.environ["GPU_PRUNER_LOG"] = "info"
os.environ["PROMETHEUS_TOKEN"] = "soak"

from gpu_pruner_amd import _pruner_core as core, _gpumon, probe

backend = core.SyntheticBackend(n_pods=100)
backend.start()
os.environ["GPU_PRUNER_K8S_URL"] = backend.k8s_url

# short sliding window so busy bursts age out inside the phase cadence
exporter = subprocess.Popen(["./bin/mi355-exporter", "-p", "19420", "-b", "127.0.0.1",
                             "-i", "250", "--activity-window", "3"],
                            stdout=subprocess.DEVNULL,
                            stderr=subprocess.DEVNULL)
time.sleep(2)

sampler = _gpumon.Sampler(poll_interval_ms=200)
sampler.init()
sampler.start()

def scrape_ratio():
    text = urllib.request.urlopen("http://127.0.0.1:19420/metrics", timeout=3).read().decode()
    for line in text.splitlines():
        if line.startswith("DCGM_FI_PROF_GR_ENGINE_ACTIVE{") and 'gpu="0"' in line:
            return float(line.rsplit("} ", 1)[1])
    raise AssertionError("no series")

# round 2: production daemon shape — watch informers + OTLP export on
from gpu_pruner_amd.fixtures import FakeOtlpCollector
collector = FakeOtlpCollector().start()
os.environ["OTEL_EXPORTER_OTLP_ENDPOINT"] = collector.url
os.environ["OTEL_METRIC_EXPORT_INTERVAL"] = "2000"
core.otlp_init("gpu-pruner-soak")
spans_total = [0]
cfg = json.dumps({"duration": 30, "grace_period": 300, "run_mode": "scale-down",
                  "prometheus_url": backend.prom_url, "eval_strategy": "watch"})

def rss_fds():
    rss = 0
    with open("/proc/self/status") as f:
        for line in f:
            if line.startswith("VmRSS:"):
                rss = int(line.split()[1])
    return rss, len(os.listdir("/proc/self/fd"))

phases = []   # (phase, ticks, scaled_total)
import os as _os
t_end = time.time() + int(_os.environ.get("SOAK_SECONDS", "60"))
phase_idx = 0
ok = True
try:
    while time.time() < t_end:
        busy_phase = phase_idx % 2 == 1
        if busy_phase:
            probe.start(0, 0, 25.0)
            time.sleep(2.0)   # let utilization rise + exporter window catch it
        scaled = ticks = 0
        t_phase = time.time() + 8
        while time.time() < t_phase:
            ratio = scrape_ratio()
            backend.set_series_value(ratio)
            out = core.run_tick(cfg)
            scaled += out["scaled"]
            ticks += 1
            time.sleep(1.0)
        if busy_phase:
            probe.stop()
            # drain: let utilization settle before the next idle phase
            for _ in range(60):
                time.sleep(0.25)
                if scrape_ratio() == 0.0:
                    break
        phases.append(("busy" if busy_phase else "idle", ticks, scaled))
        spans_total[0] += collector.drain()[0]  # don't measure fixture growth
        rss, fds = rss_fds()
        print(f"phase {phase_idx} ({'busy' if busy_phase else 'idle'}): "
              f"{ticks} ticks, {scaled} scale actions, rss={rss} KiB fds={fds}", flush=True)
        phase_idx += 1
finally:
    try:
        probe.stop()
    except Exception:
        pass
    exporter.terminate(); exporter.wait()
    core.otlp_shutdown()
    spans = spans_total[0] + collector.drain()[0]
    collector.stop()
    core.informers_reset()
    sampler.stop(); backend.stop()
    print(f"otlp trace export batches during soak: {spans}", flush=True)

for phase, ticks, scaled in phases:
    if phase == "busy" and scaled != 0:
        print(f"FAIL: busy phase performed {scaled} scale actions"); ok = False
    if phase == "idle" and scaled == 0:
        print(f"FAIL: idle phase culled nothing"); ok = False
print("SOAK", "PASS" if ok else "FAIL", f"({len(phases)} phases)")
EOF
tail -12 gpurun_out/soak.log
