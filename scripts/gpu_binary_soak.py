#!/usr/bin/env python3
"""Production-shape soak on an MI355X box, using the SHIPPED binaries.

Unlike scripts/gpu_soak.sh (which drives run_tick in-process), this runs the
full three-process deployment exactly as a cluster would:

    mi355-exporter  --(scrape)-->  MiniProm  <--(PromQL)--  gpu-pruner
         |                                                      |
       rocm_smi on the real GPU                     FakeApiServer (fixtures)

The gfx950 busy probe cycles load on/off; the pruner daemon (watch
informers, OTLP self-metrics off by default here) must cull the Deployment
only when the GPU has been idle for the whole 1-minute window, and never
while the probe burst is inside it. The harness resets spec.replicas to 1
after each cull so every idle window is a fresh decision. The pruner
process's RSS/fds are sampled throughout.

Env: SOAK_SECONDS (default 600).
"""

import json
import os
import subprocess
import sys
import time
import urllib.request

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from gpu_pruner_amd import probe  # noqa: E402
from gpu_pruner_amd.fixtures import FakeApiServer, FakePrometheus, MiniProm  # noqa: E402

SOAK_SECONDS = int(os.environ.get("SOAK_SECONDS", "600"))
EXPORTER_PORT = 19431


def scrape_activity():
    text = urllib.request.urlopen(
        f"http://127.0.0.1:{EXPORTER_PORT}/metrics", timeout=3).read().decode()
    for line in text.splitlines():
        if line.startswith("DCGM_FI_PROF_GR_ENGINE_ACTIVE{") and 'gpu="0"' in line:
            return float(line.rsplit("} ", 1)[1])
    raise AssertionError("no activity series from the exporter")


def rss_fds(pid):
    rss = 0
    with open(f"/proc/{pid}/status") as f:
        for line in f:
            if line.startswith("VmRSS:"):
                rss = int(line.split()[1])
    return rss, len(os.listdir(f"/proc/{pid}/fd"))


def main():
    exporter = subprocess.Popen(
        ["./bin/mi355-exporter", "-p", str(EXPORTER_PORT), "-b", "127.0.0.1",
         "-i", "250", "--activity-window", "3"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    api = FakeApiServer().start()
    prom = MiniProm().start()

    dep = api.add_deployment("train", "ml")
    rs = api.add_replicaset("train-rs", "ml", owner=dep)
    api.add_pod("train-0", "ml", owner_kind="ReplicaSet", owner_name="train-rs",
                owner_uid=rs["metadata"]["uid"], age_s=3 * 3600)

    env = dict(os.environ)
    env["GPU_PRUNER_K8S_URL"] = api.url
    env["PROMETHEUS_TOKEN"] = "soak"
    env["GPU_PRUNER_LOG"] = "warn"
    pruner = subprocess.Popen(
        ["./bin/gpu-pruner", "--prometheus-url", prom.url, "--daemon-mode",
         "--run-mode", "scale-down", "--check-interval", "2",
         "-t", "1", "--grace-period", "0", "--eval-strategy", "watch"],
        env=env, stdout=subprocess.DEVNULL, stderr=subprocess.PIPE)

    time.sleep(2)
    ok = True
    culls = 0
    busy_violations = 0
    rss_samples = []
    t_end = time.time() + SOAK_SECONDS
    phase = 0
    try:
        while time.time() < t_end:
            busy_phase = phase % 2 == 1
            if busy_phase:
                probe.start(device=0, max_seconds=85.0)
            phase_end = time.time() + (80 if busy_phase else 75)
            protected = not busy_phase  # set once the burst is IN the window
            while time.time() < min(phase_end, t_end):
                v = scrape_activity()
                prom.ingest_activity("train-0", "ml", v)
                if busy_phase and not protected and v > 0.0:
                    # the burst is now inside the lookback window: from the
                    # NEXT tick on, a cull would be a real violation. A cull
                    # in the first seconds (window still fully idle from the
                    # previous phase) is correct reference semantics — reset
                    # it and arm the invariant.
                    time.sleep(3.0)  # let any in-flight tick finish
                    api.objects[("Deployment", "ml", "train")]["spec"]["replicas"] = 1
                    protected = True
                if pruner.poll() is not None:
                    raise AssertionError(f"pruner exited rc={pruner.returncode}")
                rss_samples.append(rss_fds(pruner.pid))
                time.sleep(1.0)
            if busy_phase:
                probe.stop()
                # with the burst inside the window, the pod must be protected
                now_replicas = api.get("Deployment", "ml", "train")["spec"]["replicas"]
                if protected and now_replicas == 0:
                    busy_violations += 1
                    ok = False
                # idle-out: wait for the window to age past the burst before
                # judging the next idle phase
                drain_end = time.time() + 70
                while time.time() < min(drain_end, t_end):
                    prom.ingest_activity("train-0", "ml", scrape_activity())
                    time.sleep(1.0)
            else:
                if api.get("Deployment", "ml", "train")["spec"]["replicas"] == 0:
                    culls += 1
                    api.get("Deployment", "ml", "train")  # observed
                    api.objects[("Deployment", "ml", "train")]["spec"]["replicas"] = 1
                else:
                    print(f"phase {phase}: idle phase did not cull", flush=True)
                    ok = False
            r, f = rss_samples[-1] if rss_samples else (0, 0)
            print(f"phase {phase} ({'busy' if busy_phase else 'idle'}): culls={culls} "
                  f"violations={busy_violations} pruner rss={r} KiB fds={f}", flush=True)
            phase += 1
    finally:
        try:
            probe.stop()
        except Exception:
            pass
        pruner.terminate()
        try:
            pruner.wait(timeout=10)
        except subprocess.TimeoutExpired:
            pruner.kill()
        exporter.terminate()
        exporter.wait(timeout=10)
        prom.stop()
        api.stop()

    first_rss = rss_samples[2][0] if len(rss_samples) > 2 else 0
    last_rss = rss_samples[-1][0] if rss_samples else 0
    fds = sorted(set(f for _, f in rss_samples))
    print(f"pruner RSS first/last: {first_rss}/{last_rss} KiB; distinct fd counts: {fds}")
    print(f"culls={culls} busy_violations={busy_violations}")
    print("BINARY SOAK", "PASS" if ok and culls >= 1 and busy_violations == 0 else "FAIL")


if __name__ == "__main__":
    main()
