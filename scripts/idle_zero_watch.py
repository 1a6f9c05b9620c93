"""Long-horizon idle-zero check: with nothing on the GPU, every exporter
scrape over ~24 minutes must read DCGM_FI_PROF_GR_ENGINE_ACTIVE == 0.0
exactly (the culler's == 0 predicate must never see phantom activity), and
busy_percent == 0 on ~99%+ of raw samples (firmware housekeeping blips are
floored by the windowed ratio)."""
import subprocess, sys, time, urllib.request
sys.path.insert(0, ".")
proc = subprocess.Popen(["./bin/mi355-exporter", "-p", "19433", "-b", "127.0.0.1",
                         "-i", "500"], stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
time.sleep(3)
n = nonzero_ratio = nonzero_busy = 0
t_end = time.time() + int(sys.argv[1])
try:
    while time.time() < t_end:
        text = urllib.request.urlopen("http://127.0.0.1:19433/metrics", timeout=3).read().decode()
        ratio = busy = None
        for line in text.splitlines():
            if line.startswith("DCGM_FI_PROF_GR_ENGINE_ACTIVE{") and 'gpu="0"' in line:
                ratio = float(line.rsplit("} ", 1)[1])
            if line.startswith("DCGM_FI_DEV_GPU_UTIL{") and 'gpu="0"' in line:
                busy = float(line.rsplit("} ", 1)[1])
        n += 1
        if ratio != 0.0:
            nonzero_ratio += 1
            print(f"t={n*2}s NONZERO ratio {ratio}", flush=True)
        if busy != 0.0:
            nonzero_busy += 1
        time.sleep(2.0)
finally:
    proc.terminate(); proc.wait()
print(f"scrapes={n} nonzero_windowed_ratio={nonzero_ratio} nonzero_raw_busy={nonzero_busy}")
print("IDLE-ZERO", "PASS" if nonzero_ratio == 0 else "FAIL")
