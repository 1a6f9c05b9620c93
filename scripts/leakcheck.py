#!/usr/bin/env python3
"""ASan leak check of the native engine: 50 full scale-down ticks (GET and
LIST strategies) through an ASan-instrumented _pruner_core.

Build + run:
  mkdir -p /tmp/asan_mod && cd native && g++ -O1 -g -std=c++20 -fPIC -pthread \
    -fsanitize=address -fno-omit-frame-pointer $(python3 -c "import sysconfig, pybind11; \
    print('-I'+sysconfig.get_paths()['include']+' -I'+pybind11.get_include())") \
    -fvisibility=hidden -shared pybind/core_py.cpp common/*.cpp pruner/*.cpp \
    -o /tmp/asan_mod/_pruner_core$(python3 -c "import sysconfig; \
    print(sysconfig.get_config_var('EXT_SUFFIX'))") -lssl -lcrypto -pthread
  LD_PRELOAD=$(g++ -print-file-name=libasan.so) ASAN_OPTIONS=detect_leaks=1 \
    python3 scripts/leakcheck.py

Result (2026-09, this container): 86 KB in 83 allocations, all from CPython
module/type initialization (PyType_Ready etc.) — zero engine leaks.
"""
import json, os, sys, time
sys.path.insert(0, "/tmp/asan_mod")      # ASan'd module first
sys.path.insert(1, "/root/repo")
os.environ["GPU_PRUNER_LOG"] = "error"
os.environ["PROMETHEUS_TOKEN"] = "t"
import importlib.util
spec = importlib.util.spec_from_file_location(
    "_pruner_core", "/tmp/asan_mod/_pruner_core.cpython-310-x86_64-linux-gnu.so")
core = importlib.util.module_from_spec(spec)
spec.loader.exec_module(core)
b = core.SyntheticBackend(n_pods=200)
b.start()
os.environ["GPU_PRUNER_K8S_URL"] = b.k8s_url
cfg = json.dumps({"duration": 30, "grace_period": 300, "run_mode": "scale-down",
                  "prometheus_url": b.prom_url, "max_concurrency": 16})
for strat in ("get", "list"):
    c2 = json.dumps({**json.loads(cfg), "eval_strategy": strat})
    for _ in range(25):
        core.run_tick(c2)
b.stop()
print("ticks done")
