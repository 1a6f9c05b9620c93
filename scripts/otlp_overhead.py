#!/usr/bin/env python3
"""Compare bench throughput with and without live OTLP export (config 5)."""
import json
import subprocess
import sys

def run(extra):
    out = subprocess.run([sys.executable, "bench.py", "--steps", "15",
                          "--warmup", "3", *extra],
                         capture_output=True, text=True, timeout=600)
    return json.loads(out.stdout.strip().splitlines()[-1])

a = run([])
b = run(["--otlp"])
print(f"no-otlp: {a['value']} pods/s  {a['ms_per_step']} ms")
print(f"otlp   : {b['value']} pods/s  {b['ms_per_step']} ms | {b['config']['otlp']}")
print(f"overhead: {(a['value'] / b['value'] - 1) * 100:.1f}%")
