"""MiniProm — a tiny Prometheus double that *evaluates* the idle query.

FakePrometheus serves canned results; MiniProm instead stores raw samples
per series and executes the semantics of the culler's idle query
(native/pruner/promql.cpp): ``max_over_time`` of
DCGM_FI_PROF_GR_ENGINE_ACTIVE (or DCGM_FI_DEV_GPU_UTIL/100 as fallback)
over the ``[Nm]`` window, grouped per GPU, node_type enrichment from
node_dmi_info with bare fallback, the ``== 0`` idle predicate, and the
optional power ``unless`` clause. Parameters (window, filters, label
convention, threshold) are recovered from the query text itself, so the
fixture exercises the exact wire query the daemon sends.

This closes the one semantic no other fixture covers: *the lookback
window*. A pod that was busy earlier in the window has peak > 0 and must
not become a candidate, even if idle right now.
"""

from __future__ import annotations

import json
import re
import threading
import time
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class MiniProm:
    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self._lock = threading.Lock()
        # metric name -> { labels(frozenset of items) -> [(ts, value)] }
        self.series: dict[str, dict[frozenset, list[tuple[float, float]]]] = {}
        self.queries: list[str] = []

        fixture = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"
            disable_nagle_algorithm = True

            def log_message(self, *args):
                pass

            def _send(self, obj):
                body = json.dumps(obj).encode()
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def _query(self, q):
                with fixture._lock:
                    fixture.queries.append(q)
                    result = fixture.evaluate(q)
                self._send({"status": "success",
                            "data": {"resultType": "vector", "result": result}})

            def do_GET(self):
                parsed = urllib.parse.urlparse(self.path)
                q = urllib.parse.parse_qs(parsed.query).get("query", [""])[0]
                self._query(q)

            def do_POST(self):
                length = int(self.headers.get("Content-Length", "0"))
                body = self.rfile.read(length).decode()
                q = urllib.parse.parse_qs(body).get("query", [""])[0]
                self._query(q)

        # ThreadingHTTPServer's default listen backlog is 5: the engine's
        # 32-way fan-out (plus informer reconnects) overflows it under
        # load, surfacing as spurious connection-refused/reset
        class _Server(ThreadingHTTPServer):
            request_queue_size = 128

        self._server = _Server((host, port), Handler)
        self._server.daemon_threads = True
        self._thread = threading.Thread(
            target=lambda: self._server.serve_forever(poll_interval=0.05), daemon=True)

    # -- lifecycle -----------------------------------------------------------
    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()

    @property
    def url(self):
        host, port = self._server.server_address[:2]
        return f"http://{host}:{port}"

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()

    # -- ingestion -----------------------------------------------------------
    def ingest(self, metric: str, labels: dict, value: float, age_s: float = 0.0):
        """Record one sample `age_s` seconds in the past."""
        key = frozenset(labels.items())
        with self._lock:
            self.series.setdefault(metric, {}).setdefault(key, []).append(
                (time.time() - age_s, value))

    def ingest_activity(self, pod, namespace, value, age_s=0.0, gpu="0",
                        hostname="node-0", model_name="AMD Instinct MI355X",
                        container="main", honor_labels=False):
        prefix = "" if honor_labels else "exported_"
        self.ingest("DCGM_FI_PROF_GR_ENGINE_ACTIVE", {
            "Hostname": hostname, f"{prefix}pod": pod,
            f"{prefix}namespace": namespace, f"{prefix}container": container,
            "gpu": gpu, "modelName": model_name}, value, age_s)

    def ingest_power(self, pod, namespace, watts, age_s=0.0, gpu="0",
                     hostname="node-0", model_name="AMD Instinct MI355X",
                     container="main", honor_labels=False):
        prefix = "" if honor_labels else "exported_"
        self.ingest("DCGM_FI_DEV_POWER_USAGE", {
            "Hostname": hostname, f"{prefix}pod": pod,
            f"{prefix}namespace": namespace, f"{prefix}container": container,
            "gpu": gpu, "modelName": model_name}, watts, age_s)

    # -- evaluation of the idle-query shape -----------------------------------
    def evaluate(self, q: str):
        m = re.search(r"\[(\d+)m\]", q)
        window_s = int(m.group(1)) * 60 if m else 1800
        honor = "exported_pod" not in q
        pl = "pod" if honor else "exported_pod"
        nl = "namespace" if honor else "exported_namespace"
        ns_re = None
        nsm = re.search(re.escape(nl) + r' =~ "((?:[^"\\]|\\.)*)"', q)
        if nsm:
            ns_re = re.compile(nsm.group(1).replace('\\"', '"').replace("\\\\", "\\"))
        model_re = None
        mm = re.search(r'modelName =~ "((?:[^"\\]|\\.)*)"', q)
        if mm:
            model_re = re.compile(mm.group(1).replace('\\"', '"').replace("\\\\", "\\"))
        power_threshold = None
        pm = re.search(r">= ([0-9.]+)\s*\)", q) if "unless" in q else None
        if pm:
            power_threshold = float(pm.group(1))

        now = time.time()
        horizon = now - window_s

        def peak(samples):
            vals = [v for (ts, v) in samples if ts >= horizon]
            return max(vals) if vals else None

        def series_matches(labels):
            if not labels.get(pl):
                return False
            if ns_re and not ns_re.search(labels.get(nl, "")):
                return False
            if model_re and not model_re.search(labels.get("modelName", "")):
                return False
            return True

        # primary metric, with the /100 fallback for label-sets only present
        # in DCGM_FI_DEV_GPU_UTIL
        groups: dict[frozenset, float] = {}
        for key, samples in self.series.get("DCGM_FI_PROF_GR_ENGINE_ACTIVE", {}).items():
            p = peak(samples)
            if p is not None:
                groups[key] = p
        for key, samples in self.series.get("DCGM_FI_DEV_GPU_UTIL", {}).items():
            if key in groups:
                continue
            p = peak(samples)
            if p is not None:
                groups[key] = p / 100.0

        # power peaks per (pod, namespace) for the unless clause
        power_peaks: dict[tuple, float] = {}
        if power_threshold is not None:
            for key, samples in self.series.get("DCGM_FI_DEV_POWER_USAGE", {}).items():
                labels = dict(key)
                p = peak(samples)
                if p is None:
                    continue
                k = (labels.get(pl, ""), labels.get(nl, ""))
                power_peaks[k] = max(power_peaks.get(k, 0.0), p)

        # node_type enrichment (join on Hostname), bare fallback otherwise
        node_types = {}
        for key, samples in self.series.get("node_dmi_info", {}).items():
            labels = dict(key)
            host = labels.get("instance", labels.get("Hostname", ""))
            if host and "product_name" in labels:
                node_types[host] = labels["product_name"]

        result = []
        for key, value in groups.items():
            labels = dict(key)
            if not series_matches(labels):
                continue
            if value != 0.0:  # the == 0 idle predicate
                continue
            if power_threshold is not None:
                k = (labels.get(pl, ""), labels.get(nl, ""))
                if power_peaks.get(k, 0.0) >= power_threshold:
                    continue
            out_labels = dict(labels)
            host = labels.get("Hostname", "")
            if host in node_types:
                out_labels["node_type"] = node_types[host]
            result.append({"metric": out_labels, "value": [now, str(value)]})
        return result
