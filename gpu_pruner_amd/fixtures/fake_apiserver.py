"""FakeApiServer — in-memory kube-apiserver double.

Faithful enough for the whole culler path: namespaced GETs for Pods and the
five scalable kinds, RFC 7386 merge-PATCH on objects, the ``/scale``
subresource, and Event POSTs. Objects are plain dicts; builder helpers mirror
the reference's test fixture builders (reference lib.rs:590-652) plus pods
with owner references for e2e-style flows.
"""

from __future__ import annotations

import copy
import json
import re
import threading
import time
import uuid
from datetime import datetime, timedelta, timezone
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

# (api_prefix, plural) → kind
_ROUTES = {
    ("api/v1", "pods"): "Pod",
    ("api/v1", "events"): "Event",
    ("apis/apps/v1", "deployments"): "Deployment",
    ("apis/apps/v1", "replicasets"): "ReplicaSet",
    ("apis/apps/v1", "statefulsets"): "StatefulSet",
    ("apis/kubeflow.org/v1", "notebooks"): "Notebook",
    ("apis/serving.kserve.io/v1beta1", "inferenceservices"): "InferenceService",
    ("apis/coordination.k8s.io/v1", "leases"): "Lease",
}

_PATH_RE = re.compile(
    r"^/(api/v1|apis/apps/v1|apis/kubeflow\.org/v1|apis/serving\.kserve\.io/v1beta1"
    r"|apis/coordination\.k8s\.io/v1)"
    r"/namespaces/([^/]+)/([^/]+)(?:/([^/]+))?(?:/(scale))?$"
)


def _now_rfc3339(offset_s: float = 0.0) -> str:
    dt = datetime.now(timezone.utc) + timedelta(seconds=offset_s)
    return dt.strftime("%Y-%m-%dT%H:%M:%SZ")


def _merge_patch(target, patch):
    if not isinstance(patch, dict) or not isinstance(target, dict):
        return copy.deepcopy(patch)
    for k, v in patch.items():
        if v is None:
            target.pop(k, None)
        elif isinstance(v, dict) and isinstance(target.get(k), dict):
            target[k] = _merge_patch(target[k], v)
        else:
            target[k] = copy.deepcopy(v)
    return target


class FakeApiServer:
    def __init__(self, host: str = "127.0.0.1", port: int = 0, token: str | None = None,
                 latency_s: float = 0.0, certfile: str | None = None,
                 keyfile: str | None = None, client_ca: str | None = None):
        self._lock = threading.Lock()
        # objects[(kind, namespace, name)] = dict
        self.objects: dict[tuple[str, str, str], dict] = {}
        self.events: list[dict] = []
        self.requests: list[tuple[str, str]] = []  # (method, path)
        self.token = token
        self.latency_s = latency_s  # simulated apiserver RTT for benchmarks
        # throttle injection: respond 429 (+ Retry-After) to the next N
        # requests — exercises the client's bounded-retry path
        self.throttle_next = 0
        self.retry_after_s = 0
        self.throttled = 0  # how many 429s were actually served
        # watch support: monotonic resourceVersion + event log
        self._rv = 1
        self._events: list[tuple[int, str, str, str, dict]] = []  # (rv, type, kind, ns, obj)
        self._event_cv = threading.Condition(self._lock)
        self._closing = False
        self.watch_requests = 0  # watch connections served (test observability)
        # failure injection for informer resilience tests
        self.watch_410_next = 0  # respond 410 Gone to the next N watch requests
        # conflict injection: respond 409 to the next N PUTs regardless of
        # resourceVersion — simulates another writer winning the GET→PUT
        # race window (leader-election renew conflicts)
        self.conflict_next_put = 0

        fixture = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"
            disable_nagle_algorithm = True

            def log_message(self, *args):
                pass

            def _send(self, status: int, obj, headers=None):
                body = json.dumps(obj).encode()
                self.send_response(status)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                for k, v in (headers or {}).items():
                    self.send_header(k, v)
                self.end_headers()
                self.wfile.write(body)

            def _status(self, code, reason, message="", details=None):
                """Full metav1.Status shape, as the real apiserver returns it
                (API conventions: status=Failure, reason, code, details)."""
                body = {"kind": "Status", "apiVersion": "v1",
                        "metadata": {}, "status": "Failure",
                        "reason": reason, "code": code}
                if message:
                    body["message"] = message
                if details:
                    body["details"] = details
                return body

            def _maybe_throttle(self) -> bool:
                with fixture._lock:
                    if fixture.throttle_next <= 0:
                        return False
                    fixture.throttle_next -= 1
                    fixture.throttled += 1
                    ra = fixture.retry_after_s
                self._send(429, self._status(429, "TooManyRequests",
                                             "Too many requests, please try again later."),
                           headers={"Retry-After": str(ra)})
                return True

            def _auth_ok(self) -> bool:
                if fixture.token is None:
                    return True
                auth = self.headers.get("Authorization", "")
                return auth == f"Bearer {fixture.token}"

            def _route(self):
                m = _PATH_RE.match(self.path.split("?")[0])
                if not m:
                    return None
                prefix, ns, plural, name, sub = m.groups()
                kind = _ROUTES.get((prefix, plural))
                if kind is None:
                    return None
                return kind, ns, name, sub

            def _serve_watch(self, kind, ns, params):
                """Kubernetes watch: chunked stream of {"type","object"} JSON
                lines for events after ?resourceVersion, live until
                timeoutSeconds, closing with a BOOKMARK carrying the latest
                resourceVersion (allowWatchBookmarks semantics)."""
                try:
                    since = int(params.get("resourceVersion", ["0"])[0] or 0)
                except ValueError:
                    since = 0
                timeout_s = float(params.get("timeoutSeconds", ["30"])[0])
                deadline = time.monotonic() + timeout_s
                with fixture._lock:
                    fixture.watch_requests += 1
                    if fixture.watch_410_next > 0:
                        fixture.watch_410_next -= 1
                        gone = True
                    else:
                        gone = False
                if gone:  # resourceVersion too old: client must re-LIST
                    return self._send(410, self._status(
                        410, "Expired", "too old resource version"))
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Transfer-Encoding", "chunked")
                self.end_headers()

                def chunk(data: bytes):
                    self.wfile.write(f"{len(data):x}\r\n".encode() + data + b"\r\n")
                    self.wfile.flush()

                last = since
                try:
                    while not fixture._closing:
                        with fixture._event_cv:
                            pending = [e for e in fixture._events
                                       if e[0] > last and e[2] == kind and e[3] == ns]
                            if not pending:
                                remaining = deadline - time.monotonic()
                                if remaining <= 0:
                                    break
                                fixture._event_cv.wait(min(remaining, 0.2))
                                continue
                        for rv, etype, _k, _n, obj in pending:
                            chunk(json.dumps({"type": etype, "object": obj}).encode()
                                  + b"\n")
                            last = rv
                    with fixture._lock:
                        rv_now = str(fixture._rv)
                    chunk(json.dumps({"type": "BOOKMARK", "object": {
                        "kind": kind, "metadata": {"resourceVersion": rv_now}}}).encode()
                        + b"\n")
                    self.wfile.write(b"0\r\n\r\n")
                    self.wfile.flush()
                except (BrokenPipeError, ConnectionResetError, OSError):
                    pass  # client went away mid-stream

            def do_GET(self):
                with fixture._lock:
                    fixture.requests.append(("GET", self.path))
                if fixture.latency_s:
                    time.sleep(fixture.latency_s)
                if self._maybe_throttle():
                    return
                if not self._auth_ok():
                    return self._send(401, self._status(401, "Unauthorized", "Unauthorized"))
                # cluster-scope pod list (used by the exporter's attribution
                # cache), with optional spec.nodeName fieldSelector
                path_only, _, query = self.path.partition("?")
                if path_only == "/api/v1/pods":
                    import urllib.parse as _up
                    sel = _up.parse_qs(query).get("fieldSelector", [""])[0]
                    node = None
                    if sel.startswith("spec.nodeName="):
                        node = sel.split("=", 1)[1]
                    with fixture._lock:
                        items = [copy.deepcopy(o) for (k, _, _), o in fixture.objects.items()
                                 if k == "Pod" and (node is None or
                                                   o.get("spec", {}).get("nodeName") == node)]
                    return self._send(200, {"kind": "PodList", "items": items})
                r = self._route()
                if r is None:
                    return self._send(404, self._status(404, "NotFound",
                                                        "the server could not find the requested resource"))
                kind, ns, name, _sub = r
                if name is None:
                    import urllib.parse as _up
                    params = _up.parse_qs(query)
                    if params.get("watch", ["false"])[0] == "true":
                        return self._serve_watch(kind, ns, params)
                    with fixture._lock:
                        items = [copy.deepcopy(o) for (k, n, _), o in fixture.objects.items()
                                 if k == kind and n == ns]
                        rv = str(fixture._rv)
                    return self._send(200, {"kind": kind + "List",
                                            "metadata": {"resourceVersion": rv},
                                            "items": items})
                with fixture._lock:
                    obj = fixture.objects.get((kind, ns, name))
                    if obj is None:
                        return self._send(404, self._status(
                            404, "NotFound",
                            f'{kind.lower()}s "{name}" not found',
                            details={"name": name, "kind": kind.lower() + "s"}))
                    return self._send(200, copy.deepcopy(obj))

            def do_PATCH(self):
                with fixture._lock:
                    fixture.requests.append(("PATCH", self.path))
                if fixture.latency_s:
                    time.sleep(fixture.latency_s)
                if self._maybe_throttle():
                    return
                if not self._auth_ok():
                    return self._send(401, self._status(401, "Unauthorized", "Unauthorized"))
                length = int(self.headers.get("Content-Length", "0"))
                patch = json.loads(self.rfile.read(length) or b"{}")
                r = self._route()
                if r is None or r[2] is None:
                    return self._send(404, self._status(404, "NotFound",
                                                        "the server could not find the requested resource"))
                kind, ns, name, sub = r
                with fixture._lock:
                    obj = fixture.objects.get((kind, ns, name))
                    if obj is None:
                        return self._send(404, self._status(
                            404, "NotFound", f'{kind.lower()}s "{name}" not found',
                            details={"name": name, "kind": kind.lower() + "s"}))
                    if sub == "scale":
                        # /scale understands only spec.replicas; a merge patch
                        # without it is a no-op returning the current Scale
                        # (real-apiserver semantics)
                        replicas = (patch.get("spec") or {}).get("replicas")
                        if replicas is not None:
                            obj.setdefault("spec", {})["replicas"] = replicas
                            fixture._record_locked("MODIFIED", kind, obj)
                        current = obj.get("spec", {}).get("replicas", 0)
                        scale = {
                            "kind": "Scale", "apiVersion": "autoscaling/v1",
                            "metadata": {
                                "name": name, "namespace": ns,
                                "uid": obj.get("metadata", {}).get("uid", ""),
                                "resourceVersion": obj.get("metadata", {}).get(
                                    "resourceVersion", "1"),
                            },
                            "spec": {"replicas": current},
                            "status": {"replicas": current},
                        }
                        return self._send(200, scale)
                    _merge_patch(obj, patch)
                    fixture._record_locked("MODIFIED", kind, obj)
                    return self._send(200, copy.deepcopy(obj))

            def do_PUT(self):
                """Replace semantics with optimistic concurrency: a body
                carrying metadata.resourceVersion that does not match the
                stored object's is rejected 409 Conflict (what real Update
                calls do — Lease-based leader election depends on it)."""
                with fixture._lock:
                    fixture.requests.append(("PUT", self.path))
                if fixture.latency_s:
                    time.sleep(fixture.latency_s)
                if self._maybe_throttle():
                    return
                if not self._auth_ok():
                    return self._send(401, self._status(401, "Unauthorized", "Unauthorized"))
                length = int(self.headers.get("Content-Length", "0"))
                body = json.loads(self.rfile.read(length) or b"{}")
                r = self._route()
                if r is None or r[2] is None:
                    return self._send(404, self._status(404, "NotFound",
                                                        "the server could not find the requested resource"))
                kind, ns, name, _sub = r
                with fixture._lock:
                    if fixture.conflict_next_put > 0:
                        fixture.conflict_next_put -= 1
                        return self._send(409, self._status(
                            409, "Conflict",
                            f'Operation cannot be fulfilled on {kind.lower()}s "{name}": '
                            "the object has been modified; please apply your changes to "
                            "the latest version and try again",
                            details={"name": name, "kind": kind.lower() + "s"}))
                    obj = fixture.objects.get((kind, ns, name))
                    if obj is None:
                        return self._send(404, self._status(
                            404, "NotFound", f'{kind.lower()}s "{name}" not found',
                            details={"name": name, "kind": kind.lower() + "s"}))
                    sent_rv = body.get("metadata", {}).get("resourceVersion")
                    stored_rv = obj.get("metadata", {}).get("resourceVersion")
                    if sent_rv is not None and sent_rv != stored_rv:
                        return self._send(409, self._status(
                            409, "Conflict",
                            f'Operation cannot be fulfilled on {kind.lower()}s "{name}": '
                            "the object has been modified; please apply your changes to the "
                            "latest version and try again",
                            details={"name": name, "kind": kind.lower() + "s"}))
                    body.setdefault("metadata", {})
                    body["metadata"].setdefault("namespace", ns)
                    body["metadata"]["name"] = name
                    # preserve server-owned fields
                    for k in ("uid", "creationTimestamp"):
                        if k in obj.get("metadata", {}):
                            body["metadata"].setdefault(k, obj["metadata"][k])
                    fixture.objects[(kind, ns, name)] = body
                    fixture._record_locked("MODIFIED", kind, body)
                    return self._send(200, copy.deepcopy(body))

            def do_POST(self):
                with fixture._lock:
                    fixture.requests.append(("POST", self.path))
                if fixture.latency_s:
                    time.sleep(fixture.latency_s)
                if self._maybe_throttle():
                    return
                if not self._auth_ok():
                    return self._send(401, self._status(401, "Unauthorized", "Unauthorized"))
                length = int(self.headers.get("Content-Length", "0"))
                obj = json.loads(self.rfile.read(length) or b"{}")
                r = self._route()
                if r is None:
                    return self._send(404, {"kind": "Status", "code": 404})
                kind, ns, _name, _sub = r
                if kind == "Event":
                    obj.setdefault("metadata", {})
                    obj["metadata"].setdefault("uid", str(uuid.uuid4()))
                    obj["metadata"].setdefault("creationTimestamp", _now_rfc3339())
                    with fixture._lock:
                        obj["metadata"].setdefault("resourceVersion", str(fixture._rv))
                        fixture.events.append(obj)
                    return self._send(201, obj)
                name = obj.get("metadata", {}).get("name", "")
                with fixture._lock:
                    if (kind, ns, name) in fixture.objects:
                        return self._send(409, self._status(
                            409, "AlreadyExists",
                            f'{kind.lower()}s "{name}" already exists',
                            details={"name": name, "kind": kind.lower() + "s"}))
                    obj.setdefault("metadata", {}).setdefault("namespace", ns)
                    fixture.objects[(kind, ns, name)] = obj
                    fixture._record_locked("ADDED", kind, obj)
                return self._send(201, obj)

        # ThreadingHTTPServer's default listen backlog is 5: the engine's
        # 32-way fan-out (plus informer reconnects) overflows it under
        # load, surfacing as spurious connection-refused/reset
        class _Server(ThreadingHTTPServer):
            request_queue_size = 128

        self._server = _Server((host, port), Handler)
        self._server.daemon_threads = True
        self._tls = certfile is not None
        if certfile is not None:
            import ssl

            ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
            ctx.load_cert_chain(certfile, keyfile)
            if client_ca is not None:  # mTLS: require a client certificate
                ctx.load_verify_locations(client_ca)
                ctx.verify_mode = ssl.CERT_REQUIRED
            self._server.socket = ctx.wrap_socket(self._server.socket, server_side=True)
        self._thread = threading.Thread(
            target=lambda: self._server.serve_forever(poll_interval=0.05), daemon=True)

    # -- lifecycle -----------------------------------------------------------
    def start(self) -> "FakeApiServer":
        self._thread.start()
        return self

    def stop(self):
        with self._event_cv:
            self._closing = True
            self._event_cv.notify_all()
        self._server.shutdown()
        self._server.server_close()

    @property
    def url(self) -> str:
        host, port = self._server.server_address[:2]
        scheme = "https" if self._tls else "http"
        return f"{scheme}://{host}:{port}"

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()

    # -- object builders -----------------------------------------------------
    def _record_locked(self, etype: str, kind: str, obj: dict):
        """Bump the global resourceVersion, stamp the object, log the event.
        Caller holds self._lock."""
        self._rv += 1
        obj.setdefault("metadata", {})["resourceVersion"] = str(self._rv)
        self._events.append((self._rv, etype, kind,
                             obj["metadata"].get("namespace", ""), copy.deepcopy(obj)))
        self._event_cv.notify_all()

    def put(self, kind: str, obj: dict):
        meta = obj["metadata"]
        with self._lock:
            self.objects[(kind, meta["namespace"], meta["name"])] = obj
            self._record_locked("ADDED", kind, obj)
        return obj

    def delete_object(self, kind: str, ns: str, name: str):
        with self._lock:
            obj = self.objects.pop((kind, ns, name), None)
            if obj is not None:
                self._record_locked("DELETED", kind, obj)
        return obj

    def get(self, kind: str, ns: str, name: str) -> dict | None:
        with self._lock:
            obj = self.objects.get((kind, ns, name))
            return copy.deepcopy(obj) if obj is not None else None

    @staticmethod
    def _meta(name, ns, uid=None, owners=None, labels=None, age_s: float = 0.0):
        meta = {
            "name": name,
            "namespace": ns,
            "uid": uid or str(uuid.uuid4()),
            "resourceVersion": "1",
            "creationTimestamp": _now_rfc3339(-age_s),
        }
        if owners:
            meta["ownerReferences"] = owners
        if labels:
            meta["labels"] = labels
        return meta

    def add_deployment(self, name, ns, uid=None, replicas=1):
        return self.put("Deployment", {
            "apiVersion": "apps/v1", "kind": "Deployment",
            "metadata": self._meta(name, ns, uid),
            "spec": {"replicas": replicas},
        })

    def add_replicaset(self, name, ns, uid=None, owner=None, replicas=1):
        owners = None
        if owner is not None:
            owners = [{"apiVersion": "apps/v1", "kind": "Deployment",
                       "name": owner["metadata"]["name"], "uid": owner["metadata"]["uid"]}]
        return self.put("ReplicaSet", {
            "apiVersion": "apps/v1", "kind": "ReplicaSet",
            "metadata": self._meta(name, ns, uid, owners),
            "spec": {"replicas": replicas},
        })

    def add_statefulset(self, name, ns, uid=None, notebook_owner=None, replicas=1):
        owners = None
        if notebook_owner is not None:
            owners = [{"apiVersion": "kubeflow.org/v1", "kind": "Notebook",
                       "name": notebook_owner["metadata"]["name"],
                       "uid": notebook_owner["metadata"]["uid"]}]
        return self.put("StatefulSet", {
            "apiVersion": "apps/v1", "kind": "StatefulSet",
            "metadata": self._meta(name, ns, uid, owners),
            "spec": {"replicas": replicas},
        })

    def add_notebook(self, name, ns, uid=None):
        return self.put("Notebook", {
            "apiVersion": "kubeflow.org/v1", "kind": "Notebook",
            "metadata": self._meta(name, ns, uid),
            "spec": {"template": None},
        })

    def add_inferenceservice(self, name, ns, uid=None, min_replicas=1):
        return self.put("InferenceService", {
            "apiVersion": "serving.kserve.io/v1beta1", "kind": "InferenceService",
            "metadata": self._meta(name, ns, uid),
            "spec": {"predictor": {"minReplicas": min_replicas}},
        })

    def add_pod(self, name, ns, owner_kind=None, owner_name=None, owner_uid=None,
                labels=None, phase="Running", age_s: float = 7200.0,
                creation_timestamp: str | None = None):
        owners = None
        if owner_kind is not None:
            owners = [{"apiVersion": "apps/v1", "kind": owner_kind,
                       "name": owner_name, "uid": owner_uid or str(uuid.uuid4())}]
        meta = self._meta(name, ns, None, owners, labels, age_s or 0.0)
        if creation_timestamp is not None:
            meta["creationTimestamp"] = creation_timestamp
        elif age_s is None:  # explicitly no creation timestamp (skip-path test)
            meta.pop("creationTimestamp", None)
        pod = {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": meta,
            "spec": {},
            "status": {"phase": phase},
        }
        return self.put("Pod", pod)

    def inject_watch_error(self, kind: str, ns: str):
        """Append an in-band ERROR watch event (the apiserver's 410-inside-
        the-stream form); watchers must invalidate and re-LIST."""
        with self._lock:
            self._rv += 1
            self._events.append((self._rv, "ERROR", kind, ns,
                                 {"kind": "Status", "apiVersion": "v1",
                                  "status": "Failure", "reason": "Expired",
                                  "code": 410,
                                  "metadata": {"resourceVersion": str(self._rv),
                                               "name": "", "namespace": ns}}))
            self._event_cv.notify_all()
