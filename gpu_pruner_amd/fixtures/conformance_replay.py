"""Replay harness for the apiserver conformance transcripts.

Each transcript under ``fixtures/conformance/`` is a sequence of HTTP
request/expectation steps encoding *real kube-apiserver wire semantics*
(metav1.Status error shapes, the autoscaling/v1 Scale subresource, RFC 7386
merge-patch behavior, server-side Event defaulting, List envelopes). The
same transcript replays against:

  * the in-process ``FakeApiServer`` (always, in CI) — proving the fake the
    whole hermetic test suite rests on matches those semantics, and
  * a REAL kube-apiserver (``tests/test_envtest_e2e.py``) whenever envtest
    binaries are available (``KUBEBUILDER_ASSETS``) — closing the loop with
    genuine apiserver behavior.

Expectation vocabulary:
  status        — exact HTTP status
  body          — recursive SUBSET match (every key present and equal;
                  lists match pairwise as subsets)
  body_present  — paths that must exist (server-defaulted fields)
  body_absent   — paths that must NOT exist (merge-patch null deletion)
  headers       — response-header subset
  items_names   — exact set of .items[].metadata.name (List ordering-free)
"""

from __future__ import annotations

import json
import ssl
import urllib.request
from pathlib import Path

TRANSCRIPT_DIR = Path(__file__).parent / "conformance"


def transcripts() -> list[Path]:
    return sorted(TRANSCRIPT_DIR.glob("*.json"))


def is_subset(expected, actual, path="$") -> list[str]:
    """Returns a list of mismatch descriptions (empty = subset holds)."""
    errs = []
    if isinstance(expected, dict):
        if not isinstance(actual, dict):
            return [f"{path}: expected object, got {type(actual).__name__}"]
        for k, v in expected.items():
            if k not in actual:
                errs.append(f"{path}.{k}: missing")
            else:
                errs.extend(is_subset(v, actual[k], f"{path}.{k}"))
    elif isinstance(expected, list):
        if not isinstance(actual, list):
            return [f"{path}: expected array, got {type(actual).__name__}"]
        if len(actual) < len(expected):
            return [f"{path}: expected >= {len(expected)} items, got {len(actual)}"]
        for i, v in enumerate(expected):
            errs.extend(is_subset(v, actual[i], f"{path}[{i}]"))
    else:
        if expected != actual:
            errs.append(f"{path}: expected {expected!r}, got {actual!r}")
    return errs


def _walk(obj, path):
    for key in path:
        if not isinstance(obj, dict) or key not in obj:
            return False, None
        obj = obj[key]
    return True, obj


def http_request(base_url, method, path, body=None, headers=None,
                 content_type="application/json", insecure=False):
    data = None
    req_headers = dict(headers or {})
    if body is not None:
        data = json.dumps(body).encode()
        req_headers.setdefault("Content-Type", content_type)
    req = urllib.request.Request(base_url + path, data=data, method=method,
                                 headers=req_headers)
    ctx = None
    if base_url.startswith("https"):
        ctx = ssl.create_default_context()
        if insecure:
            ctx.check_hostname = False
            ctx.verify_mode = ssl.CERT_NONE
    try:
        resp = urllib.request.urlopen(req, timeout=15, context=ctx)
        status, rbody, rheaders = resp.status, resp.read(), dict(resp.headers)
    except urllib.error.HTTPError as e:
        status, rbody, rheaders = e.code, e.read(), dict(e.headers)
    try:
        parsed = json.loads(rbody) if rbody else {}
    except json.JSONDecodeError:
        parsed = {"_raw": rbody.decode(errors="replace")}
    return status, parsed, rheaders


def replay_step(base_url, step, default_headers=None, insecure=False):
    """Execute one step and return a list of mismatches (empty = pass)."""
    req = step["request"]
    expect = step["expect"]
    headers = dict(default_headers or {})
    headers.update(req.get("headers", {}))
    status, body, resp_headers = http_request(
        base_url, req["method"], req["path"], req.get("body"),
        headers, req.get("content_type", "application/json"), insecure=insecure)

    errs = []
    if status != expect["status"]:
        errs.append(f"status: expected {expect['status']}, got {status} ({body})")
    if "body" in expect:
        errs.extend(is_subset(expect["body"], body, "$"))
    for path in expect.get("body_present", []):
        ok, _ = _walk(body, path)
        if not ok:
            errs.append(f"body_present: {'.'.join(path)} missing")
    for path in expect.get("body_absent", []):
        ok, _ = _walk(body, path)
        if ok:
            errs.append(f"body_absent: {'.'.join(path)} present")
    for k, v in expect.get("headers", {}).items():
        got = resp_headers.get(k)
        if got != v:
            errs.append(f"header {k}: expected {v!r}, got {got!r}")
    if "items_names" in expect:
        names = sorted(i.get("metadata", {}).get("name", "")
                       for i in body.get("items", []))
        if names != sorted(expect["items_names"]):
            errs.append(f"items_names: expected {expect['items_names']}, got {names}")
    return errs
