"""Test fixtures: in-process fake Prometheus and fake kube-apiserver.

The reference's unit tests construct objects in memory and its e2e tests need
a real kind cluster ("Gaps"). These fixtures close that gap:
HTTP servers faithful enough for the full daemon path (query → parse → pod
eligibility → owner walk → scale patch → Event POST) to run hermetically on
CPU, plus a synthetic 1000-pod cluster generator for the benchmark configs.
"""

from .fake_prom import FakePrometheus
from .fake_apiserver import FakeApiServer
from .synth import build_synthetic_cluster
from .fake_otlp import FakeOtlpCollector, FakeOtlpGrpcCollector
from .miniprom import MiniProm

__all__ = ["FakePrometheus", "FakeApiServer", "build_synthetic_cluster",
           "FakeOtlpCollector", "FakeOtlpGrpcCollector", "MiniProm"]
