"""Per-RPC latency metrics.

The headline metric for this agent is p50 Allocate() latency (BASELINE.json),
so every RPC handler records into a lock-free-ish ring histogram; quantiles
are computed on read. Optionally exported via prometheus_client when the
agent runs with --metrics-port. The reference has no metrics at all
(SURVEY §5 observability)."""
from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional


class LatencyRecorder:
    """Fixed-size ring of recent latencies (seconds) + running count."""

    def __init__(self, capacity: int = 65536):
        self._ring: List[float] = [0.0] * capacity
        self._cap = capacity
        self._n = 0
        self._lock = threading.Lock()

    def observe(self, seconds: float) -> None:
        with self._lock:
            self._ring[self._n % self._cap] = seconds
            self._n += 1

    @property
    def count(self) -> int:
        return self._n

    def snapshot(self) -> List[float]:
        with self._lock:
            n = min(self._n, self._cap)
            return sorted(self._ring[:n])

    def quantile(self, q: float) -> Optional[float]:
        snap = self.snapshot()
        if not snap:
            return None
        idx = min(len(snap) - 1, int(q * len(snap)))
        return snap[idx]

    def summary_us(self) -> Dict[str, float]:
        snap = self.snapshot()
        if not snap:
            return {"count": 0}
        def q(p):
            return snap[min(len(snap) - 1, int(p * len(snap)))] * 1e6
        return {
            "count": self._n,
            "p50_us": q(0.50),
            "p90_us": q(0.90),
            "p99_us": q(0.99),
            "max_us": snap[-1] * 1e6,
            "mean_us": sum(snap) / len(snap) * 1e6,
        }


class Metrics:
    def __init__(self):
        self._recorders: Dict[str, LatencyRecorder] = {}
        self._lock = threading.Lock()

    def recorder(self, name: str) -> LatencyRecorder:
        with self._lock:
            rec = self._recorders.get(name)
            if rec is None:
                rec = self._recorders[name] = LatencyRecorder()
            return rec

    def time(self, name: str):
        rec = self.recorder(name)

        class _Ctx:
            def __enter__(self):
                self.t0 = time.perf_counter()
                return self

            def __exit__(self, *exc):
                rec.observe(time.perf_counter() - self.t0)
                return False

        return _Ctx()

    def summary(self) -> Dict[str, Dict[str, float]]:
        with self._lock:
            names = list(self._recorders)
        return {n: self._recorders[n].summary_us() for n in names}

    def serve_prometheus(self, port: int) -> None:
        """Best-effort Prometheus endpoint (summaries as gauges)."""
        from prometheus_client import REGISTRY, start_http_server
        from prometheus_client.core import GaugeMetricFamily

        metrics = self

        class _Collector:
            def collect(self):
                g = GaugeMetricFamily(
                    "egpu_rpc_latency_us", "per-RPC latency quantiles", labels=["rpc", "q"]
                )
                for name, summ in metrics.summary().items():
                    for k, v in summ.items():
                        g.add_metric([name, k], v)
                yield g

        REGISTRY.register(_Collector())
        start_http_server(port)


GLOBAL_METRICS = Metrics()
