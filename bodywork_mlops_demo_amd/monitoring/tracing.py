"""Serving-side request tracing: latency histogram + Prometheus metrics.

SURVEY.md §5 "Tracing / profiling": the reference's only self-measurement
is stage 4's hand-rolled wall-clock per request (stage_4:75-78).  Here
every scoring request is timed in-process and exported:

- a log-scale latency histogram + counters, queryable at ``GET /stats``;
- Prometheus exposition at ``GET /metrics`` when prometheus_client is
  installed (it is in this image), so a scrape target exists per replica.
"""
from __future__ import annotations


import threading
import time

try:
    from prometheus_client import (
        CONTENT_TYPE_LATEST,
        Counter,
        Histogram,
        generate_latest,
    )

    _PROM = True
except ImportError:  # pragma: no cover
    _PROM = False

_PROM_CACHE: dict = {}


class RequestTracer:
    #: log-spaced bucket upper bounds, 10 us .. 10 s
    BUCKETS = [1e-5 * (10 ** (i / 4)) for i in range(25)]

    def __init__(self, name: str = "scoring"):
        self._lock = threading.RLock()  # snapshot() -> percentile() re-enters
        self._counts = [0] * (len(self.BUCKETS) + 1)
        self._total = 0
        self._total_rows = 0
        self._total_time = 0.0
        if _PROM:
            # one collector set per metric name per process (create_app may
            # run more than once, e.g. under tests)
            cache = _PROM_CACHE.setdefault(name, {})
            if not cache:
                cache["requests"] = Counter(
                    f"{name}_requests_total", "scoring requests served")
                cache["rows"] = Counter(f"{name}_rows_total", "rows scored")
                cache["latency"] = Histogram(
                    f"{name}_request_seconds", "request latency",
                    buckets=self.BUCKETS)
            self._p_requests = cache["requests"]
            self._p_rows = cache["rows"]
            self._p_latency = cache["latency"]

    def observe(self, seconds: float, rows: int = 1) -> None:
        with self._lock:
            self._total += 1
            self._total_rows += rows
            self._total_time += seconds
            i = 0
            while i < len(self.BUCKETS) and seconds > self.BUCKETS[i]:
                i += 1
            self._counts[i] += 1
        if _PROM:
            self._p_requests.inc()
            self._p_rows.inc(rows)
            self._p_latency.observe(seconds)

    def percentile(self, q: float) -> float:
        """Approximate latency percentile from the histogram."""
        with self._lock:
            total = self._total
            if total == 0:
                return 0.0
            target = q * total
            acc = 0
            for i, c in enumerate(self._counts):
                acc += c
                if acc >= target:
                    return self.BUCKETS[min(i, len(self.BUCKETS) - 1)]
        return self.BUCKETS[-1]

    def snapshot(self) -> dict:
        with self._lock:
            total = self._total
            return {
                "requests": total,
                "rows_scored": self._total_rows,
                "mean_latency_s": self._total_time / total if total else 0.0,
                "p50_s": self.percentile(0.50),
                "p99_s": self.percentile(0.99),
                "rows_per_sec": (self._total_rows / self._total_time
                                 if self._total_time > 0 else 0.0),
            }

    @staticmethod
    def prometheus() -> tuple[bytes, str] | None:
        if not _PROM:
            return None
        return generate_latest(), CONTENT_TYPE_LATEST


class timed:
    """Context manager feeding a tracer: ``with timed(tracer, rows=n): ...``"""

    def __init__(self, tracer: RequestTracer, rows: int = 1):
        self.tracer = tracer
        self.rows = rows

    def __enter__(self):
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        self.tracer.observe(time.perf_counter() - self.t0, self.rows)
        return False
