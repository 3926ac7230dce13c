"""Observability: counters/gauges/histograms with a Prometheus text endpoint.

Metric-set parity with the reference (``cdn-proto/src/metrics.rs:18-78``,
``connection/metrics.rs:12-28``, ``cdn-broker/src/metrics.rs:13-21``):
``total_bytes_sent``, ``total_bytes_recv``, ``latency`` (histogram of
in-broker allocation lifetime), ``running_latency`` (30 s window),
``num_users_connected``, ``num_brokers_connected`` — served at ``/metrics``.

Uses ``prometheus_client`` when importable, else a minimal local fallback
with the same text exposition format.
"""

from __future__ import annotations

import threading
import time
from typing import List, Tuple

try:
    from prometheus_client import (
        Counter as _PromCounter,
        Gauge as _PromGauge,
        Histogram as _PromHistogram,
        REGISTRY as _REGISTRY,
        generate_latest,
    )

    _HAVE_PROM = True
except Exception:  # pragma: no cover
    _HAVE_PROM = False


class _FallbackMetric:
    def __init__(self, name: str, kind: str) -> None:
        self.name = name
        self.kind = kind
        self.value = 0.0
        self.observations: List[float] = []
        self._lock = threading.Lock()

    def inc(self, n: float = 1.0) -> None:
        with self._lock:
            self.value += n

    def dec(self, n: float = 1.0) -> None:
        with self._lock:
            self.value -= n

    def set(self, v: float) -> None:
        with self._lock:
            self.value = v

    def observe(self, v: float) -> None:
        with self._lock:
            self.observations.append(v)
            self.value += v

    def expose(self) -> str:
        if self.kind == "histogram":
            count = len(self.observations)
            total = sum(self.observations)
            return (
                f"# TYPE {self.name} histogram\n"
                f"{self.name}_count {count}\n"
                f"{self.name}_sum {total}\n"
            )
        return f"# TYPE {self.name} {self.kind}\n{self.name} {self.value}\n"


_fallback_registry: List[_FallbackMetric] = []


def _counter(name: str, doc: str):
    if _HAVE_PROM:
        try:
            return _PromCounter(name, doc)
        except ValueError:
            pass  # duplicated registration under pytest re-imports
    m = _FallbackMetric(name, "counter")
    _fallback_registry.append(m)
    return m


def _gauge(name: str, doc: str):
    if _HAVE_PROM:
        try:
            return _PromGauge(name, doc)
        except ValueError:
            pass
    m = _FallbackMetric(name, "gauge")
    _fallback_registry.append(m)
    return m


class _LatencyHistogram:
    """Histogram + a 30 s running-latency window (reference metrics.rs:43-78)."""

    def __init__(self, name: str, doc: str) -> None:
        self._window: List[Tuple[float, float]] = []
        self._lock = threading.Lock()
        if _HAVE_PROM:
            try:
                self._hist = _PromHistogram(name, doc)
            except ValueError:
                self._hist = None
        else:
            self._hist = _FallbackMetric(name, "histogram")
            _fallback_registry.append(self._hist)

    def observe(self, v: float) -> None:
        if self._hist is not None:
            self._hist.observe(v)
        now = time.monotonic()
        with self._lock:
            self._window.append((now, v))
            cutoff = now - 30.0
            while self._window and self._window[0][0] < cutoff:
                self._window.pop(0)

    def running_latency(self) -> float:
        with self._lock:
            if not self._window:
                return 0.0
            return sum(v for _, v in self._window) / len(self._window)


BYTES_SENT = _counter("total_bytes_sent", "Total bytes sent over all connections")
BYTES_RECV = _counter("total_bytes_recv", "Total bytes received over all connections")
LATENCY = _LatencyHistogram("latency", "In-broker message allocation lifetime (s)")
NUM_USERS_CONNECTED = _gauge("num_users_connected", "Users connected to this broker")
NUM_BROKERS_CONNECTED = _gauge("num_brokers_connected", "Brokers connected to this broker")


def render_metrics() -> bytes:
    extra = f"running_latency {LATENCY.running_latency()}\n".encode()
    if _HAVE_PROM:
        return generate_latest(_REGISTRY) + extra
    body = "".join(m.expose() for m in _fallback_registry)
    return body.encode() + extra


async def serve_metrics(host: str, port: int):
    """Serve GET /metrics (reference metrics.rs:18-39). Returns the asyncio server."""
    import asyncio

    async def handle(reader: "asyncio.StreamReader", writer: "asyncio.StreamWriter") -> None:
        try:
            await reader.readline()  # request line; drain rest lazily
            body = render_metrics()
            writer.write(
                b"HTTP/1.1 200 OK\r\nContent-Type: text/plain; version=0.0.4\r\n"
                + f"Content-Length: {len(body)}\r\n\r\n".encode()
                + body
            )
            await writer.drain()
        finally:
            writer.close()

    return await asyncio.start_server(handle, host, port)
