"""Observability: Prometheus metrics facade.

Mirrors the reference ``persia-metrics`` crate (persia-metrics/src/lib.rs:40-210):
a process-wide manager, gated by ``enable_metrics``, with const labels
``instance``/``ip`` and a background push loop to
``PERSIA_METRICS_GATEWAY_ADDR`` (default ``metrics_gateway:9091``).

Metric names keep the reference's vocabulary so its Grafana dashboard panels
(resources/grafana/dashboards/perisa-training.json) resolve:

* engine (ex-embedding-worker, mod.rs:49-105): ``staleness``,
  ``num_pending_batches``, ``batch_unique_indices_rate``,
  ``lookup_preprocess_time_cost_sec``, ``lookup_rpc_time_cost_sec`` (the
  all-to-all), ``lookup_postprocess_time_cost_sec``,
  ``update_gradient_time_cost_sec``, ``nan_count``, ``nan_grad_skipped``
* store (ex-parameter-server, parameter mod.rs:27-79): ``index_miss_count``,
  ``index_miss_ratio``, ``gradient_id_miss_count``,
  ``lookup_hashmap_time_cost_sec``, ``evicted_count``
"""
import os
import socket
import threading
import time
from typing import Dict, Optional

from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.metrics")

_manager = None
_lock = threading.Lock()


class MetricsManager:
    def __init__(self, job_name: str = "persia_amd", enable: bool = False,
                 push_interval_seconds: int = 10):
        self.enable = enable
        self.job_name = job_name
        self.push_interval = push_interval_seconds
        self._gauges: Dict[str, object] = {}
        self._counters: Dict[str, object] = {}
        self._histograms: Dict[str, object] = {}
        self._registry = None
        self._push_thread: Optional[threading.Thread] = None
        if not enable:
            return
        try:
            from prometheus_client import CollectorRegistry

            self._registry = CollectorRegistry()
            self._const_labels = {
                "instance": os.environ.get("HOSTNAME", socket.gethostname()),
                "job": job_name,
            }
            gateway = os.environ.get("PERSIA_METRICS_GATEWAY_ADDR", "metrics_gateway:9091")
            self._gateway = gateway
            self._push_thread = threading.Thread(
                target=self._push_loop, daemon=True, name="persia-metrics-push"
            )
            self._push_thread.start()
        except ImportError:
            _logger.warning("prometheus_client not available; metrics disabled")
            self.enable = False

    def _push_loop(self):
        from prometheus_client import push_to_gateway

        while True:
            time.sleep(self.push_interval)
            try:
                push_to_gateway(self._gateway, job=self.job_name, registry=self._registry)
            except Exception:
                pass  # gateway absent: keep collecting locally

    def gauge(self, name: str, doc: str = ""):
        if not self.enable:
            return _Noop()
        if name not in self._gauges:
            from prometheus_client import Gauge

            self._gauges[name] = Gauge(name, doc or name, ["feat"], registry=self._registry)
        return _Wrapped(self._gauges[name])

    def counter(self, name: str, doc: str = ""):
        if not self.enable:
            return _Noop()
        if name not in self._counters:
            from prometheus_client import Counter

            self._counters[name] = Counter(name, doc or name, ["feat"], registry=self._registry)
        return _Wrapped(self._counters[name])

    def histogram(self, name: str, doc: str = ""):
        if not self.enable:
            return _Noop()
        if name not in self._histograms:
            from prometheus_client import Histogram

            self._histograms[name] = Histogram(name, doc or name, ["feat"], registry=self._registry)
        return _Wrapped(self._histograms[name])


class _Noop:
    def set(self, *a, **k):
        pass

    def inc(self, *a, **k):
        pass

    def observe(self, *a, **k):
        pass

    def labels(self, *a, **k):
        return self


class _Wrapped:
    def __init__(self, metric):
        self._m = metric

    def labels(self, feat: str = ""):
        return self._m.labels(feat=feat)

    def set(self, v):
        self._m.labels(feat="").set(v)

    def inc(self, v=1):
        self._m.labels(feat="").inc(v)

    def observe(self, v):
        self._m.labels(feat="").observe(v)


def get_metrics_manager(enable: Optional[bool] = None) -> MetricsManager:
    global _manager
    with _lock:
        if _manager is None or (enable and not _manager.enable):
            if enable is None:
                enable = os.environ.get("PERSIA_ENABLE_METRICS", "0") in ("1", "true")
            _manager = MetricsManager(enable=bool(enable))
        return _manager


def reset_metrics_manager() -> None:
    """Drop the process-wide manager (tests)."""
    global _manager
    with _lock:
        _manager = None


class EngineMetrics:
    """Per-engine holder with the reference metric names."""

    def __init__(self, enable: bool):
        m = get_metrics_manager(enable)
        self.staleness = m.gauge("staleness", "lookups in flight ahead of updates")
        self.num_pending_batches = m.gauge("num_pending_batches")
        self.batch_unique_indices_rate = m.gauge("batch_unique_indices_rate")
        self.lookup_preprocess_time_cost_sec = m.gauge("lookup_preprocess_time_cost_sec")
        self.lookup_rpc_time_cost_sec = m.gauge("lookup_rpc_time_cost_sec")
        self.lookup_postprocess_time_cost_sec = m.gauge("lookup_postprocess_time_cost_sec")
        self.update_gradient_time_cost_sec = m.gauge("update_gradient_time_cost_sec")
        self.nan_count = m.counter("nan_count")
        self.nan_grad_skipped = m.counter("nan_grad_skipped")
        self.index_miss_count = m.counter("index_miss_count")
        self.gradient_id_miss_count = m.counter("gradient_id_miss_count")
        self.lookup_hashmap_time_cost_sec = m.gauge("lookup_hashmap_time_cost_sec")
        self.evicted_count = m.counter("evicted_count")
        self.distinct_id_estimate = m.gauge(
            "distinct_id_estimate", "HLL distinct-sign estimate per feature"
        )
        self.inc_packets_dumped = m.counter("inc_packets_dumped")
        self.inc_update_delay_sec = m.gauge(
            "inc_update_delay_sec",
            "age of the newest applied incremental packet (infer side)",
        )

    def sample_values(self) -> Dict[str, float]:
        """Current values of the unlabeled series (tests/debug)."""
        out: Dict[str, float] = {}
        for name, attr in vars(self).items():
            m = getattr(attr, "_m", None)
            if m is None:
                continue
            try:
                out[name] = m.labels(feat="")._value.get()
            except Exception:
                pass
        return out
