"""Prometheus metrics — same metric names as the reference.

Reference definitions:
  pkg/cachemanager/cachemanager.go:24-43 (cache counters/histograms,
  labeled {model,version}, collapsed to {"all_models","-1"} when
  metrics.modelLabels=false — cachemanager.go:92-111)
  pkg/tfservingproxy/tfservingproxy.go:25-32 (proxy counters, {protocol});
  note the reference increments the failure counter on BOTH paths
  (tfservingproxy.go:61-66) — fixed here: failures only on failure.

MI355X additions are namespaced tfservingcache_engine_*.
"""
from __future__ import annotations

from prometheus_client import (CollectorRegistry, Counter, Gauge, Histogram,
                               generate_latest, CONTENT_TYPE_LATEST)

REGISTRY = CollectorRegistry()

cache_total = Counter(
    "tfservingcache_cache", "The total number of cache misses and hits",
    ["model", "version"], registry=REGISTRY)
cache_hits = Counter(
    "tfservingcache_cache_hits", "The total number of cache hits",
    ["model", "version"], registry=REGISTRY)
cache_misses = Counter(
    "tfservingcache_cache_misses", "The total number of cache misses",
    ["model", "version"], registry=REGISTRY)
cache_duration = Histogram(
    "tfservingcache_cache_duration_seconds",
    "The duration of cache requests, including hits and misses",
    ["model", "version"], registry=REGISTRY)
cache_fetch_duration = Histogram(
    "tfservingcache_cache_fetch_duration_seconds",
    "The duration of cache fetches (when cache miss)",
    ["model", "version"], registry=REGISTRY)

proxy_requests_total = Counter(
    "tfservingcache_proxy_requests", "The total number of requests",
    ["protocol"], registry=REGISTRY)
proxy_requests_failed = Counter(
    "tfservingcache_proxy_failures", "The total number of failed requests",
    ["protocol"], registry=REGISTRY)

# -- engine (MI355X) -------------------------------------------------------
engine_predict_duration = Histogram(
    "tfservingcache_engine_predict_duration_seconds",
    "End-to-end predict execution time in the native engine",
    ["model", "version", "device"], registry=REGISTRY)
engine_load_duration = Histogram(
    "tfservingcache_engine_model_load_duration_seconds",
    "SavedModel compile+upload time into the GPU pool",
    ["device"], registry=REGISTRY)
engine_pool_bytes = Gauge(
    "tfservingcache_engine_pool_bytes",
    "Bytes of model weights resident in the pool", ["device"],
    registry=REGISTRY)
plane_transfers = Counter(
    "tfservingcache_replica_plane_transfers_total",
    "Model byte-pushes over the RCCL/xGMI replica plane", ["role"],
    registry=REGISTRY)
plane_bytes = Counter(
    "tfservingcache_replica_plane_bytes_total",
    "Bytes moved over the RCCL/xGMI replica plane", ["role"],
    registry=REGISTRY)
engine_pool_models = Gauge(
    "tfservingcache_engine_pool_models",
    "Number of models resident in the pool", ["device"], registry=REGISTRY)


# per-stage fast-path time: queue->pinned staging, GPU (H2D DMA +
# kernels + D2H, stream-synchronized), response serialization —
# aggregated at scrape time from every resident model's C++ counters
# (SURVEY.md §5 "per-stage timing" tracing suggestion)
class _EngineStageCollector:
    def __init__(self):
        import weakref
        self._pools = []
        self._weakref = weakref

    def add_pool(self, pool) -> None:
        self._pools.append(self._weakref.ref(pool))

    def collect(self):
        from prometheus_client.core import CounterMetricFamily
        fam = CounterMetricFamily(
            "tfservingcache_engine_stage_seconds",
            "Cumulative fast-path time per stage across resident models",
            labels=["stage"])
        totals = {"stage_in": 0, "gpu": 0, "serialize": 0}
        alive = []
        for ref in self._pools:
            pool = ref()
            if pool is None:
                continue
            alive.append(ref)
            try:
                for entry in list(pool._entries.values()):  # noqa: SLF001
                    fast = getattr(getattr(entry.model, "_gpu", None),
                                   "_fast", None)
                    if fast is None:
                        continue
                    si, gp, se = fast.stage_ns()
                    totals["stage_in"] += si
                    totals["gpu"] += gp
                    totals["serialize"] += se
            except Exception:       # noqa: BLE001
                pass
        self._pools = alive
        for stage, ns in totals.items():
            fam.add_metric([stage], ns / 1e9)
        yield fam


engine_stages = _EngineStageCollector()
REGISTRY.register(engine_stages)


def render() -> bytes:
    return generate_latest(REGISTRY)


CONTENT_TYPE = CONTENT_TYPE_LATEST


def model_labels(enabled: bool, name: str, version) -> tuple:
    """metrics.modelLabels switch (cachemanager.go:92-111)."""
    if enabled:
        return name, str(version)
    return "all_models", "-1"
