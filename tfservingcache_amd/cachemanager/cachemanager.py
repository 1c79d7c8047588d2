"""CacheManager — node-local orchestrator.

The reference's CacheManager (pkg/cachemanager/cachemanager.go:56-322)
ensures, per request, that a model is (a) in the on-disk LRU and (b)
loaded in TF Serving, then directs the proxy at the local TF Serving.
Here (b) becomes "resident in the in-process model pool" and the proxy
hop disappears: the request handler calls straight into the engine.

Concurrency model (fixes SURVEY.md §2.3's biggest reference bug — one
RWMutex held across whole downloads, cachemanager.go:114-115):
  * hits take no global lock (LRU + pool are internally locked);
  * misses take a PER-MODEL-VERSION single-flight lock, so concurrent
    requests for the same cold model trigger one fetch, and fetches of
    different models proceed in parallel.
"""
from __future__ import annotations

import logging
import os
import threading
import time
from typing import Callable, Dict, List, Optional

from ..engine.model import LoadedModel, load_model_from_dir
from ..utils import metrics as mt
from .lrucache import LRUCache, Model, ModelId
from .modelpool import ModelPool
from .modelprovider import ModelProvider, validate_model_name

log = logging.getLogger("tfsc.cache")


class CacheError(Exception):
    pass


class ModelUnavailableError(CacheError):
    pass


class CacheManager:
    def __init__(self, provider: ModelProvider, cache: LRUCache,
                 pool: ModelPool, model_fetch_timeout: float = 10.0,
                 model_labels: bool = False):
        self.provider = provider
        self.cache = cache
        self.pool = pool
        self.fetch_timeout = model_fetch_timeout
        self.model_labels = model_labels
        self._flight_lock = threading.Lock()
        # mid -> [lock, refcount]; the entry stays registered while any
        # thread holds OR waits on the lock (a pop while waiters were
        # still queued on a stale lock object would let a later thread
        # create a fresh lock and fetch the same model concurrently)
        self._in_flight: Dict[ModelId, list] = {}
        # fired after a provider fetch lands on disk (before the pool
        # load) — main.py uses it to push the bytes to the model's other
        # replica slots over the RCCL plane
        self.on_cold_load: Optional[Callable[[str, int], None]] = None

    # -- the hot path ------------------------------------------------------
    def ensure_loaded(self, name: str, version: int) -> LoadedModel:
        """Returns the AVAILABLE model, fetching/loading on miss."""
        validate_model_name(name)
        labels = mt.model_labels(self.model_labels, name, version)
        start = time.monotonic()
        mt.cache_total.labels(*labels).inc()
        try:
            model = self.pool.get_model(name, version)
            # cache.get (not contains) so hits refresh the MRU order
            if model is not None and self.cache.get(name, version) is not None:
                mt.cache_hits.labels(*labels).inc()
                return model
            mt.cache_misses.labels(*labels).inc()
            with mt.cache_fetch_duration.labels(*labels).time():
                return self._fetch_model(name, version)
        finally:
            mt.cache_duration.labels(*labels).observe(
                time.monotonic() - start)

    def _fetch_model(self, name: str, version: int) -> LoadedModel:
        mid: ModelId = (name, version)
        # single-flight: one concurrent fetch per model-version
        with self._flight_lock:
            flight = self._in_flight.get(mid)
            if flight is None:
                flight = [threading.Lock(), 0]
                self._in_flight[mid] = flight
            flight[1] += 1
            lock = flight[0]
        with lock:
            try:
                # re-check after winning the lock
                model = self.pool.get_model(name, version)
                if model is not None and self.cache.contains(name, version):
                    return model
                if not self.cache.contains(name, version):
                    size = self.provider.model_size(name, version)
                    self.cache.ensure_free_bytes(size)
                    entry = self.provider.load_model(name, version,
                                                     self.cache.base_dir)
                    self.cache.put(entry)
                    if self.on_cold_load is not None:
                        try:
                            self.on_cold_load(name, version)
                        except Exception:       # noqa: BLE001
                            log.exception("on_cold_load hook failed")
                # a concurrent reload may evict this model between our
                # reload and the wait, or its MRU-prefix truncation may
                # exclude it entirely — wait in SHORT chunks and re-front
                # the model in the MRU between chunks, so an eviction
                # race costs ~1s instead of a full fetch-timeout chunk
                deadline = time.monotonic() + self.fetch_timeout
                failures = 0
                while True:
                    self.cache.get(name, version)     # keep it MRU
                    self._reload_pool()
                    remaining = deadline - time.monotonic()
                    if remaining <= 0:
                        raise TimeoutError(
                            f"model {name}:{version} not available after "
                            f"{self.fetch_timeout}s")
                    try:
                        return self.pool.wait_available(
                            name, version, min(remaining, 1.0))
                    except TimeoutError:
                        continue        # re-front + reload, bounded by
                                        # the deadline above
                    except RuntimeError:
                        # the pool load FAILED (e.g. a ReloadConfig
                        # raced the disk fetch) — the END entry is
                        # re-created by the next reload; genuine load
                        # errors shouldn't loop forever
                        failures += 1
                        if failures >= 4:
                            raise
            finally:
                with self._flight_lock:
                    flight[1] -= 1
                    if flight[1] == 0:
                        self._in_flight.pop(mid, None)

    def _reload_pool(self) -> None:
        """Declarative reload: pool gets the MRU-first prefix of the disk
        cache (cachemanager.go:167-175)."""
        desired = [(e.name, e.version) for e in self.cache.list_models()]
        self.pool.reload(desired, self._version_dir)

    def _version_dir(self, name: str, version: int) -> str:
        return os.path.join(self.cache.base_dir, name, str(version))

    # -- health (cachemanager.go:76-89) ------------------------------------
    def is_healthy(self, probe_model_name: str) -> bool:
        """Probe-model trick: the probe model must NOT exist; any state
        other than 'not found' means the serving backend is wedged —
        plus the provider's own check."""
        try:
            states = self.pool.get_status(probe_model_name)
            if states:
                log.warning("probe model %s unexpectedly present",
                            probe_model_name)
                return False
        except Exception:       # noqa: BLE001
            return False
        try:
            return bool(self.provider.check())
        except Exception:       # noqa: BLE001
            return False

    # -- admin -------------------------------------------------------------
    def list_cached(self) -> List[Model]:
        return self.cache.list_models()

    def close(self) -> None:
        pass


def make_cpu_loader(cache: LRUCache) -> Callable[[str, int], LoadedModel]:
    def loader(name: str, version: int) -> LoadedModel:
        vdir = os.path.join(cache.base_dir, name, str(version))
        return load_model_from_dir(vdir, name, version)
    return loader


def make_gpu_loader(cache: LRUCache, device: str = "cuda:0",
                    max_batch: int = 64, use_graphs: bool = True,
                    batching: bool = False,
                    batch_timeout_s: float = 0.001,
                    devices: Optional[List[str]] = None,
                    n_streams: int = 6,
                    dtype: str = "bf16",
                    prewarm_batch: Optional[int] = None,
                    prewarm_ctxs: int = 1
                    ) -> Callable[[str, int], LoadedModel]:
    """Loader that compiles the SavedModel onto an MI355X: weights land
    in the GPU's HBM pool (bf16, GEMM layouts pre-transformed) and the
    predict path runs the CDNA4 HIP kernels. With batching=True,
    concurrent Predicts merge server-side (TF Serving --enable_batching
    analog). With `devices` (several GPUs in one process), each model is
    assigned a GPU by consistent hash of model##version — the
    single-process analog of the ring's (node, GPU) slots."""
    import zlib

    def pick_device(name: str, version: int) -> str:
        if not devices:
            return device
        key = f"{name}##{version}"
        return devices[zlib.crc32(key.encode()) % len(devices)]

    import time as _time
    timing = os.environ.get("TFSC_LOAD_TIMING")

    def loader(name: str, version: int) -> LoadedModel:
        from ..engine.gpu import GpuModel
        from ..engine.warmup import run_warmup
        t0 = _time.monotonic()
        vdir = os.path.join(cache.base_dir, name, str(version))
        lm = load_model_from_dir(vdir, name, version)
        t1 = _time.monotonic()
        dev = pick_device(name, version)
        lm._gpu = GpuModel(lm.plan, device=dev, max_batch=max_batch,
                           use_graphs=use_graphs, n_streams=n_streams,
                           model_name=name, model_version=version,
                           dtype=dtype)
        t2 = _time.monotonic()
        if timing:
            log.warning("load timing %s: plan=%.1fms gpu=%.1fms",
                        name, (t1 - t0) * 1e3, (t2 - t1) * 1e3)
        lm.device = dev
        if batching:
            # merging happens inside the C++ fast path (leader-follower
            # on raw requests); the Python batcher stays off so
            # predict_bytes keeps taking the fast path. Requests that
            # fall back to the Python path run unmerged.
            lm._gpu._fast.enable_batching(
                max_batch, int(batch_timeout_s * 1e6))
            # the merged bucket must exist before the C++ path can merge
            lm._gpu.prewarm(max_batch)
        # TF Serving warmup files (assets.extra/tf_serving_warmup_requests):
        # contexts build + hipGraphs capture before AVAILABLE
        run_warmup(lm, vdir)
        if prewarm_batch and not batching:
            # build ONE fast context for the expected bucket on THIS
            # (loader) thread: under LRU churn most requests are a
            # model's first, and a prewarmed context lets them take the
            # C++ fast path instead of the Python build path
            t3 = _time.monotonic()
            lm._gpu.prewarm(prewarm_batch, prewarm_ctxs)
            if timing:
                log.warning("prewarm %.1fms", (_time.monotonic() - t3)
                            * 1e3)
        return lm
    return loader
