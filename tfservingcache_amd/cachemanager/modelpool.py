"""Model pool manager — the reference's TFServingController, re-cast.

In the reference, a sidecar rewrites an external TF Serving's model
config over gRPC and POLLS GetModelStatus every 500 ms until AVAILABLE
(pkg/cachemanager/servingcontroller.go:88-157,
 pkg/cachemanager/cachemanager.go:167-195). Here the engine is
in-process: `reload()` applies the same declarative semantics (the
desired set is the first `max_concurrent_models` of the MRU list), loads
are EVENT-driven (condition variable, no polling), and the same
state machine is kept: START -> LOADING -> AVAILABLE -> UNLOADING -> END
(mirrors the enum copied at servingcontroller.go:29-54).

On GPU nodes each pool slot is a device-resident model (weights in the
HBM3E arena via engine/gpu.py); on CPU the pool holds compiled plans.
"""
from __future__ import annotations

import logging
import threading
import time
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

from ..utils import metrics as mt
from ..wire import messages as m
from .lrucache import ModelId

log = logging.getLogger("tfsc.pool")

# states (wire values from tensorflow.serving.ModelVersionStatus.State)
UNKNOWN, START, LOADING, AVAILABLE, UNLOADING, END = (
    m.STATE_UNKNOWN, m.STATE_START, m.STATE_LOADING, m.STATE_AVAILABLE,
    m.STATE_UNLOADING, m.STATE_END)


@dataclass
class PoolEntry:
    name: str
    version: int
    state: int = START
    error: str = ""
    model: object = None            # LoadedModel when AVAILABLE
    load_started: float = 0.0
    load_finished: float = 0.0


class ModelPool:
    """Holds loaded models up to max_concurrent_models; declarative reload."""

    def __init__(self, loader: Callable[[str, int], object],
                 max_concurrent_models: int = 2,
                 device: str = "cpu",
                 max_bytes: Optional[int] = None,
                 size_hint: Optional[Callable[[str, int], int]] = None):
        """loader(name, version) -> LoadedModel (blocking compile+upload).

        max_bytes: optional HBM budget for resident models (the per-GPU
        pool is sized for 288 GB of HBM3E; `serving.maxConcurrentModels`
        caps the COUNT like the reference, `engine.hbmPoolBytes` caps the
        BYTES). size_hint(name, version) estimates a model's device
        footprint before it is loaded (e.g. 3x its on-disk size for
        fp32->bf16 masters + transformed GEMM layouts)."""
        self._loader = loader
        self.max_concurrent = max_concurrent_models
        self.max_bytes = max_bytes
        self.size_hint = size_hint
        self.device = device
        self._lock = threading.Lock()
        self._cond = threading.Condition(self._lock)
        self._entries: Dict[ModelId, PoolEntry] = {}
        # seconds per completed load — benchmark/observability hook
        self.load_durations: List[float] = []
        # lifecycle hooks (e.g. the native front-end's FastModel
        # registry). on_available runs outside the pool lock;
        # on_unload runs under it (before release) — keep it light
        self.on_available: Optional[Callable[[str, int, object], None]] \
            = None
        self.on_unload: Optional[Callable[[str, int, object], None]] \
            = None

    # -- introspection (GetModelStatus semantics) --------------------------
    def get_status(self, name: str, version: Optional[int] = None
                   ) -> List[PoolEntry]:
        with self._lock:
            out = []
            for (n, v), e in self._entries.items():
                if n == name and (version is None or version == 0 or v == version):
                    out.append(PoolEntry(n, v, e.state, e.error))
            return out

    def available(self, name: str, version: int) -> bool:
        with self._lock:
            e = self._entries.get((name, version))
            return e is not None and e.state == AVAILABLE

    def get_model(self, name: str, version: int):
        with self._lock:
            e = self._entries.get((name, version))
            if e is not None and e.state == AVAILABLE:
                return e.model
            return None

    def resident_bytes(self) -> int:
        with self._lock:
            return sum(e.model.weight_bytes() for e in self._entries.values()
                       if e.state == AVAILABLE and e.model is not None)

    # -- declarative reload (ReloadConfig semantics) -----------------------
    def reload(self, desired: List[Tuple[str, int]],
               resolve_dir: Callable[[str, int], str]) -> None:
        """Bring the pool to `desired` (truncated to max_concurrent_models,
        MRU-first — cachemanager.go:168-170). Models leaving the set are
        unloaded; new ones load on background threads, signalled via the
        condition variable (replaces the reference's 500 ms status poll)."""
        desired = list(desired)[: self.max_concurrent]
        if self.max_bytes is not None and self.size_hint is not None:
            budgeted = []
            total = 0
            for mid in desired:
                try:
                    est = self.size_hint(*mid)
                except Exception:       # noqa: BLE001
                    est = 0
                if budgeted and total + est > self.max_bytes:
                    break
                budgeted.append(mid)
                total += est
            desired = budgeted
        want = set(desired)
        to_load: List[Tuple[str, int]] = []
        with self._lock:
            for mid in list(self._entries):
                # never cancel an in-flight load: a concurrent request is
                # blocked in wait_available on it; it will be reconciled by
                # the next reload once AVAILABLE (the pool may transiently
                # exceed max_concurrent_models by in-flight loads)
                if mid not in want and self._entries[mid].state == AVAILABLE:
                    self._unload_locked(mid)
            for mid in desired:
                e = self._entries.get(mid)
                if e is None or e.state in (END, UNKNOWN):
                    self._entries[mid] = PoolEntry(mid[0], mid[1], START)
                    to_load.append(mid)
        for mid in to_load:
            t = threading.Thread(target=self._load_one,
                                 args=(mid, resolve_dir(*mid)), daemon=True)
            t.start()

    def _unload_locked(self, mid: ModelId) -> None:
        e = self._entries.get(mid)
        if e is None:
            return
        e.state = UNLOADING
        model, e.model = e.model, None
        e.state = END
        del self._entries[mid]
        if model is not None and self.on_unload is not None:
            try:
                self.on_unload(mid[0], mid[1], model)
            except Exception:       # noqa: BLE001
                log.exception("on_unload hook failed for %s", mid)
        if model is not None and hasattr(model, "release"):
            try:
                model.release()
            except Exception:
                log.exception("release of %s failed", mid)
        mt.engine_pool_models.labels(self.device).set(len(self._entries))

    def _load_one(self, mid: ModelId, version_dir: str) -> None:
        name, version = mid
        with self._lock:
            e = self._entries.get(mid)
            if e is None or e.state not in (START,):
                return
            e.state = LOADING
            e.load_started = time.monotonic()
        try:
            with mt.engine_load_duration.labels(self.device).time():
                model = self._loader(name, version)
            err = ""
        except Exception as ex:         # noqa: BLE001
            log.exception("load of %s:%d failed", name, version)
            model, err = None, str(ex)
        with self._lock:
            e = self._entries.get(mid)
            if e is None or e.state != LOADING:
                # concurrently unloaded — drop the work
                if model is not None and hasattr(model, "release"):
                    model.release()
                return
            if model is None:
                e.state = END
                e.error = err
            else:
                e.model = model
                e.state = AVAILABLE
                e.load_finished = time.monotonic()
                self.load_durations.append(e.load_finished - e.load_started)
                if len(self.load_durations) > 200_000:
                    # long-lived servers: keep the recent half
                    del self.load_durations[:100_000]
            avail = [x for x in self._entries.values()
                     if x.state == AVAILABLE and x.model is not None]
            mt.engine_pool_models.labels(self.device).set(len(avail))
            try:
                mt.engine_pool_bytes.labels(self.device).set(
                    sum(x.model.weight_bytes() for x in avail))
            except Exception:       # noqa: BLE001
                pass
            self._cond.notify_all()
        if model is not None and self.on_available is not None:
            try:
                self.on_available(name, version, model)
            except Exception:       # noqa: BLE001
                log.exception("on_available hook failed for %s", mid)

    # -- event-driven wait (replaces the 500 ms poll loop) -----------------
    def wait_available(self, name: str, version: int,
                       timeout: float = 10.0):
        """Block until (name, version) is AVAILABLE; returns the model or
        raises TimeoutError / RuntimeError on failed load."""
        deadline = time.monotonic() + timeout
        seen = False
        with self._lock:
            while True:
                e = self._entries.get((name, version))
                if e is not None:
                    seen = True
                    if e.state == AVAILABLE:
                        return e.model
                    if e.state == END and e.error:
                        raise RuntimeError(
                            f"load of {name}:{version} failed: {e.error}")
                elif seen:
                    # the entry existed and vanished: it was unloaded
                    # between becoming AVAILABLE and this waiter waking
                    # (a concurrent reload's MRU truncation). Waiting
                    # longer is pointless — time out NOW so the caller
                    # re-fronts the model in the MRU and reloads.
                    raise TimeoutError(
                        f"model {name}:{version} evicted before use")
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    raise TimeoutError(
                        f"model {name}:{version} not available after "
                        f"{timeout}s")
                self._cond.wait(remaining)

    def model_states(self) -> Dict[ModelId, int]:
        with self._lock:
            return {mid: e.state for mid, e in self._entries.items()}
