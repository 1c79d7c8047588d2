"""LoadedModel: a compiled SavedModel ready to serve predictions.

CPU path: plan interpreted by executor_cpu (numpy fp32).
GPU path: plan compiled by the CDNA4 HIP engine (engine/gpu.py) into a
per-GPU resident model with bf16 MFMA kernels; LoadedModel is the common
facade the cache tier talks to.
"""
from __future__ import annotations

import collections
import os
import threading
from typing import Dict, Optional

import numpy as np

from .executor_cpu import CpuExecutor
from .planner import Plan, compile_graph, is_sym
from .savedmodel import read_saved_model
from ..wire import graph as g
from ..wire import messages as m


class ModelExecError(Exception):
    pass


class LoadedModel:
    def __init__(self, name: str, version: int, plan: Plan,
                 device: str = "cpu", gpu_model=None):
        self.name = name
        self.version = version
        self.plan = plan
        self.device = device
        self._cpu = CpuExecutor(plan)
        self._gpu = gpu_model       # engine/gpu.py GpuModel, if on GPU
        self._lock = threading.Lock()
        self._batcher = None

    def enable_batching(self, max_batch: int = 64,
                        timeout_s: float = 0.002) -> None:
        """Server-side dynamic batching (TF Serving --enable_batching
        analog): concurrent Predicts merge into one plan execution."""
        from .batching import DynamicBatcher
        batch_dims = {}
        for alias, idx in self.plan.sig_inputs.items():
            shape = self.plan.tensors[idx].shape
            if not shape or not is_sym(shape[0]):
                batch_dims = {}
                break
            batch_dims[alias] = 0
        self._batcher = DynamicBatcher(self._predict_impl, batch_dims,
                                       max_batch=max_batch,
                                       timeout_s=timeout_s)

    # -- introspection ------------------------------------------------------
    @property
    def signature_def(self) -> Optional[m.SignatureDef]:
        return self.plan.signature_def

    def weight_bytes(self) -> int:
        if self._gpu is not None:
            return self._gpu.weight_bytes()
        return self.plan.weight_bytes()

    def release(self) -> None:
        if self._gpu is not None:
            self._gpu.release()
            self._gpu = None

    # -- execution ----------------------------------------------------------
    def _infer_batch(self, feeds: Dict[int, np.ndarray]) -> int:
        for idx, arr in feeds.items():
            shape = self.plan.tensors[idx].shape
            for i, d in enumerate(shape):
                if is_sym(d):
                    if i >= arr.ndim:
                        raise ModelExecError(
                            f"input rank {arr.ndim} < expected {len(shape)}")
                    k = -d
                    if arr.shape[i] % k:
                        raise ModelExecError(
                            f"input dim {arr.shape[i]} not divisible by {k}")
                    return arr.shape[i] // k
        return 1

    def _check_feeds(self, feeds: Dict[int, np.ndarray], batch: int) -> None:
        for idx, arr in feeds.items():
            want = self.plan.resolve_shape(self.plan.tensors[idx].shape, batch)
            if tuple(arr.shape) != want:
                raise ModelExecError(
                    f"input {self.plan.tensors[idx].name}: shape "
                    f"{tuple(arr.shape)} != expected {want}")

    def predict(self, inputs: Dict[str, np.ndarray],
                output_filter=None) -> Dict[str, np.ndarray]:
        """Predict with signature-alias-keyed inputs/outputs (batched
        server-side when enable_batching was called)."""
        if self._batcher is not None:
            return self._batcher.predict(
                {k: np.asarray(v) if not hasattr(v, "segments") else v
                 for k, v in inputs.items()},
                output_filter)
        return self._predict_impl(inputs, output_filter)

    def _predict_impl(self, inputs: Dict[str, np.ndarray],
                      output_filter=None) -> Dict[str, np.ndarray]:
        from ..utils import metrics as mt
        with mt.engine_predict_duration.labels(self.name,
                                               str(self.version),
                                               self.device).time():
            return self._predict_inner(inputs, output_filter)

    def _predict_inner(self, inputs: Dict[str, np.ndarray],
                       output_filter=None) -> Dict[str, np.ndarray]:
        plan = self.plan
        feeds: Dict[int, np.ndarray] = {}
        for alias, arr in inputs.items():
            idx = plan.sig_inputs.get(alias)
            if idx is None:
                # also accept raw graph tensor names
                idx = plan.by_name.get(alias if ":" in alias else alias + ":0")
            if idx is None:
                raise ModelExecError(f"unknown input {alias!r}")
            feeds[idx] = arr if hasattr(arr, "segments") else np.asarray(arr)
        missing = [a for a, i in plan.sig_inputs.items() if i not in feeds]
        if missing:
            raise ModelExecError(f"missing inputs: {missing}")
        batch = self._infer_batch(feeds)
        self._check_feeds(feeds, batch)

        out_aliases = list(plan.sig_outputs)
        if output_filter:
            out_aliases = [a for a in out_aliases if a in set(output_filter)]
        fetch = [plan.sig_outputs[a] for a in out_aliases]

        if self._gpu is not None:
            vals = self._gpu.run(feeds, batch, fetch)
        else:
            vals = self._cpu.run(feeds, batch, fetch)
        return {a: vals[plan.sig_outputs[a]] for a in out_aliases}

    def session_run(self, feeds_by_name: Dict[str, np.ndarray],
                    fetch_names) -> Dict[str, np.ndarray]:
        plan = self.plan
        feeds: Dict[int, np.ndarray] = {}
        for name, arr in feeds_by_name.items():
            tname = name if ":" in name else name + ":0"
            idx = plan.by_name.get(tname)
            if idx is None:
                raise ModelExecError(f"unknown feed {name!r}")
            feeds[idx] = np.asarray(arr)
        fetch = []
        for name in fetch_names:
            tname = name if ":" in name else name + ":0"
            idx = plan.by_name.get(tname)
            if idx is None:
                raise ModelExecError(f"unknown fetch {name!r}")
            fetch.append(idx)
        batch = self._infer_batch(feeds) if feeds else 1
        if self._gpu is not None:
            vals = self._gpu.run(feeds, batch, fetch)
        else:
            vals = self._cpu.run(feeds, batch, fetch)
        return {name: vals[idx] for name, idx in zip(fetch_names, fetch)}


# content-identity plan cache: a fleet serving many copies/versions of
# one architecture (hardlinked model repos, blue/green rollouts of the
# same graph) should not recompile the identical SavedModel per name.
# Keyed by (dev, inode, size, mtime_ns) of saved_model.pb — with the
# disk provider's hardlink fetch a cache copy shares the repo's inode,
# so hits skip reading the file entirely. When the inode key misses
# (e.g. S3/AzBlob downloads produce fresh inodes for byte-identical
# content), a CONTENT key (xxh3 of the file bytes, ~10 GB/s) gives the
# same dedup at the cost of one hash pass. Plans are immutable after
# compile (executors only read them), so sharing is safe.
_PLAN_CACHE_CAP = 32
_plan_cache_lock = threading.Lock()
_plan_cache: "collections.OrderedDict[tuple, Plan]" = \
    collections.OrderedDict()


def _compile_cached(version_dir: str, signature_name: str) -> Plan:
    from .savedmodel import SAVED_MODEL_FILENAME
    var_prefix = os.path.join(version_dir, "variables", "variables")
    key = None
    try:
        st = os.stat(os.path.join(version_dir, SAVED_MODEL_FILENAME))
        key = (st.st_dev, st.st_ino, st.st_size, st.st_mtime_ns,
               signature_name)
        if os.path.exists(var_prefix + ".index"):
            sv = os.stat(var_prefix + ".index")
            key += (sv.st_dev, sv.st_ino, sv.st_size, sv.st_mtime_ns)
    except OSError:
        pass
    if key is not None:
        with _plan_cache_lock:
            plan = _plan_cache.get(key)
            if plan is not None:
                _plan_cache.move_to_end(key)
                return plan
    # content-hash fallback key (distinct inodes, identical bytes)
    ckey = None
    try:
        import xxhash
        h = xxhash.xxh3_64()
        with open(os.path.join(version_dir, SAVED_MODEL_FILENAME),
                  "rb") as f:
            for chunk in iter(lambda: f.read(1 << 22), b""):
                h.update(chunk)
        if os.path.exists(var_prefix + ".index"):
            for suffix in (".index", ".data-00000-of-00001"):
                fp = var_prefix + suffix
                if os.path.exists(fp):
                    with open(fp, "rb") as f:
                        for chunk in iter(lambda: f.read(1 << 22), b""):
                            h.update(chunk)
        ckey = ("content", h.intdigest(), signature_name)
        with _plan_cache_lock:
            plan = _plan_cache.get(ckey)
            if plan is not None:
                _plan_cache.move_to_end(ckey)
                if key is not None:
                    _plan_cache[key] = plan      # promote the inode key
                return plan
    except Exception:       # noqa: BLE001 — hashing is best-effort
        ckey = None
    graph_def, signatures = read_saved_model(version_dir)
    sig = signatures.get(signature_name)
    if sig is None and signatures:
        sig = next(iter(signatures.values()))
    if sig is None:
        raise ModelExecError(f"no signatures in {version_dir}")
    variables = None
    if os.path.exists(var_prefix + ".index"):
        from .tensor_bundle import read_bundle
        variables = read_bundle(var_prefix)
    plan = compile_graph(graph_def, sig, variables)
    with _plan_cache_lock:
        if key is not None:
            _plan_cache[key] = plan
        if ckey is not None:
            _plan_cache[ckey] = plan
        while len(_plan_cache) > _PLAN_CACHE_CAP:
            _plan_cache.popitem(last=False)
    return plan


def load_model_from_dir(version_dir: str, name: str, version: int,
                        signature_name: str = g.DEFAULT_SERVING_SIGNATURE
                        ) -> LoadedModel:
    plan = _compile_cached(version_dir, signature_name)
    return LoadedModel(name, version, plan)
