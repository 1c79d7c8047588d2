"""LocalServingHandler — the in-process "TF Serving" backend.

Implements the semantics TF Serving provides behind the reference
(Predict / Classify / Regress / GetModelMetadata / GetModelStatus /
HandleReloadConfigRequest / SessionRun) on top of CacheManager + the
native engine. Both the REST facade (rest.py) and the gRPC servicers
(grpc_server.py) call into this, mirroring how the reference's two
protocol fronts shared one CacheManager (cachemanager.go:268-309).
"""
from __future__ import annotations

import logging
import os
from typing import Dict, List, Optional, Tuple

import numpy as np

from ..cachemanager import CacheManager
from ..cachemanager.modelpool import AVAILABLE
from ..engine.model import LoadedModel, ModelExecError
from ..wire import graph as g
from ..wire import messages as m
from ..wire.tensor import numpy_to_tensorproto, tensorproto_to_numpy

log = logging.getLogger("tfsc.serving")


class ServingError(Exception):
    def __init__(self, message: str, code: int = m.ERROR_INVALID_ARGUMENT):
        super().__init__(message)
        self.code = code


class LocalServingHandler:
    def __init__(self, cache_manager: CacheManager):
        self.cm = cache_manager
        # model name -> {label: version}, populated by ReloadConfig
        # (ModelConfig.version_labels, model_server_config.proto field 8)
        self._version_labels: Dict[str, Dict[str, int]] = {}

    # -- helpers -----------------------------------------------------------
    def _resolve_label(self, name: str, label: str) -> Optional[int]:
        return self._version_labels.get(name, {}).get(label)

    def _resolve_version(self, name: str, version: int) -> int:
        if version:
            return version
        # version 0/unset -> latest: prefer what's already resident, else
        # ask the provider
        states = self.cm.pool.get_status(name)
        avail = [e.version for e in states if e.state == AVAILABLE]
        if avail:
            return max(avail)
        prov = self.cm.provider
        if hasattr(prov, "latest_version"):
            v = prov.latest_version(name)
            if v is not None:
                return v
        raise ServingError(f"no versions of model {name} found",
                           m.ERROR_NOT_FOUND)

    def get_model(self, name: str, version: int,
                  version_label: str = "") -> Tuple[LoadedModel, int]:
        if not name:
            raise ServingError("missing model name")
        if not version and version_label:
            labeled = self._resolve_label(name, version_label)
            if labeled is None:
                raise ServingError(
                    f"unknown version label {version_label!r} for model "
                    f"{name}", m.ERROR_NOT_FOUND)
            version = labeled
        version = self._resolve_version(name, version)
        try:
            return self.cm.ensure_loaded(name, version), version
        except TimeoutError as e:
            raise ServingError(str(e), m.ERROR_UNAVAILABLE)
        except FileNotFoundError as e:
            raise ServingError(str(e), m.ERROR_NOT_FOUND)
        except Exception as e:          # noqa: BLE001
            from ..cachemanager.modelprovider import ModelNotFoundError
            if isinstance(e, ModelNotFoundError):
                raise ServingError(str(e), m.ERROR_NOT_FOUND)
            log.exception("load failed for %s:%s", name, version)
            raise ServingError(str(e), m.ERROR_UNKNOWN)

    # -- Predict -----------------------------------------------------------
    def predict_arrays(self, name: str, version: int,
                       inputs: Dict[str, np.ndarray],
                       output_filter=None,
                       version_label: str = ""
                       ) -> Tuple[Dict[str, np.ndarray], int]:
        model, version = self.get_model(name, version, version_label)
        sig = model.signature_def
        # anonymous single input ('' key) -> sole signature input
        if "" in inputs:
            anon = inputs.pop("")
            if sig is not None and len(sig.inputs) == 1:
                inputs[next(iter(sig.inputs))] = anon
            elif len(model.plan.sig_inputs) == 1:
                inputs[next(iter(model.plan.sig_inputs))] = anon
            else:
                raise ServingError(
                    "anonymous input given but model has multiple inputs")
        for alias, arr in inputs.items():
            if getattr(arr, "dtype", None) is not None and \
                    arr.dtype == object:
                # DT_STRING decodes fine (wire/tensor.py) but the native
                # engine has no string ops — reject cleanly instead of
                # failing deep in the planner (reference passes strings
                # through to TF Serving untouched: tfservingproxy.go:201)
                raise ServingError(
                    f"input {alias!r} is DT_STRING: string tensors are "
                    "proxied at the routing tier but not executable by "
                    "the native engine", m.ERROR_INVALID_ARGUMENT)
        try:
            out = model.predict(inputs, output_filter)
        except ModelExecError as e:
            raise ServingError(str(e))
        except Exception as e:          # noqa: BLE001
            # request raced a pool eviction -> one re-fetch
            from ..engine.gpu import ModelReleasedError
            if not isinstance(e, ModelReleasedError):
                raise
            model, version = self.get_model(name, version)
            try:
                out = model.predict(inputs, output_filter)
            except ModelExecError as e2:
                raise ServingError(str(e2))
        return out, version

    def predict(self, req: m.PredictRequest) -> m.PredictResponse:
        spec = req.model_spec or m.ModelSpec()
        inputs = {}
        for alias, tp in req.inputs.items():
            try:
                inputs[alias] = tensorproto_to_numpy(tp)
            except Exception as e:      # noqa: BLE001
                raise ServingError(f"bad tensor for input {alias!r}: {e}")
        outputs, version = self.predict_arrays(
            spec.name, spec.version_value(), inputs,
            list(req.output_filter) or None,
            version_label=spec.version_label)
        resp = m.PredictResponse(model_spec=m.ModelSpec(
            name=spec.name, version=m.Int64Value(value=version),
            signature_name=spec.signature_name or
            g.DEFAULT_SERVING_SIGNATURE))
        for alias, arr in outputs.items():
            resp.outputs[alias] = numpy_to_tensorproto(
                np.ascontiguousarray(arr, dtype=np.float32)
                if arr.dtype.kind == "f" else arr)
        return resp

    # -- Classify / Regress (tf.Example inputs) ----------------------------
    def _examples_to_features(self, inp: m.Input) -> Dict[str, np.ndarray]:
        examples: List[m.Example] = []
        if inp is None:
            raise ServingError("missing input")
        if inp.example_list is not None:
            examples = inp.example_list.examples
        elif inp.example_list_with_context is not None:
            examples = inp.example_list_with_context.examples
        if not examples:
            raise ServingError("no examples in input")
        cols: Dict[str, list] = {}
        for ex in examples:
            feats = ex.features.feature if ex.features else {}
            for k, f in feats.items():
                if f.float_list is not None and f.float_list.value:
                    cols.setdefault(k, []).append(f.float_list.value)
                elif f.int64_list is not None and f.int64_list.value:
                    cols.setdefault(k, []).append(f.int64_list.value)
        out = {}
        for k, rows in cols.items():
            arr = np.asarray(rows, dtype=np.float32)
            if arr.shape[-1] == 1:
                arr = arr[..., 0]
            out[k] = arr
        if not out:
            raise ServingError("examples carry no numeric features")
        return out

    def classify(self, req: m.ClassificationRequest) -> m.ClassificationResponse:
        spec = req.model_spec or m.ModelSpec()
        feats = self._examples_to_features(req.input)
        if len(feats) == 1:
            feats = {"": next(iter(feats.values()))}
        outputs, version = self.predict_arrays(
            spec.name, spec.version_value(), feats,
            version_label=spec.version_label)
        # TF classification signatures name their outputs 'scores' (and
        # 'classes'); fall back to the first output otherwise
        scores = None
        for key in ("scores", "probabilities", "probs"):
            if key in outputs:
                scores = outputs[key]
                break
        if scores is None:
            scores = next(iter(outputs.values()))
        scores = np.atleast_2d(np.asarray(scores, dtype=np.float32))
        result = m.ClassificationResult()
        for row in scores:
            cl = m.Classifications(classes=[
                m.Class(label=str(i), score=float(s))
                for i, s in enumerate(np.atleast_1d(row))])
            result.classifications.append(cl)
        return m.ClassificationResponse(
            result=result,
            model_spec=m.ModelSpec(name=spec.name,
                                   version=m.Int64Value(value=version)))

    def regress(self, req: m.RegressionRequest) -> m.RegressionResponse:
        spec = req.model_spec or m.ModelSpec()
        feats = self._examples_to_features(req.input)
        if len(feats) == 1:
            feats = {"": next(iter(feats.values()))}
        outputs, version = self.predict_arrays(
            spec.name, spec.version_value(), feats,
            version_label=spec.version_label)
        vals = np.asarray(next(iter(outputs.values())),
                          dtype=np.float32).reshape(-1)
        return m.RegressionResponse(
            result=m.RegressionResult(
                regressions=[m.Regression(value=float(v)) for v in vals]),
            model_spec=m.ModelSpec(name=spec.name,
                                   version=m.Int64Value(value=version)))

    # -- metadata ----------------------------------------------------------
    def get_model_metadata(self, req: m.GetModelMetadataRequest
                           ) -> m.GetModelMetadataResponse:
        spec = req.model_spec or m.ModelSpec()
        model, version = self.get_model(spec.name, spec.version_value(),
                                        spec.version_label)
        resp = m.GetModelMetadataResponse(model_spec=m.ModelSpec(
            name=spec.name, version=m.Int64Value(value=version)))
        sig = model.signature_def
        if sig is not None:
            sd_map = m.SignatureDefMap()
            sd_map.signature_def[g.DEFAULT_SERVING_SIGNATURE] = sig
            resp.metadata["signature_def"] = m.Any(
                type_url="type.googleapis.com/"
                         "tensorflow.serving.SignatureDefMap",
                value=sd_map.encode())
        return resp

    # -- status ------------------------------------------------------------
    def get_model_status(self, req: m.GetModelStatusRequest
                         ) -> m.GetModelStatusResponse:
        spec = req.model_spec or m.ModelSpec()
        if not spec.name:
            raise ServingError("missing model name")
        version = spec.version_value()
        if not version and spec.version_label:
            labeled = self._resolve_label(spec.name, spec.version_label)
            if labeled is None:
                raise ServingError(
                    f"unknown version label {spec.version_label!r} for "
                    f"model {spec.name}", m.ERROR_NOT_FOUND)
            version = labeled
        entries = self.cm.pool.get_status(spec.name, version or None)
        resp = m.GetModelStatusResponse()
        if not entries:
            raise ServingError(
                f"Could not find any versions of model {spec.name}",
                m.ERROR_NOT_FOUND)
        for e in entries:
            resp.model_version_status.append(m.ModelVersionStatus(
                version=e.version, state=e.state,
                status=m.StatusProto(
                    error_code=m.ERROR_UNKNOWN if e.error else m.ERROR_OK,
                    error_message=e.error)))
        return resp

    # -- reload ------------------------------------------------------------
    def handle_reload_config(self, req: m.ReloadConfigRequest
                             ) -> m.ReloadConfigResponse:
        cfg = req.config
        desired: List[Tuple[str, int]] = []
        if cfg is not None and cfg.model_config_list is not None:
            for mc in cfg.model_config_list.config:
                if mc.version_labels:
                    self._version_labels[mc.name] = {
                        k: int(v) for k, v in mc.version_labels.items()}
                pol = mc.model_version_policy
                versions: List[int] = []
                if pol is not None and pol.specific is not None:
                    versions = [int(v) for v in pol.specific.versions]
                elif pol is not None and pol.latest is not None and \
                        pol.latest.num_versions > 1 and \
                        hasattr(self.cm.provider, "latest_version"):
                    # latest{num_versions: N}: the N newest versions
                    base = os.path.join(self.cm.provider.base_dir, mc.name) \
                        if hasattr(self.cm.provider, "base_dir") else None
                    found = []
                    if base and os.path.isdir(base):
                        for entry in os.listdir(base):
                            try:
                                found.append(int(entry))
                            except ValueError:
                                pass
                    versions = sorted(found)[-pol.latest.num_versions:]
                if not versions:
                    v = self._resolve_version(mc.name, 0)
                    versions = [v]
                for v in versions:
                    desired.append((mc.name, v))
        # fetch missing models to the disk cache in the background, then
        # the pool reload picks them up (declarative desired-state push)
        import threading
        for name, v in desired:
            if not self.cm.cache.contains(name, v):
                threading.Thread(target=self._prefetch, args=(name, v),
                                 daemon=True).start()
        self.cm.pool.reload(desired, self.cm._version_dir)
        return m.ReloadConfigResponse(status=m.StatusProto(
            error_code=m.ERROR_OK))

    def _prefetch(self, name: str, version: int) -> None:
        try:
            self.cm.ensure_loaded(name, version)
        except Exception:       # noqa: BLE001
            log.warning("prefetch of %s:%d failed", name, version,
                        exc_info=True)

    # -- SessionRun --------------------------------------------------------
    def session_run(self, req: m.SessionRunRequest) -> m.SessionRunResponse:
        spec = req.model_spec or m.ModelSpec()
        model, version = self.get_model(spec.name, spec.version_value(),
                                        spec.version_label)
        feeds = {nt.name: tensorproto_to_numpy(nt.tensor)
                 for nt in req.feed}
        try:
            out = model.session_run(feeds, list(req.fetch))
        except ModelExecError as e:
            raise ServingError(str(e))
        resp = m.SessionRunResponse(model_spec=m.ModelSpec(
            name=spec.name, version=m.Int64Value(value=version)))
        for name, arr in out.items():
            resp.tensor.append(m.NamedTensorProto(
                name=name, tensor=numpy_to_tensorproto(
                    np.ascontiguousarray(arr, dtype=np.float32)
                    if arr.dtype.kind == "f" else arr)))
        return resp


def _fast_fallback_cls():
    try:
        from ..engine import _tfsc_engine as ext
        return ext.FastFallback
    except Exception:       # noqa: BLE001
        return None


def _peek_fn():
    """C++ model_spec peek (~0.4 us vs ~11 us for the Python wire
    peek) when the engine extension is importable."""
    try:
        from ..engine import _tfsc_engine as ext
        return ext.peek_spec
    except Exception:       # noqa: BLE001
        return None


def _predict_bytes(self, data: bytes) -> bytes:
    """Raw-request Predict: C++ end-to-end fast path when the model's
    engine has a registered fast context for the request's batch bucket;
    Python decode/execute/encode otherwise (also the warm-up path that
    builds and registers the context)."""
    peek = _peek_fn()
    if peek is not None:
        name, ver, label = peek(data)
        ver = int(ver) if ver else 0
    else:
        spec = m.peek_model_spec(data)
        name, ver, label = (spec.name, spec.version_value(),
                            spec.version_label)
    model, _version = self.get_model(name, ver, label)
    gpu = getattr(model, "_gpu", None)
    if gpu is not None and model._batcher is None:
        fb = _fast_fallback_cls()
        if fb is not None:
            from ..utils import metrics as mt
            try:
                with mt.engine_predict_duration.labels(
                        model.name, str(model.version),
                        model.device).time():
                    return gpu.fast_predict(data)
            except fb:
                pass
            except Exception as e:      # noqa: BLE001
                from ..engine.gpu import ModelReleasedError
                if not isinstance(e, ModelReleasedError):
                    raise
                model, _version = self.get_model(name, ver)
                gpu = getattr(model, "_gpu", None)
                if gpu is not None:
                    try:
                        return gpu.fast_predict(data)
                    except fb:
                        pass
    req = m.PredictRequest.decode(data)
    return self.predict(req).encode()


LocalServingHandler.predict_bytes = _predict_bytes
