"""SavedModel warmup — TF Serving's assets.extra mechanism.

TF Serving executes the PredictionLog records in
`assets.extra/tf_serving_warmup_requests` (a TFRecord file) when a model
loads, so the first real request doesn't pay lazy-initialization costs.
Here that means: batch-bucket ExecContexts get built, hipGraphs get
captured and the C++ fast path gets registered BEFORE the model is
marked AVAILABLE.

TFRecord framing: [uint64 len][u32 masked-crc(len)][data][u32
masked-crc(data)] — lengths are honored, CRCs are not verified (no
crc32c in this environment's stdlib).
"""
from __future__ import annotations

import logging
import os
import struct
from typing import Iterator, List

from ..wire import messages as m
from ..wire.pb import Message

log = logging.getLogger("tfsc.warmup")

WARMUP_PATH = os.path.join("assets.extra", "tf_serving_warmup_requests")


class PredictLog(Message):
    FIELDS = [
        ("request", 1, "message", dict(msg_cls=m.PredictRequest)),
        ("response", 2, "message", dict(msg_cls=m.PredictResponse)),
    ]


class ClassifyLog(Message):
    FIELDS = [
        ("request", 1, "message", dict(msg_cls=m.ClassificationRequest)),
    ]


class RegressLog(Message):
    FIELDS = [
        ("request", 1, "message", dict(msg_cls=m.RegressionRequest)),
    ]


class PredictionLog(Message):
    # prediction_log.proto: log_metadata=1; oneof {classify=2, regress=3,
    # multi_inference=4, session_run=5, predict=6}
    FIELDS = [
        ("classify_log", 2, "message", dict(msg_cls=ClassifyLog)),
        ("regress_log", 3, "message", dict(msg_cls=RegressLog)),
        ("predict_log", 6, "message", dict(msg_cls=PredictLog)),
    ]


def read_tfrecords(path: str) -> Iterator[bytes]:
    with open(path, "rb") as f:
        while True:
            header = f.read(8)
            if len(header) < 8:
                return
            (length,) = struct.unpack("<Q", header)
            if length > 1 << 30:        # corrupt length field: stop
                log.warning("warmup record length %d exceeds 1 GiB; "
                            "treating file as corrupt", length)
                return
            f.read(4)                   # length crc (unverified)
            data = f.read(length)
            if len(data) < length:
                return
            f.read(4)                   # data crc (unverified)
            yield data


def write_tfrecord(path: str, records: List[bytes]) -> None:
    """Writer for fixtures/tests (CRC fields zeroed)."""
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with open(path, "wb") as f:
        for rec in records:
            f.write(struct.pack("<Q", len(rec)))
            f.write(b"\0\0\0\0")
            f.write(rec)
            f.write(b"\0\0\0\0")


def run_warmup(model, version_dir: str, max_requests: int = 64) -> int:
    """Execute the model's warmup requests (if any). Returns the number
    executed. Failures are logged, not fatal (TF Serving behavior)."""
    path = os.path.join(version_dir, WARMUP_PATH)
    if not os.path.exists(path):
        return 0
    from ..wire.tensor import tensorproto_to_numpy
    n = 0
    try:
        for raw in read_tfrecords(path):
            if n >= max_requests:
                break
            plog = PredictionLog.decode(raw)
            req = None
            if plog.predict_log is not None:
                req = plog.predict_log.request
            if req is None:
                continue
            inputs = {}
            for alias, tp in req.inputs.items():
                inputs[alias] = tensorproto_to_numpy(tp)
            try:
                model.predict(inputs,
                              list(req.output_filter) or None)
                n += 1
            except Exception:       # noqa: BLE001
                log.warning("warmup request %d failed", n, exc_info=True)
    except Exception:       # noqa: BLE001
        log.warning("reading warmup records failed", exc_info=True)
    if n:
        log.info("executed %d warmup requests from %s", n, path)
    return n
