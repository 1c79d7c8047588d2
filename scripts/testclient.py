#!/usr/bin/env python3
"""Manual gRPC test client (the reference's cmd/testclient/main.go:12-42
sends a Classify request against the proxy port; same here, plus
Predict).

    python scripts/testclient.py [host:port] [model] [version]
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import grpc  # noqa: E402
import numpy as np  # noqa: E402

from tfservingcache_amd.wire import messages as m  # noqa: E402
from tfservingcache_amd.wire.tensor import (numpy_to_tensorproto,  # noqa: E402
                                            tensorproto_to_numpy)


def main():
    target = sys.argv[1] if len(sys.argv) > 1 else "127.0.0.1:8100"
    model = sys.argv[2] if len(sys.argv) > 2 else "half_plus_two"
    version = int(sys.argv[3]) if len(sys.argv) > 3 else 0
    ch = grpc.insecure_channel(target)
    spec = m.ModelSpec(name=model,
                       version=m.Int64Value(value=version) if version
                       else None)

    # Classify with one tf.Example (like the reference's test client)
    ex = m.Example(features=m.Features(feature={
        "x": m.Feature(float_list=m.FloatList(value=[1.0]))}))
    classify = ch.unary_unary(
        "/tensorflow.serving.PredictionService/Classify",
        request_serializer=lambda r: r.encode(),
        response_deserializer=m.ClassificationResponse.decode)
    try:
        resp = classify(m.ClassificationRequest(
            model_spec=spec,
            input=m.Input(example_list=m.ExampleList(examples=[ex]))),
            timeout=30)
        print("Classify:", [(c.label, c.score) for c in
                            resp.result.classifications[0].classes])
    except grpc.RpcError as e:
        print("Classify failed:", e.code(), e.details())

    predict = ch.unary_unary(
        "/tensorflow.serving.PredictionService/Predict",
        request_serializer=lambda r: r.encode(),
        response_deserializer=m.PredictResponse.decode)
    try:
        resp = predict(m.PredictRequest(
            model_spec=spec,
            inputs={"x": numpy_to_tensorproto(
                np.array([1.0, 2.0, 5.0], dtype=np.float32))}), timeout=30)
        for name, tp in resp.outputs.items():
            print(f"Predict output {name}:",
                  tensorproto_to_numpy(tp).tolist())
    except grpc.RpcError as e:
        print("Predict failed:", e.code(), e.details())


if __name__ == "__main__":
    main()
