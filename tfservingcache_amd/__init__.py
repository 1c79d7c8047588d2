"""tfservingcache_amd — MI355X-native multi-model serving cache.

A from-scratch rebuild of the capabilities of mKaloer/TFServingCache
(Go sidecar + router over external TF Serving) as an MI355X-first
framework: the TF-Serving-compatible REST/gRPC front-end and the
consistent-hash routing tier are retained at the wire level, while the
compute tier is an in-process CDNA4 HIP inference engine with a per-GPU
HBM3E model pool instead of an external tensorflow_model_server.
"""

__version__ = "0.1.0"
