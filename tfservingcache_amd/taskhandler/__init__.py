from .cluster import ClusterConnection, model_key  # noqa: F401
from .metrics import MetricsMerger  # noqa: F401
from .ring import ConsistentHashRing  # noqa: F401
