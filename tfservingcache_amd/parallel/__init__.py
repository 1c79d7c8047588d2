from .replica import ReplicaPlane, clone_gpu_model  # noqa: F401
