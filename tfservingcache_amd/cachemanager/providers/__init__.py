from .disk import DiskModelProvider  # noqa: F401
