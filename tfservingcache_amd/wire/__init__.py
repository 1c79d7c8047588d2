from . import messages, pb, tensor  # noqa: F401
