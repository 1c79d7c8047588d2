"""ModelProvider interface (reference: pkg/cachemanager/modelprovider.go:3-7)."""
from __future__ import annotations

import abc

from .lrucache import Model


class ModelProviderError(Exception):
    pass


class ModelNotFoundError(ModelProviderError):
    pass


class InvalidModelNameError(ModelProviderError):
    pass


def validate_model_name(name: str) -> str:
    """Reject model names that could traverse the filesystem when joined
    into cache/provider paths (gRPC model_spec.name arrives unfiltered;
    REST is regex-guarded but validate centrally anyway). Empty names,
    path separators and '..'/'.' components are refused before they can
    reach a provider's copytree or the LRU eviction's rmtree."""
    if not name or len(name) > 512:
        raise InvalidModelNameError(f"invalid model name: {name!r}")
    if "/" in name or "\\" in name or "\x00" in name:
        raise InvalidModelNameError(f"invalid model name: {name!r}")
    if name in (".", ".."):
        raise InvalidModelNameError(f"invalid model name: {name!r}")
    return name


class ModelProvider(abc.ABC):
    """Pluggable model store: fetches model files into the local cache dir."""

    @abc.abstractmethod
    def load_model(self, model_name: str, version: int, dest_base_dir: str) -> Model:
        """Copy <model>/<version> into dest_base_dir/<model>/<version>,
        returning its Model record (path relative to dest_base_dir)."""

    @abc.abstractmethod
    def model_size(self, model_name: str, version: int) -> int:
        """Byte size of the model in the store (recursive)."""

    @abc.abstractmethod
    def check(self) -> bool:
        """Health check of the backing store."""
