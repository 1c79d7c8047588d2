"""ModelProvider interface (reference: pkg/cachemanager/modelprovider.go:3-7)."""
from __future__ import annotations

import abc

from .lrucache import Model


class ModelProviderError(Exception):
    pass


class ModelNotFoundError(ModelProviderError):
    pass


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
