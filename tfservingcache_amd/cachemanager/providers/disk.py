"""Disk model provider.

Mirrors the reference's DiskModelProvider behavior
(/root/reference/pkg/cachemanager/modelproviders/diskmodelprovider/diskmodelprovider.go):
  * version directories are matched NUMERICALLY, so "000000042" serves
    version 42 (findSrcPathForModel, diskmodelprovider.go:46-69);
  * load_model recursively copies baseDir/<model>/<version-dir>/ into the
    cache dir — by HARDLINK when the cache shares a filesystem with the
    repo (eviction unlinks, refcounts keep the repo intact), byte copy
    otherwise;
  * model_size is the RECURSIVE content size (the reference stat'ed the
    directory inode — a known bug, SURVEY.md §2.3 — fixed here).
"""
from __future__ import annotations

import os
import shutil

from ..lrucache import Model, dir_size
from ..modelprovider import (ModelNotFoundError, ModelProvider,
                             validate_model_name)


def _link_or_copy(src: str, dst: str) -> None:
    try:
        os.link(src, dst)
    except OSError:                    # cross-device / FS without links
        shutil.copy2(src, dst)


class DiskModelProvider(ModelProvider):
    def __init__(self, base_dir: str):
        self.base_dir = base_dir

    def _find_src_dir(self, model_name: str, version: int) -> str:
        validate_model_name(model_name)
        model_dir = os.path.join(self.base_dir, model_name)
        if not os.path.isdir(model_dir):
            raise ModelNotFoundError(f"model dir not found: {model_dir}")
        for entry in sorted(os.listdir(model_dir)):
            full = os.path.join(model_dir, entry)
            if not os.path.isdir(full):
                continue
            try:
                if int(entry) == version:
                    return full
            except ValueError:
                continue
        raise ModelNotFoundError(
            f"version {version} of model {model_name} not found")

    def load_model(self, model_name: str, version: int, dest_base_dir: str) -> Model:
        validate_model_name(model_name)
        src = self._find_src_dir(model_name, version)
        rel = os.path.join(model_name, str(version))
        dst = os.path.join(dest_base_dir, rel)
        if os.path.exists(dst):
            shutil.rmtree(dst)
        shutil.copytree(src, dst, copy_function=_link_or_copy)
        return Model(name=model_name, version=version, path=rel,
                     size_on_disk=dir_size(dst))

    def model_size(self, model_name: str, version: int) -> int:
        return dir_size(self._find_src_dir(model_name, version))

    def check(self) -> bool:
        return os.path.isdir(self.base_dir)

    def latest_version(self, model_name: str):
        validate_model_name(model_name)
        model_dir = os.path.join(self.base_dir, model_name)
        if not os.path.isdir(model_dir):
            return None
        versions = []
        for entry in os.listdir(model_dir):
            if os.path.isdir(os.path.join(model_dir, entry)):
                try:
                    versions.append(int(entry))
                except ValueError:
                    pass
        return max(versions) if versions else None
