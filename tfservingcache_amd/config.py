"""Config loader — byte-compatible with the reference's viper setup.

Reads `config.yaml` from the working directory (same file format as
/root/reference/config.yaml), with environment-variable overrides using
the `TFSC_` prefix and `.` -> `_` key replacement, matching viper's
behavior in /root/reference/cmd/taskhandler/cfg.go:10-26
(e.g. `TFSC_SERVING_GRPCHOST` overrides `serving.grpcHost`).

Key lookup is case-insensitive (viper lower-cases keys), and missing keys
return type-appropriate zero values (GetString -> "", GetInt -> 0), which
the composition root relies on ("proxy disabled without serviceDiscovery",
main.go:103-105).
"""
from __future__ import annotations

import logging
import os
from typing import Any, Dict, List, Optional

import yaml

log = logging.getLogger("tfsc")

_DEFAULTS = {
    # cfg.go:65
    "healthprobe.modelname": "__TFSERVINGCACHE_PROBE_CHECK__",
    # MI355X additions (absent keys keep reference behavior)
    "engine.gpus": -1,            # -1 = all visible GPUs
    "engine.dtype": "bf16",
    "engine.maxbatch": 64,
    "engine.streamsppergpu": 2,
}


def _flatten(prefix: str, node: Any, out: Dict[str, Any]) -> None:
    if isinstance(node, dict):
        for k, v in node.items():
            key = f"{prefix}.{str(k).lower()}" if prefix else str(k).lower()
            _flatten(key, v, out)
        if prefix:
            out.setdefault(prefix, node)
    else:
        out[prefix] = node


class Config:
    def __init__(self, values: Optional[Dict[str, Any]] = None):
        self._values: Dict[str, Any] = {}
        self._defaults = dict(_DEFAULTS)
        if values:
            _flatten("", values, self._values)

    # -- loading -----------------------------------------------------------
    @classmethod
    def load(cls, path: str = "config.yaml") -> "Config":
        cfg = cls()
        if os.path.exists(path):
            with open(path) as f:
                data = yaml.safe_load(f) or {}
            _flatten("", data, cfg._values)
        else:
            log.info("No config file found. Reading from env vars")
        cfg.configure_logging()
        return cfg

    def configure_logging(self) -> None:
        level_name = (self.get_string("logging.level") or "info").lower()
        fmt = self.get_string("logging.format")
        level = {"panic": logging.CRITICAL, "fatal": logging.CRITICAL,
                 "warning": logging.WARNING, "debug": logging.DEBUG,
                 "info": logging.INFO}.get(level_name, logging.INFO)
        if fmt == "json":
            fmt_str = ('{"time":"%(asctime)s","level":"%(levelname)s",'
                       '"msg":"%(message)s"}')
        else:
            fmt_str = "%(asctime)s %(levelname)s %(name)s: %(message)s"
        logging.basicConfig(level=level, format=fmt_str)

    # -- lookup ------------------------------------------------------------
    def _raw(self, key: str) -> Any:
        key = key.lower()
        env_key = "TFSC_" + key.replace(".", "_").upper()
        if env_key in os.environ:
            return os.environ[env_key]
        if key in self._values:
            return self._values[key]
        return self._defaults.get(key)

    def is_set(self, key: str) -> bool:
        return self._raw(key) is not None

    def get(self, key: str) -> Any:
        return self._raw(key)

    def get_string(self, key: str) -> str:
        v = self._raw(key)
        return "" if v is None else str(v)

    def get_int(self, key: str) -> int:
        v = self._raw(key)
        if v is None or v == "":
            return 0
        return int(v)

    def get_float(self, key: str) -> float:
        v = self._raw(key)
        if v is None or v == "":
            return 0.0
        return float(v)

    def get_bool(self, key: str) -> bool:
        v = self._raw(key)
        if isinstance(v, bool):
            return v
        if v is None:
            return False
        return str(v).lower() in ("1", "true", "yes", "on")

    def get_dict(self, key: str) -> Dict[str, Any]:
        v = self._raw(key)
        return v if isinstance(v, dict) else {}

    def get_list(self, key: str) -> List[Any]:
        v = self._raw(key)
        if isinstance(v, list):
            return v
        if isinstance(v, str) and v:
            return [s.strip() for s in v.strip("[]").split(",") if s.strip()]
        return []

    def set(self, key: str, value: Any) -> None:
        self._values[key.lower()] = value

    def set_default(self, key: str, value: Any) -> None:
        self._defaults[key.lower()] = value
