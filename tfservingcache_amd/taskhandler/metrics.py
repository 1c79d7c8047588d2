"""Metrics merger — one scrape endpoint for cache + engine metrics.

The reference merges its own Prometheus registry with a live scrape of
TF Serving's metrics endpoint (pkg/taskhandler/metrics.go:16-53). The
engine here is in-process so its metrics live in the same registry; the
merger remains for an optional external scrape target (e.g. a ROCm/SMI
exporter running beside the node), preserving the reference's
single-endpoint behavior.
"""
from __future__ import annotations

import logging
from typing import Optional

import requests

from ..utils import metrics as mt

log = logging.getLogger("tfsc.metrics")


class MetricsMerger:
    def __init__(self, extra_scrape_url: Optional[str] = None,
                 timeout: float = 3.0):
        self.extra_url = extra_scrape_url
        self.timeout = timeout
        self._session = requests.Session()

    def render(self) -> bytes:
        own = mt.render()
        if not self.extra_url:
            return own
        try:
            r = self._session.get(self.extra_url, timeout=self.timeout)
            r.raise_for_status()
            extra = r.content
            if not extra.endswith(b"\n"):
                extra += b"\n"
            return own + extra
        except requests.RequestException:
            log.warning("merging external metrics from %s failed",
                        self.extra_url, exc_info=True)
            return own
