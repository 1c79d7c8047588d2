"""Azure Blob model provider (plain HTTP + SharedKey; no Azure SDK).

Behavior mirrors the reference's AZBlobModelProvider
(pkg/cachemanager/modelproviders/azblobmodelprovider/azblobmodelprovider.go):
  * blobs under <container>/<basePath>/<model>/<version>/...;
  * load_model lists the prefix (List Blobs flat, marker-paginated —
    azblobmodelprovider.go:125-172) and downloads each blob;
  * errors when the prefix lists zero blobs (azblobmodelprovider.go:157-159);
  * model_size sums Content-Length; check() lists one blob.

Auth: account SharedKey (the reference's NewSharedKeyCredential path,
azblobmodelprovider.go:32-47) or anonymous/custom endpoint (Azurite and
the HTTP mock in tests).
"""
from __future__ import annotations

import base64
import datetime
import hashlib
import hmac
import os
import urllib.parse
import xml.etree.ElementTree as ET
from typing import Iterator, Optional, Tuple

import requests

from ..lrucache import Model, dir_size
from ..modelprovider import ModelNotFoundError, ModelProvider, \
    ModelProviderError, validate_model_name


class AZBlobModelProvider(ModelProvider):
    def __init__(self, account: str, container: str, base_path: str = "",
                 account_key: str = "", endpoint: Optional[str] = None):
        self.account = account
        self.container = container
        self.base_path = base_path.strip("/")
        self.endpoint = (endpoint.rstrip("/") if endpoint
                         else f"https://{account}.blob.core.windows.net")
        self.key = account_key
        self._session = requests.Session()

    # -- SharedKey auth ----------------------------------------------------
    def _auth_headers(self, method: str, url: str) -> dict:
        if not self.key:
            return {}
        parsed = urllib.parse.urlparse(url)
        now = datetime.datetime.utcnow().strftime(
            "%a, %d %b %Y %H:%M:%S GMT")
        headers = {"x-ms-date": now, "x-ms-version": "2020-10-02"}
        canon_headers = "".join(
            f"{k}:{v}\n" for k, v in sorted(headers.items()))
        canon_resource = f"/{self.account}{parsed.path}"
        if parsed.query:
            for k, v in sorted(urllib.parse.parse_qsl(parsed.query)):
                canon_resource += f"\n{k}:{v}"
        string_to_sign = (f"{method}\n\n\n\n\n\n\n\n\n\n\n\n"
                          f"{canon_headers}{canon_resource}")
        sig = base64.b64encode(hmac.new(
            base64.b64decode(self.key), string_to_sign.encode("utf-8"),
            hashlib.sha256).digest()).decode()
        headers["Authorization"] = f"SharedKey {self.account}:{sig}"
        return headers

    def _get(self, url: str, stream: bool = False) -> requests.Response:
        return self._session.get(url, headers=self._auth_headers("GET", url),
                                 timeout=60, stream=stream)

    # -- listing -----------------------------------------------------------
    def _prefix(self, model_name: str, version: int) -> str:
        parts = [p for p in (self.base_path, model_name, str(version)) if p]
        return "/".join(parts) + "/"

    def _list_blobs(self, prefix: str) -> Iterator[Tuple[str, int]]:
        marker = ""
        while True:
            q = ("restype=container&comp=list&prefix=" +
                 urllib.parse.quote(prefix, safe=""))
            if marker:
                q += "&marker=" + urllib.parse.quote(marker, safe="")
            url = f"{self.endpoint}/{self.container}?{q}"
            r = self._get(url)
            if r.status_code != 200:
                raise ModelProviderError(
                    f"azblob list failed: {r.status_code} {r.text[:200]}")
            root = ET.fromstring(r.content)
            for b in root.iter("Blob"):
                name = b.findtext("Name")
                size = int(b.findtext("Properties/Content-Length") or 0)
                yield name, size
            marker = root.findtext("NextMarker") or ""
            if not marker:
                return

    # -- ModelProvider -----------------------------------------------------
    def load_model(self, model_name: str, version: int,
                   dest_base_dir: str) -> Model:
        validate_model_name(model_name)
        prefix = self._prefix(model_name, version)
        rel = os.path.join(model_name, str(version))
        dst_root = os.path.join(dest_base_dir, rel)
        n = 0
        for name, _size in self._list_blobs(prefix):
            sub = name[len(prefix):]
            if not sub or sub.endswith("/"):
                continue
            if ".." in sub.split("/") or sub.startswith("/"):
                raise ModelProviderError(
                    f"refusing traversal in blob name: {name!r}")
            dst = os.path.join(dst_root, sub)
            os.makedirs(os.path.dirname(dst), exist_ok=True)
            url = f"{self.endpoint}/{self.container}/" + \
                urllib.parse.quote(name)
            r = self._get(url, stream=True)
            if r.status_code != 200:
                raise ModelProviderError(
                    f"azblob get {name} failed: {r.status_code}")
            with open(dst, "wb") as f:
                for chunk in r.iter_content(1 << 20):
                    f.write(chunk)
            n += 1
        if n == 0:
            # reference errors on zero blobs (azblobmodelprovider.go:157)
            raise ModelNotFoundError(
                f"no blobs under {self.container}/{prefix}")
        return Model(name=model_name, version=version, path=rel,
                     size_on_disk=dir_size(dst_root))

    def model_size(self, model_name: str, version: int) -> int:
        total, found = 0, False
        for _name, size in self._list_blobs(
                self._prefix(model_name, version)):
            total += size
            found = True
        if not found:
            raise ModelNotFoundError(
                f"model {model_name}:{version} not in container")
        return total

    def check(self) -> bool:
        try:
            next(self._list_blobs(self.base_path + "/"
                                  if self.base_path else ""), None)
            return True
        except Exception:       # noqa: BLE001
            return False

    def latest_version(self, model_name: str) -> Optional[int]:
        parts = [p for p in (self.base_path, model_name) if p]
        prefix = "/".join(parts) + "/"
        versions = set()
        for name, _ in self._list_blobs(prefix):
            sub = name[len(prefix):].split("/", 1)[0]
            try:
                versions.add(int(sub))
            except ValueError:
                continue
        return max(versions) if versions else None
