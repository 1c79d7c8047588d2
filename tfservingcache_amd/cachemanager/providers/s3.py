"""S3 model provider (plain HTTP + SigV4; no AWS SDK in this env).

Behavior mirrors the reference's S3ModelProvider
(pkg/cachemanager/modelproviders/s3modelprovider/s3modelprovider.go):
  * objects live under  s3://<bucket>/<basePath>/<model>/<version>/...
    (getKeyForModel, s3modelprovider.go:161-170);
  * load_model walks the prefix with paginated ListObjectsV2 and
    downloads every object, mkdir-ing subpaths (modelObjectApply,
    s3modelprovider.go:124-159);
  * model_size sums the listed object sizes (s3modelprovider.go:108-122);
  * check() lists with MaxKeys=1 (s3modelprovider.go:172-180).

Credentials come from env (AWS_ACCESS_KEY_ID/AWS_SECRET_ACCESS_KEY) or
constructor args; anonymous access is used when absent. A custom
endpoint_url supports MinIO-style services (and the HTTP mock in tests).
"""
from __future__ import annotations

import datetime
import hashlib
import hmac
import os
import urllib.parse
import xml.etree.ElementTree as ET
from typing import Dict, Iterator, Optional, Tuple

import requests

from ..lrucache import Model, dir_size
from ..modelprovider import ModelNotFoundError, ModelProvider, \
    ModelProviderError, validate_model_name


def _sha256(data: bytes) -> str:
    return hashlib.sha256(data).hexdigest()


class SigV4Signer:
    def __init__(self, access_key: str, secret_key: str, region: str,
                 service: str = "s3"):
        self.access_key = access_key
        self.secret_key = secret_key
        self.region = region
        self.service = service

    def sign(self, method: str, url: str,
             headers: Optional[Dict[str, str]] = None,
             payload: bytes = b"") -> Dict[str, str]:
        parsed = urllib.parse.urlparse(url)
        now = datetime.datetime.utcnow()
        amz_date = now.strftime("%Y%m%dT%H%M%SZ")
        datestamp = now.strftime("%Y%m%d")
        payload_hash = _sha256(payload)
        headers = dict(headers or {})
        headers["host"] = parsed.netloc
        headers["x-amz-date"] = amz_date
        headers["x-amz-content-sha256"] = payload_hash

        canonical_qs = "&".join(
            f"{k}={urllib.parse.quote(v, safe='')}"
            for k, v in sorted(urllib.parse.parse_qsl(
                parsed.query, keep_blank_values=True)))
        signed_names = sorted(headers)
        canonical_headers = "".join(
            f"{k}:{headers[k].strip()}\n" for k in signed_names)
        signed_headers = ";".join(signed_names)
        canonical_request = "\n".join([
            method, urllib.parse.quote(parsed.path or "/"), canonical_qs,
            canonical_headers, signed_headers, payload_hash])
        scope = f"{datestamp}/{self.region}/{self.service}/aws4_request"
        string_to_sign = "\n".join([
            "AWS4-HMAC-SHA256", amz_date, scope,
            _sha256(canonical_request.encode())])

        def _hmac(key, msg):
            return hmac.new(key, msg.encode(), hashlib.sha256).digest()

        k = _hmac(("AWS4" + self.secret_key).encode(), datestamp)
        k = _hmac(k, self.region)
        k = _hmac(k, self.service)
        k = _hmac(k, "aws4_request")
        signature = hmac.new(k, string_to_sign.encode(),
                             hashlib.sha256).hexdigest()
        headers["Authorization"] = (
            f"AWS4-HMAC-SHA256 Credential={self.access_key}/{scope}, "
            f"SignedHeaders={signed_headers}, Signature={signature}")
        del headers["host"]      # requests sets it
        return headers


class S3ModelProvider(ModelProvider):
    def __init__(self, bucket: str, base_path: str = "",
                 endpoint_url: Optional[str] = None,
                 region: str = "us-east-1",
                 access_key: Optional[str] = None,
                 secret_key: Optional[str] = None):
        self.bucket = bucket
        self.base_path = base_path.strip("/")
        self.endpoint = (endpoint_url.rstrip("/") if endpoint_url
                         else f"https://{bucket}.s3.{region}.amazonaws.com")
        self.path_style = endpoint_url is not None
        self.region = region
        access_key = access_key or os.environ.get("AWS_ACCESS_KEY_ID", "")
        secret_key = secret_key or os.environ.get("AWS_SECRET_ACCESS_KEY", "")
        self.signer = (SigV4Signer(access_key, secret_key, region)
                       if access_key else None)
        self._session = requests.Session()

    # -- HTTP helpers ------------------------------------------------------
    def _url(self, key: str = "", query: str = "") -> str:
        path = f"/{self.bucket}" if self.path_style else ""
        if key:
            path += "/" + urllib.parse.quote(key)
        elif self.path_style:
            path += "/"
        return f"{self.endpoint}{path}" + (f"?{query}" if query else "")

    def _get(self, url: str, stream: bool = False) -> requests.Response:
        headers = {}
        if self.signer:
            headers = self.signer.sign("GET", url)
        r = self._session.get(url, headers=headers, timeout=60,
                              stream=stream)
        return r

    # -- listing -----------------------------------------------------------
    def _key_prefix(self, model_name: str, version: int) -> str:
        parts = [p for p in (self.base_path, model_name, str(version)) if p]
        return "/".join(parts) + "/"

    def _list_objects(self, prefix: str,
                      max_keys: int = 1000) -> Iterator[Tuple[str, int]]:
        token = None
        while True:
            q = (f"list-type=2&max-keys={max_keys}"
                 f"&prefix={urllib.parse.quote(prefix, safe='')}")
            if token:
                q += f"&continuation-token={urllib.parse.quote(token, safe='')}"
            r = self._get(self._url(query=q))
            if r.status_code != 200:
                raise ModelProviderError(
                    f"S3 list failed: {r.status_code} {r.text[:200]}")
            root = ET.fromstring(r.content)
            ns = ""
            if root.tag.startswith("{"):
                ns = root.tag[:root.tag.index("}") + 1]
            for c in root.findall(f"{ns}Contents"):
                key = c.findtext(f"{ns}Key")
                size = int(c.findtext(f"{ns}Size") or 0)
                yield key, size
            truncated = (root.findtext(f"{ns}IsTruncated") or "false"
                         ) == "true"
            token = root.findtext(f"{ns}NextContinuationToken")
            if not truncated or not token:
                return

    # -- ModelProvider -----------------------------------------------------
    def load_model(self, model_name: str, version: int,
                   dest_base_dir: str) -> Model:
        validate_model_name(model_name)
        prefix = self._key_prefix(model_name, version)
        rel = os.path.join(model_name, str(version))
        dst_root = os.path.join(dest_base_dir, rel)
        n = 0
        for key, _size in self._list_objects(prefix):
            sub = key[len(prefix):]
            if not sub or sub.endswith("/"):
                continue
            if ".." in sub.split("/") or sub.startswith("/"):
                raise ModelProviderError(
                    f"refusing traversal in object key: {key!r}")
            dst = os.path.join(dst_root, sub)
            os.makedirs(os.path.dirname(dst), exist_ok=True)
            r = self._get(self._url(key), stream=True)
            if r.status_code != 200:
                raise ModelProviderError(
                    f"S3 get {key} failed: {r.status_code}")
            with open(dst, "wb") as f:
                for chunk in r.iter_content(1 << 20):
                    f.write(chunk)
            n += 1
        if n == 0:
            raise ModelNotFoundError(
                f"no objects under s3://{self.bucket}/{prefix}")
        return Model(name=model_name, version=version, path=rel,
                     size_on_disk=dir_size(dst_root))

    def model_size(self, model_name: str, version: int) -> int:
        total = 0
        found = False
        for _key, size in self._list_objects(
                self._key_prefix(model_name, version)):
            total += size
            found = True
        if not found:
            raise ModelNotFoundError(
                f"model {model_name}:{version} not in bucket")
        return total

    def check(self) -> bool:
        try:
            next(self._list_objects(self.base_path + "/"
                                    if self.base_path else "", max_keys=1),
                 None)
            return True
        except Exception:       # noqa: BLE001
            return False

    def latest_version(self, model_name: str) -> Optional[int]:
        parts = [p for p in (self.base_path, model_name) if p]
        prefix = "/".join(parts) + "/"
        versions = set()
        for key, _ in self._list_objects(prefix):
            sub = key[len(prefix):].split("/", 1)[0]
            try:
                versions.add(int(sub))
            except ValueError:
                continue
        return max(versions) if versions else None
