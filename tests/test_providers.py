"""Provider tests: disk (numeric version matching, per reference
diskmodelprovider_test.go), and S3/AzBlob against local HTTP mock stores
(the reference shipped NO tests for S3/AzBlob — SURVEY.md §4)."""
import http.server
import os
import threading
import urllib.parse

import pytest

from tfservingcache_amd.cachemanager.modelprovider import ModelNotFoundError
from tfservingcache_amd.cachemanager.providers import DiskModelProvider
from tfservingcache_amd.cachemanager.providers.azblob import AZBlobModelProvider
from tfservingcache_amd.cachemanager.providers.s3 import S3ModelProvider


def make_dummy_model(base, name, version_dir):
    d = os.path.join(base, name, version_dir)
    os.makedirs(os.path.join(d, "variables"), exist_ok=True)
    os.makedirs(os.path.join(d, "assets"), exist_ok=True)
    with open(os.path.join(d, "saved_model.pb"), "wb") as f:
        f.write(b"x" * 1000)
    with open(os.path.join(d, "variables", "variables.data"), "wb") as f:
        f.write(b"y" * 500)


# ---------------------------------------------------------------------------
# disk provider (diskmodelprovider_test.go:33-87)
# ---------------------------------------------------------------------------
def test_disk_selects_correct_version_among_decoys(tmp_path):
    base = str(tmp_path / "repo")
    make_dummy_model(base, "m", "41")
    make_dummy_model(base, "m", "42")
    make_dummy_model(base, "m", "43")
    prov = DiskModelProvider(base)
    model = prov.load_model("m", 42, str(tmp_path / "cache"))
    assert model.version == 42
    assert os.path.exists(str(tmp_path / "cache" / "m" / "42" /
                              "saved_model.pb"))


def test_disk_zero_padded_versions_match_numerically(tmp_path):
    base = str(tmp_path / "repo")
    make_dummy_model(base, "m", "000000042")
    prov = DiskModelProvider(base)
    model = prov.load_model("m", 42, str(tmp_path / "cache"))
    assert model.version == 42
    # recursive size (reference stat'ed the dir inode — fixed, §2.3)
    assert prov.model_size("m", 42) == 1500
    with pytest.raises(ModelNotFoundError):
        prov.load_model("m", 7, str(tmp_path / "cache"))
    assert prov.latest_version("m") == 42


# ---------------------------------------------------------------------------
# mock object stores
# ---------------------------------------------------------------------------
class _ObjectStoreHandler(http.server.BaseHTTPRequestHandler):
    objects = {}     # key -> bytes
    mode = "s3"

    def log_message(self, *a):
        pass

    def do_GET(self):
        parsed = urllib.parse.urlparse(self.path)
        qs = dict(urllib.parse.parse_qsl(parsed.query))
        if self.mode == "s3" and qs.get("list-type") == "2":
            prefix = qs.get("prefix", "")
            keys = sorted(k for k in self.objects if k.startswith(prefix))
            body = ['<?xml version="1.0"?><ListBucketResult>']
            for k in keys:
                body.append(f"<Contents><Key>{k}</Key>"
                            f"<Size>{len(self.objects[k])}</Size>"
                            f"</Contents>")
            body.append("<IsTruncated>false</IsTruncated>"
                        "</ListBucketResult>")
            self._send(200, "".join(body).encode())
            return
        if self.mode == "az" and qs.get("comp") == "list":
            prefix = qs.get("prefix", "")
            keys = sorted(k for k in self.objects if k.startswith(prefix))
            body = ['<?xml version="1.0"?><EnumerationResults><Blobs>']
            for k in keys:
                body.append(
                    f"<Blob><Name>{k}</Name><Properties>"
                    f"<Content-Length>{len(self.objects[k])}"
                    f"</Content-Length></Properties></Blob>")
            body.append("</Blobs><NextMarker/></EnumerationResults>")
            self._send(200, "".join(body).encode())
            return
        # object get: path = /bucket/key or /container/key
        path = urllib.parse.unquote(parsed.path).lstrip("/")
        _, _, key = path.partition("/")
        if key in self.objects:
            self._send(200, self.objects[key])
        else:
            self._send(404, b"not found")

    def _send(self, code, body):
        self.send_response(code)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)


@pytest.fixture()
def object_store():
    _ObjectStoreHandler.objects = {
        "models/m/1/saved_model.pb": b"a" * 100,
        "models/m/1/variables/variables.data": b"b" * 50,
        "models/m/3/saved_model.pb": b"c" * 70,
    }
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0),
                                          _ObjectStoreHandler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{srv.server_address[1]}"
    srv.shutdown()


def test_s3_provider(tmp_path, object_store):
    _ObjectStoreHandler.mode = "s3"
    prov = S3ModelProvider(bucket="bkt", base_path="models",
                           endpoint_url=object_store)
    assert prov.check()
    assert prov.model_size("m", 1) == 150
    model = prov.load_model("m", 1, str(tmp_path))
    assert model.size_on_disk == 150
    assert (tmp_path / "m" / "1" / "variables" /
            "variables.data").read_bytes() == b"b" * 50
    assert prov.latest_version("m") == 3
    with pytest.raises(ModelNotFoundError):
        prov.model_size("m", 9)


def test_azblob_provider(tmp_path, object_store):
    _ObjectStoreHandler.mode = "az"
    prov = AZBlobModelProvider(account="acct", container="cont",
                               base_path="models", endpoint=object_store)
    assert prov.check()
    assert prov.model_size("m", 1) == 150
    model = prov.load_model("m", 1, str(tmp_path))
    assert model.size_on_disk == 150
    assert prov.latest_version("m") == 3
    with pytest.raises(ModelNotFoundError):
        prov.load_model("m", 9, str(tmp_path))


def test_sigv4_signature_shape():
    from tfservingcache_amd.cachemanager.providers.s3 import SigV4Signer
    signer = SigV4Signer("AKID", "SECRET", "us-east-1")
    headers = signer.sign("GET", "https://bkt.s3.amazonaws.com/key?a=1")
    assert headers["Authorization"].startswith("AWS4-HMAC-SHA256 Credential=AKID/")
    assert "SignedHeaders=" in headers["Authorization"]
    assert "x-amz-date" in headers
