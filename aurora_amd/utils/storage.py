"""Object storage manager: file backend + S3-compatible backend.

Rebuilds the reference's storage layer (reference: utils/storage/
storage.py — S3-compatible manager, SeaweedFS default, AGENTS.md:39-45).
The S3 backend implements AWS SigV4 signing natively (no boto in the
image) over an injectable ``send`` callable, so request construction and
signing are fixture-testable without a server; SeaweedFS/MinIO/AWS all
speak this dialect. Selection: STORAGE_BACKEND=file|s3 (+ S3_ENDPOINT,
S3_BUCKET, S3_ACCESS_KEY, S3_SECRET_KEY, S3_REGION).
"""
from __future__ import annotations

import datetime as _dt
import hashlib
import hmac
import os
import urllib.parse
from typing import Callable, List, Optional


class StorageBackend:
    def put(self, key: str, data: bytes) -> None:
        raise NotImplementedError

    def get(self, key: str) -> Optional[bytes]:
        raise NotImplementedError

    def delete(self, key: str) -> bool:
        raise NotImplementedError

    def list(self, prefix: str = "") -> List[str]:
        raise NotImplementedError


class FileStorage(StorageBackend):
    def __init__(self, root: str):
        self.root = root
        os.makedirs(root, exist_ok=True)

    def _path(self, key: str) -> str:
        safe = key.replace("..", "_").lstrip("/")
        return os.path.join(self.root, safe)

    def put(self, key: str, data: bytes) -> None:
        p = self._path(key)
        os.makedirs(os.path.dirname(p) or self.root, exist_ok=True)
        with open(p, "wb") as f:
            f.write(data)

    def get(self, key: str) -> Optional[bytes]:
        try:
            with open(self._path(key), "rb") as f:
                return f.read()
        except OSError:
            return None

    def delete(self, key: str) -> bool:
        try:
            os.remove(self._path(key))
            return True
        except OSError:
            return False

    def list(self, prefix: str = "") -> List[str]:
        out = []
        for root, _d, files in os.walk(self.root):
            for fn in files:
                rel = os.path.relpath(os.path.join(root, fn), self.root)
                if rel.startswith(prefix):
                    out.append(rel)
        return sorted(out)


# ------------------------------------------------------------------ SigV4


def _sha256(b: bytes) -> str:
    return hashlib.sha256(b).hexdigest()


def _hmac(key: bytes, msg: str) -> bytes:
    return hmac.new(key, msg.encode(), hashlib.sha256).digest()


def sigv4_headers(method: str, url: str, region: str, access_key: str,
                  secret_key: str, payload: bytes = b"",
                  service: str = "s3",
                  now: Optional[_dt.datetime] = None) -> dict:
    """AWS Signature V4 for a single request (the subset S3 needs:
    host + x-amz-date + x-amz-content-sha256 signed headers)."""
    u = urllib.parse.urlsplit(url)
    t = now or _dt.datetime.now(_dt.timezone.utc)
    amz_date = t.strftime("%Y%m%dT%H%M%SZ")
    datestamp = t.strftime("%Y%m%d")
    payload_hash = _sha256(payload)
    canonical_query = "&".join(
        f"{urllib.parse.quote(k, safe='')}={urllib.parse.quote(v, safe='')}"
        for k, v in sorted(urllib.parse.parse_qsl(u.query,
                                                  keep_blank_values=True)))
    headers = {"host": u.netloc, "x-amz-content-sha256": payload_hash,
               "x-amz-date": amz_date}
    signed = ";".join(sorted(headers))
    canonical = "\n".join([
        method.upper(),
        urllib.parse.quote(u.path or "/", safe="/"),
        canonical_query,
        "".join(f"{k}:{headers[k]}\n" for k in sorted(headers)),
        signed,
        payload_hash,
    ])
    scope = f"{datestamp}/{region}/{service}/aws4_request"
    string_to_sign = "\n".join(["AWS4-HMAC-SHA256", amz_date, scope,
                                _sha256(canonical.encode())])
    k = _hmac(("AWS4" + secret_key).encode(), datestamp)
    k = _hmac(k, region)
    k = _hmac(k, service)
    k = _hmac(k, "aws4_request")
    signature = hmac.new(k, string_to_sign.encode(), hashlib.sha256).hexdigest()
    return {
        "Host": u.netloc,
        "x-amz-date": amz_date,
        "x-amz-content-sha256": payload_hash,
        "Authorization": (f"AWS4-HMAC-SHA256 Credential={access_key}/{scope}, "
                          f"SignedHeaders={signed}, Signature={signature}"),
    }


class S3Storage(StorageBackend):
    """S3-compatible backend (SeaweedFS / MinIO / AWS). ``send`` is an
    injectable callable(method, url, headers, body) -> (status, body);
    the default uses httpx synchronously."""

    def __init__(self, endpoint: str, bucket: str, access_key: str,
                 secret_key: str, region: str = "us-east-1",
                 send: Optional[Callable] = None):
        self.endpoint = endpoint.rstrip("/")
        self.bucket = bucket
        self.access_key = access_key
        self.secret_key = secret_key
        self.region = region
        self._send = send or self._http_send

    @staticmethod
    def _http_send(method: str, url: str, headers: dict, body: bytes):
        import httpx
        r = httpx.request(method, url, headers=headers, content=body,
                          timeout=30.0)
        return r.status_code, r.content

    def _url(self, key: str, query: str = "") -> str:
        path = f"/{self.bucket}/{urllib.parse.quote(key.lstrip('/'))}"
        return self.endpoint + path + (f"?{query}" if query else "")

    def _request(self, method: str, url: str, body: bytes = b""):
        headers = sigv4_headers(method, url, self.region, self.access_key,
                                self.secret_key, payload=body)
        return self._send(method, url, headers, body)

    def put(self, key: str, data: bytes) -> None:
        status, body = self._request("PUT", self._url(key), data)
        if status >= 300:
            raise IOError(f"S3 PUT {key}: HTTP {status}: {body[:200]!r}")

    def get(self, key: str) -> Optional[bytes]:
        status, body = self._request("GET", self._url(key))
        if status == 404:
            return None
        if status >= 300:
            raise IOError(f"S3 GET {key}: HTTP {status}")
        return body

    def delete(self, key: str) -> bool:
        status, _ = self._request("DELETE", self._url(key))
        return status < 300

    def list(self, prefix: str = "") -> List[str]:
        q = urllib.parse.urlencode({"list-type": "2", "prefix": prefix})
        url = self.endpoint + f"/{self.bucket}?{q}"
        status, body = self._request("GET", url)
        if status >= 300:
            raise IOError(f"S3 LIST: HTTP {status}")
        import re
        return re.findall(r"<Key>([^<]+)</Key>", body.decode("utf-8",
                                                             "replace"))


_storage: Optional[StorageBackend] = None


def storage_manager() -> StorageBackend:
    global _storage
    if _storage is None:
        kind = os.environ.get("STORAGE_BACKEND", "file")
        if kind == "s3":
            _storage = S3Storage(
                endpoint=os.environ.get("S3_ENDPOINT", "http://127.0.0.1:8333"),
                bucket=os.environ.get("S3_BUCKET", "aurora"),
                access_key=os.environ.get("S3_ACCESS_KEY", ""),
                secret_key=os.environ.get("S3_SECRET_KEY", ""),
                region=os.environ.get("S3_REGION", "us-east-1"))
        else:
            from .config import settings
            _storage = FileStorage(settings().storage_dir)
    return _storage


def reset_storage() -> None:
    global _storage
    _storage = None
