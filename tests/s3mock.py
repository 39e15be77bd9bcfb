"""Minimal in-memory S3 mock server for tests.

Implements enough of the S3 REST API for the elbencho_amd S3 engine: bucket
PUT/DELETE/HEAD, object PUT/GET(+Range)/HEAD/DELETE, multipart
(initiate/part/complete/abort), list-objects-v2 with continuation, POST
multi-delete. Verifies AWS SigV4 signatures against the configured secret,
so the client's signing is tested for real.
"""

from __future__ import annotations

import hashlib
import hmac
import re
import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

ACCESS_KEY = "testkey"
SECRET_KEY = "testsecret"
REGION = "us-east-1"


class S3Store:
    def __init__(self):
        self.lock = threading.Lock()
        self.buckets: dict[str, dict[str, bytes]] = {}
        self.uploads: dict[str, dict[int, bytes]] = {}  # uploadId -> parts
        self.upload_meta: dict[str, tuple[str, str]] = {}  # uploadId -> (bucket,key)
        self.next_upload = [0]
        self.acls: dict[tuple, str] = {}      # (bucket, key|None) -> canned acl
        self.tags: dict[tuple, bytes] = {}    # (bucket, key|None) -> tagging xml
        self.misc: dict[tuple, bytes] = {}    # (bucket, kind) -> config xml


def _xml(body: str) -> bytes:
    return ('<?xml version="1.0" encoding="UTF-8"?>' + body).encode()


class S3Handler(BaseHTTPRequestHandler):
    store: S3Store
    protocol_version = "HTTP/1.1"

    def log_message(self, *a):
        pass

    # ------------------------------------------------------------------
    def _verify_sig(self, body: bytes) -> bool:
        auth = self.headers.get("Authorization", "")
        m = re.match(r"AWS4-HMAC-SHA256 Credential=([^/]+)/(\d+)/([^/]+)/s3/aws4_request, "
                     r"SignedHeaders=([^,]+), Signature=([0-9a-f]+)", auth)
        if not m:
            return False
        access, datestamp, region, signed_headers, sig = m.groups()
        if access != ACCESS_KEY:
            return False

        url = urllib.parse.urlparse(self.path)
        q = urllib.parse.parse_qsl(url.query, keep_blank_values=True)
        canonical_query = "&".join(
            f"{urllib.parse.quote(k, safe='')}={urllib.parse.quote(v, safe='')}"
            for k, v in sorted(q))
        payload_hash = self.headers.get("x-amz-content-sha256", "")
        canonical_headers = "".join(
            f"{h}:{self.headers.get(h, '').strip()}\n" for h in signed_headers.split(";"))
        canonical_request = "\n".join([
            self.command, urllib.parse.quote(urllib.parse.unquote(url.path)),
            canonical_query, canonical_headers, signed_headers, payload_hash])

        scope = f"{datestamp}/{region}/s3/aws4_request"
        string_to_sign = "\n".join([
            "AWS4-HMAC-SHA256", self.headers.get("x-amz-date", ""), scope,
            hashlib.sha256(canonical_request.encode()).hexdigest()])

        def h(key, msg):
            return hmac.new(key, msg.encode(), hashlib.sha256).digest()

        k = h(("AWS4" + SECRET_KEY).encode(), datestamp)
        k = h(k, region)
        k = h(k, "s3")
        k = h(k, "aws4_request")
        expect = hmac.new(k, string_to_sign.encode(), hashlib.sha256).hexdigest()
        if expect != sig:
            return False
        # payload hash must match the body
        if payload_hash not in ("UNSIGNED-PAYLOAD", ""):
            return hashlib.sha256(body).hexdigest() == payload_hash
        return True

    # ------------------------------------------------------------------
    def _send(self, code: int, body: bytes = b"", headers: dict | None = None):
        self.send_response(code)
        for k, v in (headers or {}).items():
            self.send_header(k, v)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        if self.command != "HEAD":
            self.wfile.write(body)

    def _err(self, code: int, s3code: str):
        self._send(code, _xml(f"<Error><Code>{s3code}</Code></Error>"))

    def _parse(self):
        url = urllib.parse.urlparse(self.path)
        parts = urllib.parse.unquote(url.path).lstrip("/").split("/", 1)
        bucket = parts[0] if parts[0] else None
        key = parts[1] if len(parts) > 1 else None
        q = dict(urllib.parse.parse_qsl(url.query, keep_blank_values=True))
        return bucket, key, q

    def _body(self) -> bytes:
        n = int(self.headers.get("Content-Length", "0"))
        return self.rfile.read(n) if n else b""

    # ------------------------------------------------------------------
    def do_PUT(self):
        body = self._body()
        if not self._verify_sig(body):
            return self._err(403, "SignatureDoesNotMatch")
        bucket, key, q = self._parse()
        st = self.store
        with st.lock:
            if "acl" in q:
                st.acls[(bucket, key)] = self.headers.get("x-amz-acl", "private")
                return self._send(200)
            if "tagging" in q:
                st.tags[(bucket, key)] = body
                return self._send(200)
            if "versioning" in q:
                st.misc[(bucket, "versioning")] = body
                return self._send(200)
            if "object-lock" in q:
                st.misc[(bucket, "object-lock")] = body
                return self._send(200)
            if key is None:  # create bucket
                if bucket in st.buckets:
                    return self._err(409, "BucketAlreadyOwnedByYou")
                st.buckets[bucket] = {}
                return self._send(200)
            if bucket not in st.buckets:
                return self._err(404, "NoSuchBucket")
            if "partNumber" in q and "uploadId" in q:
                up = st.uploads.get(q["uploadId"])
                if up is None:
                    return self._err(404, "NoSuchUpload")
                up[int(q["partNumber"])] = body
                etag = hashlib.md5(body).hexdigest()
                return self._send(200, headers={"ETag": f'"{etag}"'})
            st.buckets[bucket][key] = body
            return self._send(200, headers={"ETag": f'"{hashlib.md5(body).hexdigest()}"'})

    def do_GET(self):
        if not self._verify_sig(b""):
            return self._err(403, "SignatureDoesNotMatch")
        bucket, key, q = self._parse()
        st = self.store
        with st.lock:
            if "acl" in q:
                acl = st.acls.get((bucket, key), "private")
                return self._send(200, _xml(
                    f"<AccessControlPolicy><AccessControlList><Grant>"
                    f"<Permission>{acl}</Permission></Grant></AccessControlList>"
                    f"</AccessControlPolicy>"))
            if "tagging" in q:
                return self._send(200, st.tags.get((bucket, key),
                                                   _xml("<Tagging><TagSet></TagSet></Tagging>")))
            if "versioning" in q:
                return self._send(200, st.misc.get(
                    (bucket, "versioning"), _xml("<VersioningConfiguration/>")))
            if "object-lock" in q:
                return self._send(200, st.misc.get(
                    (bucket, "object-lock"), _xml("<ObjectLockConfiguration/>")))
            if "uploads" in q and key is None:  # list multipart uploads
                ups = "".join(
                    f"<Upload><Key>{k}</Key><UploadId>{uid}</UploadId></Upload>"
                    for uid, (b, k) in st.upload_meta.items() if b == bucket
                    and k.startswith(q.get("prefix", "")))
                return self._send(200, _xml(
                    f"<ListMultipartUploadsResult>{ups}</ListMultipartUploadsResult>"))
            if "uploadId" in q and key is not None:  # list parts
                up = st.uploads.get(q["uploadId"])
                if up is None:
                    return self._err(404, "NoSuchUpload")
                parts = "".join(
                    f"<Part><PartNumber>{n}</PartNumber>"
                    f"<ETag>\"{hashlib.md5(d).hexdigest()}\"</ETag></Part>"
                    for n, d in sorted(up.items()))
                return self._send(200, _xml(f"<ListPartsResult>{parts}</ListPartsResult>"))
            if bucket not in st.buckets:
                return self._err(404, "NoSuchBucket")
            if key is None:  # list objects v2
                prefix = q.get("prefix", "")
                max_keys = int(q.get("max-keys", "1000"))
                start = q.get("continuation-token", "")
                keys = sorted(k for k in st.buckets[bucket] if k.startswith(prefix))
                if start:
                    keys = [k for k in keys if k > start]
                page = keys[:max_keys]
                truncated = len(keys) > max_keys
                contents = "".join(
                    f"<Contents><Key>{k}</Key><Size>{len(st.buckets[bucket][k])}</Size>"
                    f"</Contents>" for k in page)
                token = (f"<NextContinuationToken>{page[-1]}</NextContinuationToken>"
                         if truncated and page else "")
                return self._send(200, _xml(
                    f"<ListBucketResult><IsTruncated>{str(truncated).lower()}"
                    f"</IsTruncated>{contents}{token}</ListBucketResult>"))
            obj = st.buckets[bucket].get(key)
            if obj is None:
                return self._err(404, "NoSuchKey")
            rng = self.headers.get("Range")
            if rng:
                m = re.match(r"bytes=(\d+)-(\d+)", rng)
                lo, hi = int(m.group(1)), int(m.group(2))
                part = obj[lo:hi + 1]
                return self._send(206, part, {
                    "Content-Range": f"bytes {lo}-{hi}/{len(obj)}"})
            return self._send(200, obj)

    def do_HEAD(self):
        if not self._verify_sig(b""):
            return self._err(403, "SignatureDoesNotMatch")
        bucket, key, _ = self._parse()
        st = self.store
        with st.lock:
            if bucket not in st.buckets:
                return self._err(404, "NoSuchBucket")
            if key is None:
                return self._send(200)
            obj = st.buckets[bucket].get(key)
            if obj is None:
                return self._err(404, "NoSuchKey")
            return self._send(200, headers={"Content-Length-S3": str(len(obj))})

    def do_DELETE(self):
        if not self._verify_sig(b""):
            return self._err(403, "SignatureDoesNotMatch")
        bucket, key, q = self._parse()
        st = self.store
        with st.lock:
            if "tagging" in q:
                st.tags.pop((bucket, key), None)
                return self._send(204)
            if "uploadId" in q:  # abort multipart
                st.uploads.pop(q["uploadId"], None)
                st.upload_meta.pop(q["uploadId"], None)
                return self._send(204)
            if bucket not in st.buckets:
                return self._err(404, "NoSuchBucket")
            if key is None:
                if st.buckets[bucket]:
                    return self._err(409, "BucketNotEmpty")
                del st.buckets[bucket]
                return self._send(204)
            if key not in st.buckets[bucket]:
                return self._err(404, "NoSuchKey")
            del st.buckets[bucket][key]
            return self._send(204)

    def do_POST(self):
        body = self._body()
        if not self._verify_sig(body):
            return self._err(403, "SignatureDoesNotMatch")
        bucket, key, q = self._parse()
        st = self.store
        with st.lock:
            if "uploads" in q:  # initiate multipart
                st.next_upload[0] += 1
                uid = f"upload-{st.next_upload[0]}"
                st.uploads[uid] = {}
                st.upload_meta[uid] = (bucket, key)
                return self._send(200, _xml(
                    f"<InitiateMultipartUploadResult><UploadId>{uid}</UploadId>"
                    f"</InitiateMultipartUploadResult>"))
            if "uploadId" in q:  # complete multipart
                uid = q["uploadId"]
                up = st.uploads.pop(uid, None)
                st.upload_meta.pop(uid, None)
                if up is None:
                    return self._err(404, "NoSuchUpload")
                data = b"".join(up[n] for n in sorted(up))
                st.buckets.setdefault(bucket, {})[key] = data
                return self._send(200, _xml(
                    "<CompleteMultipartUploadResult></CompleteMultipartUploadResult>"))
            if "delete" in q:  # multi-delete
                keys = re.findall(r"<Key>([^<]+)</Key>", body.decode())
                for k in keys:
                    st.buckets.get(bucket, {}).pop(k, None)
                return self._send(200, _xml("<DeleteResult></DeleteResult>"))
        return self._err(400, "BadRequest")


def start_mock(port: int = 0) -> tuple[ThreadingHTTPServer, int]:
    store = S3Store()
    S3Handler.store = store
    server = ThreadingHTTPServer(("127.0.0.1", port), S3Handler)
    server.daemon_threads = True
    t = threading.Thread(target=server.serve_forever, daemon=True)
    t.start()
    return server, server.server_address[1]
