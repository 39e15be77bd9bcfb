"""In-memory WebHDFS mock: MKDIRS/CREATE/APPEND/OPEN/GETFILESTATUS/DELETE
with optional namenode->datanode 307 redirects (to itself), exercising the
client's redirect handling like a real Hadoop deployment."""

from __future__ import annotations

import json
import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

PREFIX = "/webhdfs/v1"


class HdfsStore:
    def __init__(self):
        self.lock = threading.Lock()
        self.files: dict[str, bytearray] = {}
        self.dirs: set[str] = {"/"}


class WebHdfsHandler(BaseHTTPRequestHandler):
    store = HdfsStore()
    protocol_version = "HTTP/1.1"
    redirect_data_ops = True  # 307 CREATE/APPEND/OPEN like a real namenode
    port = 0

    def log_message(self, *a):
        pass

    def _parse(self):
        u = urllib.parse.urlparse(self.path)
        path = urllib.parse.unquote(u.path[len(PREFIX):]) or "/"
        q = dict(urllib.parse.parse_qsl(u.query))
        return path, q

    def _send(self, code: int, body: bytes = b"",
              headers: dict | None = None):
        self.send_response(code)
        for k, v in (headers or {}).items():
            self.send_header(k, v)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _json(self, obj, code=200):
        self._send(code, json.dumps(obj).encode(),
                   {"Content-Type": "application/json"})

    def _redirect(self):
        # redirect to ourselves with a marker so the second hop executes
        self._send(307, b"", {"Location":
                              f"http://127.0.0.1:{self.port}{self.path}"
                              f"&datanode=1"})

    def _body(self) -> bytes:
        ln = int(self.headers.get("Content-Length", 0))
        return self.rfile.read(ln) if ln else b""

    def do_PUT(self):
        path, q = self._parse()
        st = self.store
        op = q.get("op", "").upper()
        if op == "MKDIRS":
            with st.lock:
                st.dirs.add(path)
            return self._json({"boolean": True})
        if op == "CREATE":
            if self.redirect_data_ops and "datanode" not in q:
                self._body()
                return self._redirect()
            body = self._body()
            with st.lock:
                if q.get("overwrite", "true") != "true" and path in st.files:
                    return self._json({"RemoteException":
                                       {"exception": "FileAlreadyExists"}}, 403)
                st.files[path] = bytearray(body)
            return self._send(201)
        self._json({"RemoteException": {"exception": "UnsupportedOp"}}, 400)

    def do_POST(self):
        path, q = self._parse()
        st = self.store
        if q.get("op", "").upper() == "APPEND":
            if self.redirect_data_ops and "datanode" not in q:
                self._body()
                return self._redirect()
            body = self._body()
            with st.lock:
                if path not in st.files:
                    return self._json({"RemoteException":
                                       {"exception": "FileNotFound"}}, 404)
                st.files[path].extend(body)
            return self._send(200)
        self._json({"RemoteException": {"exception": "UnsupportedOp"}}, 400)

    def do_GET(self):
        path, q = self._parse()
        st = self.store
        op = q.get("op", "").upper()
        if op == "OPEN":
            if self.redirect_data_ops and "datanode" not in q:
                return self._redirect()
            with st.lock:
                if path not in st.files:
                    return self._json({"RemoteException":
                                       {"exception": "FileNotFound"}}, 404)
                data = bytes(st.files[path])
            off = int(q.get("offset", 0))
            ln = q.get("length")
            end = off + int(ln) if ln else len(data)
            return self._send(200, data[off:end],
                              {"Content-Type": "application/octet-stream"})
        if op == "GETFILESTATUS":
            with st.lock:
                if path in st.files:
                    return self._json({"FileStatus": {
                        "type": "FILE", "length": len(st.files[path])}})
                if path in st.dirs:
                    return self._json({"FileStatus": {"type": "DIRECTORY",
                                                      "length": 0}})
            return self._json({"RemoteException":
                               {"exception": "FileNotFound"}}, 404)
        self._json({"RemoteException": {"exception": "UnsupportedOp"}}, 400)

    def do_DELETE(self):
        path, q = self._parse()
        st = self.store
        if q.get("op", "").upper() == "DELETE":
            recursive = q.get("recursive") == "true"
            with st.lock:
                if path in st.files:
                    del st.files[path]
                    return self._json({"boolean": True})
                if path in st.dirs or any(f.startswith(path + "/")
                                          for f in st.files):
                    if recursive:
                        st.files = {f: v for f, v in st.files.items()
                                    if not f.startswith(path + "/")}
                        st.dirs = {d for d in st.dirs
                                   if d != path and not d.startswith(path + "/")}
                    else:
                        st.dirs.discard(path)
                    return self._json({"boolean": True})
            return self._json({"boolean": False})
        self._json({"RemoteException": {"exception": "UnsupportedOp"}}, 400)


def start_mock(port: int = 0):
    WebHdfsHandler.store = HdfsStore()
    server = ThreadingHTTPServer(("127.0.0.1", port), WebHdfsHandler)
    WebHdfsHandler.port = server.server_address[1]
    t = threading.Thread(target=server.serve_forever, daemon=True)
    t.start()
    return server, server.server_address[1]
