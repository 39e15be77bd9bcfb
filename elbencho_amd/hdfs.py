"""HDFS benchmark engine over WebHDFS (REST).

Reference analogue: the libhdfs engine
(/root/reference/source/workers/LocalWorker.cpp:7488 hdfsDirModeIterateDirs,
:7617 hdfsDirModeIterateFiles, initHDFS :592): dir-mode namespaces
(r{rank}/d{dir}/r{rank}-f{file}), blockwise create/read, stat, delete.
The target image has no libhdfs/JVM, so this engine speaks the WebHDFS
REST protocol directly (op=MKDIRS/CREATE/APPEND/OPEN/GETFILESTATUS/DELETE
with the namenode's 307 datanode redirects) — plain HTTP, no Hadoop client.

Paths: ``hdfs://namenode:port/base/dir`` (WebHDFS HTTP port, usually 9870).
"""

from __future__ import annotations

import http.client
import json
import threading
import time
import urllib.parse
from typing import Any, Optional

from elbencho_amd import load_core
from elbencho_amd.config import BenchConfig
from elbencho_amd.histogram import Histogram
from elbencho_amd.s3 import _Counters, _add_lat  # same counter/latency plumbing
from elbencho_amd.stats import WorkerStats

HDFS_PREFIX = "hdfs://"


class HdfsError(RuntimeError):
    pass


class WebHdfsClient:
    """Minimal WebHDFS client: one persistent namenode connection per worker,
    follows 307 redirects to datanodes (or serves direct responses)."""

    def __init__(self, host: str, port: int, user: str = "root",
                 timeout: float = 60.0):
        self.host = host
        self.port = port
        self.user = user
        self.timeout = timeout
        self._conns: dict[tuple[str, int], http.client.HTTPConnection] = {}

    def close(self):
        for c in self._conns.values():
            c.close()
        self._conns.clear()

    def _conn(self, host: str, port: int) -> http.client.HTTPConnection:
        key = (host, port)
        if key not in self._conns:
            self._conns[key] = http.client.HTTPConnection(host, port,
                                                          timeout=self.timeout)
        return self._conns[key]

    def _url(self, path: str, op: str, **params) -> str:
        q = {"op": op, "user.name": self.user}
        q.update({k: str(v) for k, v in params.items() if v is not None})
        return ("/webhdfs/v1" + urllib.parse.quote(path) + "?" +
                urllib.parse.urlencode(q))

    def _request(self, method: str, path: str, op: str, body: bytes = b"",
                 follow: bool = True, **params) -> tuple[int, bytes]:
        url = self._url(path, op, **params)
        host, port = self.host, self.port
        for _hop in range(3):  # namenode -> datanode redirect chain
            conn = self._conn(host, port)
            try:
                conn.request(method, url, body=body or None,
                             headers={"Content-Type": "application/octet-stream"}
                             if body else {})
                resp = conn.getresponse()
                data = resp.read()
            except (ConnectionError, http.client.HTTPException, OSError):
                conn.close()
                self._conns.pop((host, port), None)
                raise
            if resp.status == 307 and follow:
                loc = resp.getheader("Location", "")
                u = urllib.parse.urlparse(loc)
                host, port = u.hostname or host, u.port or port
                url = u.path + ("?" + u.query if u.query else "")
                continue
            return resp.status, data
        raise HdfsError(f"too many redirects for {op} {path}")

    def _check(self, status: int, data: bytes, what: str):
        if status >= 300:
            raise HdfsError(f"{what} failed: HTTP {status}: "
                            f"{data[:300].decode(errors='replace')}")

    # --- operations ---
    def mkdirs(self, path: str):
        status, data = self._request("PUT", path, "MKDIRS")
        self._check(status, data, f"mkdirs {path}")

    def create(self, path: str, body: bytes, overwrite: bool = True):
        status, data = self._request("PUT", path, "CREATE", body=body,
                                     overwrite="true" if overwrite else "false")
        self._check(status, data, f"create {path}")

    def append(self, path: str, body: bytes):
        status, data = self._request("POST", path, "APPEND", body=body)
        self._check(status, data, f"append {path}")

    def open(self, path: str, offset: int = 0,
             length: Optional[int] = None) -> bytes:
        status, data = self._request("GET", path, "OPEN", offset=offset,
                                     length=length)
        self._check(status, data, f"open {path}")
        return data

    def status(self, path: str) -> dict:
        st, data = self._request("GET", path, "GETFILESTATUS")
        self._check(st, data, f"stat {path}")
        return json.loads(data)["FileStatus"]

    def delete(self, path: str, recursive: bool = False) -> bool:
        st, data = self._request("DELETE", path, "DELETE",
                                 recursive="true" if recursive else "false")
        self._check(st, data, f"delete {path}")
        return json.loads(data).get("boolean", False)


def parse_hdfs_path(p: str) -> tuple[str, int, str]:
    """"hdfs://host:port/base" -> (host, port, "/base")."""
    if not p.startswith(HDFS_PREFIX):
        raise HdfsError(f"HDFS paths must start with hdfs:// — got {p!r}")
    rest = p[len(HDFS_PREFIX):]
    hostport, _, base = rest.partition("/")
    host, _, port = hostport.partition(":")
    return host, int(port or 9870), "/" + base.rstrip("/")


class HdfsWorker(threading.Thread):
    def __init__(self, runner: "HdfsRunner", local_rank: int, phase: str):
        super().__init__(daemon=True)
        self.r = runner
        self.local_rank = local_rank
        self.rank = runner.cfg.rank_offset + local_rank
        self.phase = phase
        self.ops = _Counters()
        self.sw: Optional[_Counters] = None
        self.sw_elapsed_us = 0
        self.io_lat = Histogram()
        self.entry_lat = Histogram()
        self.error = ""
        self.elapsed_us = 0
        # --rwmixthr parity with the reference's hdfs isRWMixedReader
        # (LocalWorker.cpp:7631): first K threads of a WRITE phase read
        self.is_mix_reader = (phase == "WRITE" and
                              local_rank < runner.cfg._rwmix_threads_effective())
        self.rm_ops = _Counters()
        self.rm_sw: Optional[_Counters] = None
        self.io_lat_rm = Histogram()
        self.entry_lat_rm = Histogram()
        host, port, _ = runner.endpoint
        self.client = WebHdfsClient(host, port)
        self.core = load_core()

    # dir-mode namespace identical to the posix/S3 engines (reference
    # hdfsDirModeIterateFiles path layout)
    def _file_paths(self):
        cfg = self.r.cfg
        base = self.r.base
        for d in range(max(cfg.dirs, 1)):
            for f in range(cfg.files):
                if cfg.dirs > 0:
                    yield f"{base}/r{self.rank}/d{d}/r{self.rank}-f{f}"
                else:
                    yield f"{base}/r{self.rank}-f{f}"

    def _check_interrupt(self):
        if self.r.interrupt_flag.is_set():
            raise KeyboardInterrupt

    def _block(self, length: int, off: int) -> bytes:
        cfg = self.r.cfg
        if cfg.verify >= 0:
            return self.core.fill_checksum(length, off, cfg.verify)
        return bytes(self.r.rand_block[:length])

    def run(self):
        try:
            self.r.start_gate.wait()
            t0 = time.monotonic()
            self._run_phase()
            self.elapsed_us = int((time.monotonic() - t0) * 1e6)
        except KeyboardInterrupt:
            self.error = "interrupted"
        except Exception as e:  # noqa: BLE001
            self.error = str(e)
            self.r.interrupt_flag.set()
        finally:
            self.client.close()
            self.elapsed_us = self.elapsed_us or int(
                (time.monotonic() - self.r.phase_start) * 1e6)
            self.r.on_worker_done(self)

    def _run_phase(self):
        cfg = self.r.cfg
        ph = self.phase
        base = self.r.base
        if ph == "MKDIRS":
            for d in range(cfg.dirs):
                self._check_interrupt()
                t0 = time.monotonic()
                self.client.mkdirs(f"{base}/r{self.rank}/d{d}")
                self.entry_lat.vec = _add_lat(self.entry_lat, t0)
                self.ops.entries += 1
        elif ph == "RMDIRS":
            for d in range(cfg.dirs):
                self._check_interrupt()
                t0 = time.monotonic()
                ok = self.client.delete(f"{base}/r{self.rank}/d{d}",
                                        recursive=False)
                if not ok and not cfg.ignore_del_errors:
                    raise HdfsError(f"rmdir {base}/r{self.rank}/d{d} failed")
                self.entry_lat.vec = _add_lat(self.entry_lat, t0)
                self.ops.entries += 1
            # rank dir itself
            self.client.delete(f"{base}/r{self.rank}", recursive=True)
        elif ph == "WRITE" and self.is_mix_reader:
            self.ops, self.rm_ops = self.rm_ops, self.ops
            self.io_lat, self.io_lat_rm = self.io_lat_rm, self.io_lat
            self.entry_lat, self.entry_lat_rm = self.entry_lat_rm, self.entry_lat
            try:
                self.phase = "READ"
                self._run_phase()
            finally:
                self.phase = "WRITE"
                self.ops, self.rm_ops = self.rm_ops, self.ops
                self.io_lat, self.io_lat_rm = self.io_lat_rm, self.io_lat
                self.entry_lat, self.entry_lat_rm = \
                    self.entry_lat_rm, self.entry_lat
        elif ph == "WRITE":
            size, bs = cfg.file_size, cfg.block_size
            for path in self._file_paths():
                self._check_interrupt()
                te = time.monotonic()
                off = 0
                first = True
                while True:
                    ln = min(bs, size - off)
                    t0 = time.monotonic()
                    if first:  # CREATE writes block 0 (or an empty file)
                        self.client.create(path, self._block(ln, off))
                        first = False
                    else:
                        self.client.append(path, self._block(ln, off))
                    self.io_lat.vec = _add_lat(self.io_lat, t0)
                    self.ops.bytes += ln
                    self.ops.iops += 1
                    off += ln
                    if off >= size:
                        break
                self.entry_lat.vec = _add_lat(self.entry_lat, te)
                self.ops.entries += 1
        elif ph == "READ":
            size, bs = cfg.file_size, cfg.block_size
            for path in self._file_paths():
                self._check_interrupt()
                te = time.monotonic()
                off = 0
                while off < size:
                    ln = min(bs, size - off)
                    t0 = time.monotonic()
                    data = self.client.open(path, offset=off, length=ln)
                    self.io_lat.vec = _add_lat(self.io_lat, t0)
                    if len(data) != ln:
                        raise HdfsError(f"short read of {path}: {len(data)} != {ln}")
                    if cfg.verify >= 0:
                        bad = self.core.verify_checksum(data, off, cfg.verify)
                        if bad != 2**64 - 1:
                            raise HdfsError(f"HDFS data verification failed for "
                                            f"{path} at offset {bad}")
                    self.ops.bytes += ln
                    self.ops.iops += 1
                    off += ln
                self.entry_lat.vec = _add_lat(self.entry_lat, te)
                self.ops.entries += 1
        elif ph == "STAT":
            for path in self._file_paths():
                self._check_interrupt()
                t0 = time.monotonic()
                self.client.status(path)
                self.entry_lat.vec = _add_lat(self.entry_lat, t0)
                self.ops.entries += 1
        elif ph == "RMFILES":
            for path in self._file_paths():
                self._check_interrupt()
                t0 = time.monotonic()
                ok = self.client.delete(path)
                if not ok and not cfg.ignore_del_errors:
                    raise HdfsError(f"delete {path} failed")
                self.entry_lat.vec = _add_lat(self.entry_lat, t0)
                self.ops.entries += 1
        else:
            raise HdfsError(f"HDFS phase not supported: {ph}")


class HdfsRunner:
    """Coordinator backend for --hdfs (bench_mode == "hdfs"); same runner
    interface and stonewall semantics as the S3/posix runners."""

    def __init__(self, cfg: BenchConfig):
        self.cfg = cfg
        if not cfg.paths:
            raise HdfsError("HDFS mode requires an hdfs://namenode:port/base path")
        host, port, base = parse_hdfs_path(cfg.paths[0])
        self.endpoint = (host, port, base)
        self.base = base
        self.workers: list[HdfsWorker] = []
        self.interrupt_flag = threading.Event()
        self.start_gate = threading.Event()
        self.done_count = 0
        self.done_cv = threading.Condition()
        self.stonewalled = False
        self.phase_start = 0.0
        import os as _os
        blk = _os.urandom(min(max(cfg.block_size, 1), 1 << 22))
        while len(blk) < cfg.block_size:
            blk = blk + blk
        self.rand_block = blk[:cfg.block_size]

    def start(self, phase_name: str) -> None:
        self.interrupt_flag.clear()
        self.start_gate.clear()
        self.done_count = 0
        self.stonewalled = False
        self.workers = [HdfsWorker(self, i, phase_name)
                        for i in range(self.cfg.threads)]
        for w in self.workers:
            w.start()
        self.phase_start = time.monotonic()
        self.start_gate.set()

    def on_worker_done(self, w: HdfsWorker) -> None:
        with self.done_cv:
            if not self.stonewalled and not w.error and (
                    w.ops.bytes or w.ops.entries or w.ops.iops or
                    w.rm_ops.bytes or w.rm_ops.iops):
                elapsed = int((time.monotonic() - self.phase_start) * 1e6)
                for peer in self.workers:
                    peer.sw = _Counters(peer.ops.entries, peer.ops.bytes,
                                        peer.ops.iops)
                    peer.rm_sw = _Counters(peer.rm_ops.entries,
                                           peer.rm_ops.bytes, peer.rm_ops.iops)
                    peer.sw_elapsed_us = elapsed
                self.stonewalled = True
            self.done_count += 1
            self.done_cv.notify_all()

    def wait(self, timeout_ms: int) -> bool:
        with self.done_cv:
            return self.done_cv.wait_for(
                lambda: self.done_count >= len(self.workers),
                timeout=None if timeout_ms < 0 else timeout_ms / 1000.0)

    def poll(self) -> dict[str, Any]:
        agg = {"entries": 0, "bytes": 0, "iops": 0,
               "workers_done": self.done_count,
               "workers_total": len(self.workers),
               "workers_with_error": sum(1 for w in self.workers if w.error),
               "elapsed_usec": int((time.monotonic() - self.phase_start) * 1e6),
               "stonewall_triggered": self.stonewalled,
               "lat_num_ios": 0, "lat_sum_ios": 0, "lat_num_entries": 0,
               "lat_sum_entries": 0}
        for w in self.workers:
            agg["entries"] += w.ops.entries + w.rm_ops.entries
            agg["bytes"] += w.ops.bytes + w.rm_ops.bytes
            agg["iops"] += w.ops.iops + w.rm_ops.iops
        return agg

    def poll_workers(self):
        return [{"rank": w.rank, "entries": w.ops.entries, "bytes": w.ops.bytes,
                 "iops": w.ops.iops} for w in self.workers]

    def interrupt(self) -> None:
        self.interrupt_flag.set()

    def trigger_stonewall(self) -> None:
        with self.done_cv:
            if not self.stonewalled:
                elapsed = int((time.monotonic() - self.phase_start) * 1e6)
                for peer in self.workers:
                    peer.sw = _Counters(peer.ops.entries, peer.ops.bytes,
                                        peer.ops.iops)
                    peer.sw_elapsed_us = elapsed
                self.stonewalled = True

    def finish(self) -> list[WorkerStats]:
        self.wait(-1)
        out = []
        for w in self.workers:
            sw = w.sw or _Counters()
            rm_sw = w.rm_sw or w.rm_ops
            out.append(WorkerStats(
                rank=w.rank, elapsed_usec=w.elapsed_us,
                entries=w.ops.entries, bytes=w.ops.bytes, iops=w.ops.iops,
                stonewall_elapsed_usec=w.sw_elapsed_us or w.elapsed_us,
                stonewall_entries=sw.entries, stonewall_bytes=sw.bytes,
                stonewall_iops=sw.iops,
                rm_entries=w.rm_ops.entries, rm_bytes=w.rm_ops.bytes,
                rm_iops=w.rm_ops.iops,
                rm_stonewall_entries=rm_sw.entries,
                rm_stonewall_bytes=rm_sw.bytes, rm_stonewall_iops=rm_sw.iops,
                io_lat=list(w.io_lat.vec), entry_lat=list(w.entry_lat.vec),
                io_lat_rm=list(w.io_lat_rm.vec),
                entry_lat_rm=list(w.entry_lat_rm.vec),
                error=w.error))
        return out

    def planned_work(self, phase_name: str) -> tuple[int, int]:
        cfg = self.cfg
        nfiles = max(cfg.dirs, 1) * cfg.files * cfg.threads
        if phase_name in ("WRITE", "READ"):
            return nfiles, nfiles * cfg.file_size
        if phase_name in ("STAT", "RMFILES"):
            return nfiles, 0
        if phase_name in ("MKDIRS", "RMDIRS"):
            return cfg.dirs * cfg.threads, 0
        return 0, 0

    def close(self) -> None:
        self.interrupt_flag.set()
