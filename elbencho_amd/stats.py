"""Statistics: phase result aggregation, console table, CSV/JSON output,
live stats.

Behavior parity with the reference's Statistics layer
(/root/reference/source/Statistics.cpp): first-done ("stonewall") vs
last-done aggregate columns (generatePhaseResults :1695, console format
string "%|-11| %|-17|%|1| %|11| %|11|" Statistics.h:138), CSV schema
(docs/csv-docs.md), JSON result tree (printPhaseResultsAsJSON :2485).
Independent implementation.
"""

from __future__ import annotations

import csv
import datetime
import json
import os
import sys
import time
from dataclasses import dataclass, field
from typing import Any

from elbencho_amd import VERSION
from elbencho_amd.config import BenchConfig, PATH_DIR
from elbencho_amd.histogram import Histogram
from elbencho_amd.units import elapsed_ms_to_human


# ---------------------------------------------------------------------------
# CPU utilization (reference analogue: CPUUtil.{h,cpp})
# ---------------------------------------------------------------------------

class CpuUtil:
    def __init__(self):
        self._last = self._read()

    @staticmethod
    def _read() -> tuple[int, int]:
        try:
            with open("/proc/stat") as f:
                parts = f.readline().split()[1:]
            vals = [int(x) for x in parts]
            idle = vals[3] + (vals[4] if len(vals) > 4 else 0)
            total = sum(vals)
            return total, idle
        except OSError:
            return 0, 0

    def percent_since_last(self) -> int:
        """CPU busy percentage since the previous call."""
        cur = self._read()
        last, self._last = self._last, cur
        dt = cur[0] - last[0]
        didle = cur[1] - last[1]
        if dt <= 0:
            return 0
        return int(round(100.0 * (dt - didle) / dt))


# ---------------------------------------------------------------------------
# per-worker / per-service stats containers
# ---------------------------------------------------------------------------

@dataclass
class WorkerStats:
    """Final per-worker (or per-remote-service) stats for one phase."""
    rank: int = 0
    elapsed_usec: int = 0
    entries: int = 0
    bytes: int = 0
    iops: int = 0
    stonewall_elapsed_usec: int = 0
    stonewall_entries: int = 0
    stonewall_bytes: int = 0
    stonewall_iops: int = 0
    rm_entries: int = 0
    rm_bytes: int = 0
    rm_iops: int = 0
    rm_stonewall_entries: int = 0
    rm_stonewall_bytes: int = 0
    rm_stonewall_iops: int = 0
    io_lat: list[int] = field(default_factory=list)
    entry_lat: list[int] = field(default_factory=list)
    io_lat_rm: list[int] = field(default_factory=list)
    entry_lat_rm: list[int] = field(default_factory=list)
    error: str = ""
    num_workers: int = 1  # >1 when this row aggregates a remote service

    @classmethod
    def from_engine(cls, d: dict[str, Any]) -> "WorkerStats":
        return cls(
            rank=d["rank"],
            elapsed_usec=d["elapsed_usec"],
            entries=d["entries"],
            bytes=d["bytes"],
            iops=d["iops"],
            stonewall_elapsed_usec=d["stonewall_elapsed_usec"],
            stonewall_entries=d["stonewall_entries"],
            stonewall_bytes=d["stonewall_bytes"],
            stonewall_iops=d["stonewall_iops"],
            rm_entries=d.get("rm_entries", 0),
            rm_bytes=d.get("rm_bytes", 0),
            rm_iops=d.get("rm_iops", 0),
            rm_stonewall_entries=d.get("rm_stonewall_entries", 0),
            rm_stonewall_bytes=d.get("rm_stonewall_bytes", 0),
            rm_stonewall_iops=d.get("rm_stonewall_iops", 0),
            io_lat=list(d["io_lat"]),
            entry_lat=list(d["entry_lat"]),
            io_lat_rm=list(d.get("io_lat_rm", [])),
            entry_lat_rm=list(d.get("entry_lat_rm", [])),
            error=d["error"],
        )


@dataclass
class PhaseResults:
    phase_name: str = ""
    phase_id: str = ""
    start_time: float = 0.0  # epoch seconds
    first_finish_usec: int = 0
    last_finish_usec: int = 0
    # last-done totals
    entries: int = 0
    bytes: int = 0
    iops: int = 0
    # first-done (stonewall) totals
    sw_entries: int = 0
    sw_bytes: int = 0
    sw_iops: int = 0
    # rwmix read totals (reads within a write phase)
    rm_entries: int = 0
    rm_bytes: int = 0
    rm_iops: int = 0
    rm_sw_entries: int = 0
    rm_sw_bytes: int = 0
    rm_sw_iops: int = 0
    io_lat: Histogram = field(default_factory=Histogram)
    entry_lat: Histogram = field(default_factory=Histogram)
    io_lat_rm: Histogram = field(default_factory=Histogram)
    entry_lat_rm: Histogram = field(default_factory=Histogram)
    cpu_first: int = 0
    cpu_last: int = 0
    worker_elapsed_usec: list[int] = field(default_factory=list)
    errors: list[str] = field(default_factory=list)

    # --- derived ---
    def per_sec_last(self, value: int) -> int:
        return int(value * 1_000_000 / self.last_finish_usec) if self.last_finish_usec else 0

    def per_sec_first(self, value: int) -> int:
        return int(value * 1_000_000 / self.first_finish_usec) if self.first_finish_usec else 0


def aggregate_phase(phase_name: str, phase_id: str, start_time: float,
                    workers: list[WorkerStats], cpu_first: int = 0,
                    cpu_last: int = 0) -> PhaseResults:
    """Aggregate per-worker stats into the first-done/last-done result pair
    (reference Statistics::generatePhaseResults, Statistics.cpp:1695)."""
    r = PhaseResults(phase_name=phase_name, phase_id=phase_id, start_time=start_time)
    if not workers:
        return r

    r.first_finish_usec = min(w.stonewall_elapsed_usec or w.elapsed_usec for w in workers)
    r.last_finish_usec = max(w.elapsed_usec for w in workers)
    r.cpu_first, r.cpu_last = cpu_first, cpu_last

    for w in workers:
        r.entries += w.entries
        r.bytes += w.bytes
        r.iops += w.iops
        r.sw_entries += w.stonewall_entries
        r.sw_bytes += w.stonewall_bytes
        r.sw_iops += w.stonewall_iops
        r.rm_entries += w.rm_entries
        r.rm_bytes += w.rm_bytes
        r.rm_iops += w.rm_iops
        r.rm_sw_entries += w.rm_stonewall_entries
        r.rm_sw_bytes += w.rm_stonewall_bytes
        r.rm_sw_iops += w.rm_stonewall_iops
        if w.io_lat:
            r.io_lat.merge(w.io_lat)
        if w.entry_lat:
            r.entry_lat.merge(w.entry_lat)
        if w.io_lat_rm:
            r.io_lat_rm.merge(w.io_lat_rm)
        if w.entry_lat_rm:
            r.entry_lat_rm.merge(w.entry_lat_rm)
        r.worker_elapsed_usec.append(w.elapsed_usec)
        if w.error:
            r.errors.append(f"Rank {w.rank}: {w.error}")
    return r


# ---------------------------------------------------------------------------
# console output
# ---------------------------------------------------------------------------

def _fmt_row(op: str, label: str, first, last, colon: str | None = None) -> str:
    # reference format string: "%|-11| %|-17|%|1| %|11| %|11|" (Statistics.h:138)
    if colon is None:
        colon = ":" if label else ""
    return f"{op:<11} {label:<17}{colon:>1} {str(first):>11} {str(last):>11}"


def print_results_table_header(out=None) -> None:
    out = out or sys.stdout
    print(_fmt_row("OPERATION", "RESULT TYPE", "FIRST DONE", "LAST DONE", colon=""), file=out)
    print(_fmt_row("===========", "================", "==========", "=========", colon=""),
          file=out)


def print_phase_results(cfg: BenchConfig, r: PhaseResults, out=None) -> None:
    out = out or sys.stdout
    if cfg.bench_mode == "s3":
        entry_type = ("Buckets" if r.phase_name in ("MKBUCKETS", "RMBUCKETS")
                      else "Objects")
    else:
        entry_type = "Dirs" if r.phase_name in ("MKDIRS", "RMDIRS") else "Files"

    rows: list[tuple[str, Any, Any]] = []
    rows.append(("Elapsed time",
                 elapsed_ms_to_human(r.first_finish_usec // 1000),
                 elapsed_ms_to_human(r.last_finish_usec // 1000)))

    if r.entries:
        rows.append((f"{entry_type}/s", r.per_sec_first(r.sw_entries),
                     r.per_sec_last(r.entries)))
        if cfg.show_dir_stats and cfg.path_type == PATH_DIR and cfg.files:
            rows.append(("Dirs/s", r.per_sec_first(r.sw_entries) // cfg.files,
                         r.per_sec_last(r.entries) // cfg.files))

    is_rwmix = bool(r.rm_bytes or r.rm_iops or r.rm_entries)
    # --base10: MB/s (10^6) instead of MiB/s (2^20) in console output
    # (reference Statistics.cpp:190-191; CSV schema stays MiB)
    mib = 1000 * 1000 if cfg.show_base10 else 1024 * 1024
    tp_unit = "MB/s" if cfg.show_base10 else "MiB/s"
    tot_unit = "MB" if cfg.show_base10 else "MiB"

    if r.iops:
        # suppress IOPS when it would equal files/s (dir mode, 1 block per file)
        if (cfg.path_type != PATH_DIR) or (cfg.block_size != cfg.file_size) or not r.entries:
            rows.append(("IOPS write" if is_rwmix else "IOPS",
                         r.per_sec_first(r.sw_iops), r.per_sec_last(r.iops)))
    if is_rwmix and r.rm_iops:
        rows.append(("IOPS read", r.per_sec_first(r.rm_sw_iops), r.per_sec_last(r.rm_iops)))
        rows.append(("IOPS total", r.per_sec_first(r.sw_iops + r.rm_sw_iops),
                     r.per_sec_last(r.iops + r.rm_iops)))

    if r.bytes:
        rows.append((f"{tp_unit} write" if is_rwmix else f"Throughput {tp_unit}",
                     r.per_sec_first(r.sw_bytes) // mib, r.per_sec_last(r.bytes) // mib))
    if is_rwmix and r.rm_bytes:
        rows.append((f"{tp_unit} read", r.per_sec_first(r.rm_sw_bytes) // mib,
                     r.per_sec_last(r.rm_bytes) // mib))
        rows.append((f"{tp_unit} total", r.per_sec_first(r.sw_bytes + r.rm_sw_bytes) // mib,
                     r.per_sec_last(r.bytes + r.rm_bytes) // mib))

    if r.bytes:
        rows.append((f"{tot_unit} write" if is_rwmix else f"Total {tot_unit}",
                     r.sw_bytes // mib, r.bytes // mib))
    if is_rwmix and r.rm_bytes:
        rows.append((f"{tot_unit} read", r.rm_sw_bytes // mib, r.rm_bytes // mib))

    if r.entries:
        rows.append((f"{entry_type} total", r.sw_entries, r.entries))

    if cfg.cpu_util:
        rows.append(("CPU util %", r.cpu_first, r.cpu_last))

    first = True
    for label, a, b in rows:
        print(_fmt_row(r.phase_name if first else "", label, a, b), file=out)
        first = False

    if cfg.lat and r.entry_lat.num_values:
        _print_latency(cfg, "Ent lat us", r.entry_lat, out)
    if cfg.lat and r.io_lat.num_values:
        _print_latency(cfg, "IO lat us", r.io_lat, out)
    if cfg.lat and r.io_lat_rm.num_values:
        _print_latency(cfg, "IO lat rd us", r.io_lat_rm, out)

    if cfg.all_elapsed and r.worker_elapsed_usec:
        vals = " ".join(elapsed_ms_to_human(us // 1000) for us in r.worker_elapsed_usec)
        print(_fmt_row("", "Threads elapsed", "", vals), file=out)

    for e in r.errors:
        print(f"ERROR: {e}", file=out)


def _print_latency(cfg: BenchConfig, label: str, h: Histogram, out) -> None:
    s = f"[ min={h.min_us} avg={int(h.avg_us)} max={h.max_us} ]"
    print(_fmt_row("", label, "", s), file=out)
    if cfg.lat_percent:
        p = f"[ p50={h.percentile(50)} p75={h.percentile(75)} p99={h.percentile(99)} ]"
        print(_fmt_row("", label + " %ile", "", p), file=out)
        if cfg.lat_percent_9s:
            nines = []
            v = 99.0
            for i in range(cfg.lat_percent_9s):
                v = 99.0 + (1 - 10 ** -(i + 1)) if i else 99.9
                v = float(f"99.{'9' * (i + 1)}")
                nines.append(f"p{v}={h.percentile(v)}")
            print(_fmt_row("", label + " 9s", "", "[ " + " ".join(nines) + " ]"), file=out)
    if cfg.lat_histo:
        buckets = " ".join(f"{lo}us:{c}" for lo, c in h.nonzero_buckets())
        print(_fmt_row("", label + " histo", "", buckets), file=out)


# ---------------------------------------------------------------------------
# CSV output (schema per reference docs/csv-docs.md)
# ---------------------------------------------------------------------------

CSV_COLUMNS = [
    "ISO date", "label", "path type", "paths", "hosts", "threads", "dirs", "files",
    "file size", "block size", "direct IO", "random", "random aligned", "IO depth",
    "shared paths", "truncate", "operation",
    "time ms [first]", "time ms [last]", "CPU% [first]", "CPU% [last]",
    "entries/s [first]", "entries/s [last]", "IOPS [first]", "IOPS [last]",
    "MiB/s [first]", "MiB/s [last]", "entries [first]", "entries [last]",
    "MiB [first]", "MiB [last]",
    "Ent lat us [min]", "Ent lat us [avg]", "Ent lat us [max]",
    "IO lat us [min]", "IO lat us [avg]", "IO lat us [max]",
    "rwmix read entries/s [first]", "rwmix read entries/s [last]",
    "rwmix read IOPS [first]", "rwmix read IOPS [last]",
    "rwmix read MiB/s [first]", "rwmix read MiB/s [last]",
    "rwmix read entries [first]", "rwmix read entries [last]",
    "rwmix read MiB [first]", "rwmix read MiB [last]",
    "rwmix read Ent lat us [min]", "rwmix read Ent lat us [avg]",
    "rwmix read Ent lat us [max]",
    "rwmix read IO lat us [min]", "rwmix read IO lat us [avg]",
    "rwmix read IO lat us [max]",
    "version", "command",
]


def csv_check_compatibility(path: str) -> None:
    """Refuse to append rows to a CSV with a mismatching header (reference
    ProgArgs::checkCSVFileCompatibility, ProgArgs.cpp:4303)."""
    if not os.path.exists(path) or os.path.getsize(path) == 0:
        return
    with open(path, newline="") as f:
        header = next(csv.reader(f), None)
    if header != CSV_COLUMNS:
        raise RuntimeError(
            f"CSV file {path} has an incompatible column set; refusing to append")


def append_csv_result(cfg: BenchConfig, r: PhaseResults, path: str) -> None:
    csv_check_compatibility(path)
    new_file = not os.path.exists(path) or os.path.getsize(path) == 0
    mib = 1024 * 1024
    row = {
        "ISO date": datetime.datetime.now().astimezone().isoformat(timespec="milliseconds"),
        "label": cfg.label,
        "path type": cfg.path_type,
        "paths": len(cfg.paths),
        "hosts": len(cfg.hosts) if cfg.hosts else 1,
        "threads": cfg.threads,
        "dirs": cfg.dirs,
        "files": cfg.files,
        "file size": cfg.file_size,
        "block size": cfg.block_size,
        "direct IO": int(cfg.direct),
        "random": int(cfg.random),
        "random aligned": (int(cfg.rand_aligned) if cfg.random else ""),
        "IO depth": cfg.iodepth,
        "shared paths": int(not cfg.no_svc_share),
        "truncate": int(cfg.truncate),
        "operation": r.phase_name,
        "time ms [first]": r.first_finish_usec // 1000,
        "time ms [last]": r.last_finish_usec // 1000,
        "CPU% [first]": r.cpu_first,
        "CPU% [last]": r.cpu_last,
        "entries/s [first]": r.per_sec_first(r.sw_entries),
        "entries/s [last]": r.per_sec_last(r.entries),
        "IOPS [first]": r.per_sec_first(r.sw_iops),
        "IOPS [last]": r.per_sec_last(r.iops),
        "MiB/s [first]": r.per_sec_first(r.sw_bytes) // mib,
        "MiB/s [last]": r.per_sec_last(r.bytes) // mib,
        "entries [first]": r.sw_entries,
        "entries [last]": r.entries,
        "MiB [first]": r.sw_bytes // mib,
        "MiB [last]": r.bytes // mib,
        "Ent lat us [min]": r.entry_lat.min_us,
        "Ent lat us [avg]": int(r.entry_lat.avg_us),
        "Ent lat us [max]": r.entry_lat.max_us,
        "IO lat us [min]": r.io_lat.min_us,
        "IO lat us [avg]": int(r.io_lat.avg_us),
        "IO lat us [max]": r.io_lat.max_us,
        "version": VERSION,
        "command": " ".join(sys.argv),
    }
    if r.rm_bytes or r.rm_iops or r.rm_entries:
        row.update({
            "rwmix read entries/s [first]": r.per_sec_first(r.rm_sw_entries),
            "rwmix read entries/s [last]": r.per_sec_last(r.rm_entries),
            "rwmix read IOPS [first]": r.per_sec_first(r.rm_sw_iops),
            "rwmix read IOPS [last]": r.per_sec_last(r.rm_iops),
            "rwmix read MiB/s [first]": r.per_sec_first(r.rm_sw_bytes) // mib,
            "rwmix read MiB/s [last]": r.per_sec_last(r.rm_bytes) // mib,
            "rwmix read entries [first]": r.rm_sw_entries,
            "rwmix read entries [last]": r.rm_entries,
            "rwmix read MiB [first]": r.rm_sw_bytes // mib,
            "rwmix read MiB [last]": r.rm_bytes // mib,
            "rwmix read Ent lat us [min]": r.entry_lat_rm.min_us,
            "rwmix read Ent lat us [avg]": int(r.entry_lat_rm.avg_us),
            "rwmix read Ent lat us [max]": r.entry_lat_rm.max_us,
            "rwmix read IO lat us [min]": r.io_lat_rm.min_us,
            "rwmix read IO lat us [avg]": int(r.io_lat_rm.avg_us),
            "rwmix read IO lat us [max]": r.io_lat_rm.max_us,
        })
    for c in CSV_COLUMNS:
        row.setdefault(c, "")
    with open(path, "a", newline="") as f:
        w = csv.DictWriter(f, fieldnames=CSV_COLUMNS)
        if new_file and not cfg.no_csv_labels:
            w.writeheader()
        w.writerow(row)


# ---------------------------------------------------------------------------
# JSON output (shape per reference Statistics::printPhaseResultsAsJSON :2485)
# ---------------------------------------------------------------------------

def phase_results_json(cfg: BenchConfig, r: PhaseResults) -> dict[str, Any]:
    """Reference-schema per-phase JSON document (--jsonfile): key names and
    nesting follow Statistics::printPhaseResultsAsJSON
    (/root/reference/source/Statistics.cpp:2485-2770) — first_done/last_done
    subtrees with entries/s, iops, bytes/s, cpu%, nested rwmix_read and
    latency — plus additive keys (mib_per_sec, percentiles) kept for the
    bundled summarize tool."""
    mib = 1024 * 1024
    start_dt = datetime.datetime.fromtimestamp(r.start_time).astimezone()

    def done_tree(elapsed_usec, entries, iops, nbytes, cpu):
        t: dict[str, Any] = {"elapsed_time_ms": elapsed_usec // 1000}
        per_sec = (lambda v: v * 1_000_000 // elapsed_usec) if elapsed_usec \
            else (lambda v: 0)
        if r.entries:
            t["entries/s"] = per_sec(entries)
            t["entries"] = entries
        if r.iops:
            t["iops"] = per_sec(iops)
        if r.bytes:
            t["bytes/s"] = per_sec(nbytes)
            t["bytes"] = nbytes
            t["mib_per_sec"] = per_sec(nbytes) // mib
        t["entries_per_sec"] = t.get("entries/s", 0)
        t["cpu%"] = cpu
        return t

    first = done_tree(r.first_finish_usec, r.sw_entries, r.sw_iops, r.sw_bytes,
                      r.cpu_first)
    last = done_tree(r.last_finish_usec, r.entries, r.iops, r.bytes, r.cpu_last)

    if r.rm_bytes or r.rm_iops:
        first["rwmix_read"] = {
            "iops": r.per_sec_first(r.rm_sw_iops),
            "bytes/s": r.per_sec_first(r.rm_sw_bytes),
            "bytes": r.rm_sw_bytes,
        }
        last["rwmix_read"] = {
            "iops": r.per_sec_last(r.rm_iops),
            "bytes/s": r.per_sec_last(r.rm_bytes),
            "bytes": r.rm_bytes,
        }

    latency: dict[str, Any] = {}
    if r.entry_lat.num_values:
        latency["entries"] = _lat_json(r.entry_lat)
    if r.io_lat.num_values:
        latency["IO"] = _lat_json(r.io_lat)
    if r.io_lat_rm.num_values:
        latency.setdefault("IO", {})["rwmix_read"] = _lat_json(r.io_lat_rm)
    if cfg.lat_histo:
        if r.entry_lat.num_values:
            latency["entries"]["histogram"] = r.entry_lat.nonzero_buckets()
        if r.io_lat.num_values:
            latency["IO"]["histogram"] = r.io_lat.nonzero_buckets()
    if latency:
        last["latency"] = latency

    doc: dict[str, Any] = {
        "phase_type": r.phase_name,
        "phase_id": r.phase_id,
        "iso_start_date": start_dt.isoformat(timespec="milliseconds"),
        "config": {
            "version": VERSION,
            "command": " ".join(sys.argv),
            "path_type": cfg.path_type,
            "paths": len(cfg.paths),
            "hosts": len(cfg.hosts) if cfg.hosts else 1,
            "threads": cfg.threads,
            "dirs": cfg.dirs,
            "files": cfg.files,
            "file_size": cfg.file_size,
            "block_size": cfg.block_size,
            "direct_io": cfg.direct,
            "random_offsets": cfg.random,
            "random_aligned": cfg.rand_aligned,
            "truncate_files": cfg.truncate,
            "shared_service_paths": not cfg.no_svc_share,
            "io_depth": cfg.iodepth,
            "gpu_ids": cfg.gpu_ids,
        },
        "first_done": first,
        "last_done": last,
    }
    if cfg.label:
        doc["label"] = cfg.label
    if r.errors:
        doc["errors"] = r.errors
    return doc


def _lat_json(h: Histogram) -> dict[str, Any]:
    return {
        "min_us": h.min_us,
        "avg_us": int(h.avg_us),
        "max_us": h.max_us,
        "p50_us": h.percentile(50),
        "p99_us": h.percentile(99),
        "num_values": h.num_values,
    }


def append_json_result(cfg: BenchConfig, r: PhaseResults, path: str) -> None:
    with open(path, "a") as f:
        f.write(json.dumps(phase_results_json(cfg, r)) + "\n")


# ---------------------------------------------------------------------------
# live stats
# ---------------------------------------------------------------------------

class LiveStatsPrinter:
    """Single-line live statistics (reference Statistics.cpp:241 style)."""

    def __init__(self, cfg: BenchConfig, phase_name: str,
                 planned_entries: int, planned_bytes: int, out=None):
        out = out or sys.stderr
        self.cfg = cfg
        self.phase_name = phase_name
        self.planned_entries = planned_entries
        self.planned_bytes = planned_bytes
        self.out = out
        self.last_entries = 0
        self.last_bytes = 0
        self.last_iops = 0
        self.last_t = time.monotonic()
        self.enabled = (not cfg.no_live) and out.isatty()
        self._printed = False

    def update(self, poll: dict[str, Any], worker_rows=None) -> None:
        if not self.enabled:
            return
        now = time.monotonic()
        dt = max(now - self.last_t, 1e-6)
        eps = int((poll["entries"] - self.last_entries) / dt)
        bps = int((poll["bytes"] - self.last_bytes) / dt)
        iops = int((poll["iops"] - self.last_iops) / dt)
        self.last_entries, self.last_bytes = poll["entries"], poll["bytes"]
        self.last_iops, self.last_t = poll["iops"], now

        elapsed_s = poll["elapsed_usec"] // 1_000_000
        parts = [f"{self.phase_name}: {elapsed_s}s"]
        if self.planned_bytes:
            pct = min(100, 100 * poll["bytes"] // self.planned_bytes)
            parts.append(f"{pct}%")
        elif self.planned_entries:
            pct = min(100, 100 * poll["entries"] // self.planned_entries)
            parts.append(f"{pct}%")
        if bps:
            if self.cfg.show_base10:
                parts.append(f"{bps // (1000 * 1000)} MB/s")
            else:
                parts.append(f"{bps // (1024 * 1024)} MiB/s")
        if iops:
            parts.append(f"{iops} IOPS")
        if eps:
            parts.append(f"{eps} Files/s")
        parts.append(f"threads done: {poll['workers_done']}/{poll['workers_total']}")
        line = "; ".join(parts)
        print("\r\x1b[2K" + line, end="", file=self.out, flush=True)
        self._printed = True

    def finish(self) -> None:
        if self.enabled and self._printed:
            print("\r\x1b[2K", end="", file=self.out, flush=True)


class LiveCsvWriter:
    """Time-series live stats CSV (reference --livecsv, Statistics.cpp:3000)."""

    COLS = ["ISO date", "label", "operation", "elapsed ms", "entries", "entries/s",
            "IOPS", "MiB/s", "bytes", "threads done"]

    def __init__(self, path: str, cfg: BenchConfig, phase_name: str):
        self.path = path
        self.cfg = cfg
        self.phase_name = phase_name
        self.last = (0, 0, 0, time.monotonic())
        new = not os.path.exists(path) or os.path.getsize(path) == 0
        self.f = open(path, "a", newline="")
        self.w = csv.writer(self.f)
        if new and not cfg.no_csv_labels:
            self.w.writerow(self.COLS)

    def update(self, poll: dict[str, Any]) -> None:
        now = time.monotonic()
        le, lb, li, lt = self.last
        dt = max(now - lt, 1e-6)
        self.w.writerow([
            datetime.datetime.now().astimezone().isoformat(timespec="milliseconds"),
            self.cfg.label, self.phase_name, poll["elapsed_usec"] // 1000,
            poll["entries"], int((poll["entries"] - le) / dt),
            int((poll["iops"] - li) / dt),
            int((poll["bytes"] - lb) / dt) // (1024 * 1024),
            poll["bytes"], poll["workers_done"],
        ])
        self.f.flush()
        self.last = (poll["entries"], poll["bytes"], poll["iops"], now)

    def close(self) -> None:
        self.f.close()
