"""Live statistics displays: fullscreen per-worker dashboard, single-line,
newline mode, and extended per-worker live CSV.

Reference analogue: the Statistics live-stats family
(/root/reference/source/Statistics.cpp — fullscreen ftxui table :716-1249,
single line :241, --livecsv(ex) :3000-3290). Independent implementation on
ANSI escape sequences (no curses dependency) so it works on any terminal.
"""

from __future__ import annotations

import csv
import datetime
import os
import sys
import time
from typing import Any, Optional

from elbencho_amd.config import BenchConfig


class FullscreenLiveStats:
    """Per-worker (or per-service) live dashboard, redrawn per tick."""

    def __init__(self, cfg: BenchConfig, phase_name: str, planned_bytes: int,
                 planned_entries: int, out=None):
        self.cfg = cfg
        self.phase = phase_name
        self.planned_bytes = planned_bytes
        self.planned_entries = planned_entries
        self.out = out or sys.stderr
        self.enabled = self.out.isatty() and not cfg.no_live
        self.last_rows: dict[int, tuple[int, int, int]] = {}
        self.last_total = (0, 0, 0)
        self.last_t = time.monotonic()
        self._drawn_lines = 0

    def update(self, poll: dict[str, Any],
               worker_rows: Optional[list[dict[str, Any]]] = None) -> None:
        if not self.enabled:
            return
        now = time.monotonic()
        dt = max(now - self.last_t, 1e-6)
        mib = 1024 * 1024

        lines = []
        elapsed_s = poll["elapsed_usec"] // 1_000_000
        pct = ""
        if self.planned_bytes:
            pct = f" {min(100, 100 * poll['bytes'] // self.planned_bytes)}%"
        elif self.planned_entries:
            pct = f" {min(100, 100 * poll['entries'] // self.planned_entries)}%"
        bps = int((poll["bytes"] - self.last_total[1]) / dt)
        iops = int((poll["iops"] - self.last_total[2]) / dt)
        lines.append(f"Phase: {self.phase}{pct}  Elapsed: {elapsed_s}s  "
                     f"{bps // mib} MiB/s  {iops} IOPS  "
                     f"done: {poll['workers_done']}/{poll['workers_total']}")
        self.last_total = (poll["entries"], poll["bytes"], poll["iops"])

        if worker_rows:
            with_ping = any("ping_us" in row for row in worker_rows)
            hdr = (f"{'RANK':>5} {'ENTRIES':>10} {'MiB':>10} {'MiB/s':>10} "
                   f"{'IOPS':>10}")
            lines.append(hdr + (f" {'PING us':>9}" if with_ping else ""))
            for row in worker_rows[:40]:  # cap at terminal-ish height
                rank = row["rank"]
                prev = self.last_rows.get(rank, (0, 0, 0))
                wbps = int((row["bytes"] - prev[1]) / dt)
                wiops = int((row["iops"] - prev[2]) / dt)
                line = (f"{rank:>5} {row['entries']:>10} "
                        f"{row['bytes'] // mib:>10} {wbps // mib:>10} {wiops:>10}")
                if with_ping:
                    line += f" {row.get('ping_us', 0):>9}"
                lines.append(line)
                self.last_rows[rank] = (row["entries"], row["bytes"], row["iops"])

        # redraw in place
        buf = ""
        if self._drawn_lines:
            buf += f"\x1b[{self._drawn_lines}F"  # cursor up to redraw start
        for ln in lines:
            buf += "\x1b[2K" + ln + "\n"
        self.out.write(buf)
        self.out.flush()
        self._drawn_lines = len(lines)
        self.last_t = now

    def finish(self) -> None:
        if self.enabled and self._drawn_lines:
            # clear the dashboard area
            buf = f"\x1b[{self._drawn_lines}F"
            buf += "\x1b[2K\n" * self._drawn_lines
            buf += f"\x1b[{self._drawn_lines}F"
            self.out.write(buf)
            self.out.flush()
            self._drawn_lines = 0


class NewlineLiveStats:
    """--live1n: one line per update on its own row (for logs/pipes)."""

    def __init__(self, cfg: BenchConfig, phase_name: str, out=None):
        self.phase = phase_name
        self.out = out or sys.stderr
        self.last = (0, 0, 0, time.monotonic())
        self.enabled = not cfg.no_live

    def update(self, poll: dict[str, Any], worker_rows=None) -> None:
        if not self.enabled:
            return
        now = time.monotonic()
        le, lb, li, lt = self.last
        dt = max(now - lt, 1e-6)
        mib = 1024 * 1024
        print(f"{self.phase}: {poll['elapsed_usec'] // 1_000_000}s; "
              f"{int((poll['bytes'] - lb) / dt) // mib} MiB/s; "
              f"{int((poll['iops'] - li) / dt)} IOPS; "
              f"{int((poll['entries'] - le) / dt)} entries/s; "
              f"done {poll['workers_done']}/{poll['workers_total']}",
              file=self.out, flush=True)
        self.last = (poll["entries"], poll["bytes"], poll["iops"], now)

    def finish(self) -> None:
        pass


class LiveCsvExWriter:
    """--livecsvex: per-worker time series rows."""

    COLS = ["ISO date", "label", "operation", "elapsed ms", "rank", "entries",
            "bytes", "MiB/s", "IOPS"]

    def __init__(self, path: str, cfg: BenchConfig, phase_name: str):
        self.cfg = cfg
        self.phase = phase_name
        new = not os.path.exists(path) or os.path.getsize(path) == 0
        self.f = open(path, "a", newline="")
        self.w = csv.writer(self.f)
        if new and not cfg.no_csv_labels:
            self.w.writerow(self.COLS)
        self.last: dict[int, tuple[int, int, float]] = {}

    def update(self, poll: dict[str, Any], worker_rows) -> None:
        now = time.monotonic()
        ts = datetime.datetime.now().astimezone().isoformat(timespec="milliseconds")
        for row in worker_rows or []:
            rank = row["rank"]
            lb, li, lt = self.last.get(rank, (0, 0, now - 1))
            dt = max(now - lt, 1e-6)
            self.w.writerow([
                ts, self.cfg.label, self.phase, poll["elapsed_usec"] // 1000, rank,
                row["entries"], row["bytes"],
                int((row["bytes"] - lb) / dt) // (1024 * 1024),
                int((row["iops"] - li) / dt),
            ])
            self.last[rank] = (row["bytes"], row["iops"], now)
        self.f.flush()

    def close(self) -> None:
        self.f.close()
