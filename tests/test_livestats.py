"""Live stats displays (fullscreen/newline/live CSV ex) against a fake tty."""

import io

from elbencho_amd.config import BenchConfig
from elbencho_amd.livestats import (FullscreenLiveStats, LiveCsvExWriter,
                                    NewlineLiveStats)


class FakeTty(io.StringIO):
    def isatty(self):
        return True


def _poll(entries=10, bytes_=1 << 20, iops=16, done=1, total=4):
    return {"entries": entries, "bytes": bytes_, "iops": iops,
            "workers_done": done, "workers_total": total,
            "elapsed_usec": 2_000_000, "stonewall_triggered": False}


def test_fullscreen_renders_worker_rows():
    cfg = BenchConfig()
    out = FakeTty()
    fs = FullscreenLiveStats(cfg, "WRITE", planned_bytes=4 << 20,
                             planned_entries=0, out=out)
    rows = [{"rank": 0, "entries": 5, "bytes": 1 << 19, "iops": 8},
            {"rank": 1, "entries": 5, "bytes": 1 << 19, "iops": 8}]
    fs.update(_poll(), rows)
    fs.update(_poll(bytes_=2 << 20), rows)
    fs.finish()
    s = out.getvalue()
    assert "Phase: WRITE" in s
    assert "RANK" in s
    assert "done: 1/4" in s


def test_fullscreen_disabled_without_tty():
    cfg = BenchConfig()
    out = io.StringIO()
    fs = FullscreenLiveStats(cfg, "WRITE", 0, 0, out=out)
    fs.update(_poll(), [])
    assert out.getvalue() == ""


def test_newline_mode():
    cfg = BenchConfig()
    out = io.StringIO()
    nl = NewlineLiveStats(cfg, "READ", out=out)
    nl.update(_poll())
    nl.update(_poll(bytes_=3 << 20))
    lines = out.getvalue().strip().splitlines()
    assert len(lines) == 2
    assert all(ln.startswith("READ:") for ln in lines)


def test_livecsvex(tmp_path):
    cfg = BenchConfig()
    path = tmp_path / "live.csv"
    w = LiveCsvExWriter(str(path), cfg, "WRITE")
    rows = [{"rank": 0, "entries": 1, "bytes": 100, "iops": 2},
            {"rank": 1, "entries": 2, "bytes": 200, "iops": 4}]
    w.update(_poll(), rows)
    w.update(_poll(), rows)
    w.close()
    content = path.read_text().splitlines()
    assert content[0].startswith("ISO date")
    assert len(content) == 1 + 4  # header + 2 ticks x 2 workers


def test_fullscreen_svcping_column():
    """--svcping adds a PING us column to per-service rows."""
    cfg = BenchConfig()
    cfg.svc_ping = True
    out = FakeTty()
    fs = FullscreenLiveStats(cfg, "WRITE", planned_bytes=4 << 20,
                             planned_entries=0, out=out)
    rows = [{"rank": 0, "entries": 5, "bytes": 1 << 19, "iops": 8,
             "ping_us": 321}]
    fs.update(_poll(), rows)
    fs.finish()
    s = out.getvalue()
    assert "PING us" in s
    assert "321" in s
