"""Post-processing tools: chart, summarize-json, scan-path."""

import subprocess
import sys
import os

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_cli(args):
    from elbencho_amd.cli import main
    return main(args)


def test_summarize_json(tmp_path):
    f = tmp_path / "file1"
    jsonf = tmp_path / "res.json"
    assert _run_cli(["-w", "-r", "-t", "2", "-b", "64k", "-s", "1m", "--nolive",
                     "--jsonfile", str(jsonf), str(f)]) == 0
    res = subprocess.run([sys.executable, os.path.join(REPO, "tools",
                                                       "elbencho-amd-summarize-json"),
                          str(jsonf)], capture_output=True, text=True)
    assert res.returncode == 0, res.stderr
    assert "WRITE" in res.stdout and "READ" in res.stdout


def test_chart(tmp_path):
    f = tmp_path / "file1"
    csvf = tmp_path / "res.csv"
    assert _run_cli(["-w", "-r", "-t", "2", "-b", "64k", "-s", "1m", "--nolive",
                     "--csvfile", str(csvf), str(f)]) == 0
    res = subprocess.run([sys.executable, os.path.join(REPO, "tools", "elbencho-amd-chart"),
                          str(csvf), "--gnuplot", str(tmp_path / "out")],
                         capture_output=True, text=True)
    assert res.returncode == 0, res.stderr
    assert "#" in res.stdout  # ASCII bars
    assert (tmp_path / "out.gnuplot").exists()
    assert (tmp_path / "out.dat").exists()


def test_scan_path_tool(tmp_path):
    src = tmp_path / "src"
    (src / "d").mkdir(parents=True)
    (src / "d" / "f").write_bytes(b"x" * 123)
    out = tmp_path / "t.tree"
    res = subprocess.run([sys.executable, os.path.join(REPO, "tools",
                                                       "elbencho-amd-scan-path"),
                          str(src), str(out)], capture_output=True, text=True)
    assert res.returncode == 0, res.stderr
    assert "1 dirs, 1 files, 123 bytes" in res.stdout


def test_launcher_script():
    res = subprocess.run([os.path.join(REPO, "bin", "elbencho-amd"), "--version"],
                         capture_output=True, text=True)
    assert res.returncode == 0
    assert "elbencho-amd" in res.stdout


def test_hdfs_unsupported(tmp_path):
    assert _run_cli(["--hdfs", "-w", "-s", "1m", "--nolive", str(tmp_path / "f")]) == 1
