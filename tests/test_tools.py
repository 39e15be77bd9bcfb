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


def test_sweep_dry_run(tmp_path):
    """elbencho-amd-sweep (contrib/storage_sweep analogue): dry run prints one
    command per power-of-two point with scaled file counts."""
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "elbencho-amd-sweep"), str(tmp_path),
         "-r", "s", "--budget", "8K", "-t", "2", "-n"],
        capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    lines = [ln for ln in out.stdout.splitlines() if ln.strip()]
    assert len(lines) == 4  # 1K, 2K, 4K, 8K points within the 8K budget
    assert "-s 1024" in lines[0] and "--dirsharing" in lines[0]
    assert "--direct" in lines[-1]      # >= fs block size
    assert "--direct" not in lines[0]   # tiny files skip O_DIRECT


def test_sweep_real_micro(tmp_path):
    csvf = tmp_path / "sweep.csv"
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "elbencho-amd-sweep"),
         str(tmp_path / "bench"), "-r", "s", "--budget", "4K", "-t", "1",
         "-B", "--csv", str(csvf), "--plot", str(tmp_path / "plot")],
        capture_output=True, text=True)
    assert out.returncode == 0, out.stderr + out.stdout
    assert "Gbps" in out.stdout
    assert csvf.exists()
    assert (tmp_path / "plot.gp").exists()
