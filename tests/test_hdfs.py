"""WebHDFS engine tests against the in-memory mock (with 307 redirects)."""

import pytest

from elbencho_amd.cli import main
from elbencho_amd.hdfs import HdfsError, WebHdfsClient, parse_hdfs_path

from tests.webhdfsmock import WebHdfsHandler, start_mock


@pytest.fixture
def mock_hdfs():
    server, port = start_mock()
    yield port
    server.shutdown()


def test_parse_hdfs_path():
    assert parse_hdfs_path("hdfs://nn:9870/bench/x") == ("nn", 9870, "/bench/x")
    assert parse_hdfs_path("hdfs://nn/b") == ("nn", 9870, "/b")
    with pytest.raises(HdfsError):
        parse_hdfs_path("/not/hdfs")


def test_client_roundtrip(mock_hdfs):
    c = WebHdfsClient("127.0.0.1", mock_hdfs)
    c.mkdirs("/bench/d1")
    c.create("/bench/d1/f1", b"hello ")
    c.append("/bench/d1/f1", b"world")
    assert c.open("/bench/d1/f1") == b"hello world"
    assert c.open("/bench/d1/f1", offset=6, length=5) == b"world"
    assert c.status("/bench/d1/f1")["length"] == 11
    assert c.delete("/bench/d1/f1")
    with pytest.raises(HdfsError):
        c.open("/bench/d1/f1")
    c.close()


def test_hdfs_full_lifecycle_cli(mock_hdfs, capsys):
    """mkdirs -> write (create+append blocks) -> stat -> verified read ->
    delete files -> delete dirs, through the reference CLI surface."""
    rc = main(["-d", "-w", "--stat", "-r", "-F", "-D", "-t", "2", "-n", "1",
               "-N", "3", "-s", "96k", "-b", "32k", "--verify", "7", "--lat",
               "--nolive", f"hdfs://127.0.0.1:{mock_hdfs}/bench"])
    assert rc == 0
    out = capsys.readouterr().out
    assert "MKDIRS" in out and "WRITE" in out and "READ" in out
    for line in out.splitlines():
        if "Files total" in line:
            assert line.split()[-1] == "6"  # 2 threads x 1 dir x 3 files
            break
    else:
        raise AssertionError("no Files total row:\n" + out)
    with WebHdfsHandler.store.lock:
        assert not WebHdfsHandler.store.files  # all deleted


def test_hdfs_verify_detects_corruption(mock_hdfs):
    rc = main(["-d", "-w", "-t", "1", "-N", "1", "-s", "32k", "-b", "32k",
               "--verify", "3", "--nolive",
               f"hdfs://127.0.0.1:{mock_hdfs}/vbench"])
    assert rc == 0
    with WebHdfsHandler.store.lock:
        path = next(iter(WebHdfsHandler.store.files))
        WebHdfsHandler.store.files[path][100] ^= 0xFF
    rc = main(["-r", "-t", "1", "-N", "1", "-s", "32k", "-b", "32k",
               "--verify", "3", "--nolive",
               f"hdfs://127.0.0.1:{mock_hdfs}/vbench"])
    assert rc != 0


def test_hdfs_no_redirect_servers(mock_hdfs):
    """Servers that answer data ops directly (no 307) also work."""
    WebHdfsHandler.redirect_data_ops = False
    try:
        c = WebHdfsClient("127.0.0.1", mock_hdfs)
        c.create("/direct/f", b"abc")
        assert c.open("/direct/f") == b"abc"
        c.close()
    finally:
        WebHdfsHandler.redirect_data_ops = True


def test_hdfs_requires_hdfs_path(tmp_path):
    rc = main(["--hdfs", "-w", "-t", "1", "-N", "1", "-s", "4k", "--nolive",
               str(tmp_path)])
    assert rc != 0  # clear config error, not a crash


def test_hdfs_rwmix_dedicated_readers(mock_hdfs, capsys):
    rc = main(["-d", "-w", "-t", "2", "-N", "2", "-s", "32k", "-b", "32k",
               "--verify", "4", "--nolive",
               f"hdfs://127.0.0.1:{mock_hdfs}/mix"])
    assert rc == 0
    capsys.readouterr()
    rc = main(["-w", "-t", "2", "--rwmixthr", "1", "-N", "2", "-s", "32k",
               "-b", "32k", "--verify", "4", "--nolive",
               f"hdfs://127.0.0.1:{mock_hdfs}/mix"])
    assert rc == 0
    out = capsys.readouterr().out
    assert "MiB/s read" in out or "IOPS read" in out
