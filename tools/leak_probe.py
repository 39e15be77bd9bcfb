import os, subprocess, sys
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tests")

def vram():
    out = subprocess.run(["rocm-smi", "--showmeminfo", "vram", "--csv"],
                         capture_output=True, text=True).stdout
    for line in out.splitlines():
        if line.startswith("card"):
            return int(line.split(",")[2])
    return -1

from elbencho_amd.cli import main

base = "/dev/shm/leakp"
os.makedirs(base + "/dirs", exist_ok=True)

def run(tag, args, n=30):
    v0 = vram()
    for i in range(n):
        rc = main(args + ["--nolive"])
        assert rc == 0, (tag, rc)
    v1 = vram()
    print(f"{tag}: {(v1-v0)/n/1024:.0f} KiB/run (v0={v0>>20}M v1={v1>>20}M)", flush=True)

run("dirmode+gpu+verify", ["-t", "4", "-d", "-n", "2", "-w", "-r", "-N", "4",
    "-s", "1m", "-b", "256k", "--verify", "1", "--gpuids", "0", "-F", "-D",
    base + "/dirs"])
run("file+gpu mmap", ["-w", "-r", "-t", "8", "-b", "4m", "-s", "256m",
    "--gpuids", "0", "--mmap", base + "/big"])
run("file rand qd32 gpu", ["-r", "-t", "8", "--iodepth", "32", "-b", "4k",
    "--rand", "--randamount", "64m", "--gpuids", "0", base + "/big"])
from s3mock import ACCESS_KEY, SECRET_KEY, start_mock
server, port = start_mock()
run("s3 gpu verify", ["--s3endpoints", f"http://127.0.0.1:{port}",
    "--s3key", ACCESS_KEY, "--s3secret", SECRET_KEY, "-d", "-w", "-r", "-F",
    "-D", "-t", "4", "-N", "2", "-s", "16m", "-b", "4m", "--verify", "5",
    "--gpuids", "0", "s3://leakbkt"])
server.shutdown()
