"""VRAM trajectory probe: repeat ONE workload config many times and print
the VRAM level every 20 runs — distinguishes a linear leak from the HIP
allocator's per-config arena warm-up plateau."""
import os
import subprocess
import sys

sys.path.insert(0, "/root/repo")


def vram() -> int:
    out = subprocess.run(["rocm-smi", "--showmeminfo", "vram", "--csv"],
                         capture_output=True, text=True).stdout
    for line in out.splitlines():
        if line.startswith("card"):
            return int(line.split(",")[2])
    return -1


from elbencho_amd.cli import main  # noqa: E402

base = "/dev/shm/leakp"
os.makedirs(base + "/dirs", exist_ok=True)
args = ["-t", "4", "-d", "-n", "2", "-w", "-r", "-N", "4", "-s", "1m",
        "-b", "256k", "--verify", "1", "--gpuids", "0", "-F", "-D",
        base + "/dirs", "--nolive"]
n = int(sys.argv[1]) if len(sys.argv) > 1 else 200
for i in range(n):
    assert main(args) == 0
    if i % 20 == 0:
        print(f"run {i}: vram {vram() >> 20} MiB", flush=True)
print(f"run {n}: vram {vram() >> 20} MiB", flush=True)
