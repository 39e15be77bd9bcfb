"""Does core binding stabilize the 4K random-read IOPS at the fast mode?"""
import time

from elbencho_amd import load_core

core = load_core()
PATH = "/dev/shm/dbg_iops.bin"
SIZE = 1024 * 1024 * 1024

weng = core.Engine(dict(paths=[PATH], path_type="file", threads=8,
                        num_dataset_threads=8, file_size=SIZE,
                        block_size=1 << 20))
weng.prepare()
weng.start_phase(core.PHASES["WRITE"])
weng.wait_phase_done(-1)
weng.finish_phase()

import os
ncpu = os.cpu_count()
print(f"ncpu={ncpu}")

base = dict(paths=[PATH], path_type="file", threads=16, num_dataset_threads=16,
            file_size=SIZE, block_size=4096, random=True, gpu_ids=[0],
            blockvar_pct=0, iodepth=1)

for tag, cfg in [
    ("unbound", base),
    ("cores 0-15", dict(base, cores=list(range(16)))),
    ("cores even 0-30", dict(base, cores=list(range(0, 32, 2)))),
]:
    eng = core.Engine(cfg)
    eng.prepare()
    times = []
    for i in range(6):
        t0 = time.monotonic()
        eng.start_phase(core.PHASES["READ"])
        eng.wait_phase_done(-1)
        res = eng.finish_phase()
        dt = time.monotonic() - t0
        iops = sum(r["iops"] for r in res)
        times.append(iops / dt / 1e6)
    print(f"{tag}: " + " ".join(f"{x:.1f}" for x in times) + " M IOPS", flush=True)
