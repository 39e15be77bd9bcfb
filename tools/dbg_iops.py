"""Warm-engine 4K random-read IOPS: sync (iodepth 1) vs uring batched
(iodepth 32/128), 16 threads, GPU-staged."""
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


def run(tag, cfg, reps=3):
    eng = core.Engine(cfg)
    eng.prepare()
    best = 0.0
    for _ in range(reps):
        t0 = time.monotonic()
        eng.start_phase(core.PHASES["READ"])
        eng.wait_phase_done(-1)
        res = eng.finish_phase()
        dt = time.monotonic() - t0
        errs = [r["error"] for r in res if r["error"]]
        assert not errs, errs
        iops = sum(r["iops"] for r in res)
        best = max(best, iops / dt)
    print(f"{tag}: best {best/1e6:.2f} M IOPS", flush=True)


base = dict(paths=[PATH], path_type="file", threads=16,
            num_dataset_threads=16, file_size=SIZE, block_size=4096,
            random=True, gpu_ids=[0], blockvar_pct=0)
run("gpu sync qd1   t16", dict(base, iodepth=1))
run("gpu uring qd32 t16", dict(base, iodepth=32))
run("gpu uring qd128 t16", dict(base, iodepth=128))
nog = {k: v for k, v in base.items() if k != "gpu_ids"}
run("cpu uring qd128 t16", dict(nog, iodepth=128))
run("cpu sync qd1   t16", dict(nog, iodepth=1))

import os
os.environ["EB_GPU_BATCH_BYTES"] = "0"  # disable half-ring batching
run("gpu uring qd128 t16 UNBATCHED", dict(base, iodepth=128))
del os.environ["EB_GPU_BATCH_BYTES"]
os.environ["EB_URING_NOFIXED"] = "1"
run("gpu uring qd128 t16 nofixedbuf", dict(base, iodepth=128))
del os.environ["EB_URING_NOFIXED"]
