#!/usr/bin/env python3
"""4K random-read IOPS micro-benchmark of the engine's staging paths.

Used for the measurements in profiles/r01_uring_iops.md: warm repeated READ
passes over a tmpfs file, comparing the sync half-ring batched staging path
(iodepth 1) against io_uring at several queue depths, CPU vs GPU-staged.

Usage: python3 tools/dbg_iops.py [passes-per-config]
"""

from __future__ import annotations

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from elbencho_amd import load_core  # noqa: E402

PATH = "/dev/shm/dbg_iops.bin"
SIZE = 1024 * 1024 * 1024


def main() -> int:
    reps = int(sys.argv[1]) if len(sys.argv) > 1 else 3
    core = load_core()

    weng = core.Engine(dict(paths=[PATH], path_type="file", threads=8,
                            num_dataset_threads=8, file_size=SIZE,
                            block_size=1 << 20))
    weng.prepare()
    weng.start_phase(core.PHASES["WRITE"])
    weng.wait_phase_done(-1)
    weng.finish_phase()

    base = dict(paths=[PATH], path_type="file", threads=16,
                num_dataset_threads=16, file_size=SIZE, block_size=4096,
                random=True, blockvar_pct=0)
    have_gpu = core.gpu_device_count() > 0

    configs = [("cpu sync qd1", dict(base, iodepth=1)),
               ("cpu uring qd128", dict(base, iodepth=128))]
    if have_gpu:
        configs += [("gpu sync qd1", dict(base, iodepth=1, gpu_ids=[0])),
                    ("gpu uring qd32", dict(base, iodepth=32, gpu_ids=[0])),
                    ("gpu uring qd128", dict(base, iodepth=128, gpu_ids=[0]))]

    for tag, cfg in configs:
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
            best = max(best, sum(r["iops"] for r in res) / dt)
        print(f"{tag}: best {best / 1e6:.2f} M IOPS", flush=True)

    os.unlink(PATH)
    return 0


if __name__ == "__main__":
    sys.exit(main())
