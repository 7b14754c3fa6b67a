#!/usr/bin/env python3
"""Judged config 2: 4 KiB random reads from a 1 GiB object resident in the
HBM chunk cache (1 MI355X). Reports IOPS + GiB/s. Uses the object API when
the box denies /dev/fuse (same data path minus the kernel hop)."""

import argparse
import random
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from edgehip.lib import Engine  # noqa: E402
from tests.conftest import rand_bytes  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size-gib", type=int, default=1)
    ap.add_argument("--seconds", type=float, default=10.0)
    ap.add_argument("--jobs", type=int, default=8)
    ap.add_argument("--qd", type=int, default=32, help="reads per engine batch per job")
    args = ap.parse_args()

    import threading

    with Engine("engine=gpu,fp=blake3,compress=lz4,dedup=1,arena=8g,spill=0") as eng:
        o = eng.object()
        mb = 1 << 20
        total = args.size_gib << 30
        print(f"populating {args.size_gib} GiB...", file=sys.stderr)
        data = rand_bytes(64 * mb, seed=1)
        for off in range(0, total, 64 * mb):
            o.write(off, data if off else rand_bytes(64 * mb, seed=off))
        o.flush()

        stop = time.perf_counter() + args.seconds
        counts = [0] * args.jobs

        def job(j):
            rng = random.Random(j)
            n = 0
            buf = bytearray(args.qd * 4096)
            lens = [4096] * args.qd
            while time.perf_counter() < stop:
                offs = [rng.randrange(0, (total - 4096) // 4096) * 4096
                        for _ in range(args.qd)]
                got = o.preadv(offs, lens, buf)
                assert got == args.qd * 4096
                n += args.qd
            counts[j] = n

        ts = [threading.Thread(target=job, args=(j,)) for j in range(args.jobs)]
        t0 = time.perf_counter()
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        dt = time.perf_counter() - t0
        iops = sum(counts) / dt
        print(f"randread 4k: {iops:,.0f} IOPS, {iops * 4096 / (1 << 30):.3f} GiB/s "
              f"({args.jobs} jobs, {dt:.1f}s)")
        o.close()


if __name__ == "__main__":
    main()
