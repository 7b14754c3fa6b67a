#!/usr/bin/env python3
"""Judged config 5: dedup ingest — N overlapping files (same content with a
distinct prefix each) through the engine; reports ingest GiB/s and the dedup
ratio from the fingerprint index."""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from edgehip.lib import Engine  # noqa: E402
from tests.conftest import rand_bytes  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--files", type=int, default=10)
    ap.add_argument("--gib-per-file", type=int, default=2)
    ap.add_argument("--ngpus", type=int, default=1)
    args = ap.parse_args()

    mb = 1 << 20
    per = args.gib_per_file << 30
    base = memoryview(bytearray(rand_bytes(256 * mb, seed=7)))  # shared ring
    # (memoryview slices pass to ctypes zero-copy; bytes slicing would copy
    # under the GIL and serialize the concurrent writers)

    import threading

    with Engine(f"engine=gpu,ngpus={args.ngpus},fp=blake3,compress=lz4,dedup=1,"
                "arena=16g,spill=0") as eng:
        t0 = time.perf_counter()
        objs = [eng.object() for _ in range(args.files)]
        errs = []

        def ingest(f):
            # fio-style: all files ingest concurrently (ctypes drops the GIL,
            # so writer memcpys and engine batches overlap across files)
            try:
                o = objs[f]
                # distinct 1 MiB head, identical body => ~100% dedup vs file 0
                o.write(0, rand_bytes(mb, seed=1000 + f))
                for off in range(mb, per, 64 * mb):
                    n = min(64 * mb, per - off)
                    src_off = (off - mb) % (192 * mb)
                    o.write(off, base[src_off:src_off + n])
                o.flush()
            except Exception as e:  # pragma: no cover
                errs.append(e)

        # seed: file 0 alone commits the shared chunks; the rest then ingest
        # concurrently and dedup against COMMITTED entries (concurrent
        # identical batches would all publish pending entries and convoy on
        # each other's pipeline syncs — measured 3x slower than seeding)
        ingest(0)
        ts = [threading.Thread(target=ingest, args=(f,))
              for f in range(1, args.files)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        if errs:
            raise errs[0]
        dt = time.perf_counter() - t0
        logical = args.files * per
        stored = eng.stored_bytes
        print(f"dedup ingest: {logical / (1 << 30) / dt:.2f} GiB/s logical "
              f"({dt:.1f}s, {args.files} x {args.gib_per_file} GiB)")
        print(f"stored {stored / (1 << 30):.2f} GiB, dedup ratio "
              f"{logical / max(1, stored):.1f}x, chunks {eng.stored_chunks}")
        for o in objs:
            o.close()


if __name__ == "__main__":
    main()
