"""GPU engine concurrency soak (docs/concurrency.md): concurrent writers and
readers over a deliberately small arena force every hazardous interleaving at
once — pending-batch publishes, LRU eviction, spill reloads, cross-thread
dedup hits — and every byte read back must be exact."""

import threading

import pytest

from edgehip.lib import Engine
from tests.conftest import rand_bytes

MB = 1 << 20

pytestmark = pytest.mark.gpu


def test_concurrent_writers_spill_verify():
    # 6 writers × 96 MiB over a 128 MiB arena: constant eviction while
    # batches are publishing. Content unique per (job, offset).
    with Engine("engine=gpu,fp=blake3,compress=lz4,dedup=1,arena=128m,"
                "spill=2g") as eng:
        errs = []

        def job(j):
            try:
                with eng.object() as o:
                    data = rand_bytes(96 * MB, seed=100 + j)
                    for off in range(0, len(data), MB):
                        o.write(off, data[off:off + MB])
                    o.flush()
                    got = o.read(0, len(data))
                    assert got == data, f"job {j}: readback mismatch"
            except Exception as e:  # pragma: no cover
                errs.append(e)

        ts = [threading.Thread(target=job, args=(j,)) for j in range(6)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        assert not errs, errs[0]


def test_concurrent_shared_content_dedup():
    # All writers stream the SAME content: cross-thread dedup hits land on
    # other batches' pending entries (the cross_fixups path); refcounts must
    # keep the shared chunks alive until the last object is closed.
    with Engine("engine=gpu,fp=blake3,compress=none,dedup=1,arena=512m,"
                "spill=0") as eng:
        shared = rand_bytes(64 * MB, seed=7)
        errs = []

        def job(j):
            try:
                with eng.object() as o:
                    for off in range(0, len(shared), MB):
                        o.write(off, shared[off:off + MB])
                    o.flush()
                    assert o.read(0, len(shared)) == shared, f"job {j}"
            except Exception as e:  # pragma: no cover
                errs.append(e)

        ts = [threading.Thread(target=job, args=(j,)) for j in range(6)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        assert not errs, errs[0]
        # 6 objects, one physical copy
        assert eng.stored_chunks == 64
        assert eng.stored_bytes <= 64 * MB + (1 << 20)


def test_readers_race_eviction():
    # Writers keep evicting while readers hammer random offsets of a
    # committed object: classify-revalidate (spilled mid-read) and the
    # pending wait must never return stale or short data.
    with Engine("engine=gpu,fp=blake3,compress=none,dedup=0,arena=192m,"
                "spill=2g") as eng:
        with eng.object() as committed:
            base = rand_bytes(128 * MB, seed=42)
            for off in range(0, len(base), MB):
                committed.write(off, base[off:off + MB])
            committed.flush()

            stop = threading.Event()
            errs = []

            def churn(j):
                try:
                    with eng.object() as o:
                        k = 0
                        while not stop.is_set():
                            data = rand_bytes(8 * MB, seed=1000 + j * 97 + k)
                            o.write(0, data)
                            o.flush()
                            k += 1
                except Exception as e:  # pragma: no cover
                    errs.append(e)

            def reader(j):
                try:
                    import random
                    rng = random.Random(j)
                    for _ in range(200):
                        off = rng.randrange(0, 127) * MB
                        assert committed.read(off, MB) == base[off:off + MB]
                except Exception as e:  # pragma: no cover
                    errs.append(e)

            cs = [threading.Thread(target=churn, args=(j,)) for j in range(2)]
            rs = [threading.Thread(target=reader, args=(j,)) for j in range(4)]
            for t in cs + rs:
                t.start()
            for t in rs:
                t.join()
            stop.set()
            for t in cs:
                t.join()
            assert not errs, errs[0]


def test_dedup_off_duplicate_content_spill():
    """dedup=0 with repeated content: same-fp chunks must share one map
    entry (content-addressed) — a silent emplace failure here corrupted
    refcounts and broke reads under spill (round-1 regression)."""
    with Engine("engine=gpu,fp=blake3,compress=lz4,dedup=0,arena=128m,"
                "spill=2g") as eng:
        ring = rand_bytes(32 * MB, seed=77)
        objs = [eng.object() for _ in range(3)]
        try:
            for j, o in enumerate(objs):
                for off in range(0, 96 * MB, MB):
                    src = (off + j * MB) % (32 * MB)
                    o.write(off, ring[src:src + MB])
                o.flush()
            for j, o in enumerate(objs):
                for off in range(0, 96 * MB, 7 * MB):
                    src = (off + j * MB) % (32 * MB)
                    assert o.read(off, MB) == ring[src:src + MB], (j, off)
        finally:
            for o in objs:
                o.truncate(0)
                o.close()
