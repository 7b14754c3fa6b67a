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
                import errno as errno_mod
                import time as time_mod
                with eng.object() as o:
                    data = rand_bytes(96 * MB, seed=100 + j)
                    for off in range(0, len(data), MB):
                        # transient ENOSPC is legitimate backpressure when 6
                        # writers outrun eviction on a cold box (a real app
                        # retries); CORRUPTION below still fails hard
                        for attempt in range(4):
                            try:
                                o.write(off, data[off:off + MB])
                                break
                            except OSError as e:
                                if e.errno != errno_mod.ENOSPC or attempt == 3:
                                    raise
                                time_mod.sleep(0.3)
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


def test_threaded_random_ops_stress_gpu():
    """Randomized threaded op mix against the GPU engine with a small arena
    and spill: write/read/punch/truncate/copy_range per-object models under
    constant eviction. Seeds in failure messages."""
    import random

    with Engine("engine=gpu,fp=blake3,compress=lz4,dedup=1,arena=256m,"
                "spill=2g") as eng:
        errs = []

        def job(seed):
            rng = random.Random(seed)
            try:
                with eng.object() as o:
                    model = bytearray()
                    for _ in range(60):
                        op = rng.choice(["write", "read", "punch", "trunc",
                                         "flush", "copy"])
                        if op == "write":
                            off = rng.randrange(0, 24 * MB)
                            n = rng.randrange(1, 2 * MB)
                            pat = rand_bytes(min(n, 8192), seed=seed ^ n)
                            pat = (pat * (n // len(pat) + 1))[:n]
                            o.write(off, pat)
                            if len(model) < off + n:
                                model.extend(b"\x00" * (off + n - len(model)))
                            model[off:off + n] = pat
                        elif op == "read":
                            off = rng.randrange(0, 26 * MB)
                            n = rng.randrange(1, 2 * MB)
                            assert o.read(off, n) == bytes(model[off:off + n]), \
                                f"seed={seed}"
                        elif op == "punch":
                            off = rng.randrange(0, 24 * MB)
                            n = rng.randrange(1, 2 * MB)
                            o.punch_hole(off, n)
                            end = min(off + n, len(model))
                            if off < len(model):
                                model[off:end] = b"\x00" * (end - off)
                        elif op == "trunc":
                            sz = rng.randrange(0, 24 * MB)
                            o.truncate(sz)
                            if sz <= len(model):
                                del model[sz:]
                            else:
                                model.extend(b"\x00" * (sz - len(model)))
                        elif op == "flush":
                            o.flush()
                        elif op == "copy":
                            src = rng.randrange(0, 12 * MB)
                            dst = rng.randrange(0, 24 * MB)
                            n = rng.randrange(1, 2 * MB)
                            if abs(dst - src) >= n:
                                got = o.copy_range(o, src, dst, n)
                                want = min(n, max(0, len(model) - src)) \
                                    if src < len(model) else 0
                                assert got == want, f"seed={seed}"
                                if got:
                                    if len(model) < dst + got:
                                        model.extend(b"\x00" * (dst + got - len(model)))
                                    model[dst:dst + got] = model[src:src + got]
                    assert o.read(0, len(model) + 1) == bytes(model), f"seed={seed}"
                    o.truncate(0)
            except Exception as e:  # pragma: no cover
                errs.append(e)

        ts = [threading.Thread(target=job, args=(2000 + j,)) for j in range(6)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        assert not errs, errs[0]


def test_single_object_concurrent_writers_verify():
    """Config 3/4's exact shape: N threads write offset-partitioned slices
    of ONE object (payload copies run with the object lock dropped — r2),
    readers race them, then every byte is verified."""
    jobs, per_mb = 6, 24
    mb = 1 << 20
    with Engine("engine=gpu,fp=blake3,compress=lz4,dedup=1,arena=2g,"
                "spill=0,rs=6+2") as eng, eng.object() as o:
        datas = [rand_bytes(per_mb * mb, seed=900 + j) for j in range(jobs)]
        errs = []

        def wjob(j):
            try:
                base = j * per_mb * mb
                for i in range(per_mb):
                    o.write(base + i * mb, datas[j][i * mb:(i + 1) * mb])
            except Exception as e:
                errs.append(e)

        def rjob(j):
            try:
                for i in range(per_mb):
                    o.read(((j * 13 + i * 7) % (jobs * per_mb)) * mb, mb)
            except Exception as e:
                errs.append(e)

        ts = [threading.Thread(target=wjob, args=(j,)) for j in range(jobs)]
        ts += [threading.Thread(target=rjob, args=(j,)) for j in range(2)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        assert not errs, errs
        o.flush()
        for j in range(jobs):
            got = o.read(j * per_mb * mb, per_mb * mb)
            assert got == datas[j], f"job {j} slice mismatch"
        o.truncate(0)
