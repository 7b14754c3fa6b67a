"""Async write-behind flush (src/obj.cc FlushPool): writes hand full chunks to
background workers; reads must see in-flight bytes (inflight_view_), rewrites
must drain, and async put errors must surface on flush — never corrupt."""

import pytest

from edgehip.lib import Engine
from tests.conftest import rand_bytes

MB = 1 << 20
HOST = "engine=host,fp=blake3,compress=lz4,dedup=1"


def test_read_sees_inflight_chunks():
    # A large streaming write triggers background flushes; immediate reads
    # (before flush()) must return the just-written bytes from either the
    # dirty map, the in-flight view, or the committed manifest.
    with Engine(HOST) as eng, eng.object() as o:
        data = rand_bytes(48 * MB, seed=11)
        for off in range(0, len(data), MB):
            o.write(off, data[off:off + MB])
            # read back a stripe of earlier offsets every few writes
            if off and off % (8 * MB) == 0:
                probe = off - 3 * MB
                assert o.read(probe, 4096) == data[probe:probe + 4096]
        assert o.read(0, len(data)) == data
        o.flush()
        assert o.read(0, len(data)) == data


def test_rewrite_during_inflight_is_ordered():
    with Engine(HOST) as eng, eng.object() as o:
        a = rand_bytes(40 * MB, seed=12)
        for off in range(0, len(a), MB):
            o.write(off, a[off:off + MB])
        # immediately rewrite an early region (likely still in flight)
        b = rand_bytes(2 * MB, seed=13)
        o.write(0, b)
        o.flush()
        assert o.read(0, 2 * MB) == b
        assert o.read(2 * MB, 2 * MB) == a[2 * MB:4 * MB]


def test_async_put_error_surfaces_on_flush():
    with Engine("engine=host,fp=blake3,compress=none,dedup=0,"
                "fault=put_io:1.0") as eng, eng.object() as o:
        data = rand_bytes(40 * MB, seed=14)
        with pytest.raises(OSError):
            for off in range(0, len(data), MB):
                o.write(off, data[off:off + MB])
            o.flush()


def test_truncate_waits_for_inflight():
    with Engine(HOST) as eng, eng.object() as o:
        data = rand_bytes(40 * MB, seed=15)
        for off in range(0, len(data), MB):
            o.write(off, data[off:off + MB])
        o.truncate(10 * MB + 123)
        assert o.size == 10 * MB + 123
        assert o.read(0, 10 * MB + 123) == data[:10 * MB + 123]
        o.flush()
        assert o.read(9 * MB, MB) == data[9 * MB:10 * MB]


def test_threaded_random_ops_stress():
    """Randomized multi-threaded ops on separate objects sharing one host
    engine (dedup index, store shards, flush pool): every object must match
    its own model. Seeds printed on failure."""
    import random
    import threading
    with Engine(HOST) as eng:
        errs = []

        def job(seed):
            rng = random.Random(seed)
            try:
                with eng.object() as o:
                    model = bytearray()
                    for _ in range(40):
                        op = rng.choice(["write", "read", "punch", "trunc",
                                         "flush", "copy"])
                        if op == "write":
                            off = rng.randrange(0, 4 * MB)
                            n = rng.randrange(1, MB)
                            pat = bytes([rng.randrange(256)]) * n
                            o.write(off, pat)
                            if len(model) < off + n:
                                model.extend(b"\x00" * (off + n - len(model)))
                            model[off:off + n] = pat
                        elif op == "read":
                            off = rng.randrange(0, 5 * MB)
                            n = rng.randrange(1, MB)
                            assert o.read(off, n) == bytes(model[off:off + n]), \
                                f"seed={seed}"
                        elif op == "punch":
                            off = rng.randrange(0, 4 * MB)
                            n = rng.randrange(1, MB)
                            o.punch_hole(off, n)
                            end = min(off + n, len(model))
                            if off < len(model):
                                model[off:end] = b"\x00" * (end - off)
                        elif op == "trunc":
                            sz = rng.randrange(0, 4 * MB)
                            o.truncate(sz)
                            if sz <= len(model):
                                del model[sz:]
                            else:
                                model.extend(b"\x00" * (sz - len(model)))
                        elif op == "flush":
                            o.flush()
                        elif op == "copy":
                            src = rng.randrange(0, 2 * MB)
                            dst = rng.randrange(0, 4 * MB)
                            n = rng.randrange(1, MB)
                            if abs(dst - src) >= n:  # non-overlapping only
                                got = o.copy_range(o, src, dst, n)
                                want = min(n, max(0, len(model) - src)) \
                                    if src < len(model) else 0
                                assert got == want, f"seed={seed}"
                                if got:
                                    if len(model) < dst + got:
                                        model.extend(b"\x00" * (dst + got - len(model)))
                                    model[dst:dst + got] = model[src:src + got]
                    assert o.read(0, len(model) + 1) == bytes(model), f"seed={seed}"
            except Exception as e:  # pragma: no cover
                errs.append(e)

        ts = [threading.Thread(target=job, args=(1000 + j,)) for j in range(6)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        assert not errs, errs[0]
