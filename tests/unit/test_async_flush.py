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
