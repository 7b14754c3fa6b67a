"""Fault injection: injected chunk-read losses are healed by
RS reconstruction; injected put failures surface as EIO, never corruption."""

import pytest

from edgehip.lib import Engine
from tests.conftest import rand_bytes

MB = 1 << 20


def test_get_drop_healed_by_rs():
    # 2% of chunk reads report the chunk missing; RS 6+2 must reconstruct.
    with Engine("engine=host,fp=blake3,compress=lz4,dedup=1,rs=6+2,"
                "fault=get_drop:0.02") as eng, eng.object() as o:
        data = rand_bytes(24 * MB, seed=1)  # 4 stripes
        o.write(0, data)
        o.flush()
        for _ in range(10):
            assert o.read(0, len(data)) == data


def test_get_drop_without_rs_errors():
    with Engine("engine=host,fp=blake3,compress=none,dedup=0,"
                "fault=get_drop:1.0") as eng, eng.object() as o:
        data = rand_bytes(2 * MB, seed=2)
        o.write(0, data)
        o.flush()
        with pytest.raises(OSError):
            o.read(0, len(data))


def test_put_io_surfaces():
    with Engine("engine=host,fp=blake3,compress=none,dedup=0,"
                "fault=put_io:1.0") as eng, eng.object() as o:
        data = rand_bytes(40 * MB, seed=3)
        with pytest.raises(OSError):
            # enough data to force a batch flush inside write()
            o.write(0, data)
            o.flush()


def test_faults_off_by_default():
    with Engine("engine=host,fp=blake3,compress=none,dedup=0,fault=") as eng, \
            eng.object() as o:
        data = rand_bytes(4 * MB, seed=4)
        o.write(0, data)
        o.flush()
        assert o.read(0, len(data)) == data


def test_corrupt_detected_and_healed_by_rs():
    """Silent corruption on read: verify_read catches it (fingerprint
    mismatch) and the object layer reconstructs via RS."""
    with Engine("engine=host,fp=blake3,compress=none,dedup=1,rs=6+2,"
                "verify_read=1,fault=corrupt:0.05") as eng, eng.object() as o:
        data = rand_bytes(12 * MB, seed=9)
        o.write(0, data)
        o.commit()
        for _ in range(5):
            assert o.read(0, len(data)) == data


def test_corrupt_undetected_without_verify():
    with Engine("engine=host,fp=blake3,compress=none,dedup=1,"
                "fault=corrupt:1.0") as eng, eng.object() as o:
        data = rand_bytes(MB, seed=10)
        o.write(0, data)
        o.flush()
        assert o.read(0, MB) != data  # corruption passes through silently
