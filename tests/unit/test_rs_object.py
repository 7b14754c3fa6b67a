"""RS 6+2 through the object layer (judged config 4 semantics): parity
maintenance on write, reconstruction + self-repair on degraded read.
Runs on the host engine; tests/kernels/test_gpu_rs_object.py repeats on GPU."""

import ctypes

import pytest

from edgehip.lib import Engine
from tests.conftest import rand_bytes

RS = "engine=host,fp=blake3,compress=lz4,dedup=1,rs=6+2"
MB = 1 << 20


def drop_chunk(eng, fp_bytes):
    so = eng.lib._so
    so.eh_engine_drop_chunk.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    so.eh_engine_drop_chunk.restype = ctypes.c_int
    return so.eh_engine_drop_chunk(eng.h, fp_bytes)


def manifest_fps(mbytes):
    """Parse data-chunk (fp, flags) pairs + parity count from manifest bytes."""
    import struct
    nchunks = struct.unpack_from("<Q", mbytes, 40)[0]
    refs = []
    for i in range(nchunks):
        base = 64 + 64 * i
        fp = mbytes[base:base + 32]
        flags = struct.unpack_from("<I", mbytes, base + 40)[0]
        refs.append((fp, flags))
    data = [r for r in refs if not r[1] & 4]
    parity = [r for r in refs if r[1] & 4]
    return data, parity


def test_parity_written_and_flagged():
    with Engine(RS) as eng, eng.object() as o:
        data = rand_bytes(12 * MB, seed=1)  # exactly 2 stripes
        o.write(0, data)
        m = o.commit()
        d, p = manifest_fps(m)
        assert len(d) == 12
        assert len(p) == 4  # 2 stripes x 2 parity
        assert all(f & 2 for _, f in d)  # kFlagRs on data
        # 12 data + 4 parity chunks stored
        assert eng.stored_chunks == 16


def test_partial_tail_stripe_protected():
    with Engine(RS) as eng, eng.object() as o:
        data = rand_bytes(8 * MB + 12345, seed=2)  # 9 chunks = 1.5 stripes
        o.write(0, data)
        m = o.commit()
        d, p = manifest_fps(m)
        assert len(d) == 9
        assert len(p) == 4  # ceil(9/6)=2 stripes x 2


def test_degraded_read_single_loss():
    with Engine(RS) as eng, eng.object() as o:
        data = rand_bytes(12 * MB, seed=3)
        o.write(0, data)
        m = o.commit()
        d, _ = manifest_fps(m)
        # kill data chunk 4 (middle of stripe 0)
        assert drop_chunk(eng, d[4][0]) == 0
        # read through the damaged range: must reconstruct transparently
        assert o.read(0, len(data)) == data


def test_degraded_read_double_loss_same_stripe():
    with Engine(RS) as eng, eng.object() as o:
        data = rand_bytes(6 * MB, seed=4)
        o.write(0, data)
        m = o.commit()
        d, _ = manifest_fps(m)
        assert drop_chunk(eng, d[1][0]) == 0
        assert drop_chunk(eng, d[5][0]) == 0
        assert o.read(0, len(data)) == data


def test_triple_loss_fails():
    with Engine(RS) as eng, eng.object() as o:
        data = rand_bytes(6 * MB, seed=5)
        o.write(0, data)
        m = o.commit()
        d, _ = manifest_fps(m)
        for i in (0, 2, 3):
            assert drop_chunk(eng, d[i][0]) == 0
        with pytest.raises(OSError):
            o.read(0, len(data))


def test_repair_after_reconstruct():
    """A degraded read repairs the store: the second read needs no parity."""
    with Engine(RS) as eng, eng.object() as o:
        data = rand_bytes(6 * MB, seed=6)
        o.write(0, data)
        m = o.commit()
        d, p = manifest_fps(m)
        assert drop_chunk(eng, d[2][0]) == 0
        before = eng.stored_chunks
        assert o.read(2 * MB, MB) == data[2 * MB:3 * MB]
        assert eng.stored_chunks == before + 1  # chunk re-ingested
        # now drop ALL parity; the repaired chunk must still read
        for fp, _ in p:
            drop_chunk(eng, fp)
        assert o.read(2 * MB, MB) == data[2 * MB:3 * MB]


def test_overwrite_refreshes_parity():
    with Engine(RS) as eng, eng.object() as o:
        data = bytearray(rand_bytes(6 * MB, seed=7))
        o.write(0, bytes(data))
        o.commit()
        patch = rand_bytes(MB, seed=8)
        o.write(3 * MB, patch)
        data[3 * MB:4 * MB] = patch
        m2 = o.commit()
        d, _ = manifest_fps(m2)
        # drop the NEW chunk 3; parity must reflect the new content
        assert drop_chunk(eng, d[3][0]) == 0
        assert o.read(0, len(data)) == bytes(data)


def test_rs_with_holes():
    with Engine(RS) as eng, eng.object() as o:
        o.write(5 * MB, b"tail after hole")
        m = o.commit()
        d, p = manifest_fps(m)
        assert len(p) == 2  # one stripe
        # chunk 5 is real; 0-4 are holes; drop 5 and reconstruct from parity
        real = [fp for fp, f in d if fp != b"\x00" * 32]
        assert len(real) == 1
        assert drop_chunk(eng, real[0]) == 0
        assert o.read(5 * MB, 15) == b"tail after hole"
