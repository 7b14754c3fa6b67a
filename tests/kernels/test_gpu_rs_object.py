"""RS 6+2 through the object layer on the GPU engine (judged config 4 path:
parity via the gfx950 GF(256) kernel, reconstruct + repair on read)."""

import ctypes

import pytest

from edgehip.lib import Engine
from tests.conftest import rand_bytes

pytestmark = pytest.mark.gpu

RS = "engine=gpu,fp=blake3,compress=lz4,dedup=1,rs=6+2,arena=2g,spill=0"
MB = 1 << 20


def drop_chunk(eng, fp_bytes):
    so = eng.lib._so
    so.eh_engine_drop_chunk.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    so.eh_engine_drop_chunk.restype = ctypes.c_int
    return so.eh_engine_drop_chunk(eng.h, fp_bytes)


def manifest_fps(mbytes):
    import struct
    nchunks = struct.unpack_from("<Q", mbytes, 40)[0]
    refs = []
    for i in range(nchunks):
        base = 64 + 64 * i
        refs.append((mbytes[base:base + 32],
                     struct.unpack_from("<I", mbytes, base + 40)[0]))
    return ([r for r in refs if not r[1] & 4], [r for r in refs if r[1] & 4])


def test_gpu_rs_write_and_degraded_read():
    with Engine(RS) as eng, eng.object() as o:
        assert eng.is_gpu
        data = rand_bytes(12 * MB, seed=31)
        o.write(0, data)
        m = o.commit()
        d, p = manifest_fps(m)
        assert len(d) == 12 and len(p) == 4
        # double loss in stripe 1
        assert drop_chunk(eng, d[7][0]) == 0
        assert drop_chunk(eng, d[10][0]) == 0
        assert o.read(0, len(data)) == data
        # repair happened: drop parity, re-read still fine
        for fp, _ in p:
            drop_chunk(eng, fp)
        assert o.read(0, len(data)) == data
