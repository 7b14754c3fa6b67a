"""Multi-GPU engine paths (striping, peer copies, RCCL publish). These need
>= 2 devices; on the 1-GPU dev pool they skip, on an 8-GPU round-end box
they exercise the xGMI paths."""

import ctypes

import pytest

from edgehip import get_lib
from edgehip.lib import Engine
from tests.conftest import rand_bytes

pytestmark = pytest.mark.gpu


def ndev():
    so = get_lib()._so
    so.eh_gpu_available.restype = ctypes.c_int
    return so.eh_gpu_available()


@pytest.mark.skipif("ndev() < 2")
def test_two_gpu_striping():
    with Engine("engine=gpu,ngpus=2,fp=blake3,compress=lz4,dedup=1,arena=2g,"
                "spill=0") as eng, eng.object() as o:
        data = rand_bytes(64 << 20, seed=1)
        mb = 1 << 20
        for off in range(0, len(data), 8 * mb):
            o.write(off, data[off:off + 8 * mb])
        o.flush()
        assert o.read(0, len(data)) == data


@pytest.mark.skipif("ndev() < 2")
def test_two_gpu_rs_and_publish():
    with Engine("engine=gpu,ngpus=2,fp=blake3,compress=lz4,dedup=1,rs=6+2,"
                "arena=2g,spill=0") as eng, eng.object() as o:
        data = rand_bytes(12 << 20, seed=2)
        o.write(0, data)
        o.commit()  # parity over xGMI-gathered shards + RCCL publish
        assert o.read(0, len(data)) == data
