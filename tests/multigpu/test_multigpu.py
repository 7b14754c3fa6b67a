"""Multi-GPU engine paths (striping, placement, peer copies, RCCL publish,
device failure). These need >= 2 devices; on the 1-GPU dev pool they skip,
on an 8-GPU round-end box they exercise the xGMI paths (C1-C6) and the
stripe-aware placement invariants.
"""

import ctypes
import struct

import pytest

from edgehip import get_lib
from edgehip.lib import Engine
from tests.conftest import rand_bytes

pytestmark = pytest.mark.gpu

MIB = 1 << 20


def ndev():
    so = get_lib()._so
    so.eh_gpu_available.restype = ctypes.c_int
    return so.eh_gpu_available()


def decode_manifest(b: bytes):
    """Decode the EHM1 wire format (src/manifest.h) -> (data_refs, parity_refs).
    Each ref: dict(fp, logical_len, stored_len, flags, crc, shard_map)."""
    magic, version, fp_algo, compress = struct.unpack_from("<IHBB", b, 0)
    assert magic == 0x314D4845 and version == 1
    nchunks, logical_size, chunk_size = struct.unpack_from("<QQQ", b, 40)
    nrefs = (len(b) - 64 - 4) // 64
    data, parity = [], []
    for i in range(nrefs):
        off = 64 + i * 64
        fp = b[off:off + 32]
        ll, sl, flags, crc, smap, _res = struct.unpack_from("<IIIIQQ", b, off + 32)
        ref = dict(fp=fp, logical_len=ll, stored_len=sl, flags=flags, crc=crc,
                   shard_map=smap)
        (parity if flags & 4 else data).append(ref)
    assert len(data) == nchunks
    return data, parity


def drop_chunk(eng, fp: bytes) -> int:
    so = eng.lib._so
    so.eh_engine_drop_chunk.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    so.eh_engine_drop_chunk.restype = ctypes.c_int
    return so.eh_engine_drop_chunk(eng.h, fp)


@pytest.mark.skipif("ndev() < 2")
def test_two_gpu_striping():
    with Engine("engine=gpu,ngpus=2,fp=blake3,compress=lz4,dedup=1,arena=2g,"
                "spill=0") as eng, eng.object() as o:
        assert eng.ndev == 2
        data = rand_bytes(64 << 20, seed=1)
        for off in range(0, len(data), 8 * MIB):
            o.write(off, data[off:off + 8 * MIB])
        o.flush()
        assert o.read(0, len(data)) == data
        # chunks actually live on BOTH devices (fp placement spreads them)
        devs = set()
        mf = o.commit()
        drefs, _ = decode_manifest(mf)
        for r in drefs:
            devs.add(eng.chunk_device(r["fp"]))
        assert devs == {0, 1}, f"placement collapsed to {devs}"
        o.truncate(0)


@pytest.mark.skipif("ndev() < 2")
def test_two_gpu_rs_and_publish():
    with Engine("engine=gpu,ngpus=2,fp=blake3,compress=lz4,dedup=1,rs=6+2,"
                "arena=2g,spill=0") as eng, eng.object() as o:
        data = rand_bytes(12 << 20, seed=2)
        o.write(0, data)
        o.commit()  # parity over xGMI-gathered shards + RCCL publish (C4-C6)
        assert o.read(0, len(data)) == data
        o.truncate(0)


@pytest.mark.skipif("ndev() < 2")
def test_stripe_placement_invariant_and_shard_map():
    """Every stripe's k+m shards spread round-robin over devices: balanced
    (counts differ by <= 1), and when ndev >= 2 no device holds more than
    ceil((k+m)/ndev) shards — the manifest's shard_map records the ACTUAL
    device per shard and must agree with the store."""
    n = ndev()
    k, m = 6, 2
    with Engine(f"engine=gpu,ngpus={n},fp=blake3,compress=lz4,dedup=1,rs=6+2,"
                "arena=2g,spill=0") as eng, eng.object() as o:
        data = rand_bytes(24 << 20, seed=3)  # 4 stripes of 6 data chunks
        o.write(0, data)
        mf = o.commit()
        drefs, prefs = decode_manifest(mf)
        assert len(prefs) == (len(drefs) + k - 1) // k * m
        import math
        cap = math.ceil((k + m) / n)
        for s in range((len(drefs) + k - 1) // k):
            stripe = drefs[s * k:(s + 1) * k] + prefs[s * m:(s + 1) * m]
            devs = []
            smap = stripe[0]["shard_map"]
            for j, ref in enumerate(stripe):
                d = eng.chunk_device(ref["fp"])
                assert d >= 0, f"stripe {s} shard {j} not on a device: {d}"
                devs.append(d)
                # shard_map byte j = device of shard j, same map on every ref
                assert (smap >> (8 * j)) & 0xFF == d, \
                    f"shard_map mismatch stripe {s} shard {j}"
                assert ref["shard_map"] == smap
            counts = [devs.count(i) for i in range(n)]
            assert max(counts) <= cap, f"stripe {s} unbalanced: {devs}"
            if n >= k + m:
                assert len(set(devs)) == k + m, f"stripe {s} devices collide: {devs}"
        assert o.read(0, len(data)) == data
        o.truncate(0)


@pytest.mark.skipif("ndev() < 2")
def test_degraded_read_device_shards_dropped():
    """Drop device-1 data chunks — AT MOST m=2 per stripe (RS 6+2 tolerates
    m losses per stripe; with 2 devices a whole device holds 4 shards of
    each stripe, beyond tolerance — that case needs >= 4 devices and is
    covered by test_device_failure...). Reads must reconstruct each lost
    chunk from the surviving shards (C2 gather + G6 decode) and repair it."""
    k, m = 6, 2
    with Engine("engine=gpu,ngpus=2,fp=blake3,compress=lz4,dedup=1,rs=6+2,"
                "arena=2g,spill=0") as eng, eng.object() as o:
        data = rand_bytes(24 << 20, seed=4)
        o.write(0, data)
        mf = o.commit()
        drefs, _ = decode_manifest(mf)
        dropped = 0
        per_stripe = {}
        for i, r in enumerate(drefs):
            s = i // k
            if eng.chunk_device(r["fp"]) == 1 and per_stripe.get(s, 0) < m:
                assert drop_chunk(eng, r["fp"]) == 0
                per_stripe[s] = per_stripe.get(s, 0) + 1
                dropped += 1
        assert dropped > 0, "nothing placed on device 1?"
        assert o.read(0, len(data)) == data  # reconstructs via RS
        o.truncate(0)


@pytest.mark.skipif("ndev() < 4")
def test_device_failure_reads_degrade_writes_restripe():
    """SURVEY §6.3 device-level failure: mark device 1 failed. Whole-device
    loss is reconstructable only when no device holds more than m=2 shards
    of a stripe — 6+2 round-robin over 4 devices puts exactly 2 per device,
    so this needs >= 4 GPUs (over 2 it would hold 4 > m). Reads of the
    failed device's chunks degrade to RS reconstruct + in-place repair;
    new writes place only on survivors."""
    with Engine("engine=gpu,ngpus=4,fp=blake3,compress=lz4,dedup=1,rs=6+2,"
                "arena=2g,spill=0") as eng:
        with eng.object() as o:
            data = rand_bytes(24 << 20, seed=5)
            o.write(0, data)
            mf = o.commit()
            drefs, _ = decode_manifest(mf)
            on1 = sum(1 for r in drefs if eng.chunk_device(r["fp"]) == 1)
            assert on1 > 0
            eng.mark_device_failed(1)
            # reads reconstruct the failed device's chunks from survivors
            assert o.read(0, len(data)) == data
            o.truncate(0)
        # writes after the failure avoid the failed device
        with eng.object() as o2:
            data2 = rand_bytes(12 << 20, seed=6)
            o2.write(0, data2)
            mf2 = o2.commit()
            drefs2, prefs2 = decode_manifest(mf2)
            for r in drefs2 + prefs2:
                assert eng.chunk_device(r["fp"]) != 1
            assert o2.read(0, len(data2)) == data2
            o2.truncate(0)


@pytest.mark.skipif("ndev() < 2")
def test_cross_device_dedup_single_copy():
    """Two objects with identical content: the second write dedups against
    the first wherever its chunks landed (content-addressed store, C5
    broadcast pre-warms the index shards)."""
    with Engine("engine=gpu,ngpus=2,fp=blake3,compress=lz4,dedup=1,"
                "arena=2g,spill=0") as eng:
        data = rand_bytes(16 << 20, seed=7)
        with eng.object() as a, eng.object() as b:
            a.write(0, data)
            a.flush()
            stored_after_first = eng.stored_bytes
            b.write(0, data)
            b.flush()
            assert eng.stored_bytes == stored_after_first  # no second copy
            assert b.read(0, len(data)) == data
            a.truncate(0)
            b.truncate(0)


@pytest.mark.skipif("ndev() < 2")
def test_striped_spill_roundtrip():
    """Striping + host spill tier together (config 4's 1 TiB shape in
    miniature): arena too small for the object, eviction runs per device."""
    with Engine("engine=gpu,ngpus=2,fp=blake3,compress=lz4,dedup=0,rs=6+2,"
                "arena=256m,spill=4g") as eng, eng.object() as o:
        data = rand_bytes(768 << 20, seed=8)
        step = 32 * MIB
        for off in range(0, len(data), step):
            o.write(off, data[off:off + step])
        o.flush()
        for off in range(0, len(data), 256 * MIB):
            assert o.read(off, 8 * MIB) == data[off:off + 8 * MIB]
        o.truncate(0)


@pytest.mark.skipif("ndev() < 2")
def test_manifest_ring_wrap_saturates_not_zeroes():
    """DistCtx manifest table is a RING: with a tiny
    injected capacity, publishing past it must keep the monotonic total
    growing while the live count saturates at capacity (the r1 modulo
    zeroed the count at exactly-full)."""
    with Engine("engine=gpu,ngpus=2,fp=blake3,compress=lz4,dedup=1,arena=1g,"
                "spill=0,dist_table_slots=256") as eng:
        so = eng.lib._so
        so.eh_engine_manifest_records.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                                  ctypes.c_int]
        so.eh_engine_manifest_records.restype = ctypes.c_uint64
        # each commit publishes in 64-record padded batches over 2 ranks
        # (128 table records per publish); 4 commits x 1 batch wraps 256
        for r in range(6):
            with eng.object() as o:
                o.write(0, rand_bytes(4 * MIB, seed=100 + r))
                o.commit()
                o.truncate(0)
        total = so.eh_engine_manifest_records(eng.h, 0, 1)
        live = so.eh_engine_manifest_records(eng.h, 0, 0)
        assert total > 256, f"expected wrap past capacity, total={total}"
        assert live == 256, f"live count must saturate at capacity, got {live}"
