"""GPU engine end-to-end (object layer on the HBM arena). Needs an MI355X."""

import pytest

from edgehip.lib import Engine
from tests.conftest import rand_bytes

pytestmark = pytest.mark.gpu

GPU = "engine=gpu,fp=blake3,compress=lz4,dedup=1,arena=4g,spill=8g"


def make_engine(opts=GPU):
    try:
        return Engine(opts)
    except RuntimeError as e:
        pytest.fail(f"GPU engine must initialize on a GPU box: {e}")


def test_engine_is_gpu():
    with make_engine() as eng:
        assert eng.is_gpu


def test_seq_write_read_1gib():
    with make_engine() as eng, eng.object() as o:
        mb = 1 << 20
        n = 256  # 256 MiB
        datas = [rand_bytes(mb, seed=i) for i in range(8)]
        for i in range(n):
            o.write(i * mb, datas[i % 8])
        o.flush()
        for i in range(0, n, 17):
            assert o.read(i * mb, mb) == datas[i % 8], f"chunk {i}"
        # dedup: only 8 unique chunks stored
        assert eng.stored_chunks == 8


def test_random_4k_reads():
    with make_engine() as eng, eng.object() as o:
        data = rand_bytes(64 << 20, seed=42)
        mb = 1 << 20
        for off in range(0, len(data), mb):
            o.write(off, data[off:off + mb])
        o.flush()
        import random
        rng = random.Random(7)
        for _ in range(256):
            off = rng.randrange(0, len(data) - 4096)
            assert o.read(off, 4096) == data[off:off + 4096], f"off {off}"


def test_compressible_content():
    with make_engine() as eng, eng.object() as o:
        data = (b"edgehip rocks! " * 80000)[:1 << 20] * 4
        o.write(0, data)
        o.flush()
        assert eng.stored_bytes < len(data) // 4
        assert o.read(0, len(data)) == data


def test_rmw_and_truncate():
    with make_engine() as eng, eng.object() as o:
        base = rand_bytes(3 << 20, seed=2)
        o.write(0, base)
        o.flush()
        patch = rand_bytes(100000, seed=3)
        off = (1 << 20) - 50000
        o.write(off, patch)
        o.flush()
        expect = base[:off] + patch + base[off + len(patch):]
        assert o.read(0, len(expect)) == expect
        o.truncate(1 << 20)
        assert o.read(0, 2 << 20) == expect[:1 << 20]


def test_spill_eviction():
    """Arena smaller than the data set forces LRU spill to host RAM."""
    with Engine("engine=gpu,fp=blake3,compress=none,dedup=0,arena=256m,spill=4g") as eng:
        with eng.object() as o:
            data = rand_bytes(512 << 20, seed=5)
            mb = 1 << 20
            for off in range(0, len(data), mb):
                o.write(off, data[off:off + mb])
            o.flush()
            # everything still readable (arena + spill tier)
            for off in range(0, len(data), 37 * mb):
                assert o.read(off, mb) == data[off:off + mb], f"off {off >> 20} MiB"


def test_commit_manifest():
    with make_engine() as eng, eng.object() as o:
        data = rand_bytes((2 << 20) + 777, seed=9)
        o.write(0, data)
        m = o.commit()
        assert len(m) == 64 + 3 * 64 + 4
        assert o.read(0, len(data)) == data


def test_gpu_cdc_mode():
    from edgehip.lib import Engine
    from tests.conftest import rand_bytes as rb
    with Engine("engine=gpu,fp=blake3,compress=lz4,dedup=1,cdc=1,chunk_size=256k,"
                "arena=2g,spill=0") as eng, eng.object() as o:
        data = rb(16 << 20, seed=9)
        o.write(0, data)
        o.flush()
        assert o.read(0, len(data)) == data
        n1 = eng.stored_bytes
        with eng.object() as o2:
            o2.write(0, b"Y" * 123 + data)
            o2.flush()
        assert eng.stored_bytes - n1 < 5 << 20  # shifted content re-dedups


def test_copy_range_gpu_addref():
    """GPU engine addref path: server-side copy shares HBM-resident chunks."""
    from edgehip.lib import Engine
    with Engine("engine=gpu,fp=blake3,compress=lz4,dedup=1,arena=2g,"
                "spill=0") as eng:
        with eng.object() as a, eng.object() as b:
            data = rand_bytes(8 << 20, seed=44)
            a.write(0, data)
            a.flush()
            before = eng.stored_chunks
            assert b.copy_range(a, 0, 0, len(data)) == len(data)
            assert eng.stored_chunks == before
            assert b.read(0, len(data)) == data
            a.truncate(0)  # b's refs keep the chunks alive
            assert b.read(0, len(data)) == data
            b.truncate(0)


def test_punch_hole_gpu_reclaims():
    from edgehip.lib import Engine
    with Engine("engine=gpu,fp=blake3,compress=none,dedup=0,arena=1g,"
                "spill=0") as eng, eng.object() as o:
        data = rand_bytes(8 << 20, seed=45)
        o.write(0, data)
        o.flush()
        before = eng.stored_chunks
        o.punch_hole(1 << 20, 3 << 20)  # chunks 1..3 dropped
        assert eng.stored_chunks == before - 3
        expect = bytearray(data)
        expect[1 << 20:4 << 20] = b"\x00" * (3 << 20)
        assert o.read(0, len(data)) == bytes(expect)
        o.truncate(0)


def test_spill_disk_overflow_tier(tmp_path):
    """spill_dir: once the RAM spill budget is exhausted, evicted chunks
    land as files on disk and reads pread them back (config 4's 1 TiB >
    HBM + host RAM story). Tiny caps force all three tiers."""
    d = tmp_path / "spill"
    with Engine("engine=gpu,fp=blake3,compress=lz4,dedup=0,arena=96m,"
                f"spill=128m,spill_dir={d}") as eng, eng.object() as o:
        data = rand_bytes(512 << 20, seed=77)
        for off in range(0, len(data), 16 << 20):
            o.write(off, data[off:off + (16 << 20)])
        o.flush()
        files = list(d.iterdir()) if d.exists() else []
        assert files, "nothing overflowed to the disk tier"
        # read back across all tiers (arena + RAM spill + disk)
        for off in range(0, len(data), 64 << 20):
            assert o.read(off, 1 << 20) == data[off:off + (1 << 20)]
        assert o.read(0, len(data)) == data
        o.truncate(0)
        assert not list(d.iterdir()), "disk chunks must be unlinked on release"
