"""Object layer through ctypes: offset math, RMW, truncate, dedup accounting.
Runs on the host engine (CPU); the same tests run against the GPU engine in
tests/kernels/test_gpu_engine.py (marked gpu)."""

import pytest

from edgehip.lib import Engine
from tests.conftest import rand_bytes

HOST = "engine=host,fp=blake3,compress=lz4,dedup=1"


def test_seq_write_read():
    with Engine(HOST) as eng, eng.object() as o:
        data = rand_bytes(5 * (1 << 20) + 12345, seed=1)
        mb = 1 << 20
        for off in range(0, len(data), mb):
            o.write(off, data[off:off + mb])
        assert o.size == len(data)
        assert o.read(0, len(data)) == data
        o.flush()
        assert o.read(0, len(data)) == data
        # random 4 KiB reads
        for off in (0, 4096, 1 << 20, len(data) - 4096, 777777):
            assert o.read(off, 4096) == data[off:off + 4096]


def test_read_past_eof():
    with Engine(HOST) as eng, eng.object() as o:
        o.write(0, b"abc")
        assert o.read(0, 100) == b"abc"
        assert o.read(100, 10) == b""


def test_unaligned_rmw():
    with Engine(HOST) as eng, eng.object() as o:
        base = rand_bytes(3 * (1 << 20), seed=2)
        o.write(0, base)
        o.flush()
        patch = rand_bytes(100000, seed=3)
        off = (1 << 20) - 50000  # straddles chunk 0/1 boundary
        o.write(off, patch)
        expect = base[:off] + patch + base[off + len(patch):]
        assert o.read(0, len(expect)) == expect
        o.flush()
        assert o.read(0, len(expect)) == expect


def test_sparse_hole():
    with Engine(HOST) as eng, eng.object() as o:
        o.write(5 << 20, b"tail")
        assert o.size == (5 << 20) + 4
        assert o.read(0, 4096) == b"\x00" * 4096
        assert o.read(5 << 20, 4) == b"tail"
        o.flush()
        assert o.read((2 << 20) + 5, 10) == b"\x00" * 10


def test_truncate():
    with Engine(HOST) as eng, eng.object() as o:
        data = rand_bytes(2 << 20, seed=4)
        o.write(0, data)
        o.flush()
        o.truncate(1000000)
        assert o.size == 1000000
        assert o.read(0, 1 << 21) == data[:1000000]
        o.truncate(2 << 20)  # extend: zeros
        assert o.read(999996, 100) == data[999996:1000000] + b"\x00" * 96


def test_dedup_accounting():
    with Engine(HOST) as eng:
        chunk = rand_bytes(1 << 20, seed=5)
        with eng.object() as o1, eng.object() as o2:
            for rep in range(4):
                o1.write(rep << 20, chunk)
            o1.flush()
            o2.write(0, chunk)
            o2.flush()
            # identical content stored once
            assert eng.stored_chunks == 1
            assert eng.stored_bytes <= (1 << 20) + 4096


def test_overwrite_releases_old_chunk():
    with Engine(HOST) as eng, eng.object() as o:
        a = rand_bytes(1 << 20, seed=6)
        b = rand_bytes(1 << 20, seed=7)
        o.write(0, a)
        o.flush()
        assert eng.stored_chunks == 1
        o.write(0, b)
        o.flush()
        assert eng.stored_chunks == 1  # old chunk freed
        assert o.read(0, 1 << 20) == b


def test_commit_manifest_decodes():
    with Engine(HOST) as eng, eng.object() as o:
        data = rand_bytes((1 << 20) + 999, seed=8)
        o.write(0, data)
        mbytes = o.commit()
        assert len(mbytes) == 64 + 2 * 64 + 4  # header + 2 refs + crc
        from edgehip import get_lib
        import ctypes
        buf = ctypes.create_string_buffer(len(mbytes) + 64)
        n = get_lib()._so.eh_manifest_roundtrip(mbytes, len(mbytes), buf, len(mbytes) + 64)
        assert n == len(mbytes)


def test_fp_none_mode():
    with Engine("engine=host,fp=none,compress=none,dedup=0") as eng, eng.object() as o:
        data = rand_bytes(1 << 20, seed=9)
        o.write(0, data)
        o.flush()
        assert o.read(0, 1 << 20) == data


def test_compressible_data_shrinks():
    with Engine(HOST) as eng, eng.object() as o:
        data = (b"edgehip! " * 200000)[:1 << 20]
        o.write(0, data)
        o.flush()
        assert eng.stored_bytes < (1 << 20) // 4
        assert o.read(0, 1 << 20) == data


def test_preadv_batched():
    from edgehip.lib import Engine as E
    with E(HOST) as eng, eng.object() as o:
        data = rand_bytes(8 << 20, seed=21)
        o.write(0, data)
        o.flush()
        offs = [0, 4096, (3 << 20) + 8192, (7 << 20) + 12288, 5 << 20]
        lens = [4096] * 5
        buf = bytearray(5 * 4096)
        got = o.preadv(offs, lens, buf)
        assert got == 5 * 4096
        for i, off in enumerate(offs):
            assert bytes(buf[i * 4096:(i + 1) * 4096]) == data[off:off + 4096], i


def test_copy_range_shares_chunks():
    """Object-level server-side copy: aligned spans share chunk refs — the
    store's physical chunk count must not grow."""
    with Engine(HOST) as eng:
        with eng.object() as a, eng.object() as b:
            data = rand_bytes(4 << 20, seed=30)
            a.write(0, data)
            a.flush()
            before = eng.stored_chunks
            got = b.copy_range(a, 0, 0, len(data))
            assert got == len(data)
            assert b.read(0, len(data)) == data
            assert eng.stored_chunks == before  # shared, not re-stored
            # unaligned copy goes through RMW but stays correct
            got = b.copy_range(a, 4096 + 3, (2 << 20) + 11, 300000)
            assert got == 300000
            expect = bytearray(data)
            expect[(2 << 20) + 11:(2 << 20) + 11 + 300000] = \
                data[4096 + 3:4096 + 3 + 300000]
            assert b.read(0, len(expect)) == bytes(expect)
            # source EOF clamp
            assert b.copy_range(a, len(data) - 100, 0, 5000) == 100


def test_copy_range_then_source_delete():
    with Engine(HOST) as eng:
        data = rand_bytes(2 << 20, seed=33)
        b = eng.object()
        with eng.object() as a:
            a.write(0, data)
            a.flush()
            assert b.copy_range(a, 0, 0, len(data)) == len(data)
            a.truncate(0)  # source drops its refs
        assert b.read(0, len(data)) == data  # copy keeps the chunks alive
        b.truncate(0)
        b.close()


def test_punch_hole_subchunk_at_zero():
    # regression (hypothesis-found): aligned sub-chunk punch at offset 0 was
    # skipped by a bogus head/tail dedup guard
    with Engine(HOST) as eng, eng.object() as o:
        o.write(0, b"A")
        o.punch_hole(0, 1)
        assert o.read(0, 1) == b"\x00"
        data = rand_bytes(1 << 20, seed=77)
        o.write(0, data)
        o.punch_hole(0, 1000)
        assert o.read(0, 2000) == b"\x00" * 1000 + data[1000:2000]


def test_host_disk_overflow_tier(tmp_path):
    """spill_dir on the host engine: chunks past the RAM budget (arena) land
    as files and reads pread them back — CPU-testable mirror of the GPU
    engine's three-tier shape."""
    from edgehip.lib import Engine
    from tests.conftest import rand_bytes
    d = tmp_path / "overflow"
    with Engine(f"engine=host,fp=blake3,compress=none,dedup=0,arena=16m,"
                f"spill_dir={d}") as eng, eng.object() as o:
        data = rand_bytes(64 << 20, seed=88)
        o.write(0, data)
        o.flush()
        assert list(d.iterdir()), "nothing overflowed to disk"
        assert o.read(0, len(data)) == data
        assert o.read(40 << 20, 1 << 20) == data[40 << 20:41 << 20]
        o.truncate(0)
        assert not list(d.iterdir()), "disk chunks must unlink on release"
