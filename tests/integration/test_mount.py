"""Integration: real FUSE mount on the host engine (judged config 1 shape).
Needs /dev/fuse + root (the 'fuse' marker skips otherwise)."""

import os
import subprocess

import pytest

from edgehip.mount import Mount
from tests.conftest import rand_bytes

pytestmark = pytest.mark.fuse

HOSTOPTS = "engine=host,fp=none,compress=none,dedup=0,log=3"
CONTENTOPTS = "engine=host,fp=blake3,compress=lz4,dedup=1,log=3"


def test_mount_basic_posix(tmp_mount):
    with Mount(tmp_mount, HOSTOPTS):
        p = tmp_mount / "hello.txt"
        p.write_text("hello edgehip\n")
        assert p.read_text() == "hello edgehip\n"
        assert p.stat().st_size == 14
        (tmp_mount / "d").mkdir()
        (tmp_mount / "d" / "x").write_bytes(b"x" * 100)
        assert sorted(os.listdir(tmp_mount)) == ["d", "hello.txt"]
        os.rename(tmp_mount / "hello.txt", tmp_mount / "d" / "hello2.txt")
        assert sorted(os.listdir(tmp_mount / "d")) == ["hello2.txt", "x"]
        (tmp_mount / "d" / "x").unlink()
        (tmp_mount / "d" / "hello2.txt").unlink()
        os.rmdir(tmp_mount / "d")
        assert os.listdir(tmp_mount) == []
    assert not os.path.ismount(tmp_mount)


def test_large_file_roundtrip(tmp_mount):
    with Mount(tmp_mount, HOSTOPTS):
        data = rand_bytes(32 << 20, seed=1)
        p = tmp_mount / "big.bin"
        p.write_bytes(data)
        assert p.stat().st_size == len(data)
        assert p.read_bytes() == data


def test_random_4k_reads(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        data = rand_bytes(8 << 20, seed=2)
        p = tmp_mount / "obj.bin"
        p.write_bytes(data)
        import random
        rng = random.Random(3)
        with open(p, "rb") as f:
            for _ in range(200):
                off = rng.randrange(0, len(data) - 4096)
                f.seek(off)
                assert f.read(4096) == data[off:off + 4096]


def test_append_and_rewrite(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        p = tmp_mount / "log.txt"
        with open(p, "ab") as f:
            for i in range(100):
                f.write(f"line {i}\n".encode())
        lines = p.read_bytes().decode().splitlines()
        assert lines[0] == "line 0" and lines[99] == "line 99"
        # in-place rewrite
        with open(p, "r+b") as f:
            f.seek(7)
            f.write(b"XXXX")
        assert p.read_bytes()[7:11] == b"XXXX"


def test_truncate_via_mount(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        p = tmp_mount / "t.bin"
        data = rand_bytes(3 << 20, seed=4)
        p.write_bytes(data)
        os.truncate(p, 1 << 20)
        assert p.stat().st_size == 1 << 20
        assert p.read_bytes() == data[:1 << 20]


def test_dedup_through_mount(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        data = rand_bytes(4 << 20, seed=5)
        for i in range(3):
            (tmp_mount / f"copy{i}.bin").write_bytes(data)
        for i in range(3):
            assert (tmp_mount / f"copy{i}.bin").read_bytes() == data


def test_many_small_files(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        for i in range(200):
            (tmp_mount / f"f{i:03}.txt").write_bytes(b"v" * i)
        names = sorted(os.listdir(tmp_mount))
        assert len(names) == 200
        assert (tmp_mount / "f199.txt").stat().st_size == 199


def test_cp_and_cmp_binary(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        src = tmp_mount.parent / "src.bin"
        src.write_bytes(rand_bytes(16 << 20, seed=6))
        subprocess.run(["cp", str(src), str(tmp_mount / "dst.bin")], check=True)
        rc = subprocess.run(["cmp", str(src), str(tmp_mount / "dst.bin")]).returncode
        assert rc == 0


def test_statfs(tmp_mount):
    with Mount(tmp_mount, HOSTOPTS):
        st = os.statvfs(tmp_mount)
        assert st.f_bsize == 4096
        assert st.f_blocks > 0


def test_symlink_and_hardlink(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        p = tmp_mount / "orig.txt"
        p.write_text("payload\n")
        (tmp_mount / "sym").symlink_to("orig.txt")
        assert (tmp_mount / "sym").read_text() == "payload\n"
        assert os.readlink(tmp_mount / "sym") == "orig.txt"
        os.link(p, tmp_mount / "hard")
        assert (tmp_mount / "hard").read_text() == "payload\n"
        assert p.stat().st_nlink == 2
        p.unlink()
        assert (tmp_mount / "hard").read_text() == "payload\n"  # survives
        assert (tmp_mount / "hard").stat().st_nlink == 1


def test_readdirplus_listing_with_attrs(tmp_mount):
    """`ls -l` pattern: READDIRPLUS (offered via FUSE_DO_READDIRPLUS) returns
    entry+attr in one trip; every stat must be correct and later FORGETs must
    not underflow (unmount at fixture teardown would hang/crash if so)."""
    with Mount(tmp_mount, CONTENTOPTS):
        d = os.path.join(tmp_mount, "plusdir")
        os.mkdir(d)
        sizes = {}
        for i in range(200):
            p = os.path.join(d, f"f{i:03d}")
            with open(p, "wb") as f:
                f.write(b"x" * (i * 7))
            sizes[f"f{i:03d}"] = i * 7
        with os.scandir(d) as it:  # scandir consumes d_type + stat per entry
            seen = {}
            for e in it:
                st = e.stat()
                assert e.is_file()
                seen[e.name] = st.st_size
        assert seen == sizes
        # second pass straight after FORGET churn
        assert sorted(os.listdir(d)) == sorted(sizes)
        for name, sz in sizes.items():
            assert os.stat(os.path.join(d, name)).st_size == sz


def test_xattr_roundtrip(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        p = os.path.join(tmp_mount, "xf")
        with open(p, "wb") as f:
            f.write(b"hi")
        os.setxattr(p, "user.color", b"blue")
        os.setxattr(p, "user.blob", bytes(range(256)))
        assert os.getxattr(p, "user.color") == b"blue"
        assert os.getxattr(p, "user.blob") == bytes(range(256))
        assert sorted(os.listxattr(p)) == ["user.blob", "user.color"]
        os.setxattr(p, "user.color", b"red", os.XATTR_REPLACE)
        assert os.getxattr(p, "user.color") == b"red"
        with pytest.raises(OSError):
            os.setxattr(p, "user.color", b"x", os.XATTR_CREATE)  # EEXIST
        os.removexattr(p, "user.color")
        with pytest.raises(OSError):
            os.getxattr(p, "user.color")  # ENODATA
        assert os.listxattr(p) == ["user.blob"]


def test_copy_file_range_instant_dedup(tmp_mount):
    """FUSE_COPY_FILE_RANGE: content-addressed server-side copy — aligned
    ranges share chunk refs (no data movement), edges fall back to RMW."""
    with Mount(tmp_mount, CONTENTOPTS):
        a = os.path.join(tmp_mount, "cfr_a")
        b = os.path.join(tmp_mount, "cfr_b")
        data = os.urandom(5 * (1 << 20) + 12345)
        with open(a, "wb") as f:
            f.write(data)
        os.sync()
        fin = os.open(a, os.O_RDONLY)
        fout = os.open(b, os.O_WRONLY | os.O_CREAT)
        try:
            total, off = 0, 0
            while total < len(data):
                n = os.copy_file_range(fin, fout, len(data) - total, off, off)
                assert n > 0
                total += n
                off += n
        finally:
            os.close(fin)
            os.close(fout)
        with open(b, "rb") as f:
            assert f.read() == data
        # unaligned cross-copy into the middle of b
        fin = os.open(a, os.O_RDONLY)
        fout = os.open(b, os.O_WRONLY)
        try:
            n = os.copy_file_range(fin, fout, 100000, 4096 + 7, (1 << 20) + 13)
            assert n == 100000
        finally:
            os.close(fin)
            os.close(fout)
        expect = bytearray(data)
        expect[(1 << 20) + 13:(1 << 20) + 13 + 100000] = \
            data[4096 + 7:4096 + 7 + 100000]
        with open(b, "rb") as f:
            assert f.read() == bytes(expect)


def test_fallocate_punch_hole(tmp_mount):
    import ctypes
    libc = ctypes.CDLL("libc.so.6", use_errno=True)
    FALLOC_FL_KEEP_SIZE, FALLOC_FL_PUNCH_HOLE = 0x1, 0x2
    with Mount(tmp_mount, CONTENTOPTS):
        p = os.path.join(tmp_mount, "holey")
        data = bytearray(rand_bytes(4 * (1 << 20) + 999, seed=31))
        with open(p, "wb") as f:
            f.write(data)
        os.sync()
        fd = os.open(p, os.O_RDWR)
        try:
            # punch an unaligned 2.5 MiB hole
            off, ln = 123456, (5 << 19) + 777
            rc = libc.fallocate(fd, FALLOC_FL_PUNCH_HOLE | FALLOC_FL_KEEP_SIZE,
                                ctypes.c_long(off), ctypes.c_long(ln))
            assert rc == 0, ctypes.get_errno()
        finally:
            os.close(fd)
        data[off:off + ln] = b"\x00" * ln
        with open(p, "rb") as f:
            got = f.read()
        assert len(got) == len(data)
        assert got == bytes(data)


def test_lseek_data_hole(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        p = os.path.join(tmp_mount, "sparse")
        with open(p, "wb") as f:
            f.write(b"A" * (1 << 20))       # chunk 0: data
            f.seek(3 << 20)
            f.write(b"B" * (1 << 20))       # chunk 3: data; 1-2 never written
        os.sync()
        fd = os.open(p, os.O_RDONLY)
        try:
            assert os.lseek(fd, 0, os.SEEK_DATA) == 0
            hole = os.lseek(fd, 0, os.SEEK_HOLE)
            assert hole == (1 << 20)
            data = os.lseek(fd, hole, os.SEEK_DATA)
            assert data == (3 << 20)
            assert os.lseek(fd, data, os.SEEK_HOLE) == (4 << 20)  # EOF
        finally:
            os.close(fd)


def test_rename_exchange(tmp_mount):
    import ctypes
    libc = ctypes.CDLL("libc.so.6", use_errno=True)
    RENAME_EXCHANGE, AT_FDCWD = 2, -100
    with Mount(tmp_mount, CONTENTOPTS):
        a = os.path.join(tmp_mount, "xa")
        b = os.path.join(tmp_mount, "sub")
        os.mkdir(b)
        bf = os.path.join(b, "xb")
        with open(a, "w") as f:
            f.write("AAA")
        with open(bf, "w") as f:
            f.write("BBBB")
        rc = libc.renameat2(AT_FDCWD, a.encode(), AT_FDCWD, bf.encode(),
                            RENAME_EXCHANGE)
        assert rc == 0, ctypes.get_errno()
        assert open(a).read() == "BBBB"
        assert open(bf).read() == "AAA"
        # exchanging with a missing target must fail (ENOENT)
        rc = libc.renameat2(AT_FDCWD, a.encode(), AT_FDCWD,
                            os.path.join(tmp_mount, "nope").encode(),
                            RENAME_EXCHANGE)
        assert rc != 0


def test_stats_latency_histograms(tmp_mount):
    with Mount(tmp_mount, CONTENTOPTS):
        p = tmp_mount / "lat.bin"
        p.write_bytes(rand_bytes(2 << 20, seed=70))
        # O_DIRECT bypasses the page cache so the read reaches the daemon
        import mmap as _mmap
        fd = os.open(p, os.O_RDONLY | os.O_DIRECT)
        buf = _mmap.mmap(-1, 1 << 20)
        try:
            os.readv(fd, [buf])
        finally:
            os.close(fd)
        stats = (tmp_mount / ".edgehip-stats").read_text()
        assert "read_latency p50<=" in stats
        assert "write_latency p50<=" in stats
        assert "n=0" not in stats.split("write_latency")[1]


def test_cdc_mode_through_mount(tmp_mount):
    """CDC (content-defined chunking) behind the FUSE surface: append-stream
    writes, dedup across shifted content, correct readback."""
    with Mount(tmp_mount, "engine=host,fp=blake3,compress=lz4,dedup=1,"
                          "cdc=1,chunk_size=256k,log=3"):
        base = rand_bytes(4 << 20, seed=80)
        (tmp_mount / "a.bin").write_bytes(base)
        # shifted copy: CDC re-finds the shared boundaries
        (tmp_mount / "b.bin").write_bytes(b"PREFIX-12345" + base)
        assert (tmp_mount / "a.bin").read_bytes() == base
        assert (tmp_mount / "b.bin").read_bytes() == b"PREFIX-12345" + base
        # append workload
        with open(tmp_mount / "log.bin", "ab") as f:
            for i in range(32):
                f.write(rand_bytes(100000, seed=90 + i))
        got = (tmp_mount / "log.bin").read_bytes()
        want = b"".join(rand_bytes(100000, seed=90 + i) for i in range(32))
        assert got == want
