"""Durability: journal + chunk mirror -> remount-after-crash preserves every
committed byte. Uses real mounts (fuse marker)."""

import os
import signal
import subprocess
import time

import pytest

from edgehip.mount import Mount
from tests.conftest import rand_bytes

pytestmark = pytest.mark.fuse


def opts(jdir):
    return f"engine=host,fp=blake3,compress=lz4,dedup=1,journal={jdir},log=3"


def test_clean_remount(tmp_mount, tmp_path):
    jdir = tmp_path / "jrnl"
    data = rand_bytes(5 << 20, seed=1)
    with Mount(tmp_mount, opts(jdir)):
        (tmp_mount / "a" ).mkdir()
        (tmp_mount / "a" / "big.bin").write_bytes(data)
        (tmp_mount / "hello.txt").write_text("persistent\n")
        os.rename(tmp_mount / "hello.txt", tmp_mount / "a" / "hi.txt")
        (tmp_mount / "doomed.txt").write_text("x")
        (tmp_mount / "doomed.txt").unlink()
    # fresh process, same journal
    with Mount(tmp_mount, opts(jdir)):
        assert sorted(os.listdir(tmp_mount)) == ["a"]
        assert sorted(os.listdir(tmp_mount / "a")) == ["big.bin", "hi.txt"]
        assert (tmp_mount / "a" / "big.bin").read_bytes() == data
        assert (tmp_mount / "a" / "hi.txt").read_text() == "persistent\n"


def test_crash_kill9_preserves_fsynced(tmp_mount, tmp_path):
    jdir = tmp_path / "jrnl"
    data = rand_bytes(3 << 20, seed=2)
    m = Mount(tmp_mount, opts(jdir))
    m.__enter__()
    try:
        fd = os.open(tmp_mount / "synced.bin", os.O_WRONLY | os.O_CREAT)
        os.write(fd, data)
        os.fsync(fd)  # -> commit + journal fdatasync
        os.close(fd)
        (tmp_mount / "unsynced.bin").write_bytes(b"y" * 100)  # never fsynced/closed? (closed, so committed)
    finally:
        # hard kill: no unmount path runs
        m.proc.send_signal(signal.SIGKILL)
        m.proc.wait()
        subprocess.run(["umount", "-l", str(tmp_mount)], check=False,
                       stderr=subprocess.DEVNULL)
        time.sleep(0.2)
    with Mount(tmp_mount, opts(jdir)):
        assert (tmp_mount / "synced.bin").read_bytes() == data


def test_remount_after_overwrite(tmp_mount, tmp_path):
    jdir = tmp_path / "jrnl"
    v1 = rand_bytes(2 << 20, seed=3)
    v2 = rand_bytes(2 << 20, seed=4)
    with Mount(tmp_mount, opts(jdir)):
        (tmp_mount / "f.bin").write_bytes(v1)
    with Mount(tmp_mount, opts(jdir)):
        assert (tmp_mount / "f.bin").read_bytes() == v1
        (tmp_mount / "f.bin").write_bytes(v2)
    with Mount(tmp_mount, opts(jdir)):
        assert (tmp_mount / "f.bin").read_bytes() == v2


def test_symlink_hardlink_survive_remount(tmp_mount, tmp_path):
    jdir = tmp_path / "jrnl"
    with Mount(tmp_mount, opts(jdir)):
        (tmp_mount / "f.txt").write_text("x\n")
        (tmp_mount / "s").symlink_to("f.txt")
        os.link(tmp_mount / "f.txt", tmp_mount / "h")
    with Mount(tmp_mount, opts(jdir)):
        assert os.readlink(tmp_mount / "s") == "f.txt"
        assert (tmp_mount / "s").read_text() == "x\n"
        assert (tmp_mount / "h").read_text() == "x\n"
        assert (tmp_mount / "f.txt").stat().st_nlink == 2


def test_xattrs_survive_remount(tmp_mount, tmp_path):
    jdir = tmp_path / "jrnl"
    with Mount(tmp_mount, opts(jdir)):
        p = tmp_mount / "tagged.txt"
        p.write_text("data")
        os.setxattr(p, "user.keep", b"v1")
        os.setxattr(p, "user.gone", b"tmp")
        os.removexattr(p, "user.gone")
        os.setxattr(p, "user.keep", b"v2")  # overwrite: last write wins
    with Mount(tmp_mount, opts(jdir)):
        p = tmp_mount / "tagged.txt"
        assert os.listxattr(p) == ["user.keep"]
        assert os.getxattr(p, "user.keep") == b"v2"


def test_fsck_gc_orphans(tmp_mount, tmp_path):
    """fsck --gc removes chunk-mirror files no live manifest references
    (overwritten data and planted orphans), and live data survives."""
    import subprocess
    jdir = tmp_path / "jrnl"
    data = rand_bytes(3 << 20, seed=60)
    with Mount(tmp_mount, opts(jdir)):
        (tmp_mount / "keep.bin").write_bytes(data)
        (tmp_mount / "gone.bin").write_bytes(rand_bytes(2 << 20, seed=61))
        (tmp_mount / "gone.bin").unlink()
    # plant a fake orphan chunk file
    orphan = jdir / "chunks" / ("ff" * 32)
    orphan.write_bytes(b"junk")
    r = subprocess.run(["./bin/edgehip", "fsck", str(jdir)], capture_output=True,
                       text=True, timeout=60)
    assert r.returncode == 0, r.stderr
    assert "orphan" in r.stdout
    r = subprocess.run(["./bin/edgehip", "fsck", str(jdir), "--gc"],
                       capture_output=True, text=True, timeout=60)
    assert r.returncode == 0, r.stderr
    assert not orphan.exists()
    with Mount(tmp_mount, opts(jdir)):
        assert (tmp_mount / "keep.bin").read_bytes() == data


def test_rename_exchange_survives_remount(tmp_mount, tmp_path):
    import ctypes
    libc = ctypes.CDLL("libc.so.6", use_errno=True)
    RENAME_EXCHANGE, AT_FDCWD = 2, -100
    jdir = tmp_path / "jrnl"
    with Mount(tmp_mount, opts(jdir)):
        (tmp_mount / "p").write_text("PPP")
        (tmp_mount / "q").write_text("QQ")
        rc = libc.renameat2(AT_FDCWD, str(tmp_mount / "p").encode(),
                            AT_FDCWD, str(tmp_mount / "q").encode(),
                            RENAME_EXCHANGE)
        assert rc == 0, ctypes.get_errno()
        assert (tmp_mount / "p").read_text() == "QQ"
    with Mount(tmp_mount, opts(jdir)):
        assert (tmp_mount / "p").read_text() == "QQ"
        assert (tmp_mount / "q").read_text() == "PPP"


def test_crash_torture_random_kill(tmp_mount, tmp_path):
    """Randomized crash-consistency torture: kill -9 the daemon at a random
    moment mid-workload, remount, fsck must pass (rc 0/1, no crash) and every
    file that completed an fsync BEFORE the kill must read back exactly."""
    import random
    import threading
    rng = random.Random(99)
    jdir = tmp_path / "jrnl"
    for it in range(5):
        m = Mount(tmp_mount, opts(jdir))
        m.__enter__()
        synced = {}
        stop = threading.Event()

        def workload():
            i = 0
            while not stop.is_set():
                name = f"t{it}_{i}.bin"
                data = rand_bytes(rng.randrange(1, 3 << 20), seed=it * 1000 + i)
                try:
                    fd = os.open(tmp_mount / name, os.O_WRONLY | os.O_CREAT)
                    os.write(fd, data)
                    os.fsync(fd)
                    os.close(fd)
                    synced[name] = data  # fsync returned -> durable
                except OSError:
                    return  # daemon died mid-op: expected
                i += 1

        t = threading.Thread(target=workload)
        t.start()
        time.sleep(rng.uniform(0.05, 0.6))
        m.proc.send_signal(signal.SIGKILL)  # exact PID, never a pattern
        m.proc.wait()
        stop.set()
        t.join(timeout=30)
        subprocess.run(["umount", "-l", str(tmp_mount)], check=False,
                       stderr=subprocess.DEVNULL)
        time.sleep(0.1)
        r = subprocess.run(["./bin/edgehip", "fsck", str(jdir)],
                           capture_output=True, timeout=60)
        assert r.returncode in (0, 1), r.stderr[-300:]
        with Mount(tmp_mount, opts(jdir)):
            for name, data in synced.items():
                assert (tmp_mount / name).read_bytes() == data, (it, name)


def test_replay_resyncs_past_midlog_corruption(tmp_mount, tmp_path):
    """Mid-log damage must not silently discard every later fsync-acknowledged
    record: replay scans for the next record magic and
    continues; only the damaged record's effect is lost."""
    jdir = tmp_path / "jrnl"
    with Mount(tmp_mount, opts(jdir)):
        for n in ("early.txt", "victim.txt", "late.txt"):
            (tmp_mount / n).write_text(f"content of {n}\n")
            fd = os.open(tmp_mount / n, os.O_RDONLY)
            os.fsync(fd)
            os.close(fd)
    log = jdir / "journal.log"
    raw = bytearray(log.read_bytes())
    # corrupt one byte inside the record that mentions victim.txt
    at = raw.find(b"victim.txt")
    assert at > 0
    raw[at + 3] ^= 0xFF
    log.write_bytes(raw)
    with Mount(tmp_mount, opts(jdir)):
        names = set(os.listdir(tmp_mount))
        assert "early.txt" in names
        assert "late.txt" in names, "records after the corruption were dropped"
        assert (tmp_mount / "late.txt").read_text() == "content of late.txt\n"
        stats = (tmp_mount / ".edgehip-stats").read_text()
        resyncs = [l for l in stats.splitlines() if l.startswith("journal_resyncs")]
        assert resyncs and int(resyncs[0].split()[1]) >= 1


def test_real_file_named_reserved_tmp_survives_exchange(tmp_mount, tmp_path):
    """r1 journaled RENAME_EXCHANGE via a literal '.eh_xchg_tmp' temp name, so
    a real file with that name was clobbered on replay.
    Dedicated EXCHANGE records must leave it untouched."""
    import ctypes
    jdir = tmp_path / "jrnl"
    libc = ctypes.CDLL("libc.so.6", use_errno=True)
    with Mount(tmp_mount, opts(jdir)):
        (tmp_mount / ".eh_xchg_tmp").write_text("I am a real file\n")
        (tmp_mount / "x1").write_text("one\n")
        (tmp_mount / "x2").write_text("two\n")
        # renameat2(AT_FDCWD, x1, AT_FDCWD, x2, RENAME_EXCHANGE=2)
        rc = libc.renameat2(-100, str(tmp_mount / "x1").encode(), -100,
                            str(tmp_mount / "x2").encode(), 2)
        assert rc == 0, os.strerror(ctypes.get_errno())
        for n in ("x1", "x2", ".eh_xchg_tmp"):
            fd = os.open(tmp_mount / n, os.O_RDONLY)
            os.fsync(fd)
            os.close(fd)
    with Mount(tmp_mount, opts(jdir)):
        assert (tmp_mount / "x1").read_text() == "two\n"
        assert (tmp_mount / "x2").read_text() == "one\n"
        assert (tmp_mount / ".eh_xchg_tmp").read_text() == "I am a real file\n"
