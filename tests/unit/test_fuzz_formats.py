"""Format-robustness fuzz: decoding never crashes or over-reads on
truncated/corrupted inputs — manifests, LZ4 frames, and journal replay."""

import ctypes
import os
import random
import subprocess

import pytest

from edgehip import get_lib
from edgehip.lib import Engine
from tests.conftest import rand_bytes

MB = 1 << 20


def test_manifest_decode_fuzz():
    so = get_lib()._so
    with Engine("engine=host,fp=blake3,compress=lz4,dedup=1") as eng, \
            eng.object() as o:
        o.write(0, rand_bytes(3 * MB + 999, seed=50))
        m = o.commit()
    rng = random.Random(1)
    buf = ctypes.create_string_buffer(len(m) + 64)
    for _ in range(300):
        mb = bytearray(m)
        mode = rng.randrange(3)
        if mode == 0:  # truncate
            mb = mb[:rng.randrange(len(mb))]
        elif mode == 1:  # flip bytes
            for _ in range(rng.randrange(1, 8)):
                mb[rng.randrange(len(mb))] ^= rng.randrange(1, 256)
        else:  # random garbage
            mb = bytearray(rng.getrandbits(8) for _ in range(rng.randrange(200)))
        # must return an error or a bounded success — never crash
        so.eh_manifest_roundtrip(bytes(mb), len(mb), buf, len(m) + 64)


def test_lz4_chunk_decode_fuzz():
    so = get_lib()._so
    data = rand_bytes(MB, seed=51)
    comp = ctypes.create_string_buffer(MB + 4096)
    n = so.eh_lz4_chunk_compress(data, len(data), comp, MB + 4096)
    frame = comp.raw[:n] if n > 0 else bytes(data)
    rng = random.Random(2)
    out = ctypes.create_string_buffer(MB)
    for _ in range(200):
        fb = bytearray(frame)
        if rng.randrange(2):
            fb = fb[:rng.randrange(len(fb))]
        else:
            for _ in range(rng.randrange(1, 6)):
                fb[rng.randrange(len(fb))] ^= rng.randrange(1, 256)
        so.eh_lz4_chunk_decompress(bytes(fb), len(fb), out, MB)  # no crash


@pytest.mark.fuse
@pytest.mark.skipif("asan" in os.environ.get("EDGEHIP_LIB", "")
                    or "tsan" in os.environ.get("EDGEHIP_LIB", ""),
                    reason="spawns bin/edgehip; LD_PRELOADed sanitizer "
                           "runtime breaks the un-instrumented child")
def test_journal_replay_truncation_fuzz(tmp_path):
    """Replay of a journal truncated at any byte must not crash fsck and must
    yield a consistent (prefix) namespace."""
    from edgehip.mount import Mount
    mnt = tmp_path / "mnt"
    mnt.mkdir()
    jdir = tmp_path / "jrnl"
    with Mount(mnt, f"engine=host,fp=blake3,compress=lz4,dedup=1,journal={jdir}"):
        (mnt / "a").mkdir()
        (mnt / "a" / "f1").write_bytes(rand_bytes(2 * MB + 5, seed=52))
        os.setxattr(mnt / "a" / "f1", "user.k", b"v")
        (mnt / "a" / "f1").rename(mnt / "f2")
        (mnt / "f3").write_text("x" * 1000)
        (mnt / "f3").unlink()
    log = (jdir / "journal.log").read_bytes()
    rng = random.Random(3)
    cuts = [rng.randrange(len(log) + 1) for _ in range(25)] + [0, 1, len(log)]
    for cut in cuts:
        (jdir / "journal.log").write_bytes(log[:cut])
        r = subprocess.run(["./bin/edgehip", "fsck", str(jdir)],
                           capture_output=True, timeout=60)
        assert r.returncode in (0, 1), (cut, r.stderr[-300:])
    (jdir / "journal.log").write_bytes(log)
    r = subprocess.run(["./bin/edgehip", "fsck", str(jdir)], capture_output=True,
                       timeout=60)
    assert r.returncode == 0, r.stderr[-300:]
