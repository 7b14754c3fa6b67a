"""Manifest wire format: golden bytes frozen at v1 (docs/FORMATS.md §manifest).
If these bytes ever change, the on-wire format broke compatibility."""

import struct

from edgehip import get_lib

MAGIC = 0x314D4845  # 'EHM1'


def build_manifest(oid: bytes, chunks, logical_size, chunk_size=1 << 20,
                   fp_algo=1, compress=0):
    """Independent Python encoder of the v1 wire format."""
    out = bytearray()
    out += struct.pack("<IHBB", MAGIC, 1, fp_algo, compress)
    out += oid
    out += struct.pack("<QQQ", len(chunks), logical_size, chunk_size)
    for fp, logical_len, stored_len, flags, crc, shard_map in chunks:
        out += fp
        out += struct.pack("<IIIIQQ", logical_len, stored_len, flags, crc, shard_map, 0)
    out += struct.pack("<I", get_lib().crc32c(bytes(out)))
    return bytes(out)


def test_roundtrip_empty():
    lib = get_lib()
    enc = build_manifest(b"\x01" * 32, [], 0)
    out = bytearray(len(enc))
    import ctypes
    buf = ctypes.create_string_buffer(len(enc) + 64)
    n = lib._so.eh_manifest_roundtrip(enc, len(enc), buf, len(enc) + 64)
    assert n == len(enc)
    assert buf.raw[:n] == enc


def test_roundtrip_chunks():
    lib = get_lib()
    chunks = [
        (bytes(range(32)), 1 << 20, 1 << 20, 0, 0xDEADBEEF, 0),
        (bytes(reversed(range(32))), 1 << 20, 12345, 1, 0x12345678, 0x0706050403020100),
        (b"\xAA" * 32, 777, 777, 2, 0, 0xFF),
    ]
    enc = build_manifest(b"\x42" * 32, chunks, (2 << 20) + 777)
    import ctypes
    buf = ctypes.create_string_buffer(len(enc) + 64)
    n = lib._so.eh_manifest_roundtrip(enc, len(enc), buf, len(enc) + 64)
    assert n == len(enc)
    assert buf.raw[:n] == enc


def test_corrupt_rejected():
    lib = get_lib()
    enc = bytearray(build_manifest(b"\x01" * 32, [], 0))
    enc[10] ^= 1
    import ctypes
    buf = ctypes.create_string_buffer(len(enc) + 64)
    n = lib._so.eh_manifest_roundtrip(bytes(enc), len(enc), buf, len(enc) + 64)
    assert n == -1


def test_golden_bytes_frozen():
    """The exact v1 bytes for a known manifest — locks the wire format."""
    enc = build_manifest(b"\x07" * 32,
                         [(b"\x01" * 32, 100, 50, 1, 0xABCD, 3)], 100)
    assert enc[:8].hex() == "45484d31" + "0100" + "01" + "00"
    assert len(enc) == 64 + 64 + 4
    # header fields at fixed offsets
    assert struct.unpack_from("<Q", enc, 40)[0] == 1      # nchunks
    assert struct.unpack_from("<Q", enc, 48)[0] == 100    # logical_size
    assert struct.unpack_from("<Q", enc, 56)[0] == 1 << 20
