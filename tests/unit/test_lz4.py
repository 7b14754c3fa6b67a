"""LZ4: C++ compressor output must decode with the independent Python decoder,
round-trip through the C++ decoder, and short-circuit incompressible input."""

import pytest

from edgehip import get_lib
from edgehip.golden import lz4 as g
from tests.conftest import rand_bytes


def mk_compressible(n, seed=1):
    import numpy as np
    rng = np.random.default_rng(seed)
    # repetitive text-like data with runs
    pieces = []
    total = 0
    while total < n:
        word = bytes(rng.integers(97, 123, size=rng.integers(3, 9), dtype=np.uint8))
        rep = int(rng.integers(1, 20))
        pieces.append(word * rep)
        total += len(pieces[-1])
    return b"".join(pieces)[:n]


@pytest.mark.parametrize("n", [0, 1, 5, 12, 13, 64, 100, 4096, 65536])
def test_roundtrip_compressible(n):
    lib = get_lib()
    data = mk_compressible(n) if n else b""
    comp = lib.lz4_compress_block(data)
    if not comp:
        assert n == 0
        return
    assert lib.lz4_decompress_block(comp, n) == data
    assert g.decompress_block(comp, n) == data  # independent decoder


@pytest.mark.parametrize("n", [100, 4096, 65536])
def test_roundtrip_random(n):
    lib = get_lib()
    data = rand_bytes(n, seed=n)
    comp = lib.lz4_compress_block(data)
    assert comp is not None
    assert lib.lz4_decompress_block(comp, n) == data
    assert g.decompress_block(comp, n) == data


def test_zeros_compress_well():
    lib = get_lib()
    data = b"\x00" * 65536
    comp = lib.lz4_compress_block(data)
    assert comp and len(comp) < 300
    assert g.decompress_block(comp, 65536) == data


@pytest.mark.parametrize("n", [4096, 65536, 1 << 20, (1 << 20) - 333])
def test_chunk_framing_compressible(n):
    lib = get_lib()
    data = mk_compressible(n, seed=n)
    framed = lib.lz4_chunk_compress(data)
    assert framed is not None and len(framed) < n
    assert lib.lz4_chunk_decompress(framed, n) == data
    assert g.decompress_chunk(framed, n) == data


@pytest.mark.parametrize("n", [65536, 1 << 20])
def test_chunk_framing_incompressible_short_circuits(n):
    lib = get_lib()
    data = rand_bytes(n, seed=n + 7)
    # random bytes: framed form would be larger -> compressor declines
    assert lib.lz4_chunk_compress(data) is None


def test_mixed_chunk():
    """Chunk with one compressible and one incompressible 64 KiB block."""
    lib = get_lib()
    data = mk_compressible(64 * 1024, seed=3) + rand_bytes(64 * 1024, seed=4)
    framed = lib.lz4_chunk_compress(data)
    assert framed is not None and len(framed) < len(data)
    assert lib.lz4_chunk_decompress(framed, len(data)) == data
    assert g.decompress_chunk(framed, len(data)) == data


def test_corrupt_rejected():
    lib = get_lib()
    data = mk_compressible(8192, seed=9)
    framed = bytearray(lib.lz4_chunk_compress(data))
    framed[len(framed) // 2] ^= 0xFF
    out = lib.lz4_chunk_decompress(bytes(framed), 8192)
    # either decode error (None) or wrong bytes — never a crash
    assert out is None or out != data
