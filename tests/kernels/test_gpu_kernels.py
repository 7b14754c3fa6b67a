"""HIP kernel numerics vs CPU goldens (bit-for-bit). Needs an MI355X."""

import ctypes

import pytest

from edgehip import get_lib
from edgehip.golden import gf256 as ggf
from edgehip.golden import lz4 as glz4
from edgehip.golden.blake3 import blake3 as py_blake3
from tests.conftest import rand_bytes

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def lib():
    lib = get_lib()
    so = lib._so
    so.eh_gpu_available.restype = ctypes.c_int
    so.eh_gpu_blake3.argtypes = [ctypes.c_void_p, ctypes.c_uint64, ctypes.c_int, ctypes.c_void_p]
    so.eh_gpu_blake3_one.argtypes = [ctypes.c_void_p, ctypes.c_uint64, ctypes.c_void_p]
    so.eh_gpu_lz4_compress.argtypes = [ctypes.c_void_p, ctypes.c_uint32, ctypes.c_void_p]
    so.eh_gpu_lz4_compress.restype = ctypes.c_int64
    so.eh_gpu_lz4_decompress.argtypes = [ctypes.c_void_p, ctypes.c_uint32, ctypes.c_void_p, ctypes.c_uint32]
    so.eh_gpu_lz4_decompress.restype = ctypes.c_int
    so.eh_gpu_gf_matmul.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
                                    ctypes.c_void_p, ctypes.c_uint64, ctypes.c_void_p]
    so.eh_gpu_gf_matmul.restype = ctypes.c_int
    so.eh_gpu_chunk_crc.argtypes = [ctypes.c_void_p, ctypes.c_uint32, ctypes.c_void_p]
    so.eh_gpu_chunk_crc.restype = ctypes.c_int
    if not so.eh_gpu_available():
        pytest.fail("no GPU visible — gpu tests must not silently skip on a GPU box")
    return lib


def mk_compressible(n, seed=1):
    import numpy as np
    rng = np.random.default_rng(seed)
    pieces, total = [], 0
    while total < n:
        word = bytes(rng.integers(97, 123, size=rng.integers(3, 9), dtype=np.uint8))
        pieces.append(word * int(rng.integers(1, 20)))
        total += len(pieces[-1])
    return b"".join(pieces)[:n]


# ---- BLAKE3 -----------------------------------------------------------------
@pytest.mark.parametrize("n", [0, 1, 63, 64, 65, 1023, 1024, 1025, 4096, 65536,
                               (1 << 18) - 1, 1 << 18, (1 << 18) + 1,  # 1/2-group edge
                               (1 << 19) + 12345, 700000, (3 << 18) + 1,  # partial groups
                               (1 << 20) - 1, 1 << 20])
def test_blake3_one_vs_host(lib, n):
    data = rand_bytes(n, seed=n)
    out = ctypes.create_string_buffer(32)
    assert lib._so.eh_gpu_blake3_one(data, n, out) == 0
    assert out.raw == lib.blake3(data), f"len={n}"


def test_blake3_batch(lib):
    n, cl = 16, 1 << 20
    data = rand_bytes(n * cl, seed=7)
    out = ctypes.create_string_buffer(32 * n)
    assert lib._so.eh_gpu_blake3(data, cl, n, out) == 0
    for i in range(n):
        chunk = data[i * cl:(i + 1) * cl]
        assert out.raw[32 * i:32 * (i + 1)] == lib.blake3(chunk), f"chunk {i}"


def test_blake3_official_vector_1mib(lib):
    # i % 251 pattern, 1 MiB — cross-checked against the pure-Python golden
    data = bytes(i % 251 for i in range(1 << 20))
    out = ctypes.create_string_buffer(32)
    assert lib._so.eh_gpu_blake3_one(data, len(data), out) == 0
    assert out.raw == py_blake3(data)


# ---- LZ4 --------------------------------------------------------------------
@pytest.mark.parametrize("n", [4096, 65536, 65536 + 4096, 1 << 20, (1 << 20) - 777])
def test_lz4_gpu_compress_host_decode(lib, n):
    data = mk_compressible(n, seed=n)
    out = ctypes.create_string_buffer(n + 4096)
    sz = lib._so.eh_gpu_lz4_compress(data, n, out)
    assert sz > 0, "compressible data must compress"
    framed = out.raw[:sz]
    # decode with host C++ and with the independent python golden
    assert lib.lz4_chunk_decompress(framed, n) == data
    assert glz4.decompress_chunk(framed, n) == data


@pytest.mark.parametrize("n", [65536, 1 << 20])
def test_lz4_gpu_incompressible_short_circuits(lib, n):
    data = rand_bytes(n, seed=n + 5)
    out = ctypes.create_string_buffer(n + 4096)
    sz = lib._so.eh_gpu_lz4_compress(data, n, out)
    assert sz == 0, "random bytes must be declined (stored raw by the engine)"


@pytest.mark.parametrize("n", [65536, 1 << 20])
def test_lz4_gpu_decompress_gpu_stream(lib, n):
    data = mk_compressible(n, seed=n + 9)
    out = ctypes.create_string_buffer(n + 4096)
    sz = lib._so.eh_gpu_lz4_compress(data, n, out)
    assert sz > 0
    dec = ctypes.create_string_buffer(n)
    assert lib._so.eh_gpu_lz4_decompress(out.raw[:sz], sz, dec, n) == 0
    assert dec.raw == data


def test_lz4_gpu_decompress_host_stream(lib):
    """GPU decoder must also read host-written (non-segmented) frames."""
    n = 1 << 20
    data = mk_compressible(n, seed=3)
    framed = lib.lz4_chunk_compress(data)
    assert framed is not None
    dec = ctypes.create_string_buffer(n)
    assert lib._so.eh_gpu_lz4_decompress(framed, len(framed), dec, n) == 0
    assert dec.raw == data


def test_lz4_mixed_blocks(lib):
    n = 256 * 1024
    data = (mk_compressible(64 * 1024, seed=1) + rand_bytes(64 * 1024, seed=2)
            + mk_compressible(64 * 1024, seed=3) + rand_bytes(64 * 1024, seed=4))
    out = ctypes.create_string_buffer(n + 4096)
    sz = lib._so.eh_gpu_lz4_compress(data, n, out)
    assert 0 < sz < n
    assert lib.lz4_chunk_decompress(out.raw[:sz], n) == data
    assert glz4.decompress_chunk(out.raw[:sz], n) == data


# ---- Reed-Solomon GF(256) ---------------------------------------------------
@pytest.mark.parametrize("k,m,ln", [(6, 2, 1 << 20), (6, 2, 65536), (6, 1, 4096),
                                    (4, 3, 8192), (10, 4, 4096), (6, 2, 4096 + 48)])
def test_gf_matmul_vs_golden(lib, k, m, ln):
    shards = [rand_bytes(ln, seed=100 + i) for i in range(k)]
    M = ggf.parity_matrix(k, m)
    coef = bytes(c for row in M for c in row)
    out = ctypes.create_string_buffer(ln * m)
    assert lib._so.eh_gpu_gf_matmul(b"".join(shards), k, m, coef, ln, out) == 0
    golden = ggf.encode(k, m, shards)
    for i in range(m):
        assert out.raw[i * ln:(i + 1) * ln] == golden[i], f"parity {i}"


def test_gf_matmul_reconstruct_matrix(lib):
    """Decode path: inverted-matrix rows recover erased shards exactly."""
    k, m, ln = 6, 2, 1 << 18
    data = [rand_bytes(ln, seed=i) for i in range(k)]
    parity = ggf.encode(k, m, data)
    full = data + parity
    # erase shards 1 and 4; surviving rows = identity rows + parity rows
    lost = [1, 4]
    rows, srcs = [], []
    for i in range(k + m):
        if i in lost:
            continue
        if len(rows) == k:
            break
        rows.append([1 if j == i else 0 for j in range(k)] if i < k else
                    ggf.parity_matrix(k, m)[i - k])
        srcs.append(full[i])
    Ainv = ggf.matrix_invert(rows, k)
    coef = bytes(Ainv[d][j] for d in lost for j in range(k))
    out = ctypes.create_string_buffer(ln * len(lost))
    assert lib._so.eh_gpu_gf_matmul(b"".join(srcs), k, len(lost), coef, ln, out) == 0
    for x, d in enumerate(lost):
        assert out.raw[x * ln:(x + 1) * ln] == data[d], f"reconstructed shard {d}"


# ---- crc --------------------------------------------------------------------
@pytest.mark.parametrize("n", [1, 4095, 4096, 4097, 65536, 1 << 20, (1 << 20) - 3])
def test_chunk_crc_vs_host(lib, n):
    import numpy as np
    data = rand_bytes(n, seed=n)
    out = ctypes.c_uint32(0)
    assert lib._so.eh_gpu_chunk_crc(data, n, ctypes.byref(out)) == 0
    # host: XOR of per-4K-page crc32c
    acc = 0
    for off in range(0, n, 4096):
        acc ^= lib.crc32c(data[off:off + 4096])
    assert out.value == acc


# ---- dedup index (G7) -------------------------------------------------------
def test_dedup_index_hbm(lib):
    """HBM hash table: first upsert of unique fps -> new; second pass -> dup;
    duplicate fps within one batch -> exactly one inserter."""
    import numpy as np
    so = lib._so
    so.eh_gpu_dedup_test.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
                                     ctypes.c_void_p]
    so.eh_gpu_dedup_test.restype = ctypes.c_int
    n = 4096
    rng = np.random.default_rng(5)
    uniq = rng.integers(0, 256, size=(n // 2, 32), dtype=np.uint8)
    fps = np.vstack([uniq, uniq]).tobytes()  # every fp appears exactly twice
    out1 = (ctypes.c_uint32 * n)()
    out2 = (ctypes.c_uint32 * n)()
    assert so.eh_gpu_dedup_test(fps, n, out1, out2) == 0
    # pass 1: each duplicated pair has exactly one inserter (dup sum == n/2)
    assert sum(out1) == n // 2
    # pass 2: everything is a dup
    assert sum(out2) == n


def test_gf_matmul_mfma_vs_golden(lib):
    """North-star-mandated MFMA bit-plane GF(256) variant: bit-exact vs the
    golden AND vs the v_perm table kernel."""
    so = lib._so
    so.eh_gpu_gf_matmul_mfma.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
                                         ctypes.c_void_p, ctypes.c_uint64, ctypes.c_void_p]
    so.eh_gpu_gf_matmul_mfma.restype = ctypes.c_int
    k, m, ln = 6, 2, 1 << 20
    shards = [rand_bytes(ln, seed=200 + i) for i in range(k)]
    M = ggf.parity_matrix(k, m)
    coef = bytes(c for row in M for c in row)
    out = ctypes.create_string_buffer(ln * m)
    assert so.eh_gpu_gf_matmul_mfma(b"".join(shards), k, m, coef, ln, out) == 0
    golden = ggf.encode(k, m, shards)
    for i in range(m):
        assert out.raw[i * ln:(i + 1) * ln] == golden[i], f"parity {i}"


# ---- CDC candidates (G2) ----------------------------------------------------
def test_cdc_gpu_boundaries_match_host(lib):
    from edgehip.golden import lz4 as _unused  # noqa: F401
    import ctypes as ct
    so = lib._so
    so.eh_gpu_cdc_boundaries.argtypes = [ct.c_void_p, ct.c_uint64, ct.c_uint64,
                                         ct.c_void_p, ct.c_int]
    so.eh_gpu_cdc_boundaries.restype = ct.c_int
    data = rand_bytes(8 << 20, seed=77)
    avg = 256 * 1024
    out = (ct.c_uint64 * 4096)()
    n = so.eh_gpu_cdc_boundaries(data, len(data), avg, out, 4096)
    assert n > 0
    gpu_ends = [out[i] for i in range(n)]
    # host reference through the same selection (capi has no host hook; use
    # the object layer indirectly): recompute via python? compare count sanity
    # + exact equality against host engine cut by writing through a host
    # engine with identical avg:
    from edgehip.lib import Engine
    with Engine(f"engine=host,fp=blake3,compress=none,dedup=0,cdc=1,"
                f"chunk_size={avg}") as eng, eng.object() as o:
        o.write(0, data)
        m = o.commit()
    import struct
    nchunks = struct.unpack_from("<Q", m, 40)[0]
    host_ends, acc = [], 0
    for i in range(nchunks):
        ll = struct.unpack_from("<I", m, 64 + 64 * i + 32)[0]
        acc += ll
        host_ends.append(acc)
    assert gpu_ends == host_ends
