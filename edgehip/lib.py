"""ctypes bindings onto lib/libedgehip.so (built by `make`)."""

import ctypes
import os
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


class EhLib:
    def __init__(self, path=None):
        path = path or os.environ.get("EDGEHIP_LIB") or str(REPO / "lib" / "libedgehip.so")
        self._so = ctypes.CDLL(path)
        so = self._so
        so.eh_blake3.argtypes = [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_void_p]
        so.eh_crc32c.argtypes = [ctypes.c_void_p, ctypes.c_size_t]
        so.eh_crc32c.restype = ctypes.c_uint32
        so.eh_lz4_compress_block.argtypes = [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_void_p, ctypes.c_size_t]
        so.eh_lz4_compress_block.restype = ctypes.c_size_t
        so.eh_lz4_decompress_block.argtypes = [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_void_p, ctypes.c_size_t]
        so.eh_lz4_decompress_block.restype = ctypes.c_size_t
        so.eh_lz4_chunk_compress.argtypes = [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_void_p, ctypes.c_size_t]
        so.eh_lz4_chunk_compress.restype = ctypes.c_int64
        so.eh_lz4_chunk_decompress.argtypes = [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_void_p, ctypes.c_size_t]
        so.eh_lz4_chunk_decompress.restype = ctypes.c_int
        so.eh_rs_encode.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
        so.eh_rs_encode.restype = ctypes.c_int
        so.eh_rs_reconstruct.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
        so.eh_rs_reconstruct.restype = ctypes.c_int
        so.eh_rs_parity_matrix.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_void_p]
        so.eh_manifest_roundtrip.argtypes = [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_void_p, ctypes.c_size_t]
        so.eh_manifest_roundtrip.restype = ctypes.c_int64
        so.eh_engine_create.argtypes = [ctypes.c_char_p]
        so.eh_engine_create.restype = ctypes.c_void_p
        so.eh_engine_destroy.argtypes = [ctypes.c_void_p]
        so.eh_engine_stored_bytes.argtypes = [ctypes.c_void_p]
        so.eh_engine_stored_bytes.restype = ctypes.c_uint64
        so.eh_engine_stored_chunks.argtypes = [ctypes.c_void_p]
        so.eh_engine_stored_chunks.restype = ctypes.c_uint64
        so.eh_engine_is_gpu.argtypes = [ctypes.c_void_p]
        so.eh_engine_is_gpu.restype = ctypes.c_int
        so.eh_object_create.argtypes = [ctypes.c_void_p]
        so.eh_object_create.restype = ctypes.c_void_p
        so.eh_object_destroy.argtypes = [ctypes.c_void_p]
        so.eh_object_write.argtypes = [ctypes.c_void_p, ctypes.c_uint64, ctypes.c_void_p, ctypes.c_uint64]
        so.eh_object_write.restype = ctypes.c_int
        so.eh_object_read.argtypes = [ctypes.c_void_p, ctypes.c_uint64, ctypes.c_void_p, ctypes.c_uint64]
        so.eh_object_read.restype = ctypes.c_int64
        so.eh_object_flush.argtypes = [ctypes.c_void_p]
        so.eh_object_flush.restype = ctypes.c_int
        so.eh_object_truncate.argtypes = [ctypes.c_void_p, ctypes.c_uint64]
        so.eh_object_truncate.restype = ctypes.c_int
        so.eh_object_size.argtypes = [ctypes.c_void_p]
        so.eh_object_size.restype = ctypes.c_uint64
        so.eh_object_commit.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
        so.eh_object_commit.restype = ctypes.c_int64
        so.eh_object_preadv.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
                                        ctypes.c_void_p, ctypes.c_void_p]
        so.eh_object_preadv.restype = ctypes.c_int64
        so.eh_object_copy_range.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                            ctypes.c_uint64, ctypes.c_uint64,
                                            ctypes.c_uint64]
        so.eh_object_copy_range.restype = ctypes.c_int64
        so.eh_object_punch_hole.argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                            ctypes.c_uint64]
        so.eh_object_punch_hole.restype = ctypes.c_int
        so.eh_engine_ndev.argtypes = [ctypes.c_void_p]
        so.eh_engine_ndev.restype = ctypes.c_int
        so.eh_engine_chunk_device.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
        so.eh_engine_chunk_device.restype = ctypes.c_int
        so.eh_engine_mark_device_failed.argtypes = [ctypes.c_void_p, ctypes.c_int]
        so.eh_engine_mark_device_failed.restype = ctypes.c_int
        so.eh_metric.argtypes = [ctypes.c_char_p]
        so.eh_metric.restype = ctypes.c_uint64
        so.eh_mem_pin.argtypes = [ctypes.c_void_p, ctypes.c_size_t]
        so.eh_mem_pin.restype = ctypes.c_int
        so.eh_mem_unpin.argtypes = [ctypes.c_void_p]
        so.eh_mem_unpin.restype = ctypes.c_int

    # ---- convenience wrappers ----
    def blake3(self, data: bytes) -> bytes:
        out = ctypes.create_string_buffer(32)
        self._so.eh_blake3(data, len(data), out)
        return out.raw

    def crc32c(self, data: bytes) -> int:
        return self._so.eh_crc32c(data, len(data))

    def lz4_compress_block(self, data: bytes, max_out=None) -> bytes | None:
        cap = max_out if max_out is not None else len(data) + len(data) // 255 + 64
        out = ctypes.create_string_buffer(cap)
        n = self._so.eh_lz4_compress_block(data, len(data), out, cap)
        return out.raw[:n] if n else None

    def lz4_decompress_block(self, comp: bytes, logical: int) -> bytes | None:
        out = ctypes.create_string_buffer(logical)
        n = self._so.eh_lz4_decompress_block(comp, len(comp), out, logical)
        if n == ctypes.c_size_t(-1).value:
            return None
        return out.raw[:n]

    def lz4_chunk_compress(self, data: bytes) -> bytes | None:
        cap = len(data) + len(data) // 4 + 4096
        out = ctypes.create_string_buffer(cap)
        n = self._so.eh_lz4_chunk_compress(data, len(data), out, cap)
        if n <= 0:
            return None
        return out.raw[:n]

    def lz4_chunk_decompress(self, framed: bytes, logical: int) -> bytes | None:
        out = ctypes.create_string_buffer(logical)
        st = self._so.eh_lz4_chunk_decompress(framed, len(framed), out, logical)
        return out.raw if st == 0 else None

    def rs_parity_matrix(self, k: int, m: int) -> bytes:
        out = ctypes.create_string_buffer(k * m)
        self._so.eh_rs_parity_matrix(k, m, out)
        return out.raw

    def rs_encode(self, data_shards: list[bytes]) -> None:
        raise NotImplementedError("use rs_encode_km")

    def rs_encode_km(self, k: int, m: int, data_shards: list[bytes]) -> list[bytes]:
        assert len(data_shards) == k
        ln = len(data_shards[0])
        dbufs = [ctypes.create_string_buffer(s, ln) for s in data_shards]
        pbufs = [ctypes.create_string_buffer(ln) for _ in range(m)]
        darr = (ctypes.c_void_p * k)(*[ctypes.cast(b, ctypes.c_void_p) for b in dbufs])
        parr = (ctypes.c_void_p * m)(*[ctypes.cast(b, ctypes.c_void_p) for b in pbufs])
        rc = self._so.eh_rs_encode(k, m, darr, parr, ln)
        assert rc == 0
        return [b.raw[:ln] for b in pbufs]

    def rs_reconstruct_km(self, k: int, m: int, shards: list[bytes | None], ln: int) -> list[bytes]:
        bufs = []
        present = (ctypes.c_uint8 * (k + m))()
        for i, s in enumerate(shards):
            if s is None:
                bufs.append(ctypes.create_string_buffer(ln))
                present[i] = 0
            else:
                bufs.append(ctypes.create_string_buffer(s, ln))
                present[i] = 1
        arr = (ctypes.c_void_p * (k + m))(*[ctypes.cast(b, ctypes.c_void_p) for b in bufs])
        rc = self._so.eh_rs_reconstruct(k, m, arr, present, ln)
        if rc != 0:
            raise RuntimeError(f"rs_reconstruct rc={rc}")
        return [b.raw[:ln] for b in bufs]


_lib = None


def get_lib() -> EhLib:
    global _lib
    if _lib is None:
        _lib = EhLib()
    return _lib


def metric(name: str) -> int:
    """Global metrics counter by name (e.g. dedup_hits); -1 if unknown."""
    v = get_lib()._so.eh_metric(name.encode())
    return -1 if v == (1 << 64) - 1 else v


class Engine:
    """Engine+Object harness over ctypes (tests only)."""

    def __init__(self, opts: str = ""):
        self.lib = get_lib()
        self.h = self.lib._so.eh_engine_create(opts.encode())
        if not self.h:
            raise RuntimeError(f"engine create failed: {opts!r}")

    def close(self):
        if self.h:
            self.lib._so.eh_engine_destroy(self.h)
            self.h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    @property
    def stored_bytes(self):
        return self.lib._so.eh_engine_stored_bytes(self.h)

    @property
    def stored_chunks(self):
        return self.lib._so.eh_engine_stored_chunks(self.h)

    @property
    def is_gpu(self):
        return bool(self.lib._so.eh_engine_is_gpu(self.h))

    @property
    def ndev(self):
        return self.lib._so.eh_engine_ndev(self.h)

    def chunk_device(self, fp: bytes) -> int:
        """Device slot storing this chunk (-2 host/spill tier, -1 absent)."""
        return self.lib._so.eh_engine_chunk_device(self.h, fp)

    def mark_device_failed(self, dev: int) -> None:
        rc = self.lib._so.eh_engine_mark_device_failed(self.h, dev)
        if rc != 0:
            raise OSError(rc, f"mark_device_failed({dev}) rc={rc}")

    def pin(self, buf, n: int) -> bool:
        """hipHostRegister a writable buffer (mmap/bytearray) for DMA reads."""
        mv = memoryview(buf)
        b = (ctypes.c_ubyte * n).from_buffer(mv.cast("B"))
        return self.lib._so.eh_mem_pin(b, n) == 0

    def unpin(self, buf) -> None:
        mv = memoryview(buf)
        b = (ctypes.c_ubyte * mv.nbytes).from_buffer(mv.cast("B"))
        self.lib._so.eh_mem_unpin(b)

    def object(self):
        return Obj(self)


class Obj:
    def __init__(self, eng: Engine):
        self.eng = eng
        self.lib = eng.lib
        self.h = self.lib._so.eh_object_create(eng.h)

    def close(self):
        if self.h:
            self.lib._so.eh_object_destroy(self.h)
            self.h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def write(self, off: int, data):
        if isinstance(data, memoryview):
            n = data.nbytes
            buf = (ctypes.c_ubyte * n).from_buffer(data.cast("B"))
            rc = self.lib._so.eh_object_write(self.h, off, buf, n)
        else:
            rc = self.lib._so.eh_object_write(self.h, off, data, len(data))
        if rc != 0:
            raise OSError(-rc, "object write failed")

    def read(self, off: int, n: int) -> bytes:
        out = ctypes.create_string_buffer(n)
        got = self.lib._so.eh_object_read(self.h, off, out, n)
        if got < 0:
            raise OSError(-got, "object read failed")
        return out.raw[:got]

    def read_into(self, off: int, n: int, buf) -> int:
        """Read into a reusable writable buffer (no per-call allocation)."""
        mv = memoryview(buf)
        b = (ctypes.c_ubyte * n).from_buffer(mv.cast("B"))
        got = self.lib._so.eh_object_read(self.h, off, b, n)
        if got < 0:
            raise OSError(-got, "object read failed")
        return got

    def punch_hole(self, off: int, n: int) -> None:
        rc = self.lib._so.eh_object_punch_hole(self.h, off, n)
        if rc != 0:
            raise OSError(-rc, "punch_hole failed")

    def copy_range(self, src, soff: int, doff: int, n: int) -> int:
        """Server-side copy from `src` object: aligned spans share chunk refs
        (no data movement)."""
        got = self.lib._so.eh_object_copy_range(self.h, src.h, soff, doff, n)
        if got < 0:
            raise OSError(int(-got), "copy_range failed")
        return got

    def preadv(self, offs, lens, buf) -> int:
        """Batched scatter read of len(offs) ranges into buf (concatenated)."""
        n = len(offs)
        oa = (ctypes.c_uint64 * n)(*offs)
        la = (ctypes.c_uint32 * n)(*lens)
        mv = memoryview(buf)
        b = (ctypes.c_ubyte * mv.nbytes).from_buffer(mv.cast("B"))
        got = self.lib._so.eh_object_preadv(self.h, n, oa, la, b)
        if got < 0:
            raise OSError(int(-got), "preadv failed")
        return got

    def flush(self):
        rc = self.lib._so.eh_object_flush(self.h)
        if rc != 0:
            raise OSError(-rc, "flush failed")

    def truncate(self, size: int):
        rc = self.lib._so.eh_object_truncate(self.h, size)
        if rc != 0:
            raise OSError(-rc, "truncate failed")

    def commit(self) -> bytes:
        cap = 1 << 22
        out = ctypes.create_string_buffer(cap)
        n = self.lib._so.eh_object_commit(self.h, out, cap)
        if n < 0:
            raise OSError(int(-n), "commit failed")
        return out.raw[:n]

    @property
    def size(self):
        return self.lib._so.eh_object_size(self.h)
