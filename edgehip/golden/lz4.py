"""Pure-Python LZ4 block decoder + chunk-frame decoder, written from the public
LZ4 block-format spec. Independent of src/lz4.cc; used to validate that the
C++/HIP compressors emit spec-conformant streams."""

import struct

LZ4_MAGIC = 0x5A4C4845
LZ4_BLOCK = 64 * 1024


def decompress_block(src: bytes, expected: int) -> bytes:
    out = bytearray()
    i = 0
    n = len(src)
    while i < n:
        tok = src[i]
        i += 1
        lit = tok >> 4
        if lit == 15:
            while True:
                b = src[i]
                i += 1
                lit += b
                if b != 255:
                    break
        out += src[i:i + lit]
        i += lit
        if i == n:
            break
        off = src[i] | (src[i + 1] << 8)
        i += 2
        if off == 0 or off > len(out):
            raise ValueError("bad offset")
        mlen = (tok & 15) + 4
        if (tok & 15) == 15:
            while True:
                b = src[i]
                i += 1
                mlen += b
                if b != 255:
                    break
        for _ in range(mlen):
            out.append(out[-off])
    if len(out) != expected:
        raise ValueError(f"decompressed {len(out)} != expected {expected}")
    return bytes(out)


def decompress_chunk(framed: bytes, logical: int) -> bytes:
    magic, nblocks = struct.unpack_from("<II", framed, 0)
    if magic != LZ4_MAGIC:
        raise ValueError("bad magic")
    header = 8 + nblocks * 8
    payload = framed[header:]
    out = bytearray()
    for b in range(nblocks):
        off, len_flags = struct.unpack_from("<II", framed, 8 + 8 * b)
        blen = len_flags & 0x3FFFFFFF
        raw = bool(len_flags & 0x80000000)
        seg = bool(len_flags & 0x40000000)
        want = min(LZ4_BLOCK, logical - b * LZ4_BLOCK)
        blob = payload[off:off + blen]
        if raw:
            if blen != want:
                raise ValueError("raw block length mismatch")
            out += blob
        elif seg:
            # GPU segmented: u16 length table, segments packed = one stream;
            # also verify per-segment decode (the parallel property)
            ns = (want + 4095) // 4096
            lens = struct.unpack_from(f"<{ns}H", blob, 0)
            pos = ns * 2
            for s, ln in enumerate(lens):
                swant = min(4096, want - s * 4096)
                out += decompress_block(blob[pos:pos + ln], swant)
                pos += ln
            if pos != blen:
                raise ValueError("segment table inconsistent")
        else:
            out += decompress_block(blob, want)
    if len(out) != logical:
        raise ValueError("length mismatch")
    return bytes(out)
