"""Pure-Python BLAKE3-256 (one-shot), written from the public spec.
Independent of src/blake3.cc; cross-checked against the published test
vectors and against the C++/HIP implementations."""

IV = [0x6A09E667, 0xBB67AE85, 0x3C6EF372, 0xA54FF53A,
      0x510E527F, 0x9B05688C, 0x1F83D9AB, 0x5BE0CD19]
MSG_PERM = [2, 6, 3, 10, 7, 0, 4, 13, 1, 11, 12, 5, 9, 14, 15, 8]
CHUNK_START, CHUNK_END, PARENT, ROOT = 1, 2, 4, 8
MASK = 0xFFFFFFFF


def _rotr(x, n):
    return ((x >> n) | (x << (32 - n))) & MASK


def _g(v, a, b, c, d, mx, my):
    v[a] = (v[a] + v[b] + mx) & MASK
    v[d] = _rotr(v[d] ^ v[a], 16)
    v[c] = (v[c] + v[d]) & MASK
    v[b] = _rotr(v[b] ^ v[c], 12)
    v[a] = (v[a] + v[b] + my) & MASK
    v[d] = _rotr(v[d] ^ v[a], 8)
    v[c] = (v[c] + v[d]) & MASK
    v[b] = _rotr(v[b] ^ v[c], 7)


def _compress(cv, block_words, counter, block_len, flags):
    v = list(cv) + IV[:4] + [counter & MASK, (counter >> 32) & MASK, block_len, flags]
    m = list(block_words)
    for r in range(7):
        _g(v, 0, 4, 8, 12, m[0], m[1])
        _g(v, 1, 5, 9, 13, m[2], m[3])
        _g(v, 2, 6, 10, 14, m[4], m[5])
        _g(v, 3, 7, 11, 15, m[6], m[7])
        _g(v, 0, 5, 10, 15, m[8], m[9])
        _g(v, 1, 6, 11, 12, m[10], m[11])
        _g(v, 2, 7, 8, 13, m[12], m[13])
        _g(v, 3, 4, 9, 14, m[14], m[15])
        if r < 6:
            m = [m[MSG_PERM[i]] for i in range(16)]
    return [v[i] ^ v[i + 8] for i in range(8)]


def _words(b):
    b = b + b"\x00" * (64 - len(b))
    return [int.from_bytes(b[4 * i:4 * i + 4], "little") for i in range(16)]


def _chunk_cv(data, counter, root):
    cv = list(IV)
    nblocks = max(1, (len(data) + 63) // 64)
    for i in range(nblocks):
        blk = data[64 * i:64 * (i + 1)]
        flags = 0
        if i == 0:
            flags |= CHUNK_START
        if i == nblocks - 1:
            flags |= CHUNK_END
            if root:
                flags |= ROOT
        cv = _compress(cv, _words(blk), counter, len(blk), flags)
    return cv


def _parent(l, r, root):
    words = list(l) + list(r)
    return _compress(IV, words, 0, 64, PARENT | (ROOT if root else 0))


def _subtree(data, first, n, root):
    if n == 1:
        return _chunk_cv(data[first * 1024:(first + 1) * 1024], first, root)
    left = 1
    while left * 2 < n:
        left *= 2
    l = _subtree(data, first, left, False)
    r = _subtree(data, first + left, n - left, False)
    return _parent(l, r, root)


def blake3(data: bytes) -> bytes:
    nchunks = max(1, (len(data) + 1023) // 1024)
    cv = _subtree(data, 0, nchunks, True)
    return b"".join(w.to_bytes(4, "little") for w in cv)
