"""Pure-Python GF(256) + Reed-Solomon golden model (poly 0x11D, Cauchy parity
matrix c_ij = inv((k+i) ^ j)). Independent of src/gf256.cc."""

import numpy as np

_EXP = [0] * 512
_LOG = [0] * 256


def _init():
    x = 1
    for i in range(255):
        _EXP[i] = x
        _LOG[x] = i
        x <<= 1
        if x & 0x100:
            x ^= 0x11D
    for i in range(255, 512):
        _EXP[i] = _EXP[i - 255]


_init()


def gmul(a, b):
    if a == 0 or b == 0:
        return 0
    return _EXP[_LOG[a] + _LOG[b]]


def ginv(a):
    return _EXP[255 - _LOG[a]]


def parity_matrix(k, m):
    return [[ginv((k + i) ^ j) for j in range(k)] for i in range(m)]


def _mul_table(c):
    return np.array([gmul(c, x) for x in range(256)], dtype=np.uint8)


def encode(k, m, data_shards):
    """data_shards: list of k equal-length bytes. Returns m parity bytes."""
    M = parity_matrix(k, m)
    n = len(data_shards[0])
    data = [np.frombuffer(s, dtype=np.uint8) for s in data_shards]
    out = []
    for i in range(m):
        acc = np.zeros(n, dtype=np.uint8)
        for j in range(k):
            acc ^= _mul_table(M[i][j])[data[j]]
        out.append(acc.tobytes())
    return out


def matrix_invert(A, n):
    A = [row[:] for row in A]
    I = [[1 if i == j else 0 for j in range(n)] for i in range(n)]
    for col in range(n):
        piv = next((r for r in range(col, n) if A[r][col]), None)
        if piv is None:
            raise ValueError("singular")
        A[col], A[piv] = A[piv], A[col]
        I[col], I[piv] = I[piv], I[col]
        d = ginv(A[col][col])
        A[col] = [gmul(x, d) for x in A[col]]
        I[col] = [gmul(x, d) for x in I[col]]
        for r in range(n):
            if r == col or not A[r][col]:
                continue
            c0 = A[r][col]
            A[r] = [a ^ gmul(c0, b) for a, b in zip(A[r], A[col])]
            I[r] = [a ^ gmul(c0, b) for a, b in zip(I[r], I[col])]
    return I


def reconstruct(k, m, shards):
    """shards: list of k+m entries, bytes or None. Returns fully repaired list."""
    n = next(len(s) for s in shards if s is not None)
    present = [s is not None for s in shards]
    if sum(present) < k:
        raise ValueError("too few shards")
    P = parity_matrix(k, m)
    rows, srcs = [], []
    for i in range(k + m):
        if not present[i]:
            continue
        if len(rows) == k:
            break
        rows.append([1 if j == i else 0 for j in range(k)] if i < k else P[i - k])
        srcs.append(np.frombuffer(shards[i], dtype=np.uint8))
    Ainv = matrix_invert(rows, k)
    out = list(shards)
    for d in range(k):
        if present[d]:
            continue
        acc = np.zeros(n, dtype=np.uint8)
        for j in range(k):
            c = Ainv[d][j]
            if c:
                acc ^= _mul_table(c)[srcs[j]]
        out[d] = acc.tobytes()
    datavecs = [np.frombuffer(out[j], dtype=np.uint8) for j in range(k)]
    for i in range(m):
        if present[k + i]:
            continue
        acc = np.zeros(n, dtype=np.uint8)
        for j in range(k):
            c = P[i][j]
            if c:
                acc ^= _mul_table(c)[datavecs[j]]
        out[k + i] = acc.tobytes()
    return out
