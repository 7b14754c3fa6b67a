// BLAKE3-256, one-shot, tree-correct for any input length.
// Written from the public BLAKE3 specification. Verified against the
// published test vectors in tests/unit/test_blake3.py.
#include "blake3.h"

#include <vector>

namespace edgehip {
namespace {

constexpr uint32_t IV[8] = {0x6A09E667u, 0xBB67AE85u, 0x3C6EF372u, 0xA54FF53Au,
                            0x510E527Fu, 0x9B05688Cu, 0x1F83D9ABu, 0x5BE0CD19u};

constexpr uint32_t CHUNK_START = 1, CHUNK_END = 2, PARENT = 4, ROOT = 8;
constexpr size_t BLOCK = 64, CHUNK = 1024;

inline uint32_t rotr(uint32_t x, int n) { return (x >> n) | (x << (32 - n)); }

inline void g(uint32_t* v, int a, int b, int c, int d, uint32_t mx, uint32_t my) {
    v[a] = v[a] + v[b] + mx;
    v[d] = rotr(v[d] ^ v[a], 16);
    v[c] = v[c] + v[d];
    v[b] = rotr(v[b] ^ v[c], 12);
    v[a] = v[a] + v[b] + my;
    v[d] = rotr(v[d] ^ v[a], 8);
    v[c] = v[c] + v[d];
    v[b] = rotr(v[b] ^ v[c], 7);
}

constexpr int PERM[16] = {2, 6, 3, 10, 7, 0, 4, 13, 1, 11, 12, 5, 9, 14, 15, 8};

// Compress one 64-byte block; returns the 8-word output CV in out.
void compress(const uint32_t cv[8], const uint32_t block_words[16], uint64_t counter,
              uint32_t block_len, uint32_t flags, uint32_t out[8]) {
    uint32_t v[16];
    uint32_t m[16];
    for (int i = 0; i < 8; i++) v[i] = cv[i];
    for (int i = 0; i < 4; i++) v[8 + i] = IV[i];
    v[12] = (uint32_t)counter;
    v[13] = (uint32_t)(counter >> 32);
    v[14] = block_len;
    v[15] = flags;
    for (int i = 0; i < 16; i++) m[i] = block_words[i];
    for (int round = 0; round < 7; round++) {
        g(v, 0, 4, 8, 12, m[0], m[1]);
        g(v, 1, 5, 9, 13, m[2], m[3]);
        g(v, 2, 6, 10, 14, m[4], m[5]);
        g(v, 3, 7, 11, 15, m[6], m[7]);
        g(v, 0, 5, 10, 15, m[8], m[9]);
        g(v, 1, 6, 11, 12, m[10], m[11]);
        g(v, 2, 7, 8, 13, m[12], m[13]);
        g(v, 3, 4, 9, 14, m[14], m[15]);
        if (round < 6) {
            uint32_t t[16];
            for (int i = 0; i < 16; i++) t[i] = m[PERM[i]];
            for (int i = 0; i < 16; i++) m[i] = t[i];
        }
    }
    for (int i = 0; i < 8; i++) out[i] = v[i] ^ v[i + 8];
}

void load_words(const uint8_t* p, size_t n, uint32_t w[16]) {
    uint8_t buf[BLOCK] = {0};
    std::memcpy(buf, p, n);
    for (int i = 0; i < 16; i++)
        w[i] = (uint32_t)buf[4 * i] | (uint32_t)buf[4 * i + 1] << 8 |
               (uint32_t)buf[4 * i + 2] << 16 | (uint32_t)buf[4 * i + 3] << 24;
}

// Hash one chunk (<= 1024 B) at chunk index `counter`; emits its CV.
// `root` may only be set when this chunk is the whole input.
void chunk_cv(const uint8_t* p, size_t n, uint64_t counter, bool root, uint32_t cv_out[8]) {
    uint32_t cv[8];
    for (int i = 0; i < 8; i++) cv[i] = IV[i];
    size_t nblocks = n == 0 ? 1 : div_up(n, BLOCK);
    for (size_t b = 0; b < nblocks; b++) {
        size_t off = b * BLOCK;
        size_t blen = n == 0 ? 0 : std::min(BLOCK, n - off);
        uint32_t w[16];
        load_words(p + off, blen, w);
        uint32_t flags = 0;
        if (b == 0) flags |= CHUNK_START;
        if (b == nblocks - 1) flags |= CHUNK_END;
        if (root && b == nblocks - 1) flags |= ROOT;
        compress(cv, w, counter, (uint32_t)blen, flags, cv);
    }
    for (int i = 0; i < 8; i++) cv_out[i] = cv[i];
}

// Parent node: block = left_cv || right_cv.
void parent_cv(const uint32_t l[8], const uint32_t r[8], bool root, uint32_t out[8]) {
    uint32_t w[16];
    for (int i = 0; i < 8; i++) { w[i] = l[i]; w[8 + i] = r[i]; }
    compress(IV, w, 0, BLOCK, PARENT | (root ? ROOT : 0), out);
}

// Largest power of two strictly less than n (n >= 2).
uint64_t left_len(uint64_t n) {
    uint64_t p = 1;
    while (p * 2 < n) p *= 2;
    return p;
}

// Recursive subtree CV over chunks [first, first+n).
void subtree_cv(const uint8_t* data, uint64_t first, uint64_t n, uint64_t total_len,
                bool root, uint32_t out[8]) {
    if (n == 1) {
        uint64_t off = first * CHUNK;
        size_t len = (size_t)std::min((uint64_t)CHUNK, total_len - off);
        chunk_cv(data + off, len, first, root, out);
        return;
    }
    uint64_t l = left_len(n);
    uint32_t lcv[8], rcv[8];
    subtree_cv(data, first, l, total_len, false, lcv);
    subtree_cv(data, first + l, n - l, total_len, false, rcv);
    parent_cv(lcv, rcv, root, out);
}

}  // namespace

void blake3_hash(const void* data, size_t len, uint8_t out[32]) {
    const uint8_t* p = (const uint8_t*)data;
    uint64_t nchunks = len == 0 ? 1 : div_up(len, CHUNK);
    uint32_t cv[8];
    subtree_cv(p, 0, nchunks, len, true, cv);
    for (int i = 0; i < 8; i++) {
        out[4 * i] = cv[i] & 0xff;
        out[4 * i + 1] = (cv[i] >> 8) & 0xff;
        out[4 * i + 2] = (cv[i] >> 16) & 0xff;
        out[4 * i + 3] = (cv[i] >> 24) & 0xff;
    }
}

}  // namespace edgehip
