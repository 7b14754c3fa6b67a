// BLAKE3 device implementation for CDNA4 (integer-ALU bound; no MFMA —
// SURVEY.md §3.2 G3). One workgroup hashes one chunk: leaves (1 KiB each)
// lane-parallel, parent tree reduced through LDS with pairwise-promotion
// (provably equal to the BLAKE3 left-largest-power-of-two tree).
// Bit-identical to src/blake3.cc (host) and edgehip/golden/blake3.py.
#pragma once

#include "dev_common.h"

namespace edgehip {

EH_DEV uint32_t b3_rotr(uint32_t x, int n) { return (x >> n) | (x << (32 - n)); }

__constant__ __device__ const uint32_t kB3IV[8] = {
    0x6A09E667u, 0xBB67AE85u, 0x3C6EF372u, 0xA54FF53Au,
    0x510E527Fu, 0x9B05688Cu, 0x1F83D9ABu, 0x5BE0CD19u};

constexpr uint32_t B3_CHUNK_START = 1, B3_CHUNK_END = 2, B3_PARENT = 4, B3_ROOT = 8;

EH_DEV void b3_g(uint32_t* v, int a, int b, int c, int d, uint32_t mx, uint32_t my) {
    v[a] = v[a] + v[b] + mx;
    v[d] = b3_rotr(v[d] ^ v[a], 16);
    v[c] = v[c] + v[d];
    v[b] = b3_rotr(v[b] ^ v[c], 12);
    v[a] = v[a] + v[b] + my;
    v[d] = b3_rotr(v[d] ^ v[a], 8);
    v[c] = v[c] + v[d];
    v[b] = b3_rotr(v[b] ^ v[c], 7);
}

// One compression; m is permuted in place (clobbered).
EH_DEV void b3_compress(const uint32_t cv[8], uint32_t m[16], uint64_t counter,
                        uint32_t block_len, uint32_t flags, uint32_t out[8]) {
    uint32_t v[16];
#pragma unroll
    for (int i = 0; i < 8; i++) v[i] = cv[i];
#pragma unroll
    for (int i = 0; i < 4; i++) v[8 + i] = kB3IV[i];
    v[12] = (uint32_t)counter;
    v[13] = (uint32_t)(counter >> 32);
    v[14] = block_len;
    v[15] = flags;
    // 7 rounds with the fixed message schedule unrolled (PERM applied 6x)
#pragma unroll
    for (int round = 0; round < 7; round++) {
        b3_g(v, 0, 4, 8, 12, m[0], m[1]);
        b3_g(v, 1, 5, 9, 13, m[2], m[3]);
        b3_g(v, 2, 6, 10, 14, m[4], m[5]);
        b3_g(v, 3, 7, 11, 15, m[6], m[7]);
        b3_g(v, 0, 5, 10, 15, m[8], m[9]);
        b3_g(v, 1, 6, 11, 12, m[10], m[11]);
        b3_g(v, 2, 7, 8, 13, m[12], m[13]);
        b3_g(v, 3, 4, 9, 14, m[14], m[15]);
        if (round < 6) {
            uint32_t t0 = m[0];
            uint32_t t1 = m[1];
            m[0] = m[2]; m[2] = m[3]; m[3] = m[10]; m[10] = m[12]; m[12] = m[9];
            m[9] = m[11]; m[11] = m[5]; m[5] = t0;
            t0 = m[4];
            m[1] = m[6]; m[6] = t0; m[4] = m[7]; m[7] = m[13]; m[13] = m[14];
            m[14] = m[15]; m[15] = m[8]; m[8] = t1;
        }
    }
#pragma unroll
    for (int i = 0; i < 8; i++) out[i] = v[i] ^ v[i + 8];
}

// Hash one <=1024-byte leaf at tree position `counter`. Data length `len`
// (0 only valid for an empty whole-input chunk). root => whole input.
EH_DEV void b3_leaf_cv(const uint8_t* p, uint32_t len, uint64_t counter, bool root,
                       uint32_t cv[8]) {
#pragma unroll
    for (int i = 0; i < 8; i++) cv[i] = kB3IV[i];
    uint32_t nblocks = len == 0 ? 1 : (len + 63) / 64;
    for (uint32_t b = 0; b < nblocks; b++) {
        uint32_t off = b * 64;
        uint32_t blen = len == 0 ? 0 : min(64u, len - off);
        uint32_t m[16];
        if (blen == 64 && (((uintptr_t)(p + off)) & 3) == 0) {
            const uint32_t* w = (const uint32_t*)(p + off);
#pragma unroll
            for (int i = 0; i < 16; i++) m[i] = w[i];
        } else {
#pragma unroll
            for (int i = 0; i < 16; i++) m[i] = 0;
            for (uint32_t i = 0; i < blen; i++)
                m[i >> 2] |= (uint32_t)p[off + i] << (8 * (i & 3));
        }
        uint32_t flags = 0;
        if (b == 0) flags |= B3_CHUNK_START;
        if (b == nblocks - 1) {
            flags |= B3_CHUNK_END;
            if (root) flags |= B3_ROOT;
        }
        b3_compress(cv, m, counter, blen, flags, cv);
    }
}

EH_DEV void b3_parent_cv(const uint32_t l[8], const uint32_t r[8], bool root, uint32_t out[8]) {
    uint32_t m[16];
#pragma unroll
    for (int i = 0; i < 8; i++) { m[i] = l[i]; m[8 + i] = r[i]; }
    b3_compress(kB3IV, m, 0, 64, B3_PARENT | (root ? B3_ROOT : 0), out);
}

}  // namespace edgehip
