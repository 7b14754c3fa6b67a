// BLAKE3-256 — host implementation (fingerprint algorithm).
// Used by the host engine, by golden tests, and for manifest CHIDs; the GPU
// engine computes the same function in src/kernels/fingerprint.hip (one
// workgroup per chunk, leaf CVs lane-parallel, LDS tree reduce) — both must
// agree bit-for-bit (tests/kernels/test_fingerprint.py).
#pragma once

#include <cstddef>
#include <cstdint>

#include "common.h"

namespace edgehip {

// One-shot BLAKE3 of [data, data+len) -> 32-byte digest.
void blake3_hash(const void* data, size_t len, uint8_t out[32]);

inline Fp blake3_fp(const void* data, size_t len) {
    Fp f;
    blake3_hash(data, len, f.b.data());
    return f;
}

}  // namespace edgehip
