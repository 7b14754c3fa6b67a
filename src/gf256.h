// GF(256) arithmetic + Reed–Solomon k+m erasure coding — host implementation
// (SURVEY.md §3.2 G6 golden model; judged config 4 is RS 6+2).
//
// Field: GF(2^8) with the storage-standard primitive polynomial 0x11D
// (x^8+x^4+x^3+x^2+1). Encoding matrix: [ I_k ; C ] where C is the m×k Cauchy
// matrix c_ij = 1/(x_i ^ y_j), x_i = k+i, y_j = j — every k×k submatrix of a
// Cauchy-extended identity is invertible, so ANY k surviving shards recover
// the stripe. The GPU path (src/kernels/rs_gf256.hip) computes the identical
// matrices; tests/kernels/test_rs.py cross-checks byte-for-byte.
#pragma once

#include <cstdint>
#include <vector>

#include "common.h"

namespace edgehip {

struct GF256 {
    uint8_t exp[512];
    uint8_t log[256];
    GF256();
    uint8_t mul(uint8_t a, uint8_t b) const {
        if (!a || !b) return 0;
        return exp[log[a] + log[b]];
    }
    uint8_t inv(uint8_t a) const { return exp[255 - log[a]]; }
};

const GF256& gf256();

// m×k Cauchy parity matrix for RS(k, m) (row-major).
std::vector<uint8_t> rs_parity_matrix(int k, int m);

// parity[i][*] = sum_j M[i*k+j] * data[j][*]; each shard is `len` bytes.
void rs_encode(int k, int m, const uint8_t* const* data, uint8_t* const* parity, size_t len);

// Reconstruct missing shards. shards[0..k-1] = data, shards[k..k+m-1] = parity;
// present[i] says which pointers hold valid data. Missing DATA shards are
// recomputed in place (their pointers must be writable). Returns St::inval if
// fewer than k shards are present. Missing parity shards are re-encoded.
St rs_reconstruct(int k, int m, uint8_t* const* shards, const bool* present, size_t len);

// Invert a k×k matrix over GF(256) (row-major). Returns false if singular.
bool gf_matrix_invert(std::vector<uint8_t>& a, int n);

}  // namespace edgehip
