#include "gf256.h"

namespace edgehip {

GF256::GF256() {
    uint32_t x = 1;
    for (int i = 0; i < 255; i++) {
        exp[i] = (uint8_t)x;
        log[x] = (uint8_t)i;
        x <<= 1;
        if (x & 0x100) x ^= 0x11D;
    }
    for (int i = 255; i < 512; i++) exp[i] = exp[i - 255];
    log[0] = 0;  // undefined; mul() guards
}

const GF256& gf256() {
    static GF256 f;
    return f;
}

std::vector<uint8_t> rs_parity_matrix(int k, int m) {
    const GF256& f = gf256();
    std::vector<uint8_t> M((size_t)m * k);
    for (int i = 0; i < m; i++)
        for (int j = 0; j < k; j++)
            M[(size_t)i * k + j] = f.inv((uint8_t)((k + i) ^ j));
    return M;
}

void rs_encode(int k, int m, const uint8_t* const* data, uint8_t* const* parity, size_t len) {
    const GF256& f = gf256();
    auto M = rs_parity_matrix(k, m);
    for (int i = 0; i < m; i++) {
        uint8_t* out = parity[i];
        std::memset(out, 0, len);
        for (int j = 0; j < k; j++) {
            uint8_t c = M[(size_t)i * k + j];
            const uint8_t* in = data[j];
            if (c == 1) {
                for (size_t b = 0; b < len; b++) out[b] ^= in[b];
            } else {
                // split-table multiply (same scheme as the GPU kernel)
                uint8_t lo[16], hi[16];
                for (int t = 0; t < 16; t++) {
                    lo[t] = f.mul(c, (uint8_t)t);
                    hi[t] = f.mul(c, (uint8_t)(t << 4));
                }
                for (size_t b = 0; b < len; b++) out[b] ^= (uint8_t)(lo[in[b] & 15] ^ hi[in[b] >> 4]);
            }
        }
    }
}

bool gf_matrix_invert(std::vector<uint8_t>& a, int n) {
    const GF256& f = gf256();
    std::vector<uint8_t> inv((size_t)n * n, 0);
    for (int i = 0; i < n; i++) inv[(size_t)i * n + i] = 1;
    for (int col = 0; col < n; col++) {
        int piv = -1;
        for (int r = col; r < n; r++)
            if (a[(size_t)r * n + col]) { piv = r; break; }
        if (piv < 0) return false;
        if (piv != col) {
            for (int c = 0; c < n; c++) {
                std::swap(a[(size_t)piv * n + c], a[(size_t)col * n + c]);
                std::swap(inv[(size_t)piv * n + c], inv[(size_t)col * n + c]);
            }
        }
        uint8_t d = f.inv(a[(size_t)col * n + col]);
        for (int c = 0; c < n; c++) {
            a[(size_t)col * n + c] = f.mul(a[(size_t)col * n + c], d);
            inv[(size_t)col * n + c] = f.mul(inv[(size_t)col * n + c], d);
        }
        for (int r = 0; r < n; r++) {
            if (r == col) continue;
            uint8_t c0 = a[(size_t)r * n + col];
            if (!c0) continue;
            for (int c = 0; c < n; c++) {
                a[(size_t)r * n + c] ^= f.mul(c0, a[(size_t)col * n + c]);
                inv[(size_t)r * n + c] ^= f.mul(c0, inv[(size_t)col * n + c]);
            }
        }
    }
    a = std::move(inv);
    return true;
}

St rs_reconstruct(int k, int m, uint8_t* const* shards, const bool* present, size_t len) {
    const GF256& f = gf256();
    int navail = 0;
    for (int i = 0; i < k + m; i++) navail += present[i];
    if (navail < k) return St::inval;

    bool any_data_missing = false;
    for (int i = 0; i < k; i++) any_data_missing |= !present[i];

    if (any_data_missing) {
        // Build the k×k matrix of encoding rows for the first k present shards.
        auto P = rs_parity_matrix(k, m);
        std::vector<uint8_t> A((size_t)k * k, 0);
        std::vector<const uint8_t*> src(k);
        int r = 0;
        for (int i = 0; i < k + m && r < k; i++) {
            if (!present[i]) continue;
            if (i < k) A[(size_t)r * k + i] = 1;  // identity row
            else std::memcpy(&A[(size_t)r * k], &P[(size_t)(i - k) * k], k);
            src[r] = shards[i];
            r++;
        }
        if (!gf_matrix_invert(A, k)) return St::io;
        // missing data shard d = row d of A^{-1} × available shards
        for (int d = 0; d < k; d++) {
            if (present[d]) continue;
            uint8_t* out = shards[d];
            std::memset(out, 0, len);
            for (int j = 0; j < k; j++) {
                uint8_t c = A[(size_t)d * k + j];
                if (!c) continue;
                const uint8_t* in = src[j];
                if (c == 1) {
                    for (size_t b = 0; b < len; b++) out[b] ^= in[b];
                } else {
                    uint8_t lo[16], hi[16];
                    for (int t = 0; t < 16; t++) {
                        lo[t] = f.mul(c, (uint8_t)t);
                        hi[t] = f.mul(c, (uint8_t)(t << 4));
                    }
                    for (size_t b = 0; b < len; b++) out[b] ^= (uint8_t)(lo[in[b] & 15] ^ hi[in[b] >> 4]);
                }
            }
        }
    }
    // re-encode any missing parity from (now complete) data
    bool any_parity_missing = false;
    for (int i = 0; i < m; i++) any_parity_missing |= !present[k + i];
    if (any_parity_missing) {
        auto P = rs_parity_matrix(k, m);
        for (int i = 0; i < m; i++) {
            if (present[k + i]) continue;
            uint8_t* out = shards[k + i];
            std::memset(out, 0, len);
            for (int j = 0; j < k; j++) {
                uint8_t c = P[(size_t)i * k + j];
                const uint8_t* in = shards[j];
                uint8_t lo[16], hi[16];
                for (int t = 0; t < 16; t++) {
                    lo[t] = f.mul(c, (uint8_t)t);
                    hi[t] = f.mul(c, (uint8_t)(t << 4));
                }
                for (size_t b = 0; b < len; b++) out[b] ^= (uint8_t)(lo[in[b] & 15] ^ hi[in[b] >> 4]);
            }
        }
    }
    return St::ok;
}

}  // namespace edgehip
