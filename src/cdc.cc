#include "cdc.h"

namespace edgehip {

const uint64_t* gear_table() {
    static uint64_t t[256];
    static bool init = [] {
        uint64_t x = 0x9E3779B97F4A7C15ull;  // fixed seed: format-stable
        for (int i = 0; i < 256; i++) {
            x += 0x9E3779B97F4A7C15ull;
            uint64_t z = x;
            z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
            z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
            t[i] = z ^ (z >> 31);
        }
        return true;
    }();
    (void)init;
    return t;
}

void cdc_candidates_host(const uint8_t* data, size_t len, uint64_t avg,
                         std::vector<uint32_t>* bitmap) {
    bitmap->assign((len + 31) / 32, 0);
    const uint64_t* T = gear_table();
    const uint64_t mask = avg - 1;
    uint64_t h = 0;
    for (size_t i = 0; i < len; i++) {
        h = (h << 1) + T[data[i]];
        if ((h & mask) == 0) (*bitmap)[i >> 5] |= 1u << (i & 31);
    }
}

void cdc_select(const uint32_t* bitmap, size_t len, uint64_t avg, bool final_tail,
                std::vector<size_t>* ends) {
    ends->clear();
    const size_t minc = avg / 4, maxc = avg * 4;
    size_t start = 0;
    while (start + minc <= len) {
        // first candidate bit in [start+minc-1, start+maxc-2], else forced
        size_t lo = start + minc - 1;
        size_t hi = std::min(start + maxc - 1, len);  // exclusive bit bound
        size_t found = SIZE_MAX;
        for (size_t w = lo >> 5; w <= (hi - 1) >> 5 && found == SIZE_MAX; w++) {
            uint32_t bits = bitmap[w];
            if (w == lo >> 5) bits &= ~0u << (lo & 31);
            if (w == (hi - 1) >> 5 && (hi & 31))
                bits &= (1u << (hi & 31)) - 1;
            if (bits) found = (w << 5) + (size_t)__builtin_ctz(bits);
        }
        size_t end;
        if (found != SIZE_MAX) {
            end = found + 1;
        } else if (start + maxc <= len) {
            end = start + maxc;  // forced max cut
        } else {
            break;  // tail shorter than max and no candidate
        }
        ends->push_back(end);
        start = end;
    }
    if (final_tail && start < len) ends->push_back(len);
}

void cdc_boundaries(const uint8_t* data, size_t len, uint64_t avg, bool final_tail,
                    std::vector<size_t>* ends) {
    std::vector<uint32_t> bm;
    cdc_candidates_host(data, len, avg, &bm);
    cdc_select(bm.data(), len, avg, final_tail, ends);
}

}  // namespace edgehip
