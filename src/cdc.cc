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

void cdc_boundaries(const uint8_t* data, size_t len, uint64_t avg, bool final_tail,
                    std::vector<size_t>* ends) {
    ends->clear();
    const uint64_t* T = gear_table();
    const uint64_t mask = avg - 1;
    const size_t minc = avg / 4, maxc = avg * 4;
    size_t start = 0;
    uint64_t h = 0;
    size_t i = 0;
    while (i < len) {
        h = (h << 1) + T[data[i]];
        i++;
        size_t clen = i - start;
        if ((clen >= minc && (h & mask) == 0) || clen >= maxc) {
            ends->push_back(i);
            start = i;
            h = 0;
        }
    }
    if (final_tail && start < len) ends->push_back(len);
}

}  // namespace edgehip
