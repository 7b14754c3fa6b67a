#include "bufpool.h"

#include <thread>

namespace edgehip {

void BufPool::prealloc(uint64_t bytes, uint32_t block_sz, int nthreads) {
    if (!bytes) return;
    uint64_t nblocks = div_up(bytes, block_sz);
    std::vector<std::vector<Blob>> held((size_t)nthreads);
    std::vector<std::thread> ts;
    for (int t = 0; t < nthreads; t++) {
        ts.emplace_back([&, t] {
            uint64_t mine = nblocks / nthreads + (t < (int)(nblocks % nthreads) ? 1 : 0);
            held[t].reserve(mine);
            for (uint64_t i = 0; i < mine; i++) {
                Blob b = get(block_sz);
                std::memset(b.p, 0, block_sz);  // first touch
                held[t].push_back(std::move(b));
            }
        });
    }
    for (auto& th : ts) th.join();
    // destructors return everything to the free lists
}

}  // namespace edgehip
