#include "bufpool.h"

#include <hip/hip_runtime.h>

#include <map>
#include <thread>

#include "obs.h"

namespace edgehip {

void* BufPool::alloc_block(uint32_t cls) {
    // Prefer pinned host memory when a GPU runtime is live: hipMemcpyAsync
    // from these blocks is true async DMA (pageable memory would stage).
    if (!probed_) {
        probed_ = true;
        int n = 0;
        pinned_ = hipGetDeviceCount(&n) == hipSuccess && n > 0;
        if (pinned_) LOGI("bufpool: pinned host blocks (GPU present)");
    }
    if (pinned_) {
        void* p = nullptr;
        if (hipHostMalloc(&p, cls, hipHostMallocDefault) == hipSuccess) return p;
        pinned_ = false;  // fall back for good
        LOGW("bufpool: hipHostMalloc failed; falling back to pageable");
    }
    void* p = nullptr;
    if (posix_memalign(&p, 2 * MiB, cls) != 0) return nullptr;
    return p;
}

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

namespace {
std::mutex g_pinreg_mu;
std::map<uintptr_t, size_t> g_pinreg;  // start -> len
}  // namespace

void pinreg_add(const void* p, size_t n) {
    std::lock_guard<std::mutex> g(g_pinreg_mu);
    g_pinreg[(uintptr_t)p] = n;
}
void pinreg_del(const void* p) {
    std::lock_guard<std::mutex> g(g_pinreg_mu);
    g_pinreg.erase((uintptr_t)p);
}
bool pinreg_contains(const void* p, size_t n) {
    std::lock_guard<std::mutex> g(g_pinreg_mu);
    if (g_pinreg.empty()) return false;
    auto it = g_pinreg.upper_bound((uintptr_t)p);
    if (it == g_pinreg.begin()) return false;
    --it;
    return (uintptr_t)p >= it->first && (uintptr_t)p + n <= it->first + it->second;
}

}  // namespace edgehip
