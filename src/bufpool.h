// Pooled chunk buffers. Every 1 MiB chunk that flows write->store would
// otherwise cost malloc + first-touch page faults + memset + copy; the pool
// hands out 2 MiB-aligned, MADV_HUGEPAGE, already-faulted buffers that cycle
// between the object layer's dirty buffers and the host store without copies.
#pragma once

#include <cstdlib>
#include <mutex>
#include <sys/mman.h>
#include <vector>

#include "common.h"

namespace edgehip {

// A chunk-sized buffer: either pool-owned (cap = pool block size) or an exact
// malloc'd blob (cap == 0 marks malloc; used for compressed payloads).
struct Blob {
    uint8_t* p = nullptr;
    uint32_t len = 0;   // valid bytes
    uint32_t cap = 0;   // pool block capacity; 0 => plain malloc of len bytes
    Blob() = default;
    Blob(uint8_t* ptr, uint32_t l, uint32_t c) : p(ptr), len(l), cap(c) {}
    Blob(const Blob&) = delete;
    Blob& operator=(const Blob&) = delete;
    Blob(Blob&& o) noexcept : p(o.p), len(o.len), cap(o.cap) { o.p = nullptr; o.len = o.cap = 0; }
    Blob& operator=(Blob&& o) noexcept {
        std::swap(p, o.p); std::swap(len, o.len); std::swap(cap, o.cap);
        return *this;
    }
    ~Blob();  // returns pool blocks to the pool, frees malloc blobs
    void release();
};

class BufPool {
  public:
    static BufPool& inst() {
        static BufPool p;
        return p;
    }

    // Uninitialized block of >= sz bytes (sz <= 16 MiB). Returned cap is the
    // block class size. On a GPU box blocks are hipHostMalloc-pinned so the
    // object layer's dirty buffers DMA straight into the engine (no staging
    // memcpy). NOTE: no MADV_HUGEPAGE on the pageable fallback — with THP
    // defrag=madvise the first touch does synchronous compaction (measured
    // 0.4 GB/s ingest vs 4+ GB/s with plain 4K faults).
    Blob get(uint32_t sz) {
        uint32_t cls = block_class(sz);
        {
            std::lock_guard<std::mutex> g(mu_);
            auto& fl = free_[class_idx(cls)];
            if (!fl.empty()) {
                uint8_t* p = fl.back();
                fl.pop_back();
                return Blob(p, sz, cls);
            }
        }
        void* p = alloc_block(cls);
        if (!p) throw std::bad_alloc();
        return Blob((uint8_t*)p, sz, cls);
    }

    // True when pool blocks are pinned (GPU runtime present).
    bool pinned() const { return pinned_; }

    void put(Blob&& b) {
        if (!b.p) return;
        if (b.cap == 0) {
            free(b.p);
        } else {
            std::lock_guard<std::mutex> g(mu_);
            free_[class_idx(b.cap)].push_back(b.p);
        }
        b.p = nullptr;
    }

    static Blob make_exact(uint32_t len) {  // malloc blob (compressed payloads)
        auto* p = (uint8_t*)malloc(len ? len : 1);
        if (!p) throw std::bad_alloc();
        return Blob(p, len, 0);
    }

    // Pre-fault `bytes` of pool memory (parallel first-touch), so ingest does
    // not pay page faults inline. Used by bench presets (conf.prealloc).
    void prealloc(uint64_t bytes, uint32_t block_sz, int threads = 8);

  private:
    // block classes: 64K,128K,256K,512K,1M,2M,4M,8M,16M
    static uint32_t block_class(uint32_t sz) {
        uint32_t c = 64 * 1024;
        while (c < sz) c <<= 1;
        return c;
    }
    static int class_idx(uint32_t cls) {
        int i = 0;
        uint32_t c = 64 * 1024;
        while (c < cls) { c <<= 1; i++; }
        return i < 9 ? i : 8;
    }
    void* alloc_block(uint32_t cls);  // bufpool.cc (may pin via HIP)

    std::mutex mu_;
    std::vector<uint8_t*> free_[9];
    bool pinned_ = false;
    bool probed_ = false;
};

inline void Blob::release() {
    if (!p) return;
    if (cap == 0) {
        free(p);
        p = nullptr;
    } else {
        Blob tmp = std::move(*this);
        BufPool::inst().put(std::move(tmp));
    }
    len = cap = 0;
}
inline Blob::~Blob() { release(); }

// Registry of CALLER-registered (hipHostRegister) host ranges that device
// kernels may write directly. hipHostGetDevicePointer alone is NOT a valid
// probe: on XNACK-capable boxes it succeeds for arbitrary pageable memory,
// and kernel writes to such pages are silently lost on others (read-path
// zeros corruption, caught by tests/kernels/test_concurrency.py on a cold
// box). Only ranges explicitly pinned through eh_mem_pin are eligible.
void pinreg_add(const void* p, size_t n);
void pinreg_del(const void* p);
bool pinreg_contains(const void* p, size_t n);

}  // namespace edgehip
