// G7: the fingerprint dedup index in HBM — an open-addressing
// hash table of 8-byte fingerprint tags, one lane per key, insert-if-absent
// with device-scope atomicCAS (per-XCD L2s are not coherent; global atomics
// are agent-scope — guide §6 G12/G16). Powers judged config 5's ingest:
// the ingest pipeline consults the HBM index for dup/new before placing
// chunks; the host map stays authoritative for refcounts/locations, so a
// stale tag (after deletes, which do not shrink the index) only costs a
// confirmed-miss, never corrupts.
#include "dev_common.h"

namespace edgehip {

// table: capacity (power of two) u64 tags, 0 = empty. fps: 32 B each; the
// tag is the first 8 bytes (forced nonzero). out_dup[i] = 1 if the tag was
// already present, 0 if this thread inserted it. Full table => treated as
// dup=0 (host dedup still catches real dups).
__global__ __launch_bounds__(256) void k_dedup_upsert(
    unsigned long long* __restrict__ table, uint64_t mask,
    const uint8_t* __restrict__ fps, uint32_t* __restrict__ out_dup, int n) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    unsigned long long tag;
    __builtin_memcpy(&tag, fps + 32 * i, 8);
    if (tag == 0) tag = 1;  // reserve 0 for empty
    uint64_t h = tag * 0x9E3779B97F4A7C15ull;
    uint32_t dup = 0;
    for (uint64_t probe = 0; probe <= mask; probe++) {
        uint64_t slot = (h + probe) & mask;
        unsigned long long cur = table[slot];
        if (cur == tag) {
            dup = 1;
            break;
        }
        if (cur == 0) {
            unsigned long long prev = atomicCAS(&table[slot], 0ull, tag);
            if (prev == 0) break;          // inserted
            if (prev == tag) {
                dup = 1;
                break;
            }
            // raced with another insert of a different tag: keep probing
        }
    }
    out_dup[i] = dup;
}

// Batched presence check (no insert) — read path / tests.
__global__ __launch_bounds__(256) void k_dedup_lookup(
    const unsigned long long* __restrict__ table, uint64_t mask,
    const uint8_t* __restrict__ fps, uint32_t* __restrict__ out_hit, int n) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    unsigned long long tag;
    __builtin_memcpy(&tag, fps + 32 * i, 8);
    if (tag == 0) tag = 1;
    uint64_t h = tag * 0x9E3779B97F4A7C15ull;
    uint32_t hit = 0;
    for (uint64_t probe = 0; probe <= mask; probe++) {
        uint64_t slot = (h + probe) & mask;
        unsigned long long cur = table[slot];
        if (cur == tag) {
            hit = 1;
            break;
        }
        if (cur == 0) break;
    }
    out_hit[i] = hit;
}

void launch_dedup_upsert(unsigned long long* table, uint64_t mask, const uint8_t* fps,
                         uint32_t* out_dup, int n, hipStream_t stream) {
    if (n <= 0) return;
    hipLaunchKernelGGL(k_dedup_upsert, dim3((n + 255) / 256), dim3(256), 0, stream, table,
                       mask, fps, out_dup, n);
}

void launch_dedup_lookup(const unsigned long long* table, uint64_t mask, const uint8_t* fps,
                         uint32_t* out_hit, int n, hipStream_t stream) {
    if (n <= 0) return;
    hipLaunchKernelGGL(k_dedup_lookup, dim3((n + 255) / 256), dim3(256), 0, stream, table,
                       mask, fps, out_hit, n);
}

}  // namespace edgehip
