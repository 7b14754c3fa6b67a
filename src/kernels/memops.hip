// G9: batched device-side scatter/gather copies — chunk
// staging <-> arena moves and page slicing for small random reads, replacing
// per-chunk hipMemcpyDtoD launches with one batch kernel.
#include "dev_common.h"

namespace edgehip {

// descriptors: src_off/dst_off are absolute byte offsets into src/dst bases.
__global__ __launch_bounds__(256) void k_batch_copy(
    const uint8_t* __restrict__ src, uint8_t* __restrict__ dst,
    const DevChunk* __restrict__ descs, int ndescs) {
    int d = blockIdx.y;
    if (d >= ndescs) return;
    const DevChunk c = descs[d];
    const uint8_t* s = src + c.src_off;
    uint8_t* o = dst + c.dst_off;
    uint32_t len = c.len;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if ((((uintptr_t)s | (uintptr_t)o) & 15) == 0) {
        uint32_t vec = len / 16;
        for (uint64_t v = t; v < vec; v += stride)
            ((uint4*)o)[v] = ((const uint4*)s)[v];
        for (uint64_t i = (uint64_t)vec * 16 + t; i < len; i += stride) o[i] = s[i];
    } else {
        for (uint64_t i = t; i < len; i += stride) o[i] = s[i];
    }
}

void launch_batch_copy(const uint8_t* src, uint8_t* dst, const DevChunk* descs, int ndescs,
                       hipStream_t stream, uint32_t max_len) {
    if (ndescs <= 0) return;
    // x-dim sized for the largest descriptor (4096 B per WG per pass);
    // y-dim = descriptor index
    uint32_t bx = (max_len + 256 * 16 - 1) / (256 * 16);
    if (bx < 1) bx = 1;
    if (bx > 32) bx = 32;
    hipLaunchKernelGGL(k_batch_copy, dim3(bx, ndescs), dim3(256), 0, stream, src, dst,
                       descs, ndescs);
}

// Pull a batch of scattered PINNED host buffers into device staging with ONE
// kernel launch: the ingest H2D wall was per-hipMemcpyAsync SDMA setup
// (~40 us per 1 MiB chunk — profiles/engine_bench_r01; 64 copies/batch).
// gfx950 reads mapped pinned memory over PCIe at near-link rate when enough
// loads are in flight; 8 WGs x 1 wave per 1 MiB chunk with dwordx4 loads
// keeps ~2 KB per lane outstanding.
__global__ __launch_bounds__(256) void k_gather_host(
    const PtrChunk* __restrict__ descs, uint8_t* __restrict__ dst, int ndescs) {
    int d = blockIdx.y;
    if (d >= ndescs) return;
    const PtrChunk c = descs[d];
    const uint8_t* s = (const uint8_t*)c.src;
    uint8_t* o = dst + c.dst_off;
    uint32_t len = c.len;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if ((((uintptr_t)s | (uintptr_t)o) & 15) == 0) {
        uint32_t vec = len / 16;
        for (uint64_t v = t; v < vec; v += stride)
            ((uint4*)o)[v] = ((const uint4*)s)[v];
        for (uint64_t i = (uint64_t)vec * 16 + t; i < len; i += stride) o[i] = s[i];
    } else {
        for (uint64_t i = t; i < len; i += stride) o[i] = s[i];
    }
}

// PCIe copy kernels are latency-bound, not CU-bound: ~256 outstanding WGs
// saturate the link, and MORE just crowd the compute kernels off the CUs
// (profiled: 512-WG gathers slowed concurrent blake3/crc/lz4 4x). Budget
// the x-dim so grid ~= 256 WGs total.
static uint32_t copy_bx(int ndescs, uint32_t max_len) {
    uint32_t by_budget = ndescs > 0 ? 256u / (uint32_t)ndescs : 1;
    uint32_t by_len = (max_len + 256 * 512 - 1) / (256 * 512);  // 128 KiB/WG
    uint32_t bx = by_budget < by_len ? by_budget : by_len;
    return bx < 1 ? 1 : bx > 16 ? 16 : bx;
}

void launch_gather_host(const PtrChunk* descs, uint8_t* dst, int ndescs,
                        uint32_t max_len, hipStream_t stream) {
    if (ndescs <= 0) return;
    hipLaunchKernelGGL(k_gather_host, dim3(copy_bx(ndescs, max_len), ndescs),
                       dim3(256), 0, stream, descs, dst, ndescs);
}

// Mirror of k_gather_host for reads: push device bytes into scattered PINNED
// host destinations with one launch (posted PCIe writes run at link rate).
// Descriptor reuse: PtrChunk.src = host DESTINATION pointer,
// PtrChunk.dst_off = ABSOLUTE device source address (arena or staging).
__global__ __launch_bounds__(256) void k_scatter_host(
    const PtrChunk* __restrict__ descs, int ndescs) {
    int d = blockIdx.y;
    if (d >= ndescs) return;
    const PtrChunk c = descs[d];
    const uint8_t* s = (const uint8_t*)(uintptr_t)c.dst_off;
    uint8_t* o = (uint8_t*)c.src;
    uint32_t len = c.len;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if ((((uintptr_t)s | (uintptr_t)o) & 15) == 0) {
        uint32_t vec = len / 16;
        for (uint64_t v = t; v < vec; v += stride)
            ((uint4*)o)[v] = ((const uint4*)s)[v];
        for (uint64_t i = (uint64_t)vec * 16 + t; i < len; i += stride) o[i] = s[i];
    } else {
        for (uint64_t i = t; i < len; i += stride) o[i] = s[i];
    }
}

void launch_scatter_host(const PtrChunk* descs, int ndescs, uint32_t max_len,
                         hipStream_t stream) {
    if (ndescs <= 0) return;
    hipLaunchKernelGGL(k_scatter_host, dim3(copy_bx(ndescs, max_len), ndescs),
                       dim3(256), 0, stream, descs, ndescs);
}

}  // namespace edgehip
