// G2: content-defined chunking candidate scan on gfx950.
// The gear hash h_i = sum over the trailing 64 bytes (older bytes shift out
// of the 64-bit register), so candidates are computable in parallel: each
// thread warms up on the 64 bytes before its run, then rolls through its run
// emitting a bit per candidate position ((h & (avg-1)) == 0). Boundary
// SELECTION (min/max walk) stays on the host and is byte-identical to
// src/cdc.cc because the host hash also only depends on the trailing 64
// bytes once a chunk is at least 64 bytes old — and min chunk = avg/4 >= 64.
#include "dev_common.h"

namespace edgehip {

__constant__ __device__ uint64_t c_gear[256];

// bitmap: 1 bit per byte position, little-endian within u32 words.
__global__ __launch_bounds__(256) void k_cdc_candidates(
    const uint8_t* __restrict__ data, uint64_t len, uint64_t mask,
    uint32_t* __restrict__ bitmap, uint64_t run /* bytes per thread */) {
    uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint64_t start = t * run;
    if (start >= len) return;
    uint64_t end = min(start + run, len);
    uint64_t h = 0;
    uint64_t warm = start >= 64 ? start - 64 : 0;
    for (uint64_t i = warm; i < start; i++) h = (h << 1) + c_gear[data[i]];
    // roll through the run, accumulating candidate bits per 32-position word
    uint64_t i = start;
    while (i < end) {
        uint64_t word = i >> 5;
        uint32_t bits = 0;
        uint64_t wend = min(((word + 1) << 5), end);
        for (; i < wend; i++) {
            h = (h << 1) + c_gear[data[i]];
            if ((h & mask) == 0) bits |= 1u << (i & 31);
        }
        if (bits) atomicOr(&bitmap[word], bits);  // runs may share edge words
    }
}

void launch_cdc_candidates(const uint8_t* data, uint64_t len, uint64_t avg,
                           uint32_t* bitmap, hipStream_t stream,
                           const uint64_t* gear_host) {
    static bool gear_loaded = false;
    if (!gear_loaded) {
        (void)hipMemcpyToSymbol(HIP_SYMBOL(c_gear), gear_host, 256 * 8);
        gear_loaded = true;
    }
    (void)hipMemsetAsync(bitmap, 0, ((len + 31) / 32) * 4, stream);
    const uint64_t run = 4096;
    uint64_t threads = (len + run - 1) / run;
    int blocks = (int)((threads + 255) / 256);
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(k_cdc_candidates, dim3(blocks), dim3(256), 0, stream, data, len,
                       avg - 1, bitmap, run);
}

}  // namespace edgehip
