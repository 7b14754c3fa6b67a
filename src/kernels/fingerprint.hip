// G3: batched BLAKE3-256 chunk fingerprints on gfx950.
// Grid = one 256-thread workgroup per chunk; each lane hashes leaves
// (1 KiB) with stride 256, CVs tree-reduce through double-buffered LDS.
// Max chunk 1 MiB (1024 leaves; the engine enforces this). Integer-ALU
// bound: ~12 ops/byte puts the ceiling near the HBM read rate.
#include "blake3_dev.h"

namespace edgehip {

constexpr int kMaxLeavesLds = 1024;

__global__ __launch_bounds__(256) void k_blake3_chunks(
    const uint8_t* __restrict__ src, const DevChunk* __restrict__ chunks,
    uint8_t* __restrict__ out_fps /* 32 B per chunk */) {
    __shared__ uint32_t cvA[kMaxLeavesLds][8];
    __shared__ uint32_t cvB[kMaxLeavesLds / 2][8];

    const DevChunk c = chunks[blockIdx.x];
    const uint8_t* base = src + c.src_off;
    const uint32_t len = c.len;
    const int tid = threadIdx.x;

    uint32_t nleaves = len == 0 ? 1 : (len + 1023) / 1024;
    if (nleaves > kMaxLeavesLds) {  // engine caps chunk_size at 1 MiB; guard
        if (tid == 0)
            for (int i = 0; i < 32; i++) out_fps[blockIdx.x * 32 + i] = 0xEE;
        return;
    }

    const bool single = nleaves == 1;
    for (uint32_t leaf = tid; leaf < nleaves; leaf += blockDim.x) {
        uint32_t cv[8];
        uint32_t loff = leaf * 1024;
        uint32_t llen = len == 0 ? 0 : min(1024u, len - loff);
        b3_leaf_cv(base + loff, llen, leaf, single, cv);
#pragma unroll
        for (int i = 0; i < 8; i++) cvA[leaf][i] = cv[i];
    }

    // Pairwise reduce with promotion of a trailing odd node; the final parent
    // carries ROOT. (Equal to BLAKE3's split-at-largest-power-of-two tree.)
    uint32_t (*from)[8] = cvA;
    uint32_t (*to)[8] = cvB;
    uint32_t n = nleaves;
    while (n > 1) {
        __syncthreads();
        uint32_t pairs = n / 2;
        bool promote = n & 1;
        bool root_level = (pairs == 1) && !promote;
        for (uint32_t i = tid; i < pairs; i += blockDim.x)
            b3_parent_cv(from[2 * i], from[2 * i + 1], root_level, to[i]);
        if (promote && tid == 0) {
#pragma unroll
            for (int k = 0; k < 8; k++) to[pairs][k] = from[n - 1][k];
        }
        n = pairs + (promote ? 1 : 0);
        uint32_t(*t)[8] = from;
        from = to;
        to = t;
    }
    __syncthreads();

    if (tid < 8) {
        uint32_t w = from[0][tid];
        uint8_t* o = out_fps + blockIdx.x * 32 + tid * 4;
        o[0] = w & 0xff;
        o[1] = (w >> 8) & 0xff;
        o[2] = (w >> 16) & 0xff;
        o[3] = (w >> 24) & 0xff;
    }
}

void launch_blake3_chunks(const uint8_t* src, const DevChunk* chunks, uint8_t* out_fps,
                          int nchunks, hipStream_t stream) {
    if (nchunks <= 0) return;
    hipLaunchKernelGGL(k_blake3_chunks, dim3(nchunks), dim3(256), 0, stream, src, chunks,
                       out_fps);
}

}  // namespace edgehip
