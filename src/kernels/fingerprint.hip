// G3: batched BLAKE3-256 chunk fingerprints on gfx950.
// Grid = (groups_per_chunk, nchunks): each 128-thread workgroup hashes a
// 128-leaf (128 KiB) aligned subtree — 8 WGs per 1 MiB chunk, so a 64-chunk
// pipeline batch launches 512 WGs (r1's 4-WG split at 32-chunk batches put
// 128 WGs on the 256-CU chip and ran at ~254 GB/s in-pipeline vs ~2.2 TB/s
// standalone — profiles/engine_bench_r01_pipeline.md).
// A trailing finisher kernel combines the ≤8 subtree CVs (parent chain,
// ROOT on top). Subtree split correctness: BLAKE3's tree splits at the
// largest power of two ≤ n leaves, so 128-aligned groups are exact subtrees
// and the group CVs combine with the same pairwise-with-promotion rule
// (bit-exact against the official vectors in tests/kernels).
// Max chunk 1 MiB (1024 leaves; the engine enforces this).
#include "blake3_dev.h"

#include "launchers.h"

namespace edgehip {

constexpr int kLeavesPerGroup = 128;
constexpr int kGroupsPerChunk = kB3Groups;  // 8

__global__ __launch_bounds__(128) void k_blake3_chunks(
    const uint8_t* __restrict__ src, const DevChunk* __restrict__ chunks,
    uint8_t* __restrict__ out_fps /* 32 B per chunk */,
    uint32_t* __restrict__ group_cvs /* 8 u32 per (chunk, group) */) {
    __shared__ uint32_t cvA[kLeavesPerGroup][8];
    __shared__ uint32_t cvB[kLeavesPerGroup / 2][8];

    const int chunk = blockIdx.y;
    const int group = blockIdx.x;
    const DevChunk c = chunks[chunk];
    const uint8_t* base = src + c.src_off;
    const uint32_t len = c.len;
    const int tid = threadIdx.x;

    uint32_t nleaves = len == 0 ? 1 : (len + 1023) / 1024;
    uint32_t ngroups = (nleaves + kLeavesPerGroup - 1) / kLeavesPerGroup;
    if (ngroups > kGroupsPerChunk) {  // engine caps chunk_size at 1 MiB; guard
        if (group == 0 && tid == 0)
            for (int i = 0; i < 32; i++) out_fps[chunk * 32 + i] = 0xEE;
        return;
    }
    if ((uint32_t)group >= ngroups) return;

    uint32_t lo = (uint32_t)group * kLeavesPerGroup;
    uint32_t hi = min(nleaves, lo + kLeavesPerGroup);
    uint32_t gl = hi - lo;  // leaves in this group

    const bool single = nleaves == 1;
    for (uint32_t leaf = lo + tid; leaf < hi; leaf += blockDim.x) {
        uint32_t cv[8];
        uint32_t loff = leaf * 1024;
        uint32_t llen = len == 0 ? 0 : min(1024u, len - loff);
        b3_leaf_cv(base + loff, llen, leaf, single, cv);
#pragma unroll
        for (int i = 0; i < 8; i++) cvA[leaf - lo][i] = cv[i];
    }

    // Pairwise reduce with promotion of a trailing odd node. When this group
    // is the WHOLE chunk (ngroups == 1) the top parent carries ROOT and the
    // result goes straight to out_fps; otherwise the group CV is an internal
    // subtree CV (never ROOT) for the finisher kernel.
    const bool whole = ngroups == 1;
    uint32_t (*from)[8] = cvA;
    uint32_t (*to)[8] = cvB;
    uint32_t n = gl;
    while (n > 1) {
        __syncthreads();
        uint32_t pairs = n / 2;
        bool promote = n & 1;
        bool root_level = whole && (pairs == 1) && !promote;
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

    if (whole) {
        if (tid < 8) {
            uint32_t w = from[0][tid];
            uint8_t* o = out_fps + chunk * 32 + tid * 4;
            o[0] = w & 0xff;
            o[1] = (w >> 8) & 0xff;
            o[2] = (w >> 16) & 0xff;
            o[3] = (w >> 24) & 0xff;
        }
    } else if (tid < 8) {
        group_cvs[(chunk * kGroupsPerChunk + group) * 8 + tid] = from[0][tid];
    }
}

// Combine ≤4 subtree CVs per chunk: pairwise with promotion, ROOT on top.
__global__ __launch_bounds__(64) void k_blake3_finish(
    const DevChunk* __restrict__ chunks, const uint32_t* __restrict__ group_cvs,
    uint8_t* __restrict__ out_fps, int nchunks) {
    int chunk = blockIdx.x * blockDim.x + threadIdx.x;
    if (chunk >= nchunks) return;
    const DevChunk c = chunks[chunk];
    uint32_t nleaves = c.len == 0 ? 1 : (c.len + 1023) / 1024;
    uint32_t ngroups = (nleaves + kLeavesPerGroup - 1) / kLeavesPerGroup;
    if (ngroups <= 1 || ngroups > kGroupsPerChunk) return;  // done in pass 1

    uint32_t cv[kGroupsPerChunk][8];
    for (uint32_t g = 0; g < ngroups; g++)
        for (int i = 0; i < 8; i++)
            cv[g][i] = group_cvs[(chunk * kGroupsPerChunk + g) * 8 + i];
    uint32_t n = ngroups;
    while (n > 1) {
        uint32_t pairs = n / 2;
        bool promote = n & 1;
        bool root_level = (pairs == 1) && !promote;
        for (uint32_t i = 0; i < pairs; i++) {
            uint32_t out[8];
            b3_parent_cv(cv[2 * i], cv[2 * i + 1], root_level, out);
            for (int k = 0; k < 8; k++) cv[i][k] = out[k];
        }
        if (promote)
            for (int k = 0; k < 8; k++) cv[pairs][k] = cv[n - 1][k];
        n = pairs + (promote ? 1 : 0);
    }
    uint8_t* o = out_fps + chunk * 32;
    for (int i = 0; i < 8; i++) {
        uint32_t w = cv[0][i];
        o[4 * i] = w & 0xff;
        o[4 * i + 1] = (w >> 8) & 0xff;
        o[4 * i + 2] = (w >> 16) & 0xff;
        o[4 * i + 3] = (w >> 24) & 0xff;
    }
}

void launch_blake3_chunks(const uint8_t* src, const DevChunk* chunks, uint8_t* out_fps,
                          int nchunks, hipStream_t stream, uint32_t* group_scratch) {
    if (nchunks <= 0) return;
    hipLaunchKernelGGL(k_blake3_chunks, dim3(kGroupsPerChunk, nchunks),
                       dim3(kLeavesPerGroup), 0, stream, src, chunks, out_fps,
                       group_scratch);
    hipLaunchKernelGGL(k_blake3_finish, dim3((nchunks + 63) / 64), dim3(64), 0, stream,
                       chunks, group_scratch, out_fps, nchunks);
}

}  // namespace edgehip
