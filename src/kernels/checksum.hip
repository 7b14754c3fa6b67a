// G8: verify-lite integrity checksum. ChunkRef.crc is the
// XOR of crc32c over each 4 KiB page of the stored bytes (docs/FORMATS.md
// §checksum) — XOR-combining makes the computation page-parallel on the GPU
// while the host side stays a cheap SSE4.2 loop (src/obs.cc chunk_crc).
#include "dev_common.h"

namespace edgehip {

// Grid = (kCrcSlices, nchunks), block = one wave (64): the page-XOR
// structure makes the whole checksum order-free, so each chunk splits
// across kCrcSlices workgroups that atomicXor their partial into
// out[chunk] (zeroed by the launcher). A 1 MiB chunk has 256 pages =
// 4 slices x 64 threads, exactly one page per lane. r1's one-256-thread-
// WG-per-chunk shape put 32 WGs on the 256-CU chip and ran at 137 GB/s
// in-pipeline; 4 x 64 chunks = 256 single-wave WGs.
constexpr int kCrcSlices = 4;
constexpr int kCrcBlock = 64;

__global__ __launch_bounds__(kCrcBlock) void k_crc32c_chunks(
    const uint8_t* __restrict__ src, const DevChunk* __restrict__ chunks,
    uint32_t* __restrict__ out /* one per chunk */, int nchunks) {
    (void)nchunks;
    __shared__ uint32_t tab[8][256];  // slice-by-8: 8 B per iteration
    // build tables cooperatively
    for (int i = threadIdx.x; i < 256; i += blockDim.x) {
        uint32_t c = i;
        for (int k = 0; k < 8; k++) c = (c & 1) ? (0x82F63B78u ^ (c >> 1)) : (c >> 1);
        tab[0][i] = c;
    }
    __syncthreads();
    for (int t = 1; t < 8; t++) {
        for (int i = threadIdx.x; i < 256; i += blockDim.x)
            tab[t][i] = (tab[t - 1][i] >> 8) ^ tab[0][tab[t - 1][i] & 0xff];
        __syncthreads();
    }

    const DevChunk c = chunks[blockIdx.y];
    const uint8_t* base = src + c.src_off;
    uint32_t npages = (c.len + 4095) / 4096;
    uint32_t acc = 0;
    for (uint32_t p = blockIdx.x * blockDim.x + threadIdx.x; p < npages;
         p += gridDim.x * blockDim.x) {
        uint32_t off = p * 4096;
        uint32_t len = min(4096u, c.len - off);
        const uint8_t* q = base + off;
        uint32_t crc = 0xFFFFFFFFu;
        uint32_t i = 0;
        if ((((uintptr_t)q) & 7) == 0) {
            for (; i + 8 <= len; i += 8) {
                uint32_t w0, w1;
                __builtin_memcpy(&w0, q + i, 4);
                __builtin_memcpy(&w1, q + i + 4, 4);
                uint32_t v = w0 ^ crc;
                crc = tab[7][v & 0xff] ^ tab[6][(v >> 8) & 0xff] ^
                      tab[5][(v >> 16) & 0xff] ^ tab[4][(v >> 24) & 0xff] ^
                      tab[3][w1 & 0xff] ^ tab[2][(w1 >> 8) & 0xff] ^
                      tab[1][(w1 >> 16) & 0xff] ^ tab[0][(w1 >> 24) & 0xff];
            }
        }
        for (; i < len; i++) crc = tab[0][(crc ^ q[i]) & 0xff] ^ (crc >> 8);
        acc ^= ~crc;
    }
    // single-wave block: the wave XOR-reduce IS the block reduce
#pragma unroll
    for (int s = 32; s >= 1; s >>= 1) acc ^= __shfl_xor(acc, s, 64);
    if (threadIdx.x == 0) {
        if (gridDim.x == 1)
            out[blockIdx.y] = acc;
        else if (acc)  // out is pre-zeroed; XOR-combine slice partials
            atomicXor(out + blockIdx.y, acc);
    }
}

void launch_crc32c_chunks(const uint8_t* src, const DevChunk* chunks, uint32_t* out,
                          int nchunks, hipStream_t stream) {
    if (nchunks <= 0) return;
    (void)hipMemsetAsync(out, 0, sizeof(uint32_t) * nchunks, stream);
    hipLaunchKernelGGL(k_crc32c_chunks, dim3(kCrcSlices, nchunks), dim3(kCrcBlock), 0,
                       stream, src, chunks, out, nchunks);
}

}  // namespace edgehip
