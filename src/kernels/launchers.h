// Host-visible launch wrappers for the edgehip HIP kernels.
#pragma once

#include <hip/hip_runtime.h>

#include "dev_common.h"

namespace edgehip {

// group_scratch: 32 B x kB3Groups per chunk (subtree chaining values)
constexpr int kB3Groups = 8;
void launch_blake3_chunks(const uint8_t* src, const DevChunk* chunks, uint8_t* out_fps,
                          int nchunks, hipStream_t stream, uint32_t* group_scratch);

void launch_lz4_compress_segs(const uint8_t* src, const ulong2* seg_meta, uint8_t* scratch,
                              uint32_t* sizes, int nsegs, hipStream_t stream);
void launch_lz4_emit(const uint8_t* src, const uint8_t* scratch, const uint32_t* seg_sizes,
                     const DevChunk* chunks, uint8_t* dst_base, uint32_t* out_sizes,
                     int nchunks, hipStream_t stream);
void launch_lz4_decompress_chunks(const uint8_t* src, const DevChunk* chunks,
                                  uint8_t* dst_base, uint32_t* status, int nchunks,
                                  hipStream_t stream);

void launch_crc32c_chunks(const uint8_t* src, const DevChunk* chunks, uint32_t* out,
                          int nchunks, hipStream_t stream);

void launch_batch_copy(const uint8_t* src, uint8_t* dst, const DevChunk* descs, int ndescs,
                       hipStream_t stream, uint32_t max_len = 1 << 20);
// one-launch pull of scattered pinned host buffers into device staging
// (replaces per-chunk hipMemcpyAsync SDMA setup on the ingest path)
void launch_gather_host(const PtrChunk* descs, uint8_t* dst, int ndescs,
                        uint32_t max_len, hipStream_t stream);
// reads: device bytes -> scattered pinned host destinations, one launch
// (PtrChunk.src = host dst pointer, PtrChunk.dst_off = abs device src addr)
void launch_scatter_host(const PtrChunk* descs, int ndescs, uint32_t max_len,
                         hipStream_t stream);

void launch_cdc_candidates(const uint8_t* data, uint64_t len, uint64_t avg,
                           uint32_t* bitmap, hipStream_t stream, const uint64_t* gear_host);

void launch_dedup_upsert(unsigned long long* table, uint64_t mask, const uint8_t* fps,
                         uint32_t* out_dup, int n, hipStream_t stream);
void launch_dedup_lookup(const unsigned long long* table, uint64_t mask, const uint8_t* fps,
                         uint32_t* out_hit, int n, hipStream_t stream);

constexpr int kRsMaxK = 16;
constexpr int kRsMaxM = 4;
struct RsDesc {
    const uint8_t* in[kRsMaxK];
    uint8_t* out[kRsMaxM];
    uint8_t coef[kRsMaxM][kRsMaxK];
    int k, m;
    uint64_t len;
};
void launch_gf_matmul(const RsDesc& d, hipStream_t stream);
void launch_gf_matmul_mfma(const RsDesc& d, hipStream_t stream);  // bit-plane MFMA variant

}  // namespace edgehip
