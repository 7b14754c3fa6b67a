// Shared device-side helpers for edgehip HIP kernels (gfx950 / CDNA4 only).
// Wave = 64 lanes, block = 256 threads throughout (guide: Appendix A).
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace edgehip {

#define EH_DEV __device__ __forceinline__

constexpr int kWave = 64;  // CDNA4 wavefront — hard-coded per guide §1

// Chunk descriptor passed to batch kernels (device-visible array).
struct DevChunk {
    uint64_t src_off;   // byte offset into the batch source buffer
    uint64_t dst_off;   // byte offset into the batch destination buffer
    uint32_t len;       // logical length
    uint32_t aux;       // kernel-specific
};

// pointer-source copy descriptor (k_gather_host: pinned host -> device)
struct PtrChunk {
    const void* src;    // device-accessible (pinned) host pointer
    uint64_t dst_off;   // byte offset into the destination base
    uint32_t len;
    uint32_t pad;
};

#define HIP_CHECK(expr)                                                              \
    do {                                                                             \
        hipError_t _e = (expr);                                                      \
        if (_e != hipSuccess) {                                                      \
            throw std::runtime_error(std::string("HIP error: ") +                    \
                                     hipGetErrorString(_e) + " at " __FILE__ ":" +   \
                                     std::to_string(__LINE__));                      \
        }                                                                            \
    } while (0)

}  // namespace edgehip
