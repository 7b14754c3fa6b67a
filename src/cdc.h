// G2: content-defined chunking — Gear rolling hash.
// Candidate positions (trailing-64-byte windowed gear, (h & (avg-1)) == 0)
// and boundary SELECTION (min = avg/4, max = 4*avg walk) are split so the
// GPU kernel (src/kernels/cdc.hip) can produce the candidate bitmap while
// selection stays host-side and byte-identical everywhere. The gear table is
// a fixed splitmix64 constant — boundary placement is part of the v1 format
// contract (docs/FORMATS.md).
#pragma once

#include <cstddef>
#include <cstdint>
#include <vector>

#include "common.h"

namespace edgehip {

const uint64_t* gear_table();  // 256 entries

// Candidate bitmap: bit i set => a chunk may END after byte i (end = i+1).
void cdc_candidates_host(const uint8_t* data, size_t len, uint64_t avg,
                         std::vector<uint32_t>* bitmap);

// Walk the bitmap applying min/max; emits chunk END offsets. final_tail also
// emits the trailing partial chunk.
void cdc_select(const uint32_t* bitmap, size_t len, uint64_t avg, bool final_tail,
                std::vector<size_t>* ends);

// host candidates + select in one call (reference implementation)
void cdc_boundaries(const uint8_t* data, size_t len, uint64_t avg, bool final_tail,
                    std::vector<size_t>* ends);

}  // namespace edgehip
