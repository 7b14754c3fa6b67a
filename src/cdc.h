// G2: content-defined chunking — Gear rolling hash.
// Boundary where (hash & (avg-1)) == 0, subject to min <= len <= max
// (min = avg/4, max = 4*avg). The gear table is a fixed pseudo-random
// constant (seeded splitmix64) so boundaries are stable across versions —
// part of the v1 format contract (docs/FORMATS.md).
#pragma once

#include <cstddef>
#include <cstdint>
#include <vector>

#include "common.h"

namespace edgehip {

const uint64_t* gear_table();  // 256 entries

// Scan [data, data+len) and return chunk END offsets (ascending). `final_tail`
// = true also emits the trailing partial chunk; false leaves the tail for the
// next call (streaming). avg must be a power of two.
void cdc_boundaries(const uint8_t* data, size_t len, uint64_t avg, bool final_tail,
                    std::vector<size_t>* ends);

}  // namespace edgehip
