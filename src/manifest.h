// Chunk manifests — the on-wire format.
//
// The reference's manifest bytes were unreadable at survey time; this is the
// v1 format, frozen by golden tests (tests/unit/test_manifest.py golden bytes).
// Layout (all little-endian), defined in docs/FORMATS.md:
//
//   header (64 B):
//     u32 magic        = 'EHM1' (0x314D4845)
//     u16 version      = 1
//     u8  fp_algo      (FpAlgo)
//     u8  compress     (Compress)
//     u8  oid[32]      object id (content hash of first manifest version, or uuid)
//     u64 nchunks
//     u64 logical_size object byte length
//     u64 chunk_size   nominal chunk size (last chunk may be short)
//   chunk refs (64 B each):
//     u8  fp[32]       content fingerprint of the UNCOMPRESSED chunk (§9 Q6)
//     u32 logical_len  uncompressed length
//     u32 stored_len   stored (possibly compressed) length
//     u32 flags        bit0 = lz4-compressed; bit1 = rs-striped
//     u32 crc32c       of the stored bytes (verify-lite on read)
//     u64 shard_map    RS: 8 × u8 ACTUAL device slot per shard (k data then
//                      m parity; byte j = shard j's device; 0xFE = host/spill
//                      tier, 0xFF = hole/unknown). Recorded at parity-encode
//                      time from the store's placement (obj.cc
//                      rs_update_locked), identical across a stripe's refs.
//     u64 reserved     = 0
//   trailer (4 B):
//     u32 crc32c over header + refs
#pragma once

#include "common.h"

namespace edgehip {

struct ChunkRef {
    Fp fp;
    uint32_t logical_len = 0;
    uint32_t stored_len = 0;
    uint32_t flags = 0;
    uint32_t crc = 0;
    uint64_t shard_map = 0;

    static constexpr uint32_t kFlagLz4 = 1u << 0;
    static constexpr uint32_t kFlagRs = 1u << 1;      // member of an RS stripe
    static constexpr uint32_t kFlagParity = 1u << 2;  // parity chunk (not logical data)
    bool compressed() const { return flags & kFlagLz4; }
    bool parity() const { return flags & kFlagParity; }
};

struct Manifest {
    static constexpr uint32_t kMagic = 0x314D4845;  // 'EHM1'
    static constexpr uint16_t kVersion = 1;
    static constexpr size_t kHeaderSize = 64;
    static constexpr size_t kRefSize = 64;

    Fp oid;
    uint8_t fp_algo = 1;   // FpAlgo::blake3
    uint8_t compress = 0;
    uint64_t logical_size = 0;
    uint64_t chunk_size = 1 * MiB;
    std::vector<ChunkRef> chunks;   // data chunks, index = offset / chunk_size
    std::vector<ChunkRef> parity;   // RS parity (flag kFlagParity), appended on wire

    size_t encoded_size() const {
        return kHeaderSize + (chunks.size() + parity.size()) * kRefSize + 4;
    }
    void encode(std::vector<uint8_t>* out) const;
    static St decode(const uint8_t* data, size_t len, Manifest* out);
};

}  // namespace edgehip
