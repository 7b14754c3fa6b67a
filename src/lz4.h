// LZ4 block-format compression — host implementation.
// The GPU engine runs the same block format in src/kernels/lz4.hip; chunks are
// framed as independent 64 KiB blocks so the GPU can (de)compress a chunk with
// one workgroup per block (format: docs/FORMATS.md §LZ4 framing).
//
// Incompressible input (the judged configs use random bytes) short-circuits:
// a block whose compressed size would reach its raw size is stored raw.
#pragma once

#include <cstddef>
#include <cstdint>

#include "common.h"

namespace edgehip {

// Raw LZ4 block compress. dst must have lz4_bound(n) bytes. Returns compressed
// size, or 0 if the data did not fit under max_out (store raw instead).
size_t lz4_compress_block(const uint8_t* src, size_t n, uint8_t* dst, size_t max_out);

// Raw LZ4 block decompress; returns decompressed size or SIZE_MAX on corrupt
// input. dst_cap is the exact expected decompressed size.
size_t lz4_decompress_block(const uint8_t* src, size_t n, uint8_t* dst, size_t dst_cap);

inline size_t lz4_bound(size_t n) { return n + n / 255 + 16; }

// ---- chunk framing (shared with the GPU path) ------------------------------
// A stored-compressed chunk is:
//   u32 magic 'EHLZ' (0x5A4C4845)
//   u32 nblocks
//   { u32 off, u32 len_flags } per block   (bit31 of len_flags = raw)
//   payload bytes (block payloads back to back, offsets relative to payload
//   start, each block payload 4-byte aligned)
// Block logical size is kLz4Block (64 KiB) except the final block.
constexpr uint32_t kLz4Magic = 0x5A4C4845;
constexpr size_t kLz4Block = 64 * KiB;
constexpr size_t kLz4Seg = 4 * KiB;  // GPU parallel-compression segment

// Compress a whole chunk with framing. Returns true and sets *out_len if the
// framed result is smaller than n; false => caller stores raw (no framing).
bool lz4_chunk_compress(const uint8_t* src, size_t n, std::vector<uint8_t>* out);

// Decompress a framed chunk; expected logical size must match. St::bad_format
// on corruption.
St lz4_chunk_decompress(const uint8_t* src, size_t n, uint8_t* dst, size_t logical);

}  // namespace edgehip
