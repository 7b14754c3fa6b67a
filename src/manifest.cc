#include "manifest.h"

namespace edgehip {

static void put_u16(std::vector<uint8_t>* o, uint16_t v) { o->push_back(v & 0xff); o->push_back(v >> 8); }
static void put_u32(std::vector<uint8_t>* o, uint32_t v) { for (int i = 0; i < 4; i++) o->push_back((v >> (8 * i)) & 0xff); }
static void put_u64(std::vector<uint8_t>* o, uint64_t v) { for (int i = 0; i < 8; i++) o->push_back((v >> (8 * i)) & 0xff); }
static uint16_t get_u16(const uint8_t* p) { return (uint16_t)(p[0] | p[1] << 8); }
static uint32_t get_u32(const uint8_t* p) { return p[0] | p[1] << 8 | p[2] << 16 | (uint32_t)p[3] << 24; }
static uint64_t get_u64(const uint8_t* p) { return get_u32(p) | (uint64_t)get_u32(p + 4) << 32; }

void Manifest::encode(std::vector<uint8_t>* out) const {
    out->clear();
    out->reserve(encoded_size());
    put_u32(out, kMagic);
    put_u16(out, kVersion);
    out->push_back(fp_algo);
    out->push_back(compress);
    out->insert(out->end(), oid.b.begin(), oid.b.end());
    put_u64(out, chunks.size() + parity.size());
    put_u64(out, logical_size);
    put_u64(out, chunk_size);
    auto put_ref = [&](const ChunkRef& c) {
        out->insert(out->end(), c.fp.b.begin(), c.fp.b.end());
        put_u32(out, c.logical_len);
        put_u32(out, c.stored_len);
        put_u32(out, c.flags);
        put_u32(out, c.crc);
        put_u64(out, c.shard_map);
        put_u64(out, 0);  // reserved
    };
    for (const auto& c : chunks) put_ref(c);
    for (const auto& c : parity) put_ref(c);
    uint32_t crc = crc32c(out->data(), out->size());
    put_u32(out, crc);
}

St Manifest::decode(const uint8_t* data, size_t len, Manifest* out) {
    if (len < kHeaderSize + 4) return St::bad_format;
    if (get_u32(data) != kMagic) return St::bad_format;
    if (get_u16(data + 4) != kVersion) return St::bad_format;
    out->fp_algo = data[6];
    out->compress = data[7];
    std::memcpy(out->oid.b.data(), data + 8, 32);
    uint64_t nchunks = get_u64(data + 40);
    out->logical_size = get_u64(data + 48);
    out->chunk_size = get_u64(data + 56);
    size_t need = kHeaderSize + nchunks * kRefSize + 4;
    if (len != need) return St::bad_format;
    if (crc32c(data, len - 4) != get_u32(data + len - 4)) return St::bad_format;
    out->chunks.clear();
    out->parity.clear();
    const uint8_t* p = data + kHeaderSize;
    for (uint64_t i = 0; i < nchunks; i++, p += kRefSize) {
        ChunkRef c;
        std::memcpy(c.fp.b.data(), p, 32);
        c.logical_len = get_u32(p + 32);
        c.stored_len = get_u32(p + 36);
        c.flags = get_u32(p + 40);
        c.crc = get_u32(p + 44);
        c.shard_map = get_u64(p + 48);
        if (c.parity()) {
            out->parity.push_back(c);
        } else {
            if (!out->parity.empty()) return St::bad_format;  // parity must trail
            out->chunks.push_back(c);
        }
    }
    return St::ok;
}

}  // namespace edgehip
