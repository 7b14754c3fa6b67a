// C API for the Python test/bench harness (ctypes). The product data path is
// the native CLI + library; Python is glue only.
#include <memory>
#include <new>

#include "blake3.h"
#include "conf.h"
#include "engine/engine.h"
#include "gf256.h"
#include "lz4.h"
#include "manifest.h"
#include "obj.h"
#include "obs.h"

using namespace edgehip;

extern "C" {

void eh_blake3(const void* data, size_t len, uint8_t* out32) { blake3_hash(data, len, out32); }

uint32_t eh_crc32c(const void* data, size_t len) { return crc32c(data, len); }

size_t eh_lz4_compress_block(const uint8_t* src, size_t n, uint8_t* dst, size_t cap) {
    return lz4_compress_block(src, n, dst, cap);
}
size_t eh_lz4_decompress_block(const uint8_t* src, size_t n, uint8_t* dst, size_t cap) {
    size_t r = lz4_decompress_block(src, n, dst, cap);
    return r == SIZE_MAX ? (size_t)-1 : r;
}
// framed chunk compress: returns framed size, 0 if incompressible, -1 on error
int64_t eh_lz4_chunk_compress(const uint8_t* src, size_t n, uint8_t* dst, size_t cap) {
    std::vector<uint8_t> out;
    if (!lz4_chunk_compress(src, n, &out)) return 0;
    if (out.size() > cap) return -1;
    std::memcpy(dst, out.data(), out.size());
    return (int64_t)out.size();
}
int eh_lz4_chunk_decompress(const uint8_t* src, size_t n, uint8_t* dst, size_t logical) {
    return (int)lz4_chunk_decompress(src, n, dst, logical);
}

int eh_rs_encode(int k, int m, const uint8_t* const* data, uint8_t* const* parity, size_t len) {
    rs_encode(k, m, data, parity, len);
    return 0;
}
int eh_rs_reconstruct(int k, int m, uint8_t* const* shards, const uint8_t* present, size_t len) {
    bool pres[64];
    for (int i = 0; i < k + m; i++) pres[i] = present[i] != 0;
    return (int)rs_reconstruct(k, m, shards, pres, len);
}
void eh_rs_parity_matrix(int k, int m, uint8_t* out) {
    auto M = rs_parity_matrix(k, m);
    std::memcpy(out, M.data(), M.size());
}

// ---- manifest ----
int64_t eh_manifest_roundtrip(const uint8_t* in, size_t len, uint8_t* out, size_t cap) {
    Manifest m;
    if (Manifest::decode(in, len, &m) != St::ok) return -1;
    std::vector<uint8_t> enc;
    m.encode(&enc);
    if (enc.size() > cap) return -2;
    std::memcpy(out, enc.data(), enc.size());
    return (int64_t)enc.size();
}

// ---- engine/object harness ----
struct EhEngine {
    Conf conf;
    std::unique_ptr<Engine> eng;
};

void* eh_engine_create(const char* opts) {
    auto* e = new (std::nothrow) EhEngine;
    if (!e) return nullptr;
    if (opts && *opts && e->conf.apply_opts(opts) != St::ok) {
        delete e;
        return nullptr;
    }
    try {
        e->eng = make_engine(e->conf);
    } catch (const std::exception& ex) {
        LOGE("eh_engine_create: %s", ex.what());
        delete e;
        return nullptr;
    }
    return e;
}
void eh_engine_destroy(void* e) { delete (EhEngine*)e; }
uint64_t eh_engine_stored_bytes(void* e) { return ((EhEngine*)e)->eng->stored_bytes(); }
uint64_t eh_engine_stored_chunks(void* e) { return ((EhEngine*)e)->eng->stored_chunks(); }
int eh_engine_is_gpu(void* e) { return ((EhEngine*)e)->eng->is_gpu() ? 1 : 0; }

void* eh_object_create(void* e) {
    auto* ee = (EhEngine*)e;
    return new Object(ee->eng.get(), &ee->conf);
}
void eh_object_destroy(void* o) { delete (Object*)o; }
int eh_object_write(void* o, uint64_t off, const void* data, uint64_t len) {
    return (int)((Object*)o)->write(off, (const uint8_t*)data, len);
}
int64_t eh_object_read(void* o, uint64_t off, void* out, uint64_t len) {
    uint64_t got = 0;
    St s = ((Object*)o)->read(off, len, (uint8_t*)out, &got);
    return s == St::ok ? (int64_t)got : (int64_t)st_errno(s);
}
int eh_object_flush(void* o) { return (int)((Object*)o)->flush(); }
int eh_object_truncate(void* o, uint64_t size) { return (int)((Object*)o)->truncate(size); }
uint64_t eh_object_size(void* o) { return ((Object*)o)->size(); }
int64_t eh_object_commit(void* o, uint8_t* manifest_out, size_t cap) {
    std::vector<uint8_t> bytes;
    St s = ((Object*)o)->commit(&bytes);
    if (s != St::ok) return (int64_t)st_errno(s);
    if (manifest_out) {
        if (bytes.size() > cap) return -1000;
        std::memcpy(manifest_out, bytes.data(), bytes.size());
    }
    return (int64_t)bytes.size();
}

}  // extern "C"

// batched scatter read: n ranges concatenated into out (sum(lens) bytes).
// One object-level pass per range but ranges run engine batches via the
// object's internal batching; used by the random-read bench (fio iodepth).
extern "C" int64_t eh_object_preadv(void* o, int n, const uint64_t* offs,
                                    const uint32_t* lens, void* out) {
    uint64_t got = 0;
    St s = ((Object*)o)->preadv(offs, lens, n, (uint8_t*)out, &got);
    if (s != St::ok) return (int64_t)st_errno(s);
    return (int64_t)got;
}

extern "C" int eh_object_punch_hole(void* o, uint64_t off, uint64_t len) {
    return (int)st_errno(((Object*)o)->punch_hole(off, len));
}

extern "C" int64_t eh_object_copy_range(void* dst, void* src, uint64_t soff,
                                        uint64_t doff, uint64_t len) {
    uint64_t copied = 0;
    St s = ((Object*)dst)->copy_range((Object*)src, soff, doff, len, &copied);
    if (s != St::ok) return (int64_t)st_errno(s);
    return (int64_t)copied;
}

extern "C" int eh_engine_drop_chunk(void* e, const uint8_t* fp32) {
    Fp f;
    std::memcpy(f.b.data(), fp32, 32);
    return (int)((EhEngine*)e)->eng->debug_drop_chunk(f);
}

// ---- multi-device topology (tests/multigpu + bench striped mode) ----
extern "C" int eh_engine_ndev(void* e) { return ((EhEngine*)e)->eng->ndev(); }
// >=0 device slot; -2 host/spill tier; -1 absent
extern "C" int eh_engine_chunk_device(void* e, const uint8_t* fp32) {
    Fp f;
    std::memcpy(f.b.data(), fp32, 32);
    return ((EhEngine*)e)->eng->chunk_device(f);
}
extern "C" int eh_engine_mark_device_failed(void* e, int dev) {
    return (int)((EhEngine*)e)->eng->mark_device_failed(dev);
}
extern "C" uint64_t eh_engine_manifest_records(void* e, int dev_slot, int total) {
    return ((EhEngine*)e)->eng->manifest_records(dev_slot, total != 0);
}

// global metrics counter by name (bench honesty: exports dedup_hits etc.)
extern "C" uint64_t eh_metric(const char* name) {
    Metrics& m = metrics();
    std::string n(name ? name : "");
    if (n == "dedup_hits") return m.dedup_hits.load();
    if (n == "dedup_bytes_saved") return m.dedup_bytes_saved.load();
    if (n == "chunks_put") return m.chunks_put.load();
    if (n == "chunks_get") return m.chunks_get.load();
    if (n == "cache_hits") return m.cache_hits.load();
    if (n == "cache_misses") return m.cache_misses.load();
    if (n == "spilled_chunks") return m.spilled_chunks.load();
    if (n == "spill_bytes") return m.spill_bytes.load();
    if (n == "rs_encoded_stripes") return m.rs_encoded_stripes.load();
    if (n == "rs_reconstructs") return m.rs_reconstructs.load();
    if (n == "h2d_bytes") return m.h2d_bytes.load();
    if (n == "d2h_bytes") return m.d2h_bytes.load();
    if (n == "lz4_in_bytes") return m.lz4_in_bytes.load();
    if (n == "lz4_out_bytes") return m.lz4_out_bytes.load();
    if (n == "journal_resyncs") return m.journal_resyncs.load();
    return (uint64_t)-1;
}
