// HostEngine — CPU data plane (judged config 1 + golden path for tests).
#include "engine.h"

#include <atomic>

#include "../blake3.h"
#include "../lz4.h"
#include "../obs.h"
#include "../store.h"

namespace edgehip {

namespace {

class HostEngine : public Engine {
  public:
    explicit HostEngine(const Conf& conf) : conf_(conf), store_(0) {
        if (conf.prealloc) BufPool::inst().prealloc(conf.prealloc, (uint32_t)conf.chunk_size);
    }

    St put_chunks(const PutChunk* put, ChunkRef* refs, int n) override {
        for (int i = 0; i < n; i++) {
            const PutChunk& p = put[i];
            ChunkRef& r = refs[i];
            r = ChunkRef{};
            r.logical_len = p.len;
            if (conf_.fp == FpAlgo::blake3) {
                r.fp = blake3_fp(p.data, p.len);
            } else {
                // fp=none: synthesized unique id (no content addressing, no dedup)
                static std::atomic<uint64_t> seq{1};
                uint64_t s = seq.fetch_add(1);
                std::memcpy(r.fp.b.data(), &s, 8);
                r.fp.b[31] = 0xFE;  // mark non-content id
            }
            if (conf_.dedup && conf_.fp != FpAlgo::none && store_.has(r.fp)) {
                if (store_.addref(r.fp) == St::ok) {
                    metrics().dedup_hits++;
                    metrics().dedup_bytes_saved += p.len;
                    r.stored_len = store_.stored_len(r.fp);
                    // flags of the stored copy decide compression; re-derive:
                    r.flags = (r.stored_len != p.len) ? ChunkRef::kFlagLz4 : 0;
                    if (p.donate) p.donate->release();
                    continue;
                }
            }
            std::vector<uint8_t> comp;
            bool compressed = conf_.compress == Compress::lz4 && p.len >= 4096 &&
                              lz4_chunk_compress(p.data, p.len, &comp);
            metrics().lz4_in_bytes += p.len;
            St s;
            if (compressed) {
                r.flags |= ChunkRef::kFlagLz4;
                r.stored_len = (uint32_t)comp.size();
                if (conf_.crc) r.crc = chunk_crc(comp.data(), comp.size());
                s = store_.put_copy(r.fp, comp.data(), (uint32_t)comp.size());
                if (p.donate) p.donate->release();
            } else {
                r.stored_len = p.len;
                if (conf_.crc) r.crc = chunk_crc(p.data, p.len);
                if (p.donate && p.donate->p) {
                    p.donate->len = p.len;
                    s = store_.put(r.fp, std::move(*p.donate));  // zero-copy ingest
                } else {
                    s = store_.put_copy(r.fp, p.data, p.len);
                }
            }
            metrics().lz4_out_bytes += r.stored_len;
            if (s != St::ok && s != St::exists) return s;
            metrics().chunks_put++;
        }
        return St::ok;
    }

    St get_chunks(const GetChunk* get, int n) override {
        std::vector<uint8_t> tmp;
        for (int i = 0; i < n; i++) {
            const GetChunk& g = get[i];
            const ChunkRef& r = *g.ref;
            if ((uint64_t)g.off + g.len > r.logical_len) return St::inval;
            metrics().chunks_get++;
            if (!r.compressed()) {
                St s = store_.get(r.fp, g.out, g.off, g.len);
                if (s != St::ok) return s;
            } else {
                // decompress only the 64 KiB blocks covering [off, off+len)
                tmp.resize(r.stored_len);
                St s = store_.get(r.fp, tmp.data(), 0, r.stored_len);
                if (s != St::ok) return s;
                if (g.off == 0 && g.len == r.logical_len) {
                    s = lz4_chunk_decompress(tmp.data(), tmp.size(), g.out, r.logical_len);
                    if (s != St::ok) return s;
                } else {
                    std::vector<uint8_t> full(r.logical_len);
                    s = lz4_chunk_decompress(tmp.data(), tmp.size(), full.data(), r.logical_len);
                    if (s != St::ok) return s;
                    std::memcpy(g.out, full.data() + g.off, g.len);
                }
            }
        }
        return St::ok;
    }

    St release_chunk(const ChunkRef& ref) override { return store_.decref(ref.fp); }

    uint64_t stored_bytes() const override { return store_.bytes(); }
    uint64_t stored_chunks() const override { return store_.count(); }
    bool is_gpu() const override { return false; }

  private:
    Conf conf_;
    HostStore store_;
};

}  // namespace

std::unique_ptr<Engine> make_host_engine(const Conf& conf) {
    return std::make_unique<HostEngine>(conf);
}

std::unique_ptr<Engine> make_engine(const Conf& conf) {
    if (conf.engine == EngineKind::gpu) return make_gpu_engine(conf);
    return make_host_engine(conf);
}

}  // namespace edgehip
