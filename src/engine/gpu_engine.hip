// GpuEngine — the MI355X data plane.
//
// Chunks live in a per-GPU HBM arena (the chunk store IS device memory —
// 288 GB HBM3E per GPU); ingest pipelines H2D staging, BLAKE3 fingerprints,
// LZ4 compression and crc32c on the GPU (src/kernels/*.hip), then batch-copies
// stored bytes into arena slots. Reads decompress on-device and stage D2H
// through pinned buffers. Cold chunks spill to a host RAM tier when the arena
// fills (config 4's 1 TiB object exceeds one GPU's HBM).
//
// Fails LOUDLY when no GPU is present — no silent CPU fallback on the GPU
// path (a GPU test passing on eager fallback is a failure, not a feature).
#include <algorithm>
#include <list>
#include <mutex>
#include <stdexcept>
#include <unordered_map>
#include <vector>

#include "../kernels/launchers.h"
#include "../journal.h"
#include "../lz4.h"
#include "../obs.h"
#include "../store.h"
#include "engine.h"

namespace edgehip {

namespace {

constexpr int kMaxBatch = 32;            // chunks per pipeline pass
constexpr uint64_t kGranule = 64 * KiB;  // arena allocation granule
constexpr uint32_t kSegSz = 4096;        // LZ4 segment (lz4.hip kSeg)
constexpr uint32_t kSegStride = 4352;
constexpr uint32_t kMaxSegs = kMaxBatch * 256;  // 1 MiB chunks -> 256 segs each

// pinned metadata buffer layout (256 KiB)
constexpr uint64_t kOffChunks = 0;                 // DevChunk[kMaxBatch]
constexpr uint64_t kOffChunks2 = 4 * KiB;          // DevChunk[kMaxBatch]
constexpr uint64_t kOffSegMeta = 8 * KiB;          // ulong2[kMaxSegs] = 128 KiB
constexpr uint64_t kOffFps = 140 * KiB;            // 32 B * kMaxBatch
constexpr uint64_t kOffSizes = 144 * KiB;          // u32 * kMaxBatch
constexpr uint64_t kOffCrcs = 148 * KiB;           // u32 * kMaxBatch
constexpr uint64_t kOffCopy = 152 * KiB;           // DevChunk[kMaxBatch]
constexpr uint64_t kOffStatus = 156 * KiB;         // u32 * kMaxBatch
constexpr uint64_t kPinMetaSz = 256 * KiB;

struct ArenaLoc {
    uint64_t off = 0;
    uint32_t len = 0;  // stored bytes
    uint32_t crc = 0;
    uint32_t refs = 1;
    bool spilled = false;
    std::list<Fp>::iterator lru;  // valid when !spilled
};

class GpuEngine : public Engine {
  public:
    explicit GpuEngine(const Conf& conf) : conf_(conf), spill_(conf.spill_bytes) {
        if (conf.chunk_size > 1 * MiB)
            throw std::runtime_error("GpuEngine: chunk_size > 1 MiB not supported");
        int ndev = 0;
        hipError_t e = hipGetDeviceCount(&ndev);
        if (e != hipSuccess || ndev <= conf.gpu_first)
            throw std::runtime_error("GpuEngine: no MI355X visible (hipGetDeviceCount: " +
                                     std::string(hipGetErrorString(e)) + ")");
        dev_ = conf.gpu_first;
        HIP_CHECK(hipSetDevice(dev_));
        HIP_CHECK(hipStreamCreateWithFlags(&s_pipe_, hipStreamNonBlocking));
        HIP_CHECK(hipStreamCreateWithFlags(&s_read_, hipStreamNonBlocking));

        arena_sz_ = conf.arena_bytes;
        HIP_CHECK(hipMalloc(&arena_, arena_sz_));

        const uint64_t cs = conf.chunk_size;
        stage_sz_ = (uint64_t)kMaxBatch * cs;
        HIP_CHECK(hipMalloc(&d_stage_, stage_sz_));
        HIP_CHECK(hipMalloc(&d_comp_, stage_sz_));
        HIP_CHECK(hipMalloc(&d_rdstage_, stage_sz_));
        HIP_CHECK(hipMalloc(&d_segscratch_, (uint64_t)kMaxSegs * kSegStride));
        HIP_CHECK(hipMalloc(&d_chunks_, sizeof(DevChunk) * kMaxBatch * 2));
        HIP_CHECK(hipMalloc(&d_segmeta_, sizeof(ulong2) * kMaxSegs));
        HIP_CHECK(hipMalloc(&d_segsizes_, sizeof(uint32_t) * kMaxSegs));
        HIP_CHECK(hipMalloc(&d_sizes_, sizeof(uint32_t) * kMaxBatch));
        HIP_CHECK(hipMalloc(&d_fps_, 32 * kMaxBatch));
        HIP_CHECK(hipMalloc(&d_crcs_, sizeof(uint32_t) * kMaxBatch));
        HIP_CHECK(hipMalloc(&d_copy_, sizeof(DevChunk) * kMaxBatch));

        HIP_CHECK(hipHostMalloc(&pin_in_, stage_sz_, hipHostMallocDefault));
        HIP_CHECK(hipHostMalloc(&pin_out_, stage_sz_, hipHostMallocDefault));
        HIP_CHECK(hipHostMalloc(&pin_meta_, kPinMetaSz, hipHostMallocDefault));

        LOGI("GpuEngine: dev %d arena %.1f GiB, staging %.0f MiB x3, spill cap %.1f GiB",
             dev_, arena_sz_ / (double)GiB, stage_sz_ / (double)MiB,
             conf.spill_bytes / (double)GiB);
    }

    ~GpuEngine() override {
        (void)hipSetDevice(dev_);
        (void)hipDeviceSynchronize();
        for (void* p : {(void*)arena_, (void*)d_stage_, (void*)d_comp_, (void*)d_rdstage_,
                        (void*)d_segscratch_, (void*)d_chunks_, (void*)d_segmeta_,
                        (void*)d_segsizes_, (void*)d_sizes_, (void*)d_fps_, (void*)d_crcs_,
                        (void*)d_copy_})
            (void)hipFree(p);
        (void)hipHostFree(pin_in_);
        (void)hipHostFree(pin_out_);
        (void)hipHostFree(pin_meta_);
        (void)hipStreamDestroy(s_pipe_);
        (void)hipStreamDestroy(s_read_);
    }

    St put_chunks(const PutChunk* put, ChunkRef* refs, int n) override {
        for (int i = 0; i < n; i += kMaxBatch) {
            St s = flush_batch(put + i, refs + i, std::min(kMaxBatch, n - i));
            if (s != St::ok) return s;
        }
        return St::ok;
    }

    St get_chunks(const GetChunk* get, int n) override;
    St release_chunk(const ChunkRef& ref) override;

    St debug_drop_chunk(const Fp& fp) override {
        std::lock_guard<std::mutex> g(map_mu_);
        auto it = map_.find(fp);
        if (it == map_.end()) return spill_.remove(fp);
        if (it->second.spilled) {
            spill_.remove(fp);
        } else {
            arena_free(it->second.off, it->second.len);
            lru_.erase(it->second.lru);
        }
        map_.erase(it);
        return St::ok;
    }

    uint64_t stored_bytes() const override { return arena_used_.load() + spill_.bytes(); }
    uint64_t stored_chunks() const override {
        std::lock_guard<std::mutex> g(map_mu_);
        return map_.size();
    }
    bool is_gpu() const override { return true; }

  private:
    int size_class(uint32_t bytes) const {
        return (int)((bytes + kGranule - 1) / kGranule) - 1;
    }

    bool arena_alloc(uint32_t bytes, uint64_t* off) {
        int cls = size_class(bytes);
        std::lock_guard<std::mutex> g(alloc_mu_);
        if (cls < (int)free_.size() && !free_[cls].empty()) {
            *off = free_[cls].back();
            free_[cls].pop_back();
            arena_used_ += (uint64_t)(cls + 1) * kGranule;
            return true;
        }
        uint64_t need = (uint64_t)(cls + 1) * kGranule;
        if (bump_ + need <= arena_sz_) {
            *off = bump_;
            bump_ += need;
            arena_used_ += need;
            return true;
        }
        return false;
    }

    void arena_free(uint64_t off, uint32_t bytes) {
        int cls = size_class(bytes);
        std::lock_guard<std::mutex> g(alloc_mu_);
        if ((int)free_.size() <= cls) free_.resize(cls + 1);
        free_[cls].push_back(off);
        arena_used_ -= (uint64_t)(cls + 1) * kGranule;
    }

    St evict_for(uint32_t need_bytes);  // map_mu_ held
    St flush_batch(const PutChunk* put, ChunkRef* refs, int n);

    Conf conf_;
    int dev_ = 0;
    hipStream_t s_pipe_{}, s_read_{};

    uint8_t* arena_ = nullptr;
    uint64_t arena_sz_ = 0;
    std::mutex alloc_mu_;
    std::vector<std::vector<uint64_t>> free_;
    uint64_t bump_ = 0;
    std::atomic<uint64_t> arena_used_{0};

    uint8_t *d_stage_ = nullptr, *d_comp_ = nullptr, *d_rdstage_ = nullptr,
            *d_segscratch_ = nullptr, *d_fps_ = nullptr;
    DevChunk *d_chunks_ = nullptr, *d_copy_ = nullptr;
    ulong2* d_segmeta_ = nullptr;
    uint32_t *d_segsizes_ = nullptr, *d_sizes_ = nullptr, *d_crcs_ = nullptr;
    uint64_t stage_sz_ = 0;

    uint8_t *pin_in_ = nullptr, *pin_out_ = nullptr, *pin_meta_ = nullptr;

    mutable std::mutex map_mu_;
    std::unordered_map<Fp, ArenaLoc, FpHash> map_;
    std::list<Fp> lru_;  // front = hottest

    std::mutex pipe_mu_;  // staging buffers + streams

    HostStore spill_;
};

St GpuEngine::flush_batch(const PutChunk* put, ChunkRef* refs, int n) {
    const uint32_t cs = (uint32_t)conf_.chunk_size;
    const bool do_lz4 = conf_.compress == Compress::lz4;
    const bool do_fp = conf_.fp == FpAlgo::blake3;

    std::lock_guard<std::mutex> pg(pipe_mu_);

    auto* h_chunks = (DevChunk*)(pin_meta_ + kOffChunks);
    auto* h_chunks2 = (DevChunk*)(pin_meta_ + kOffChunks2);
    auto* h_segmeta = (ulong2*)(pin_meta_ + kOffSegMeta);
    int nsegs = 0;
    uint64_t off = 0;
    for (int i = 0; i < n; i++) {
        std::memcpy(pin_in_ + off, put[i].data, put[i].len);
        h_chunks[i] = DevChunk{off, off, put[i].len, (uint32_t)nsegs};
        h_chunks2[i] = DevChunk{off, (uint64_t)i * cs, put[i].len, (uint32_t)nsegs};
        uint32_t ns = (put[i].len + kSegSz - 1) / kSegSz;
        for (uint32_t s = 0; s < ns; s++)
            h_segmeta[nsegs + s] =
                ulong2{off + (uint64_t)s * kSegSz, std::min(kSegSz, put[i].len - s * kSegSz)};
        nsegs += ns;
        off += round_up(put[i].len, 64);
    }
    HIP_CHECK(hipMemcpyAsync(d_stage_, pin_in_, off, hipMemcpyHostToDevice, s_pipe_));
    HIP_CHECK(hipMemcpyAsync(d_chunks_, h_chunks, sizeof(DevChunk) * n, hipMemcpyHostToDevice,
                             s_pipe_));
    metrics().h2d_bytes += off;

    if (do_fp) launch_blake3_chunks(d_stage_, d_chunks_, d_fps_, n, s_pipe_);
    if (do_lz4) {
        HIP_CHECK(hipMemcpyAsync(d_segmeta_, h_segmeta, sizeof(ulong2) * nsegs,
                                 hipMemcpyHostToDevice, s_pipe_));
        launch_lz4_compress_segs(d_stage_, d_segmeta_, d_segscratch_, d_segsizes_, nsegs,
                                 s_pipe_);
        HIP_CHECK(hipMemcpyAsync(d_chunks_ + kMaxBatch, h_chunks2, sizeof(DevChunk) * n,
                                 hipMemcpyHostToDevice, s_pipe_));
        launch_lz4_emit(d_stage_, d_segscratch_, d_segsizes_, d_chunks_ + kMaxBatch, d_comp_,
                        d_sizes_, n, s_pipe_);
    }

    auto* h_fps = (uint8_t*)(pin_meta_ + kOffFps);
    auto* h_sizes = (uint32_t*)(pin_meta_ + kOffSizes);
    auto* h_crcs = (uint32_t*)(pin_meta_ + kOffCrcs);
    if (do_fp)
        HIP_CHECK(hipMemcpyAsync(h_fps, d_fps_, 32 * n, hipMemcpyDeviceToHost, s_pipe_));
    if (do_lz4)
        HIP_CHECK(hipMemcpyAsync(h_sizes, d_sizes_, 4 * n, hipMemcpyDeviceToHost, s_pipe_));
    HIP_CHECK(hipStreamSynchronize(s_pipe_));

    struct Pend {
        int i;
        uint64_t src_off;
        bool from_comp;
        uint32_t stored;
        uint64_t arena_off;
    };
    std::vector<Pend> pend;
    for (int i = 0; i < n; i++) {
        ChunkRef& r = refs[i];
        r = ChunkRef{};
        r.logical_len = put[i].len;
        if (do_fp) {
            std::memcpy(r.fp.b.data(), h_fps + 32 * i, 32);
        } else {
            static std::atomic<uint64_t> seq{1};
            uint64_t s = seq.fetch_add(1);
            std::memcpy(r.fp.b.data(), &s, 8);
            r.fp.b[31] = 0xFE;
        }
        uint32_t framed = do_lz4 ? h_sizes[i] : 0;
        bool compressed = framed != 0 && framed < put[i].len;
        uint32_t stored = compressed ? framed : put[i].len;
        if (compressed) r.flags |= ChunkRef::kFlagLz4;
        r.stored_len = stored;
        metrics().lz4_in_bytes += put[i].len;
        metrics().lz4_out_bytes += stored;

        bool dup = false;
        {
            std::lock_guard<std::mutex> g(map_mu_);
            if (conf_.dedup && do_fp) {
                auto it = map_.find(r.fp);
                if (it != map_.end()) {
                    it->second.refs++;
                    r.stored_len = it->second.len;
                    r.crc = it->second.crc;
                    r.flags = (it->second.len != r.logical_len) ? ChunkRef::kFlagLz4 : 0;
                    metrics().dedup_hits++;
                    metrics().dedup_bytes_saved += put[i].len;
                    dup = true;
                }
            }
            if (!dup) {
                uint64_t aoff;
                if (!arena_alloc(stored, &aoff)) {
                    St s = evict_for(stored);
                    if (s != St::ok) return s;
                    if (!arena_alloc(stored, &aoff)) return St::no_space;
                }
                ArenaLoc loc;
                loc.off = aoff;
                loc.len = stored;
                lru_.push_front(r.fp);
                loc.lru = lru_.begin();
                map_.emplace(r.fp, loc);
                pend.push_back({i, compressed ? (uint64_t)i * cs : h_chunks[i].src_off,
                                compressed, stored, aoff});
                metrics().chunks_put++;
            }
        }
        if (!dup && conf_.journal_ptr)
            conf_.journal_ptr->put_chunk(r.fp, put[i].data, put[i].len);
        if (put[i].donate) put[i].donate->release();
    }

    if (!pend.empty()) {
        auto* h_copy = (DevChunk*)(pin_meta_ + kOffCopy);
        for (int pass = 0; pass < 2; pass++) {
            int m = 0;
            for (auto& p : pend) {
                if ((int)p.from_comp != pass) continue;
                h_copy[m++] = DevChunk{p.src_off, p.arena_off, p.stored, 0};
            }
            if (!m) continue;
            HIP_CHECK(hipMemcpyAsync(d_copy_, h_copy, sizeof(DevChunk) * m,
                                     hipMemcpyHostToDevice, s_pipe_));
            launch_batch_copy(pass ? d_comp_ : d_stage_, arena_, d_copy_, m, s_pipe_);
            HIP_CHECK(hipStreamSynchronize(s_pipe_));
        }
        if (conf_.crc) {
            for (size_t j = 0; j < pend.size(); j++)
                h_copy[j] = DevChunk{pend[j].arena_off, 0, pend[j].stored, 0};
            HIP_CHECK(hipMemcpyAsync(d_copy_, h_copy, sizeof(DevChunk) * pend.size(),
                                     hipMemcpyHostToDevice, s_pipe_));
            launch_crc32c_chunks(arena_, d_copy_, d_crcs_, (int)pend.size(), s_pipe_);
            HIP_CHECK(hipMemcpyAsync(h_crcs, d_crcs_, 4 * pend.size(), hipMemcpyDeviceToHost,
                                     s_pipe_));
            HIP_CHECK(hipStreamSynchronize(s_pipe_));
            std::lock_guard<std::mutex> g(map_mu_);
            for (size_t j = 0; j < pend.size(); j++) {
                refs[pend[j].i].crc = h_crcs[j];
                auto it = map_.find(refs[pend[j].i].fp);
                if (it != map_.end()) it->second.crc = h_crcs[j];
            }
        }
    }
    return St::ok;
}

St GpuEngine::evict_for(uint32_t need_bytes) {
    if (conf_.spill_bytes == 0) return St::no_space;
    uint64_t freed = 0;
    while (freed < (uint64_t)need_bytes * 4 && !lru_.empty()) {
        Fp victim = lru_.back();
        auto it = map_.find(victim);
        if (it == map_.end() || it->second.spilled) {
            lru_.pop_back();
            continue;
        }
        ArenaLoc& loc = it->second;
        Blob b = BufPool::inst().get(loc.len);
        b.len = loc.len;
        HIP_CHECK(hipMemcpyAsync(b.p, arena_ + loc.off, loc.len, hipMemcpyDeviceToHost,
                                 s_read_));
        HIP_CHECK(hipStreamSynchronize(s_read_));
        St s = spill_.put(victim, std::move(b));
        if (s != St::ok && s != St::exists) return s;
        arena_free(loc.off, loc.len);
        freed += round_up(loc.len, kGranule);
        loc.spilled = true;
        lru_.pop_back();
        metrics().spilled_chunks++;
        metrics().spill_bytes += loc.len;
    }
    return freed ? St::ok : St::no_space;
}

St GpuEngine::get_chunks(const GetChunk* get, int n) {
    const uint32_t cs = (uint32_t)conf_.chunk_size;
    struct DevReq {
        const GetChunk* g;
        uint64_t arena_off;
        uint32_t stored;
        bool compressed;
    };
    std::vector<DevReq> dev;
    dev.reserve(n);
    for (int i = 0; i < n; i++) {
        const GetChunk& g = get[i];
        const ChunkRef& r = *g.ref;
        if ((uint64_t)g.off + g.len > r.logical_len) return St::inval;
        metrics().chunks_get++;
        bool spilled = false;
        uint64_t aoff = 0;
        uint32_t stored = 0;
        bool found;
        {
            std::lock_guard<std::mutex> lk(map_mu_);
            found = map_.count(r.fp) != 0;
        }
        if (!found && conf_.journal_ptr) {
            // remount: re-ingest the chunk's logical bytes from the journal
            std::vector<uint8_t> bytes;
            if (conf_.journal_ptr->get_chunk(r.fp, &bytes) == St::ok &&
                bytes.size() == r.logical_len) {
                PutChunk p{bytes.data(), (uint32_t)bytes.size(), nullptr};
                ChunkRef nr;
                (void)put_chunks(&p, &nr, 1);
            }
        }
        {
            std::lock_guard<std::mutex> lk(map_mu_);
            auto it = map_.find(r.fp);
            if (it == map_.end()) return St::not_found;
            ArenaLoc& loc = it->second;
            spilled = loc.spilled;
            aoff = loc.off;
            stored = loc.len;
            if (!spilled) {
                metrics().cache_hits++;
                lru_.erase(loc.lru);
                lru_.push_front(r.fp);
                loc.lru = lru_.begin();
            } else {
                metrics().cache_misses++;
            }
        }
        if (spilled) {
            if (!r.compressed()) {
                St s = spill_.get(r.fp, g.out, g.off, g.len);
                if (s != St::ok) return s;
            } else {
                std::vector<uint8_t> tmp(stored), full(r.logical_len);
                St s = spill_.get(r.fp, tmp.data(), 0, stored);
                if (s != St::ok) return s;
                s = lz4_chunk_decompress(tmp.data(), tmp.size(), full.data(), r.logical_len);
                if (s != St::ok) return s;
                std::memcpy(g.out, full.data() + g.off, g.len);
            }
        } else {
            dev.push_back({&g, aoff, stored, r.compressed()});
        }
    }
    if (dev.empty()) return St::ok;

    std::lock_guard<std::mutex> pg(pipe_mu_);
    auto* h_dc = (DevChunk*)(pin_meta_ + kOffChunks);
    auto* h_status = (uint32_t*)(pin_meta_ + kOffStatus);

    // sub-batches: <= kMaxBatch compressed chunks (d_rdstage_ slots) and
    // pinned-out bytes <= stage_sz_
    size_t j0 = 0;
    while (j0 < dev.size()) {
        int m = 0;  // compressed chunks in this sub-batch
        uint64_t outb = 0;
        size_t j1 = j0;
        while (j1 < dev.size()) {
            const DevReq& q = dev[j1];
            int nm = m + (q.compressed ? 1 : 0);
            uint64_t no = outb + round_up(q.g->len, 64);
            if (nm > kMaxBatch || no > stage_sz_) break;
            m = nm;
            outb = no;
            j1++;
        }
        // decompress pass
        int mi = 0;
        for (size_t j = j0; j < j1; j++) {
            if (!dev[j].compressed) continue;
            h_dc[mi] = DevChunk{dev[j].arena_off, (uint64_t)mi * cs, dev[j].g->ref->logical_len,
                                dev[j].stored};
            mi++;
        }
        if (mi) {
            HIP_CHECK(hipMemcpyAsync(d_chunks_, h_dc, sizeof(DevChunk) * mi,
                                     hipMemcpyHostToDevice, s_read_));
            launch_lz4_decompress_chunks(arena_, d_chunks_, d_rdstage_, d_sizes_, mi, s_read_);
            HIP_CHECK(hipMemcpyAsync(h_status, d_sizes_, 4 * mi, hipMemcpyDeviceToHost,
                                     s_read_));
        }
        // D2H requested ranges
        uint64_t pout = 0;
        mi = 0;
        for (size_t j = j0; j < j1; j++) {
            const GetChunk& g = *dev[j].g;
            const uint8_t* src;
            if (dev[j].compressed) {
                src = d_rdstage_ + (uint64_t)mi * cs + g.off;
                mi++;
            } else {
                src = arena_ + dev[j].arena_off + g.off;
            }
            HIP_CHECK(hipMemcpyAsync(pin_out_ + pout, src, g.len, hipMemcpyDeviceToHost,
                                     s_read_));
            pout += round_up(g.len, 64);
        }
        HIP_CHECK(hipStreamSynchronize(s_read_));
        metrics().d2h_bytes += pout;
        for (int k = 0; k < mi; k++)
            if (h_status[k]) return St::bad_format;
        pout = 0;
        for (size_t j = j0; j < j1; j++) {
            const GetChunk& g = *dev[j].g;
            std::memcpy(g.out, pin_out_ + pout, g.len);
            pout += round_up(g.len, 64);
        }
        j0 = j1;
    }
    return St::ok;
}

St GpuEngine::release_chunk(const ChunkRef& ref) {
    std::lock_guard<std::mutex> g(map_mu_);
    auto it = map_.find(ref.fp);
    if (it == map_.end()) return St::not_found;
    if (--it->second.refs == 0) {
        if (it->second.spilled) {
            spill_.decref(ref.fp);
        } else {
            arena_free(it->second.off, it->second.len);
            lru_.erase(it->second.lru);
        }
        map_.erase(it);
        if (conf_.journal_ptr) conf_.journal_ptr->del_chunk(ref.fp);
    }
    return St::ok;
}

}  // namespace

std::unique_ptr<Engine> make_gpu_engine(const Conf& conf) {
    return std::make_unique<GpuEngine>(conf);
}

}  // namespace edgehip
