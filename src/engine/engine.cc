// HostEngine — CPU data plane (judged config 1 + golden path for tests).
#include "engine.h"

#include <atomic>
#include <thread>

#include "../blake3.h"
#include "../gf256.h"
#include "../lz4.h"
#include "../obs.h"
#include "../journal.h"
#include "../store.h"

namespace edgehip {

namespace {

class HostEngine : public Engine {
  public:
    // arena acts as the host-RAM store budget (the host engine's "HBM");
    // with spill_dir set, chunks past it overflow to the disk tier — the
    // same three-tier shape as the GPU engine, CPU-testable.
    explicit HostEngine(const Conf& conf)
        : conf_(conf), store_(conf.spill_dir.empty() ? 0 : conf.arena_bytes,
                              conf.spill_dir) {
        if (conf.prealloc) BufPool::inst().prealloc(conf.prealloc, (uint32_t)conf.chunk_size);
    }

    St put_chunks(const PutChunk* put, ChunkRef* refs, int n) override {
        if (Faults::inst().fire("put_io")) return St::io;
        // CPU-heavy per-chunk work (blake3 + lz4) fans out across threads;
        // the store/dedup bookkeeping below stays sequential (it is cheap
        // and batch-order dedup accounting is easier to reason about).
        const bool do_fp = conf_.fp == FpAlgo::blake3;
        const bool do_lz4 = conf_.compress == Compress::lz4;
        struct Pre {
            Fp fp;
            std::vector<uint8_t> comp;
            bool compressed = false;
        };
        std::vector<Pre> pre((size_t)n);
        auto prep = [&](int i) {
            const PutChunk& p = put[i];
            if (do_fp) pre[i].fp = blake3_fp(p.data, p.len);
            // probe-only dedup peek: on a hit the compression is wasted work,
            // so skip it (racy reads are fine — worst case we compress)
            bool likely_dup =
                do_fp && conf_.dedup && store_.has(pre[i].fp);
            pre[i].compressed = !likely_dup && do_lz4 && p.len >= 4096 &&
                                lz4_chunk_compress(p.data, p.len, &pre[i].comp);
        };
        if (n > 1) {
            int nt = std::min(8, n);
            std::vector<std::thread> ts;
            std::atomic<int> next{0};
            for (int t = 0; t < nt; t++)
                ts.emplace_back([&] {
                    for (int i = next.fetch_add(1); i < n; i = next.fetch_add(1))
                        prep(i);
                });
            for (auto& t : ts) t.join();
        } else if (n == 1) {
            prep(0);
        }
        for (int i = 0; i < n; i++) {
            const PutChunk& p = put[i];
            ChunkRef& r = refs[i];
            r = ChunkRef{};
            r.logical_len = p.len;
            if (do_fp) {
                r.fp = pre[i].fp;
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
            std::vector<uint8_t> comp = std::move(pre[i].comp);
            bool compressed = pre[i].compressed;
            if (!compressed && comp.empty() && conf_.dedup && do_fp && do_lz4 &&
                p.len >= 4096) {
                // the dedup peek guessed "dup" but it is not stored: compress now
                compressed = lz4_chunk_compress(p.data, p.len, &comp);
            }
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
            if (conf_.journal_ptr && s == St::ok)
                conf_.journal_ptr->put_chunk(r.fp, p.data, p.len);  // logical bytes
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
            if (Faults::inst().fire("get_drop")) return St::not_found;
            uint32_t stored = store_.stored_len(r.fp);
            if (stored == 0) {
                // miss: try the journal chunk mirror (remount path)
                if (!load_from_journal(r)) return St::not_found;
                stored = store_.stored_len(r.fp);
                if (stored == 0) return St::not_found;
            }
            // the stored copy may differ from the ref (re-ingest after
            // remount); compression is decided by stored != logical
            if (stored == r.logical_len) {
                St s = store_.get(r.fp, g.out, g.off, g.len);
                if (s != St::ok) return s;
            } else {
                tmp.resize(stored);
                St s = store_.get(r.fp, tmp.data(), 0, stored);
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
            if (Faults::inst().fire("corrupt")) g.out[g.len / 2] ^= 0x5A;
            if (conf_.verify_read && g.off == 0 && g.len == r.logical_len &&
                conf_.fp == FpAlgo::blake3) {
                // full verification: re-hash and compare the content address
                if (blake3_fp(g.out, g.len) != r.fp) {
                    LOGW("verify_read: fingerprint mismatch (chunk %s)",
                         r.fp.hex().c_str());
                    return St::not_found;  // object layer reconstructs via RS
                }
            }
        }
        return St::ok;
    }

    // Remount: re-ingest a chunk's logical bytes from the journal mirror
    // (content addressing reproduces the same fingerprint).
    bool load_from_journal(const ChunkRef& r) {
        if (!conf_.journal_ptr) return false;
        std::vector<uint8_t> bytes;
        if (conf_.journal_ptr->get_chunk(r.fp, &bytes) != St::ok) return false;
        if (bytes.size() != r.logical_len) return false;
        PutChunk p{bytes.data(), (uint32_t)bytes.size(), nullptr};
        ChunkRef nr;
        if (put_chunks(&p, &nr, 1) != St::ok) return false;
        return nr.fp == r.fp;
    }

    St release_chunk(const ChunkRef& ref) override {
        St s = store_.decref(ref.fp);
        if (conf_.journal_ptr && s == St::ok && !store_.has(ref.fp))
            conf_.journal_ptr->del_chunk(ref.fp);
        return s;
    }

    St addref_chunk(const ChunkRef& ref) override { return store_.addref(ref.fp); }

    St debug_drop_chunk(const Fp& fp) override { return store_.remove(fp); }

    uint64_t stored_bytes() const override { return store_.bytes(); }
    uint64_t stored_chunks() const override { return store_.count(); }
    bool is_gpu() const override { return false; }
    int chunk_device(const Fp& fp) const override {
        return store_.has(fp) ? kDevHost : kDevAbsent;
    }

  private:
    Conf conf_;
    HostStore store_;
};

}  // namespace

// ---- Engine base: RS via the generic chunk APIs (host GF(256) math) --------
// Works for any engine; GpuEngine overrides with the on-device kernel path.
St Engine::rs_encode_stripes(const ChunkRef* data, int k, int m, uint32_t shard_len,
                             int nstripes, ChunkRef* parity_out,
                             const int16_t* parity_place) {
    for (int s = 0; s < nstripes; s++) {
        St st = rs_encode_stripe(data + (size_t)s * k, k, m, shard_len,
                                 parity_out + (size_t)s * m,
                                 parity_place ? parity_place + (size_t)s * m
                                              : nullptr);
        if (st != St::ok) return st;
    }
    return St::ok;
}

St Engine::rs_encode_stripe(const ChunkRef* data, int k, int m, uint32_t shard_len,
                            ChunkRef* parity_out, const int16_t* parity_place) {
    std::vector<std::vector<uint8_t>> shards((size_t)k);
    std::vector<GetChunk> gets;
    for (int j = 0; j < k; j++) {
        shards[j].assign(shard_len, 0);
        if (!data[j].fp.is_zero())
            gets.push_back({&data[j], shards[j].data(), 0, data[j].logical_len});
    }
    if (!gets.empty()) {
        St s = get_chunks(gets.data(), (int)gets.size());
        if (s != St::ok) return s;
    }
    std::vector<const uint8_t*> in((size_t)k);
    for (int j = 0; j < k; j++) in[j] = shards[j].data();
    std::vector<std::vector<uint8_t>> par((size_t)m, std::vector<uint8_t>(shard_len));
    std::vector<uint8_t*> out((size_t)m);
    for (int i = 0; i < m; i++) out[i] = par[i].data();
    rs_encode(k, m, in.data(), out.data(), shard_len);
    std::vector<PutChunk> puts((size_t)m);
    for (int i = 0; i < m; i++) {
        puts[i] = {par[i].data(), shard_len, nullptr};
        if (parity_place) puts[i].place_dev = parity_place[i];
    }
    St s = put_chunks(puts.data(), parity_out, m);
    if (s != St::ok) return s;
    for (int i = 0; i < m; i++)
        parity_out[i].flags |= ChunkRef::kFlagParity | ChunkRef::kFlagRs;
    metrics().rs_encoded_stripes++;
    return St::ok;
}

St Engine::rs_reconstruct(const ChunkRef* refs, const bool* present_in, int k, int m,
                          uint32_t shard_len, int target, uint8_t* out, uint32_t out_len) {
    int n = k + m;
    std::vector<std::vector<uint8_t>> shards((size_t)n);
    std::vector<uint8_t*> ptrs((size_t)n);
    std::vector<bool> present((size_t)n, false);
    int navail = 0;
    for (int i = 0; i < n && navail < k; i++) {
        shards[i].assign(shard_len, 0);
        ptrs[i] = shards[i].data();
        if (i == target) continue;
        if (refs[i].fp.is_zero()) {  // hole = zero shard, always available
            present[i] = true;
            navail++;
            continue;
        }
        if (present_in && !present_in[i]) continue;
        GetChunk g{&refs[i], shards[i].data(), 0, refs[i].logical_len};
        if (get_chunks(&g, 1) == St::ok) {
            present[i] = true;
            navail++;
        }
    }
    for (int i = 0; i < n; i++)
        if (shards[i].empty()) {
            shards[i].assign(shard_len, 0);
            ptrs[i] = shards[i].data();
        }
    if (navail < k) return St::io;
    std::vector<bool> pres2 = present;
    // convert to plain bool array for rs_reconstruct()
    std::vector<char> pc((size_t)n);
    for (int i = 0; i < n; i++) pc[i] = present[i] ? 1 : 0;
    std::vector<bool> dummy;
    bool pb[64];
    for (int i = 0; i < n; i++) pb[i] = pc[i] != 0;
    St s = ::edgehip::rs_reconstruct(k, m, ptrs.data(), pb, shard_len);
    if (s != St::ok) return s;
    if (out_len > shard_len) return St::inval;
    std::memcpy(out, shards[target].data(), out_len);
    metrics().rs_reconstructs++;
    return St::ok;
}

std::unique_ptr<Engine> make_host_engine(const Conf& conf) {
    return std::make_unique<HostEngine>(conf);
}

std::unique_ptr<Engine> make_engine(const Conf& conf) {
    Faults::inst().configure(conf.fault);
    if (conf.engine == EngineKind::gpu) return make_gpu_engine(conf);
    return make_host_engine(conf);
}

}  // namespace edgehip
