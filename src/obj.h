// Object layer: offset→chunk mapping, write buffering with
// batched engine puts, read planning, commit (manifest build).
// One Object per file inode; FUSE worker threads serialize on obj mutex (the
// engine below parallelizes across objects and within batches).
#pragma once

#include <condition_variable>
#include <map>
#include <memory>
#include <mutex>
#include <set>

#include "common.h"
#include "engine/engine.h"
#include "manifest.h"

namespace edgehip {

class Object {
  public:
    Object(Engine* eng, const Conf* conf) : eng_(eng), conf_(conf) {
        mf_.chunk_size = conf->chunk_size;
        mf_.fp_algo = (uint8_t)conf->fp;
        mf_.compress = (uint8_t)conf->compress;
    }
    ~Object();

    St read(uint64_t off, uint64_t len, uint8_t* out, uint64_t* got);
    // Batched scatter read: n ranges concatenated into out — ONE engine batch
    // (the 4 KiB random-read path: fio iodepth analog).
    // SEEK_DATA/SEEK_HOLE support: next offset >= off that is (or is not)
    // backed by stored/dirty bytes. Returns size_ when no further data.
    uint64_t next_boundary(uint64_t off, bool want_data);

    // FALLOC_FL_PUNCH_HOLE: whole chunks in range drop their refs (space
    // reclaimed, reads return zeros); partial edge chunks are zeroed in place.
    St punch_hole(uint64_t off, uint64_t len);

    // Server-side copy (FUSE_COPY_FILE_RANGE): chunk-aligned spans share the
    // source's chunk refs (content-addressed — no data moves); edges fall
    // back to read+write. Clamps at the source EOF; returns bytes copied.
    St copy_range(Object* src, uint64_t soff, uint64_t doff, uint64_t len,
                  uint64_t* copied);

    St preadv(const uint64_t* offs, const uint32_t* lens, int n, uint8_t* out,
              uint64_t* got);
    St write(uint64_t off, const uint8_t* data, uint64_t len);
    St flush();                    // drain write buffers into the engine
    St truncate(uint64_t size);
    uint64_t size() const { return size_; }

    // commit(): flush + rebuild the manifest bytes (the durable form).
    St commit(std::vector<uint8_t>* manifest_bytes = nullptr);

    const Manifest& manifest() const { return mf_; }
    St set_manifest(const Manifest& m);   // adopt (remount / clone)

  private:
    struct Dirty {
        Blob data;            // pool buffer, chunk_size capacity
        uint32_t len = 0;     // valid prefix length
        bool loaded = false;  // true once committed content was merged
        int busy = 0;         // writers copying payload with mu_ DROPPED:
                              // the entry (and its blob) must not be erased,
                              // flushed, merged or snapshotted while busy>0
    };

    St flush_locked(std::unique_lock<std::mutex>& lk);
    void submit_async_flush_locked(std::vector<uint64_t> idxs);
    void wait_inflight_locked(std::unique_lock<std::mutex>& lk);
    St load_chunk_for_rmw(uint64_t idx, Dirty* d);
    void drop_ref(const ChunkRef& r);
    // RS (judged config 4): refresh parity for stripes whose data changed;
    // reconstruct a lost chunk from stripe survivors and repair it in place.
    void mark_stripe_dirty(uint64_t chunk_idx);
    St rs_update_locked();
    St rs_reconstruct_chunk_locked(uint64_t idx, uint8_t* chunk_buf);
    // async parity pipeline: encode stable dirty stripes in the background
    // while their chunks are still HBM-resident (flush-time encode after
    // spill pulled every shard back from host RAM — the config-4 wall)
    void maybe_async_rs_locked();
    uint64_t stripe_shard_map(const ChunkRef* data, const ChunkRef* par) const;
    // stripe-aware placement: device slot for shard `shard` of `stripe`
    // (data shards 0..k-1, parity k..k+m-1); -1 = no hint (fp placement)
    int16_t shard_place_(uint64_t stripe, int shard) const;
    int16_t data_place_(uint64_t chunk_idx) const;

    Engine* eng_;
    const Conf* conf_;
    std::mutex mu_;
    Manifest mf_;                    // committed chunk refs (index = chunk idx)
    std::map<uint64_t, Dirty> dirty_;
    std::set<uint64_t> rs_dirty_;    // stripe indices with stale parity
    std::set<uint64_t> rs_inflight_; // stripes encoding in the background
    int rs_busy_ = 0;                // background rs-encode tasks outstanding
                                     // (separate from inflight_ so writer
                                     // backpressure ignores parity work)
    int write_busy_ = 0;             // writers in an unlocked payload copy
                                     // (flush/truncate/punch wait for 0)
    int inflight_ = 0;               // background flush batches outstanding
    std::condition_variable flush_cv_;
    St async_err_ = St::ok;
    // chunks handed to a background flush: non-owning views (task owns the
    // blobs; valid while the entry exists — both maintained under mu_)
    std::map<uint64_t, std::pair<const uint8_t*, uint32_t>> inflight_view_;
    std::vector<ChunkRef> pending_publish_;  // C4: refs committed since last publish
    // CDC mode (conf.cdc): append-only stream, variable chunk boundaries
    std::vector<uint64_t> cum_;   // cumulative END offset per committed chunk
    // CDC reorder window: FUSE writeback can deliver a sequentially-written
    // file's 1 MiB requests out of order across reader threads; stash
    // ahead-of-stream writes until the gap fills (bounded).
    std::map<uint64_t, std::vector<uint8_t>> cdc_stash_;
    uint64_t cdc_stash_bytes_ = 0;
    std::vector<uint8_t> tail_;   // uncut stream tail
    St cdc_write_locked(uint64_t off, const uint8_t* data, uint64_t len);
    St cdc_cut_locked(bool final_tail);
    St cdc_read_locked(uint64_t off, uint64_t len, uint8_t* out);
    uint64_t size_ = 0;              // logical size incl. dirty extension
};

}  // namespace edgehip
