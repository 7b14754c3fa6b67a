// Chunk persistence — host RAM store with refcounts.
// Used as: (a) the whole store for the host engine (config 1), (b) the spill
// tier behind the GPU HBM arena (config 4's 1 TiB object > 288 GB/GPU).
// Sharded by fingerprint to keep lock contention off the FUSE thread pool.
#pragma once

#include <atomic>
#include <memory>
#include <mutex>
#include <unordered_map>
#include <vector>

#include "bufpool.h"
#include "common.h"

namespace edgehip {

class HostStore {
  public:
    // cap_bytes: RAM-resident budget (0 = unlimited). dir: optional disk
    // overflow tier — once RAM is at cap, chunks land as files under
    // dir/<fp-hex> (config 4's 1 TiB object > HBM + host RAM). Disk-tier
    // reads pread(2) on demand.
    explicit HostStore(uint64_t cap_bytes = 0, const std::string& dir = "");

    // Stores (or bumps the refcount of) fp, CONSUMING the blob (zero-copy
    // ingest: pool buffers cycle object layer -> store -> pool). Returns
    // exists if already present (refcount bumped, blob recycled — dedup hit),
    // ok if newly stored.
    St put(const Fp& fp, Blob&& data);
    St put_copy(const Fp& fp, const uint8_t* data, uint32_t len);
    St get(const Fp& fp, uint8_t* out, uint32_t off, uint32_t len) const;
    bool has(const Fp& fp) const;
    uint32_t stored_len(const Fp& fp) const;  // 0 if absent
    St addref(const Fp& fp);
    St decref(const Fp& fp);  // frees at 0
    St remove(const Fp& fp);  // unconditional erase (fault injection)
    uint64_t bytes() const { return bytes_.load() + disk_bytes_.load(); }
    uint64_t disk_bytes() const { return disk_bytes_.load(); }
    uint64_t count() const;

  private:
    struct Chunk {
        Blob data;           // RAM tier (data.p != nullptr)
        uint32_t disk_len = 0;  // disk tier: stored file length
        uint32_t refs = 1;
        bool on_disk() const { return data.p == nullptr; }
    };
    std::string chunk_path(const Fp& fp) const;
    St write_file(const Fp& fp, const uint8_t* p, uint32_t len);
    static constexpr int kShards = 64;
    struct Shard {
        mutable std::mutex mu;
        std::unordered_map<Fp, Chunk, FpHash> map;
    };
    Shard& shard(const Fp& fp) const { return shards_[fp.b[0] & (kShards - 1)]; }
    mutable Shard shards_[kShards];
    std::atomic<uint64_t> bytes_{0};       // RAM-resident bytes
    std::atomic<uint64_t> disk_bytes_{0};  // disk-tier bytes
    uint64_t cap_;
    std::string dir_;
};

}  // namespace edgehip
