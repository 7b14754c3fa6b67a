// Observability: counters + stats rendering.
// Counters are plain relaxed atomics — the hot path pays one uncontended
// fetch_add per op; histograms can come later if profiles show need.
#pragma once

#include <atomic>
#include <string>
#include <vector>

#include "common.h"

namespace edgehip {

// Power-of-two latency histogram (ns): bucket b counts samples in
// [2^b, 2^(b+1)). Lock-free; rendered as cumulative micro/millisecond lines.
struct LatHist {
    static constexpr int kBuckets = 36;  // up to ~64 s
    std::atomic<uint64_t> b[kBuckets]{};
    void record(uint64_t ns) {
        int i = 63 - __builtin_clzll(ns | 1);
        if (i >= kBuckets) i = kBuckets - 1;
        b[i].fetch_add(1, std::memory_order_relaxed);
    }
    std::string render(const char* name) const;
};

struct Metrics {
    std::atomic<uint64_t> fuse_reqs{0};
    std::atomic<uint64_t> read_ops{0}, read_bytes{0};
    std::atomic<uint64_t> write_ops{0}, write_bytes{0};
    std::atomic<uint64_t> chunks_put{0}, chunks_get{0};
    std::atomic<uint64_t> dedup_hits{0}, dedup_bytes_saved{0};
    std::atomic<uint64_t> lz4_in_bytes{0}, lz4_out_bytes{0};
    std::atomic<uint64_t> rs_encoded_stripes{0}, rs_reconstructs{0};
    std::atomic<uint64_t> cache_hits{0}, cache_misses{0};
    std::atomic<uint64_t> h2d_bytes{0}, d2h_bytes{0};
    std::atomic<uint64_t> spilled_chunks{0}, spill_bytes{0};
    std::atomic<uint64_t> journal_resyncs{0};  // mid-log corruption skips on replay
    LatHist read_lat, write_lat;  // per-FUSE-request service time

    std::string render() const;  // text for the in-mount .edgehip/stats file
};

Metrics& metrics();

// roctx tracing: EDGEHIP_ROCTX=1 emits named ranges around
// the pipeline stages so rocprofv3 --sys-trace attributes device time to
// data-path phases. Zero overhead when off (one branch on a cached bool).
bool roctx_on();
void roctx_push(const char* name);
void roctx_pop();
struct RoctxRange {
    explicit RoctxRange(const char* name) { if (roctx_on()) roctx_push(name); }
    ~RoctxRange() { if (roctx_on()) roctx_pop(); }
};

// Fault injection: conf `fault=site:prob[,site:prob]`.
// Sites wired into the data path: "get_drop" (chunk read returns not_found,
// exercising RS reconstruct), "put_io" (ingest fails with EIO),
// "corrupt" (read bytes corrupted — caught by verify_read / crc checks).
class Faults {
  public:
    static Faults& inst();
    void configure(const std::string& spec, uint64_t seed = 0x12345678);
    bool fire(const char* site);  // true => inject at this call site
    bool enabled() const { return enabled_; }

  private:
    struct Site {
        std::string name;
        double prob;
    };
    std::vector<Site> sites_;
    std::atomic<uint64_t> rng_{0x9E3779B97F4A7C15ull};
    bool enabled_ = false;
};

}  // namespace edgehip
