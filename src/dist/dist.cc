#include "dist.h"

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include "../obs.h"

namespace edgehip {

#define NCCL_CHECK(expr)                                                        \
    do {                                                                        \
        ncclResult_t _r = (expr);                                               \
        if (_r != ncclSuccess) {                                                \
            LOGE("rccl: %s at %s:%d", ncclGetErrorString(_r), __FILE__, __LINE__); \
            return St::io;                                                      \
        }                                                                       \
    } while (0)

constexpr uint64_t kTableSlots = 1 << 20;  // 64 MiB of ChunkRef records per GPU

DistCtx::~DistCtx() {
    for (size_t i = 0; i < comms_.size(); i++) {
        if (comms_[i]) ncclCommDestroy((ncclComm_t)comms_[i]);
    }
    for (size_t i = 0; i < d_tables_.size(); i++) {
        (void)hipSetDevice(first_dev_ + (int)i);
        if (d_tables_[i]) (void)hipFree(d_tables_[i]);
        if (d_scratch_[i]) (void)hipFree(d_scratch_[i]);
        if (streams_[i]) (void)hipStreamDestroy((hipStream_t)streams_[i]);
    }
}

St DistCtx::init_local(int ngpus, int first_dev, uint64_t table_slots) {
    ndev_ = ngpus;
    first_dev_ = first_dev;
    if (ngpus <= 1) return St::ok;  // single device: no collectives needed
    comms_.assign(ngpus, nullptr);
    streams_.assign(ngpus, nullptr);
    d_tables_.assign(ngpus, nullptr);
    d_scratch_.assign(ngpus, nullptr);
    totals_.assign(ngpus, 0);
    table_cap_ = table_slots ? table_slots : kTableSlots;
    std::vector<int> devs(ngpus);
    for (int i = 0; i < ngpus; i++) devs[i] = first_dev + i;
    // one communicator spanning the node's GPUs (RCCL over xGMI)
    NCCL_CHECK(ncclCommInitAll((ncclComm_t*)comms_.data(), ngpus, devs.data()));
    for (int i = 0; i < ngpus; i++) {
        if (hipSetDevice(devs[i]) != hipSuccess) return St::gpu;
        hipStream_t s;
        if (hipStreamCreateWithFlags(&s, hipStreamNonBlocking) != hipSuccess) return St::gpu;
        streams_[i] = s;
        if (hipMalloc(&d_tables_[i], table_cap_ * sizeof(ChunkRef)) != hipSuccess)
            return St::gpu;
        if (hipMalloc(&d_scratch_[i], 64 * sizeof(ChunkRef) * (ngpus + 1)) != hipSuccess)
            return St::gpu;
    }
    LOGI("dist: RCCL communicator over %d GPUs (first %d)", ngpus, first_dev);
    return St::ok;
}

St DistCtx::allgather_manifests(int dev_slot, const ChunkRef* recs, int n, int slots) {
    if (ndev_ <= 1) return St::ok;
    if (n > slots || slots > 64) return St::inval;
    // stage this rank's records (zero-padded to `slots`)
    std::vector<uint8_t> h(slots * sizeof(ChunkRef), 0);
    // ChunkRef is trivially copyable on the wire here (intra-node, same ABI)
    std::memcpy(h.data(), recs, n * sizeof(ChunkRef));
    if (hipSetDevice(first_dev_ + dev_slot) != hipSuccess) return St::gpu;
    uint8_t* d_in = (uint8_t*)d_scratch_[dev_slot];
    if (hipMemcpyAsync(d_in, h.data(), h.size(), hipMemcpyHostToDevice,
                       (hipStream_t)streams_[dev_slot]) != hipSuccess)
        return St::gpu;
    // C4: ncclAllGather — every device receives ndev_*slots records
    NCCL_CHECK(ncclGroupStart());
    for (int i = 0; i < ndev_; i++) {
        if (hipSetDevice(first_dev_ + i) != hipSuccess) return St::gpu;
        uint8_t* in = (uint8_t*)d_scratch_[i];
        uint8_t* out = in + slots * sizeof(ChunkRef);
        NCCL_CHECK(ncclAllGather(i == dev_slot ? d_in : in, out,
                                 slots * sizeof(ChunkRef), ncclInt8,
                                 (ncclComm_t)comms_[i], (hipStream_t)streams_[i]));
    }
    NCCL_CHECK(ncclGroupEnd());
    // Append the gathered records into each device's manifest RING. The
    // write head wraps at capacity (oldest records overwritten) — totals_
    // stays monotonic so manifest_count() saturates instead of zeroing.
    for (int i = 0; i < ndev_; i++) {
        if (hipSetDevice(first_dev_ + i) != hipSuccess) return St::gpu;
        uint64_t nrec = (uint64_t)ndev_ * slots;
        const uint8_t* src = (uint8_t*)d_scratch_[i] + slots * sizeof(ChunkRef);
        uint64_t head = totals_[i] % table_cap_;
        if (!warned_wrap_ && totals_[i] + nrec > table_cap_) {
            warned_wrap_ = true;
            LOGW("dist: manifest table wrapped at %llu records — oldest "
                 "entries are recycled (ring semantics)",
                 (unsigned long long)table_cap_);
        }
        while (nrec) {
            uint64_t take = std::min(nrec, table_cap_ - head);
            if (hipMemcpyAsync((uint8_t*)d_tables_[i] + head * sizeof(ChunkRef),
                               src, take * sizeof(ChunkRef),
                               hipMemcpyDeviceToDevice,
                               (hipStream_t)streams_[i]) != hipSuccess)
                return St::gpu;
            src += take * sizeof(ChunkRef);
            totals_[i] += take;
            head = (head + take) % table_cap_;
            nrec -= take;
        }
    }
    for (int i = 0; i < ndev_; i++) {
        if (hipSetDevice(first_dev_ + i) != hipSuccess) return St::gpu;
        if (hipStreamSynchronize((hipStream_t)streams_[i]) != hipSuccess) return St::gpu;
    }
    return St::ok;
}

St DistCtx::broadcast_tags(int root_slot, const uint64_t* tags, int n) {
    if (ndev_ <= 1 || n <= 0) return St::ok;
    if (n > 64 * 8) return St::inval;
    if (hipSetDevice(first_dev_ + root_slot) != hipSuccess) return St::gpu;
    uint8_t* d_root = (uint8_t*)d_scratch_[root_slot];
    if (hipMemcpyAsync(d_root, tags, n * 8, hipMemcpyHostToDevice,
                       (hipStream_t)streams_[root_slot]) != hipSuccess)
        return St::gpu;
    NCCL_CHECK(ncclGroupStart());
    for (int i = 0; i < ndev_; i++) {
        if (hipSetDevice(first_dev_ + i) != hipSuccess) return St::gpu;
        NCCL_CHECK(ncclBroadcast((uint8_t*)d_scratch_[i], (uint8_t*)d_scratch_[i], n * 8,
                                 ncclInt8, root_slot, (ncclComm_t)comms_[i],
                                 (hipStream_t)streams_[i]));
    }
    NCCL_CHECK(ncclGroupEnd());
    for (int i = 0; i < ndev_; i++) {
        if (hipSetDevice(first_dev_ + i) != hipSuccess) return St::gpu;
        if (hipStreamSynchronize((hipStream_t)streams_[i]) != hipSuccess) return St::gpu;
    }
    return St::ok;
}

St DistCtx::epoch_barrier(uint64_t epoch, uint64_t* max_epoch) {
    if (ndev_ <= 1) {
        if (max_epoch) *max_epoch = epoch;
        return St::ok;
    }
    NCCL_CHECK(ncclGroupStart());
    for (int i = 0; i < ndev_; i++) {
        if (hipSetDevice(first_dev_ + i) != hipSuccess) return St::gpu;
        uint64_t* d = (uint64_t*)d_scratch_[i];
        if (i == 0) {
            if (hipMemcpyAsync(d, &epoch, 8, hipMemcpyHostToDevice,
                               (hipStream_t)streams_[i]) != hipSuccess)
                return St::gpu;
        }
        NCCL_CHECK(ncclAllReduce(d, d + 1, 1, ncclUint64, ncclMax, (ncclComm_t)comms_[i],
                                 (hipStream_t)streams_[i]));
    }
    NCCL_CHECK(ncclGroupEnd());
    uint64_t out = 0;
    if (hipSetDevice(first_dev_) != hipSuccess) return St::gpu;
    if (hipMemcpy(&out, (uint64_t*)d_scratch_[0] + 1, 8, hipMemcpyDeviceToHost) != hipSuccess)
        return St::gpu;
    if (max_epoch) *max_epoch = out;
    return St::ok;
}

void* DistCtx::manifest_table(int dev_slot) const {
    return dev_slot < (int)d_tables_.size() ? d_tables_[dev_slot] : nullptr;
}
uint64_t DistCtx::manifest_count(int dev_slot) const {
    if (dev_slot >= (int)totals_.size()) return 0;
    return std::min(totals_[dev_slot], table_cap_);
}
uint64_t DistCtx::manifest_total(int dev_slot) const {
    return dev_slot < (int)totals_.size() ? totals_[dev_slot] : 0;
}

}  // namespace edgehip
