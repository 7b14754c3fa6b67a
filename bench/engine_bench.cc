// Native ingest/read micro-bench: the bench.py object-api workload with zero
// Python in the loop, for rocprofv3 attribution of the engine pipeline
// (SURVEY.md §7). Usage: engine_bench [gib_per_job [jobs [opts]]]
#include <atomic>
#include <chrono>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <random>
#include <thread>
#include <vector>

#include "../include/edgehip.h"

static constexpr uint64_t MB = 1ull << 20;

int main(int argc, char** argv) {
    double gib = argc > 1 ? atof(argv[1]) : 2.0;
    int jobs = argc > 2 ? atoi(argv[2]) : 6;
    const char* opts = argc > 3 ? argv[3]
                               : "engine=gpu,fp=blake3,compress=lz4,dedup=1,"
                                 "arena=32g,spill=0";
    void* eng = eh_engine_create(opts);
    if (!eng) {
        fprintf(stderr, "engine create failed (%s)\n", opts);
        return 1;
    }
    // content ring shared across jobs, like bench.py: unique per MiB
    uint64_t ring = 1024 * MB;
    std::vector<uint8_t> buf(ring);
    {
        std::vector<std::thread> ts;
        for (int t = 0; t < 8; t++)
            ts.emplace_back([&, t] {
                std::mt19937_64 g(t);
                auto* p = (uint64_t*)(buf.data() + t * (ring / 8));
                for (uint64_t i = 0; i < ring / 8 / 8; i++) p[i] = g();
            });
        for (auto& t : ts) t.join();
    }

    uint64_t per_job = (uint64_t)(gib * 1024.0) / jobs * MB;
    const uint64_t unit = 8 * MB;
    std::vector<double> wsec(jobs), rsec(jobs);
    std::vector<int> errs(jobs, 0);

    // globally-unique content stamping (bench.py stamp_blocks analog): each
    // job owns a disjoint ring region and stamps every 1 MiB block with a
    // fresh counter before writing — zero dedup hits by construction
    std::atomic<uint64_t> stamp{1};
    const uint64_t region_units = ring / unit / jobs;
    auto worker = [&](int j) {
        void* o = eh_object_create(eng);
        auto t0 = std::chrono::steady_clock::now();
        for (uint64_t off = 0; off < per_job; off += unit) {
            uint64_t src = ((uint64_t)j * region_units +
                            (off / unit) % region_units) * unit;
            for (uint64_t b = 0; b < unit; b += MB) {
                uint64_t s = stamp.fetch_add(1);
                std::memcpy(buf.data() + src + b, &s, 8);
            }
            if (eh_object_write(o, off, buf.data() + src, unit) != 0) errs[j]++;
        }
        // keep a copy of the final unit's content for read-back verification
        std::vector<uint8_t> expect;
        uint64_t expect_off = (uint64_t)-1;
        if (per_job >= unit) {
            uint64_t last = (per_job - unit) / unit;
            uint64_t src = ((uint64_t)j * region_units + last % region_units) * unit;
            expect.assign(buf.data() + src, buf.data() + src + unit);
            expect_off = last * unit;
        }
        if (eh_object_flush(o) != 0) errs[j]++;
        auto t1 = std::chrono::steady_clock::now();
        std::vector<uint8_t> rbuf(unit);
        // pinned destination: direct-DMA reads land at link rate (D2H into
        // pageable memory degrades to runtime-staged copies)
        bool pinned = eh_mem_pin(rbuf.data(), unit) == 0;
        for (uint64_t off = 0; off < per_job; off += unit) {
            if (eh_object_read(o, off, rbuf.data(), unit) != (int64_t)unit) errs[j]++;
            // verify bytes (not just sizes): a silently-faulted kernel must
            // show up as corruption, not as a clean-looking rate
            if (off == expect_off && std::memcmp(rbuf.data(), expect.data(), unit)) {
                fprintf(stderr, "job %d: READBACK MISMATCH at off %llu\n", j,
                        (unsigned long long)off);
                errs[j]++;
            }
        }
        if (pinned) eh_mem_unpin(rbuf.data());
        auto t2 = std::chrono::steady_clock::now();
        wsec[j] = std::chrono::duration<double>(t1 - t0).count();
        rsec[j] = std::chrono::duration<double>(t2 - t1).count();
        eh_object_truncate(o, 0);
        eh_object_destroy(o);
    };

    // warmup: half the data once (grows the pinned pool to steady state)
    {
        std::vector<std::thread> ts;
        for (int j = 0; j < jobs; j++) ts.emplace_back(worker, j);
        for (auto& t : ts) t.join();
    }
    std::vector<std::thread> ts;
    auto T0 = std::chrono::steady_clock::now();
    for (int j = 0; j < jobs; j++) ts.emplace_back(worker, j);
    for (auto& t : ts) t.join();
    auto T1 = std::chrono::steady_clock::now();

    double wall = std::chrono::duration<double>(T1 - T0).count();
    double w = 0, r = 0;
    int e = 0;
    for (int j = 0; j < jobs; j++) {
        w = std::max(w, wsec[j]);
        r = std::max(r, rsec[j]);
        e += errs[j];
    }
    double gibs = gib;
    printf("engine_bench: write %.2f GiB/s  read %.2f GiB/s  combined %.2f GiB/s"
           "  (%.2f GiB x2, %d jobs, errs %d)\n",
           gibs / w, gibs / r, 2 * gibs / wall, gibs, jobs, e);
    eh_engine_destroy(eng);
    return e ? 1 : 0;
}
