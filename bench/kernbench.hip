// Kernel microbenchmarks: per-kernel GB/s on device-resident data (no PCIe),
// the numbers the data-plane design is judged against (HBM ceiling ~6.3 TB/s).
// Run on an MI355X:  bin/kernbench [mib_per_round [rounds]]
// rocprof evidence:  rocprofv3 --kernel-trace --stats -- bin/kernbench 1024 3
#include <chrono>
#include <cstdio>
#include <vector>

#include "../src/gf256.h"
#include "../src/kernels/launchers.h"

using namespace edgehip;
using clk = std::chrono::steady_clock;

static double secs(clk::time_point a, clk::time_point b) {
    return std::chrono::duration<double>(b - a).count();
}

int main(int argc, char** argv) {
    uint64_t mib = argc > 1 ? atoll(argv[1]) : 1024;
    int rounds = argc > 2 ? atoi(argv[2]) : 3;
    const uint64_t cs = 1 << 20;
    const int nchunks = (int)mib;
    const uint64_t total = mib * cs;

    uint8_t *d_src, *d_dst, *d_fps, *d_scratch;
    DevChunk* d_chunks;
    ulong2* d_segmeta;
    uint32_t *d_sizes, *d_segsizes, *d_status;
    HIP_CHECK(hipMalloc(&d_src, total));
    HIP_CHECK(hipMalloc(&d_dst, total + (64 << 20)));
    HIP_CHECK(hipMalloc(&d_fps, 32 * nchunks));
    HIP_CHECK(hipMalloc(&d_chunks, sizeof(DevChunk) * nchunks));
    HIP_CHECK(hipMalloc(&d_segmeta, sizeof(ulong2) * nchunks * 256));
    HIP_CHECK(hipMalloc(&d_segsizes, 4 * nchunks * 256));
    HIP_CHECK(hipMalloc(&d_scratch, (uint64_t)nchunks * 256 * 4352));
    HIP_CHECK(hipMalloc(&d_sizes, 4 * nchunks));
    HIP_CHECK(hipMalloc(&d_status, 4 * nchunks));

    // fill with pseudo-random (incompressible) data on device
    {
        std::vector<uint8_t> h(total);
        uint64_t x = 0x243F6A8885A308D3ull;
        for (uint64_t i = 0; i < total; i += 8) {
            x ^= x << 13;
            x ^= x >> 7;
            x ^= x << 17;
            memcpy(&h[i], &x, 8);
        }
        HIP_CHECK(hipMemcpy(d_src, h.data(), total, hipMemcpyHostToDevice));
    }
    std::vector<DevChunk> chunks(nchunks);
    for (int i = 0; i < nchunks; i++) chunks[i] = DevChunk{cs * i, cs * i, (uint32_t)cs, (uint32_t)(i * 256)};
    HIP_CHECK(hipMemcpy(d_chunks, chunks.data(), sizeof(DevChunk) * nchunks,
                        hipMemcpyHostToDevice));
    std::vector<ulong2> segs((size_t)nchunks * 256);
    for (int i = 0; i < nchunks; i++)
        for (int s = 0; s < 256; s++)
            segs[(size_t)i * 256 + s] = ulong2{cs * i + (uint64_t)s * 4096, 4096};
    HIP_CHECK(hipMemcpy(d_segmeta, segs.data(), sizeof(ulong2) * segs.size(),
                        hipMemcpyHostToDevice));

    auto bench = [&](const char* name, double bytes, auto&& fn) {
        fn();  // warmup
        HIP_CHECK(hipDeviceSynchronize());
        double best = 1e30;
        for (int r = 0; r < rounds; r++) {
            auto t0 = clk::now();
            fn();
            HIP_CHECK(hipDeviceSynchronize());
            best = std::min(best, secs(t0, clk::now()));
        }
        printf("%-22s %8.1f GB/s  (%.2f ms, %llu MiB)\n", name, bytes / best / 1e9,
               best * 1e3, (unsigned long long)(bytes / (1 << 20)));
        return best;
    };

    uint32_t* d_fpg = nullptr;
    (void)hipMalloc(&d_fpg, 32ull * kB3Groups * nchunks);
    bench("blake3_fingerprint", (double)total,
          [&] { launch_blake3_chunks(d_src, d_chunks, d_fps, nchunks, nullptr, d_fpg); });

    bench("crc32c_pages", (double)total,
          [&] { launch_crc32c_chunks(d_src, d_chunks, d_sizes, nchunks, nullptr); });

    bench("lz4_compress(random)", (double)total, [&] {
        launch_lz4_compress_segs(d_src, d_segmeta, d_scratch, d_segsizes, nchunks * 256,
                                 nullptr);
    });

    bench("batch_copy", 2.0 * total,
          [&] { launch_batch_copy(d_src, d_dst, d_chunks, nchunks, nullptr); });

    // zero-copy PCIe kernels: pull from / push to scattered pinned host
    // buffers (the ingest / direct-read / eviction transfer path)
    {
        void* pin = nullptr;
        if (hipHostMalloc(&pin, total, hipHostMallocDefault) == hipSuccess) {
            void* dp = nullptr;
            bool mapped = hipHostGetDevicePointer(&dp, pin, 0) == hipSuccess &&
                          dp == pin;
            if (mapped) {
                std::vector<PtrChunk> h_pd(nchunks);
                for (uint64_t i = 0; i < nchunks; i++)
                    h_pd[i] = PtrChunk{(uint8_t*)pin + i * (1 << 20),
                                       i * (1ull << 20), 1 << 20, 0};
                PtrChunk* d_pd = nullptr;
                (void)hipMalloc(&d_pd, sizeof(PtrChunk) * nchunks);
                (void)hipMemcpy(d_pd, h_pd.data(), sizeof(PtrChunk) * nchunks,
                                hipMemcpyHostToDevice);
                bench("gather_host(PCIe pull)", (double)total, [&] {
                    launch_gather_host(d_pd, d_dst, (int)nchunks, 1 << 20, nullptr);
                });
                // scatter: src = device addr in dst_off, dst = host ptr
                for (uint64_t i = 0; i < nchunks; i++)
                    h_pd[i].dst_off = (uint64_t)(uintptr_t)(d_src + i * (1 << 20));
                (void)hipMemcpy(d_pd, h_pd.data(), sizeof(PtrChunk) * nchunks,
                                hipMemcpyHostToDevice);
                bench("scatter_host(PCIe push)", (double)total, [&] {
                    launch_scatter_host(d_pd, (int)nchunks, 1 << 20, nullptr);
                });
                (void)hipFree(d_pd);
            }
            (void)hipHostFree(pin);
        }
    }

    // RS 6+2 over the whole buffer (k shards of total/6)
    {
        uint64_t shard = total / 8 & ~15ull;
        RsDesc d{};
        auto M = rs_parity_matrix(6, 2);
        for (int j = 0; j < 6; j++) d.in[j] = d_src + shard * j;
        for (int i = 0; i < 2; i++) d.out[i] = d_dst + shard * i;
        for (int i = 0; i < 2; i++)
            for (int j = 0; j < 6; j++) d.coef[i][j] = M[(size_t)i * 6 + j];
        d.k = 6;
        d.m = 2;
        d.len = shard;
        // traffic: read 6 shards + write 2
        bench("rs_6+2_encode", (double)shard * 8, [&] { launch_gf_matmul(d, nullptr); });
        // decode: 2 lost data shards from 4 data + 2 parity
        RsDesc r{};
        for (int j = 0; j < 6; j++) r.in[j] = d_src + shard * j;
        for (int i = 0; i < 2; i++) r.out[i] = d_dst + shard * i;
        for (int i = 0; i < 2; i++)
            for (int j = 0; j < 6; j++) r.coef[i][j] = M[(size_t)i * 6 + j];  // shape proxy
        r.k = 6;
        r.m = 2;
        r.len = shard;
        bench("rs_6+2_decode", (double)shard * 8, [&] { launch_gf_matmul(r, nullptr); });
        // the MFMA bit-plane variant (north-star mandated comparison)
        bench("rs_6+2_encode_mfma", (double)shard * 8,
              [&] { launch_gf_matmul_mfma(d, nullptr); });
    }

    // dedup index upsert
    {
        unsigned long long* d_tbl;
        uint64_t slots = 1ull << 24;
        HIP_CHECK(hipMalloc(&d_tbl, slots * 8));
        HIP_CHECK(hipMemset(d_tbl, 0, slots * 8));
        int nfp = nchunks;
        bench("dedup_upsert", (double)nfp * 32, [&] {
            launch_dedup_upsert(d_tbl, slots - 1, d_fps, d_status, nfp, nullptr);
        });
        (void)hipFree(d_tbl);
    }

    // decompress a compressible buffer (text-like): build on host
    {
        std::vector<uint8_t> h(total);
        const char* words[] = {"edge", "hip", "mi355x", "gfx950", "chunk", "lz4 "};
        uint64_t x = 12345, o = 0;
        while (o < total) {
            const char* w = words[x % 6];
            size_t L = strlen(w);
            size_t rep = 1 + (x >> 8) % 12;
            for (size_t r2 = 0; r2 < rep && o + L <= total; r2++) {
                memcpy(&h[o], w, L);
                o += L;
            }
            x = x * 6364136223846793005ull + 1442695040888963407ull;
            if (o + 8 >= total) break;
        }
        HIP_CHECK(hipMemcpy(d_src, h.data(), total, hipMemcpyHostToDevice));
        launch_lz4_compress_segs(d_src, d_segmeta, d_scratch, d_segsizes, nchunks * 256,
                                 nullptr);
        bench("lz4_compress(text)", (double)total, [&] {
            launch_lz4_compress_segs(d_src, d_segmeta, d_scratch, d_segsizes, nchunks * 256,
                                     nullptr);
        });
        // emit frames then decompress them
        launch_lz4_emit(d_src, d_scratch, d_segsizes, d_chunks, d_dst, d_sizes, nchunks,
                        nullptr);
        HIP_CHECK(hipDeviceSynchronize());
        std::vector<uint32_t> sizes(nchunks);
        HIP_CHECK(hipMemcpy(sizes.data(), d_sizes, 4 * nchunks, hipMemcpyDeviceToHost));
        int ok = 0;
        uint64_t framed_bytes = 0;
        std::vector<DevChunk> dec(nchunks);
        for (int i = 0; i < nchunks; i++) {
            if (!sizes[i]) continue;
            dec[ok] = DevChunk{chunks[i].dst_off, cs * (uint64_t)ok, (uint32_t)cs, sizes[i]};
            framed_bytes += sizes[i];
            ok++;
        }
        printf("  (text compressed %d/%d chunks, ratio %.2fx)\n", ok, nchunks,
               ok ? (double)ok * cs / framed_bytes : 0.0);
        if (ok) {
            DevChunk* d_dec;
            HIP_CHECK(hipMalloc(&d_dec, sizeof(DevChunk) * ok));
            HIP_CHECK(hipMemcpy(d_dec, dec.data(), sizeof(DevChunk) * ok,
                                hipMemcpyHostToDevice));
            uint8_t* d_out2 = d_scratch;  // reuse
            bench("lz4_decompress(text)", (double)ok * cs, [&] {
                launch_lz4_decompress_chunks(d_dst, d_dec, d_out2, d_status, ok, nullptr);
            });
            (void)hipFree(d_dec);
        }
    }
    return 0;
}
