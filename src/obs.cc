// Observability: leveled logging + lock-free-ish counters.
#include "obs.h"

#include <cstdarg>
#include <ctime>
#include <malloc.h>
#include <mutex>
#include <unistd.h>

namespace edgehip {

// 1 MiB chunk buffers churn constantly; without this every large alloc is an
// mmap/munmap pair plus page-fault storm (~4x slowdown on the write path).
__attribute__((constructor)) static void tune_malloc() {
    mallopt(M_MMAP_THRESHOLD, 256 * 1024 * 1024);
    mallopt(M_TRIM_THRESHOLD, 256 * 1024 * 1024);
}

LogLevel g_log_level = LogLevel::info;

uint64_t now_ns() {
    timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return (uint64_t)ts.tv_sec * 1000000000ull + ts.tv_nsec;
}
uint64_t wall_ns() {
    timespec ts;
    clock_gettime(CLOCK_REALTIME, &ts);
    return (uint64_t)ts.tv_sec * 1000000000ull + ts.tv_nsec;
}

void log_line(LogLevel lvl, const char* fmt, ...) {
    static std::mutex mu;
    static const char* names[] = {"TRC", "DBG", "INF", "WRN", "ERR"};
    char body[2048];
    va_list ap;
    va_start(ap, fmt);
    vsnprintf(body, sizeof(body), fmt, ap);
    va_end(ap);
    char line[2200];
    double t = now_ns() * 1e-9;
    int n = snprintf(line, sizeof(line), "[edgehip %s %.6f] %s\n", names[(int)lvl], t, body);
    std::lock_guard<std::mutex> g(mu);
    ssize_t w = write(2, line, (size_t)n);
    (void)w;
}

Metrics& metrics() {
    static Metrics m;
    return m;
}

#include <roctracer/roctx.h>
bool roctx_on() {
    static bool on = [] {
        const char* e = getenv("EDGEHIP_ROCTX");
        return e && *e && *e != '0';
    }();
    return on;
}
void roctx_push(const char* name) { roctxRangePushA(name); }
void roctx_pop() { roctxRangePop(); }

std::string LatHist::render(const char* name) const {
    // cumulative percentile summary from power-of-two buckets
    uint64_t total = 0;
    for (int i = 0; i < kBuckets; i++) total += b[i].load();
    char buf[256];
    if (!total) {
        snprintf(buf, sizeof(buf), "%s p50=- p90=- p99=- n=0\n", name);
        return buf;
    }
    auto pct = [&](double q) {
        uint64_t need = (uint64_t)(q * total);
        if (need == 0) need = 1;
        uint64_t acc = 0;
        for (int i = 0; i < kBuckets; i++) {
            acc += b[i].load();
            if (acc >= need) return (double)(1ull << (i + 1)) / 1e3;  // -> us
        }
        return (double)(1ull << kBuckets) / 1e3;
    };
    snprintf(buf, sizeof(buf), "%s p50<=%.0fus p90<=%.0fus p99<=%.0fus n=%llu\n",
             name, pct(0.50), pct(0.90), pct(0.99), (unsigned long long)total);
    return buf;
}

std::string Metrics::render() const {
    char buf[4096];
    snprintf(buf, sizeof(buf),
             "fuse_reqs %llu\n"
             "read_ops %llu\nread_bytes %llu\n"
             "write_ops %llu\nwrite_bytes %llu\n"
             "chunks_put %llu\nchunks_get %llu\n"
             "dedup_hits %llu\ndedup_bytes_saved %llu\n"
             "lz4_in_bytes %llu\nlz4_out_bytes %llu\n"
             "rs_encoded_stripes %llu\nrs_reconstructs %llu\n"
             "cache_hits %llu\ncache_misses %llu\n"
             "h2d_bytes %llu\nd2h_bytes %llu\n"
             "spilled_chunks %llu\nspill_bytes %llu\n"
             "journal_resyncs %llu\n",
             (unsigned long long)fuse_reqs.load(),
             (unsigned long long)read_ops.load(), (unsigned long long)read_bytes.load(),
             (unsigned long long)write_ops.load(), (unsigned long long)write_bytes.load(),
             (unsigned long long)chunks_put.load(), (unsigned long long)chunks_get.load(),
             (unsigned long long)dedup_hits.load(), (unsigned long long)dedup_bytes_saved.load(),
             (unsigned long long)lz4_in_bytes.load(), (unsigned long long)lz4_out_bytes.load(),
             (unsigned long long)rs_encoded_stripes.load(), (unsigned long long)rs_reconstructs.load(),
             (unsigned long long)cache_hits.load(), (unsigned long long)cache_misses.load(),
             (unsigned long long)h2d_bytes.load(), (unsigned long long)d2h_bytes.load(),
             (unsigned long long)spilled_chunks.load(), (unsigned long long)spill_bytes.load(),
             (unsigned long long)journal_resyncs.load());
    std::string out(buf);
    out += read_lat.render("read_latency");
    out += write_lat.render("write_latency");
    return out;
}

Faults& Faults::inst() {
    static Faults f;
    return f;
}

void Faults::configure(const std::string& spec, uint64_t seed) {
    sites_.clear();
    enabled_ = false;
    rng_ = seed ? seed : 0x9E3779B97F4A7C15ull;
    size_t pos = 0;
    while (pos < spec.size()) {
        size_t comma = spec.find(',', pos);
        if (comma == std::string::npos) comma = spec.size();
        std::string item = spec.substr(pos, comma - pos);
        pos = comma + 1;
        size_t colon = item.find(':');
        if (colon == std::string::npos) continue;
        Site s;
        s.name = item.substr(0, colon);
        s.prob = atof(item.c_str() + colon + 1);
        if (s.prob > 0) {
            sites_.push_back(s);
            enabled_ = true;
            LOGW("faults: %s armed at p=%.4f", s.name.c_str(), s.prob);
        }
    }
}

bool Faults::fire(const char* site) {
    if (!enabled_) return false;
    for (auto& s : sites_) {
        if (s.name != site) continue;
        // xorshift64* per call (thread-coarse; probabilistic tests only)
        uint64_t x = rng_.fetch_add(0x9E3779B97F4A7C15ull) ^ rng_.load();
        x ^= x >> 12;
        x ^= x << 25;
        x ^= x >> 27;
        double u = (double)(x * 0x2545F4914F6CDD1Dull >> 11) / 9007199254740992.0;
        return u < s.prob;
    }
    return false;
}

// ---------------------------------------------------------------- crc32c
// Castagnoli polynomial 0x1EDC6F41, reflected 0x82F63B78. Table-sliced by 8.
static uint32_t crc_tab[8][256];
static bool crc_init_done = [] {
    for (uint32_t i = 0; i < 256; i++) {
        uint32_t c = i;
        for (int k = 0; k < 8; k++) c = (c & 1) ? (0x82F63B78u ^ (c >> 1)) : (c >> 1);
        crc_tab[0][i] = c;
    }
    for (uint32_t i = 0; i < 256; i++)
        for (int t = 1; t < 8; t++)
            crc_tab[t][i] = (crc_tab[t - 1][i] >> 8) ^ crc_tab[0][crc_tab[t - 1][i] & 0xff];
    return true;
}();

#if defined(__x86_64__)
__attribute__((target("sse4.2"))) static uint32_t crc32c_hw(const uint8_t* p, size_t len, uint32_t c) {
    while (len && ((uintptr_t)p & 7)) { c = __builtin_ia32_crc32qi(c, *p++); len--; }
    uint64_t c64 = c;
    while (len >= 8) {
        uint64_t v;
        std::memcpy(&v, p, 8);
        c64 = __builtin_ia32_crc32di(c64, v);
        p += 8; len -= 8;
    }
    c = (uint32_t)c64;
    while (len--) c = __builtin_ia32_crc32qi(c, *p++);
    return c;
}
static const bool have_sse42 = __builtin_cpu_supports("sse4.2");
#endif

uint32_t crc32c(const void* data, size_t len, uint32_t seed) {
    const uint8_t* p = (const uint8_t*)data;
    uint32_t c = ~seed;
#if defined(__x86_64__)
    if (have_sse42) return ~crc32c_hw(p, len, c);
#endif
    while (len && ((uintptr_t)p & 7)) { c = crc_tab[0][(c ^ *p++) & 0xff] ^ (c >> 8); len--; }
    while (len >= 8) {
        uint64_t v;
        std::memcpy(&v, p, 8);
        v ^= c;
        c = crc_tab[7][v & 0xff] ^ crc_tab[6][(v >> 8) & 0xff] ^ crc_tab[5][(v >> 16) & 0xff] ^
            crc_tab[4][(v >> 24) & 0xff] ^ crc_tab[3][(v >> 32) & 0xff] ^ crc_tab[2][(v >> 40) & 0xff] ^
            crc_tab[1][(v >> 48) & 0xff] ^ crc_tab[0][(v >> 56) & 0xff];
        p += 8; len -= 8;
    }
    while (len--) c = crc_tab[0][(c ^ *p++) & 0xff] ^ (c >> 8);
    return ~c;
}

uint32_t chunk_crc(const void* data, size_t len) {
    const uint8_t* p = (const uint8_t*)data;
    uint32_t acc = 0;
    for (size_t off = 0; off < len; off += 4096)
        acc ^= crc32c(p + off, std::min<size_t>(4096, len - off));
    return acc;
}

}  // namespace edgehip
