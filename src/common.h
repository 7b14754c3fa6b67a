// edgehip — MI355X-native FUSE object-storage data path.
// Common types and helpers shared by every layer.
//
// Reference parity note: the reference (Nexenta/edge-fuse, a C libfuse client
// over libccow) was unreadable at survey time; the capability
// surface implemented here is the one BASELINE.json names. No file:line
// citations are possible — formats defined here are v1-frozen (docs/FORMATS.md).
#pragma once

#include <cstdint>
#include <cstring>
#include <cstdio>
#include <cstdlib>
#include <string>
#include <vector>
#include <array>
#include <stdexcept>

namespace edgehip {

// ---------------------------------------------------------------- fingerprint
// 32-byte content fingerprint (BLAKE3-256 by default; algo byte lives in
// ChunkRef.flags so other algorithms can be added — SURVEY.md §9 Q3).
struct Fp {
    std::array<uint8_t, 32> b{};
    bool operator==(const Fp& o) const { return std::memcmp(b.data(), o.b.data(), 32) == 0; }
    bool operator!=(const Fp& o) const { return !(*this == o); }
    bool is_zero() const {
        for (auto x : b) if (x) return false;
        return true;
    }
    std::string hex() const {
        static const char* d = "0123456789abcdef";
        std::string s; s.reserve(64);
        for (auto x : b) { s.push_back(d[x >> 4]); s.push_back(d[x & 15]); }
        return s;
    }
};

struct FpHash {
    size_t operator()(const Fp& f) const {
        // fingerprints are uniformly distributed — first 8 bytes are a fine hash
        uint64_t h; std::memcpy(&h, f.b.data(), 8); return (size_t)h;
    }
};

// ---------------------------------------------------------------- error codes
// Host-side status. GPU kernel status travels separately (engine).
enum class St : int {
    ok = 0,
    not_found = -2,      // maps to -ENOENT at the FUSE boundary
    exists = -17,
    inval = -22,
    no_space = -28,
    io = -5,
    not_dir = -20,
    is_dir = -21,
    not_empty = -39,
    bad_format = -100,
    gpu = -101,
};
inline int st_errno(St s) {
    int v = (int)s;
    return (v <= -100) ? -5 /*EIO*/ : v;
}

// ---------------------------------------------------------------- logging
enum class LogLevel : int { trace = 0, debug = 1, info = 2, warn = 3, error = 4 };
extern LogLevel g_log_level;  // set from conf/env (obs.cc)
void log_line(LogLevel lvl, const char* fmt, ...) __attribute__((format(printf, 2, 3)));

#define LOGT(...) do { if (::edgehip::g_log_level <= ::edgehip::LogLevel::trace) ::edgehip::log_line(::edgehip::LogLevel::trace, __VA_ARGS__); } while (0)
#define LOGD(...) do { if (::edgehip::g_log_level <= ::edgehip::LogLevel::debug) ::edgehip::log_line(::edgehip::LogLevel::debug, __VA_ARGS__); } while (0)
#define LOGI(...) do { if (::edgehip::g_log_level <= ::edgehip::LogLevel::info)  ::edgehip::log_line(::edgehip::LogLevel::info,  __VA_ARGS__); } while (0)
#define LOGW(...) do { if (::edgehip::g_log_level <= ::edgehip::LogLevel::warn)  ::edgehip::log_line(::edgehip::LogLevel::warn,  __VA_ARGS__); } while (0)
#define LOGE(...) do { ::edgehip::log_line(::edgehip::LogLevel::error, __VA_ARGS__); } while (0)

// ---------------------------------------------------------------- misc
constexpr uint64_t KiB = 1024ull;
constexpr uint64_t MiB = 1024ull * 1024;
constexpr uint64_t GiB = 1024ull * 1024 * 1024;

inline uint64_t div_up(uint64_t a, uint64_t b) { return (a + b - 1) / b; }
inline uint64_t round_up(uint64_t a, uint64_t b) { return div_up(a, b) * b; }

uint64_t now_ns();   // CLOCK_MONOTONIC
uint64_t wall_ns();  // CLOCK_REALTIME

// crc32c (Castagnoli, SSE4.2 when available, table fallback) — manifest trailer
uint32_t crc32c(const void* data, size_t len, uint32_t seed = 0);

// ChunkRef.crc: XOR of crc32c over each 4 KiB page of the stored bytes
// (page-parallel on the GPU, docs/FORMATS.md §checksum).
uint32_t chunk_crc(const void* data, size_t len);

}  // namespace edgehip
