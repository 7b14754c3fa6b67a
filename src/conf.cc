#include "conf.h"

#include <cctype>
#include <cstring>
#include <fstream>
#include <sstream>

extern char** environ;

namespace edgehip {

static bool parse_bool(const std::string& v, bool* out) {
    if (v == "1" || v == "true" || v == "on" || v == "yes" || v.empty()) { *out = true; return true; }
    if (v == "0" || v == "false" || v == "off" || v == "no") { *out = false; return true; }
    return false;
}

// sizes accept suffixes k/m/g/t (binary)
static bool parse_size(const std::string& v, uint64_t* out) {
    if (v.empty()) return false;
    char* end = nullptr;
    double x = strtod(v.c_str(), &end);
    if (end == v.c_str() || x < 0) return false;
    uint64_t mult = 1;
    if (*end) {
        switch (tolower(*end)) {
            case 'k': mult = KiB; break;
            case 'm': mult = MiB; break;
            case 'g': mult = GiB; break;
            case 't': mult = GiB * 1024; break;
            default: return false;
        }
        if (end[1] && !(tolower(end[1]) == 'i' && tolower(end[2]) == 'b' && !end[3]) &&
            !(tolower(end[1]) == 'b' && !end[2]))
            return false;
    }
    *out = (uint64_t)(x * (double)mult);
    return true;
}

static bool parse_int(const std::string& v, int* out) {
    char* end = nullptr;
    long x = strtol(v.c_str(), &end, 10);
    if (end == v.c_str() || *end) return false;
    *out = (int)x;
    return true;
}

St Conf::apply_kv(const std::string& k, const std::string& v) {
    auto bad = [&]() { LOGE("conf: bad value for %s: '%s'", k.c_str(), v.c_str()); return St::inval; };
    if (k == "chunk_size") return parse_size(v, &chunk_size) && chunk_size >= 4 * KiB ? St::ok : bad();
    if (k == "cdc") return parse_bool(v, &cdc) ? St::ok : bad();
    if (k == "fp") {
        if (v == "none") fp = FpAlgo::none;
        else if (v == "blake3") fp = FpAlgo::blake3;
        else return bad();
        return St::ok;
    }
    if (k == "compress") {
        if (v == "none") compress = Compress::none;
        else if (v == "lz4") compress = Compress::lz4;
        else return bad();
        return St::ok;
    }
    if (k == "dedup") return parse_bool(v, &dedup) ? St::ok : bad();
    if (k == "verify_read") return parse_bool(v, &verify_read) ? St::ok : bad();
    if (k == "clone") return parse_bool(v, &fuse_clone) ? St::ok : bad();
    if (k == "crc") return parse_bool(v, &crc) ? St::ok : bad();
    if (k == "prealloc") return parse_size(v, &prealloc) ? St::ok : bad();
    if (k == "rs") {  // "6+2" or "0"
        if (v == "0" || v == "off") { rs_k = rs_m = 0; return St::ok; }
        size_t plus = v.find('+');
        if (plus == std::string::npos) return bad();
        int kk, mm;
        if (!parse_int(v.substr(0, plus), &kk) || !parse_int(v.substr(plus + 1), &mm)) return bad();
        if (kk < 1 || kk > 32 || mm < 0 || mm > 8) return bad();
        rs_k = kk; rs_m = mm;
        return St::ok;
    }
    if (k == "engine") {
        if (v == "host") engine = EngineKind::host;
        else if (v == "gpu") engine = EngineKind::gpu;
        else return bad();
        return St::ok;
    }
    if (k == "ngpus") return parse_int(v, &ngpus) && ngpus >= 1 && ngpus <= 64 ? St::ok : bad();
    if (k == "gpu_first") return parse_int(v, &gpu_first) && gpu_first >= 0 ? St::ok : bad();
    if (k == "arena") return parse_size(v, &arena_bytes) ? St::ok : bad();
    if (k == "pinned") return parse_size(v, &pinned_bytes) ? St::ok : bad();
    if (k == "spill") return parse_size(v, &spill_bytes) ? St::ok : bad();
    if (k == "dist_table_slots")
        return parse_size(v, &dist_table_slots) ? St::ok : bad();
    if (k == "spill_dir") { spill_dir = v; return St::ok; }
    if (k == "threads") return parse_int(v, &fuse_threads) && fuse_threads >= 0 && fuse_threads <= 128 ? St::ok : bad();
    if (k == "max_write") {
        uint64_t x;
        if (!parse_size(v, &x) || x < 4 * KiB || x > 16 * MiB) return bad();
        max_write = (uint32_t)x;
        return St::ok;
    }
    if (k == "writeback") return parse_bool(v, &writeback) ? St::ok : bad();
    if (k == "read_direct") return parse_bool(v, &read_direct) ? St::ok : bad();
    if (k == "direct_io") return parse_bool(v, &direct_io) ? St::ok : bad();
    if (k == "kernel_cache") return parse_bool(v, &kernel_cache) ? St::ok : bad();
    if (k == "log") {
        if (!parse_int(v, &log_level) || log_level < 0 || log_level > 4) return bad();
        g_log_level = (LogLevel)log_level;
        return St::ok;
    }
    if (k == "journal") { journal = v; return St::ok; }
    if (k == "fault") { fault = v; return St::ok; }
    // standard mount options we accept and ignore (mount(8) may add them)
    if (k == "rw" || k == "ro" || k == "nosuid" || k == "nodev" || k == "noexec" ||
        k == "relatime" || k == "noatime" || k == "default_permissions" || k == "allow_other")
        return St::ok;
    LOGE("conf: unknown option '%s'", k.c_str());
    return St::inval;
}

St Conf::apply_opts(const std::string& opts) {
    size_t pos = 0;
    while (pos < opts.size()) {
        size_t comma = opts.find(',', pos);
        if (comma == std::string::npos) comma = opts.size();
        std::string kv = opts.substr(pos, comma - pos);
        pos = comma + 1;
        if (kv.empty()) continue;
        size_t eq = kv.find('=');
        std::string k = kv.substr(0, eq == std::string::npos ? kv.size() : eq);
        std::string v = eq == std::string::npos ? "" : kv.substr(eq + 1);
        St s = apply_kv(k, v);
        if (s != St::ok) return s;
    }
    return St::ok;
}

St Conf::apply_env() {
    for (char** e = environ; *e; e++) {
        const char* s = *e;
        if (strncmp(s, "EDGEHIP_", 8) != 0) continue;
        const char* eq = strchr(s, '=');
        if (!eq) continue;
        std::string key(s + 8, eq - s - 8);
        for (auto& c : key) c = (char)tolower(c);
        if (key == "fault_seed") continue;  // consumed by obs
        St st = apply_kv(key, eq + 1);
        if (st != St::ok) return st;
    }
    return St::ok;
}

St Conf::load_file(const std::string& path) {
    std::ifstream f(path);
    if (!f) { LOGE("conf: cannot open %s", path.c_str()); return St::not_found; }
    std::string line;
    while (std::getline(f, line)) {
        size_t h = line.find('#');
        if (h != std::string::npos) line = line.substr(0, h);
        // trim
        size_t a = line.find_first_not_of(" \t\r");
        if (a == std::string::npos) continue;
        size_t b = line.find_last_not_of(" \t\r");
        line = line.substr(a, b - a + 1);
        size_t eq = line.find('=');
        if (eq == std::string::npos) { LOGE("conf: bad line '%s'", line.c_str()); return St::inval; }
        auto strip = [](std::string s) {
            size_t a2 = s.find_first_not_of(" \t");
            size_t b2 = s.find_last_not_of(" \t");
            return a2 == std::string::npos ? std::string() : s.substr(a2, b2 - a2 + 1);
        };
        St st = apply_kv(strip(line.substr(0, eq)), strip(line.substr(eq + 1)));
        if (st != St::ok) return st;
    }
    return St::ok;
}

std::string Conf::dump() const {
    std::ostringstream o;
    o << "chunk_size=" << chunk_size
      << " fp=" << (fp == FpAlgo::none ? "none" : "blake3")
      << " compress=" << (compress == Compress::none ? "none" : "lz4")
      << " dedup=" << dedup << " rs=" << rs_k << "+" << rs_m
      << " engine=" << (engine == EngineKind::host ? "host" : "gpu")
      << " ngpus=" << ngpus << " arena=" << arena_bytes
      << " threads=" << fuse_threads << " max_write=" << max_write
      << " writeback=" << writeback;
    return o.str();
}

}  // namespace edgehip
