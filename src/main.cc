// edgehip CLI: mount / version / selftest subcommands.
#include <csignal>
#include <cstring>
#include <fcntl.h>
#include <malloc.h>
#include <unistd.h>
#include <vector>
#include <dirent.h>
#include <map>
#include <set>
#include <thread>

#include "manifest.h" 

#include "conf.h"
#include "journal.h"
#include "engine/engine.h"
#include "fuse.h"
#include "obs.h"
#include "vfs.h"

using namespace edgehip;

static FuseMount* g_mount = nullptr;

static void on_signal(int) {
    if (g_mount) g_mount->unmount();
}

static int usage() {
    fprintf(stderr,
            "usage: edgehip mount <mountpoint> [-o key=val,...] [-c conf_file] [-d]\n"
            "       edgehip stat <mountpoint>\n"
            "       edgehip fsck <journal_dir> [--gc]\n"
            "       edgehip mkfs <journal_dir>\n"
            "       edgehip bench <mountpoint> [size_mib [block_kib [jobs]]]\n"
            "       edgehip version\n"
            "options: see docs/options.md; presets in conf/presets/\n");
    return 2;
}

// fio-style sequential write+read through an existing mount (native driver,
// no Python overhead): reports GiB/s per phase.
static int bench_cmd(const char* mnt, long size_mib, long block_kib, int jobs) {
    size_t bs = (size_t)block_kib * 1024;
    size_t per_job = (size_t)size_mib * 1024 * 1024 / jobs;
    std::vector<double> wsec(jobs), rsec(jobs);
    std::vector<int> errs(jobs, 0);
    auto worker = [&](int j) {
        std::string path =
            std::string(mnt) + "/.edgehip-bench." + std::to_string(j) + ".tmp";
        std::vector<uint8_t> buf(bs);
        for (size_t i = 0; i < bs; i += 8) {
            uint64_t v = 0x9E3779B97F4A7C15ull * (i + 1 + j * 77);
            memcpy(&buf[i], &v, std::min((size_t)8, bs - i));
        }
        int fd = ::open(path.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0644);
        if (fd < 0) { errs[j]++; return; }
        uint64_t t0 = now_ns();
        for (size_t off = 0; off < per_job; off += bs) {
            uint64_t tag = (off + j) * 0x2545F4914F6CDD1Dull;
            memcpy(buf.data(), &tag, 8);  // vary content: dedup finds nothing
            if (::write(fd, buf.data(), bs) != (ssize_t)bs) { errs[j]++; return; }
        }
        ::fsync(fd);
        ::close(fd);
        uint64_t t1 = now_ns();
        wsec[j] = (t1 - t0) * 1e-9;
        fd = ::open(path.c_str(), O_RDONLY | O_DIRECT);
        if (fd < 0) fd = ::open(path.c_str(), O_RDONLY);
        void* rbuf = nullptr;
        if (posix_memalign(&rbuf, 4096, bs)) { errs[j]++; return; }
        t0 = now_ns();
        size_t got = 0;
        while (got < per_job) {
            ssize_t r = ::read(fd, rbuf, bs);
            if (r <= 0) { errs[j]++; break; }
            got += (size_t)r;
        }
        t1 = now_ns();
        ::close(fd);
        free(rbuf);
        rsec[j] = (t1 - t0) * 1e-9;
        ::unlink(path.c_str());
    };
    std::vector<std::thread> ts;
    for (int j = 0; j < jobs; j++) ts.emplace_back(worker, j);
    for (auto& t : ts) t.join();
    double w = 0, r = 0;
    int e = 0;
    for (int j = 0; j < jobs; j++) {
        w = std::max(w, wsec[j]);
        r = std::max(r, rsec[j]);
        e += errs[j];
    }
    if (e || w == 0 || r == 0) {
        fprintf(stderr, "bench: %d job errors\n", e);
        return 1;
    }
    double total = (double)per_job * jobs;
    printf("seq write: %.2f GiB/s (%ld MiB, %ld KiB blocks, %d jobs)\n",
           total / w / (1ull << 30), size_mib, block_kib, jobs);
    printf("seq read:  %.2f GiB/s\n", total / r / (1ull << 30));
    return 0;
}

static int fsck_cmd(const char* dir, bool gc) {
    Journal j;
    if (j.open(dir) != St::ok) return 1;
    std::vector<Journal::Replay> recs;
    if (j.replay(&recs) != St::ok) {
        fprintf(stderr, "fsck: journal replay failed (corrupt log)\n");
        return 1;
    }
    // last-wins live view: overwritten/unlinked versions legitimately freed
    // their chunks, so only the surviving manifest per path is checked
    std::map<std::string, std::vector<uint8_t>> live;
    for (auto& r : recs) {
        switch (r.type) {
            case 2:
                live[r.path] = r.manifest;
                break;
            case 3:
                live.erase(r.path);
                break;
            case 4: {
                if (!r.path.empty() && r.path[0] == '\x01') break;  // pseudo
                auto it = live.find(r.path);
                if (it != live.end()) {
                    live[r.path2] = std::move(it->second);
                    live.erase(it);
                }
                // directory rename: re-prefix every file underneath
                std::string pre = r.path + "/";
                std::vector<std::pair<std::string, std::vector<uint8_t>>> moved;
                for (auto i2 = live.begin(); i2 != live.end();) {
                    if (i2->first.rfind(pre, 0) == 0) {
                        moved.emplace_back(r.path2 + "/" + i2->first.substr(pre.size()),
                                           std::move(i2->second));
                        i2 = live.erase(i2);
                    } else {
                        ++i2;
                    }
                }
                for (auto& mv : moved) live[mv.first] = std::move(mv.second);
                break;
            }
            default:
                break;
        }
    }
    size_t files = 0, chunks = 0, missing = 0, bad_manifest = 0;
    std::set<std::string> referenced;
    for (auto& kv : live) {
        files++;
        Manifest m;
        if (Manifest::decode(kv.second.data(), kv.second.size(), &m) != St::ok) {
            fprintf(stderr, "fsck: bad manifest for %s\n", kv.first.c_str());
            bad_manifest++;
            continue;
        }
        for (auto& c : m.chunks) {
            if (c.fp.is_zero()) continue;
            chunks++;
            referenced.insert(c.fp.hex());
            std::vector<uint8_t> bytes;
            if (j.get_chunk(c.fp, &bytes) != St::ok) {
                fprintf(stderr, "fsck: %s: missing chunk %s\n", kv.first.c_str(),
                        c.fp.hex().c_str());
                missing++;
            } else if (bytes.size() != c.logical_len) {
                fprintf(stderr, "fsck: %s: chunk %s size %zu != %u\n",
                        kv.first.c_str(), c.fp.hex().c_str(), bytes.size(),
                        c.logical_len);
                missing++;
            }
        }
    }
    // GC: chunk-mirror files not referenced by any live manifest (e.g. from a
    // kill -9 between chunk write and manifest commit) are safe to remove
    size_t orphans = 0;
    std::string cdir = std::string(dir) + "/chunks";
    if (DIR* d = opendir(cdir.c_str())) {
        while (struct dirent* e = readdir(d)) {
            if (e->d_name[0] == '.') continue;
            std::string name(e->d_name);
            if (referenced.count(name)) continue;
            orphans++;
            if (gc) ::unlink((cdir + "/" + name).c_str());
        }
        closedir(d);
    }
    printf("fsck: %zu records, %zu live files, %zu chunks, %zu missing, %zu bad, "
           "%zu orphan chunk files%s\n",
           recs.size(), files, chunks, missing, bad_manifest, orphans,
           gc && orphans ? " (removed)" : "");
    return (missing || bad_manifest) ? 1 : 0;
}

int main(int argc, char** argv) {
    // Large chunk buffers (1 MiB) churn through malloc constantly; keep them on
    // the heap instead of per-allocation mmap/munmap (page-fault storms).
    mallopt(M_MMAP_THRESHOLD, 256 * 1024 * 1024);
    mallopt(M_TRIM_THRESHOLD, 256 * 1024 * 1024);
    if (argc < 2) return usage();
    std::string cmd = argv[1];
    if (cmd == "version") {
        printf("edgehip 0.1 (MI355X-native FUSE object store; gfx950)\n");
        return 0;
    }
    if (cmd == "stat") {
        if (argc < 3) return usage();
        std::string p = std::string(argv[2]) + "/.edgehip-stats";
        FILE* f = fopen(p.c_str(), "r");
        if (!f) { perror("stat"); return 1; }
        char buf[4096];
        size_t n;
        while ((n = fread(buf, 1, sizeof(buf), f)) > 0) fwrite(buf, 1, n, stdout);
        fclose(f);
        return 0;
    }
    if (cmd == "fsck") {
        if (argc < 3) return usage();
        bool gc = argc > 3 && std::string(argv[3]) == "--gc";
        return fsck_cmd(argv[2], gc);
    }
    if (cmd == "mkfs") {
        // initialize (or wipe) a journal directory for a durable mount
        if (argc < 3) return usage();
        std::string dir = argv[2];
        std::string cmd2 = "rm -rf '" + dir + "/journal.log' '" + dir + "/chunks'";
        if (system(cmd2.c_str()) != 0) LOGW("mkfs: cleanup failed (fresh dir?)");
        Journal j;
        if (j.open(dir) != St::ok) return 1;
        printf("mkfs: initialized journal at %s\n", dir.c_str());
        return 0;
    }
    if (cmd == "bench") {
        if (argc < 3) return usage();
        long mib = argc > 3 ? atol(argv[3]) : 2048;
        long bk = argc > 4 ? atol(argv[4]) : 1024;
        int jobs = argc > 5 ? atoi(argv[5]) : 4;
        return bench_cmd(argv[2], mib, bk, jobs > 0 ? jobs : 1);
    }
    if (cmd == "mount") {
        if (argc < 3) return usage();
        std::string mnt = argv[2];
        Conf conf;
        bool daemonize = false;
        if (conf.apply_env() != St::ok) return 2;
        for (int i = 3; i < argc; i++) {
            if (!strcmp(argv[i], "-o") && i + 1 < argc) {
                if (conf.apply_opts(argv[++i]) != St::ok) return 2;
            } else if (!strcmp(argv[i], "-c") && i + 1 < argc) {
                if (conf.load_file(argv[++i]) != St::ok) return 2;
            } else if (!strcmp(argv[i], "-d") || !strcmp(argv[i], "--daemon")) {
                daemonize = true;
            } else {
                return usage();
            }
        }
        if (daemonize) {
            // classic double-fork AFTER arg parsing, BEFORE engine init (HIP
            // contexts do not survive fork) — parent exits once the child is off
            pid_t pid = fork();
            if (pid < 0) { perror("fork"); return 1; }
            if (pid > 0) return 0;
            setsid();
            pid = fork();
            if (pid < 0) _exit(1);
            if (pid > 0) _exit(0);
            int devnull = ::open("/dev/null", O_RDWR);
            if (devnull >= 0) {
                dup2(devnull, 0);
                if (conf.log_level < 3) { dup2(devnull, 1); dup2(devnull, 2); }
            }
        }
        g_log_level = (LogLevel)conf.log_level;
        LOGI("conf: %s", conf.dump().c_str());
        Journal journal;
        if (!conf.journal.empty()) {
            if (journal.open(conf.journal) != St::ok) return 1;
            conf.journal_ptr = &journal;
        }
        std::unique_ptr<Engine> eng;
        try {
            eng = make_engine(conf);
        } catch (const std::exception& e) {
            LOGE("engine init failed: %s", e.what());
            return 1;
        }
        Vfs vfs(eng.get(), &conf, conf.journal_ptr);
        if (vfs.restore() != St::ok) {
            LOGE("journal replay failed");
            return 1;
        }
        FuseMount fm(&vfs, &conf);
        g_mount = &fm;
        signal(SIGINT, on_signal);
        signal(SIGTERM, on_signal);
        St s = fm.mount(mnt);
        if (s != St::ok) return 1;
        fm.wait();
        LOGI("unmounted; stats:\n%s", metrics().render().c_str());
        return 0;
    }
    return usage();
}
