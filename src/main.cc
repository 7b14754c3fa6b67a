// edgehip CLI: mount / version / selftest subcommands.
#include <csignal>
#include <cstring>
#include <malloc.h>

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
            "usage: edgehip mount <mountpoint> [-o key=val,...] [-c conf_file]\n"
            "       edgehip version\n"
            "options: see docs/options.md; presets in conf/presets/\n");
    return 2;
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
    if (cmd == "mount") {
        if (argc < 3) return usage();
        std::string mnt = argv[2];
        Conf conf;
        if (conf.apply_env() != St::ok) return 2;
        for (int i = 3; i < argc; i++) {
            if (!strcmp(argv[i], "-o") && i + 1 < argc) {
                if (conf.apply_opts(argv[++i]) != St::ok) return 2;
            } else if (!strcmp(argv[i], "-c") && i + 1 < argc) {
                if (conf.load_file(argv[++i]) != St::ok) return 2;
            } else {
                return usage();
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
