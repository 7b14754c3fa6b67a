#include "fuse.h"

#include <algorithm>
#include <errno.h>
#include <fcntl.h>
#include <linux/fuse.h>
#include <poll.h>
#include <sys/ioctl.h>
#include <sys/stat.h>
#include <sys/mount.h>
#include <sys/sysmacros.h>
#include <sys/uio.h>
#include <unistd.h>

#include "obs.h"

namespace edgehip {

namespace {

constexpr uint64_t kAttrValidSec = 1;
constexpr uint64_t kEntryValidSec = 1;
// hidden virtual file /.edgehip-stats exposing live counters
constexpr uint64_t kStatsIno = 0x7FFFFFFFFFFFFFF0ull;
constexpr const char* kStatsName = ".edgehip-stats";

void fill_attr(const Inode* in, fuse_attr* a) {
    std::memset(a, 0, sizeof(*a));
    a->ino = in->ino;
    a->size = S_ISLNK(in->mode) ? in->symlink_target.size() : in->size();
    a->blocks = div_up(a->size, 512);
    a->atime = (uint64_t)(in->atime / 1000000000ll);
    a->mtime = (uint64_t)(in->mtime / 1000000000ll);
    a->ctime = (uint64_t)(in->ctime / 1000000000ll);
    a->atimensec = (uint32_t)(in->atime % 1000000000ll);
    a->mtimensec = (uint32_t)(in->mtime % 1000000000ll);
    a->ctimensec = (uint32_t)(in->ctime % 1000000000ll);
    a->mode = in->mode;
    a->nlink = in->nlink;
    a->uid = in->uid;
    a->gid = in->gid;
    a->blksize = 1u << 20;  // advertise 1 MiB preferred I/O
}

void fill_entry(const Inode* in, fuse_entry_out* e) {
    std::memset(e, 0, sizeof(*e));
    e->nodeid = in->ino;
    e->generation = 1;
    e->entry_valid = kEntryValidSec;
    e->attr_valid = kAttrValidSec;
    fill_attr(in, &e->attr);
}

}  // namespace

FuseMount::~FuseMount() { unmount(); }

St FuseMount::mount(const std::string& mountpoint) {
    fd_ = ::open("/dev/fuse", O_RDWR | O_CLOEXEC);
    if (fd_ < 0 && errno == ENOENT) {
        // GPU boxes ship without the device node; create it (char 10:229).
        if (mknod("/dev/fuse", S_IFCHR | 0600, makedev(10, 229)) == 0)
            fd_ = ::open("/dev/fuse", O_RDWR | O_CLOEXEC);
    }
    if (fd_ < 0) {
        LOGE("fuse: open /dev/fuse failed: %s", strerror(errno));
        return St::io;
    }
    char opts[256];
    snprintf(opts, sizeof(opts), "fd=%d,rootmode=40000,user_id=0,group_id=0,max_read=%u",
             fd_, conf_->max_write);
    unsigned long mflags = MS_NOSUID | MS_NODEV | MS_NOATIME;
    if (::mount("edgehip", mountpoint.c_str(), "fuse.edgehip", mflags, opts)) {
        LOGE("fuse: mount(%s) failed: %s", mountpoint.c_str(), strerror(errno));
        ::close(fd_);
        fd_ = -1;
        return St::io;
    }
    mountpoint_ = mountpoint;

    // --- INIT handshake (synchronous, before worker threads start) ---
    std::vector<uint8_t> buf(64 * KiB);
    ssize_t n = ::read(fd_, buf.data(), buf.size());
    if (n < (ssize_t)(sizeof(fuse_in_header) + sizeof(fuse_init_in))) {
        LOGE("fuse: bad INIT read (%zd): %s", n, strerror(errno));
        unmount();
        return St::io;
    }
    auto* h = (fuse_in_header*)buf.data();
    auto* ii = (fuse_init_in*)(buf.data() + sizeof(fuse_in_header));
    if (h->opcode != FUSE_INIT) {
        LOGE("fuse: first request not INIT (op %u)", h->opcode);
        unmount();
        return St::io;
    }
    kernel_minor_ = ii->minor;
    struct {
        fuse_out_header h;
        fuse_init_out o;
    } rep;
    std::memset(&rep, 0, sizeof(rep));
    rep.h.len = sizeof(rep);
    rep.h.unique = h->unique;
    rep.o.major = FUSE_KERNEL_VERSION;          // 7
    rep.o.minor = FUSE_KERNEL_MINOR_VERSION;    // 34 (header's); kernel negotiates min
    rep.o.max_readahead = conf_->max_readahead;
    uint32_t want = FUSE_ASYNC_READ | FUSE_BIG_WRITES | FUSE_PARALLEL_DIROPS |
                    FUSE_DO_READDIRPLUS | FUSE_READDIRPLUS_AUTO |
                    FUSE_MAX_PAGES | FUSE_AUTO_INVAL_DATA |
                    FUSE_CACHE_SYMLINKS |       // kernel caches readlink results
                    FUSE_NO_OPENDIR_SUPPORT;    // our OPENDIR is stateless
    if (conf_->writeback) want |= FUSE_WRITEBACK_CACHE;
    rep.o.flags = want & ii->flags;
    rep.o.max_background = 64;
    rep.o.congestion_threshold = 48;
    rep.o.max_write = conf_->max_write;
    rep.o.time_gran = 1;
    rep.o.max_pages = (uint16_t)div_up(conf_->max_write, 4096);
    if (::write(fd_, &rep, sizeof(rep)) != (ssize_t)sizeof(rep)) {
        LOGE("fuse: INIT reply failed: %s", strerror(errno));
        unmount();
        return St::io;
    }
    // The kernel clamps the fuse bdi's readahead to min(128 KiB, max_readahead);
    // raising it needs an explicit write to the bdi knob. Without this, buffered
    // sequential reads arrive as 128 KiB requests instead of max_write-sized
    // ones (measured 5-7x slower).
    // (Never stat() through the mount here — no reader threads yet => deadlock.
    // The device number comes from /proc/self/mountinfo instead.)
    {
        FILE* mi = fopen("/proc/self/mountinfo", "r");
        char line[1024];
        char devnum[32] = {0};
        while (mi && fgets(line, sizeof(line), mi)) {
            // fields: id parent major:minor root mountpoint ...
            unsigned id, parent, maj, min;
            char root[256], mp[512];
            if (sscanf(line, "%u %u %u:%u %255s %511s", &id, &parent, &maj, &min, root, mp) == 6 &&
                mountpoint == mp) {
                snprintf(devnum, sizeof(devnum), "%u:%u", maj, min);
            }
        }
        if (mi) fclose(mi);
        if (devnum[0]) {
            auto bdi_set = [&](const char* knob, const char* val) {
                char path[128];
                snprintf(path, sizeof(path), "/sys/class/bdi/%s/%s", devnum, knob);
                int bfd = ::open(path, O_WRONLY);
                if (bfd < 0) {
                    LOGW("fuse: cannot open %s: %s", path, strerror(errno));
                    return;
                }
                if (::write(bfd, val, strlen(val)) != (ssize_t)strlen(val))
                    LOGW("fuse: bdi %s=%s failed: %s", knob, val, strerror(errno));
                ::close(bfd);
            };
            char ra[32];
            snprintf(ra, sizeof(ra), "%u", conf_->max_readahead / 1024);
            bdi_set("read_ahead_kb", ra);
            // fuse bdis default to strict dirty limiting at max_ratio=1% of
            // dirtyable memory, which throttles buffered writes to ~0.15 GB/s
            // regardless of filesystem speed; lift both.
            bdi_set("max_ratio", "90");
            bdi_set("strict_limit", "0");
        } else {
            LOGW("fuse: bdi for %s not found in mountinfo", mountpoint.c_str());
        }
    }
    // threads=0 -> auto-size by cores: each reader thread is a full request
    // pipeline (read(2) copy + handle + reply) and oversubscription collapses
    // throughput (measured 2.5x on an 8-core box); half the cores leaves the
    // other half for flush workers / engine threads.
    int nthreads = conf_->fuse_threads;
    if (nthreads <= 0) {
        long np = sysconf(_SC_NPROCESSORS_ONLN);
        nthreads = (int)std::min(12l, std::max(4l, np / 2));
    }
    LOGI("fuse: mounted %s (kernel 7.%u, max_write %u, flags 0x%x, %d threads)",
         mountpoint.c_str(), kernel_minor_, conf_->max_write, rep.o.flags, nthreads);
    mounted_ = true;
    stop_ = false;
    for (int i = 0; i < nthreads; i++)
        threads_.emplace_back([this, i] { reader_loop(i); });
    return St::ok;
}

void FuseMount::unmount() {
    if (!mountpoint_.empty()) {
        ::umount2(mountpoint_.c_str(), MNT_DETACH);
        mountpoint_.clear();
    }
    stop_ = true;
    for (auto& t : threads_) t.join();
    threads_.clear();
    if (fd_ >= 0) {
        ::close(fd_);
        fd_ = -1;
    }
    mounted_ = false;
}

void FuseMount::wait() {
    for (auto& t : threads_) t.join();
    threads_.clear();
    mounted_ = false;
}

// Per-thread device channel: FUSE_DEV_IOC_CLONE gives each reader its own
// fd sharing the connection — each clone has its own processing queue and
// lock, so threads stop contending on one fd's queue (SURVEY H1's
// multi-channel plan). Falls back to the shared fd if the ioctl refuses.
int FuseMount::clone_channel() {
    int nfd = ::open("/dev/fuse", O_RDWR | O_CLOEXEC);
    if (nfd < 0) return -1;
    uint32_t src = (uint32_t)fd_;
    if (::ioctl(nfd, FUSE_DEV_IOC_CLONE, &src)) {
        ::close(nfd);
        return -1;
    }
    return nfd;
}

void FuseMount::reader_loop(int tid) {
    size_t bufsz = (size_t)conf_->max_write + 128 * KiB;
    std::vector<uint8_t> buf(bufsz);
    std::vector<uint8_t> reply;
    reply.reserve((size_t)conf_->max_write + 4 * KiB);
    int own = conf_->fuse_clone ? clone_channel() : -1;
    int ch = own >= 0 ? own : fd_;
    if (tid == 0)
        LOGI("fuse: reader channels: %s", own >= 0 ? "per-thread clones" : "shared fd");
    while (!stop_.load(std::memory_order_relaxed)) {
        ssize_t n = ::read(ch, buf.data(), bufsz);
        if (n < 0) {
            if (errno == EINTR || errno == EAGAIN || errno == ENOENT) continue;
            if (errno == ENODEV) break;  // unmounted
            LOGE("fuse: read error: %s", strerror(errno));
            break;
        }
        if (n < (ssize_t)sizeof(fuse_in_header)) continue;
        handle(buf.data(), (size_t)n, &reply, ch);
    }
    if (own >= 0) ::close(own);
    mounted_ = false;
}

void FuseMount::handle(const uint8_t* req, size_t len, std::vector<uint8_t>* reply_buf,
                       int ch_fd) {
    auto* h = (const fuse_in_header*)req;
    const uint8_t* arg = req + sizeof(fuse_in_header);
    metrics().fuse_reqs++;
    LOGD("fuse: op=%u nodeid=%lu unique=%lu len=%zu", h->opcode,
         (unsigned long)h->nodeid, (unsigned long)h->unique, len);

    auto reply_err = [&](int neg_errno) {
        fuse_out_header oh{};
        oh.len = sizeof(oh);
        oh.error = neg_errno;
        oh.unique = h->unique;
        ssize_t w = ::write(ch_fd, &oh, sizeof(oh));
        (void)w;
    };
    auto reply_body = [&](const void* body, size_t blen) {
        fuse_out_header oh{};
        oh.len = (uint32_t)(sizeof(oh) + blen);
        oh.unique = h->unique;
        iovec iov[2] = {{&oh, sizeof(oh)}, {(void*)body, blen}};
        ssize_t w = ::writev(ch_fd, iov, blen ? 2 : 1);
        (void)w;
    };
    auto reply_st = [&](St s) { reply_err(st_errno(s)); };

    switch (h->opcode) {
        case FUSE_LOOKUP: {
            std::string name((const char*)arg);
            if (h->nodeid == 1 && name == kStatsName) {
                fuse_entry_out e{};
                e.nodeid = kStatsIno;
                e.generation = 1;
                e.attr.ino = kStatsIno;
                e.attr.mode = S_IFREG | 0444;
                e.attr.nlink = 1;
                e.attr.size = metrics().render().size();
                e.attr.blksize = 4096;
                reply_body(&e, sizeof(e));
                return;
            }
            Inode* in;
            St s = vfs_->lookup(h->nodeid, name, &in);
            if (s != St::ok) { reply_st(s); return; }
            fuse_entry_out e;
            fill_entry(in, &e);
            reply_body(&e, sizeof(e));
            return;
        }
        case FUSE_FORGET: {
            auto* f = (const fuse_forget_in*)arg;
            vfs_->forget(h->nodeid, f->nlookup);
            return;  // no reply
        }
        case FUSE_BATCH_FORGET: {
            auto* bf = (const fuse_batch_forget_in*)arg;
            auto* one = (const fuse_forget_one*)(arg + sizeof(*bf));
            for (uint32_t i = 0; i < bf->count; i++) vfs_->forget(one[i].nodeid, one[i].nlookup);
            return;  // no reply
        }
        case FUSE_GETATTR: {
            if (h->nodeid == kStatsIno) {
                fuse_attr_out ao{};
                ao.attr.ino = kStatsIno;
                ao.attr.mode = S_IFREG | 0444;
                ao.attr.nlink = 1;
                ao.attr.size = metrics().render().size();
                ao.attr.blksize = 4096;
                reply_body(&ao, sizeof(ao));
                return;
            }
            Inode* in;
            St s = vfs_->getattr(h->nodeid, &in);
            if (s != St::ok) { reply_st(s); return; }
            fuse_attr_out ao{};
            ao.attr_valid = kAttrValidSec;
            fill_attr(in, &ao.attr);
            reply_body(&ao, sizeof(ao));
            return;
        }
        case FUSE_SETATTR: {
            auto* sa = (const fuse_setattr_in*)arg;
            Inode* in;
            St s = vfs_->getattr(h->nodeid, &in);
            if (s != St::ok) { reply_st(s); return; }
            if (sa->valid & FATTR_SIZE) {
                s = vfs_->setattr_size(h->nodeid, sa->size);
                if (s != St::ok) { reply_st(s); return; }
            }
            {
                std::lock_guard<std::mutex> g(vfs_->lock());
                if (sa->valid & FATTR_MODE) in->mode = (in->mode & S_IFMT) | (sa->mode & 07777);
                if (sa->valid & FATTR_UID) in->uid = sa->uid;
                if (sa->valid & FATTR_GID) in->gid = sa->gid;
                if (sa->valid & FATTR_MTIME)
                    in->mtime = (int64_t)sa->mtime * 1000000000ll + sa->mtimensec;
                if (sa->valid & FATTR_ATIME)
                    in->atime = (int64_t)sa->atime * 1000000000ll + sa->atimensec;
            }
            fuse_attr_out ao{};
            ao.attr_valid = kAttrValidSec;
            fill_attr(in, &ao.attr);
            reply_body(&ao, sizeof(ao));
            return;
        }
        case FUSE_MKNOD: {
            auto* m = (const fuse_mknod_in*)arg;
            if (!S_ISREG(m->mode)) { reply_err(-ENOSYS); return; }
            std::string name((const char*)(arg + sizeof(*m)));
            Inode* in;
            St s = vfs_->create(h->nodeid, name, m->mode & 07777, h->uid, h->gid, &in);
            if (s != St::ok) { reply_st(s); return; }
            fuse_entry_out e;
            fill_entry(in, &e);
            reply_body(&e, sizeof(e));
            return;
        }
        case FUSE_MKDIR: {
            auto* m = (const fuse_mkdir_in*)arg;
            std::string name((const char*)(arg + sizeof(*m)));
            Inode* in;
            St s = vfs_->mkdir(h->nodeid, name, m->mode, h->uid, h->gid, &in);
            if (s != St::ok) { reply_st(s); return; }
            fuse_entry_out e;
            fill_entry(in, &e);
            reply_body(&e, sizeof(e));
            return;
        }
        case FUSE_UNLINK: {
            std::string name((const char*)arg);
            reply_st(vfs_->unlink(h->nodeid, name));
            return;
        }
        case FUSE_RMDIR: {
            std::string name((const char*)arg);
            reply_st(vfs_->rmdir(h->nodeid, name));
            return;
        }
        case FUSE_RENAME: {
            auto* r = (const fuse_rename_in*)arg;
            const char* oldname = (const char*)(arg + sizeof(*r));
            const char* newname = oldname + strlen(oldname) + 1;
            reply_st(vfs_->rename(h->nodeid, oldname, r->newdir, newname, false));
            return;
        }
        case FUSE_RENAME2: {
            auto* r = (const fuse_rename2_in*)arg;
            const char* oldname = (const char*)(arg + sizeof(*r));
            const char* newname = oldname + strlen(oldname) + 1;
            if (r->flags & ~3u) { reply_err(-EINVAL); return; }  // no WHITEOUT
            if (r->flags & 2u) {  // RENAME_EXCHANGE
                reply_st(vfs_->exchange(h->nodeid, oldname, r->newdir, newname));
                return;
            }
            bool noreplace = r->flags & 1u;  // RENAME_NOREPLACE
            reply_st(vfs_->rename(h->nodeid, oldname, r->newdir, newname, noreplace));
            return;
        }
        case FUSE_OPEN: {
            if (h->nodeid == kStatsIno) {
                fuse_open_out oo{};
                oo.fh = 0;
                oo.open_flags = FOPEN_DIRECT_IO;  // size is dynamic
                reply_body(&oo, sizeof(oo));
                return;
            }
            uint64_t fh;
            St s = vfs_->open(h->nodeid, &fh);
            if (s != St::ok) { reply_st(s); return; }
            fuse_open_out oo{};
            oo.fh = fh;
            if (conf_->direct_io) oo.open_flags |= FOPEN_DIRECT_IO;
            else if (conf_->kernel_cache) oo.open_flags |= FOPEN_KEEP_CACHE;
            reply_body(&oo, sizeof(oo));
            return;
        }
        case FUSE_CREATE: {
            auto* c = (const fuse_create_in*)arg;
            std::string name((const char*)(arg + sizeof(*c)));
            Inode* in;
            St s = vfs_->create(h->nodeid, name, c->mode & 07777, h->uid, h->gid, &in);
            if (s == St::exists) {
                // O_CREAT without O_EXCL on existing file: open it
                if (!(c->flags & O_EXCL)) s = vfs_->lookup(h->nodeid, name, &in);
            }
            if (s != St::ok) { reply_st(s); return; }
            uint64_t fh;
            s = vfs_->open(in->ino, &fh);
            if (s != St::ok) { reply_st(s); return; }
            if (c->flags & O_TRUNC) vfs_->setattr_size(in->ino, 0);
            struct {
                fuse_entry_out e;
                fuse_open_out o;
            } rep;
            std::memset(&rep, 0, sizeof(rep));
            fill_entry(in, &rep.e);
            rep.o.fh = fh;
            if (conf_->direct_io) rep.o.open_flags |= FOPEN_DIRECT_IO;
            reply_body(&rep, sizeof(rep));
            return;
        }
        case FUSE_READ: {
            auto* r = (const fuse_read_in*)arg;
            if (h->nodeid == kStatsIno) {
                std::string s = metrics().render();
                size_t off = std::min((size_t)r->offset, s.size());
                size_t n2 = std::min((size_t)r->size, s.size() - off);
                reply_body(s.data() + off, n2);
                return;
            }
            reply_buf->resize(r->size);
            uint64_t got = 0;
            uint64_t t0 = now_ns();
            St s = vfs_->read(r->fh, r->offset, r->size, reply_buf->data(), &got);
            metrics().read_lat.record(now_ns() - t0);
            if (s != St::ok) { reply_st(s); return; }
            reply_body(reply_buf->data(), got);
            return;
        }
        case FUSE_WRITE: {
            auto* w = (const fuse_write_in*)arg;
            const uint8_t* data = arg + sizeof(*w);
            if (sizeof(fuse_in_header) + sizeof(*w) + w->size > len) { reply_err(-EINVAL); return; }
            uint64_t t0 = now_ns();
            St s = vfs_->write(w->fh, w->offset, data, w->size);
            metrics().write_lat.record(now_ns() - t0);
            if (s != St::ok) { reply_st(s); return; }
            fuse_write_out wo{};
            wo.size = w->size;
            reply_body(&wo, sizeof(wo));
            return;
        }
        case FUSE_FLUSH: {
            if (h->nodeid == kStatsIno) {
                reply_err(0);
                return;
            }
            auto* f = (const fuse_flush_in*)arg;
            reply_st(vfs_->flush(f->fh));
            return;
        }
        case FUSE_RELEASE: {
            if (h->nodeid == kStatsIno) {
                reply_err(0);
                return;
            }
            auto* r = (const fuse_release_in*)arg;
            reply_st(vfs_->release(r->fh));
            return;
        }
        case FUSE_FSYNC: {
            auto* f = (const fuse_fsync_in*)arg;
            reply_st(vfs_->fsync(f->fh));
            return;
        }
        case FUSE_OPENDIR: {
            fuse_open_out oo{};
            oo.fh = 0;
            reply_body(&oo, sizeof(oo));
            return;
        }
        case FUSE_READDIR: {
            auto* r = (const fuse_read_in*)arg;
            std::vector<DirEntry> ents;
            St s = vfs_->readdir(h->nodeid, &ents);
            if (s != St::ok) { reply_st(s); return; }
            reply_buf->clear();
            uint64_t idx = r->offset;  // cookie = next entry index
            while (idx < ents.size()) {
                const DirEntry& de = ents[idx];
                size_t entlen = FUSE_NAME_OFFSET + de.name.size();
                size_t entlen_aligned = FUSE_DIRENT_ALIGN(entlen);
                if (reply_buf->size() + entlen_aligned > r->size) break;
                size_t base = reply_buf->size();
                reply_buf->resize(base + entlen_aligned, 0);
                auto* d = (fuse_dirent*)(reply_buf->data() + base);
                d->ino = de.ino;
                d->off = idx + 1;
                d->namelen = (uint32_t)de.name.size();
                d->type = de.type;
                std::memcpy(d->name, de.name.data(), de.name.size());
                idx++;
            }
            reply_body(reply_buf->data(), reply_buf->size());
            return;
        }
        case FUSE_READDIRPLUS: {
            // READDIR + LOOKUP in one trip (FUSE_DO_READDIRPLUS): each entry
            // carries a full fuse_entry_out, and the kernel takes an nlookup
            // reference exactly as if it had sent a LOOKUP — except for
            // "." / "..", which go out with nodeid 0 (no reference).
            auto* r = (const fuse_read_in*)arg;
            std::vector<DirEntry> ents;
            St s = vfs_->readdir(h->nodeid, &ents);
            if (s != St::ok) { reply_st(s); return; }
            reply_buf->clear();
            uint64_t idx = r->offset;
            while (idx < ents.size()) {
                const DirEntry& de = ents[idx];
                size_t entlen = FUSE_NAME_OFFSET_DIRENTPLUS + de.name.size();
                size_t entlen_aligned = FUSE_DIRENT_ALIGN(entlen);
                if (reply_buf->size() + entlen_aligned > r->size) break;
                size_t base = reply_buf->size();
                reply_buf->resize(base + entlen_aligned, 0);
                auto* dp = (fuse_direntplus*)(reply_buf->data() + base);
                bool dotty = de.name == "." || de.name == "..";
                Inode* in = nullptr;
                if (!dotty && vfs_->getattr(de.ino, &in) == St::ok) {
                    fill_entry(in, &dp->entry_out);
                    vfs_->ref(in);  // kernel holds it until FORGET
                }
                dp->dirent.ino = de.ino;
                dp->dirent.off = idx + 1;
                dp->dirent.namelen = (uint32_t)de.name.size();
                dp->dirent.type = de.type;
                std::memcpy(dp->dirent.name, de.name.data(), de.name.size());
                idx++;
            }
            reply_body(reply_buf->data(), reply_buf->size());
            return;
        }
        case FUSE_RELEASEDIR:
        case FUSE_FSYNCDIR:
            reply_err(0);
            return;
        case FUSE_STATFS: {
            fuse_statfs_out so{};
            so.st.bsize = 4096;
            so.st.frsize = 4096;
            uint64_t cap = 1ull << 41;  // 2 TiB logical
            uint64_t used = vfs_->engine()->stored_bytes();
            so.st.blocks = cap / 4096;
            so.st.bfree = so.st.bavail = (cap - std::min(cap, used)) / 4096;
            so.st.files = 1u << 20;
            so.st.ffree = (1u << 20) - vfs_->inode_count();
            so.st.namelen = 255;
            reply_body(&so, sizeof(so));
            return;
        }
        case FUSE_ACCESS:
            reply_err(0);
            return;
        case FUSE_FALLOCATE: {
            auto* f = (const fuse_fallocate_in*)arg;
            Inode* in = vfs_->handle_inode(f->fh);
            if (!in) { reply_err(-EBADF); return; }
            if (f->mode == 0) {  // reserve: extend the logical size
                uint64_t end = f->offset + f->length;
                if (end > in->size()) {
                    St s = vfs_->setattr_size(in->ino, end);
                    if (s != St::ok) { reply_st(s); return; }
                }
                reply_err(0);
                return;
            }
            // FALLOC_FL_PUNCH_HOLE (1) requires KEEP_SIZE (2)
            if (f->mode == 0x3) {
                if (!in->obj) { reply_err(-EBADF); return; }
                reply_st(in->obj->punch_hole(f->offset, f->length));
                return;
            }
            reply_err(-EOPNOTSUPP);
            return;
        }
        case FUSE_LSEEK: {
            auto* l = (const fuse_lseek_in*)arg;
            Inode* in = vfs_->handle_inode(l->fh);
            if (!in) { reply_err(-EBADF); return; }
            // SEEK_DATA=3, SEEK_HOLE=4 — chunk-granular: punched/never-
            // written chunks are holes, everything else is data.
            uint64_t sz = in->size();
            fuse_lseek_out lo{};
            if (l->whence == 3) {
                if (l->offset >= sz) { reply_err(-ENXIO); return; }
                lo.offset = in->obj ? in->obj->next_boundary(l->offset, true)
                                    : l->offset;
                if (lo.offset >= sz) { reply_err(-ENXIO); return; }
            } else if (l->whence == 4) {
                if (l->offset >= sz) { reply_err(-ENXIO); return; }
                lo.offset = in->obj ? in->obj->next_boundary(l->offset, false)
                                    : sz;
            } else {
                reply_err(-EINVAL);
                return;
            }
            reply_body(&lo, sizeof(lo));
            return;
        }
        case FUSE_INTERRUPT:
            return;  // no reply
        case FUSE_DESTROY:
            reply_err(0);
            stop_ = true;
            return;
        case FUSE_SYMLINK: {
            const char* name = (const char*)arg;
            const char* target = name + strlen(name) + 1;
            Inode* in;
            St s = vfs_->symlink(h->nodeid, name, target, h->uid, h->gid, &in);
            if (s != St::ok) { reply_st(s); return; }
            fuse_entry_out e;
            fill_entry(in, &e);
            reply_body(&e, sizeof(e));
            return;
        }
        case FUSE_READLINK: {
            std::string t;
            St s = vfs_->readlink(h->nodeid, &t);
            if (s != St::ok) { reply_st(s); return; }
            reply_body(t.data(), t.size());
            return;
        }
        case FUSE_LINK: {
            auto* l = (const fuse_link_in*)arg;
            std::string name((const char*)(arg + sizeof(*l)));
            Inode* in;
            St s = vfs_->link(l->oldnodeid, h->nodeid, name, &in);
            if (s != St::ok) { reply_st(s); return; }
            fuse_entry_out e;
            fill_entry(in, &e);
            reply_body(&e, sizeof(e));
            return;
        }
        case FUSE_SETXATTR: {
            // no FUSE_SETXATTR_EXT negotiated -> 8-byte compat header
            auto* x = (const fuse_setxattr_in*)arg;
            const char* name = (const char*)arg + FUSE_COMPAT_SETXATTR_IN_SIZE;
            const char* val = name + strlen(name) + 1;
            St s = vfs_->setxattr(h->nodeid, name, std::string(val, x->size),
                                  x->flags);
            if (s == St::not_found) { reply_err(-ENODATA); return; }
            reply_st(s);
            return;
        }
        case FUSE_GETXATTR: {
            auto* x = (const fuse_getxattr_in*)arg;
            const char* name = (const char*)arg + sizeof(fuse_getxattr_in);
            std::string v;
            St s = vfs_->getxattr(h->nodeid, name, &v);
            if (s == St::not_found) { reply_err(-ENODATA); return; }
            if (s != St::ok) { reply_st(s); return; }
            if (x->size == 0) {  // size probe
                fuse_getxattr_out go{};
                go.size = (uint32_t)v.size();
                reply_body(&go, sizeof(go));
            } else if (v.size() > x->size) {
                reply_err(-ERANGE);
            } else {
                reply_body(v.data(), v.size());
            }
            return;
        }
        case FUSE_LISTXATTR: {
            auto* x = (const fuse_getxattr_in*)arg;
            std::string names;
            St s = vfs_->listxattr(h->nodeid, &names);
            if (s != St::ok) { reply_st(s); return; }
            if (x->size == 0) {
                fuse_getxattr_out go{};
                go.size = (uint32_t)names.size();
                reply_body(&go, sizeof(go));
            } else if (names.size() > x->size) {
                reply_err(-ERANGE);
            } else {
                reply_body(names.data(), names.size());
            }
            return;
        }
        case FUSE_REMOVEXATTR: {
            St s = vfs_->removexattr(h->nodeid, (const char*)arg);
            if (s == St::not_found) { reply_err(-ENODATA); return; }
            reply_st(s);
            return;
        }
        case FUSE_COPY_FILE_RANGE: {
            // server-side copy: content-addressed chunks make aligned copies
            // instant (the object layer shares chunk refs — no data moves)
            auto* x = (const fuse_copy_file_range_in*)arg;
            if (x->flags) { reply_err(-EINVAL); return; }
            uint64_t copied = 0;
            St s = vfs_->copy_range(x->fh_in, x->off_in, x->fh_out, x->off_out,
                                    x->len, &copied);
            if (s != St::ok) { reply_st(s); return; }
            fuse_write_out wo{};
            wo.size = (uint32_t)std::min<uint64_t>(copied, UINT32_MAX);
            reply_body(&wo, sizeof(wo));
            return;
        }
        case FUSE_GETLK:
        case FUSE_SETLK:
        case FUSE_SETLKW:
        case FUSE_BMAP:
        case FUSE_IOCTL:
        case FUSE_POLL:
        default:
            reply_err(-ENOSYS);
            return;
    }
}

}  // namespace edgehip
