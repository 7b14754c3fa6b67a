#include "journal.h"

#include <fcntl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <fstream>

#include "obs.h"

namespace edgehip {

constexpr uint32_t kJMagic = 0x314A4845;  // 'EHJ1'

Journal::~Journal() {
    {
        std::unique_lock<std::mutex> g(jmu_);
        jstop_ = true;
        jcv_.notify_all();
    }
    if (jworker_.joinable()) jworker_.join();
    if (fd_ >= 0) ::close(fd_);
    if (chunks_dirfd_ >= 0) ::close(chunks_dirfd_);
}

void Journal::mirror_loop() {
    for (;;) {
        uint64_t id;
        Fp fp;
        std::vector<uint8_t> data;
        {
            std::unique_lock<std::mutex> g(jmu_);
            jcv_.wait(g, [&] { return jstop_ || !jq_.empty(); });
            if (jq_.empty()) {
                if (jstop_) return;
                continue;
            }
            // copy, write, THEN pop: the entry stays visible to get_chunk
            // and sync() drain until its bytes are actually on disk
            id = jq_.front().id;
            fp = jq_.front().fp;
            data = jq_.front().data;
        }
        if (write_chunk_file(fp, data.data(), (uint32_t)data.size()) != St::ok) {
            // ENOSPC/EIO: the chunk is NOT durable. Mark the journal failed so
            // sync()/fsync report St::io instead of acknowledging lost data
            // (ADVICE r1 high #1); the entry is still popped (retrying forever
            // on a full disk would wedge the bounded queue and every writer).
            werr_.store(true);
            LOGE("journal: chunk mirror write failed for %s — journal marked "
                 "failed, fsync will return EIO", fp.hex().c_str());
        }
        {
            std::unique_lock<std::mutex> g(jmu_);
            if (!jq_.empty() && jq_.front().id == id) {  // del may have raced
                jq_bytes_ -= jq_.front().data.size();
                jq_.pop_front();
            }
            jcv_drain_.notify_all();
        }
    }
}

St Journal::open(const std::string& dir) {
    dir_ = dir;
    ::mkdir(dir.c_str(), 0755);
    ::mkdir((dir + "/chunks").c_str(), 0755);
    fd_ = ::open((dir + "/journal.log").c_str(), O_CREAT | O_WRONLY | O_APPEND | O_CLOEXEC, 0644);
    if (fd_ < 0) {
        LOGE("journal: open %s/journal.log failed: %s", dir.c_str(), strerror(errno));
        return St::io;
    }
    chunks_dirfd_ = ::open((dir + "/chunks").c_str(), O_RDONLY | O_DIRECTORY | O_CLOEXEC);
    if (chunks_dirfd_ < 0) {
        LOGE("journal: open %s/chunks dir failed: %s", dir.c_str(), strerror(errno));
        ::close(fd_);
        fd_ = -1;
        return St::io;
    }
    return St::ok;
}

St Journal::append(uint8_t type, const std::vector<uint8_t>& body) {
    std::vector<uint8_t> rec;
    rec.reserve(12 + body.size() + 4);
    auto put32 = [&](uint32_t v) {
        for (int i = 0; i < 4; i++) rec.push_back((v >> (8 * i)) & 0xff);
    };
    put32(kJMagic);
    rec.push_back(type);
    rec.push_back(0);
    rec.push_back(0);
    rec.push_back(0);
    put32((uint32_t)body.size());
    rec.insert(rec.end(), body.begin(), body.end());
    put32(crc32c(rec.data(), rec.size()));
    std::lock_guard<std::mutex> g(mu_);
    // Loop on EINTR/short writes; on failure ftruncate back to the record
    // start so no torn bytes stay mid-log ahead of later records (a torn
    // record would make replay silently discard everything after it —
    // ADVICE r1 medium #2).
    struct stat st;
    if (fstat(fd_, &st)) return St::io;
    const off_t start = st.st_size;  // O_APPEND: writes land at EOF
    size_t done = 0;
    while (done < rec.size()) {
        ssize_t w = ::write(fd_, rec.data() + done, rec.size() - done);
        if (w < 0) {
            if (errno == EINTR) continue;
            LOGE("journal: log write failed at %zu/%zu: %s", done, rec.size(),
                 strerror(errno));
            if (::ftruncate(fd_, start))
                werr_.store(true);  // can't even roll back: journal is toast
            return St::io;
        }
        done += (size_t)w;
    }
    return St::ok;
}

static std::vector<uint8_t> path_body(const std::string& p) {
    return std::vector<uint8_t>(p.begin(), p.end());
}

St Journal::mkdir_rec(const std::string& p) { return append(1, path_body(p)); }
St Journal::rmdir_rec(const std::string& p) { return append(5, path_body(p)); }
St Journal::unlink_rec(const std::string& p) { return append(3, path_body(p)); }

St Journal::rename_rec(const std::string& oldp, const std::string& newp) {
    std::vector<uint8_t> b;
    b.push_back((uint8_t)(oldp.size() & 0xff));
    b.push_back((uint8_t)(oldp.size() >> 8));
    b.insert(b.end(), oldp.begin(), oldp.end());
    b.insert(b.end(), newp.begin(), newp.end());
    return append(4, b);
}

St Journal::file_rec(const std::string& p, const std::vector<uint8_t>& manifest) {
    std::vector<uint8_t> b;
    b.push_back((uint8_t)(p.size() & 0xff));
    b.push_back((uint8_t)(p.size() >> 8));
    b.insert(b.end(), p.begin(), p.end());
    b.insert(b.end(), manifest.begin(), manifest.end());
    return append(2, b);
}

// two length-prefixed strings -> body {u16 a_len, a, b}
static std::vector<uint8_t> pair_body(const std::string& a, const std::string& b) {
    std::vector<uint8_t> v;
    v.push_back((uint8_t)(a.size() & 0xff));
    v.push_back((uint8_t)(a.size() >> 8));
    v.insert(v.end(), a.begin(), a.end());
    v.insert(v.end(), b.begin(), b.end());
    return v;
}

St Journal::symlink_rec(const std::string& target, const std::string& p) {
    return append(6, pair_body(target, p));
}
St Journal::link_rec(const std::string& oldp, const std::string& newp) {
    return append(7, pair_body(oldp, newp));
}
St Journal::xattr_set_rec(const std::string& p, const std::string& name,
                          const std::string& value) {
    // {u16 path_len, path, u16 name_len, name, value}
    std::vector<uint8_t> b = pair_body(p, "");
    b.push_back((uint8_t)(name.size() & 0xff));
    b.push_back((uint8_t)(name.size() >> 8));
    b.insert(b.end(), name.begin(), name.end());
    b.insert(b.end(), value.begin(), value.end());
    return append(8, b);
}
St Journal::xattr_del_rec(const std::string& p, const std::string& name) {
    return append(9, pair_body(p, name));
}
St Journal::exchange_rec(const std::string& a, const std::string& b) {
    return append(10, pair_body(a, b));
}

St Journal::sync() {
    {   // drain the chunk-mirror queue first: fsync means "it is on disk"
        std::unique_lock<std::mutex> g(jmu_);
        jcv_drain_.wait(g, [&] { return jq_.empty(); });
    }
    if (werr_.load()) return St::io;  // a mirror/log write was lost — say so
    // fsync chunk files written since the last sync (deferred from the
    // mirror loop), then the chunks/ dir once for the renames
    std::vector<std::string> files;
    {
        std::lock_guard<std::mutex> g(jmu_);
        files.swap(pending_fsync_);
    }
    for (auto& p : files) {
        int fd = ::open(p.c_str(), O_RDONLY | O_CLOEXEC);
        if (fd < 0) continue;  // deleted meanwhile: nothing to make durable
        int rc = ::fsync(fd);
        ::close(fd);
        if (rc) return St::io;
    }
    if (!files.empty() && chunks_dirfd_ >= 0 && ::fsync(chunks_dirfd_))
        return St::io;
    std::lock_guard<std::mutex> g(mu_);
    return ::fdatasync(fd_) == 0 ? St::ok : St::io;
}

std::string Journal::chunk_path(const Fp& fp) const { return dir_ + "/chunks/" + fp.hex(); }

St Journal::write_chunk_file(const Fp& fp, const uint8_t* data, uint32_t len) {
    std::string tmp = chunk_path(fp) + ".tmp";
    int fd = ::open(tmp.c_str(), O_CREAT | O_WRONLY | O_TRUNC | O_CLOEXEC, 0644);
    if (fd < 0) return St::io;
    size_t done = 0;
    while (done < len) {
        ssize_t w = ::write(fd, data + done, len - done);
        if (w < 0) {
            if (errno == EINTR) continue;
            ::close(fd);
            ::unlink(tmp.c_str());
            return St::io;
        }
        done += (size_t)w;
    }
    ::close(fd);
    if (::rename(tmp.c_str(), chunk_path(fp).c_str())) return St::io;
    // Durability is deferred to sync(): the file lands unfsynced here (the
    // mirror loop is the ingest write-behind — an fsync per chunk would
    // serialize it to disk latency), and sync() fsyncs every file written
    // since the last sync plus the chunks/ dir before acknowledging
    // (fsync must not return ok with chunk bytes only in
    // page cache). A crash between rename and sync can leave a torn chunk
    // file; restore detects that (size + content-address mismatch) and
    // treats the chunk as absent — correct, since sync never covered it.
    {
        std::lock_guard<std::mutex> g(jmu_);
        pending_fsync_.push_back(chunk_path(fp));
    }
    return St::ok;
}

St Journal::put_chunk(const Fp& fp, const uint8_t* data, uint32_t len) {
    std::unique_lock<std::mutex> g(jmu_);
    if (!jstarted_) {
        jstarted_ = true;
        jworker_ = std::thread([this] { mirror_loop(); });
    }
    // bound the write-behind queue at 256 MiB
    jcv_drain_.wait(g, [&] { return jq_bytes_ < (256ull << 20); });
    jq_.push_back({jq_next_id_++, fp, std::vector<uint8_t>(data, data + len)});
    jq_bytes_ += len;
    jcv_.notify_one();
    return St::ok;
}

St Journal::get_chunk(const Fp& fp, std::vector<uint8_t>* out) const {
    {   // still queued? serve from memory
        auto* self = const_cast<Journal*>(this);
        std::lock_guard<std::mutex> g(self->jmu_);
        for (auto& q : self->jq_)
            if (q.fp == fp) {
                *out = q.data;
                return St::ok;
            }
    }
    int fd = ::open(chunk_path(fp).c_str(), O_RDONLY | O_CLOEXEC);
    if (fd < 0) return St::not_found;
    struct stat st;
    if (fstat(fd, &st)) {
        ::close(fd);
        return St::io;
    }
    out->resize((size_t)st.st_size);
    ssize_t r = ::read(fd, out->data(), out->size());
    ::close(fd);
    return r == (ssize_t)out->size() ? St::ok : St::io;
}

St Journal::del_chunk(const Fp& fp) {
    {   // cancel any queued write of this chunk
        std::lock_guard<std::mutex> g(jmu_);
        for (auto it = jq_.begin(); it != jq_.end();) {
            if (it->fp == fp) {
                jq_bytes_ -= it->data.size();
                it = jq_.erase(it);
            } else {
                ++it;
            }
        }
        jcv_drain_.notify_all();
    }
    ::unlink(chunk_path(fp).c_str());
    return St::ok;
}

St Journal::replay(std::vector<Replay>* out) const {
    out->clear();
    std::ifstream f(dir_ + "/journal.log", std::ios::binary);
    if (!f) return St::ok;  // empty journal = fresh fs
    std::vector<uint8_t> data((std::istreambuf_iterator<char>(f)),
                              std::istreambuf_iterator<char>());
    size_t pos = 0;
    auto get32 = [&](size_t at) {
        return (uint32_t)data[at] | (uint32_t)data[at + 1] << 8 |
               (uint32_t)data[at + 2] << 16 | (uint32_t)data[at + 3] << 24;
    };
    size_t resyncs = 0;
    while (pos + 16 <= data.size()) {
        if (get32(pos) != kJMagic ||
            12 + (uint64_t)get32(pos + 8) + 4 > data.size() - pos ||
            crc32c(data.data() + pos, 12 + get32(pos + 8)) !=
                get32(pos + 12 + get32(pos + 8))) {
            // Bad magic / length / crc. A torn TAIL (crash mid-append) is
            // normal and ends replay; mid-log damage must not silently drop
            // every later fsync-acknowledged record:
            // scan forward for the next magic and resync.
            size_t next = pos + 1;
            while (next + 16 <= data.size() && get32(next) != kJMagic) next++;
            if (next + 16 > data.size()) break;  // tail — clean stop
            LOGE("journal: corrupt record at offset %zu, resynced at %zu "
                 "(records between are lost)", pos, next);
            resyncs++;
            pos = next;
            continue;
        }
        uint8_t type = data[pos + 4];
        uint32_t blen = get32(pos + 8);
        size_t total = 12 + blen + 4;
        const uint8_t* body = data.data() + pos + 12;
        Replay r;
        r.type = type;
        switch (type) {
            case 1:
            case 3:
            case 5:
                r.path.assign((const char*)body, blen);
                break;
            case 4:    // rename   {u16 old_len, old, new}
            case 6:    // symlink  {u16 target_len, target, path}
            case 7:    // link     {u16 old_len, old, new}
            case 9:    // xattr-del{u16 path_len, path, name}
            case 10: { // exchange {u16 a_len, a, b}
                uint16_t ol = body[0] | body[1] << 8;
                if (2u + ol > blen) return St::bad_format;
                r.path.assign((const char*)body + 2, ol);
                r.path2.assign((const char*)body + 2 + ol, blen - 2 - ol);
                break;
            }
            case 2: {  // file {u16 path_len, path, manifest}
                uint16_t pl = body[0] | body[1] << 8;
                if (2u + pl > blen) return St::bad_format;
                r.path.assign((const char*)body + 2, pl);
                r.manifest.assign(body + 2 + pl, body + blen);
                break;
            }
            case 8: {  // xattr-set {u16 path_len, path, u16 name_len, name, value}
                uint16_t pl = body[0] | body[1] << 8;
                if (2u + pl + 2u > blen) return St::bad_format;
                r.path.assign((const char*)body + 2, pl);
                uint16_t nl = body[2 + pl] | body[3 + pl] << 8;
                if (4u + pl + nl > blen) return St::bad_format;
                r.path2.assign((const char*)body + 4 + pl, nl);
                r.manifest.assign(body + 4 + pl + nl, body + blen);
                break;
            }
            default:
                break;
        }
        out->push_back(std::move(r));
        pos += total;
    }
    if (resyncs) metrics().journal_resyncs += resyncs;
    return St::ok;
}

}  // namespace edgehip
