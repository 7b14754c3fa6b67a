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
        (void)write_chunk_file(fp, data.data(), (uint32_t)data.size());
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
    ssize_t w = ::write(fd_, rec.data(), rec.size());
    return w == (ssize_t)rec.size() ? St::ok : St::io;
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

St Journal::sync() {
    {   // drain the chunk-mirror queue first: fsync means "it is on disk"
        std::unique_lock<std::mutex> g(jmu_);
        jcv_drain_.wait(g, [&] { return jq_.empty(); });
    }
    std::lock_guard<std::mutex> g(mu_);
    return ::fdatasync(fd_) == 0 ? St::ok : St::io;
}

std::string Journal::chunk_path(const Fp& fp) const { return dir_ + "/chunks/" + fp.hex(); }

St Journal::write_chunk_file(const Fp& fp, const uint8_t* data, uint32_t len) {
    std::string tmp = chunk_path(fp) + ".tmp";
    int fd = ::open(tmp.c_str(), O_CREAT | O_WRONLY | O_TRUNC | O_CLOEXEC, 0644);
    if (fd < 0) return St::io;
    ssize_t w = ::write(fd, data, len);
    ::close(fd);
    if (w != (ssize_t)len) {
        ::unlink(tmp.c_str());
        return St::io;
    }
    if (::rename(tmp.c_str(), chunk_path(fp).c_str())) return St::io;
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
    while (pos + 16 <= data.size()) {
        if (get32(pos) != kJMagic) break;
        uint8_t type = data[pos + 4];
        uint32_t blen = get32(pos + 8);
        size_t total = 12 + blen + 4;
        if (pos + total > data.size()) break;  // torn tail
        if (crc32c(data.data() + pos, 12 + blen) != get32(pos + 12 + blen)) break;
        const uint8_t* body = data.data() + pos + 12;
        Replay r;
        r.type = type;
        switch (type) {
            case 1:
            case 3:
            case 5:
                r.path.assign((const char*)body, blen);
                break;
            case 4: {
                uint16_t ol = body[0] | body[1] << 8;
                if (2u + ol > blen) return St::bad_format;
                r.path.assign((const char*)body + 2, ol);
                r.path2.assign((const char*)body + 2 + ol, blen - 2 - ol);
                break;
            }
            case 2: {
                uint16_t pl = body[0] | body[1] << 8;
                if (2u + pl > blen) return St::bad_format;
                r.path.assign((const char*)body + 2, pl);
                r.manifest.assign(body + 2 + pl, body + blen);
                break;
            }
            default:
                break;
        }
        out->push_back(std::move(r));
        pos += total;
    }
    return St::ok;
}

}  // namespace edgehip
