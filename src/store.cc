#include "store.h"

#include <errno.h>
#include <dirent.h>
#include <fcntl.h>
#include <sys/stat.h>
#include <unistd.h>

#include "obs.h"

namespace edgehip {

HostStore::HostStore(uint64_t cap_bytes, const std::string& dir)
    : cap_(cap_bytes), dir_(dir) {
    if (!dir_.empty()) {
        ::mkdir(dir_.c_str(), 0755);
        // the disk tier is a CACHE (durability is the journal): stale files
        // from a previous process are unreachable — start clean
        if (DIR* d = ::opendir(dir_.c_str())) {
            size_t purged = 0;
            while (dirent* e = ::readdir(d)) {
                if (e->d_name[0] == '.') continue;
                if (::unlink((dir_ + "/" + e->d_name).c_str()) == 0) purged++;
            }
            ::closedir(d);
            if (purged)
                LOGI("store: purged %zu stale disk-tier chunks", purged);
        }
        LOGI("store: disk overflow tier at %s (RAM budget %.1f GiB)",
             dir_.c_str(), cap_ / (double)GiB);
    }
}

std::string HostStore::chunk_path(const Fp& fp) const {
    return dir_ + "/" + fp.hex();
}

St HostStore::write_file(const Fp& fp, const uint8_t* p, uint32_t len) {
    std::string path = chunk_path(fp);
    int fd = ::open(path.c_str(), O_CREAT | O_WRONLY | O_TRUNC | O_CLOEXEC, 0644);
    if (fd < 0) return St::io;
    size_t done = 0;
    while (done < len) {
        ssize_t w = ::write(fd, p + done, len - done);
        if (w < 0) {
            if (errno == EINTR) continue;
            ::close(fd);
            ::unlink(path.c_str());
            return St::io;
        }
        done += (size_t)w;
    }
    ::close(fd);
    return St::ok;
}

St HostStore::put(const Fp& fp, Blob&& data) {
    Shard& s = shard(fp);
    std::lock_guard<std::mutex> g(s.mu);
    auto it = s.map.find(fp);
    if (it != s.map.end()) {
        it->second.refs++;
        data.release();  // dedup hit: recycle the buffer
        return St::exists;
    }
    if (cap_ && bytes_.load() + data.len > cap_) {
        // RAM budget exhausted: overflow to the disk tier when configured
        if (dir_.empty()) return St::no_space;
        St st = write_file(fp, data.p, data.len);
        if (st != St::ok) return st;
        Chunk c;
        c.disk_len = data.len;
        disk_bytes_ += data.len;
        data.release();  // recycle the blob — bytes live on disk now
        s.map.emplace(fp, std::move(c));
        return St::ok;
    }
    bytes_ += data.len;
    Chunk c;
    c.data = std::move(data);
    s.map.emplace(fp, std::move(c));
    return St::ok;
}

St HostStore::put_copy(const Fp& fp, const uint8_t* data, uint32_t len) {
    Blob b = BufPool::make_exact(len);
    std::memcpy(b.p, data, len);
    b.len = len;
    return put(fp, std::move(b));
}

St HostStore::get(const Fp& fp, uint8_t* out, uint32_t off, uint32_t len) const {
    Shard& s = shard(fp);
    std::lock_guard<std::mutex> g(s.mu);
    auto it = s.map.find(fp);
    if (it == s.map.end()) return St::not_found;
    if (it->second.on_disk()) {
        if ((size_t)off + len > it->second.disk_len) return St::inval;
        int fd = ::open(chunk_path(fp).c_str(), O_RDONLY | O_CLOEXEC);
        if (fd < 0) return St::io;
        ssize_t r = ::pread(fd, out, len, off);
        ::close(fd);
        return r == (ssize_t)len ? St::ok : St::io;
    }
    if ((size_t)off + len > it->second.data.len) return St::inval;
    std::memcpy(out, it->second.data.p + off, len);
    return St::ok;
}

bool HostStore::has(const Fp& fp) const {
    Shard& s = shard(fp);
    std::lock_guard<std::mutex> g(s.mu);
    return s.map.count(fp) != 0;
}

uint32_t HostStore::stored_len(const Fp& fp) const {
    Shard& s = shard(fp);
    std::lock_guard<std::mutex> g(s.mu);
    auto it = s.map.find(fp);
    if (it == s.map.end()) return 0;
    return it->second.on_disk() ? it->second.disk_len : it->second.data.len;
}

St HostStore::addref(const Fp& fp) {
    Shard& s = shard(fp);
    std::lock_guard<std::mutex> g(s.mu);
    auto it = s.map.find(fp);
    if (it == s.map.end()) return St::not_found;
    it->second.refs++;
    return St::ok;
}

St HostStore::decref(const Fp& fp) {
    Shard& s = shard(fp);
    std::lock_guard<std::mutex> g(s.mu);
    auto it = s.map.find(fp);
    if (it == s.map.end()) return St::not_found;
    if (--it->second.refs == 0) {
        if (it->second.on_disk()) {
            disk_bytes_ -= it->second.disk_len;
            ::unlink(chunk_path(fp).c_str());
        } else {
            bytes_ -= it->second.data.len;
        }
        s.map.erase(it);  // Chunk dtor recycles the blob
    }
    return St::ok;
}

St HostStore::remove(const Fp& fp) {
    Shard& s = shard(fp);
    std::lock_guard<std::mutex> g(s.mu);
    auto it = s.map.find(fp);
    if (it == s.map.end()) return St::not_found;
    if (it->second.on_disk()) {
        disk_bytes_ -= it->second.disk_len;
        ::unlink(chunk_path(fp).c_str());
    } else {
        bytes_ -= it->second.data.len;
    }
    s.map.erase(it);
    return St::ok;
}

uint64_t HostStore::count() const {
    uint64_t n = 0;
    for (int i = 0; i < kShards; i++) {
        std::lock_guard<std::mutex> g(shards_[i].mu);
        n += shards_[i].map.size();
    }
    return n;
}

}  // namespace edgehip
