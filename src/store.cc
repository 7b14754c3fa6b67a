#include "store.h"

#include "obs.h"

namespace edgehip {

St HostStore::put(const Fp& fp, Blob&& data) {
    Shard& s = shard(fp);
    std::lock_guard<std::mutex> g(s.mu);
    auto it = s.map.find(fp);
    if (it != s.map.end()) {
        it->second.refs++;
        data.release();  // dedup hit: recycle the buffer
        return St::exists;
    }
    if (cap_ && bytes_.load() + data.len > cap_) return St::no_space;
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
    return it == s.map.end() ? 0 : it->second.data.len;
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
        bytes_ -= it->second.data.len;
        s.map.erase(it);  // Chunk dtor recycles the blob
    }
    return St::ok;
}

St HostStore::remove(const Fp& fp) {
    Shard& s = shard(fp);
    std::lock_guard<std::mutex> g(s.mu);
    auto it = s.map.find(fp);
    if (it == s.map.end()) return St::not_found;
    bytes_ -= it->second.data.len;
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
