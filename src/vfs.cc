#include "vfs.h"

#include <dirent.h>

#include "obs.h"

namespace edgehip {

Vfs::Vfs(Engine* eng, const Conf* conf) : eng_(eng), conf_(conf) {
    auto root = std::make_unique<Inode>();
    root->ino = 1;
    root->mode = S_IFDIR | 0755;
    root->nlink = 2;
    root->nlookup = 1;  // root never forgotten
    root->parent = 1;
    root->atime = root->mtime = root->ctime = (int64_t)wall_ns();
    inodes_.emplace(1, std::move(root));
}

Inode* Vfs::get(uint64_t ino) {
    auto it = inodes_.find(ino);
    return it == inodes_.end() ? nullptr : it->second.get();
}

St Vfs::lookup(uint64_t parent, const std::string& name, Inode** out) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* p = get(parent);
    if (!p) return St::not_found;
    if (!p->is_dir()) return St::not_dir;
    auto it = p->children.find(name);
    if (it == p->children.end()) return St::not_found;
    Inode* c = get(it->second);
    if (!c) return St::not_found;
    c->nlookup++;
    *out = c;
    return St::ok;
}

St Vfs::getattr(uint64_t ino, Inode** out) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    if (!in) return St::not_found;
    *out = in;
    return St::ok;
}

St Vfs::setattr_size(uint64_t ino, uint64_t size) {
    std::shared_ptr<Object> obj;
    {
        std::lock_guard<std::mutex> g(mu_);
        Inode* in = get(ino);
        if (!in) return St::not_found;
        if (in->is_dir()) return St::is_dir;
        obj = in->obj;
        in->mtime = in->ctime = (int64_t)wall_ns();
    }
    return obj ? obj->truncate(size) : St::io;
}

St Vfs::create(uint64_t parent, const std::string& name, uint32_t mode, uint32_t uid,
               uint32_t gid, Inode** out) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* p = get(parent);
    if (!p) return St::not_found;
    if (!p->is_dir()) return St::not_dir;
    if (p->children.count(name)) return St::exists;
    if (name.size() > 255) return St::inval;
    auto in = std::make_unique<Inode>();
    in->ino = alloc_ino();
    in->mode = S_IFREG | (mode & 07777);
    in->uid = uid;
    in->gid = gid;
    in->nlookup = 1;
    in->parent = parent;
    in->atime = in->mtime = in->ctime = (int64_t)wall_ns();
    in->obj = std::make_shared<Object>(eng_, conf_);
    Inode* raw = in.get();
    p->children[name] = in->ino;
    p->mtime = p->ctime = in->ctime;
    inodes_.emplace(raw->ino, std::move(in));
    *out = raw;
    return St::ok;
}

St Vfs::mkdir(uint64_t parent, const std::string& name, uint32_t mode, uint32_t uid,
              uint32_t gid, Inode** out) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* p = get(parent);
    if (!p) return St::not_found;
    if (!p->is_dir()) return St::not_dir;
    if (p->children.count(name)) return St::exists;
    auto in = std::make_unique<Inode>();
    in->ino = alloc_ino();
    in->mode = S_IFDIR | (mode & 07777);
    in->uid = uid;
    in->gid = gid;
    in->nlink = 2;
    in->nlookup = 1;
    in->parent = parent;
    in->atime = in->mtime = in->ctime = (int64_t)wall_ns();
    Inode* raw = in.get();
    p->children[name] = in->ino;
    p->nlink++;
    p->mtime = p->ctime = in->ctime;
    inodes_.emplace(raw->ino, std::move(in));
    *out = raw;
    return St::ok;
}

void Vfs::maybe_free(uint64_t ino) {
    // caller holds mu_
    Inode* in = get(ino);
    if (!in || ino == 1) return;
    if (in->unlinked && in->nlookup == 0) inodes_.erase(ino);
}

St Vfs::unlink(uint64_t parent, const std::string& name) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* p = get(parent);
    if (!p) return St::not_found;
    auto it = p->children.find(name);
    if (it == p->children.end()) return St::not_found;
    Inode* c = get(it->second);
    if (!c) return St::not_found;
    if (c->is_dir()) return St::is_dir;
    LOGD("vfs: unlink %s (ino %lu)", name.c_str(), (unsigned long)c->ino);
    c->unlinked = true;
    c->nlink--;
    p->children.erase(it);
    p->mtime = p->ctime = (int64_t)wall_ns();
    maybe_free(c->ino);
    return St::ok;
}

St Vfs::rmdir(uint64_t parent, const std::string& name) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* p = get(parent);
    if (!p) return St::not_found;
    auto it = p->children.find(name);
    if (it == p->children.end()) return St::not_found;
    Inode* c = get(it->second);
    if (!c) return St::not_found;
    if (!c->is_dir()) return St::not_dir;
    if (!c->children.empty()) return St::not_empty;
    c->unlinked = true;
    p->children.erase(it);
    p->nlink--;
    p->mtime = p->ctime = (int64_t)wall_ns();
    maybe_free(c->ino);
    return St::ok;
}

St Vfs::rename(uint64_t parent, const std::string& name, uint64_t newparent,
               const std::string& newname, bool noreplace) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* p = get(parent);
    Inode* np = get(newparent);
    if (!p || !np) return St::not_found;
    auto it = p->children.find(name);
    if (it == p->children.end()) return St::not_found;
    uint64_t ino = it->second;
    LOGD("vfs: rename %s -> %s", name.c_str(), newname.c_str());
    auto dst = np->children.find(newname);
    if (dst != np->children.end()) {
        if (noreplace) return St::exists;
        Inode* victim = get(dst->second);
        if (victim) {
            if (victim->is_dir() && !victim->children.empty()) return St::not_empty;
            victim->unlinked = true;
            if (victim->is_dir()) np->nlink--;
            maybe_free(victim->ino);
        }
        np->children.erase(dst);
    }
    Inode* moving = get(ino);
    p->children.erase(it);
    np->children[newname] = ino;
    if (moving && moving->is_dir() && parent != newparent) {
        p->nlink--;
        np->nlink++;
        moving->parent = newparent;
    }
    int64_t t = (int64_t)wall_ns();
    p->mtime = p->ctime = t;
    np->mtime = np->ctime = t;
    return St::ok;
}

St Vfs::readdir(uint64_t ino, std::vector<DirEntry>* out) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    if (!in) return St::not_found;
    if (!in->is_dir()) return St::not_dir;
    out->clear();
    out->push_back({".", in->ino, DT_DIR});
    out->push_back({"..", in->parent, DT_DIR});
    for (auto& kv : in->children) {
        Inode* c = get(kv.second);
        out->push_back({kv.first, kv.second, c && c->is_dir() ? (uint32_t)DT_DIR : (uint32_t)DT_REG});
    }
    return St::ok;
}

St Vfs::open(uint64_t ino, uint64_t* fh) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    if (!in) return St::not_found;
    if (in->is_dir()) return St::is_dir;
    *fh = next_fh_++;
    handles_[*fh] = ino;
    return St::ok;
}

Inode* Vfs::handle_inode(uint64_t fh) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = handles_.find(fh);
    if (it == handles_.end()) return nullptr;
    return get(it->second);
}

St Vfs::release(uint64_t fh) {
    std::shared_ptr<Object> obj;
    {
        std::lock_guard<std::mutex> g(mu_);
        auto it = handles_.find(fh);
        if (it == handles_.end()) return St::inval;
        Inode* in = get(it->second);
        if (in && in->obj) obj = in->obj;
        handles_.erase(it);
    }
    if (obj) return obj->commit();
    return St::ok;
}

St Vfs::read(uint64_t fh, uint64_t off, uint64_t len, uint8_t* out, uint64_t* got) {
    std::shared_ptr<Object> obj;
    {
        std::lock_guard<std::mutex> g(mu_);
        auto it = handles_.find(fh);
        if (it == handles_.end()) return St::inval;
        Inode* in = get(it->second);
        if (!in || !in->obj) return St::io;
        obj = in->obj;
    }
    return obj->read(off, len, out, got);
}

St Vfs::write(uint64_t fh, uint64_t off, const uint8_t* data, uint64_t len) {
    std::shared_ptr<Object> obj;
    Inode* in;
    {
        std::lock_guard<std::mutex> g(mu_);
        auto it = handles_.find(fh);
        if (it == handles_.end()) return St::inval;
        in = get(it->second);
        if (!in || !in->obj) return St::io;
        obj = in->obj;
        in->mtime = in->ctime = (int64_t)wall_ns();
    }
    return obj->write(off, data, len);
}

St Vfs::flush(uint64_t fh) {
    std::shared_ptr<Object> obj;
    {
        std::lock_guard<std::mutex> g(mu_);
        auto it = handles_.find(fh);
        if (it == handles_.end()) return St::inval;
        Inode* in = get(it->second);
        if (in && in->obj) obj = in->obj;
    }
    if (obj) return obj->commit();
    return St::ok;
}

void Vfs::forget(uint64_t ino, uint64_t n) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    if (!in || ino == 1) return;
    in->nlookup = in->nlookup > n ? in->nlookup - n : 0;
    maybe_free(ino);
}

}  // namespace edgehip
