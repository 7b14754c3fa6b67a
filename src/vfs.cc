#include "vfs.h"

#include <dirent.h>

#include "obs.h"

namespace edgehip {

Vfs::Vfs(Engine* eng, const Conf* conf, Journal* journal)
    : eng_(eng), conf_(conf), journal_(journal) {
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

Inode* Vfs::get_const(uint64_t ino) const {
    auto it = inodes_.find(ino);
    return it == inodes_.end() ? nullptr : it->second.get();
}

std::string Vfs::path_locked(uint64_t ino) const {
    std::vector<const std::string*> parts;
    const Inode* in = get_const(ino);
    int guard = 0;
    while (in && in->ino != 1 && guard++ < 256) {
        parts.push_back(&in->name);
        in = get_const(in->parent);
    }
    std::string p;
    for (auto it = parts.rbegin(); it != parts.rend(); ++it) {
        p.push_back('/');
        p += **it;
    }
    return p.empty() ? "/" : p;
}

St Vfs::resolve_parent_locked(const std::string& path, uint64_t* parent, std::string* leaf) {
    uint64_t cur = 1;
    size_t pos = 1;  // skip leading '/'
    if (path.empty() || path[0] != '/') return St::inval;
    while (true) {
        size_t slash = path.find('/', pos);
        std::string comp = path.substr(pos, slash == std::string::npos ? std::string::npos
                                                                       : slash - pos);
        if (slash == std::string::npos) {
            *parent = cur;
            *leaf = comp;
            return St::ok;
        }
        Inode* p = get(cur);
        if (!p) return St::not_found;
        auto it = p->children.find(comp);
        if (it == p->children.end()) return St::not_found;
        cur = it->second;
        pos = slash + 1;
    }
}

St Vfs::restore() {
    if (!journal_ || !journal_->enabled()) return St::ok;
    std::vector<Journal::Replay> recs;
    St s = journal_->replay(&recs);
    if (s != St::ok) return s;
    std::lock_guard<std::mutex> g(mu_);
    for (auto& r : recs) {
        uint64_t parent;
        std::string leaf;
        switch (r.type) {
            case 1: {  // mkdir
                if (resolve_parent_locked(r.path, &parent, &leaf) != St::ok) break;
                Inode* p = get(parent);
                if (!p || p->children.count(leaf)) break;
                auto in = std::make_unique<Inode>();
                in->ino = alloc_ino();
                in->mode = S_IFDIR | 0755;
                in->nlink = 2;
                in->nlookup = 0;
                in->parent = parent;
                in->name = leaf;
                in->atime = in->mtime = in->ctime = (int64_t)wall_ns();
                p->children[leaf] = in->ino;
                p->nlink++;
                inodes_.emplace(in->ino, std::move(in));
                break;
            }
            case 2: {  // file + manifest
                if (resolve_parent_locked(r.path, &parent, &leaf) != St::ok) break;
                Inode* p = get(parent);
                if (!p) break;
                Manifest m;
                if (Manifest::decode(r.manifest.data(), r.manifest.size(), &m) != St::ok)
                    break;
                auto it = p->children.find(leaf);
                Inode* f;
                if (it == p->children.end()) {
                    auto in = std::make_unique<Inode>();
                    in->ino = alloc_ino();
                    in->mode = S_IFREG | 0644;
                    in->nlookup = 0;
                    in->parent = parent;
                    in->name = leaf;
                    in->atime = in->mtime = in->ctime = (int64_t)wall_ns();
                    in->obj = std::make_shared<Object>(eng_, conf_);
                    f = in.get();
                    p->children[leaf] = in->ino;
                    inodes_.emplace(f->ino, std::move(in));
                } else {
                    f = get(it->second);
                    if (!f || !f->obj) break;
                    f->obj = std::make_shared<Object>(eng_, conf_);  // latest wins
                }
                f->obj->set_manifest(m);
                break;
            }
            case 3: {  // unlink
                if (resolve_parent_locked(r.path, &parent, &leaf) != St::ok) break;
                Inode* p = get(parent);
                if (!p) break;
                auto it = p->children.find(leaf);
                if (it == p->children.end()) break;
                inodes_.erase(it->second);
                p->children.erase(it);
                break;
            }
            case 5: {  // rmdir
                if (resolve_parent_locked(r.path, &parent, &leaf) != St::ok) break;
                Inode* p = get(parent);
                if (!p) break;
                auto it = p->children.find(leaf);
                if (it == p->children.end()) break;
                inodes_.erase(it->second);
                p->children.erase(it);
                p->nlink--;
                break;
            }
            case 6: {  // symlink {target, path}
                const std::string& target = r.path;
                if (resolve_parent_locked(r.path2, &parent, &leaf) != St::ok) break;
                Inode* p2 = get(parent);
                if (!p2 || p2->children.count(leaf)) break;
                auto in = std::make_unique<Inode>();
                in->ino = alloc_ino();
                in->mode = S_IFLNK | 0777;
                in->nlookup = 0;
                in->parent = parent;
                in->name = leaf;
                in->symlink_target = target;
                in->atime = in->mtime = in->ctime = (int64_t)wall_ns();
                p2->children[leaf] = in->ino;
                inodes_.emplace(in->ino, std::move(in));
                break;
            }
            case 7: {  // link {oldpath, newpath}
                uint64_t op2;
                std::string oleaf;
                if (resolve_parent_locked(r.path, &op2, &oleaf) != St::ok) break;
                Inode* opn = get(op2);
                if (!opn) break;
                auto oit = opn->children.find(oleaf);
                if (oit == opn->children.end()) break;
                if (resolve_parent_locked(r.path2, &parent, &leaf) != St::ok) break;
                Inode* p2 = get(parent);
                if (!p2 || p2->children.count(leaf)) break;
                p2->children[leaf] = oit->second;
                Inode* t = get(oit->second);
                if (t) t->nlink++;
                break;
            }
            case 8:    // xattr-set {path, name, value(manifest field)}
            case 9: {  // xattr-del {path, name}
                Inode* in = nullptr;
                if (r.path == "/") {
                    in = get(1);
                } else if (resolve_parent_locked(r.path, &parent, &leaf) == St::ok) {
                    Inode* p2 = get(parent);
                    if (p2) {
                        auto cit = p2->children.find(leaf);
                        if (cit != p2->children.end()) in = get(cit->second);
                    }
                }
                if (in) {
                    if (r.type == 9)
                        in->xattrs.erase(r.path2);
                    else
                        in->xattrs[r.path2] =
                            std::string(r.manifest.begin(), r.manifest.end());
                }
                break;
            }
            case 10: {  // exchange {path_a, path_b}: swap the two entries
                uint64_t pb;
                std::string bleaf;
                if (resolve_parent_locked(r.path, &parent, &leaf) != St::ok) break;
                if (resolve_parent_locked(r.path2, &pb, &bleaf) != St::ok) break;
                Inode* pa = get(parent);
                Inode* pbn = get(pb);
                if (!pa || !pbn) break;
                auto ia = pa->children.find(leaf);
                auto ib = pbn->children.find(bleaf);
                if (ia == pa->children.end() || ib == pbn->children.end()) break;
                std::swap(ia->second, ib->second);
                Inode* a = get(ia->second);
                Inode* b = get(ib->second);
                if (a) { a->parent = parent; a->name = leaf; }
                if (b) { b->parent = pb; b->name = bleaf; }
                break;
            }
            case 4: {  // rename {old, new}
                uint64_t np;
                std::string nleaf;
                if (resolve_parent_locked(r.path, &parent, &leaf) != St::ok) break;
                if (resolve_parent_locked(r.path2, &np, &nleaf) != St::ok) break;
                Inode* p = get(parent);
                Inode* q = get(np);
                if (!p || !q) break;
                auto it = p->children.find(leaf);
                if (it == p->children.end()) break;
                uint64_t ino = it->second;
                auto dst = q->children.find(nleaf);
                if (dst != q->children.end()) {
                    inodes_.erase(dst->second);
                    q->children.erase(dst);
                }
                p->children.erase(it);
                q->children[nleaf] = ino;
                Inode* moving = get(ino);
                if (moving) {
                    moving->parent = np;
                    moving->name = nleaf;
                }
                break;
            }
        }
    }
    LOGI("vfs: journal replayed: %zu records, %zu inodes", recs.size(), inodes_.size());
    return St::ok;
}

St Vfs::commit_and_journal(uint64_t ino, const std::shared_ptr<Object>& obj) {
    std::vector<uint8_t> bytes;
    St s = obj->commit(journal_ && journal_->enabled() ? &bytes : nullptr);
    if (s != St::ok) return s;
    if (journal_ && journal_->enabled()) {
        // Path resolution AND the journal append happen under mu_, in program
        // order with rename/unlink records. FUSE RELEASE is asynchronous: a
        // commit can run AFTER the file was unlinked or renamed, and a
        // file_rec appended out of order would resurrect the old path on
        // replay (observed as a rare remount flake).
        std::lock_guard<std::mutex> g(mu_);
        Inode* in = get(ino);
        if (!in || (in->unlinked && in->nlink == 0)) return St::ok;
        journal_->file_rec(path_locked(ino), bytes);
    }
    return St::ok;
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
    in->name = name;
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
    in->name = name;
    in->atime = in->mtime = in->ctime = (int64_t)wall_ns();
    Inode* raw = in.get();
    p->children[name] = in->ino;
    p->nlink++;
    p->mtime = p->ctime = in->ctime;
    inodes_.emplace(raw->ino, std::move(in));
    if (journal_ && journal_->enabled()) journal_->mkdir_rec(path_locked(raw->ino));
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
    if (journal_ && journal_->enabled()) journal_->unlink_rec(path_locked(c->ino));
    c->nlink--;
    if (c->nlink == 0) c->unlinked = true;
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
    if (journal_ && journal_->enabled()) journal_->rmdir_rec(path_locked(c->ino));
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
    if (journal_ && journal_->enabled()) {
        std::string oldp = path_locked(ino);
        std::string newp = path_locked(newparent);
        if (newp == "/") newp.clear();
        journal_->rename_rec(oldp, newp + "/" + newname);
    }
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
    if (moving) {
        moving->name = newname;
        moving->parent = newparent;
        if (moving->is_dir() && parent != newparent) {
            p->nlink--;
            np->nlink++;
        }
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
        uint32_t t = DT_REG;
        if (c && c->is_dir()) t = DT_DIR;
        else if (c && S_ISLNK(c->mode)) t = DT_LNK;
        out->push_back({kv.first, kv.second, t});
    }
    return St::ok;
}

St Vfs::setxattr(uint64_t ino, const std::string& name, const std::string& value,
                 uint32_t flags) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    if (!in) return St::not_found;
    bool has = in->xattrs.count(name) != 0;
    if ((flags & 1 /*XATTR_CREATE*/) && has) return St::exists;
    if ((flags & 2 /*XATTR_REPLACE*/) && !has) return St::not_found;
    in->xattrs[name] = value;
    in->ctime = (int64_t)wall_ns();
    if (journal_) journal_->xattr_set_rec(path_locked(ino), name, value);
    return St::ok;
}

St Vfs::getxattr(uint64_t ino, const std::string& name, std::string* out) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    if (!in) return St::not_found;
    auto it = in->xattrs.find(name);
    if (it == in->xattrs.end()) return St::not_found;
    *out = it->second;
    return St::ok;
}

St Vfs::listxattr(uint64_t ino, std::string* out) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    if (!in) return St::not_found;
    out->clear();
    for (auto& kv : in->xattrs) {
        out->append(kv.first);
        out->push_back('\0');
    }
    return St::ok;
}

St Vfs::removexattr(uint64_t ino, const std::string& name) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    if (!in) return St::not_found;
    if (!in->xattrs.erase(name)) return St::not_found;
    in->ctime = (int64_t)wall_ns();
    if (journal_) journal_->xattr_del_rec(path_locked(ino), name);
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
    uint64_t ino = 0;
    {
        std::lock_guard<std::mutex> g(mu_);
        auto it = handles_.find(fh);
        if (it == handles_.end()) return St::inval;
        ino = it->second;
        Inode* in = get(ino);
        if (in && in->obj && !in->unlinked) obj = in->obj;
        handles_.erase(it);
    }
    if (obj) return commit_and_journal(ino, obj);
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

St Vfs::copy_range(uint64_t fh_in, uint64_t off_in, uint64_t fh_out,
                   uint64_t off_out, uint64_t len, uint64_t* copied) {
    std::shared_ptr<Object> src, dst;
    Inode* din;
    {
        std::lock_guard<std::mutex> g(mu_);
        auto i1 = handles_.find(fh_in);
        auto i2 = handles_.find(fh_out);
        if (i1 == handles_.end() || i2 == handles_.end()) return St::inval;
        Inode* sin = get(i1->second);
        din = get(i2->second);
        if (!sin || !sin->obj || !din || !din->obj) return St::io;
        src = sin->obj;
        dst = din->obj;
        din->mtime = din->ctime = (int64_t)wall_ns();
    }
    return dst->copy_range(src.get(), off_in, off_out, len, copied);
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
    uint64_t ino = 0;
    {
        std::lock_guard<std::mutex> g(mu_);
        auto it = handles_.find(fh);
        if (it == handles_.end()) return St::inval;
        ino = it->second;
        Inode* in = get(ino);
        if (in && in->obj && !in->unlinked) obj = in->obj;
    }
    if (obj) return commit_and_journal(ino, obj);
    return St::ok;
}

St Vfs::fsync(uint64_t fh) {
    St s = flush(fh);
    if (s != St::ok) return s;
    if (journal_ && journal_->enabled()) return journal_->sync();
    return St::ok;
}

St Vfs::exchange(uint64_t parent, const std::string& name, uint64_t newparent,
                 const std::string& newname) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* p = get(parent);
    Inode* q = get(newparent);
    if (!p || !q || !p->is_dir() || !q->is_dir()) return St::not_dir;
    auto it1 = p->children.find(name);
    auto it2 = q->children.find(newname);
    if (it1 == p->children.end() || it2 == q->children.end()) return St::not_found;
    std::swap(it1->second, it2->second);
    Inode* a = get(it1->second);
    Inode* b = get(it2->second);
    if (a) { a->parent = parent; a->name = name; a->ctime = (int64_t)wall_ns(); }
    if (b) { b->parent = newparent; b->name = newname; b->ctime = (int64_t)wall_ns(); }
    if (journal_) {  // dedicated EXCHANGE record (no reserved temp names)
        std::string pa = path_locked(parent), pb = path_locked(newparent);
        auto j = [&](const std::string& x) { return x == "/" ? std::string() : x; };
        journal_->exchange_rec(j(pa) + "/" + name, j(pb) + "/" + newname);
    }
    return St::ok;
}

St Vfs::symlink(uint64_t parent, const std::string& name, const std::string& target,
                uint32_t uid, uint32_t gid, Inode** out) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* p = get(parent);
    if (!p) return St::not_found;
    if (!p->is_dir()) return St::not_dir;
    if (p->children.count(name)) return St::exists;
    auto in = std::make_unique<Inode>();
    in->ino = alloc_ino();
    in->mode = S_IFLNK | 0777;
    in->uid = uid;
    in->gid = gid;
    in->nlookup = 1;
    in->parent = parent;
    in->name = name;
    in->symlink_target = target;
    in->atime = in->mtime = in->ctime = (int64_t)wall_ns();
    Inode* raw = in.get();
    p->children[name] = in->ino;
    p->mtime = p->ctime = in->ctime;
    inodes_.emplace(raw->ino, std::move(in));
    if (journal_ && journal_->enabled())
        journal_->symlink_rec(target, path_locked(raw->ino));
    *out = raw;
    return St::ok;
}

St Vfs::readlink(uint64_t ino, std::string* target) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    if (!in) return St::not_found;
    if (!S_ISLNK(in->mode)) return St::inval;
    *target = in->symlink_target;
    return St::ok;
}

St Vfs::link(uint64_t ino, uint64_t newparent, const std::string& newname, Inode** out) {
    std::lock_guard<std::mutex> g(mu_);
    Inode* in = get(ino);
    Inode* np = get(newparent);
    if (!in || !np) return St::not_found;
    if (in->is_dir()) return St::is_dir;
    if (np->children.count(newname)) return St::exists;
    np->children[newname] = ino;
    in->nlink++;
    in->nlookup++;
    np->mtime = np->ctime = (int64_t)wall_ns();
    if (journal_ && journal_->enabled())
        journal_->link_rec(path_locked(ino), path_locked(newparent) == "/"
                                                 ? "/" + newname
                                                 : path_locked(newparent) + "/" + newname);
    *out = in;
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
