// VFS: inode table, directory tree, open-handle table.
// POSIX semantics for the libfuse-compatible callback surface (§1.1): lookup,
// getattr, setattr, create, open, read, write, flush, release, fsync, unlink,
// mkdir, rmdir, rename, readdir, statfs.
#pragma once

#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <sys/stat.h>
#include <unordered_map>
#include <vector>

#include "common.h"
#include "conf.h"
#include "engine/engine.h"
#include "journal.h"
#include "obj.h"

namespace edgehip {

// extended attributes live on the inode (journal-persisted via pseudo-records)
using XattrMap = std::map<std::string, std::string>;

struct Inode {
    uint64_t ino = 0;
    uint32_t mode = 0;      // type + perms
    uint32_t uid = 0, gid = 0;
    uint32_t nlink = 1;
    uint64_t nlookup = 0;   // kernel references (FORGET decrements)
    int64_t atime = 0, mtime = 0, ctime = 0;  // ns since epoch
    std::map<std::string, uint64_t> children; // dirs only
    std::shared_ptr<Object> obj;              // files only
    std::string symlink_target;               // symlinks only
    uint64_t parent = 0;
    std::string name;                         // dentry name (journal paths)
    bool unlinked = false;

    bool is_dir() const { return S_ISDIR(mode); }
    uint64_t size() const { return obj ? obj->size() : 0; }
    XattrMap xattrs;
};

struct DirEntry {
    std::string name;
    uint64_t ino;
    uint32_t type;  // DT_*
};

class Vfs {
  public:
    Vfs(Engine* eng, const Conf* conf, Journal* journal = nullptr);

    // Remount: rebuild the namespace + manifests from the journal.
    St restore();

    // All methods return St; outputs via pointers. ino arguments are FUSE
    // nodeids (root = 1).
    St lookup(uint64_t parent, const std::string& name, Inode** out);
    St getattr(uint64_t ino, Inode** out);
    St setattr_size(uint64_t ino, uint64_t size);
    St create(uint64_t parent, const std::string& name, uint32_t mode, uint32_t uid,
              uint32_t gid, Inode** out);
    St mkdir(uint64_t parent, const std::string& name, uint32_t mode, uint32_t uid,
             uint32_t gid, Inode** out);
    St unlink(uint64_t parent, const std::string& name);
    St rmdir(uint64_t parent, const std::string& name);
    St rename(uint64_t parent, const std::string& name, uint64_t newparent,
              const std::string& newname, bool noreplace);
    // RENAME_EXCHANGE: atomically swap two existing entries.
    St exchange(uint64_t parent, const std::string& name, uint64_t newparent,
                const std::string& newname);
    St symlink(uint64_t parent, const std::string& name, const std::string& target,
               uint32_t uid, uint32_t gid, Inode** out);
    St readlink(uint64_t ino, std::string* target);
    St link(uint64_t ino, uint64_t newparent, const std::string& newname, Inode** out);
    St readdir(uint64_t ino, std::vector<DirEntry>* out);

    // xattrs (flags: XATTR_CREATE=1 / XATTR_REPLACE=2 semantics)
    St setxattr(uint64_t ino, const std::string& name, const std::string& value,
                uint32_t flags);
    St getxattr(uint64_t ino, const std::string& name, std::string* out);
    St listxattr(uint64_t ino, std::string* out);  // NUL-joined names
    St removexattr(uint64_t ino, const std::string& name);

    // handles
    St open(uint64_t ino, uint64_t* fh);
    St release(uint64_t fh);
    Inode* handle_inode(uint64_t fh);

    St read(uint64_t fh, uint64_t off, uint64_t len, uint8_t* out, uint64_t* got);
    St copy_range(uint64_t fh_in, uint64_t off_in, uint64_t fh_out, uint64_t off_out,
                  uint64_t len, uint64_t* copied);
    St write(uint64_t fh, uint64_t off, const uint8_t* data, uint64_t len);
    St flush(uint64_t fh);
    St fsync(uint64_t fh);  // flush + journal fdatasync

    void forget(uint64_t ino, uint64_t n);
    Inode* get(uint64_t ino);
    Inode* get_const(uint64_t ino) const;
    void ref(Inode* in) { in->nlookup++; }

    uint64_t inode_count() const { return inodes_.size(); }
    Engine* engine() { return eng_; }
    std::mutex& lock() { return mu_; }  // big VFS lock (data path is in obj/engine)

  private:
    uint64_t alloc_ino() { return next_ino_++; }
    void maybe_free(uint64_t ino);
    std::string path_locked(uint64_t ino) const;
    St commit_and_journal(uint64_t ino, const std::shared_ptr<Object>& obj);
    St resolve_parent_locked(const std::string& path, uint64_t* parent,
                             std::string* leaf);

    Engine* eng_;
    const Conf* conf_;
    Journal* journal_;
    std::mutex mu_;
    std::unordered_map<uint64_t, std::unique_ptr<Inode>> inodes_;
    std::unordered_map<uint64_t, uint64_t> handles_;  // fh -> ino
    uint64_t next_ino_ = 2;  // 1 = root
    uint64_t next_fh_ = 1;
};

}  // namespace edgehip
