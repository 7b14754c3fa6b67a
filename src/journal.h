// Durability journal (SURVEY.md §3.1 H10, §6.4 — the storage-framework
// checkpoint/resume): an append-only, crc-protected log of namespace
// operations and committed manifests, plus a chunk mirror directory.
// `edgehip mount` with journal=<dir> replays the log to rebuild the
// filesystem; chunk bytes are mirrored to <dir>/chunks/<fp-hex> on commit
// (crash discards only un-fsynced data — the POSIX contract).
//
// Record format (little-endian):
//   u32 magic 'EHJ1' | u8 type | u8 pad[3] | u32 body_len | body | u32 crc32c
// Types: 1 MKDIR {path}  2 FILE {u16 path_len, path, manifest}  3 UNLINK
//        {path}  4 RENAME {u16 old_len, old, new}  5 RMDIR {path}
//        6 SYMLINK {u16 target_len, target, path}
//        7 LINK {u16 old_len, old, new}
//        8 XATTR_SET {u16 path_len, path, u16 name_len, name, value}
//        9 XATTR_DEL {u16 path_len, path, name}
//        10 EXCHANGE {u16 a_len, path_a, path_b}
// (6-10 are dedicated types — earlier builds overloaded RENAME with reserved
// in-band names, which a real file named ".eh_xchg_tmp" could collide with;
// ADVICE r1 low #4.)
#pragma once

#include <atomic>
#include <condition_variable>
#include <deque>
#include <mutex>
#include <thread>
#include <string>
#include <vector>

#include "common.h"

namespace edgehip {

class Journal {
  public:
    ~Journal();
    // Opens (creates) <dir>/journal.log and <dir>/chunks/. Returns ok.
    St open(const std::string& dir);
    bool enabled() const { return fd_ >= 0; }
    const std::string& dir() const { return dir_; }

    St mkdir_rec(const std::string& path);
    St rmdir_rec(const std::string& path);
    St unlink_rec(const std::string& path);
    St rename_rec(const std::string& oldp, const std::string& newp);
    St file_rec(const std::string& path, const std::vector<uint8_t>& manifest);
    St symlink_rec(const std::string& target, const std::string& path);
    St link_rec(const std::string& oldp, const std::string& newp);
    St xattr_set_rec(const std::string& path, const std::string& name,
                     const std::string& value);
    St xattr_del_rec(const std::string& path, const std::string& name);
    St exchange_rec(const std::string& a, const std::string& b);
    // fdatasync the log; St::io if any mirror/log write failed since open
    // (fsync must never acknowledge data the mirror failed to persist —
    // ADVICE r1 high #1)
    St sync();

    // Chunk mirror: store/load/remove stored chunk bytes by fingerprint.
    St put_chunk(const Fp& fp, const uint8_t* data, uint32_t len);
    St get_chunk(const Fp& fp, std::vector<uint8_t>* out) const;
    St del_chunk(const Fp& fp);

    struct Replay {
        uint8_t type;
        std::string path, path2;
        std::vector<uint8_t> manifest;
    };
    // Read every valid record (stops at the first torn/corrupt one).
    St replay(std::vector<Replay>* out) const;

  private:
    St append(uint8_t type, const std::vector<uint8_t>& body);
    std::string chunk_path(const Fp& fp) const;
    St write_chunk_file(const Fp& fp, const uint8_t* data, uint32_t len);
    void mirror_loop();

    std::string dir_;
    int fd_ = -1;
    int chunks_dirfd_ = -1;  // fsync target so renames into chunks/ are durable
    std::mutex mu_;
    std::atomic<bool> werr_{false};  // sticky: a mirror/log write failed

    // chunk-mirror write-behind: the ingest path enqueues; one worker writes.
    // sync() drains the queue, so fsync durability is unchanged — only the
    // put_chunks hot path stops paying per-chunk file I/O inline.
    std::mutex jmu_;
    std::condition_variable jcv_, jcv_drain_;
    struct JItem {
        uint64_t id;
        Fp fp;
        std::vector<uint8_t> data;
    };
    std::deque<JItem> jq_;
    std::vector<std::string> pending_fsync_;  // chunk files awaiting sync()
    uint64_t jq_bytes_ = 0;
    uint64_t jq_next_id_ = 1;
    bool jstop_ = false;
    bool jstarted_ = false;
    std::thread jworker_;
};

}  // namespace edgehip
