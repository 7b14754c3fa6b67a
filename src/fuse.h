// Raw /dev/fuse frontend: mount(2) with fd=, decode
// fuse_in_header requests, dispatch to the VFS, encode replies with writev.
// No libfuse — the environment has only the kernel protocol header, and the
// from-scratch rule forbids porting the reference's libfuse client anyway.
//
// Threading: conf.fuse_threads reader threads share the /dev/fuse fd (the
// kernel hands each read(2) one request); each thread owns its request and
// reply buffers and handles the request synchronously — concurrency across
// requests comes from the pool, matching the libfuse multithreaded loop the
// reference's callback surface implies.
#pragma once

#include <atomic>
#include <string>
#include <thread>
#include <vector>

#include "common.h"
#include "conf.h"
#include "vfs.h"

namespace edgehip {

class FuseMount {
  public:
    FuseMount(Vfs* vfs, const Conf* conf) : vfs_(vfs), conf_(conf) {}
    ~FuseMount();

    // mount(2) onto mountpoint and start reader threads. Returns St::ok once
    // INIT has been answered.
    St mount(const std::string& mountpoint);
    // Unmount and join reader threads.
    void unmount();
    // Block until the filesystem is unmounted externally (umount(8)).
    void wait();
    bool mounted() const { return mounted_.load(); }

  private:
    void reader_loop(int tid);
    // ch_fd: the /dev/fuse channel the request arrived on (per-thread clone
    // via FUSE_DEV_IOC_CLONE when available) — replies go back on it
    void handle(const uint8_t* req, size_t len, std::vector<uint8_t>* reply_buf,
                int ch_fd);
    int clone_channel();  // -1 if the kernel/ioctl refuses

    Vfs* vfs_;
    const Conf* conf_;
    int fd_ = -1;
    std::string mountpoint_;
    std::atomic<bool> mounted_{false};
    std::atomic<bool> stop_{false};
    std::vector<std::thread> threads_;
    uint32_t kernel_minor_ = 0;
};

}  // namespace edgehip
