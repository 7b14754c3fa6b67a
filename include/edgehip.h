/* edgehip public C API (implemented in src/capi.cc / src/capi_gpu.hip,
 * exported by lib/libedgehip.so). The POSIX surface is the FUSE mount
 * (bin/edgehip mount); this API embeds the same native data path directly
 * (objects over the host or MI355X GPU engine). All functions return 0 /
 * positive on success and a negative errno-style value on failure unless
 * noted. Thread-safe. */
#ifndef EDGEHIP_H
#define EDGEHIP_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- primitives ---- */
void eh_blake3(const void* data, size_t len, uint8_t out32[32]);
uint32_t eh_crc32c(const void* data, size_t len);
size_t eh_lz4_compress_block(const uint8_t* src, size_t n, uint8_t* dst, size_t cap);
size_t eh_lz4_decompress_block(const uint8_t* src, size_t n, uint8_t* dst, size_t cap);
int64_t eh_lz4_chunk_compress(const uint8_t* src, size_t n, uint8_t* dst, size_t cap);
int eh_lz4_chunk_decompress(const uint8_t* src, size_t n, uint8_t* dst, size_t logical);
int eh_rs_encode(int k, int m, const uint8_t* const* data, uint8_t* const* parity,
                 size_t len);
int eh_rs_reconstruct(int k, int m, uint8_t* const* shards, const uint8_t* present,
                      size_t len);
void eh_rs_parity_matrix(int k, int m, uint8_t* out /* m*k */);

/* ---- engine + objects ----
 * opts: the same key=val list as `edgehip mount -o` (docs/options.md),
 * e.g. "engine=gpu,ngpus=8,fp=blake3,compress=lz4,dedup=1,rs=6+2". */
void* eh_engine_create(const char* opts);          /* NULL on failure */
void eh_engine_destroy(void* engine);
uint64_t eh_engine_stored_bytes(void* engine);
uint64_t eh_engine_stored_chunks(void* engine);
int eh_engine_is_gpu(void* engine);
int eh_engine_drop_chunk(void* engine, const uint8_t fp32[32]); /* fault hook */

void* eh_object_create(void* engine);
void eh_object_destroy(void* obj);
int eh_object_write(void* obj, uint64_t off, const void* data, uint64_t len);
int64_t eh_object_read(void* obj, uint64_t off, void* out, uint64_t len);
int64_t eh_object_preadv(void* obj, int n, const uint64_t* offs, const uint32_t* lens,
                         void* out);
int eh_object_flush(void* obj);
int eh_object_truncate(void* obj, uint64_t size);
uint64_t eh_object_size(void* obj);
int64_t eh_object_commit(void* obj, uint8_t* manifest_out, size_t cap);
int eh_object_punch_hole(void* obj, uint64_t off, uint64_t len);
/* server-side copy: returns bytes copied (aligned spans share chunk refs) */
int64_t eh_object_copy_range(void* dst_obj, void* src_obj, uint64_t src_off,
                             uint64_t dst_off, uint64_t len);

/* ---- GPU kernel test hooks (tests/kernels) ---- */
int eh_gpu_available(void);
int eh_mem_pin(void* p, size_t len);  /* hipHostRegister caller buffer */
int eh_mem_unpin(void* p);

/* multi-device topology (striped store) */
int eh_engine_ndev(void* engine);
int eh_engine_chunk_device(void* engine, const uint8_t fp32[32]); /* -2 host, -1 absent */
int eh_engine_mark_device_failed(void* engine, int dev);
uint64_t eh_metric(const char* name); /* global counter, (uint64_t)-1 unknown */

#ifdef __cplusplus
}
#endif
#endif /* EDGEHIP_H */
