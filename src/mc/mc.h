/* Memory component: allocation, pointer classification and synchronous
 * copies across host/device spaces. Parity: reference components/mc/
 * (mc_cpu.c, mc_rocm.c mpool + hipPointerGetAttributes). The device pool
 * lives in tl/cdna4 (co-designed scratch); this layer is the thin
 * query/alloc/copy service. All HIP use is runtime-gated so the library
 * loads and runs on GPU-less hosts. */
#ifndef UCC_AMD_MC_H_
#define UCC_AMD_MC_H_

#include <cstddef>
#include "../api/ucc.h"

namespace ucc {
namespace mc {

bool hip_available();   /* true iff a HIP device is present */
int  hip_device_count();

ucc_status_t mem_query(const void *ptr, ucc_memory_type_t *mt);
ucc_status_t alloc(void **ptr, size_t size, ucc_memory_type_t mt);
ucc_status_t mem_free(void *ptr, ucc_memory_type_t mt);

/* Scratch mpool (parity: reference mc_rocm.c:97-108, 1MB elems / max 8
 * cached): short-lived staging buffers for asymmetric-memtype and pack
 * paths. Requests <= the elem size are served from a per-memtype cache
 * of pre-sized buffers (hipMalloc in a collective's init path is a
 * latency hazard); larger requests fall through to plain alloc.
 * scratch_free() returns pooled buffers to the cache. Thread-safe. */
ucc_status_t scratch_alloc(void **ptr, size_t size, ucc_memory_type_t mt);
void         scratch_free(void *ptr, size_t size, ucc_memory_type_t mt);
/* observability for tests: number of raw device/host allocations done
 * by scratch_alloc since process start (cache hits do not count) */
size_t       scratch_raw_allocs();
/* Synchronous copy between any host/device combination. */
ucc_status_t copy(void *dst, ucc_memory_type_t dst_mt, const void *src,
                  ucc_memory_type_t src_mt, size_t bytes);
ucc_status_t memset_(void *ptr, ucc_memory_type_t mt, int value,
                     size_t bytes);

/* HIP-IPC handle export/import for ucc_mem_map (device memory). The
 * handle blob is hipIpcMemHandle_t-sized (64B). */
constexpr size_t kIpcHandleBytes = 64;
ucc_status_t ipc_export(const void *ptr, void *handle_out,
                        size_t *base_off_out);
ucc_status_t ipc_import(const void *handle, void **mapped);
ucc_status_t ipc_close(void *mapped);

/* Synchronize the device's default stream (no-op without a device). */
ucc_status_t device_sync();

/* HIP event helpers for the EE completion-event flow (gated: no-ops
 * without a device). ev handles are opaque (hipEvent_t). */
ucc_status_t stream_event_record(void *stream, void **ev_out);
/* 1 = complete, 0 = pending, <0 = error */
int          event_query(void *ev);
void         event_free(void *ev);

static inline bool is_device_mt(ucc_memory_type_t mt)
{
    return mt == UCC_MEMORY_TYPE_CUDA || mt == UCC_MEMORY_TYPE_ROCM ||
           mt == UCC_MEMORY_TYPE_CUDA_MANAGED ||
           mt == UCC_MEMORY_TYPE_ROCM_MANAGED;
}

} // namespace mc
} // namespace ucc

#endif
