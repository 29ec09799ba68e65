#include "mc.h"
#include "../utils/log.h"

#include <atomic>
#include <cstring>
#include <mutex>
#include <vector>

#include <hip/hip_runtime.h>

namespace ucc {
namespace mc {

int hip_device_count()
{
    /* cache positive answers only: a failed early probe (e.g. before
     * the process's primary HIP runtime finished loading) must not
     * latch the library into host-only mode forever */
    static std::atomic<int> cached{-1};
    int c = cached.load(std::memory_order_relaxed);
    if (c >= 0) {
        return c;
    }
    int        n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess) {
        UCC_LOG(LogLevel::DEBUG, "mc", "hipGetDeviceCount: %s",
                hipGetErrorString(e));
        return 0; /* not cached */
    }
    cached.store(n, std::memory_order_relaxed);
    return n;
}

extern "C" int ucc_amd_hip_device_count_c() { return hip_device_count(); }

bool hip_available() { return hip_device_count() > 0; }

ucc_status_t mem_query(const void *ptr, ucc_memory_type_t *mt)
{
    if (!hip_available()) {
        *mt = UCC_MEMORY_TYPE_HOST;
        return UCC_OK;
    }
    hipPointerAttribute_t attr;
    hipError_t            err = hipPointerGetAttributes(&attr, ptr);
    if (err != hipSuccess) {
        (void)hipGetLastError();
        *mt = UCC_MEMORY_TYPE_HOST;
        return UCC_OK;
    }
    switch (attr.type) {
    case hipMemoryTypeDevice: *mt = UCC_MEMORY_TYPE_CUDA; break;
    case hipMemoryTypeManaged: *mt = UCC_MEMORY_TYPE_CUDA_MANAGED; break;
    default: *mt = UCC_MEMORY_TYPE_HOST; break;
    }
    return UCC_OK;
}

ucc_status_t alloc(void **ptr, size_t size, ucc_memory_type_t mt)
{
    if (is_device_mt(mt)) {
        if (hipMalloc(ptr, size) != hipSuccess) {
            return UCC_ERR_NO_MEMORY;
        }
        return UCC_OK;
    }
    *ptr = malloc(size);
    return *ptr ? UCC_OK : UCC_ERR_NO_MEMORY;
}

ucc_status_t mem_free(void *ptr, ucc_memory_type_t mt)
{
    if (is_device_mt(mt)) {
        return hipFree(ptr) == hipSuccess ? UCC_OK : UCC_ERR_INVALID_PARAM;
    }
    free(ptr);
    return UCC_OK;
}

ucc_status_t copy(void *dst, ucc_memory_type_t dst_mt, const void *src,
                  ucc_memory_type_t src_mt, size_t bytes)
{
    if (!is_device_mt(dst_mt) && !is_device_mt(src_mt)) {
        memcpy(dst, src, bytes);
        return UCC_OK;
    }
    hipError_t err = hipMemcpy(dst, src, bytes, hipMemcpyDefault);
    return err == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

ucc_status_t memset_(void *ptr, ucc_memory_type_t mt, int value, size_t bytes)
{
    if (!is_device_mt(mt)) {
        memset(ptr, value, bytes);
        return UCC_OK;
    }
    return hipMemset(ptr, value, bytes) == hipSuccess ? UCC_OK
                                                      : UCC_ERR_NO_RESOURCE;
}

/* ----------------------------------------------------- scratch mpool */
namespace {
constexpr size_t kScratchElem = 1 << 20; /* 1 MiB (ref mc_rocm.c:20-25) */
constexpr size_t kScratchMax  = 8;       /* cached elems per pool       */

struct ScratchPool {
    std::mutex          mtx;
    std::vector<void *> free_list;
};
ScratchPool g_pool_dev, g_pool_host;
std::atomic<size_t> g_raw_allocs{0};

ScratchPool &pool_for(ucc_memory_type_t mt)
{
    return is_device_mt(mt) ? g_pool_dev : g_pool_host;
}
} // namespace

ucc_status_t scratch_alloc(void **ptr, size_t size, ucc_memory_type_t mt)
{
    if (size > kScratchElem) {
        g_raw_allocs.fetch_add(1, std::memory_order_relaxed);
        return alloc(ptr, size, mt);
    }
    ScratchPool &p = pool_for(mt);
    {
        std::lock_guard<std::mutex> lk(p.mtx);
        if (!p.free_list.empty()) {
            *ptr = p.free_list.back();
            p.free_list.pop_back();
            return UCC_OK;
        }
    }
    g_raw_allocs.fetch_add(1, std::memory_order_relaxed);
    return alloc(ptr, kScratchElem, mt);
}

void scratch_free(void *ptr, size_t size, ucc_memory_type_t mt)
{
    if (!ptr) {
        return;
    }
    if (size <= kScratchElem) { /* pooled size class */
        ScratchPool &p = pool_for(mt);
        std::lock_guard<std::mutex> lk(p.mtx);
        if (p.free_list.size() < kScratchMax) {
            p.free_list.push_back(ptr);
            return;
        }
    }
    (void)mem_free(ptr, mt);
}

size_t scratch_raw_allocs()
{
    return g_raw_allocs.load(std::memory_order_relaxed);
}

ucc_status_t device_sync()
{
    if (!hip_available()) {
        return UCC_OK;
    }
    return hipStreamSynchronize(nullptr) == hipSuccess
               ? UCC_OK
               : UCC_ERR_NO_RESOURCE;
}

ucc_status_t stream_event_record(void *stream, void **ev_out)
{
    if (!hip_available()) {
        *ev_out = nullptr;
        return UCC_ERR_NOT_SUPPORTED;
    }
    hipEvent_t e = nullptr;
    if (hipEventCreateWithFlags(&e, hipEventDisableTiming) !=
            hipSuccess ||
        hipEventRecord(e, (hipStream_t)stream) != hipSuccess) {
        if (e) {
            (void)hipEventDestroy(e);
        }
        return UCC_ERR_NO_RESOURCE;
    }
    *ev_out = (void *)e;
    return UCC_OK;
}

int event_query(void *ev)
{
    hipError_t e = hipEventQuery((hipEvent_t)ev);
    if (e == hipSuccess) {
        return 1;
    }
    if (e == hipErrorNotReady) {
        return 0;
    }
    return -1;
}

void event_free(void *ev)
{
    if (ev) {
        (void)hipEventDestroy((hipEvent_t)ev);
    }
}

ucc_status_t ipc_export(const void *ptr, void *handle_out,
                        size_t *base_off_out)
{
    static_assert(sizeof(hipIpcMemHandle_t) <= kIpcHandleBytes,
                  "handle blob too small");
    hipDeviceptr_t base  = nullptr;
    size_t         bsize = 0;
    if (hipMemGetAddressRange(&base, &bsize, (hipDeviceptr_t)ptr) !=
        hipSuccess) {
        return UCC_ERR_INVALID_PARAM;
    }
    hipIpcMemHandle_t h;
    if (hipIpcGetMemHandle(&h, (void *)base) != hipSuccess) {
        return UCC_ERR_NO_RESOURCE;
    }
    memcpy(handle_out, &h, sizeof(h));
    *base_off_out = (size_t)((uintptr_t)ptr - (uintptr_t)base);
    return UCC_OK;
}

ucc_status_t ipc_import(const void *handle, void **mapped)
{
    hipIpcMemHandle_t h;
    memcpy(&h, handle, sizeof(h));
    hipError_t e =
        hipIpcOpenMemHandle(mapped, h, hipIpcMemLazyEnablePeerAccess);
    return e == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

ucc_status_t ipc_close(void *mapped)
{
    return hipIpcCloseMemHandle(mapped) == hipSuccess ? UCC_OK
                                                      : UCC_ERR_INVALID_PARAM;
}

} // namespace mc
} // namespace ucc
