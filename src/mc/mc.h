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

static inline bool is_device_mt(ucc_memory_type_t mt)
{
    return mt == UCC_MEMORY_TYPE_CUDA || mt == UCC_MEMORY_TYPE_ROCM ||
           mt == UCC_MEMORY_TYPE_CUDA_MANAGED ||
           mt == UCC_MEMORY_TYPE_ROCM_MANAGED;
}

} // namespace mc
} // namespace ucc

#endif
