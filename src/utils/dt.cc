/* Datatype/op/status helpers. Parity target: reference ucc.h dtype table
 * (src/ucc/api/ucc.h:203-221) and ucc_status.h strings. */
#include "../api/ucc.h"

#include <atomic>

extern "C" {

size_t ucc_dt_size(ucc_datatype_t dt)
{
    switch (dt) {
    case UCC_DT_INT8:
    case UCC_DT_UINT8:
    case UCC_DT_FLOAT8_E4M3:
    case UCC_DT_FLOAT8_E5M2: return 1;
    case UCC_DT_INT16:
    case UCC_DT_UINT16:
    case UCC_DT_FLOAT16:
    case UCC_DT_BFLOAT16: return 2;
    case UCC_DT_INT32:
    case UCC_DT_UINT32:
    case UCC_DT_FLOAT32: return 4;
    case UCC_DT_INT64:
    case UCC_DT_UINT64:
    case UCC_DT_FLOAT64:
    case UCC_DT_FLOAT32_COMPLEX: return 8;
    case UCC_DT_INT128:
    case UCC_DT_UINT128:
    case UCC_DT_FLOAT128:
    case UCC_DT_FLOAT64_COMPLEX: return 16;
    case UCC_DT_FLOAT128_COMPLEX: return 32;
    default: {
        const ucc_generic_dt_ops_t *g = ucc_dt_generic_ops(dt);
        if (g && (g->flags & UCC_GENERIC_DT_OPS_FLAG_CONTIG)) {
            return g->contig_size;
        }
        return 0;
    }
    }
}

/* ------------------------------------------------- generic datatypes */
/* Registered in a process-global slab; the public handle is
 * UCC_DT_USERDEFINED_BASE + index (predefined ids stay small ints). */
#define UCC_DT_USERDEFINED_BASE 0x2000

struct ucc_dt_generic {
    ucc_generic_dt_ops_t ops;
    void                *cookie;
    /* 0 = free, 1 = claimed (being filled), 2 = live. Slots are claimed
     * with CAS so THREAD_MULTIPLE creators never share a slot, and ops
     * are published with a release store so readers doing an acquire
     * load of 'used' see a fully-written struct. */
    std::atomic<int>     used;
};

static struct ucc_dt_generic g_generic_dts[64];

ucc_status_t ucc_dt_create_generic(const ucc_generic_dt_ops_t *ops,
                                   void *cookie, ucc_datatype_t *dt)
{
    if (!ops || !dt) {
        return UCC_ERR_INVALID_PARAM;
    }
    for (int i = 0; i < 64; i++) {
        int expect = 0;
        if (g_generic_dts[i].used.compare_exchange_strong(
                expect, 1, std::memory_order_acq_rel)) {
            g_generic_dts[i].ops    = *ops;
            g_generic_dts[i].cookie = cookie;
            g_generic_dts[i].used.store(2, std::memory_order_release);
            *dt = (ucc_datatype_t)(UCC_DT_USERDEFINED_BASE + i);
            return UCC_OK;
        }
    }
    return UCC_ERR_NO_RESOURCE;
}

void ucc_dt_destroy(ucc_datatype_t dt)
{
    int i = (int)dt - UCC_DT_USERDEFINED_BASE;
    if (i >= 0 && i < 64) {
        g_generic_dts[i].used.store(0, std::memory_order_release);
    }
}

int ucc_dt_is_predefined(ucc_datatype_t dt)
{
    return (int)dt < UCC_DT_PREDEFINED_LAST;
}

const ucc_generic_dt_ops_t *ucc_dt_generic_ops(ucc_datatype_t dt)
{
    int i = (int)dt - UCC_DT_USERDEFINED_BASE;
    if (i >= 0 && i < 64 &&
        g_generic_dts[i].used.load(std::memory_order_acquire) == 2) {
        return &g_generic_dts[i].ops;
    }
    return 0;
}

const char *ucc_status_string(ucc_status_t status)
{
    switch (status) {
    case UCC_OK: return "Success";
    case UCC_INPROGRESS: return "Operation in progress";
    case UCC_OPERATION_INITIALIZED: return "Operation initialized";
    case UCC_ERR_NOT_SUPPORTED: return "Not supported";
    case UCC_ERR_NOT_IMPLEMENTED: return "Not implemented";
    case UCC_ERR_INVALID_PARAM: return "Invalid parameter";
    case UCC_ERR_NO_MEMORY: return "Out of memory";
    case UCC_ERR_NO_RESOURCE: return "No resource";
    case UCC_ERR_NO_MESSAGE: return "No message";
    case UCC_ERR_NOT_FOUND: return "Not found";
    case UCC_ERR_TIMED_OUT: return "Timed out";
    default: return "Unknown error";
    }
}

void ucc_get_version(unsigned *major, unsigned *minor, unsigned *release)
{
    *major   = UCC_API_MAJOR;
    *minor   = UCC_API_MINOR;
    *release = 0;
}

const char *ucc_get_version_string(void) { return "ucc_amd 1.3.0 (gfx950)"; }

} /* extern "C" */
