/* Datatype/op/status helpers. Parity target: reference ucc.h dtype table
 * (src/ucc/api/ucc.h:203-221) and ucc_status.h strings. */
#include "../api/ucc.h"

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
    default: return 0;
    }
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
