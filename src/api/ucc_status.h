/* ucc_amd — MI355X-native collective communication library.
 *
 * Status codes. API-compatible with the reference surface
 * (see /root/reference/src/ucc/api/ucc_status.h for the parity target);
 * implementation is original.
 */
#ifndef UCC_AMD_STATUS_H_
#define UCC_AMD_STATUS_H_

#ifdef __cplusplus
extern "C" {
#endif

typedef enum {
    UCC_OK                          = 0,
    UCC_INPROGRESS                  = 1,
    UCC_OPERATION_INITIALIZED       = 2,
    UCC_ERR_NOT_SUPPORTED           = -1,
    UCC_ERR_NOT_IMPLEMENTED         = -2,
    UCC_ERR_INVALID_PARAM           = -3,
    UCC_ERR_NO_MEMORY               = -4,
    UCC_ERR_NO_RESOURCE             = -5,
    UCC_ERR_NO_MESSAGE              = -6,
    UCC_ERR_NOT_FOUND               = -7,
    UCC_ERR_TIMED_OUT               = -8,
    UCC_ERR_LAST                    = -100,
} ucc_status_t;

const char *ucc_status_string(ucc_status_t status);

#ifdef __cplusplus
}
#endif

#endif /* UCC_AMD_STATUS_H_ */
