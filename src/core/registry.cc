/* Static TL registry (replaces the reference's dlopen plugin loader,
 * core/ucc_constructor.c — design deviation documented in SURVEY.md §7). */
#include "core.h"

namespace ucc {

Tl *tl_self_iface();
Tl *tl_shm_iface();
Tl *tl_tcp_iface();
#ifdef UCC_AMD_HAS_TL_CDNA4
Tl *tl_cdna4_iface();
#endif
#ifdef UCC_AMD_HAS_TL_RCCL
Tl *tl_rccl_iface();
#endif

void ensure_builtin_tls()
{
    static bool done = [] {
        register_tl(tl_self_iface());
        register_tl(tl_shm_iface());
        register_tl(tl_tcp_iface());
#ifdef UCC_AMD_HAS_TL_CDNA4
        register_tl(tl_cdna4_iface());
#endif
#ifdef UCC_AMD_HAS_TL_RCCL
        register_tl(tl_rccl_iface());
#endif
        return true;
    }();
    (void)done;
}

} // namespace ucc
