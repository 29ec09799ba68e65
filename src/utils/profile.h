/* Request-event profiler.
 * Reference parity: utils/profile/ (UCC_PROFILE_REQUEST_{NEW,EVENT,FREE}
 * macros over UCS profile, ucc_profile_on.h:34-98) — native
 * re-implementation: lock-free-ish in-process ring of fixed-size records,
 * enabled by UCC_PROFILE_MODE=log, dumped to UCC_PROFILE_FILE (default
 * ucc_profile.log) at process exit or ucc_profile_dump(). */
#ifndef UCC_AMD_PROFILE_H_
#define UCC_AMD_PROFILE_H_

#include <cstdint>

namespace ucc {
namespace prof {

enum class Ev : uint8_t {
    REQUEST_NEW = 0,
    REQUEST_EVENT,
    REQUEST_FREE,
    SCOPE_BEGIN,
    SCOPE_END,
};

bool enabled();
/* name must be a string literal / static string (stored by pointer). */
void record(Ev type, const char *name, uint64_t req_id);
void dump(); /* write + reset; also installed via atexit */

} // namespace prof
} // namespace ucc

#define UCC_PROFILE_REQUEST_NEW(name, id)                                    \
    do {                                                                     \
        if (::ucc::prof::enabled())                                          \
            ::ucc::prof::record(::ucc::prof::Ev::REQUEST_NEW, (name),        \
                                (uint64_t)(id));                             \
    } while (0)
#define UCC_PROFILE_REQUEST_EVENT(name, id)                                  \
    do {                                                                     \
        if (::ucc::prof::enabled())                                          \
            ::ucc::prof::record(::ucc::prof::Ev::REQUEST_EVENT, (name),      \
                                (uint64_t)(id));                             \
    } while (0)
#define UCC_PROFILE_REQUEST_FREE(name, id)                                   \
    do {                                                                     \
        if (::ucc::prof::enabled())                                          \
            ::ucc::prof::record(::ucc::prof::Ev::REQUEST_FREE, (name),       \
                                (uint64_t)(id));                             \
    } while (0)

#endif
