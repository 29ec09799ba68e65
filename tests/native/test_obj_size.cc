/* Object-size regression guard (reference test/gtest/common/
 * test_obj_size.cc role, re-derived): the hot-path objects the progress
 * engine touches per collective must stay within cache-friendly bounds.
 * Growth past a bound is usually an accidentally-added fat member. */
#include <cstdio>
#include <cstdlib>

#include "../../src/core/core.h"

using namespace ucc;

#define CHECK_SIZE(T, max)                                                 \
    do {                                                                   \
        if (sizeof(T) > (max)) {                                           \
            fprintf(stderr, "sizeof(%s) = %zu > %zu\n", #T, sizeof(T),     \
                    (size_t)(max));                                        \
            fail = 1;                                                      \
        } else {                                                           \
            printf("%-24s %5zu / %zu\n", #T, sizeof(T), (size_t)(max));    \
        }                                                                  \
    } while (0)

int main()
{
    int fail = 0;
    /* progress-engine per-op objects */
    CHECK_SIZE(Task, 160);
    CHECK_SIZE(Schedule, 224);
    CHECK_SIZE(CollRequest, 512);
    CHECK_SIZE(OobRound, 160);
    CHECK_SIZE(ScoreRange, 128);
    /* team/context are per-entity (not per-op) but still bounded */
    CHECK_SIZE(ProcInfo, 32);
    CHECK_SIZE(ucc_coll_args_t, 384);
    if (fail) {
        return 1;
    }
    printf("OBJ_SIZE_OK\n");
    return 0;
}
