/* Object-size regression guard (reference test/gtest/common/
 * test_obj_size.cc role, re-derived): the hot-path objects the progress
 * engine touches per collective must stay within cache-friendly bounds.
 * Growth past a bound is usually an accidentally-added fat member. */
#include <cstdio>
#include <cstdlib>

#include "../../src/core/core.h"

using namespace ucc;

/* tiny API-attr smoke rides along with the size checks: lib/context/
 * team attr queries through the public C API (reference attr gtest
 * role) */
static int api_attr_smoke()
{
    ucc_lib_h        lib = nullptr;
    ucc_lib_params_t lp{};
    if (ucc_init(&lp, nullptr, &lib) != UCC_OK) {
        return 1;
    }
    ucc_lib_attr_t la{};
    la.mask = UCC_LIB_ATTR_FIELD_THREAD_MODE |
              UCC_LIB_ATTR_FIELD_COLL_TYPES;
    if (ucc_lib_get_attr(lib, &la) != UCC_OK ||
        la.coll_types != UCC_COLL_TYPE_ALL) {
        return 1;
    }
    ucc_context_h        ctx = nullptr;
    ucc_context_params_t cp{};
    if (ucc_context_create(lib, &cp, nullptr, &ctx) != UCC_OK) {
        return 1;
    }
    ucc_context_attr_t ca{};
    ca.mask = UCC_CONTEXT_ATTR_FIELD_TYPE;
    if (ucc_context_get_attr(ctx, &ca) != UCC_OK) {
        return 1;
    }
    ucc_team_h        team = nullptr;
    ucc_team_params_t tp{};
    tp.mask      = UCC_TEAM_PARAM_FIELD_TEAM_SIZE;
    tp.team_size = 1;
    if (ucc_team_create_post(&ctx, 1, &tp, &team) != UCC_OK) {
        return 1;
    }
    while (ucc_team_create_test(team) == UCC_INPROGRESS) {
    }
    ucc_team_attr_t ta{};
    ta.mask = UCC_TEAM_ATTR_FIELD_SIZE | UCC_TEAM_ATTR_FIELD_EP;
    if (ucc_team_get_attr(team, &ta) != UCC_OK || ta.size != 1 ||
        ta.ep != 0) {
        return 1;
    }
    ucc_team_destroy(team);
    ucc_context_destroy(ctx);
    ucc_finalize(lib);
    return 0;
}

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
    if (api_attr_smoke() != 0) {
        fprintf(stderr, "api attr smoke failed\n");
        return 1;
    }
    printf("OBJ_SIZE_OK\n");
    return 0;
}
