/* ucc_info — version / component / config introspection.
 * Reference parity: tools/info/ucc_info.c (-v version, -c configs,
 * -s score map of a simulated 2-rank team). */
#include <cstdio>
#include <cstring>
#include <vector>

#include "../src/api/ucc.h"
#include "../src/core/core.h"
#include "../src/utils/config.h"

using namespace ucc;

static void print_version()
{
    printf("# UCC-AMD (MI355X-native) version %s\n",
           ucc_get_version_string());
    printf("#  arch: gfx950 (CDNA4), transports: ");
    ensure_builtin_tls();
    for (auto *tl : tl_registry()) {
        printf("%s ", tl->name());
    }
    printf("\n");
}

static void print_configs()
{
    /* create a lib+context so every TL's context_create runs its
     * declare() calls (the dump should list ALL knobs, not just the
     * core table) */
    {
        ucc_lib_h        lib = nullptr;
        ucc_lib_params_t lp{};
        if (ucc_init(&lp, nullptr, &lib) == UCC_OK) {
            ucc_context_h        ctx = nullptr;
            ucc_context_params_t cp{};
            if (ucc_context_create(lib, &cp, nullptr, &ctx) == UCC_OK) {
                ucc_context_destroy(ctx);
            }
            ucc_finalize(lib);
        }
    }
    Config &c = Config::instance();
    printf("#\n# configuration variables (UCC_<COMPONENT>_<NAME>)\n#\n");
    auto entries = c.entries();
    for (auto &e : entries) {
        std::string key = Config::key(e.component, e.name);
        printf("%s=%s\n", key.c_str(),
               c.get(e.component, e.name, e.dflt).c_str());
        if (!e.doc.empty()) {
            printf("# %s\n", e.doc.c_str());
        }
    }
    if (entries.empty()) {
        printf("# (no components touched their config yet)\n");
    }
}

/* -s: create a simulated 2-rank team (in-process memcpy OOB) and print
 * its score map — what the selection logic will do on this machine. */
#include <cstring>
#include <map>

namespace {
struct Loob {
    struct Round {
        std::vector<std::vector<uint8_t>> blobs;
        int                               arrived = 0;
    };
    std::map<uint64_t, Round> rounds;
    uint64_t                  next[2] = {0, 0};
};
Loob g_loob;
struct LoobReq {
    uint64_t round;
    void    *recv;
    size_t   size;
};
ucc_status_t lo_ag(void *src, void *recv, size_t size, void *info,
                   void **req)
{
    int      rank  = (int)(intptr_t)info;
    uint64_t round = g_loob.next[rank]++;
    auto    &r     = g_loob.rounds[round];
    if (r.blobs.empty()) {
        r.blobs.resize(2);
    }
    r.blobs[rank].assign((uint8_t *)src, (uint8_t *)src + size);
    r.arrived++;
    *req = new LoobReq{round, recv, size};
    return UCC_OK;
}
ucc_status_t lo_test(void *req)
{
    auto *r  = (LoobReq *)req;
    auto &rd = g_loob.rounds[r->round];
    if (rd.arrived < 2) {
        return UCC_INPROGRESS;
    }
    for (int i = 0; i < 2; i++) {
        memcpy((uint8_t *)r->recv + i * r->size, rd.blobs[i].data(),
               r->size);
    }
    return UCC_OK;
}
ucc_status_t lo_free(void *req)
{
    delete (LoobReq *)req;
    return UCC_OK;
}
} // namespace

static void print_score_map()
{
    ucc_lib_h     libs[2];
    ucc_context_h ctxs[2];
    ucc_team_h    teams[2];
    for (int r = 0; r < 2; r++) {
        ucc_lib_params_t lp{};
        ucc_init(&lp, nullptr, &libs[r]);
        ucc_context_params_t cp{};
        ucc_context_create(libs[r], &cp, nullptr, &ctxs[r]);
        ucc_team_params_t tp{};
        tp.mask          = UCC_TEAM_PARAM_FIELD_OOB;
        tp.oob.allgather = lo_ag;
        tp.oob.req_test  = lo_test;
        tp.oob.req_free  = lo_free;
        tp.oob.coll_info = (void *)(intptr_t)r;
        tp.oob.n_oob_eps = 2;
        tp.oob.oob_ep    = r;
        ucc_team_create_post(&ctxs[r], 1, &tp, &teams[r]);
    }
    while (true) {
        ucc_status_t s0 = ucc_team_create_test(teams[0]);
        ucc_status_t s1 = ucc_team_create_test(teams[1]);
        if (s0 < 0 || s1 < 0) {
            printf("# team create failed (%d/%d)\n", s0, s1);
            return;
        }
        if (s0 == UCC_OK && s1 == UCC_OK) {
            break;
        }
    }
    auto *team = reinterpret_cast<Team *>(teams[0]);
    printf("#\n# score map of a simulated 2-rank team "
           "(coll:mem:range:@tl/alg:score)\n#\n%s",
           team->score_map.to_string().c_str());
    for (int r = 0; r < 2; r++) {
        ucc_team_destroy(teams[r]);
        ucc_context_destroy(ctxs[r]);
        ucc_finalize(libs[r]);
    }
}

int main(int argc, char **argv)
{
    bool ver = true, cfg = false, smap = false;
    for (int i = 1; i < argc; i++) {
        if (!strcmp(argv[i], "-c") || !strcmp(argv[i], "-caf")) {
            cfg = true;
        } else if (!strcmp(argv[i], "-v")) {
            ver = true;
        } else if (!strcmp(argv[i], "-s")) {
            smap = true;
        } else if (!strcmp(argv[i], "-h")) {
            printf("ucc_info [-v] [-c] [-s]\n");
            return 0;
        }
    }
    if (ver) {
        print_version();
    }
    if (cfg) {
        print_configs();
    }
    if (smap) {
        print_score_map();
    }
    return 0;
}
