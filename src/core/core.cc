/* Public API implementation: lib/context/team/collective lifecycle.
 * Parity targets: reference core/ucc_lib.c, core/ucc_context.c,
 * core/ucc_team.c (nonblocking create state machine), core/ucc_coll.c
 * (score-map dispatch, zero-size fast path, persistent re-post). */
#include "core.h"
#include "../mc/mc.h"
#include "../utils/profile.h"

#include <atomic>
#include <algorithm>
#include <cstdlib>
#include <random>
#include <unistd.h>

namespace ucc {

/* ------------------------------------------------------------- ProcInfo */
static uint64_t str_hash(const char *s)
{
    uint64_t h = 1469598103934665603ull;
    while (*s) {
        h = (h ^ (uint8_t)*s++) * 1099511628211ull;
    }
    return h;
}

/* read one small integer file (sysfs); -1 on failure */
static int read_int_file(const char *path)
{
    FILE *f = fopen(path, "r");
    if (!f) {
        return -1;
    }
    int  v  = -1;
    int  ok = fscanf(f, "%d", &v);
    fclose(f);
    return ok == 1 ? v : -1;
}

ProcInfo local_proc_info()
{
    ProcInfo pi;
    char     host[256] = {0};
    gethostname(host, sizeof(host) - 1);
    pi.host_hash = str_hash(host);
    pi.pid       = (int32_t)getpid();
    pi.device    = -1;
    /* socket/NUMA of the cpu this process currently runs on
     * (reference ucc_proc_info sysfs role). Best effort: stays -1
     * where sysfs is unavailable. */
    int cpu = sched_getcpu();
    if (cpu >= 0) {
        char path[128];
        snprintf(path, sizeof(path),
                 "/sys/devices/system/cpu/cpu%d/topology/"
                 "physical_package_id", cpu);
        pi.socket_id = (int16_t)read_int_file(path);
        for (int nd = 0; nd < 64; nd++) {
            snprintf(path, sizeof(path),
                     "/sys/devices/system/node/node%d/cpumap", nd);
            FILE *f = fopen(path, "r");
            if (!f) {
                break;
            }
            /* cpumap is a comma-separated hex mask, lowest word last */
            char mask[512] = {0};
            if (fgets(mask, sizeof(mask), f)) {
                /* walk 32-bit words from the right */
                int         bit   = 0;
                bool        found = false;
                const char *end   = mask + strlen(mask);
                const char *q     = end;
                while (q > mask && !found) {
                    const char *comma = q;
                    while (comma > mask && comma[-1] != ',') {
                        comma--;
                    }
                    uint32_t w = (uint32_t)strtoul(comma, nullptr, 16);
                    if (cpu >= bit && cpu < bit + 32 &&
                        (w >> (cpu - bit)) & 1u) {
                        pi.numa_id = (int16_t)nd;
                        found      = true;
                    }
                    bit += 32;
                    q = comma > mask ? comma - 1 : mask;
                }
            }
            fclose(f);
            if (pi.numa_id >= 0) {
                break;
            }
        }
    }
    /* CPU model consensus hash (vendor+model from /proc/cpuinfo) */
    FILE *ci = fopen("/proc/cpuinfo", "r");
    if (ci) {
        char line[256];
        while (fgets(line, sizeof(line), ci)) {
            if (!strncmp(line, "model name", 10) ||
                !strncmp(line, "vendor_id", 9)) {
                pi.cpu_hash =
                    (uint32_t)(str_hash(line) ^ (pi.cpu_hash * 31));
                break;
            }
        }
        fclose(ci);
    }
    return pi;
}

Team::~Team()
{
    hier::destroy(this);
    if (split_oob) {
        subooob_api::free_(split_oob);
        split_oob = nullptr;
    }
}

bool Team::all_same_node() const
{
    for (const auto &p : procs) {
        if (p.host_hash != procs[0].host_hash) {
            return false;
        }
    }
    return true;
}

bool Team::all_have_device() const
{
    for (const auto &p : procs) {
        if (p.device < 0) {
            return false;
        }
    }
    return true;
}

/* ------------------------------------------------------------- OobRound */
ucc_status_t OobRound::start(const void *src, size_t size)
{
    send_.assign((const uint8_t *)src, (const uint8_t *)src + size);
    recv_.resize(size * oob_.n_oob_eps);
    ucc_status_t st = oob_.allgather(send_.data(), recv_.data(), size,
                                     oob_.coll_info, &req_);
    if (st != UCC_OK) {
        return st;
    }
    active_ = true;
    return UCC_OK;
}

ucc_status_t OobRound::test()
{
    if (!active_) {
        return UCC_ERR_INVALID_PARAM;
    }
    ucc_status_t st = oob_.req_test(req_);
    if (st == UCC_INPROGRESS) {
        return st;
    }
    oob_.req_free(req_);
    req_    = nullptr;
    active_ = false;
    return st;
}

/* ------------------------------------------------------------ registry  */
std::vector<Tl *> &tl_registry()
{
    static std::vector<Tl *> reg;
    return reg;
}

void register_tl(Tl *tl) { tl_registry().push_back(tl); }

} // namespace ucc

using namespace ucc;

/* ===================================================== lib object (C API) */
struct ucc_lib_config {
    std::string prefix;
};
struct ucc_context_config {
    Lib *lib;
};

extern "C" {

ucc_status_t ucc_lib_config_read(const char *env_prefix, const char *filename,
                                 ucc_lib_config_h *config)
{
    (void)filename;
    auto *c   = new ucc_lib_config;
    c->prefix = env_prefix ? env_prefix : "";
    *config   = c;
    return UCC_OK;
}

void ucc_lib_config_release(ucc_lib_config_h config) { delete config; }

ucc_status_t ucc_lib_config_modify(ucc_lib_config_h config, const char *name,
                                   const char *value)
{
    (void)config;
    Config::instance().set("", name, value);
    return UCC_OK;
}

void ucc_lib_config_print(const ucc_lib_config_h config, void *stream,
                          const char *title, int print_flags)
{
    (void)config;
    (void)print_flags;
    FILE *f = stream ? (FILE *)stream : stdout;
    fprintf(f, "# %s\n", title ? title : "ucc_amd config");
    for (const auto &e : Config::instance().entries()) {
        fprintf(f, "%s=%s  # %s\n",
                Config::key(e.component, e.name).c_str(),
                Config::instance().get(e.component, e.name, e.dflt).c_str(),
                e.doc.c_str());
    }
}

ucc_status_t ucc_init_version(unsigned api_major, unsigned api_minor,
                              const ucc_lib_params_t *params,
                              const ucc_lib_config_h  config,
                              ucc_lib_h *lib_p)
{
    (void)config;
    if (api_major != UCC_API_MAJOR || api_minor > UCC_API_MINOR) {
        ucc_warn("requested api %u.%u, library is %u.%u", api_major, api_minor,
                 UCC_API_MAJOR, UCC_API_MINOR);
    }
    ensure_builtin_tls();
    auto *lib = new Lib;
    if (params) {
        lib->params = *params;
        if (params->mask & UCC_LIB_PARAM_FIELD_THREAD_MODE) {
            lib->thread_mode = params->thread_mode;
        }
    }
    *lib_p = reinterpret_cast<ucc_lib_h>(lib);
    return UCC_OK;
}

ucc_status_t ucc_finalize(ucc_lib_h lib)
{
    delete reinterpret_cast<Lib *>(lib);
    return UCC_OK;
}

ucc_status_t ucc_lib_get_attr(ucc_lib_h lib, ucc_lib_attr_t *attr)
{
    auto *l = reinterpret_cast<Lib *>(lib);
    if (attr->mask & UCC_LIB_ATTR_FIELD_THREAD_MODE) {
        attr->thread_mode = l->thread_mode;
    }
    if (attr->mask & UCC_LIB_ATTR_FIELD_COLL_TYPES) {
        attr->coll_types = UCC_COLL_TYPE_ALL;
    }
    if (attr->mask & UCC_LIB_ATTR_FIELD_REDUCTION_TYPES) {
        attr->reduction_types =
            (1u << UCC_OP_SUM) | (1u << UCC_OP_PROD) | (1u << UCC_OP_MAX) |
            (1u << UCC_OP_MIN) | (1u << UCC_OP_LAND) | (1u << UCC_OP_LOR) |
            (1u << UCC_OP_LXOR) | (1u << UCC_OP_BAND) |
            (1u << UCC_OP_BOR) | (1u << UCC_OP_BXOR) | (1u << UCC_OP_AVG);
    }
    return UCC_OK;
}

/* ---------------------------------------------------------------- context */
ucc_status_t ucc_context_config_read(ucc_lib_h lib, const char *filename,
                                     ucc_context_config_h *config)
{
    (void)filename;
    auto *c = new ucc_context_config;
    c->lib  = reinterpret_cast<Lib *>(lib);
    *config = c;
    return UCC_OK;
}

void ucc_context_config_release(ucc_context_config_h config) { delete config; }

ucc_status_t ucc_context_config_modify(ucc_context_config_h config,
                                       const char *component,
                                       const char *name, const char *value)
{
    (void)config;
    Config::instance().set(component ? component : "", name, value);
    return UCC_OK;
}

void ucc_context_config_print(const ucc_context_config_h config, void *stream,
                              const char *title, int print_flags)
{
    ucc_lib_config_print(nullptr, stream, title, print_flags);
    (void)config;
}

ucc_status_t ucc_context_create(ucc_lib_h lib_h,
                                const ucc_context_params_t *params,
                                const ucc_context_config_h  config,
                                ucc_context_h *context)
{
    (void)config;
    auto *lib = reinterpret_cast<Lib *>(lib_h);
    auto *ctx = new Context;
    ctx->lib  = lib;
    if (params) {
        ctx->params = *params;
        ctx->has_oob = params->mask & UCC_CONTEXT_PARAM_FIELD_OOB;
    }
    ctx->mt   = lib->thread_mode == UCC_THREAD_MULTIPLE;
    ctx->lock_free =
        Config::instance().get_bool("", "LOCK_FREE_PROGRESS_Q", true);
    ctx->seq  = lib->next_ctx_seq++;
    ctx->proc = local_proc_info();
    ctx->proc.ctx_seq = ((uint64_t)ctx->proc.pid << 20) | ctx->seq;
    /* test hook: UCC_FAKE_NODE_SPLIT=k spreads in-process contexts over
     * k pseudo-nodes so the hier composition is exercisable on one
     * machine (tests/test_hier.py) */
    if (const char *fs = getenv("UCC_FAKE_NODE_SPLIT")) {
        int k = atoi(fs);
        if (k > 1) {
            static std::atomic<uint64_t> g_fake_ctx{0};
            uint64_t idx = g_fake_ctx.fetch_add(1);
            ctx->proc.host_hash ^=
                0x9e3779b97f4a7c15ull * (1 + idx % (uint64_t)k);
        }
    }
    /* test hook: UCC_FAKE_SOCKET_SPLIT=k spreads in-process contexts
     * over k pseudo-sockets of one node, so socket-aware staging (shm
     * two-level bcast, SOCKET sbgps) is exercisable anywhere */
    if (const char *ss = getenv("UCC_FAKE_SOCKET_SPLIT")) {
        int k = atoi(ss);
        if (k > 1) {
            static std::atomic<uint64_t> g_fake_sck{0};
            ctx->proc.socket_id =
                (int16_t)(g_fake_sck.fetch_add(1) % (uint64_t)k);
        }
    }
    for (Tl *tl : tl_registry()) {
        TlContext *tlc = tl->context_create(ctx);
        if (tlc) {
            ctx->tl_ctxs.emplace_back(tlc);
            ucc_debug("context %p: tl %s available", (void *)ctx, tl->name());
        }
    }
    *context = reinterpret_cast<ucc_context_h>(ctx);
    return UCC_OK;
}

ucc_status_t ucc_context_destroy(ucc_context_h context)
{
    delete reinterpret_cast<Context *>(context);
    return UCC_OK;
}

ucc_status_t ucc_context_get_attr(ucc_context_h context,
                                  ucc_context_attr_t *attr)
{
    (void)context;
    if (attr->mask & UCC_CONTEXT_ATTR_FIELD_CTX_ADDR_LEN) {
        attr->ctx_addr_len = 0;
    }
    return UCC_OK;
}

ucc_status_t ucc_context_progress(ucc_context_h context)
{
    return reinterpret_cast<Context *>(context)->progress();
}

/* ------------------------------------------------------------------ team */
ucc_status_t ucc_team_create_post(ucc_context_h *contexts,
                                  uint32_t num_contexts,
                                  const ucc_team_params_t *team_params,
                                  ucc_team_h *new_team)
{
    if (num_contexts != 1 || !contexts || !team_params) {
        return UCC_ERR_NOT_SUPPORTED; /* one context per process */
    }
    auto *ctx  = reinterpret_cast<Context *>(contexts[0]);
    auto *team = new Team;
    team->ctx        = ctx;
    team->is_subteam = hier::creating_subteam();
    team->params = *team_params;
    team->has_oob = team_params->mask & UCC_TEAM_PARAM_FIELD_OOB;
    if (team->has_oob) {
        team->oob  = team_params->oob;
        team->rank = team->oob.oob_ep;
        team->size = team->oob.n_oob_eps;
    } else if ((team_params->mask & UCC_TEAM_PARAM_FIELD_TEAM_SIZE) &&
               team_params->team_size == 1) {
        team->rank = 0;
        team->size = 1;
    } else {
        delete team;
        return UCC_ERR_INVALID_PARAM; /* OOB required for size > 1 */
    }
    team->id = (team_params->mask & UCC_TEAM_PARAM_FIELD_ID)
                   ? team_params->id
                   : (uint16_t)(ctx->next_team_id++);

    if (team->size == 1) {
        team->procs.assign(1, ctx->proc);
        team->team_uid = ctx->proc.ctx_seq;
        team->state    = Team::TL_CREATE;
        /* create TL teams immediately (self at least) */
        for (auto &tlc : ctx->tl_ctxs) {
            TlTeam *tt = tlc->iface()->team_create(tlc.get(), team);
            if (tt) {
                team->tl_teams.emplace_back(tt);
            }
        }
    } else {
        /* round 1: proc info + rank0's uid proposal */
        struct R1 {
            ProcInfo pi;
            uint64_t uid;
        } r1;
        r1.pi = ctx->proc;
        std::random_device rd;
        r1.uid = ((uint64_t)rd() << 32) ^ rd() ^
                 ((uint64_t)ctx->proc.pid << 16) ^ ctx->seq;
        team->oobr.init(team->oob);
        ucc_status_t st = team->oobr.start(&r1, sizeof(r1));
        if (st != UCC_OK) {
            delete team;
            return st;
        }
        team->state = Team::ADDR_EXCHANGE;
    }
    *new_team = reinterpret_cast<ucc_team_h>(team);
    return UCC_OK;
}

/* ---- team split (reference ucc.h ucc_team_create_from_parent):
 * round 0 exchanges {included, ep} over the parent team; members then
 * run the normal create flow over a SubOob (parent-padded rounds);
 * observers drive matching padded rounds and finish as a size-0 stub. */
ucc_status_t ucc_team_create_from_parent(uint64_t my_ep, uint32_t included,
                                         ucc_team_h parent_h,
                                         ucc_team_h *new_team)
{
    auto *parent = reinterpret_cast<Team *>(parent_h);
    if (!parent || parent->state != Team::ACTIVE || !parent->has_oob) {
        return UCC_ERR_INVALID_PARAM;
    }
    auto *team           = new Team;
    team->ctx            = parent->ctx;
    team->is_subteam     = true;
    team->split_parent   = parent;
    team->split_included = included;
    team->split_ep       = my_ep;
    struct M {
        uint32_t included;
        uint32_t pad;
        uint64_t ep;
    } m{included, 0, my_ep};
    team->oobr.init(parent->oob);
    ucc_status_t st = team->oobr.start(&m, sizeof(m));
    if (st != UCC_OK) {
        delete team;
        return st;
    }
    team->state = Team::SPLIT_MEMBERS;
    *new_team   = reinterpret_cast<ucc_team_h>(team);
    return UCC_OK;
}

ucc_status_t ucc_team_create_test(ucc_team_h team_h)
{
    auto *team = reinterpret_cast<Team *>(team_h);
    auto *ctx  = team->ctx;

    switch (team->state) {
    case Team::SPLIT_MEMBERS: {
        ucc_status_t st = team->oobr.test();
        if (st == UCC_INPROGRESS) {
            return UCC_INPROGRESS;
        }
        if (st != UCC_OK) {
            team->state = Team::FAILED;
            return (team->err = st);
        }
        struct M {
            uint32_t included;
            uint32_t pad;
            uint64_t ep;
        };
        const M *all = (const M *)team->oobr.data();
        Team    *par = team->split_parent;
        /* members ordered by (ep, parent rank) */
        std::vector<std::pair<uint64_t, uint32_t>> mem;
        for (uint32_t r = 0; r < par->size; r++) {
            if (all[r].included) {
                mem.push_back({all[r].ep, r});
            }
        }
        std::sort(mem.begin(), mem.end());
        std::vector<uint32_t> members;
        int                   my_idx = -1;
        for (size_t i = 0; i < mem.size(); i++) {
            members.push_back(mem[i].second);
            if (mem[i].second == par->rank) {
                my_idx = (int)i;
            }
        }
        if (!team->split_included || my_idx < 0) {
            /* observer: mirror the members' bootstrap rounds */
            team->split_oob = subooob_api::make(par, members, -1);
            team->split_observe_rounds = members.size() >= 2 ? 2 : 0;
            team->rank = 0;
            team->size = 0;
            team->state = Team::SPLIT_OBSERVE;
            return ucc_team_create_test(team_h);
        }
        team->split_oob = subooob_api::make(par, members, my_idx);
        subooob_api::fill_oob(team->split_oob, &team->oob);
        team->has_oob = true;
        team->rank    = (uint32_t)my_idx;
        team->size    = (uint32_t)members.size();
        team->id      = (uint16_t)(ctx->next_team_id++);
        if (team->size == 1) {
            team->procs.assign(1, ctx->proc);
            team->team_uid = ctx->proc.ctx_seq ^ 0x5b17;
            team->state    = Team::TL_CREATE;
            for (auto &tlc : ctx->tl_ctxs) {
                TlTeam *tt =
                    tlc->iface()->team_create(tlc.get(), team);
                if (tt) {
                    team->tl_teams.emplace_back(tt);
                }
            }
            return ucc_team_create_test(team_h);
        }
        struct R1 {
            ProcInfo pi;
            uint64_t uid;
        } r1;
        r1.pi = ctx->proc;
        std::random_device rd;
        r1.uid = ((uint64_t)rd() << 32) ^ rd() ^
                 ((uint64_t)ctx->proc.pid << 16) ^ ctx->seq;
        team->oobr.init(team->oob);
        ucc_status_t rs = team->oobr.start(&r1, sizeof(r1));
        if (rs != UCC_OK) {
            team->state = Team::FAILED;
            return (team->err = rs);
        }
        team->state = Team::ADDR_EXCHANGE;
        return UCC_INPROGRESS;
    }
    case Team::SPLIT_OBSERVE: {
        ucc_status_t st = subooob_api::observe_tick(
            team->split_oob, team->split_observe_rounds);
        if (st == UCC_INPROGRESS) {
            return UCC_INPROGRESS;
        }
        if (st != UCC_OK) {
            team->state = Team::FAILED;
            return (team->err = st);
        }
        team->state = Team::ACTIVE;
        return UCC_OK;
    }
    case Team::ADDR_EXCHANGE: {
        ucc_status_t st = team->oobr.test();
        if (st == UCC_INPROGRESS) {
            return UCC_INPROGRESS;
        }
        if (st != UCC_OK) {
            team->state = Team::FAILED;
            return (team->err = st);
        }
        struct R1 {
            ProcInfo pi;
            uint64_t uid;
        };
        const R1 *all = (const R1 *)team->oobr.data();
        team->procs.resize(team->size);
        for (uint32_t i = 0; i < team->size; i++) {
            team->procs[i] = all[i].pi;
        }
        team->team_uid = all[0].uid;
        /* create TL team objects and the combined exchange blob */
        for (auto &tlc : ctx->tl_ctxs) {
            TlTeam *tt = tlc->iface()->team_create(tlc.get(), team);
            if (tt) {
                team->tl_teams.emplace_back(tt);
            }
        }
        size_t off = 0;
        team->exchg_off.clear();
        for (auto &tt : team->tl_teams) {
            team->exchg_off.push_back(off);
            off += tt->exchg_size();
        }
        /* Always run the TL exchange round (pad zero strides): team
         * create is then EXACTLY two OOB rounds on every team, which
         * keeps parent-OOB round sequences positionally aligned for the
         * hier sub-team bootstrap (src/cl/hier.cc). */
        if (off == 0) {
            off = 8;
        }
        team->exchg_stride = off;
        team->exchg_buf.assign(off, 0);
        for (size_t i = 0; i < team->tl_teams.size(); i++) {
            team->tl_teams[i]->exchg_pack(team->exchg_buf.data() +
                                          team->exchg_off[i]);
        }
        ucc_status_t st2 =
            team->oobr.start(team->exchg_buf.data(), team->exchg_stride);
        if (st2 != UCC_OK) {
            team->state = Team::FAILED;
            return (team->err = st2);
        }
        team->state = Team::TL_EXCHANGE;
        return UCC_INPROGRESS;
    }
    case Team::TL_EXCHANGE: {
        ucc_status_t st = team->oobr.test();
        if (st == UCC_INPROGRESS) {
            return UCC_INPROGRESS;
        }
        if (st != UCC_OK) {
            team->state = Team::FAILED;
            return (team->err = st);
        }
        const uint8_t *all = (const uint8_t *)team->oobr.data();
        for (size_t i = 0; i < team->tl_teams.size();) {
            ucc_status_t us = team->tl_teams[i]->exchg_unpack(
                all + team->exchg_off[i], team->exchg_stride);
            if (us != UCC_OK) {
                ucc_warn("tl %s team exchange failed (%d), dropping",
                         team->tl_teams[i]->tlc_->iface()->name(), us);
                team->tl_teams.erase(team->tl_teams.begin() + i);
                team->exchg_off.erase(team->exchg_off.begin() + i);
            } else {
                i++;
            }
        }
        team->state = Team::TL_CREATE;
        return ucc_team_create_test(team_h);
    }
    case Team::TL_CREATE: {
        bool all_done = true;
        for (size_t i = 0; i < team->tl_teams.size();) {
            ucc_status_t st = team->tl_teams[i]->create_test();
            if (st == UCC_INPROGRESS) {
                all_done = false;
                i++;
            } else if (st != UCC_OK) {
                ucc_warn("tl %s team create failed (%d), dropping",
                         team->tl_teams[i]->tlc_->iface()->name(), st);
                team->tl_teams.erase(team->tl_teams.begin() + i);
            } else {
                i++;
            }
        }
        if (!all_done) {
            return UCC_INPROGRESS;
        }
        if (team->tl_teams.empty()) {
            team->state = Team::FAILED;
            return (team->err = UCC_ERR_NO_RESOURCE);
        }
        if (hier::wanted(team)) {
            ucc_status_t hs = hier::setup(team);
            if (hs == UCC_OK) {
                team->want_hier = true;
                team->state     = Team::HIER_CREATE;
                return ucc_team_create_test(team_h);
            }
            ucc_warn("hier setup declined (%d), flat team", hs);
        }
        for (auto &tt : team->tl_teams) {
            tt->get_scores(team, team->score_map);
        }
        std::string tune = Config::instance().get("", "TUNE", "");
        if (!tune.empty()) {
            team->score_map.apply_str(tune);
        }
        if (log_level() >= LogLevel::INFO) {
            ucc_info("team %u size %u score map:\n%s", team->id, team->size,
                     team->score_map.to_string().c_str());
        }
        team->state = Team::ACTIVE;
        return UCC_OK;
    }
    case Team::HIER_CREATE: {
        ucc_status_t st = hier::test(team);
        if (st == UCC_INPROGRESS) {
            return UCC_INPROGRESS;
        }
        if (st != UCC_OK) {
            team->state = Team::FAILED;
            return (team->err = st);
        }
        for (auto &tt : team->tl_teams) {
            tt->get_scores(team, team->score_map);
        }
        hier::add_scores(team);
        std::string tune2 = Config::instance().get("", "TUNE", "");
        if (!tune2.empty()) {
            team->score_map.apply_str(tune2);
        }
        if (log_level() >= LogLevel::INFO) {
            ucc_info("team %u size %u score map:\n%s", team->id,
                     team->size, team->score_map.to_string().c_str());
        }
        team->state = Team::ACTIVE;
        return UCC_OK;
    }
    case Team::ACTIVE: return UCC_OK;
    case Team::FAILED: return team->err;
    }
    return UCC_ERR_INVALID_PARAM;
}

ucc_status_t ucc_team_destroy(ucc_team_h team_h)
{
    delete reinterpret_cast<Team *>(team_h);
    return UCC_OK;
}

ucc_status_t ucc_team_get_attr(ucc_team_h team_h, ucc_team_attr_t *attr)
{
    auto *team = reinterpret_cast<Team *>(team_h);
    if (attr->mask & UCC_TEAM_ATTR_FIELD_EP) {
        attr->ep = team->rank;
    }
    if (attr->mask & UCC_TEAM_ATTR_FIELD_SIZE) {
        attr->size = team->size;
    }
    if (attr->mask & UCC_TEAM_ATTR_FIELD_EP_RANGE) {
        attr->ep_range = UCC_COLLECTIVE_EP_RANGE_CONTIG;
    }
    return UCC_OK;
}

ucc_status_t ucc_team_get_size(ucc_team_h team_h, uint32_t *size)
{
    *size = reinterpret_cast<Team *>(team_h)->size;
    return UCC_OK;
}

ucc_status_t ucc_team_get_my_ep(ucc_team_h team_h, uint64_t *ep)
{
    *ep = reinterpret_cast<Team *>(team_h)->rank;
    return UCC_OK;
}

ucc_status_t ucc_team_get_all_eps(ucc_team_h team_h, uint64_t **ep,
                                  uint64_t *num_eps)
{
    (void)team_h; (void)ep; (void)num_eps;
    return UCC_ERR_NOT_IMPLEMENTED;
}

/* ------------------------------------------------------------ collective */
namespace {
class StubTask final : public Task {
  public:
    using Task::Task;
    ucc_status_t post() override { return UCC_OK; }
};
} // namespace

ucc_status_t ucc_collective_init(ucc_coll_args_t *coll_args,
                                 ucc_coll_req_h *request, ucc_team_h team_h)
{
    auto *team = reinterpret_cast<Team *>(team_h);
    if (team->state != Team::ACTIVE) {
        return UCC_ERR_INVALID_PARAM;
    }
    auto  *req = new CollRequest;
    req->args  = *coll_args;
    req->team  = team;
    req->persistent = coll_args->mask & UCC_COLL_ARGS_FIELD_FLAGS &&
                      (coll_args->flags & UCC_COLL_ARGS_FLAG_PERSISTENT);
    if (!(coll_args->mask & UCC_COLL_ARGS_FIELD_FLAGS)) {
        req->args.flags = 0;
    }
    size_t msgsize = coll_args_msgsize(req->args, team->rank, team->size);
    /* asymmetric src/dst memory spaces are not staged (reference
     * ucc_coll.c:236-246 scratch staging): reject cleanly rather than
     * let a device kernel touch a host pointer */
    {
        auto fold = [](ucc_memory_type_t m) {
            if (m == UCC_MEMORY_TYPE_ROCM) {
                return UCC_MEMORY_TYPE_CUDA;
            }
            if (m == UCC_MEMORY_TYPE_ROCM_MANAGED) {
                return UCC_MEMORY_TYPE_CUDA_MANAGED;
            }
            return m;
        };
        const ucc_coll_args_t &a = req->args;
        bool has_both =
            a.coll_type != UCC_COLL_TYPE_BARRIER &&
            a.coll_type != UCC_COLL_TYPE_FANIN &&
            a.coll_type != UCC_COLL_TYPE_FANOUT &&
            a.coll_type != UCC_COLL_TYPE_BCAST &&
            !(a.mask & UCC_COLL_ARGS_FIELD_FLAGS &&
              (a.flags & UCC_COLL_ARGS_FLAG_IN_PLACE));
        if (has_both) {
            bool sv = a.coll_type == UCC_COLL_TYPE_ALLTOALLV ||
                      a.coll_type == UCC_COLL_TYPE_SCATTERV;
            bool dv = a.coll_type == UCC_COLL_TYPE_ALLTOALLV ||
                      a.coll_type == UCC_COLL_TYPE_ALLGATHERV ||
                      a.coll_type == UCC_COLL_TYPE_GATHERV ||
                      a.coll_type == UCC_COLL_TYPE_REDUCE_SCATTERV;
            ucc_memory_type_t ms =
                fold(sv ? a.src.info_v.mem_type : a.src.info.mem_type);
            ucc_memory_type_t md =
                fold(dv ? a.dst.info_v.mem_type : a.dst.info.mem_type);
            /* rooted colls: the non-root side may pass only one valid
             * buffer; only the root's pair is meaningful */
            bool rooted = a.coll_type == UCC_COLL_TYPE_REDUCE ||
                          a.coll_type == UCC_COLL_TYPE_GATHER ||
                          a.coll_type == UCC_COLL_TYPE_GATHERV ||
                          a.coll_type == UCC_COLL_TYPE_SCATTER ||
                          a.coll_type == UCC_COLL_TYPE_SCATTERV;
            bool meaningful = !rooted || a.root == team->rank;
            if (meaningful && ms != md &&
                ucc_dt_is_predefined(sv ? a.src.info_v.datatype
                                        : a.src.info.datatype)) {
                /* stage src into a scratch of the dst's memtype, copied
                 * at every post (reference ucc_coll.c:236-246 asymmetric
                 * staging, re-derived: scratch lives dst-side so no
                 * copy-back is needed and device TLs stay eligible) */
                size_t sbytes = 0;
                size_t dtsz   = ucc_dt_size(sv ? a.src.info_v.datatype
                                               : a.src.info.datatype);
                if (sv) { /* displacement layout: copy the full extent */
                    for (uint32_t r = 0; r < team->size; r++) {
                        uint64_t cn =
                            ((a.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
                                 ? ((const uint64_t *)
                                        a.src.info_v.counts)[r]
                                 : ((const uint32_t *)
                                        a.src.info_v.counts)[r]);
                        uint64_t dp =
                            ((a.flags &
                              UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
                                 ? ((const uint64_t *)
                                        a.src.info_v.displacements)[r]
                                 : ((const uint32_t *)
                                        a.src.info_v.displacements)[r]);
                        size_t e = (size_t)(dp + cn) * dtsz;
                        sbytes   = e > sbytes ? e : sbytes;
                    }
                } else {
                    sbytes = (size_t)a.src.info.count * dtsz;
                    if (a.coll_type == UCC_COLL_TYPE_ALLGATHER ||
                        a.coll_type == UCC_COLL_TYPE_GATHER) {
                        /* src side is one block; tolerate callers that
                         * set src.count to the gathered total */
                        size_t blk =
                            ((size_t)a.dst.info.count / team->size) *
                            dtsz;
                        sbytes = blk < sbytes ? blk : sbytes;
                    }
                }
                void *scratch = nullptr;
                if (sbytes &&
                    mc::scratch_alloc(&scratch, sbytes, md) != UCC_OK) {
                    delete req;
                    return UCC_ERR_NO_MEMORY;
                }
                req->asymm_scratch = scratch;
                req->asymm_bytes   = sbytes;
                req->asymm_mt      = md;
                const void *usrc   = sv ? a.src.info_v.buffer
                                        : a.src.info.buffer;
                ucc_memory_type_t umt = ms;
                req->pre_post = [scratch, usrc, sbytes, md, umt]() {
                    return sbytes ? mc::copy(scratch, md, usrc, umt,
                                             sbytes)
                                  : UCC_OK;
                };
                if (sv) {
                    req->args.src.info_v.buffer   = scratch;
                    req->args.src.info_v.mem_type =
                        a.dst.info_v.mem_type;
                } else {
                    req->args.src.info.buffer   = scratch;
                    req->args.src.info.mem_type =
                        dv ? a.dst.info_v.mem_type : a.dst.info.mem_type;
                }
            }
        }
    }
    /* Zero-size fast path only where a local zero implies a global zero
     * (fixed-count colls; v-variants may be locally empty but must still
     * take part in the exchange). Reference: ucc_coll.c:191-208. */
    const bool v_coll = req->args.coll_type == UCC_COLL_TYPE_ALLGATHERV ||
                        req->args.coll_type == UCC_COLL_TYPE_ALLTOALLV ||
                        req->args.coll_type == UCC_COLL_TYPE_GATHERV ||
                        req->args.coll_type == UCC_COLL_TYPE_SCATTERV ||
                        req->args.coll_type == UCC_COLL_TYPE_REDUCE_SCATTERV;
    const ucc_generic_dt_ops_t *ncg =
        ucc_dt_generic_ops(req->args.src.info.datatype);
    const bool noncontig_gen =
        ncg && !(ncg->flags & UCC_GENERIC_DT_OPS_FLAG_CONTIG);
    /* Non-contiguous generic dt, data-movement colls: pack the local
     * contribution into a packed byte image, run the collective on
     * plain bytes, unpack at completion (reference generic-dt
     * pack/unpack role generalized from the tl path; assumes the type's
     * packed size is content-independent — homogeneous elements — so
     * every rank computes identical packed block sizes). Host memory
     * only (pack callbacks are host functions). Bcast keeps the tl/tcp
     * packed-image tree (handles size-dynamic packs); reduce-family
     * colls cannot run on packed bytes. */
    if (noncontig_gen &&
        (req->args.coll_type == UCC_COLL_TYPE_ALLGATHER ||
         req->args.coll_type == UCC_COLL_TYPE_ALLTOALL ||
         req->args.coll_type == UCC_COLL_TYPE_GATHER ||
         req->args.coll_type == UCC_COLL_TYPE_SCATTER) &&
        req->args.src.info.mem_type == UCC_MEMORY_TYPE_HOST &&
        ncg->ops.start_pack && ncg->ops.start_unpack &&
        ncg->ops.packed_size && ncg->ops.pack && ncg->ops.unpack) {
        auto pack_all = [ncg](const void *buf, size_t count, uint8_t *out,
                              size_t psz) -> ucc_status_t {
            void *obj = ncg->ops.start_pack(ncg->cookie, buf, count);
            size_t got = 0;
            while (got < psz) {
                size_t len = psz - got;
                if (ncg->ops.pack(obj, got, out + got, &len) != UCC_OK ||
                    len == 0) {
                    if (ncg->ops.finish) {
                        ncg->ops.finish(obj);
                    }
                    return UCC_ERR_NO_MESSAGE;
                }
                got += len;
            }
            if (ncg->ops.finish) {
                ncg->ops.finish(obj);
            }
            return UCC_OK;
        };
        auto unpack_all = [ncg](void *buf, size_t count,
                                const uint8_t *in,
                                size_t psz) -> ucc_status_t {
            void *obj = ncg->ops.start_unpack(ncg->cookie, buf, count);
            ucc_status_t st = ncg->ops.unpack(obj, 0, in, psz);
            if (ncg->ops.finish) {
                ncg->ops.finish(obj);
            }
            return st;
        };
        const ucc_coll_args_t &a    = req->args;
        const uint32_t         nt   = team->size;
        const bool             root = a.root == team->rank;
        ucc_coll_type_t        ct   = a.coll_type;
        /* per-role element geometry (count conventions: rooted colls
         * carry the total at the root, one block elsewhere) */
        size_t blk_el = 0, pack_el = 0, unp_el = 0;
        bool   i_pack = false, i_unpack = false;
        switch (ct) {
        case UCC_COLL_TYPE_ALLGATHER:
            blk_el  = (size_t)a.dst.info.count / nt;
            pack_el = blk_el;
            unp_el  = a.dst.info.count;
            i_pack = i_unpack = true;
            break;
        case UCC_COLL_TYPE_ALLTOALL:
            blk_el  = (size_t)a.dst.info.count / nt;
            pack_el = a.dst.info.count;
            unp_el  = a.dst.info.count;
            i_pack = i_unpack = true;
            break;
        case UCC_COLL_TYPE_GATHER:
            blk_el  = root ? (size_t)a.dst.info.count / nt
                           : (size_t)a.src.info.count;
            pack_el = blk_el;
            unp_el  = root ? a.dst.info.count : 0;
            i_pack  = true;
            i_unpack = root;
            break;
        case UCC_COLL_TYPE_SCATTER:
            blk_el  = root ? (size_t)a.src.info.count / nt
                           : (size_t)a.dst.info.count;
            pack_el = root ? a.src.info.count : 0;
            unp_el  = blk_el;
            i_pack  = root;
            i_unpack = true;
            break;
        default:
            break;
        }
        if (blk_el == 0) {
            delete req;
            return UCC_ERR_NOT_SUPPORTED;
        }
        /* packed bytes of one block, probed locally (content-
         * independent size assumption) */
        const void *probe_buf =
            i_pack ? a.src.info.buffer : a.dst.info.buffer;
        void  *pobj  = ncg->ops.start_pack(ncg->cookie, probe_buf,
                                           blk_el);
        size_t blk_p = ncg->ops.packed_size(pobj);
        if (ncg->ops.finish) {
            ncg->ops.finish(pobj);
        }
        size_t pack_p = pack_el / blk_el * blk_p;
        size_t unp_p  = unp_el / blk_el * blk_p;
        if (i_pack) {
            req->gdt_send.resize(pack_p);
            const void *ub = a.src.info.buffer;
            req->pre_post  = [pack_all, ub, pack_el,
                             &buf = req->gdt_send, pack_p]() {
                return pack_all(ub, pack_el, buf.data(), pack_p);
            };
        }
        if (i_unpack) {
            req->gdt_recv.resize(unp_p);
            void *ub           = a.dst.info.buffer;
            req->post_complete = [unpack_all, ub, unp_el,
                                  &buf = req->gdt_recv, unp_p]() {
                return unpack_all(ub, unp_el, buf.data(), unp_p);
            };
        }
        /* rewrite to plain bytes, preserving each role's count
         * convention (counts scale by blk_p/blk_el) */
        req->args.src.info.count =
            (size_t)a.src.info.count / blk_el * blk_p;
        req->args.dst.info.count =
            (size_t)a.dst.info.count / blk_el * blk_p;
        if (i_pack) {
            req->args.src.info.buffer = req->gdt_send.data();
        }
        if (i_unpack) {
            req->args.dst.info.buffer = req->gdt_recv.data();
        }
        req->args.src.info.datatype = UCC_DT_UINT8;
        req->args.dst.info.datatype = UCC_DT_UINT8;
        msgsize = coll_args_msgsize(req->args, team->rank, team->size);
    }
    if (msgsize == 0 && !v_coll && !noncontig_gen &&
        req->args.coll_type != UCC_COLL_TYPE_BARRIER &&
        req->args.coll_type != UCC_COLL_TYPE_FANIN &&
        req->args.coll_type != UCC_COLL_TYPE_FANOUT) {
        req->task = new StubTask(team->ctx);
    } else {
        ucc_status_t st =
            team->score_map.init_coll(req->args, team, msgsize, &req->task);
        if (st != UCC_OK) {
            delete req;
            return st;
        }
    }
    req->task->req_ = req;
    if ((req->args.mask & UCC_COLL_ARGS_FIELD_FLAGS) &&
        (req->args.flags & UCC_COLL_ARGS_FLAG_TIMEOUT)) {
        req->task->timeout = req->args.timeout;
    }
    req_status_store(&req->super, UCC_OPERATION_INITIALIZED);
    *request          = &req->super;
    UCC_PROFILE_REQUEST_NEW(coll_type_name(req->args.coll_type),
                            (uintptr_t)req);
    return UCC_OK;
}

ucc_status_t ucc_collective_post(ucc_coll_req_h request)
{
    auto *req = reinterpret_cast<CollRequest *>(request);
    if (req->posted && req->task->status == UCC_INPROGRESS) {
        return UCC_ERR_INVALID_PARAM; /* re-post of in-flight coll */
    }
    req->posted       = true;
    req->seq          = req->team->coll_seq++;
    req_status_store(&req->super, UCC_INPROGRESS);
    UCC_PROFILE_REQUEST_EVENT("post", (uintptr_t)req);
    if (req->pre_post) {
        ucc_status_t pst = req->pre_post();
        if (pst != UCC_OK) {
            req_status_store(&req->super, pst);
            return pst;
        }
    }
    task_start(req->task);
    ucc_status_t ts = req->task->status.load();
    if (ts != UCC_INPROGRESS) {
        req_status_store(&req->super, ts);
    }
    return ts < 0 ? ts : UCC_OK;
}

ucc_status_t ucc_collective_init_and_post(ucc_coll_args_t *coll_args,
                                          ucc_coll_req_h *request,
                                          ucc_team_h team)
{
    ucc_status_t st = ucc_collective_init(coll_args, request, team);
    if (st != UCC_OK) {
        return st;
    }
    st = ucc_collective_post(*request);
    if (st < 0) {
        ucc_collective_finalize(*request);
        *request = nullptr;
    }
    return st;
}

ucc_status_t ucc_collective_test(ucc_coll_req_h request)
{
    auto *req = reinterpret_cast<CollRequest *>(request);
    return req_status_load(&req->super);
}

ucc_status_t ucc_collective_finalize(ucc_coll_req_h request)
{
    auto *req = reinterpret_cast<CollRequest *>(request);
    UCC_PROFILE_REQUEST_FREE("finalize", (uintptr_t)req);
    if (req->asymm_scratch) {
        mc::scratch_free(req->asymm_scratch, req->asymm_bytes,
                         req->asymm_mt);
    }
    delete req->task;
    delete req;
    return UCC_OK;
}

/* ------------------------------------------------------------------- EE */
/* Execution-engine API (reference core/ucc_ee.c + triggered-post path in
 * core/ucc_coll.c:423-659, re-derived): an EE wraps a user HIP stream;
 * ucc_collective_triggered_post launches the collective's device work
 * stream-ordered on it (tasks that support it: cdna4 fused allreduce,
 * which is also hipGraph-capture legal). */
ucc_status_t ucc_ee_create(ucc_team_h team, const ucc_ee_params_t *params,
                           ucc_ee_h *ee)
{
    if (!team || !params || !ee) {
        return UCC_ERR_INVALID_PARAM;
    }
    if (params->ee_type != UCC_EE_CUDA_STREAM &&
        params->ee_type != UCC_EE_ROCM_STREAM) {
        return UCC_ERR_NOT_SUPPORTED;
    }
    auto *e   = new Ee();
    e->team   = reinterpret_cast<Team *>(team);
    e->params = *params;
    e->stream = params->ee_context;
    *ee       = reinterpret_cast<ucc_ee_h>(e);
    return UCC_OK;
}
ucc_status_t ucc_ee_destroy(ucc_ee_h ee)
{
    delete reinterpret_cast<Ee *>(ee);
    return UCC_OK;
}
/* drain fired stream events into the user-visible queue */
static void ee_poll_pending(Ee *e)
{
    while (!e->pending.empty()) {
        auto &p = e->pending.front();
        int   q = mc::event_query(p.hip_ev);
        if (q == 0) {
            break; /* stream work still running (FIFO per stream) */
        }
        ucc_ev_t done = p.ev;
        done.ev_type  = UCC_EVENT_COLLECTIVE_COMPLETE;
        if (q < 0 && done.req) {
            req_status_store(done.req, UCC_ERR_NO_RESOURCE);
        }
        e->events.push_back(done);
        mc::event_free(p.hip_ev);
        e->pending.pop_front();
    }
}

ucc_status_t ucc_ee_get_event(ucc_ee_h ee, ucc_ev_t **ev)
{
    auto *e = reinterpret_cast<Ee *>(ee);
    ee_poll_pending(e);
    if (e->events.empty()) {
        return UCC_ERR_NOT_FOUND;
    }
    *ev = &e->events.front();
    return UCC_OK;
}
ucc_status_t ucc_ee_ack_event(ucc_ee_h ee, ucc_ev_t *ev)
{
    auto *e = reinterpret_cast<Ee *>(ee);
    if (!e->events.empty() && ev == &e->events.front()) {
        e->events.pop_front();
        return UCC_OK;
    }
    return UCC_ERR_INVALID_PARAM;
}
ucc_status_t ucc_ee_set_event(ucc_ee_h ee, ucc_ev_t *ev)
{
    auto *e = reinterpret_cast<Ee *>(ee);
    e->events.push_back(*ev);
    return UCC_OK;
}
ucc_status_t ucc_ee_wait(ucc_ee_h ee, ucc_ev_t *ev)
{
    /* block until the NEXT event is available (reference ucc_ee.c
     * role): drain pending stream completions while spinning. The
     * event is returned in *ev and stays at the queue head until
     * acked. */
    auto *e = reinterpret_cast<Ee *>(ee);
    if (!e || !ev) {
        return UCC_ERR_INVALID_PARAM;
    }
    for (;;) {
        ee_poll_pending(e);
        if (!e->events.empty()) {
            *ev = e->events.front();
            return UCC_OK;
        }
        if (e->pending.empty()) {
            return UCC_ERR_NOT_FOUND; /* nothing will ever arrive */
        }
        sched_yield();
    }
}
ucc_status_t ucc_collective_triggered_post(ucc_ee_h ee, ucc_ev_t *ev)
{
    auto *e = reinterpret_cast<Ee *>(ee);
    if (!e || !ev || !ev->req) {
        return UCC_ERR_INVALID_PARAM;
    }
    auto *req = reinterpret_cast<CollRequest *>(ev->req);
    if (req->posted && !req->persistent) {
        return UCC_ERR_INVALID_PARAM;
    }
    if (req->pre_post) { /* asymm staging copy: synchronous (the user's
                          * src must be final before a triggered post) */
        ucc_status_t pst = req->pre_post();
        if (pst != UCC_OK) {
            return pst;
        }
    }
    ucc_status_t st = req->task->triggered_post(e->stream);
    if (st != UCC_OK && st != UCC_INPROGRESS) {
        return st;
    }
    req->posted        = true;
    req_status_store(&req->super, req->task->status.load());
    ucc_ev_t done      = *ev;
    done.ev_type       = UCC_EVENT_COLLECTIVE_POST;
    done.req           = &req->super;
    e->events.push_back(done);
    /* completion event: fires when the stream-ordered work finishes */
    void *hev = nullptr;
    if (mc::stream_event_record(e->stream, &hev) == UCC_OK) {
        e->pending.push_back(Ee::Pending{hev, done});
    }
    return UCC_OK;
}

Ee::~Ee()
{
    for (auto &p : pending) {
        mc::event_free(p.hip_ev);
    }
}

/* ------------------------------------------------------------- mem_map */
/* Export: produce a transferable handle blob for the given segments
 * (device segments carry HIP-IPC handles; host segments are recorded by
 * address for same-node shm use). Import: open the blob received from a
 * peer, mapping device segments into this process.
 * Reference parity: ucc.h mem_map section + tl/cuda IPC mapping cache
 * (tl_cuda_cache.c role, collapsed: one open per imported segment). */
namespace {
struct MemMapSeg {
    uint64_t addr;   /* exporter VA            */
    uint64_t len;
    uint64_t base_off; /* addr - allocation base */
    int32_t  mt;
    int32_t  has_ipc;
    uint8_t  handle[ucc::mc::kIpcHandleBytes];
    uint64_t mapped; /* importer-local VA (base) — set on import */
};
struct MemMapBlob {
    uint32_t  magic; /* 0x4d4d4150 "MMAP" */
    uint32_t  n;
    int32_t   imported;
    int32_t   pad;
    MemMapSeg segs[];
};
constexpr uint32_t kMemMapMagic = 0x4d4d4150;
} // namespace

ucc_status_t ucc_mem_map(ucc_context_h context, ucc_mem_map_flags_t flags,
                         ucc_mem_map_params_t *params, size_t *memh_size,
                         ucc_mem_map_mem_h *memh)
{
    (void)context;
    if (!memh || !memh_size) {
        return UCC_ERR_INVALID_PARAM;
    }
    if (flags & UCC_MEM_MAP_MODE_EXPORT) {
        if (!params) {
            return UCC_ERR_INVALID_PARAM;
        }
        size_t n  = params->n_segments;
        size_t sz = sizeof(MemMapBlob) + n * sizeof(MemMapSeg);
        auto  *b  = (MemMapBlob *)calloc(1, sz);
        b->magic  = kMemMapMagic;
        b->n      = (uint32_t)n;
        for (size_t i = 0; i < n; i++) {
            MemMapSeg &s = b->segs[i];
            s.addr = (uint64_t)(uintptr_t)params->segments[i].address;
            s.len  = params->segments[i].len;
            ucc_memory_type_t mt = UCC_MEMORY_TYPE_HOST;
            ucc::mc::mem_query((const void *)(uintptr_t)s.addr, &mt);
            s.mt = (int32_t)mt;
            if (ucc::mc::is_device_mt(mt)) {
                size_t off = 0;
                if (ucc::mc::ipc_export((const void *)(uintptr_t)s.addr,
                                   s.handle, &off) == UCC_OK) {
                    s.has_ipc  = 1;
                    s.base_off = off;
                }
            }
        }
        *memh      = b;
        *memh_size = sz;
        return UCC_OK;
    }
    if (flags & UCC_MEM_MAP_MODE_IMPORT) {
        auto *b = (MemMapBlob *)*memh;
        if (!b || b->magic != kMemMapMagic) {
            return UCC_ERR_INVALID_PARAM;
        }
        for (uint32_t i = 0; i < b->n; i++) {
            MemMapSeg &s = b->segs[i];
            if (s.has_ipc) {
                void *m = nullptr;
                ucc_status_t st = ucc::mc::ipc_import(s.handle, &m);
                if (st != UCC_OK) {
                    return st;
                }
                s.mapped = (uint64_t)(uintptr_t)m;
            }
        }
        b->imported = 1;
        *memh_size  = sizeof(MemMapBlob) + b->n * sizeof(MemMapSeg);
        return UCC_OK;
    }
    return UCC_ERR_INVALID_PARAM;
}

/* TL helper (reference alltoall_onesided.c src/dst memh role): look up
 * a registered EXPORTED segment covering [addr, addr+len) and hand back
 * its IPC handle + the offset of addr from the exporting allocation's
 * base. Lets collectives on registered buffers skip the per-post
 * hipIpcGetMemHandle. */
int ucc_memh_lookup(ucc_mem_map_mem_h memh, const void *addr, size_t len,
                    void *handle_out, uint64_t *alloc_off)
{
    auto *b = (MemMapBlob *)memh;
    if (!b || b->magic != kMemMapMagic || b->imported) {
        return 0;
    }
    uint64_t a = (uint64_t)(uintptr_t)addr;
    for (uint32_t i = 0; i < b->n; i++) {
        MemMapSeg &s = b->segs[i];
        if (s.has_ipc && a >= s.addr && a + len <= s.addr + s.len) {
            memcpy(handle_out, s.handle, ucc::mc::kIpcHandleBytes);
            *alloc_off = s.base_off + (a - s.addr);
            return 1;
        }
    }
    return 0;
}

ucc_status_t ucc_mem_unmap(ucc_mem_map_mem_h *memh)
{
    if (!memh || !*memh) {
        return UCC_ERR_INVALID_PARAM;
    }
    auto *b = (MemMapBlob *)*memh;
    if (b->magic != kMemMapMagic) {
        return UCC_ERR_INVALID_PARAM;
    }
    if (b->imported) {
        for (uint32_t i = 0; i < b->n; i++) {
            if (b->segs[i].has_ipc && b->segs[i].mapped) {
                ucc::mc::ipc_close((void *)(uintptr_t)b->segs[i].mapped);
            }
        }
    } else {
        free(b);
    }
    *memh = nullptr;
    return UCC_OK;
}

} /* extern "C" */
