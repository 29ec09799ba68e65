/* Hierarchical composition — the cl/hier role (reference
 * src/components/cl/hier: allreduce RAB = node reduce -> leader
 * allreduce -> node bcast, cl_hier.h:38-57, allreduce_rab.c),
 * re-derived for this framework's collapsed CL/TL stack:
 *
 *  - After a multi-node team reaches TL_CREATE, core drives the HIER
 *    state: two INTERNAL sub-teams are created — node_team (my node's
 *    ranks, served by shm/cdna4) and leaders_team (lowest rank per
 *    node, served by tcp) — through the same public team machinery.
 *  - Sub-team bootstrap rides the PARENT team's OOB: every sub-OOB
 *    round is one parent allgather of a fixed kPad-byte slot, so
 *    non-members observe (contribute zeros) and the per-rank round
 *    sequence stays positionally aligned across all ranks. Team create
 *    is forced to exactly two OOB rounds (addr + TL exchange) for this
 *    alignment (core pads a zero-stride TL exchange to 8 bytes).
 *  - The hier allreduce is a 3-phase task over the sub-teams,
 *    registered in the parent score map at score 60 for host memory
 *    (above tcp's flat algorithms, below same-node shm which wins when
 *    the team does not span nodes — where hier is never installed).
 *
 * Testability without a cluster: UCC_FAKE_NODE_SPLIT=k makes contexts
 * hash to k pseudo-nodes (core/ucc_context_create), so the full
 * node/leader composition runs inside one machine (tests/test_hier.py).
 */
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <map>

#include "../core/core.h"
#include "../ec/ec_cpu.h"
#include "../ec/ec_hip.h"
#include "../mc/mc.h"
#include "../topo/topo.h"

namespace ucc {

static constexpr size_t kPad = 512; /* fixed parent-OOB slot per round */

/* ------------------------------------------------------------- SubOob */
/* Declared in core.h as ucc::SubOob (Team member pointers). */
struct SubOob {
    Team                 *parent = nullptr;
    std::vector<uint32_t> members; /* parent ranks, sorted              */
    int                   my_idx = -1;

    struct Req {
        SubOob              *so;
        void                *parent_req = nullptr;
        std::vector<uint8_t> full;   /* parent_n * kPad                 */
        std::vector<uint8_t> send;   /* kPad                            */
        void                *recv = nullptr;
        size_t               size = 0;
        bool                 done = false;
    };

    ucc_status_t start(const void *src, void *recv, size_t size,
                       void **request)
    {
        if (size + 8 > kPad) {
            return UCC_ERR_NO_RESOURCE;
        }
        auto *r = new Req;
        r->so   = this;
        r->send.assign(kPad, 0);
        if (src && size) {
            *(uint64_t *)r->send.data() = size;
            memcpy(r->send.data() + 8, src, size);
        }
        r->full.resize((size_t)parent->size * kPad);
        r->recv = recv;
        r->size = size;
        ucc_status_t st = parent->oob.allgather(
            r->send.data(), r->full.data(), kPad, parent->oob.coll_info,
            &r->parent_req);
        if (st != UCC_OK) {
            delete r;
            return st;
        }
        *request = r;
        return UCC_OK;
    }

    static ucc_status_t cb_allgather(void *src, void *recv, size_t size,
                                     void *info, void **request)
    {
        return ((SubOob *)info)->start(src, recv, size, request);
    }

    static ucc_status_t cb_test(void *request)
    {
        auto *r = (Req *)request;
        if (!r->done) {
            ucc_status_t st =
                r->so->parent->oob.req_test(r->parent_req);
            if (st == UCC_INPROGRESS) {
                return st;
            }
            r->so->parent->oob.req_free(r->parent_req);
            r->parent_req = nullptr;
            if (st != UCC_OK) {
                return st;
            }
            if (r->recv && r->size) {
                for (size_t i = 0; i < r->so->members.size(); i++) {
                    memcpy((uint8_t *)r->recv + i * r->size,
                           r->full.data() +
                               (size_t)r->so->members[i] * kPad + 8,
                           r->size);
                }
            }
            r->done = true;
        }
        return UCC_OK;
    }

    static ucc_status_t cb_free(void *request)
    {
        delete (Req *)request;
        return UCC_OK;
    }

    /* ------------- observer (non-member) round machinery */
    Req *obs_req = nullptr;
    int  obs_rounds_done = 0;

    ucc_status_t observe_tick(int target_rounds)
    {
        if (obs_rounds_done >= target_rounds) {
            return UCC_OK;
        }
        if (!obs_req) {
            void *rq = nullptr;
            ucc_status_t st = start(nullptr, nullptr, 0, &rq);
            if (st != UCC_OK) {
                return st;
            }
            obs_req = (Req *)rq;
        }
        ucc_status_t st = cb_test(obs_req);
        if (st == UCC_INPROGRESS) {
            return UCC_INPROGRESS;
        }
        cb_free(obs_req);
        obs_req = nullptr;
        if (st != UCC_OK) {
            return st;
        }
        obs_rounds_done++;
        return obs_rounds_done >= target_rounds ? UCC_OK : UCC_INPROGRESS;
    }
};

namespace subooob_api {

SubOob *make(Team *parent, const std::vector<uint32_t> &members,
             int my_idx)
{
    auto *so    = new SubOob;
    so->parent  = parent;
    so->members = members;
    so->my_idx  = my_idx;
    return so;
}

ucc_status_t observe_tick(SubOob *so, int target_rounds)
{
    return so->observe_tick(target_rounds);
}

void fill_oob(SubOob *so, ucc_oob_coll_t *oob)
{
    oob->allgather = SubOob::cb_allgather;
    oob->req_test  = SubOob::cb_test;
    oob->req_free  = SubOob::cb_free;
    oob->coll_info = so;
    oob->n_oob_eps = (uint32_t)so->members.size();
    oob->oob_ep    = (uint32_t)so->my_idx;
}

void free_(SubOob *so) { delete so; }

} // namespace subooob_api

namespace hier {

static thread_local bool g_creating_subteam = false;

bool creating_subteam() { return g_creating_subteam; }

/* ------------------------------------------------------------- wanted  */
bool wanted(Team *team)
{
    if (g_creating_subteam || team->is_subteam || team->size < 2 ||
        !team->has_oob) {
        return false;
    }
    auto &cfg = Config::instance();
    cfg.declare("CL_HIER", "FRAG_SIZE", "4m",
                "fragment bytes for pipelined hier allreduce "
                "(0 = monolithic; pipelined when msg >= 2*FRAG_SIZE)");
    cfg.declare("CL_HIER", "PIPELINE_DEPTH", "2",
                "fragments in flight for pipelined hier collectives");
    cfg.declare("CL_HIER", "PIPELINE_TRACE", "",
                "append 'P/C frag stage' pipeline events to this file "
                "(testing hook)");
    if (!cfg.get_bool("CL_HIER", "ENABLE", true)) {
        return false;
    }
    return !team->all_same_node();
}

/* ------------------------------------------------------------- setup   */
static ucc_status_t create_subteam(Team *parent, SubOob *so,
                                   std::unique_ptr<Team> &out)
{
    ucc_team_params_t tp{};
    tp.mask          = UCC_TEAM_PARAM_FIELD_OOB;
    tp.oob.allgather = SubOob::cb_allgather;
    tp.oob.req_test  = SubOob::cb_test;
    tp.oob.req_free  = SubOob::cb_free;
    tp.oob.coll_info = so;
    tp.oob.n_oob_eps = (uint32_t)so->members.size();
    tp.oob.oob_ep    = (uint32_t)so->my_idx;
    ucc_context_h ch = reinterpret_cast<ucc_context_h>(parent->ctx);
    ucc_team_h    th = nullptr;
    g_creating_subteam = true;
    ucc_status_t st = ucc_team_create_post(&ch, 1, &tp, &th);
    g_creating_subteam = false;
    if (st != UCC_OK) {
        return st;
    }
    out.reset(reinterpret_cast<Team *>(th));
    return UCC_OK;
}

ucc_status_t setup(Team *team)
{
    auto node = topo::build_sbgp(team, topo::SbgpType::NODE);
    auto ldr  = topo::build_sbgp(team, topo::SbgpType::NODE_LEADERS);
    team->node_ranks   = node.ranks;
    team->leader_ranks = ldr.ranks;
    if (node.ranks.size() < 2 || ldr.ranks.size() < 2 ||
        ldr.ranks.size() == team->size) {
        /* every rank on its own node (1 proc/node) or trivial split:
         * hier adds nothing over the flat inter-node path */
        return UCC_ERR_NOT_SUPPORTED;
    }
    /* parent-OOB round alignment requires every node team to run the
     * same 2 bootstrap rounds: sizes must all be >= 2 */
    {
        std::map<uint64_t, uint32_t> per_node;
        for (auto &p : team->procs) {
            per_node[p.host_hash]++;
        }
        for (auto &kv : per_node) {
            if (kv.second < 2) {
                return UCC_ERR_NOT_SUPPORTED;
            }
        }
    }
    team->node_oob          = new SubOob;
    team->node_oob->parent  = team;
    team->node_oob->members = node.ranks;
    team->node_oob->my_idx  = node.my_idx;
    team->leaders_oob          = new SubOob;
    team->leaders_oob->parent  = team;
    team->leaders_oob->members = ldr.ranks;
    team->leaders_oob->my_idx  = ldr.my_idx; /* -1 if not a leader */
    team->hier_step            = 0;
    /* rails (split_rail): rank p of every node forms rail p — needs
     * UNIFORM node sizes. Node order = leader order. */
    {
        std::map<uint64_t, std::vector<uint32_t>> per_node;
        for (uint32_t r = 0; r < team->size; r++) {
            per_node[team->procs[r].host_hash].push_back(r);
        }
        size_t nsz = per_node.begin()->second.size();
        bool uniform = true;
        for (auto &kv : per_node) {
            if (kv.second.size() != nsz) {
                uniform = false;
            }
        }
        if (uniform && node.my_idx >= 0 && (size_t)node.my_idx < nsz) {
            std::vector<uint32_t> rail;
            for (uint32_t lr : ldr.ranks) { /* node order by leader */
                uint64_t hh = team->procs[lr].host_hash;
                rail.push_back(per_node[hh][node.my_idx]);
            }
            int my_idx = -1;
            for (size_t i = 0; i < rail.size(); i++) {
                if (rail[i] == team->rank) {
                    my_idx = (int)i;
                }
            }
            team->rail_oob          = new SubOob;
            team->rail_oob->parent  = team;
            team->rail_oob->members = rail;
            team->rail_oob->my_idx  = my_idx;
            team->rails_ok          = my_idx >= 0;
        }
    }
    if (node.ranks.size() == 1) {
        /* singleton node: node team is size-1 (self TL) — skip its OOB
         * rounds entirely is NOT possible (alignment); create it anyway
         * via the sub-OOB (2 rounds, self TL serves colls). */
    }
    ucc_status_t st = create_subteam(team, team->node_oob,
                                     team->node_team);
    if (st != UCC_OK) {
        return st;
    }
    return UCC_OK;
}

/* ------------------------------------------------------------- test    */
ucc_status_t test(Team *team)
{
    const bool leader = team->leaders_oob->my_idx >= 0;
    static const bool dbg = getenv("UCC_HIER_DEBUG") != nullptr;
    if (dbg) {
        fprintf(stderr, "[hier] rank %u step %d node_state %d ldr_state %d\n",
                team->rank, team->hier_step,
                team->node_team ? (int)team->node_team->state : -1,
                team->leaders_team ? (int)team->leaders_team->state : -1);
    }
    switch (team->hier_step) {
    case 0: { /* node team consumes its 2 parent rounds */
        ucc_status_t st = ucc_team_create_test(
            reinterpret_cast<ucc_team_h>(team->node_team.get()));
        if (st < 0) {
            return st;
        }
        if (team->node_team->state < Team::TL_CREATE) {
            return UCC_INPROGRESS;
        }
        /* both node OOB rounds consumed: start leaders stage */
        if (leader) {
            ucc_status_t cs = create_subteam(team, team->leaders_oob,
                                             team->leaders_team);
            if (cs != UCC_OK) {
                return cs;
            }
        }
        team->hier_step = 1;
        [[fallthrough]];
    }
    case 1: { /* leaders team rounds (leaders) / 2 observed (others) */
        if (leader) {
            ucc_status_t st = ucc_team_create_test(
                reinterpret_cast<ucc_team_h>(team->leaders_team.get()));
            if (st < 0) {
                return st;
            }
            if (team->leaders_team->state < Team::TL_CREATE) {
                return UCC_INPROGRESS;
            }
        } else {
            ucc_status_t st = team->leaders_oob->observe_tick(2);
            if (st == UCC_INPROGRESS) {
                return UCC_INPROGRESS;
            }
            if (st != UCC_OK) {
                return st;
            }
        }
        if (team->rails_ok) { /* every rank drives its own rail: the
                               * per-rank parent round sequence stays
                               * aligned (uniform node sizes) */
            ucc_status_t cs =
                create_subteam(team, team->rail_oob, team->rail_team);
            if (cs != UCC_OK) {
                team->rails_ok = false;
            }
        }
        team->hier_step = 4;
        [[fallthrough]];
    }
    case 4: { /* rail team rounds (all ranks in lockstep) */
        if (team->rails_ok) {
            ucc_status_t st = ucc_team_create_test(
                reinterpret_cast<ucc_team_h>(team->rail_team.get()));
            if (st < 0) {
                return st;
            }
            if (team->rail_team->state < Team::TL_CREATE) {
                return UCC_INPROGRESS;
            }
        }
        team->hier_step = 2;
        [[fallthrough]];
    }
    case 2: { /* finish both create_test phases (no parent OOB) */
        ucc_status_t st = ucc_team_create_test(
            reinterpret_cast<ucc_team_h>(team->node_team.get()));
        if (st < 0) {
            return st;
        }
        if (st == UCC_INPROGRESS) {
            return UCC_INPROGRESS;
        }
        if (leader) {
            st = ucc_team_create_test(
                reinterpret_cast<ucc_team_h>(team->leaders_team.get()));
            if (st < 0) {
                return st;
            }
            if (st == UCC_INPROGRESS) {
                return UCC_INPROGRESS;
            }
        }
        if (team->rails_ok) {
            st = ucc_team_create_test(
                reinterpret_cast<ucc_team_h>(team->rail_team.get()));
            if (st < 0) {
                return st;
            }
            if (st == UCC_INPROGRESS) {
                return UCC_INPROGRESS;
            }
        }
        return UCC_OK;
    }
    }
    return UCC_ERR_INVALID_PARAM;
}

void destroy(Team *team)
{
    delete team->node_oob;
    delete team->leaders_oob;
    delete team->rail_oob;
    team->node_oob    = nullptr;
    team->leaders_oob = nullptr;
    team->rail_oob    = nullptr;
}

/* ------------------------------------------ hier allreduce (RAB) task  */
class HierAllreduceTask final : public Task {
  public:
    HierAllreduceTask(Context *ctx, Team *team,
                      const ucc_coll_args_t &args)
        : Task(ctx), team_(team), a_(args)
    {
    }
    ~HierAllreduceTask() override
    {
        if (req_h_) {
            ucc_collective_finalize(req_h_);
        }
    }

    ucc_status_t post() override
    {
        phase_   = 0;
        leader_  = team_->leaders_oob->my_idx >= 0;
        inplace_ = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        count_   = a_.dst.info.count;
        dt_      = a_.dst.info.datatype;
        status   = UCC_INPROGRESS;
        return step();
    }

    ucc_status_t progress() override { return step(); }

  private:
    ucc_status_t launch(Team *t, ucc_coll_type_t ct, uint64_t root)
    {
        ucc_coll_args_t sa{};
        sa.mask      = UCC_COLL_ARGS_FIELD_FLAGS;
        sa.flags     = a_.flags & (UCC_COLL_ARGS_FLAG_COUNT_64BIT |
                                   UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT);
        sa.coll_type = ct;
        sa.root      = root;
        sa.op = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
        if (ct == UCC_COLL_TYPE_REDUCE) {
            sa.src.info        = a_.src.info;
            if (inplace_) {
                sa.src.info.buffer   = a_.dst.info.buffer;
                sa.src.info.count    = count_;
                sa.src.info.datatype = dt_;
                sa.src.info.mem_type = a_.dst.info.mem_type;
            }
            sa.dst.info = a_.dst.info;
        } else if (ct == UCC_COLL_TYPE_ALLREDUCE) {
            sa.flags |= UCC_COLL_ARGS_FLAG_IN_PLACE;
            sa.src.info = a_.dst.info;
            sa.dst.info = a_.dst.info;
        } else { /* bcast */
            sa.src.info = a_.dst.info;
        }
        ucc_status_t st = ucc_collective_init(
            &sa, &req_h_, reinterpret_cast<ucc_team_h>(t));
        if (st != UCC_OK) {
            return st;
        }
        return ucc_collective_post(req_h_);
    }

    ucc_status_t step()
    {
        while (true) {
            if (req_h_) {
                ucc_status_t st = ucc_collective_test(req_h_);
                if (st == UCC_INPROGRESS) {
                    return UCC_INPROGRESS;
                }
                ucc_collective_finalize(req_h_);
                req_h_ = nullptr;
                if (st != UCC_OK) {
                    return st;
                }
                phase_++;
            }
            switch (phase_) {
            case 0: { /* node reduce to node leader (node rank 0) */
                if (team_->node_team->size == 1) {
                    /* singleton: local copy src->dst unless inplace */
                    if (!inplace_) {
                        memcpy(a_.dst.info.buffer, a_.src.info.buffer,
                               count_ * ucc_dt_size(dt_));
                    }
                    phase_ = 1;
                    continue;
                }
                ucc_status_t st =
                    launch(team_->node_team.get(),
                           UCC_COLL_TYPE_REDUCE, 0);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 1: { /* leaders allreduce (leaders only) */
                if (!leader_) {
                    phase_ = 2;
                    continue;
                }
                ucc_status_t st = launch(team_->leaders_team.get(),
                                         UCC_COLL_TYPE_ALLREDUCE, 0);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 2: { /* node bcast from the leader */
                if (team_->node_team->size == 1) {
                    phase_ = 3;
                    continue;
                }
                ucc_status_t st = launch(team_->node_team.get(),
                                         UCC_COLL_TYPE_BCAST, 0);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 3: { /* AVG: final local scale by 1/N */
                if (a_.op == UCC_OP_AVG) {
                    const void *srcs[1] = {a_.dst.info.buffer};
                    ec_cpu::reduce(a_.dst.info.buffer, srcs, 1, count_,
                                   dt_, UCC_OP_SUM,
                                   1.0 / (double)team_->size);
                }
                return UCC_OK;
            }
            default:
                return UCC_ERR_INVALID_PARAM;
            }
            /* a sub-collective was posted; poll it next iteration */
            if (req_h_) {
                continue;
            }
        }
    }

    Team           *team_;
    ucc_coll_args_t a_;
    ucc_coll_req_h  req_h_ = nullptr;
    int             phase_ = 0;
    bool            leader_ = false, inplace_ = false;
    uint64_t        count_ = 0;
    ucc_datatype_t  dt_ = UCC_DT_FLOAT32;
};

/* ---- RAB over the generic fragment pipeline: large messages split
 * into frags so the node-reduce of frag f+1 overlaps the leader
 * allreduce of frag f and the node-bcast of f-1 (reference
 * cl_hier.h:47-57 pipeline configs + ucc_schedule_pipelined.h). */
class HierAllreducePipeTask final : public PipelineTask {
  public:
    HierAllreducePipeTask(Context *ctx, Team *team,
                          const ucc_coll_args_t &args, size_t frag_elems,
                          size_t depth)
        : PipelineTask(ctx), team_(team), a_(args)
    {
        count_   = a_.dst.info.count;
        dt_      = a_.dst.info.datatype;
        dtsz_    = ucc_dt_size(dt_);
        inplace_ = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        dev_     = mc::is_device_mt(a_.dst.info.mem_type);
        fc_      = frag_elems ? frag_elems : count_;
        if (fc_ == 0 || fc_ > count_) {
            fc_ = count_ ? count_ : 1;
        }
        n_frags  = (count_ + fc_ - 1) / fc_;
        n_stages = a_.op == UCC_OP_AVG ? 4 : 3;
        pdepth   = depth ? depth : 2;
        pair_early = 0; /* node reduce and node bcast share node_team */
        pair_late  = 2;
        stage_post = [this](size_t f, size_t s, ucc_coll_req_h *r) {
            return do_stage(f, s, r);
        };
        /* device memory: the inter-node leader phase runs on HOST
         * buffers (tcp leaders team) — one staging buffer per in-flight
         * fragment, D2H before the leader allreduce and H2D after
         * (stage_done hook). Node phases stay on device (cdna4). */
        if (dev_) {
            stage_done = [this](size_t f, size_t s) -> ucc_status_t {
                if (s == 1 && leader_) {
                    size_t b   = f * fc_;
                    size_t cnt = count_ - b < fc_ ? count_ - b : fc_;
                    return mc::copy((uint8_t *)a_.dst.info.buffer +
                                        b * dtsz_,
                                    a_.dst.info.mem_type,
                                    hstg_[f % pdepth].data(),
                                    UCC_MEMORY_TYPE_HOST, cnt * dtsz_);
                }
                return UCC_OK;
            };
        }
    }

    ucc_status_t post() override
    {
        leader_ = team_->leaders_oob->my_idx >= 0;
        if (dev_ && leader_ && hstg_.empty()) {
            hstg_.resize(pdepth);
            for (auto &v : hstg_) {
                v.resize(fc_ * dtsz_);
            }
        }
        return PipelineTask::post();
    }

  private:
    ucc_status_t do_stage(size_t f, size_t s, ucc_coll_req_h *req)
    {
        *req         = nullptr;
        size_t   b   = f * fc_;
        size_t   cnt = count_ - b < fc_ ? count_ - b : fc_;
        uint8_t *dstp =
            (uint8_t *)a_.dst.info.buffer + b * dtsz_;
        const uint8_t *srcp =
            inplace_ ? dstp
                     : (const uint8_t *)a_.src.info.buffer + b * dtsz_;
        ucc_coll_args_t sa{};
        sa.mask  = UCC_COLL_ARGS_FIELD_FLAGS;
        sa.flags = a_.flags & (UCC_COLL_ARGS_FLAG_COUNT_64BIT |
                               UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT);
        sa.op    = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
        sa.root  = 0;
        Team *st_team = nullptr;
        switch (s) {
        case 0: /* node reduce of the slice to the node leader */
            if (team_->node_team->size == 1) {
                if (!inplace_) {
                    return mc::copy(dstp, a_.dst.info.mem_type, srcp,
                                    a_.dst.info.mem_type, cnt * dtsz_);
                }
                return UCC_OK;
            }
            sa.coll_type         = UCC_COLL_TYPE_REDUCE;
            sa.src.info          = a_.src.info;
            sa.src.info.buffer   = (void *)srcp;
            sa.src.info.count    = cnt;
            sa.src.info.datatype = dt_;
            sa.src.info.mem_type = a_.dst.info.mem_type;
            sa.dst.info          = a_.dst.info;
            sa.dst.info.buffer   = dstp;
            sa.dst.info.count    = cnt;
            st_team              = team_->node_team.get();
            break;
        case 1: /* leaders allreduce of the slice (leaders only) */
            if (!leader_) {
                return UCC_OK;
            }
            sa.coll_type       = UCC_COLL_TYPE_ALLREDUCE;
            sa.flags          |= UCC_COLL_ARGS_FLAG_IN_PLACE;
            sa.src.info        = a_.dst.info;
            sa.src.info.buffer = dstp;
            sa.src.info.count  = cnt;
            if (dev_) {
                /* inter-node over host transports: stage D2H */
                uint8_t *h = hstg_[f % pdepth].data();
                ucc_status_t cs =
                    mc::copy(h, UCC_MEMORY_TYPE_HOST, dstp,
                             a_.dst.info.mem_type, cnt * dtsz_);
                if (cs != UCC_OK) {
                    return cs;
                }
                sa.src.info.buffer   = h;
                sa.src.info.mem_type = UCC_MEMORY_TYPE_HOST;
            }
            sa.dst.info        = sa.src.info;
            st_team            = team_->leaders_team.get();
            break;
        case 2: /* node bcast of the reduced slice */
            if (team_->node_team->size == 1) {
                return UCC_OK;
            }
            sa.coll_type       = UCC_COLL_TYPE_BCAST;
            sa.src.info        = a_.dst.info;
            sa.src.info.buffer = dstp;
            sa.src.info.count  = cnt;
            st_team            = team_->node_team.get();
            break;
        case 3: { /* AVG: scale the completed slice by 1/N */
            if (dev_) {
                ec_hip::ReduceArgs ra{};
                ra.dst     = dstp;
                ra.srcs[0] = dstp;
                ra.n_srcs  = 1;
                ra.count   = cnt;
                ra.dt      = dt_;
                ra.op      = (ucc_reduction_op_t)12; /* SUM w/ alpha */
                ra.alpha   = 1.0f / (float)team_->size;
                ucc_status_t rs = ec_hip::reduce(ra, nullptr);
                if (rs != UCC_OK) {
                    return rs;
                }
                return mc::device_sync();
            }
            const void *sp[1] = {dstp};
            ec_cpu::reduce(dstp, sp, 1, cnt, dt_, UCC_OP_SUM,
                           1.0 / (double)team_->size);
            return UCC_OK;
        }
        default:
            return UCC_ERR_INVALID_PARAM;
        }
        ucc_status_t st = ucc_collective_init(
            &sa, req, reinterpret_cast<ucc_team_h>(st_team));
        if (st != UCC_OK) {
            return st;
        }
        return ucc_collective_post(*req);
    }

    Team           *team_;
    ucc_coll_args_t a_;
    bool            leader_ = false, inplace_ = false, dev_ = false;
    uint64_t        count_ = 0;
    size_t          fc_ = 0, dtsz_ = 4;
    ucc_datatype_t  dt_ = UCC_DT_FLOAT32;
    std::vector<std::vector<uint8_t>> hstg_;
};

/* ---- hier allreduce (split_rail role): node reduce_scatterv ->
 * per-rail allreduce (rank p of every node forms rail p, so ALL ranks
 * drive inter-node traffic concurrently instead of just the leaders) ->
 * node allgatherv. Requires uniform node sizes (team->rails_ok).
 * Reference parity: cl/hier allreduce split_rail
 * (components/cl/hier/allreduce/allreduce_split_rail.c) — re-derived:
 * slices are packed into the user dst and every sub-step runs in-place,
 * so the inter-node phase moves exactly count/node_size elements per
 * rank with no repack. */
class SplitRailAllreduceTask final : public Task {
  public:
    SplitRailAllreduceTask(Context *ctx, Team *team,
                           const ucc_coll_args_t &args)
        : Task(ctx), team_(team), a_(args)
    {
    }
    ~SplitRailAllreduceTask() override
    {
        if (req_h_) {
            ucc_collective_finalize(req_h_);
        }
    }

    ucc_status_t post() override
    {
        phase_   = 0;
        inplace_ = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        count_   = a_.dst.info.count;
        dt_      = a_.dst.info.datatype;
        dtsz_    = ucc_dt_size(dt_);
        const uint32_t nn = (uint32_t)team_->node_ranks.size();
        int my_nidx = -1;
        for (size_t i = 0; i < team_->node_ranks.size(); i++) {
            if (team_->node_ranks[i] == team_->rank) {
                my_nidx = (int)i;
            }
        }
        if (my_nidx < 0 || !team_->rails_ok || !team_->rail_team) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        /* packed per-node-rank slices of the dst vector */
        cnt_.resize(nn);
        dsp_.resize(nn);
        uint64_t per = count_ / nn, rem = count_ % nn, off = 0;
        for (uint32_t r = 0; r < nn; r++) {
            cnt_[r] = per + (r < rem ? 1 : 0);
            dsp_[r] = off;
            off += cnt_[r];
        }
        my_off_ = dsp_[(size_t)my_nidx];
        my_cnt_ = cnt_[(size_t)my_nidx];
        if (!inplace_) { /* all sub-steps run in-place on dst */
            memcpy(a_.dst.info.buffer, a_.src.info.buffer,
                   count_ * dtsz_);
        }
        status = UCC_INPROGRESS;
        return step();
    }

    ucc_status_t progress() override { return step(); }

  private:
    ucc_status_t step()
    {
        while (true) {
            if (req_h_) {
                ucc_status_t st = ucc_collective_test(req_h_);
                if (st == UCC_INPROGRESS) {
                    return UCC_INPROGRESS;
                }
                ucc_collective_finalize(req_h_);
                req_h_ = nullptr;
                if (st != UCC_OK) {
                    return st;
                }
                phase_++;
            }
            ucc_coll_args_t sa{};
            sa.mask  = UCC_COLL_ARGS_FIELD_FLAGS;
            sa.flags = UCC_COLL_ARGS_FLAG_IN_PLACE |
                       UCC_COLL_ARGS_FLAG_COUNT_64BIT |
                       UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT;
            sa.op    = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
            switch (phase_) {
            case 0: { /* node reduce_scatterv: my packed slice lands at
                       * dst + my_off (in-place RSV convention) */
                sa.coll_type           = UCC_COLL_TYPE_REDUCE_SCATTERV;
                sa.dst.info_v.buffer   = a_.dst.info.buffer;
                sa.dst.info_v.counts   = (ucc_count_t *)cnt_.data();
                sa.dst.info_v.datatype = dt_;
                sa.dst.info_v.mem_type = a_.dst.info.mem_type;
                ucc_status_t st        = launch(team_->node_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 1: { /* rail allreduce of my slice (inter-node) */
                if (my_cnt_ == 0) {
                    phase_ = 2;
                    continue;
                }
                sa.coll_type         = UCC_COLL_TYPE_ALLREDUCE;
                sa.dst.info.buffer   = (uint8_t *)a_.dst.info.buffer +
                                     my_off_ * dtsz_;
                sa.dst.info.count    = my_cnt_;
                sa.dst.info.datatype = dt_;
                sa.dst.info.mem_type = a_.dst.info.mem_type;
                sa.src.info          = sa.dst.info;
                ucc_status_t st      = launch(team_->rail_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 2: { /* AVG: scale just my slice before the allgather */
                if (a_.op == UCC_OP_AVG && my_cnt_) {
                    uint8_t    *b = (uint8_t *)a_.dst.info.buffer +
                                 my_off_ * dtsz_;
                    const void *srcs[1] = {b};
                    ec_cpu::reduce(b, srcs, 1, my_cnt_, dt_, UCC_OP_SUM,
                                   1.0 / (double)team_->size);
                }
                sa.coll_type                = UCC_COLL_TYPE_ALLGATHERV;
                sa.dst.info_v.buffer        = a_.dst.info.buffer;
                sa.dst.info_v.counts        = (ucc_count_t *)cnt_.data();
                sa.dst.info_v.displacements = (ucc_aint_t *)dsp_.data();
                sa.dst.info_v.datatype      = dt_;
                sa.dst.info_v.mem_type      = a_.dst.info.mem_type;
                ucc_status_t st = launch(team_->node_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 3:
                return UCC_OK;
            default:
                return UCC_ERR_INVALID_PARAM;
            }
        }
    }

    ucc_status_t launch(Team *t, ucc_coll_args_t &sa)
    {
        ucc_status_t st = ucc_collective_init(
            &sa, &req_h_, reinterpret_cast<ucc_team_h>(t));
        if (st != UCC_OK) {
            return st;
        }
        return ucc_collective_post(req_h_);
    }

    Team                 *team_;
    ucc_coll_args_t       a_;
    ucc_coll_req_h        req_h_ = nullptr;
    int                   phase_ = 0;
    bool                  inplace_ = false;
    uint64_t              count_ = 0, my_off_ = 0, my_cnt_ = 0;
    size_t                dtsz_ = 4;
    ucc_datatype_t        dt_ = UCC_DT_FLOAT32;
    std::vector<uint64_t> cnt_, dsp_;
};

/* ---- split_rail over the fragment pipeline: the node-RSV of frag f+1
 * overlaps the rail allreduce of frag f and the node-AGV of f-1, so the
 * inter-node rail phase streams instead of idling the node links
 * (reference cl_hier.h:47-57 split_rail pipeline config role). */
class HierSplitRailPipeTask final : public PipelineTask {
  public:
    HierSplitRailPipeTask(Context *ctx, Team *team,
                          const ucc_coll_args_t &args, size_t frag_elems,
                          size_t depth)
        : PipelineTask(ctx), team_(team), a_(args)
    {
        count_   = a_.dst.info.count;
        dt_      = a_.dst.info.datatype;
        dtsz_    = ucc_dt_size(dt_);
        inplace_ = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        fc_      = frag_elems ? frag_elems : count_;
        n_frags  = (count_ + fc_ - 1) / fc_;
        n_stages = 3;
        pdepth   = depth ? depth : 2;
        pair_early = 0; /* node RSV and node AGV share node_team */
        pair_late  = 2;
        stage_post = [this](size_t f, size_t s, ucc_coll_req_h *r) {
            return do_stage(f, s, r);
        };
    }

    ucc_status_t post() override
    {
        const uint32_t nn = (uint32_t)team_->node_ranks.size();
        int            my_nidx = -1;
        for (size_t i = 0; i < team_->node_ranks.size(); i++) {
            if (team_->node_ranks[i] == team_->rank) {
                my_nidx = (int)i;
            }
        }
        if (my_nidx < 0 || !team_->rails_ok || !team_->rail_team) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        nidx_ = (uint32_t)my_nidx;
        nn_   = nn;
        /* per-fragment packed slice tables (pointers into these arrays
         * must outlive the in-flight sub-collectives) */
        fcnt_.assign(n_frags, {});
        fdsp_.assign(n_frags, {});
        for (size_t f = 0; f < n_frags; f++) {
            uint64_t b   = f * fc_;
            uint64_t cnt = count_ - b < fc_ ? count_ - b : fc_;
            auto    &cv  = fcnt_[f];
            auto    &dv  = fdsp_[f];
            cv.resize(nn);
            dv.resize(nn);
            uint64_t per = cnt / nn, rem = cnt % nn, off = 0;
            for (uint32_t r = 0; r < nn; r++) {
                cv[r] = per + (r < rem ? 1 : 0);
                dv[r] = off;
                off += cv[r];
            }
        }
        if (!inplace_) { /* all sub-steps run in-place on dst */
            memcpy(a_.dst.info.buffer, a_.src.info.buffer,
                   count_ * dtsz_);
        }
        return PipelineTask::post();
    }

  private:
    ucc_status_t do_stage(size_t f, size_t s, ucc_coll_req_h *req)
    {
        *req          = nullptr;
        uint64_t b    = f * fc_;
        uint8_t *base = (uint8_t *)a_.dst.info.buffer + b * dtsz_;
        uint64_t moff = fdsp_[f][nidx_], mcnt = fcnt_[f][nidx_];
        ucc_coll_args_t sa{};
        sa.mask  = UCC_COLL_ARGS_FIELD_FLAGS;
        sa.flags = UCC_COLL_ARGS_FLAG_IN_PLACE |
                   UCC_COLL_ARGS_FLAG_COUNT_64BIT |
                   UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT;
        sa.op    = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
        Team *st_team = nullptr;
        switch (s) {
        case 0: /* node RSV of the fragment slice */
            sa.coll_type           = UCC_COLL_TYPE_REDUCE_SCATTERV;
            sa.dst.info_v.buffer   = base;
            sa.dst.info_v.counts   = (ucc_count_t *)fcnt_[f].data();
            sa.dst.info_v.datatype = dt_;
            sa.dst.info_v.mem_type = a_.dst.info.mem_type;
            st_team                = team_->node_team.get();
            break;
        case 1: /* rail allreduce of my sub-slice */
            if (mcnt == 0) {
                return UCC_OK;
            }
            sa.coll_type         = UCC_COLL_TYPE_ALLREDUCE;
            sa.dst.info.buffer   = base + moff * dtsz_;
            sa.dst.info.count    = mcnt;
            sa.dst.info.datatype = dt_;
            sa.dst.info.mem_type = a_.dst.info.mem_type;
            sa.src.info          = sa.dst.info;
            st_team              = team_->rail_team.get();
            break;
        case 2: { /* AVG scale my sub-slice, then node AGV */
            if (a_.op == UCC_OP_AVG && mcnt) {
                uint8_t    *p       = base + moff * dtsz_;
                const void *srcs[1] = {p};
                ec_cpu::reduce(p, srcs, 1, mcnt, dt_, UCC_OP_SUM,
                               1.0 / (double)team_->size);
            }
            sa.coll_type                = UCC_COLL_TYPE_ALLGATHERV;
            sa.dst.info_v.buffer        = base;
            sa.dst.info_v.counts        = (ucc_count_t *)fcnt_[f].data();
            sa.dst.info_v.displacements = (ucc_aint_t *)fdsp_[f].data();
            sa.dst.info_v.datatype      = dt_;
            sa.dst.info_v.mem_type      = a_.dst.info.mem_type;
            st_team                     = team_->node_team.get();
            break;
        }
        default:
            return UCC_ERR_INVALID_PARAM;
        }
        ucc_status_t st = ucc_collective_init(
            &sa, req, reinterpret_cast<ucc_team_h>(st_team));
        if (st != UCC_OK) {
            return st;
        }
        return ucc_collective_post(*req);
    }

    Team           *team_;
    ucc_coll_args_t a_;
    bool            inplace_ = false;
    uint64_t        count_ = 0;
    size_t          fc_ = 0, dtsz_ = 4;
    uint32_t        nidx_ = 0, nn_ = 1;
    ucc_datatype_t  dt_ = UCC_DT_FLOAT32;
    std::vector<std::vector<uint64_t>> fcnt_, fdsp_;
};

static uint64_t v_cnt_at(const ucc_coll_args_t &a, const void *counts,
                         uint32_t r)
{
    return (a.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
               ? ((const uint64_t *)counts)[r]
               : ((const uint32_t *)counts)[r];
}
static uint64_t v_dsp_at(const ucc_coll_args_t &a, const void *displs,
                         uint32_t r)
{
    return (a.flags & UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
               ? ((const uint64_t *)displs)[r]
               : ((const uint32_t *)displs)[r];
}

/* ---- hier alltoallv (node-aggregated): members pack their send data
 * and gather it (plus their count row+column) to the node leader;
 * leaders run ONE aggregated alltoallv (per-pair payload ordered
 * dst-member-major, src-member asc); leaders repack per dst member and
 * scatterv; members unpack by the node-major src order. Aggregation
 * collapses nranks^2 small inter-node messages into nleaders^2 — the
 * reference's a2av_node_thresh role (components/cl/hier/alltoallv/
 * alltoallv.c), re-derived. Small-message path: registered below
 * CL_HIER_A2AV_NODE_THRESH only. */
class HierAlltoallvTask final : public Task {
  public:
    HierAlltoallvTask(Context *ctx, Team *team,
                      const ucc_coll_args_t &args)
        : Task(ctx), team_(team), a_(args)
    {
    }
    ~HierAlltoallvTask() override
    {
        if (req_h_) {
            ucc_collective_finalize(req_h_);
        }
    }

    ucc_status_t post() override
    {
        phase_  = 0;
        leader_ = team_->leaders_oob->my_idx >= 0;
        if (a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        const uint32_t n = team_->size;
        sdtsz_ = ucc_dt_size(a_.src.info_v.datatype);
        rdtsz_ = ucc_dt_size(a_.dst.info_v.datatype);
        row_.resize(n);
        col_.resize(n);
        for (uint32_t j = 0; j < n; j++) {
            row_[j] = v_cnt_at(a_, a_.src.info_v.counts, j) * sdtsz_;
            col_[j] = v_cnt_at(a_, a_.dst.info_v.counts, j) * rdtsz_;
        }
        /* node-major global src order (nodes in leader order, members in
         * team-rank order) — the unpack order every rank agrees on */
        nodes_.assign(team_->leader_ranks.size(),
                      std::vector<uint32_t>());
        for (size_t k = 0; k < team_->leader_ranks.size(); k++) {
            uint64_t h = team_->procs[team_->leader_ranks[k]].host_hash;
            for (uint32_t r = 0; r < n; r++) {
                if (team_->procs[r].host_hash == h) {
                    nodes_[k].push_back(r);
                }
            }
        }
        /* pack my send data: dst team-rank ascending */
        size_t tot_s = 0, tot_r = 0;
        for (uint32_t j = 0; j < n; j++) {
            tot_s += row_[j];
            tot_r += col_[j];
        }
        send_pk_.resize(tot_s);
        recv_pk_.resize(tot_r);
        {
            const uint8_t *src = (const uint8_t *)a_.src.info_v.buffer;
            size_t         off = 0;
            for (uint32_t j = 0; j < n; j++) {
                if (row_[j]) {
                    memcpy(send_pk_.data() + off,
                           src + v_dsp_at(a_,
                                          a_.src.info_v.displacements,
                                          j) *
                                     sdtsz_,
                           row_[j]);
                    off += row_[j];
                }
            }
        }
        if (leader_) {
            meta_.assign((size_t)nodes_[my_node_()].size() * 2 * n, 0);
        }
        my_meta_.resize(2 * n);
        for (uint32_t j = 0; j < n; j++) {
            my_meta_[j]     = row_[j];
            my_meta_[n + j] = col_[j];
        }
        status = UCC_INPROGRESS;
        return step();
    }

    ucc_status_t progress() override { return step(); }

  private:
    size_t my_node_() const
    {
        for (size_t k = 0; k < nodes_.size(); k++) {
            for (uint32_t r : nodes_[k]) {
                if (r == team_->rank) {
                    return k;
                }
            }
        }
        return 0;
    }

    /* s_{i -> j} in bytes, from gathered node metadata (leader only;
     * i must be a member of my node) */
    uint64_t s_row(size_t member_idx, uint32_t j) const
    {
        return meta_[member_idx * 2 * team_->size + j];
    }
    /* r_{j <- i} = s_{i -> j} for my member j, any global i */
    uint64_t r_col(size_t member_idx, uint32_t i) const
    {
        return meta_[member_idx * 2 * team_->size + team_->size + i];
    }

    ucc_status_t step()
    {
        const uint32_t n  = team_->size;
        const size_t   nl = nodes_.size();
        while (true) {
            if (req_h_) {
                ucc_status_t st = ucc_collective_test(req_h_);
                if (st == UCC_INPROGRESS) {
                    return UCC_INPROGRESS;
                }
                ucc_collective_finalize(req_h_);
                req_h_ = nullptr;
                if (st != UCC_OK) {
                    return st;
                }
                phase_++;
            }
            ucc_coll_args_t sa{};
            sa.mask  = UCC_COLL_ARGS_FIELD_FLAGS;
            sa.flags = UCC_COLL_ARGS_FLAG_COUNT_64BIT |
                       UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT;
            switch (phase_) {
            case 0: { /* node gather of [row|column] metadata */
                sa.coll_type         = UCC_COLL_TYPE_GATHER;
                sa.root              = 0;
                sa.src.info.buffer   = my_meta_.data();
                sa.src.info.count    = 2 * n;
                sa.src.info.datatype = UCC_DT_UINT64;
                sa.src.info.mem_type = UCC_MEMORY_TYPE_HOST;
                sa.dst.info          = sa.src.info;
                sa.dst.info.count    = (uint64_t)2 * n *
                                    nodes_[my_node_()].size();
                if (leader_) {
                    sa.dst.info.buffer = meta_.data();
                }
                ucc_status_t st = launch(team_->node_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 1: { /* node gatherv of packed send payloads */
                const auto &mem = nodes_[my_node_()];
                sa.coll_type         = UCC_COLL_TYPE_GATHERV;
                sa.root              = 0;
                sa.src.info.buffer   = send_pk_.data();
                sa.src.info.count    = send_pk_.size();
                sa.src.info.datatype = UCC_DT_UINT8;
                sa.src.info.mem_type = UCC_MEMORY_TYPE_HOST;
                if (leader_) {
                    sub_cnt_.resize(mem.size());
                    sub_dsp_.resize(mem.size());
                    size_t off = 0;
                    for (size_t m = 0; m < mem.size(); m++) {
                        uint64_t t = 0;
                        for (uint32_t j = 0; j < n; j++) {
                            t += s_row(m, j);
                        }
                        sub_cnt_[m] = t;
                        sub_dsp_[m] = off;
                        off += t;
                    }
                    gath_.resize(off);
                    sa.dst.info_v.buffer        = gath_.data();
                    sa.dst.info_v.counts = (ucc_count_t *)sub_cnt_.data();
                    sa.dst.info_v.displacements =
                        (ucc_aint_t *)sub_dsp_.data();
                    sa.dst.info_v.datatype = UCC_DT_UINT8;
                    sa.dst.info_v.mem_type = UCC_MEMORY_TYPE_HOST;
                }
                ucc_status_t st = launch(team_->node_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 2: { /* leaders aggregated alltoallv */
                if (!leader_) {
                    phase_ = 3;
                    continue;
                }
                const auto &mem = nodes_[my_node_()];
                /* send buf to leader l: for dst member j of node l (in
                 * node order), for src member m of my node: s_m[j] */
                a2a_scnt_.assign(nl, 0);
                a2a_sdsp_.assign(nl, 0);
                a2a_rcnt_.assign(nl, 0);
                a2a_rdsp_.assign(nl, 0);
                size_t tot = 0;
                for (size_t l = 0; l < nl; l++) {
                    a2a_sdsp_[l] = tot;
                    for (uint32_t j : nodes_[l]) {
                        for (size_t m = 0; m < mem.size(); m++) {
                            a2a_scnt_[l] += s_row(m, j);
                        }
                    }
                    tot += a2a_scnt_[l];
                }
                a2a_send_.resize(tot);
                /* member m's packed block starts at sub_dsp_[m]; within
                 * it, dst j's piece is at the row prefix sum */
                {
                    std::vector<uint64_t> moff(mem.size());
                    size_t w = 0;
                    for (size_t l = 0; l < nl; l++) {
                        for (uint32_t j : nodes_[l]) {
                            for (size_t m = 0; m < mem.size(); m++) {
                                uint64_t pre = 0;
                                for (uint32_t j2 = 0; j2 < j; j2++) {
                                    pre += s_row(m, j2);
                                }
                                uint64_t len = s_row(m, j);
                                if (len) {
                                    memcpy(a2a_send_.data() + w,
                                           gath_.data() + sub_dsp_[m] +
                                               pre,
                                           len);
                                }
                                w += len;
                            }
                        }
                    }
                    (void)moff;
                }
                size_t rtot = 0;
                for (size_t l = 0; l < nl; l++) {
                    a2a_rdsp_[l] = rtot;
                    for (size_t m = 0; m < mem.size(); m++) {
                        for (uint32_t i : nodes_[l]) {
                            a2a_rcnt_[l] += r_col(m, i);
                        }
                    }
                    rtot += a2a_rcnt_[l];
                }
                a2a_recv_.resize(rtot);
                sa.coll_type                = UCC_COLL_TYPE_ALLTOALLV;
                sa.src.info_v.buffer        = a2a_send_.data();
                sa.src.info_v.counts        = (ucc_count_t *)a2a_scnt_.data();
                sa.src.info_v.displacements = (ucc_aint_t *)a2a_sdsp_.data();
                sa.src.info_v.datatype      = UCC_DT_UINT8;
                sa.src.info_v.mem_type      = UCC_MEMORY_TYPE_HOST;
                sa.dst.info_v.buffer        = a2a_recv_.data();
                sa.dst.info_v.counts        = (ucc_count_t *)a2a_rcnt_.data();
                sa.dst.info_v.displacements = (ucc_aint_t *)a2a_rdsp_.data();
                sa.dst.info_v.datatype      = UCC_DT_UINT8;
                sa.dst.info_v.mem_type      = UCC_MEMORY_TYPE_HOST;
                ucc_status_t st = launch(team_->leaders_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 3: { /* node scatterv of per-dst-member blocks, ordered
                       * by src in node-major global order */
                const auto &mem = nodes_[my_node_()];
                sa.coll_type = UCC_COLL_TYPE_SCATTERV;
                sa.root      = 0;
                if (leader_) {
                    /* repack: arriving block from node l is
                     * (dst member-major, src member asc); target is
                     * (dst member-major over ALL srcs node-major) */
                    sub_cnt_.assign(mem.size(), 0);
                    sub_dsp_.assign(mem.size(), 0);
                    size_t tot = 0;
                    for (size_t m = 0; m < mem.size(); m++) {
                        sub_dsp_[m] = tot;
                        for (uint32_t i = 0; i < n; i++) {
                            sub_cnt_[m] += r_col(m, i);
                        }
                        tot += sub_cnt_[m];
                    }
                    scat_.resize(tot);
                    std::vector<uint64_t> roff(nl);
                    for (size_t l = 0; l < nl; l++) {
                        roff[l] = a2a_rdsp_[l];
                    }
                    for (size_t m = 0; m < mem.size(); m++) {
                        size_t w = sub_dsp_[m];
                        for (size_t l = 0; l < nl; l++) {
                            for (uint32_t i : nodes_[l]) {
                                uint64_t len = r_col(m, i);
                                if (len) {
                                    memcpy(scat_.data() + w,
                                           a2a_recv_.data() + roff[l],
                                           len);
                                }
                                roff[l] += len;
                                w += len;
                            }
                        }
                    }
                    sa.src.info_v.buffer        = scat_.data();
                    sa.src.info_v.counts = (ucc_count_t *)sub_cnt_.data();
                    sa.src.info_v.displacements =
                        (ucc_aint_t *)sub_dsp_.data();
                    sa.src.info_v.datatype = UCC_DT_UINT8;
                    sa.src.info_v.mem_type = UCC_MEMORY_TYPE_HOST;
                }
                sa.dst.info.buffer   = recv_pk_.data();
                sa.dst.info.count    = recv_pk_.size();
                sa.dst.info.datatype = UCC_DT_UINT8;
                sa.dst.info.mem_type = UCC_MEMORY_TYPE_HOST;
                ucc_status_t st = launch(team_->node_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 4: { /* unpack: src blocks arrive node-major */
                uint8_t *dst = (uint8_t *)a_.dst.info_v.buffer;
                size_t   off = 0;
                for (size_t l = 0; l < nl; l++) {
                    for (uint32_t i : nodes_[l]) {
                        if (col_[i]) {
                            memcpy(dst + v_dsp_at(
                                             a_,
                                             a_.dst.info_v.displacements,
                                             i) *
                                             rdtsz_,
                                   recv_pk_.data() + off, col_[i]);
                            off += col_[i];
                        }
                    }
                }
                return UCC_OK;
            }
            default:
                return UCC_ERR_INVALID_PARAM;
            }
        }
    }

    ucc_status_t launch(Team *t, ucc_coll_args_t &sa)
    {
        ucc_status_t st = ucc_collective_init(
            &sa, &req_h_, reinterpret_cast<ucc_team_h>(t));
        if (st != UCC_OK) {
            return st;
        }
        return ucc_collective_post(req_h_);
    }

    Team                              *team_;
    ucc_coll_args_t                    a_;
    ucc_coll_req_h                     req_h_ = nullptr;
    int                                phase_ = 0;
    bool                               leader_ = false;
    size_t                             sdtsz_ = 1, rdtsz_ = 1;
    std::vector<uint64_t>              row_, col_, my_meta_, meta_;
    std::vector<std::vector<uint32_t>> nodes_;
    std::vector<uint64_t> sub_cnt_, sub_dsp_, a2a_scnt_, a2a_sdsp_,
        a2a_rcnt_, a2a_rdsp_;
    std::vector<uint8_t> send_pk_, recv_pk_, gath_, a2a_send_, a2a_recv_,
        scat_;
};

/* ---- hier allgatherv: node gatherv to the leader (packed) -> leaders
 * allgatherv of node-aggregated blocks -> node bcast of the full packed
 * vector -> local unpack into user dst positions. All per-rank counts
 * are global knowledge (dst.info_v), so every phase's geometry is
 * locally computable. Reference parity: cl/hier allgatherv
 * (components/cl/hier/allgatherv/) — re-derived with node-major packed
 * relay and a final local unpack. */
class HierAllgathervTask final : public Task {
  public:
    HierAllgathervTask(Context *ctx, Team *team,
                       const ucc_coll_args_t &args)
        : Task(ctx), team_(team), a_(args)
    {
    }
    ~HierAllgathervTask() override
    {
        if (req_h_) {
            ucc_collective_finalize(req_h_);
        }
        if (dscr_) {
            mc::scratch_free(dscr_, total_, UCC_MEMORY_TYPE_CUDA);
        }
    }

    ucc_status_t post() override
    {
        phase_   = 0;
        leader_  = team_->leaders_oob->my_idx >= 0;
        inplace_ = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        dt_      = a_.dst.info_v.datatype;
        dtsz_    = ucc_dt_size(dt_);
        dev_     = mc::is_device_mt(a_.dst.info_v.mem_type);
        const uint32_t n = team_->size;
        /* node-major packed layout: nodes in leader order, members in
         * team-rank order within each node */
        cnt_.resize(n);
        udsp_.resize(n);
        pdsp_.resize(n); /* packed offset of rank r's block            */
        node_bytes_.assign(team_->leader_ranks.size(), 0);
        node_poff_.assign(team_->leader_ranks.size(), 0);
        for (uint32_t r = 0; r < n; r++) {
            cnt_[r]  = v_cnt_at(a_, a_.dst.info_v.counts, r) * dtsz_;
            udsp_[r] = v_dsp_at(a_, a_.dst.info_v.displacements, r) * dtsz_;
        }
        size_t off = 0;
        my_node_idx_ = -1;
        for (size_t k = 0; k < team_->leader_ranks.size(); k++) {
            uint64_t h = team_->procs[team_->leader_ranks[k]].host_hash;
            node_poff_[k] = off;
            for (uint32_t r = 0; r < n; r++) {
                if (team_->procs[r].host_hash == h) {
                    pdsp_[r] = off;
                    off += cnt_[r];
                    node_bytes_[k] += cnt_[r];
                    if (r == team_->rank) {
                        my_node_idx_ = (int)k;
                    }
                }
            }
        }
        total_ = off;
        if (my_node_idx_ < 0) {
            return UCC_ERR_INVALID_PARAM;
        }
        packed_.resize(total_);
        if (dev_) { /* device run: packed layout lives in device
                     * scratch; the host packed_ stages only the
                     * leaders' inter-node hop (rab_dev pattern) */
            void *p = nullptr;
            ucc_status_t st =
                mc::scratch_alloc(&p, total_, UCC_MEMORY_TYPE_CUDA);
            if (st != UCC_OK) {
                return st;
            }
            dscr_ = (uint8_t *)p;
        }
        status = UCC_INPROGRESS;
        return step();
    }

    ucc_status_t progress() override { return step(); }

  private:
    ucc_status_t step()
    {
        while (true) {
            if (req_h_) {
                ucc_status_t st = ucc_collective_test(req_h_);
                if (st == UCC_INPROGRESS) {
                    return UCC_INPROGRESS;
                }
                ucc_collective_finalize(req_h_);
                req_h_ = nullptr;
                if (st != UCC_OK) {
                    return st;
                }
                phase_++;
            }
            ucc_coll_args_t sa{};
            sa.mask  = UCC_COLL_ARGS_FIELD_FLAGS;
            sa.flags = UCC_COLL_ARGS_FLAG_COUNT_64BIT |
                       UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT;
            const uint8_t *my_src =
                inplace_ ? (const uint8_t *)a_.dst.info_v.buffer +
                               udsp_[team_->rank]
                         : (const uint8_t *)a_.src.info.buffer;
            switch (phase_) {
            case 0: { /* node gatherv -> leader's packed block */
                sa.coll_type         = UCC_COLL_TYPE_GATHERV;
                sa.root              = 0; /* node leader = lowest rank */
                sa.src.info.buffer   = (void *)my_src;
                sa.src.info.count    = cnt_[team_->rank] / dtsz_;
                sa.src.info.datatype = dt_;
                sa.src.info.mem_type =
                    dev_ ? UCC_MEMORY_TYPE_CUDA : UCC_MEMORY_TYPE_HOST;
                if (leader_) {
                    const auto  &nr = team_->node_ranks;
                    sub_cnt_.resize(nr.size());
                    sub_dsp_.resize(nr.size());
                    size_t base = node_poff_[(size_t)my_node_idx_];
                    for (size_t j = 0; j < nr.size(); j++) {
                        sub_cnt_[j] = cnt_[nr[j]] / dtsz_;
                        sub_dsp_[j] = (pdsp_[nr[j]] - base) / dtsz_;
                    }
                    sa.dst.info_v.buffer =
                        (dev_ ? dscr_ : packed_.data()) + base;
                    sa.dst.info_v.counts = (ucc_count_t *)sub_cnt_.data();
                    sa.dst.info_v.displacements =
                        (ucc_aint_t *)sub_dsp_.data();
                    sa.dst.info_v.datatype = dt_;
                    sa.dst.info_v.mem_type = dev_ ? UCC_MEMORY_TYPE_CUDA
                                                  : UCC_MEMORY_TYPE_HOST;
                }
                ucc_status_t st = launch(team_->node_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 1: { /* leaders allgatherv of node blocks (in-place:
                       * my node's block is already at its packed spot) */
                if (!leader_) {
                    phase_ = 2;
                    continue;
                }
                if (dev_) { /* stage my node's packed block D2H */
                    size_t       base = node_poff_[(size_t)my_node_idx_];
                    ucc_status_t cs   = mc::copy(
                        packed_.data() + base, UCC_MEMORY_TYPE_HOST,
                        dscr_ + base, UCC_MEMORY_TYPE_CUDA,
                        node_bytes_[(size_t)my_node_idx_]);
                    if (cs != UCC_OK) {
                        return cs;
                    }
                }
                const size_t nl = team_->leader_ranks.size();
                sub_cnt_.resize(nl);
                sub_dsp_.resize(nl);
                for (size_t k = 0; k < nl; k++) {
                    sub_cnt_[k] = node_bytes_[k] / dtsz_;
                    sub_dsp_[k] = node_poff_[k] / dtsz_;
                }
                sa.flags |= UCC_COLL_ARGS_FLAG_IN_PLACE;
                sa.coll_type         = UCC_COLL_TYPE_ALLGATHERV;
                sa.dst.info_v.buffer = packed_.data();
                sa.dst.info_v.counts = (ucc_count_t *)sub_cnt_.data();
                sa.dst.info_v.displacements = (ucc_aint_t *)sub_dsp_.data();
                sa.dst.info_v.datatype = dt_;
                sa.dst.info_v.mem_type = UCC_MEMORY_TYPE_HOST;
                ucc_status_t st = launch(team_->leaders_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 2: { /* node bcast of the full packed vector */
                if (dev_ && leader_) { /* unstage the gathered total */
                    ucc_status_t cs = mc::copy(
                        dscr_, UCC_MEMORY_TYPE_CUDA, packed_.data(),
                        UCC_MEMORY_TYPE_HOST, total_);
                    if (cs != UCC_OK) {
                        return cs;
                    }
                }
                sa.coll_type         = UCC_COLL_TYPE_BCAST;
                sa.root              = 0;
                sa.src.info.buffer   = dev_ ? dscr_ : packed_.data();
                sa.src.info.count    = total_ / dtsz_;
                sa.src.info.datatype = dt_;
                sa.src.info.mem_type =
                    dev_ ? UCC_MEMORY_TYPE_CUDA : UCC_MEMORY_TYPE_HOST;
                ucc_status_t st = launch(team_->node_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 3: { /* local unpack into user dst positions */
                uint8_t *dst = (uint8_t *)a_.dst.info_v.buffer;
                for (uint32_t r = 0; r < team_->size; r++) {
                    if (!cnt_[r]) {
                        continue;
                    }
                    if (dev_) {
                        ucc_status_t cs = mc::copy(
                            dst + udsp_[r], UCC_MEMORY_TYPE_CUDA,
                            dscr_ + pdsp_[r], UCC_MEMORY_TYPE_CUDA,
                            cnt_[r]);
                        if (cs != UCC_OK) {
                            return cs;
                        }
                    } else {
                        memcpy(dst + udsp_[r], packed_.data() + pdsp_[r],
                               cnt_[r]);
                    }
                }
                return UCC_OK;
            }
            default:
                return UCC_ERR_INVALID_PARAM;
            }
        }
    }

    ucc_status_t launch(Team *t, ucc_coll_args_t &sa)
    {
        ucc_status_t st = ucc_collective_init(
            &sa, &req_h_, reinterpret_cast<ucc_team_h>(t));
        if (st != UCC_OK) {
            return st;
        }
        return ucc_collective_post(req_h_);
    }

    Team                 *team_;
    ucc_coll_args_t       a_;
    ucc_coll_req_h        req_h_ = nullptr;
    int                   phase_ = 0, my_node_idx_ = -1;
    bool                  leader_ = false, inplace_ = false,
                          dev_ = false;
    uint8_t              *dscr_ = nullptr;
    size_t                dtsz_ = 4, total_ = 0;
    ucc_datatype_t        dt_ = UCC_DT_FLOAT32;
    std::vector<size_t>   cnt_, udsp_, pdsp_, node_bytes_, node_poff_;
    std::vector<uint64_t> sub_cnt_, sub_dsp_;
    std::vector<uint8_t>  packed_;
};

/* ---- hier reduce (2step role): node reduce -> leaders reduce to the
 * root's node leader -> [node bcast delivers to a non-leader root].
 * Reference parity: cl/hier reduce 2step
 * (components/cl/hier/reduce/reduce_2step.c) — re-derived. Intermediate
 * partials live in task-owned host scratch; only the root's user dst is
 * ever written. */
class HierReduceTask final : public Task {
  public:
    HierReduceTask(Context *ctx, Team *team, const ucc_coll_args_t &args)
        : Task(ctx), team_(team), a_(args)
    {
    }
    ~HierReduceTask() override
    {
        if (req_h_) {
            ucc_collective_finalize(req_h_);
        }
        if (dscr_) {
            mc::scratch_free(dscr_, count_ * dtsz_,
                             UCC_MEMORY_TYPE_CUDA);
        }
    }

    ucc_status_t post() override
    {
        phase_   = 0;
        leader_  = team_->leaders_oob->my_idx >= 0;
        root_    = (uint32_t)a_.root;
        is_root_ = team_->rank == root_;
        count_   = a_.src.info.count;
        dt_      = a_.src.info.datatype;
        if (is_root_) {
            count_ = a_.dst.info.count;
            dt_    = a_.dst.info.datatype;
        }
        dtsz_ = ucc_dt_size(dt_);
        on_root_node_ = false;
        for (uint32_t nr : team_->node_ranks) {
            if (nr == root_) {
                on_root_node_ = true;
            }
        }
        root_leader_idx_ = -1;
        {
            uint64_t rhost = team_->procs[root_].host_hash;
            for (size_t i = 0; i < team_->leader_ranks.size(); i++) {
                if (team_->procs[team_->leader_ranks[i]].host_hash ==
                    rhost) {
                    root_leader_idx_ = (int)i;
                    break;
                }
            }
        }
        root_is_leader_ =
            root_leader_idx_ >= 0 &&
            team_->leader_ranks[(size_t)root_leader_idx_] == root_;
        dev_ = mc::is_device_mt(is_root_ ? a_.dst.info.mem_type
                                         : a_.src.info.mem_type);
        if (leader_ || on_root_node_) {
            if (dev_) {
                /* node/delivery phases stay on a DEVICE scratch; only
                 * the inter-node leader phase stages through the host */
                if (!dscr_ &&
                    mc::scratch_alloc(&dscr_, count_ * dtsz_,
                                      UCC_MEMORY_TYPE_CUDA) != UCC_OK) {
                    return UCC_ERR_NO_MEMORY;
                }
                if (leader_) {
                    scratch_.resize(count_ * dtsz_);
                }
            } else {
                scratch_.resize(count_ * dtsz_);
            }
        }
        status = UCC_INPROGRESS;
        return step();
    }

    ucc_status_t progress() override { return step(); }

  private:
    ucc_status_t step()
    {
        while (true) {
            if (req_h_) {
                ucc_status_t st = ucc_collective_test(req_h_);
                if (st == UCC_INPROGRESS) {
                    return UCC_INPROGRESS;
                }
                ucc_collective_finalize(req_h_);
                req_h_ = nullptr;
                if (st != UCC_OK) {
                    return st;
                }
                if (unstage_) { /* H2D: leaders result back to device */
                    unstage_ = false;
                    void *dv = root_is_leader_ ? a_.dst.info.buffer
                                               : dscr_;
                    ucc_status_t cs = mc::copy(
                        dv, UCC_MEMORY_TYPE_CUDA, scratch_.data(),
                        UCC_MEMORY_TYPE_HOST, count_ * dtsz_);
                    if (cs != UCC_OK) {
                        return cs;
                    }
                }
                phase_++;
            }
            ucc_coll_args_t sa{};
            sa.mask  = UCC_COLL_ARGS_FIELD_FLAGS;
            sa.flags = 0;
            sa.op    = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
            switch (phase_) {
            case 0: { /* node reduce to my node's leader (node rank 0) */
                sa.coll_type         = UCC_COLL_TYPE_REDUCE;
                sa.root              = 0;
                sa.src.info          = a_.src.info;
                sa.src.info.count    = count_;
                sa.src.info.datatype = dt_;
                if (is_root_ && (a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE)) {
                    sa.src.info.buffer   = a_.dst.info.buffer;
                    sa.src.info.mem_type = a_.dst.info.mem_type;
                }
                sa.dst.info          = sa.src.info;
                sa.dst.info.buffer   = dev_ ? dscr_
                                            : scratch_.data();
                ucc_status_t st = launch(team_->node_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 1: { /* leaders reduce to the root's node leader */
                if (!leader_) {
                    phase_ = 2;
                    continue;
                }
                if (dev_) { /* stage the node partial D2H */
                    ucc_status_t cs = mc::copy(
                        scratch_.data(), UCC_MEMORY_TYPE_HOST, dscr_,
                        UCC_MEMORY_TYPE_CUDA, count_ * dtsz_);
                    if (cs != UCC_OK) {
                        return cs;
                    }
                }
                sa.coll_type         = UCC_COLL_TYPE_REDUCE;
                sa.root              = (uint64_t)root_leader_idx_;
                sa.src.info.buffer   = scratch_.data();
                sa.src.info.count    = count_;
                sa.src.info.datatype = dt_;
                sa.src.info.mem_type = UCC_MEMORY_TYPE_HOST;
                sa.dst.info          = sa.src.info;
                if (team_->leaders_oob->my_idx == root_leader_idx_) {
                    sa.dst.info.buffer =
                        (root_is_leader_ && !dev_)
                            ? a_.dst.info.buffer
                            : scratch_.data();
                    if (root_is_leader_ && !dev_) {
                        sa.dst.info.mem_type = a_.dst.info.mem_type;
                    }
                    unstage_ = dev_; /* H2D the result on completion */
                }
                ucc_status_t st = launch(team_->leaders_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 2: { /* non-leader root: node bcast delivers the total
                       * (non-root node members receive into scratch) */
                if (!on_root_node_ || root_is_leader_) {
                    phase_ = 3;
                    continue;
                }
                sa.coll_type         = UCC_COLL_TYPE_BCAST;
                sa.root              = 0;
                sa.src.info.buffer   = is_root_ ? a_.dst.info.buffer
                                      : dev_    ? dscr_
                                                : scratch_.data();
                sa.src.info.count    = count_;
                sa.src.info.datatype = dt_;
                sa.src.info.mem_type =
                    is_root_ ? a_.dst.info.mem_type
                    : dev_   ? UCC_MEMORY_TYPE_CUDA
                             : UCC_MEMORY_TYPE_HOST;
                ucc_status_t st = launch(team_->node_team.get(), sa);
                if (st != UCC_OK) {
                    return st;
                }
                break;
            }
            case 3: { /* AVG post-scale at the root */
                if (is_root_ && a_.op == UCC_OP_AVG) {
                    if (dev_) {
                        ec_hip::ReduceArgs ra{};
                        ra.dst     = a_.dst.info.buffer;
                        ra.srcs[0] = a_.dst.info.buffer;
                        ra.n_srcs  = 1;
                        ra.count   = count_;
                        ra.dt      = dt_;
                        ra.op      = (ucc_reduction_op_t)12;
                        ra.alpha   = 1.0f / (float)team_->size;
                        ucc_status_t rs = ec_hip::reduce(ra, nullptr);
                        if (rs != UCC_OK) {
                            return rs;
                        }
                        ucc_status_t ds = mc::device_sync();
                        if (ds != UCC_OK) {
                            return ds;
                        }
                    } else {
                        const void *srcs[1] = {a_.dst.info.buffer};
                        ec_cpu::reduce(a_.dst.info.buffer, srcs, 1,
                                       count_, dt_, UCC_OP_SUM,
                                       1.0 / (double)team_->size);
                    }
                }
                return UCC_OK;
            }
            default:
                return UCC_ERR_INVALID_PARAM;
            }
        }
    }

    ucc_status_t launch(Team *t, ucc_coll_args_t &sa)
    {
        ucc_status_t st = ucc_collective_init(
            &sa, &req_h_, reinterpret_cast<ucc_team_h>(t));
        if (st != UCC_OK) {
            return st;
        }
        return ucc_collective_post(req_h_);
    }

    Team                *team_;
    ucc_coll_args_t      a_;
    ucc_coll_req_h       req_h_ = nullptr;
    int                  phase_ = 0;
    bool                 leader_ = false, is_root_ = false;
    bool                 on_root_node_ = false, root_is_leader_ = false;
    uint32_t             root_ = 0;
    int                  root_leader_idx_ = -1;
    uint64_t             count_ = 0;
    size_t               dtsz_ = 4;
    ucc_datatype_t       dt_ = UCC_DT_FLOAT32;
    bool                 dev_ = false, unstage_ = false;
    void                *dscr_ = nullptr;
    std::vector<uint8_t> scratch_;
};

/* ---- hier barrier: node fanin -> leaders barrier -> node fanout.     */
class HierBarrierTask final : public Task {
  public:
    HierBarrierTask(Context *ctx, Team *team, const ucc_coll_args_t &args)
        : Task(ctx), team_(team)
    {
        (void)args;
    }
    ~HierBarrierTask() override
    {
        if (req_h_) {
            ucc_collective_finalize(req_h_);
        }
    }

    ucc_status_t post() override
    {
        phase_  = 0;
        leader_ = team_->leaders_oob->my_idx >= 0;
        status  = UCC_INPROGRESS;
        return step();
    }

    ucc_status_t progress() override { return step(); }

  private:
    ucc_status_t step()
    {
        while (true) {
            if (req_h_) {
                ucc_status_t st = ucc_collective_test(req_h_);
                if (st == UCC_INPROGRESS) {
                    return UCC_INPROGRESS;
                }
                ucc_collective_finalize(req_h_);
                req_h_ = nullptr;
                if (st != UCC_OK) {
                    return st;
                }
                phase_++;
            }
            static const ucc_coll_type_t kPhase[3] = {
                UCC_COLL_TYPE_FANIN, UCC_COLL_TYPE_BARRIER,
                UCC_COLL_TYPE_FANOUT};
            if (phase_ >= 3) {
                return UCC_OK;
            }
            if (phase_ == 1 && !leader_) {
                phase_ = 2;
                continue;
            }
            ucc_coll_args_t sa{};
            sa.coll_type = kPhase[phase_];
            sa.root      = 0;
            Team *t = phase_ == 1 ? team_->leaders_team.get()
                                  : team_->node_team.get();
            ucc_status_t st = ucc_collective_init(
                &sa, &req_h_, reinterpret_cast<ucc_team_h>(t));
            if (st != UCC_OK) {
                return st;
            }
            st = ucc_collective_post(req_h_);
            if (st != UCC_OK) {
                return st;
            }
        }
    }

    Team          *team_;
    ucc_coll_req_h req_h_ = nullptr;
    int            phase_ = 0;
    bool           leader_ = false;
};

/* ---- hier bcast (2step role): [root's node bcast from root] ->
 * leaders bcast from the root's leader -> node bcast from each leader.
 * The first hop only runs on the root's node; other nodes join at the
 * leaders phase. */
class HierBcastTask final : public Task {
  public:
    HierBcastTask(Context *ctx, Team *team, const ucc_coll_args_t &args)
        : Task(ctx), team_(team), a_(args)
    {
    }
    ~HierBcastTask() override
    {
        if (req_h_) {
            ucc_collective_finalize(req_h_);
        }
    }

    ucc_status_t post() override
    {
        phase_  = 0;
        leader_ = team_->leaders_oob->my_idx >= 0;
        root_   = (uint32_t)a_.root;
        /* my node's members and the root's node leader */
        on_root_node_ = false;
        my_node_idx_  = -1;
        for (size_t i = 0; i < team_->node_ranks.size(); i++) {
            if (team_->node_ranks[i] == root_) {
                on_root_node_ = true;
            }
            if (team_->node_ranks[i] == team_->rank) {
                my_node_idx_ = (int)i;
            }
        }
        root_leader_idx_ = -1;
        {
            /* the leader of the root's node = the smallest team rank on
             * the root's host (leaders list is per-host lowest rank) */
            uint64_t rhost = team_->procs[root_].host_hash;
            for (size_t i = 0; i < team_->leader_ranks.size(); i++) {
                if (team_->procs[team_->leader_ranks[i]].host_hash ==
                    rhost) {
                    root_leader_idx_ = (int)i;
                    break;
                }
            }
        }
        status = UCC_INPROGRESS;
        return step();
    }

    ucc_status_t progress() override { return step(); }

  private:
    ucc_status_t launch_bcast(Team *t, uint64_t root_in_t,
                              bool host_staged = false)
    {
        ucc_coll_args_t sa{};
        sa.mask      = UCC_COLL_ARGS_FIELD_FLAGS;
        sa.coll_type = UCC_COLL_TYPE_BCAST;
        sa.root      = root_in_t;
        sa.src.info  = a_.src.info;
        if (host_staged) {
            /* device memory crossing nodes: the leader phase runs on a
             * host staging copy (D2H here, H2D when it completes) */
            size_t bytes = a_.src.info.count *
                           ucc_dt_size(a_.src.info.datatype);
            hstg_.resize(bytes);
            if ((int)root_in_t ==
                team_->leaders_oob->my_idx) { /* I hold the data */
                ucc_status_t cs =
                    mc::copy(hstg_.data(), UCC_MEMORY_TYPE_HOST,
                             a_.src.info.buffer, a_.src.info.mem_type,
                             bytes);
                if (cs != UCC_OK) {
                    return cs;
                }
            }
            sa.src.info.buffer   = hstg_.data();
            sa.src.info.mem_type = UCC_MEMORY_TYPE_HOST;
            staged_ = true;
        }
        ucc_status_t st = ucc_collective_init(
            &sa, &req_h_, reinterpret_cast<ucc_team_h>(t));
        if (st != UCC_OK) {
            return st;
        }
        return ucc_collective_post(req_h_);
    }

    ucc_status_t step()
    {
        while (true) {
            if (req_h_) {
                ucc_status_t st = ucc_collective_test(req_h_);
                if (st == UCC_INPROGRESS) {
                    return UCC_INPROGRESS;
                }
                ucc_collective_finalize(req_h_);
                req_h_ = nullptr;
                if (st != UCC_OK) {
                    return st;
                }
                if (staged_) { /* leaders phase on a host copy: H2D */
                    staged_ = false;
                    size_t bytes = a_.src.info.count *
                                   ucc_dt_size(a_.src.info.datatype);
                    ucc_status_t cs =
                        mc::copy(a_.src.info.buffer,
                                 a_.src.info.mem_type, hstg_.data(),
                                 UCC_MEMORY_TYPE_HOST, bytes);
                    if (cs != UCC_OK) {
                        return cs;
                    }
                }
                phase_++;
            }
            switch (phase_) {
            case 0: /* root's node: bcast from the root so the node
                     * leader holds the data */
                if (!on_root_node_ || team_->node_team->size == 1) {
                    phase_ = 1;
                    continue;
                }
                {
                    /* root's index inside its node team */
                    int ridx = -1;
                    for (size_t i = 0; i < team_->node_ranks.size();
                         i++) {
                        if (team_->node_ranks[i] == root_) {
                            ridx = (int)i;
                        }
                    }
                    ucc_status_t st = launch_bcast(
                        team_->node_team.get(), (uint64_t)ridx);
                    if (st != UCC_OK) {
                        return st;
                    }
                }
                break;
            case 1: /* leaders bcast from the root's leader */
                if (!leader_) {
                    phase_ = 2;
                    continue;
                }
                {
                    ucc_status_t st = launch_bcast(
                        team_->leaders_team.get(),
                        (uint64_t)root_leader_idx_,
                        mc::is_device_mt(a_.src.info.mem_type));
                    if (st != UCC_OK) {
                        return st;
                    }
                }
                break;
            case 2: /* every node: bcast from its leader (node rank 0);
                     * the root's node already has the data but repeats
                     * harmlessly (same values) to keep phases aligned */
                if (team_->node_team->size == 1) {
                    return UCC_OK;
                }
                {
                    ucc_status_t st =
                        launch_bcast(team_->node_team.get(), 0);
                    if (st != UCC_OK) {
                        return st;
                    }
                }
                break;
            case 3:
                return UCC_OK;
            default:
                return UCC_ERR_INVALID_PARAM;
            }
            if (req_h_) {
                continue;
            }
        }
    }

    Team           *team_;
    ucc_coll_args_t a_;
    ucc_coll_req_h  req_h_ = nullptr;
    int             phase_ = 0;
    bool            leader_ = false, on_root_node_ = false;
    bool            staged_ = false;
    std::vector<uint8_t> hstg_;
    uint32_t        root_ = 0;
    int             my_node_idx_ = -1, root_leader_idx_ = -1;
};

void add_scores(Team *team)
{
    ScoreRange r;
    r.start    = 0;
    r.end      = SIZE_MAX;
    r.score    = 60;
    r.tl_name  = "hier";
    r.alg_name = "rab";
    r.init     = [](const ucc_coll_args_t &args, Team *t,
                Task **task) -> ucc_status_t {
        if (args.op != UCC_OP_SUM && args.op != UCC_OP_MAX &&
            args.op != UCC_OP_MIN && args.op != UCC_OP_PROD &&
            args.op != UCC_OP_AVG) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        auto  &cfg = Config::instance();
        size_t fb  = cfg.get_size("CL_HIER", "FRAG_SIZE",
                                  4 * 1024 * 1024);
        size_t dtsz = ucc_dt_size(args.dst.info.datatype);
        size_t msg  = (size_t)args.dst.info.count * dtsz;
        if (fb > 0 && dtsz > 0 && msg >= 2 * fb) {
            size_t depth = (size_t)cfg.get_int(
                "CL_HIER", "PIPELINE_DEPTH", 2);
            auto *pt = new HierAllreducePipeTask(
                t->ctx, t, args, fb / dtsz, depth);
            std::string tr = cfg.get("CL_HIER", "PIPELINE_TRACE", "");
            if (!tr.empty()) {
                std::string path = tr;
                void       *id   = (void *)pt;
                pt->trace = [path, id](char ev, size_t f, size_t s) {
                    FILE *fp = fopen(path.c_str(), "a");
                    if (fp) {
                        fprintf(fp, "%c %p %zu %zu\n", ev, id, f, s);
                        fclose(fp);
                    }
                };
            }
            *task = pt;
            return UCC_OK;
        }
        *task = new HierAllreduceTask(t->ctx, t, args);
        return UCC_OK;
    };
    team->score_map.add(UCC_COLL_TYPE_ALLREDUCE, UCC_MEMORY_TYPE_HOST, r);

    {
        /* device-memory hier allreduce: node phases run on the device
         * TL (cdna4); the inter-node leader phase stages through host
         * buffers over the host transports (the multi-node GPU
         * composition the reference reaches via tl/ucp GPU-aware UCX —
         * re-derived as explicit D2H/H2D staging in the pipeline). */
        ScoreRange dr;
        dr.start    = 0;
        dr.end      = SIZE_MAX;
        dr.score    = 60;
        dr.tl_name  = "hier";
        dr.alg_name = "rab_dev";
        dr.init     = [](const ucc_coll_args_t &args, Team *t,
                     Task **task) -> ucc_status_t {
            if (args.op != UCC_OP_SUM && args.op != UCC_OP_MAX &&
                args.op != UCC_OP_MIN && args.op != UCC_OP_PROD &&
                args.op != UCC_OP_AVG) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            auto  &cfg = Config::instance();
            size_t fb  = cfg.get_size("CL_HIER", "FRAG_SIZE",
                                      4 * 1024 * 1024);
            size_t dtsz = ucc_dt_size(args.dst.info.datatype);
            if (dtsz == 0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            size_t depth = (size_t)cfg.get_int(
                "CL_HIER", "PIPELINE_DEPTH", 2);
            *task = new HierAllreducePipeTask(
                t->ctx, t, args, fb ? fb / dtsz : 0, depth);
            return UCC_OK;
        };
        team->score_map.add(UCC_COLL_TYPE_ALLREDUCE,
                            UCC_MEMORY_TYPE_CUDA, dr);
    }

    if (team->rails_ok && team->rail_team) {
        /* split_rail beats RAB on large vectors: every rank carries
         * 1/node_size of the inter-node traffic instead of leaders
         * carrying all of it */
        ScoreRange s;
        s.start    = 64 * 1024;
        s.end      = SIZE_MAX;
        s.score    = 61;
        s.tl_name  = "hier";
        s.alg_name = "split_rail";
        s.init     = [](const ucc_coll_args_t &args, Team *t,
                    Task **task) -> ucc_status_t {
            if (args.op != UCC_OP_SUM && args.op != UCC_OP_MAX &&
                args.op != UCC_OP_MIN && args.op != UCC_OP_PROD &&
                args.op != UCC_OP_AVG) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            auto  &cfg = Config::instance();
            size_t fb  = cfg.get_size("CL_HIER", "FRAG_SIZE",
                                      4 * 1024 * 1024);
            size_t dtsz = ucc_dt_size(args.dst.info.datatype);
            size_t msg  = (size_t)args.dst.info.count * dtsz;
            if (fb > 0 && dtsz > 0 && msg >= 2 * fb) {
                size_t depth = (size_t)cfg.get_int(
                    "CL_HIER", "PIPELINE_DEPTH", 2);
                auto *pt = new HierSplitRailPipeTask(
                    t->ctx, t, args, fb / dtsz, depth);
                std::string tr =
                    cfg.get("CL_HIER", "PIPELINE_TRACE", "");
                if (!tr.empty()) {
                    std::string path = tr;
                    void       *id   = (void *)pt;
                    pt->trace = [path, id](char ev, size_t f,
                                           size_t s2) {
                        FILE *fp = fopen(path.c_str(), "a");
                        if (fp) {
                            fprintf(fp, "%c %p %zu %zu\n", ev, id, f,
                                    s2);
                            fclose(fp);
                        }
                    };
                }
                *task = pt;
                return UCC_OK;
            }
            *task = new SplitRailAllreduceTask(t->ctx, t, args);
            return UCC_OK;
        };
        team->score_map.add(UCC_COLL_TYPE_ALLREDUCE, UCC_MEMORY_TYPE_HOST,
                            s);
    }

    ScoreRange b;
    b.start    = 0;
    b.end      = SIZE_MAX;
    b.score    = 60;
    b.tl_name  = "hier";
    b.alg_name = "2step";
    b.init     = [](const ucc_coll_args_t &args, Team *t,
                Task **task) -> ucc_status_t {
        if (args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        *task = new HierBcastTask(t->ctx, t, args);
        return UCC_OK;
    };
    team->score_map.add(UCC_COLL_TYPE_BCAST, UCC_MEMORY_TYPE_HOST, b);
    /* device-memory 2step bcast: node phases on the device TL, the
     * leaders hop staged through host buffers (same pattern as
     * rab_dev) */
    team->score_map.add(UCC_COLL_TYPE_BCAST, UCC_MEMORY_TYPE_CUDA, b);

    /* node-aggregated a2av: nleaders^2 sockets instead of nranks^2.
     * NOTE: alltoallv SELECTION sizing is rank-invariant (constant) —
     * per-rank row sums diverge across ranks and mixed selections
     * deadlock — so this range knob acts as an on/off switch
     * (0 disables); the small-vs-large adaptivity lives in the flat
     * hybrid's per-pair threshold when this path is off. */
    size_t a2av_thresh = Config::instance().get_size(
        "CL_HIER", "A2AV_NODE_THRESH", 256 * 1024);
    if (a2av_thresh > 0) {
        ScoreRange av;
        av.start    = 0;
        av.end      = a2av_thresh;
        av.score    = 60;
        av.tl_name  = "hier";
        av.alg_name = "node_aggregated";
        av.init     = [](const ucc_coll_args_t &args, Team *t,
                     Task **task) -> ucc_status_t {
            if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
                (args.flags & UCC_COLL_ARGS_FLAG_IN_PLACE) ||
                !ucc_dt_is_predefined(args.src.info_v.datatype) ||
                !ucc_dt_is_predefined(args.dst.info_v.datatype)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new HierAlltoallvTask(t->ctx, t, args);
            return UCC_OK;
        };
        team->score_map.add(UCC_COLL_TYPE_ALLTOALLV,
                            UCC_MEMORY_TYPE_HOST, av);
    }

    ScoreRange ag;
    ag.start    = 0;
    ag.end      = SIZE_MAX;
    ag.score    = 60;
    ag.tl_name  = "hier";
    ag.alg_name = "node_packed";
    ag.init     = [](const ucc_coll_args_t &args, Team *t,
                 Task **task) -> ucc_status_t {
        if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
            !ucc_dt_is_predefined(args.dst.info_v.datatype)) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        *task = new HierAllgathervTask(t->ctx, t, args);
        return UCC_OK;
    };
    team->score_map.add(UCC_COLL_TYPE_ALLGATHERV, UCC_MEMORY_TYPE_HOST,
                        ag);
    /* device-memory allgatherv: node gatherv + node bcast on the
     * device TL with the packed layout in device scratch; only the
     * leaders' inter-node allgatherv hop stages D2H/H2D (same shape
     * as rab_dev / the 2step device variants) */
    team->score_map.add(UCC_COLL_TYPE_ALLGATHERV, UCC_MEMORY_TYPE_CUDA,
                        ag);

    ScoreRange rd;
    rd.start    = 0;
    rd.end      = SIZE_MAX;
    rd.score    = 60;
    rd.tl_name  = "hier";
    rd.alg_name = "2step";
    rd.init     = [](const ucc_coll_args_t &args, Team *t,
                 Task **task) -> ucc_status_t {
        if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
            !ucc_dt_is_predefined(args.src.info.datatype)) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        if (args.op != UCC_OP_SUM && args.op != UCC_OP_MAX &&
            args.op != UCC_OP_MIN && args.op != UCC_OP_PROD &&
            args.op != UCC_OP_AVG) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        *task = new HierReduceTask(t->ctx, t, args);
        return UCC_OK;
    };
    team->score_map.add(UCC_COLL_TYPE_REDUCE, UCC_MEMORY_TYPE_HOST, rd);
    team->score_map.add(UCC_COLL_TYPE_REDUCE, UCC_MEMORY_TYPE_CUDA, rd);

    ScoreRange ba;
    ba.start    = 0;
    ba.end      = SIZE_MAX;
    ba.score    = 60;
    ba.tl_name  = "hier";
    ba.alg_name = "fanin_fanout";
    ba.init     = [](const ucc_coll_args_t &args, Team *t,
                 Task **task) -> ucc_status_t {
        *task = new HierBarrierTask(t->ctx, t, args);
        return UCC_OK;
    };
    team->score_map.add(UCC_COLL_TYPE_BARRIER, UCC_MEMORY_TYPE_HOST, ba);
}

} // namespace hier
} // namespace ucc
