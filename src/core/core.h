/* ucc_amd — internal object model.
 *
 * Reference parity map (structure re-derived, not ported):
 *  - Task/Schedule      <- src/schedule/ucc_schedule.h (task fn ptrs, deps,
 *                          event manager) — here: virtual post/progress +
 *                          dependency counters on a C++ class.
 *  - Context progress   <- core/ucc_context.c:1063 progress queue loop.
 *  - Team state machine <- core/ucc_team.c (ADDR_EXCHANGE->TL_CREATE->ACTIVE).
 *  - Score map dispatch <- src/coll_score/ (ranges per coll x memtype,
 *                          fallback chain).
 *  - TL static registry <- components/{cl,tl}/ dlopen zoo collapsed to a
 *                          compiled-in registry (design deviation per
 *                          SURVEY.md section 7).
 */
#ifndef UCC_AMD_CORE_H_
#define UCC_AMD_CORE_H_

#include <atomic>
#include <cstdint>
#include <cstring>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

#include "../api/ucc.h"
#include "../utils/config.h"
#include "../utils/lf_queue.h"
#include "../utils/log.h"

namespace ucc {

struct Lib;
struct Context;
struct Team;
class Task;
class Schedule;
class TlContext;
class TlTeam;
class Tl;
struct CollRequest;

double time_sec(); /* monotonic seconds */

/* ------------------------------------------------------------------ Task */
/* Unit of the progress engine. A collective algorithm is a Task (or a
 * Schedule of Tasks). status lifecycle:
 *   OPERATION_INITIALIZED -> (post) -> INPROGRESS|OK|err -> (progress)* */
class Task {
  public:
    explicit Task(Context *ctx) : ctx_(ctx) {}
    virtual ~Task() = default;

    virtual ucc_status_t post()     = 0;
    virtual ucc_status_t progress() { return status; }
    /* Triggered (stream) post: launch the collective's device work onto
     * the execution-engine stream; completion is stream-ordered. Tasks
     * that cannot run stream-ordered return NOT_SUPPORTED. */
    virtual ucc_status_t triggered_post(void *ee_stream)
    {
        (void)ee_stream;
        return UCC_ERR_NOT_SUPPORTED;
    }
    /* Called exactly once when the task reaches a terminal status. */
    virtual void on_complete() {}

    /* written by the progress thread, read by user threads through
     * ucc_collective_test / re-post checks: atomic for THREAD_MULTIPLE
     * (tsan-verified, `make tsan`). Relaxed-by-default seq_cst is fine:
     * per-collective cost is nanoseconds. */
    std::atomic<ucc_status_t> status{UCC_OPERATION_INITIALIZED};
    Context     *ctx_;
    CollRequest *req_        = nullptr; /* set on the root task           */
    Schedule    *sched       = nullptr; /* parent schedule, if any        */
    int          n_deps      = 0;
    int          n_satisfied = 0;
    std::vector<Task *> dependents; /* started when I complete            */
    double       start_time = 0;
    double       timeout    = 0; /* seconds; 0 = none                     */
    std::atomic<bool> in_pq{false};

    void depends_on(Task *producer)
    {
        producer->dependents.push_back(this);
        n_deps++;
    }
};

/* Run a task: post it and hand it to the progress queue if needed. */
void task_start(Task *t);
/* Called by the engine when t->status became terminal. */
void task_completed(Task *t);

/* -------------------------------------------------------------- Schedule */
class Schedule : public Task {
  public:
    explicit Schedule(Context *ctx) : Task(ctx) {}
    ~Schedule() override
    {
        for (auto *t : tasks_) {
            delete t;
        }
    }
    void add(Task *t)
    {
        t->sched = this;
        tasks_.push_back(t);
    }
    ucc_status_t post() override;
    ucc_status_t progress() override { return status; }
    void         subtask_completed(Task *t);

    std::vector<Task *> tasks_;
    size_t              n_completed_ = 0;
    bool                posting_     = false;
};

/* --------------------------------------------------------- PipelineTask */
/* Generic fragment pipeline (reference ucc_schedule_pipelined.h:36-78
 * role, re-derived): a collective is split into n_frags fragments, each
 * running the same n_stages-stage chain; at most pdepth fragments are in
 * flight, so stage s+1 of fragment f overlaps stage s of fragment f+1.
 * Stages are sub-collective requests produced by stage_post(frag, stage)
 * — posting happens in FRAGMENT ORDER per stage on every rank, which is
 * the matching invariant our TLs need (tl/tcp tags and tl/shm slots are
 * assigned by per-team post sequence). A stage may "skip" by returning
 * UCC_OK with *req = nullptr. */
class PipelineTask : public Task {
  public:
    using StagePost = std::function<ucc_status_t(
        size_t frag, size_t stage, ucc_coll_req_h *req)>;

    explicit PipelineTask(Context *ctx) : Task(ctx) {}
    ~PipelineTask() override;

    size_t    n_frags  = 1;
    size_t    n_stages = 1;
    size_t    pdepth   = 2;
    StagePost stage_post;
    /* Two stages that post on the SAME sub-team must publish their
     * posts in an order that is identical on every rank (sub-collective
     * matching is per-team post-sequence based). Setting pair_early /
     * pair_late pins the canonical interleave
     *   E(0), E(1), L(0), E(2), L(1), ... , L(n-1)
     * by gating L(f) on E(min(f+1, last)) having POSTED; E(f) after
     * L(f-2) holds automatically through flight recycling. Without the
     * pair, completion-timing differences between ranks can reorder the
     * shared team's posts (observed: cdna4 node-team desync in the
     * device hier RAB). */
    size_t pair_early = SIZE_MAX;
    size_t pair_late  = SIZE_MAX;
    /* optional epilogue run when (frag, stage) completes — e.g. the
     * H2D copy-back after a host-staged leader phase */
    std::function<ucc_status_t(size_t frag, size_t stage)> stage_done;
    /* optional observer for tests/tracing: ev 'P' = stage posted,
     * 'C' = stage completed */
    std::function<void(char ev, size_t frag, size_t stage)> trace;

    ucc_status_t post() override;
    ucc_status_t progress() override { return drive(); }

  private:
    struct Flight {
        size_t         frag = 0, stage = 0;
        ucc_coll_req_h req    = nullptr;
        bool           posted = false, active = false;
    };
    ucc_status_t drive();

    std::vector<Flight> fl_;
    std::vector<size_t> ord_;  /* per-stage posted-fragment count */
    size_t              next_ = 0, done_ = 0;
};

/* ------------------------------------------------------------- ScoreMap  */
using CollInitFn =
    std::function<ucc_status_t(const ucc_coll_args_t &, Team *, Task **)>;

struct ScoreRange {
    size_t     start = 0;
    size_t     end   = SIZE_MAX; /* inclusive range [start, end]           */
    int        score = 0;
    CollInitFn init;
    std::string tl_name;
    std::string alg_name;
};

class ScoreMap {
  public:
    /* indexed [coll_type_idx][mem_type] -> ranges sorted by score desc */
    std::vector<ScoreRange> ranges[UCC_COLL_TYPE_NUM][UCC_MEMORY_TYPE_LAST];

    void add(ucc_coll_type_t ct, ucc_memory_type_t mt, ScoreRange r);
    /* Apply a user tuning string: "coll:msgrange:mem:@alg:score,..." */
    ucc_status_t apply_str(const std::string &s);
    /* Dispatch with fallback: highest score first. */
    ucc_status_t init_coll(const ucc_coll_args_t &args, Team *team,
                           size_t msgsize, Task **task) const;
    std::string  to_string() const;
};

int coll_type_index(ucc_coll_type_t ct);
const char *coll_type_name(ucc_coll_type_t ct);
const char *mem_type_name(ucc_memory_type_t mt);
ucc_status_t coll_type_from_name(const std::string &s, ucc_coll_type_t *ct);

/* -------------------------------------------------------------- ProcInfo */
struct ProcInfo {
    uint64_t host_hash = 0;
    int32_t  pid       = 0;
    int32_t  device    = -1; /* hip device id, -1 = none                   */
    uint64_t ctx_seq   = 0;  /* unique per (pid, context)                  */
    /* placement within the node (reference ucc_proc_info socket/numa
     * via sysfs) and a CPU model hash for symmetric-tuning consensus
     * (reference ucc_topo.h:88-95) */
    int16_t  socket_id = -1;
    int16_t  numa_id   = -1;
    uint32_t cpu_hash  = 0;
};
ProcInfo local_proc_info();

/* --------------------------------------------------------------- OobPoll */
/* Drives repeated nonblocking OOB allgather rounds. */
class OobRound {
  public:
    void init(const ucc_oob_coll_t &oob) { oob_ = oob; }
    /* Begin an allgather of `size` bytes per rank. */
    ucc_status_t start(const void *src, size_t size);
    /* UCC_INPROGRESS / UCC_OK / error. On OK, data() is valid. */
    ucc_status_t test();
    void        *data() { return recv_.data(); }
    uint32_t     size() const { return oob_.n_oob_eps; }
    uint32_t     rank() const { return oob_.oob_ep; }

  private:
    ucc_oob_coll_t        oob_{};
    std::vector<uint8_t>  send_, recv_;
    void                 *req_ = nullptr;
    bool                  active_ = false;
};

/* ------------------------------------------------------------------- TL  */
class TlContext {
  public:
    explicit TlContext(Context *ctx) : ctx_(ctx) {}
    virtual ~TlContext() = default;
    virtual Tl *iface()  = 0;
    Context    *ctx_;
};

class TlTeam {
  public:
    TlTeam(TlContext *tlc, Team *team) : tlc_(tlc), team_(team) {}
    virtual ~TlTeam() = default;

    /* Two-phase create with one combined OOB exchange round:
     *  exchg_size/pack: contribute a fixed-size blob before the round;
     *  exchg_unpack(all): receives n_ranks blobs at my TL's offset;
     *  create_test: poll until resources are ready. */
    virtual size_t       exchg_size() { return 0; }
    virtual void         exchg_pack(void *buf) { (void)buf; }
    virtual ucc_status_t exchg_unpack(const void *all, size_t stride)
    {
        (void)all; (void)stride;
        return UCC_OK;
    }
    virtual ucc_status_t create_test() { return UCC_OK; }
    /* Populate the team score map with this TL's algorithms. */
    virtual void get_scores(Team *team, ScoreMap &map) = 0;

    TlContext *tlc_;
    Team      *team_;
};

class Tl {
  public:
    virtual ~Tl() = default;
    virtual const char *name() const   = 0;
    virtual int         default_score() const = 0;
    /* nullptr if TL cannot run in this process context. */
    virtual TlContext  *context_create(Context *ctx) = 0;
    /* nullptr if TL cannot serve this team (e.g. multi-node for shm). */
    virtual TlTeam     *team_create(TlContext *tlc, Team *team) = 0;
};

std::vector<Tl *> &tl_registry();
void               register_tl(Tl *tl);
void               ensure_builtin_tls(); /* registers self/shm/cdna4/rccl */

/* --------------------------------------------------------------- Context */
struct Context {
    Lib                 *lib = nullptr;
    ucc_context_params_t params{};
    bool                 has_oob = false;
    uint64_t             seq     = 0; /* unique id of this ctx in process  */
    ProcInfo             proc;
    std::deque<Task *>   pq;
    std::recursive_mutex pq_mtx; /* THREAD_MULTIPLE locked fallback        */
    LfQueue<Task *>      lf_pq;  /* THREAD_MULTIPLE lock-free fast path    */
    bool                 mt = false;
    bool                 lock_free = true; /* UCC_LOCK_FREE_PROGRESS_Q     */
    std::vector<std::unique_ptr<TlContext>> tl_ctxs;
    uint64_t             next_team_id = 1;
    uint32_t             n_progress_calls = 0;

    void pq_push(Task *t);
    ucc_status_t progress();
};

/* ------------------------------------------------------------------ Team */
struct Team {
    enum State {
        SPLIT_MEMBERS, /* team split: membership exchange round        */
        SPLIT_OBSERVE, /* team split: non-member observing sub rounds  */
        ADDR_EXCHANGE,
        TL_EXCHANGE,
        TL_CREATE,
        HIER_CREATE, /* internal sub-team bootstrap (src/cl/hier.cc) */
        ACTIVE,
        FAILED,
    };

    Context            *ctx = nullptr;
    ucc_team_params_t   params{};
    ucc_oob_coll_t      oob{};
    bool                has_oob = false;
    uint32_t            rank = 0, size = 1;
    uint16_t            id = 0;
    State               state = ADDR_EXCHANGE;
    ucc_status_t        err   = UCC_OK;
    std::vector<ProcInfo>                  procs;
    std::vector<std::unique_ptr<TlTeam>>   tl_teams;
    std::vector<size_t>                    exchg_off; /* per-TL blob offset */
    size_t                                 exchg_stride = 0;
    std::vector<uint8_t>                   exchg_buf;
    OobRound            oobr;
    ScoreMap            score_map;
    uint64_t            team_uid = 0; /* rank0 random, shared: shm naming  */
    uint64_t            coll_seq = 0;

    /* hierarchical composition (cl/hier role): internal sub-teams built
     * after TL_CREATE when the team spans nodes. node_team covers my
     * node's ranks; leaders_team (leaders only) covers one rank per
     * node. Sub-team OOB rounds ride on the parent OOB with fixed-size
     * padded payloads so non-members can observe (see SubOob). */
    std::unique_ptr<Team>   node_team;
    std::unique_ptr<Team>   leaders_team;
    std::unique_ptr<Team>   rail_team; /* my node-position across nodes */
    struct SubOob          *node_oob    = nullptr;
    struct SubOob          *leaders_oob = nullptr;
    struct SubOob          *rail_oob    = nullptr;
    bool                    rails_ok    = false; /* uniform node sizes */
    int                     hier_step   = 0;
    bool                    want_hier   = false;
    bool                    is_subteam  = false;
    std::vector<uint32_t>   node_ranks, leader_ranks;
    /* team split (ucc_team_create_from_parent) */
    Team                   *split_parent   = nullptr;
    struct SubOob          *split_oob      = nullptr;
    uint32_t                split_included = 0;
    uint64_t                split_ep       = 0;
    int                     split_observe_rounds = 0;
    ~Team();

    bool all_same_node() const;
    bool all_have_device() const;
};

/* ------------------------------------------------------------------ Lib  */
struct Lib {
    ucc_lib_params_t  params{};
    ucc_thread_mode_t thread_mode = UCC_THREAD_SINGLE;
    uint64_t          next_ctx_seq = 1;
};

/* -------------------------------------------------------------------- EE */
/* Execution engine: a user HIP stream colls can be triggered onto
 * (reference ucc_ee_create ucc.h events section; core/ucc_ee.c). */
struct Ee {
    Team            *team = nullptr;
    ucc_ee_params_t  params{};
    void            *stream = nullptr; /* hipStream_t */
    std::deque<ucc_ev_t> events;
    /* triggered collectives whose stream work is in flight: a HIP
     * event recorded after the launch + the COLLECTIVE_COMPLETE ev to
     * publish once it fires (reference ucc.h event flow: POST at
     * launch, COMPLETE when the device work is done) */
    struct Pending {
        void     *hip_ev;
        ucc_ev_t  ev;
    };
    std::deque<Pending> pending;
    ~Ee();
};

/* ----------------------------------------------------------- CollRequest */
/* super.status is a plain C field read by user threads while the
 * progress thread completes the task: access through atomic builtins
 * (THREAD_MULTIPLE; tsan-verified). */
static inline void req_status_store(ucc_coll_req_t *r, ucc_status_t s)
{
    __atomic_store_n((int *)&r->status, (int)s, __ATOMIC_RELEASE);
}
static inline ucc_status_t req_status_load(const ucc_coll_req_t *r)
{
    return (ucc_status_t)__atomic_load_n((const int *)&r->status,
                                         __ATOMIC_ACQUIRE);
}

struct CollRequest {
    ucc_coll_req_t  super; /* must be first: public handle casts here      */
    ucc_coll_args_t args{};
    Team           *team = nullptr;
    Task           *task = nullptr;
    bool            persistent = false;
    bool            posted     = false;
    uint64_t        seq        = 0;
    /* asymmetric src/dst memtype staging (reference ucc_coll.c:236-246):
     * src is copied into a scratch of the dst's memtype at every post */
    std::function<ucc_status_t()> pre_post;
    void             *asymm_scratch = nullptr;
    size_t            asymm_bytes   = 0;
    ucc_memory_type_t asymm_mt      = UCC_MEMORY_TYPE_HOST;
    /* non-contig generic dt: runs after the inner (packed-byte)
     * collective completes OK, before user-visible completion */
    std::function<ucc_status_t()> post_complete;
    std::vector<uint8_t>          gdt_send, gdt_recv;
};

/* msgsize used for score-map range selection (bytes, per reference
 * ucc_coll_utils msgsize conventions). */
size_t coll_args_msgsize(const ucc_coll_args_t &args, uint32_t rank,
                         uint32_t size);

/* SubOob factory/driver (defined in src/cl/hier.cc, shared with team
 * split) */
namespace subooob_api {
SubOob *make(Team *parent, const std::vector<uint32_t> &members,
             int my_idx);
ucc_status_t observe_tick(SubOob *so, int target_rounds);
void         fill_oob(SubOob *so, ucc_oob_coll_t *oob);
void         free_(SubOob *so);
} // namespace subooob_api

/* hierarchical composition hooks (src/cl/hier.cc) */
namespace hier {
bool         creating_subteam();
bool         wanted(Team *team);
ucc_status_t setup(Team *team);
ucc_status_t test(Team *team);
void         add_scores(Team *team);
void         destroy(Team *team);
} // namespace hier
ucc_memory_type_t coll_args_mem_type(const ucc_coll_args_t &args,
                                     uint32_t rank);

} // namespace ucc

/* mem_map handle lookup for TLs (defined in core.cc, C linkage): finds
 * a registered device segment covering [addr, addr+len) inside an
 * EXPORTED ucc_mem_map handle; copies its IPC handle (64B) and sets
 * the offset of addr from the exporting allocation base. Returns 1 on
 * hit. */
extern "C" int ucc_memh_lookup(ucc_mem_map_mem_h memh, const void *addr,
                               size_t len, void *handle_out,
                               uint64_t *alloc_off);

#endif
